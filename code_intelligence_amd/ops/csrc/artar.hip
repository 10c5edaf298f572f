// Fused AR/TAR activation regularization (fastai RNNTrainer alpha/beta,
// reference fit-loop semantics — SURVEY.md §2.4 loss path).
//
//   reg = alpha * mean(out^2) + beta * mean((r_t+1 - r_t)^2)
//
// Eager torch spends ~6 elementwise+reduce kernels per step re-streaming
// the (B,T,H) activations (~13 ms at the bench shape). Here ONE forward
// kernel reads each tensor once and ONE backward kernel writes both grad
// contributions. r arrives TIME-MAJOR ((T,B,H) contiguous — the LSTM
// layer's native storage, so the trainer passes the transpose view's
// base with zero copies) which makes the TAR neighbor a uniform +B*H
// stride for every element.
#include "common.h"

namespace ci {

template <typename T>
__global__ void artar_fwd_kernel(const T* __restrict__ out, long n_out,
                                 const T* __restrict__ r, long n_r,
                                 long BH,  // time stride of r
                                 float* __restrict__ acc) {  // [sq, diff]
  constexpr int VEC = 16 / sizeof(T);
  float s_sq = 0.f, s_df = 0.f;
  const long stride = (long)gridDim.x * blockDim.x * VEC;
  for (long base = ((long)blockIdx.x * blockDim.x + threadIdx.x) * VEC;
       base < n_out; base += stride) {
    if (base + VEC <= n_out) {
      float o[VEC];
      ldv<T, VEC>(out + base, o);
      #pragma unroll
      for (int e = 0; e < VEC; ++e) s_sq += o[e] * o[e];
    } else {
      for (long i = base; i < n_out; ++i) {
        const float o = ld(out + i);
        s_sq += o * o;
      }
    }
  }
  const long n_d = n_r - BH;  // elements with a t+1 neighbor
  for (long base = ((long)blockIdx.x * blockDim.x + threadIdx.x) * VEC;
       base < n_d; base += stride) {
    if (base + VEC <= n_d) {
      float a[VEC], b[VEC];
      ldv<T, VEC>(r + base, a);
      ldv<T, VEC>(r + base + BH, b);
      #pragma unroll
      for (int e = 0; e < VEC; ++e) {
        const float d = b[e] - a[e];
        s_df += d * d;
      }
    } else {
      for (long i = base; i < n_d; ++i) {
        const float d = ld(r + i + BH) - ld(r + i);
        s_df += d * d;
      }
    }
  }
  #pragma unroll
  for (int off = kWave / 2; off > 0; off >>= 1) {
    s_sq += __shfl_down(s_sq, off);
    s_df += __shfl_down(s_df, off);
  }
  __shared__ float red[2][4];
  const int wave = threadIdx.x / kWave;
  if ((threadIdx.x & (kWave - 1)) == 0) {
    red[0][wave] = s_sq;
    red[1][wave] = s_df;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    float a0 = 0.f, a1 = 0.f;
    for (int w = 0; w < (int)(blockDim.x / kWave); ++w) {
      a0 += red[0][w];
      a1 += red[1][w];
    }
    atomicAdd(acc, a0);
    atomicAdd(acc + 1, a1);
  }
}

template <typename T>
__global__ void artar_bwd_kernel(const T* __restrict__ out, long n_out,
                                 const T* __restrict__ r, long n_r, long BH,
                                 T* __restrict__ dout, T* __restrict__ dr,
                                 const float* __restrict__ dloss,
                                 float ca, float cb) {
  constexpr int VEC = 16 / sizeof(T);
  const float g = dloss[0];
  const long stride = (long)gridDim.x * blockDim.x * VEC;
  for (long base = ((long)blockIdx.x * blockDim.x + threadIdx.x) * VEC;
       base < n_out; base += stride) {
    if (base + VEC <= n_out) {
      float o[VEC];
      ldv<T, VEC>(out + base, o);
      #pragma unroll
      for (int e = 0; e < VEC; ++e) o[e] = ca * o[e] * g;
      stv<T, VEC>(dout + base, o);
    } else {
      for (long i = base; i < n_out; ++i)
        st(dout + i, ca * ld(out + i) * g);
    }
  }
  for (long base = ((long)blockIdx.x * blockDim.x + threadIdx.x) * VEC;
       base < n_r; base += stride) {
    if (base + VEC <= n_r) {
      const bool has_up = base + BH < n_r;   // whole vector: BH % VEC == 0
      const bool has_dn = base >= BH;
      float a[VEC], up[VEC], dn[VEC];
      ldv<T, VEC>(r + base, a);
      if (has_up) ldv<T, VEC>(r + base + BH, up);
      if (has_dn) ldv<T, VEC>(r + base - BH, dn);
      #pragma unroll
      for (int e = 0; e < VEC; ++e) {
        float s = 0.f;
        if (has_up) s += a[e] - up[e];
        if (has_dn) s += a[e] - dn[e];
        a[e] = cb * s * g;
      }
      stv<T, VEC>(dr + base, a);
    } else {
      for (long i = base; i < n_r; ++i) {
        float s = 0.f;
        if (i + BH < n_r) s += ld(r + i) - ld(r + i + BH);
        if (i >= BH) s += ld(r + i) - ld(r + i - BH);
        st(dr + i, cb * s * g);
      }
    }
  }
}

static int artar_blocks(long n, int threads) {
  const long b = (n + (long)threads * 8 - 1) / ((long)threads * 8);
  return (int)std::min<long>(8192, std::max<long>(b, 512));
}

at::Tensor artar_forward(at::Tensor out, at::Tensor r_tm) {
  // out: any contiguous flat tensor; r_tm: (T, B, H) contiguous
  CI_CHECK_CUDA(out); CI_CHECK_CONTIG(out); CI_CHECK_CONTIG(r_tm);
  TORCH_CHECK(r_tm.dim() == 3 && out.scalar_type() == r_tm.scalar_type());
  const long BH = (long)r_tm.size(1) * r_tm.size(2);
  const long n_out = out.numel(), n_r = r_tm.numel();
  auto acc = at::zeros({2}, out.options().dtype(at::ScalarType::Float));
  const int threads = 256;
  CI_DISPATCH_FB(out.scalar_type(), "artar_fwd", [&] {
    constexpr int VEC = 16 / sizeof(scalar_t);
    TORCH_CHECK(BH % VEC == 0, "artar: B*H must be divisible by ", VEC);
    hipLaunchKernelGGL((artar_fwd_kernel<scalar_t>),
        dim3(artar_blocks(std::max(n_out, n_r), threads)), dim3(threads), 0,
        stream(),
        reinterpret_cast<const scalar_t*>(out.data_ptr()), n_out,
        reinterpret_cast<const scalar_t*>(r_tm.data_ptr()), n_r, BH,
        acc.data_ptr<float>());
  });
  return acc;  // [sum(out^2), sum(diff^2)] — host applies alpha/N, beta/M
}

std::vector<at::Tensor> artar_backward(at::Tensor out, at::Tensor r_tm,
                                       at::Tensor dloss, double ca, double cb) {
  CI_CHECK_CUDA(out); CI_CHECK_CONTIG(out); CI_CHECK_CONTIG(r_tm);
  const long BH = (long)r_tm.size(1) * r_tm.size(2);
  const long n_out = out.numel(), n_r = r_tm.numel();
  auto dout = at::empty_like(out);
  auto dr = at::empty_like(r_tm);
  const int threads = 256;
  CI_DISPATCH_FB(out.scalar_type(), "artar_bwd", [&] {
    constexpr int VEC = 16 / sizeof(scalar_t);
    TORCH_CHECK(BH % VEC == 0, "artar: B*H must be divisible by ", VEC);
    hipLaunchKernelGGL((artar_bwd_kernel<scalar_t>),
        dim3(artar_blocks(std::max(n_out, n_r), threads)), dim3(threads), 0,
        stream(),
        reinterpret_cast<const scalar_t*>(out.data_ptr()), n_out,
        reinterpret_cast<const scalar_t*>(r_tm.data_ptr()), n_r, BH,
        reinterpret_cast<scalar_t*>(dout.data_ptr()),
        reinterpret_cast<scalar_t*>(dr.data_ptr()),
        dloss.data_ptr<float>(), (float)ca, (float)cb);
  });
  return {dout, dr};
}

}  // namespace ci
