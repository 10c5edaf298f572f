// K6 epilogues: fused tied-decoder softmax+CE row reductions.
// The (chunked) logits GEMM runs on hipBLASLt; these kernels do the
// per-row max/logsumexp pass and the in-place dlogits transform so full
// logits never round-trip more than once (SURVEY.md par.7 "hard parts").
// The decoder bias is folded in HERE (template<HAS_BIAS>, vectorized
// float4 bias loads — a per-element `bias ? bias[v] : 0` conditional made
// hipcc branch around every load and cost 5x, the par.5 trap (c) of the
// CDNA guide) so the (chunk, 60k) bias broadcast-add never materializes.
#include "common.h"

#include <hip/hip_fp8.h>

namespace ci {

typedef float f32x4 __attribute__((ext_vector_type(4)));

template <typename T, int VEC>
static __device__ __forceinline__ void load_vec(const T* p, float* out) {
  T buf[VEC];
  *reinterpret_cast<int4*>(buf) = *reinterpret_cast<const int4*>(p);
  #pragma unroll
  for (int e = 0; e < VEC; ++e) out[e] = ld(buf + e);
}

template <int VEC>
static __device__ __forceinline__ void load_bias(const float* b, float* out) {
  #pragma unroll
  for (int e = 0; e < VEC; e += 4)
    *reinterpret_cast<f32x4*>(out + e) = *reinterpret_cast<const f32x4*>(b + e);
}

// one block per row, ONLINE single-pass logsumexp (one read of the row
// instead of two): per-thread running (max, sum) merged wave-wide then
// across waves. lse[i] = log(sum(exp(x + bias - max))) + max.
template <typename T, int THREADS, bool HAS_BIAS>
__global__ void ce_rowstats_kernel(const T* __restrict__ logits, long row_stride,
                                   const long* __restrict__ targets,
                                   const float* __restrict__ bias,
                                   float* __restrict__ lse,
                                   float* __restrict__ tgt, int V) {
  constexpr int NW = THREADS / kWave;
  constexpr int VEC = 16 / sizeof(T);
  const int row = blockIdx.x;
  const T* x = logits + (long)row * row_stride;
  __shared__ float red_m[NW];
  __shared__ float red_s[NW];
  const int Vv = V / VEC * VEC;
  float v8[VEC], b8[VEC];
  float m = -3.4e38f, sum = 0.f;
  for (int v = threadIdx.x * VEC; v < Vv; v += THREADS * VEC) {
    load_vec<T, VEC>(x + v, v8);
    if (HAS_BIAS) {
      load_bias<VEC>(bias + v, b8);
      #pragma unroll
      for (int e = 0; e < VEC; ++e) v8[e] += b8[e];
    }
    float bm = v8[0];
    #pragma unroll
    for (int e = 1; e < VEC; ++e) bm = fmaxf(bm, v8[e]);
    // rescale the running sum once per 8-wide block
    if (bm > m) {
      sum *= __expf(m - bm);
      m = bm;
    }
    #pragma unroll
    for (int e = 0; e < VEC; ++e) sum += __expf(v8[e] - m);
  }
  for (int v = Vv + threadIdx.x; v < V; v += THREADS) {
    const float val = ld(x + v) + (HAS_BIAS ? bias[v] : 0.f);
    if (val > m) {
      sum *= __expf(m - val);
      m = val;
    }
    sum += __expf(val - m);
  }
  // wave merge
  #pragma unroll
  for (int off = kWave / 2; off > 0; off >>= 1) {
    const float om = __shfl_down(m, off);
    const float os = __shfl_down(sum, off);
    const float nm = fmaxf(m, om);
    sum = sum * __expf(m - nm) + os * __expf(om - nm);
    m = nm;
  }
  if ((threadIdx.x & (kWave - 1)) == 0) {
    red_m[threadIdx.x / kWave] = m;
    red_s[threadIdx.x / kWave] = sum;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    float gm = red_m[0];
    for (int w = 1; w < NW; ++w) gm = fmaxf(gm, red_m[w]);
    float gs = 0.f;
    for (int w = 0; w < NW; ++w) gs += red_s[w] * __expf(red_m[w] - gm);
    lse[row] = gm + __logf(gs);
    tgt[row] = ld(x + targets[row]) + (HAS_BIAS ? bias[targets[row]] : 0.f);
  }
}

void ce_rowstats(at::Tensor logits, at::Tensor targets, at::Tensor bias,
                 at::Tensor lse, at::Tensor tgt) {
  CI_CHECK_CUDA(logits); CI_CHECK_CONTIG(logits);
  const int N = logits.size(0), V = logits.size(1);
  constexpr int THREADS = 256;
  const bool hb = bias.numel() > 0;
  CI_DISPATCH_FB(logits.scalar_type(), "ce_rowstats", [&] {
    if (hb) {
      hipLaunchKernelGGL((ce_rowstats_kernel<scalar_t, THREADS, true>), dim3(N),
          dim3(THREADS), 0, stream(),
          reinterpret_cast<const scalar_t*>(logits.data_ptr()), (long)V,
          targets.data_ptr<long>(), bias.data_ptr<float>(),
          lse.data_ptr<float>(), tgt.data_ptr<float>(), V);
    } else {
      hipLaunchKernelGGL((ce_rowstats_kernel<scalar_t, THREADS, false>), dim3(N),
          dim3(THREADS), 0, stream(),
          reinterpret_cast<const scalar_t*>(logits.data_ptr()), (long)V,
          targets.data_ptr<long>(), nullptr,
          lse.data_ptr<float>(), tgt.data_ptr<float>(), V);
    }
  });
}

// in-place: logits <- (exp(logits + bias - lse) - onehot) * scale
template <typename T, int THREADS, bool HAS_BIAS>
__global__ void ce_dlogits_kernel(T* __restrict__ logits, long row_stride,
                                  const long* __restrict__ targets,
                                  const float* __restrict__ bias,
                                  const float* __restrict__ lse,
                                  const float* __restrict__ scale, int V) {
  constexpr int VEC = 16 / sizeof(T);
  const int row = blockIdx.x;
  T* x = logits + (long)row * row_stride;
  const float l = lse[row];
  const float sc = scale[0];
  const int tgt = (int)targets[row];
  const int Vv = V / VEC * VEC;
  float v8[VEC], b8[VEC];
  for (int v = threadIdx.x * VEC; v < Vv; v += THREADS * VEC) {
    load_vec<T, VEC>(x + v, v8);
    if (HAS_BIAS) {
      load_bias<VEC>(bias + v, b8);
      #pragma unroll
      for (int e = 0; e < VEC; ++e) v8[e] += b8[e];
    }
    T buf[VEC];
    #pragma unroll
    for (int e = 0; e < VEC; ++e) {
      float p = __expf(v8[e] - l);
      if (v + e == tgt) p -= 1.f;
      st(buf + e, p * sc);
    }
    *reinterpret_cast<int4*>(x + v) = *reinterpret_cast<const int4*>(buf);
  }
  for (int v = Vv + threadIdx.x; v < V; v += THREADS) {
    float p = __expf(ld(x + v) + (HAS_BIAS ? bias[v] : 0.f) - l);
    if (v == tgt) p -= 1.f;
    st(x + v, p * sc);
  }
}

void ce_dlogits(at::Tensor logits, at::Tensor targets, at::Tensor bias,
                at::Tensor lse, at::Tensor scale) {
  CI_CHECK_CUDA(logits); CI_CHECK_CONTIG(logits);
  const int N = logits.size(0), V = logits.size(1);
  constexpr int THREADS = 256;
  const bool hb = bias.numel() > 0;
  CI_DISPATCH_FB(logits.scalar_type(), "ce_dlogits", [&] {
    if (hb) {
      hipLaunchKernelGGL((ce_dlogits_kernel<scalar_t, THREADS, true>),
          dim3(N), dim3(THREADS), 0, stream(),
          reinterpret_cast<scalar_t*>(logits.data_ptr()), (long)V,
          targets.data_ptr<long>(), bias.data_ptr<float>(),
          lse.data_ptr<float>(), scale.data_ptr<float>(), V);
    } else {
      hipLaunchKernelGGL((ce_dlogits_kernel<scalar_t, THREADS, false>),
          dim3(N), dim3(THREADS), 0, stream(),
          reinterpret_cast<scalar_t*>(logits.data_ptr()), (long)V,
          targets.data_ptr<long>(), nullptr,
          lse.data_ptr<float>(), scale.data_ptr<float>(), V);
    }
  });
}

// ---- fp8 CE GEMM support (CI_CE_FP8R) ------------------------------------
// Measured on MI355X (scripts/fp8r_probe.py, profiles/fp8r_probe.log):
// the fp8->fp8 GEMM epilogue is UNTUNED on this stack (1.83 ms vs 1.15 ms
// for fp8->bf16 at the chunk shape) and _scaled_mm ignores scale_result,
// so a fully fp8-RESIDENT logits buffer loses. The winning composition is
// hybrid: bf16-resident logits produced by the fp8-input GEMM, and this
// dual epilogue that (in one read of the row) rewrites the bf16 buffer to
// (softmax-onehot)*dloss/N for the bf16 dW GEMM while emitting an e4m3
// copy scaled by STORE=448 for the fp8 dh GEMM (0.78 vs 1.44 ms).
template <typename T, int THREADS, bool HAS_BIAS>
__global__ void ce_dlogits_dual_kernel(T* __restrict__ logits, long row_stride,
                                       const long* __restrict__ targets,
                                       const float* __restrict__ bias,
                                       const float* __restrict__ lse,
                                       const float* __restrict__ scale,
                                       unsigned char* __restrict__ scratch8,
                                       long srs, float store_scale, int V) {
  constexpr int VEC = 16 / sizeof(T);
  const int row = blockIdx.x;
  T* x = logits + (long)row * row_stride;
  unsigned char* s8 = scratch8 + (long)row * srs;
  const float l = lse[row];
  const float sc = scale[0];
  const int tgt = (int)targets[row];
  const int Vv = V / VEC * VEC;
  float v8[VEC], b8[VEC];
  for (int v = threadIdx.x * VEC; v < Vv; v += THREADS * VEC) {
    load_vec<T, VEC>(x + v, v8);
    if (HAS_BIAS) {
      load_bias<VEC>(bias + v, b8);
      #pragma unroll
      for (int e = 0; e < VEC; ++e) v8[e] += b8[e];
    }
    T buf[VEC];
    unsigned char q8[VEC];
    #pragma unroll
    for (int e = 0; e < VEC; ++e) {
      float p = __expf(v8[e] - l);
      if (v + e == tgt) p -= 1.f;
      st(buf + e, p * sc);
      q8[e] = __hip_fp8_e4m3(p * store_scale).__x;
    }
    *reinterpret_cast<int4*>(x + v) = *reinterpret_cast<const int4*>(buf);
    if constexpr (VEC == 8) {  // bf16 path: 8 fp8 bytes
      *reinterpret_cast<uint2*>(s8 + v) = *reinterpret_cast<const uint2*>(q8);
    } else {                   // fp32 path: 4 fp8 bytes
      *reinterpret_cast<unsigned int*>(s8 + v) =
          *reinterpret_cast<const unsigned int*>(q8);
    }
  }
  for (int v = Vv + threadIdx.x; v < V; v += THREADS) {
    float p = __expf(ld(x + v) + (HAS_BIAS ? bias[v] : 0.f) - l);
    if (v == tgt) p -= 1.f;
    st(x + v, p * sc);
    s8[v] = __hip_fp8_e4m3(p * store_scale).__x;
  }
}

void ce_dlogits_dual(at::Tensor logits, at::Tensor targets, at::Tensor bias,
                     at::Tensor lse, at::Tensor scale, at::Tensor scratch8,
                     double store_scale) {
  CI_CHECK_CUDA(logits); CI_CHECK_CONTIG(logits); CI_CHECK_CONTIG(scratch8);
  const int N = logits.size(0), V = logits.size(1);
  TORCH_CHECK(scratch8.size(0) >= N && scratch8.size(1) == V,
              "ce_dlogits_dual: scratch too small");
  TORCH_CHECK(scratch8.scalar_type() == at::ScalarType::Byte ||
              scratch8.scalar_type() == at::ScalarType::Float8_e4m3fn);
  constexpr int THREADS = 256;
  const bool hb = bias.numel() > 0;
  CI_DISPATCH_FB(logits.scalar_type(), "ce_dlogits_dual", [&] {
    if (hb) {
      hipLaunchKernelGGL((ce_dlogits_dual_kernel<scalar_t, THREADS, true>),
          dim3(N), dim3(THREADS), 0, stream(),
          reinterpret_cast<scalar_t*>(logits.data_ptr()), (long)V,
          targets.data_ptr<long>(), bias.data_ptr<float>(),
          lse.data_ptr<float>(), scale.data_ptr<float>(),
          reinterpret_cast<unsigned char*>(scratch8.data_ptr()),
          (long)V, (float)store_scale, V);
    } else {
      hipLaunchKernelGGL((ce_dlogits_dual_kernel<scalar_t, THREADS, false>),
          dim3(N), dim3(THREADS), 0, stream(),
          reinterpret_cast<scalar_t*>(logits.data_ptr()), (long)V,
          targets.data_ptr<long>(), nullptr,
          lse.data_ptr<float>(), scale.data_ptr<float>(),
          reinterpret_cast<unsigned char*>(scratch8.data_ptr()),
          (long)V, (float)store_scale, V);
    }
  });
}

}  // namespace ci
