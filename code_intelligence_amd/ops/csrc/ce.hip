// K6 epilogues: fused tied-decoder softmax+CE row reductions.
// The (chunked) logits GEMM runs on hipBLASLt; these kernels do the
// per-row max/logsumexp pass and the in-place dlogits transform so full
// logits never round-trip more than once (SURVEY.md §7 "hard parts": K6).
#include "common.h"

namespace ci {

// one block per row: lse[i] = log(sum(exp(x - max))) + max ; tgt[i] = x[target]
template <typename T, int THREADS>
__global__ void ce_rowstats_kernel(const T* __restrict__ logits, long row_stride,
                                   const long* __restrict__ targets,
                                   const float* __restrict__ bias,  // or null
                                   float* __restrict__ lse,
                                   float* __restrict__ tgt, int V) {
  constexpr int NW = THREADS / kWave;
  const int row = blockIdx.x;
  const T* x = logits + (long)row * row_stride;
  __shared__ float red[NW];
  __shared__ float bcast;
  // pass 1: global row max (16-B vector loads, G13)
  constexpr int VEC = 16 / sizeof(T);
  const int Vv = V / VEC * VEC;
  float mx = -3.4e38f;
  for (int v = threadIdx.x * VEC; v < Vv; v += THREADS * VEC) {
    T buf[VEC];
    *reinterpret_cast<int4*>(buf) = *reinterpret_cast<const int4*>(x + v);
    #pragma unroll
    for (int e = 0; e < VEC; ++e)
      mx = fmaxf(mx, ld(buf + e) + (bias ? bias[v + e] : 0.f));
  }
  for (int v = Vv + threadIdx.x; v < V; v += THREADS)
    mx = fmaxf(mx, ld(x + v) + (bias ? bias[v] : 0.f));
  #pragma unroll
  for (int off = kWave / 2; off > 0; off >>= 1)
    mx = fmaxf(mx, __shfl_down(mx, off));
  if ((threadIdx.x & (kWave - 1)) == 0) red[threadIdx.x / kWave] = mx;
  __syncthreads();
  if (threadIdx.x == 0) {
    float m = red[0];
    for (int w = 1; w < NW; ++w) m = fmaxf(m, red[w]);
    bcast = m;
  }
  __syncthreads();
  mx = bcast;
  __syncthreads();  // red[] reused below
  // pass 2: sum exp(x - max)
  float s = 0.f;
  for (int v = threadIdx.x * VEC; v < Vv; v += THREADS * VEC) {
    T buf[VEC];
    *reinterpret_cast<int4*>(buf) = *reinterpret_cast<const int4*>(x + v);
    #pragma unroll
    for (int e = 0; e < VEC; ++e)
      s += __expf(ld(buf + e) + (bias ? bias[v + e] : 0.f) - mx);
  }
  for (int v = Vv + threadIdx.x; v < V; v += THREADS)
    s += __expf(ld(x + v) + (bias ? bias[v] : 0.f) - mx);
  #pragma unroll
  for (int off = kWave / 2; off > 0; off >>= 1) s += __shfl_down(s, off);
  if ((threadIdx.x & (kWave - 1)) == 0) red[threadIdx.x / kWave] = s;
  __syncthreads();
  if (threadIdx.x == 0) {
    float tot = 0.f;
    for (int w = 0; w < NW; ++w) tot += red[w];
    lse[row] = mx + __logf(tot);
    tgt[row] = ld(x + targets[row]) +
               (bias ? bias[targets[row]] : 0.f);
  }
}

void ce_rowstats(at::Tensor logits, at::Tensor targets, at::Tensor bias,
                 at::Tensor lse, at::Tensor tgt) {
  CI_CHECK_CUDA(logits); CI_CHECK_CONTIG(logits);
  const int N = logits.size(0), V = logits.size(1);
  constexpr int THREADS = 256;
  const float* bp = bias.numel() ? bias.data_ptr<float>() : nullptr;
  CI_DISPATCH_FB(logits.scalar_type(), "ce_rowstats", [&] {
    hipLaunchKernelGGL((ce_rowstats_kernel<scalar_t, THREADS>), dim3(N),
        dim3(THREADS), 0, stream(),
        reinterpret_cast<const scalar_t*>(logits.data_ptr()), (long)V,
        targets.data_ptr<long>(), bp, lse.data_ptr<float>(),
        tgt.data_ptr<float>(), V);
  });
}

// in-place: logits <- (exp(logits - lse) - onehot) * scale   (scale on device)
// one block per row, vectorized 8-wide in-place transform (G13).
template <typename T, int THREADS>
__global__ void ce_dlogits_kernel(T* __restrict__ logits, long row_stride,
                                  const long* __restrict__ targets,
                                  const float* __restrict__ bias,  // or null
                                  const float* __restrict__ lse,
                                  const float* __restrict__ scale,
                                  int V) {
  const int row = blockIdx.x;
  T* x = logits + (long)row * row_stride;
  const float l = lse[row];
  const float sc = scale[0];
  const int tgt = (int)targets[row];
  constexpr int VEC = 16 / sizeof(T);  // one 16-B vector per thread-step
  const int Vv = V / VEC * VEC;
  for (int v = threadIdx.x * VEC; v < Vv; v += THREADS * VEC) {
    T buf[VEC];
    *reinterpret_cast<int4*>(buf) = *reinterpret_cast<const int4*>(x + v);
    #pragma unroll
    for (int e = 0; e < VEC; ++e) {
      float p = __expf(ld(buf + e) + (bias ? bias[v + e] : 0.f) - l);
      if (v + e == tgt) p -= 1.f;
      st(buf + e, p * sc);
    }
    *reinterpret_cast<int4*>(x + v) = *reinterpret_cast<const int4*>(buf);
  }
  for (int v = Vv + threadIdx.x; v < V; v += THREADS) {
    float p = __expf(ld(x + v) + (bias ? bias[v] : 0.f) - l);
    if (v == tgt) p -= 1.f;
    st(x + v, p * sc);
  }
}

void ce_dlogits(at::Tensor logits, at::Tensor targets, at::Tensor bias,
                at::Tensor lse, at::Tensor scale) {
  CI_CHECK_CUDA(logits); CI_CHECK_CONTIG(logits);
  const int N = logits.size(0), V = logits.size(1);
  constexpr int THREADS = 256;
  const float* bp = bias.numel() ? bias.data_ptr<float>() : nullptr;
  CI_DISPATCH_FB(logits.scalar_type(), "ce_dlogits", [&] {
    hipLaunchKernelGGL((ce_dlogits_kernel<scalar_t, THREADS>),
        dim3(N), dim3(THREADS), 0, stream(),
        reinterpret_cast<scalar_t*>(logits.data_ptr()), (long)V,
        targets.data_ptr<long>(), bp, lse.data_ptr<float>(),
        scale.data_ptr<float>(), V);
  });
}

}  // namespace ci
