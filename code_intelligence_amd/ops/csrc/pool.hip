// K5: masked concat-pool — cat([mean, max, last], -1) over true lengths.
// Single-pass reduction over T; reference semantics inference.py:232-263
// (batch_seq_pool: padding masked per true length; "last" = h[length-1]).
#include "common.h"

namespace ci {

template <typename T>
__global__ void concat_pool_kernel(const T* __restrict__ hidden,
                                   const int* __restrict__ lengths,
                                   T* __restrict__ out, int B, int Tn, int H) {
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= (long)B * H) return;
  const int b = idx / H, h = idx % H;
  const int len = max(1, lengths[b]);
  const T* row = hidden + (long)b * Tn * H + h;
  float sum = 0.f, mx = -3.4e38f;
  for (int t = 0; t < len; ++t) {
    const float v = ld(row + (long)t * H);
    sum += v;
    mx = fmaxf(mx, v);
  }
  const float last = ld(row + (long)(len - 1) * H);
  T* orow = out + (long)b * 3 * H + h;
  st(orow, sum / len);
  st(orow + H, mx);
  st(orow + 2 * H, last);
}

at::Tensor concat_pool(at::Tensor hidden, at::Tensor lengths) {
  CI_CHECK_CUDA(hidden); CI_CHECK_CONTIG(hidden); CI_CHECK_CONTIG(lengths);
  TORCH_CHECK(lengths.scalar_type() == at::ScalarType::Int);
  const int B = hidden.size(0), Tn = hidden.size(1), H = hidden.size(2);
  auto out = at::empty({B, 3 * H}, hidden.options());
  const int threads = 256;
  const int blocks = ceil_div((long)B * H, threads);
  CI_DISPATCH_FB(hidden.scalar_type(), "concat_pool", [&] {
    hipLaunchKernelGGL((concat_pool_kernel<scalar_t>), dim3(blocks), dim3(threads),
        0, stream(),
        reinterpret_cast<const scalar_t*>(hidden.data_ptr()),
        lengths.data_ptr<int>(),
        reinterpret_cast<scalar_t*>(out.data_ptr()), B, Tn, H);
  });
  return out;
}

}  // namespace ci
