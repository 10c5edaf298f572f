// K8: fused AdamW step — fp32 master weights + moments, cast-back to the
// (bf16) working parameter. One launch per tensor (parameter count is ~20;
// multi-tensor chunking is not the bottleneck at this model size).
#include "common.h"

namespace ci {

template <typename T>
__global__ void adamw_kernel(T* __restrict__ p, const T* __restrict__ g,
                             float* __restrict__ m, float* __restrict__ ea,
                             float* __restrict__ eas, long n, float lr,
                             float b1, float b2, float eps, float wd,
                             float bc1, float bc2, bool has_master) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const float gf = ld(g + i);
  float mv = has_master ? m[i] : ld(p + i);
  mv *= (1.f - lr * wd);
  const float a = ea[i] = b1 * ea[i] + (1.f - b1) * gf;
  const float v = eas[i] = b2 * eas[i] + (1.f - b2) * gf * gf;
  mv -= lr * (a / bc1) / (sqrtf(v / bc2) + eps);
  if (has_master) m[i] = mv;
  st(p + i, mv);
}

void fused_adamw(std::vector<at::Tensor> params, std::vector<at::Tensor> grads,
                 std::vector<at::Tensor> masters, std::vector<at::Tensor> eas_,
                 std::vector<at::Tensor> eass, double lr, double b1, double b2,
                 double eps, double wd, double bc1, double bc2) {
  const int threads = 256;
  for (size_t k = 0; k < params.size(); ++k) {
    auto& p = params[k];
    const long n = p.numel();
    const bool has_master = masters[k].data_ptr() != p.data_ptr();
    CI_DISPATCH_FB(p.scalar_type(), "fused_adamw", [&] {
      hipLaunchKernelGGL((adamw_kernel<scalar_t>), dim3(ceil_div(n, threads)),
          dim3(threads), 0, stream(),
          reinterpret_cast<scalar_t*>(p.data_ptr()),
          reinterpret_cast<const scalar_t*>(grads[k].data_ptr()),
          masters[k].data_ptr<float>(), eas_[k].data_ptr<float>(),
          eass[k].data_ptr<float>(), n, (float)lr, (float)b1, (float)b2,
          (float)eps, (float)wd, (float)bc1, (float)bc2, has_master);
    });
  }
}

}  // namespace ci
