// K1: embedding lookup with fused word-level (row) dropout mask.
// fastai EmbeddingDropout masks whole rows of the table (embed_p=0.02,
// train.py:69-70); fusing the mask into the gather avoids materializing a
// masked copy of the 60k x 800 table every forward.
#include "common.h"

namespace ci {

template <typename T>
__global__ void emb_gather_kernel(const T* __restrict__ weight,
                                  const long* __restrict__ ids,
                                  const float* __restrict__ rowmask,  // (V) or null
                                  T* __restrict__ out, long n_tok, int E) {
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= n_tok * E) return;
  const long tok = idx / E;
  const int e = idx % E;
  const long row = ids[tok];
  float v = ld(weight + row * E + e);
  if (rowmask != nullptr) v *= rowmask[row];
  st(out + idx, v);
}

// dW[id] += g * mask[id]  (atomic fp32 into the master-grad buffer)
template <typename T>
__global__ void emb_scatter_kernel(const T* __restrict__ gout,
                                   const long* __restrict__ ids,
                                   const float* __restrict__ rowmask,
                                   float* __restrict__ dweight, long n_tok,
                                   int E, long pad_idx) {
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= n_tok * E) return;
  const long tok = idx / E;
  const int e = idx % E;
  const long row = ids[tok];
  if (row == pad_idx) return;
  float v = ld(gout + idx);
  if (rowmask != nullptr) v *= rowmask[row];
  atomicAdd(dweight + row * E + e, v);
}

at::Tensor emb_gather(at::Tensor weight, at::Tensor ids, at::Tensor rowmask) {
  CI_CHECK_CUDA(weight); CI_CHECK_CONTIG(weight); CI_CHECK_CONTIG(ids);
  const int E = weight.size(1);
  const long n_tok = ids.numel();
  auto sizes = ids.sizes().vec();
  sizes.push_back(E);
  auto out = at::empty(sizes, weight.options());
  const int threads = 256;
  CI_DISPATCH_FB(weight.scalar_type(), "emb_gather", [&] {
    hipLaunchKernelGGL((emb_gather_kernel<scalar_t>),
        dim3(ceil_div(n_tok * E, threads)), dim3(threads), 0, stream(),
        reinterpret_cast<const scalar_t*>(weight.data_ptr()),
        ids.data_ptr<long>(),
        rowmask.numel() ? rowmask.data_ptr<float>() : nullptr,
        reinterpret_cast<scalar_t*>(out.data_ptr()), n_tok, E);
  });
  return out;
}

at::Tensor emb_scatter(at::Tensor gout, at::Tensor ids, at::Tensor rowmask,
                       long V, long pad_idx) {
  CI_CHECK_CUDA(gout); CI_CHECK_CONTIG(ids);
  auto gc = gout.contiguous();
  const int E = gc.size(-1);
  const long n_tok = ids.numel();
  auto dw = at::zeros({V, (long)E}, gc.options().dtype(at::ScalarType::Float));
  const int threads = 256;
  CI_DISPATCH_FB(gc.scalar_type(), "emb_scatter", [&] {
    hipLaunchKernelGGL((emb_scatter_kernel<scalar_t>),
        dim3(ceil_div(n_tok * E, threads)), dim3(threads), 0, stream(),
        reinterpret_cast<const scalar_t*>(gc.data_ptr()),
        ids.data_ptr<long>(),
        rowmask.numel() ? rowmask.data_ptr<float>() : nullptr,
        dw.data_ptr<float>(), n_tok, E, pad_idx);
  });
  return dw;
}

}  // namespace ci
