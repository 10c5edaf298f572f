// K2 (lib mode): LSTM cell pointwise kernels + C++ sequence drivers.
// The recurrent GEMM runs through hipBLASLt (at::mm) per timestep; the gate
// activation + c/h update (and their backward) are fused HIP kernels.
// Reference semantics: fastai AWD_LSTM nn.LSTM cell, gate order i,f,g,o
// (/root/reference/Issue_Embeddings/train.py:88-92).
#include "common.h"

#include <hip/hip_fp8.h>

namespace ci {

// |h| < 1 exactly (sigmoid * tanh), so the fp8 copy of the hidden state
// uses a FIXED scale of 1/448: h8 = e4m3(h * 448), dequant h = h8/448.
// Emitted by the cell kernel for free (CI_LSTM_FP8 recurrent GEMM path).
constexpr float kH8Scale = 448.0f;

void quantize_e4m3(at::Tensor src, at::Tensor dst, at::Tensor scale);  // fp8util.hip

// VEC consecutive j per thread with 16-B vector loads/stores: ldv/stv
// helpers live in common.h. Scalar tail path covers H % VEC != 0.

// one thread per (b, j-block): gates = xp + rec (+bias); c' = f*c + i*g;
// h' = o*tanh(c')
template <typename T>
__global__ void lstm_cell_fwd(
    const T* __restrict__ xp, long xp_rs,      // (B,4H) view, row stride xp_rs
    const T* __restrict__ rec, long rec_rs,    // (B,4H) recurrent GEMM out
    const float* __restrict__ bias,            // (4H) b_ih + b_hh
    const float* __restrict__ c_prev, long cp_rs,
    T* __restrict__ h_out, long h_rs,
    float* __restrict__ c_out, long c_rs,
    T* __restrict__ gates_out, long g_rs,      // post-activation i,f,g,o
    unsigned char* __restrict__ h8_out,        // optional e4m3(h*448)
    int B, int H) {
  constexpr int VEC = 16 / sizeof(T);
  const int Hv = H / VEC;  // vector blocks per row (tail handled scalar)
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= (long)B * Hv) return;
  const int b = idx / Hv, j = (idx % Hv) * VEC;
  const long xo = (long)b * xp_rs + j;
  const long ro = (long)b * rec_rs + j;
  float gi[VEC], gf[VEC], gg[VEC], go[VEC], tmp[VEC], cprev[VEC];
  ldv<T, VEC>(xp + xo, gi); ldv<T, VEC>(rec + ro, tmp);
  #pragma unroll
  for (int e = 0; e < VEC; ++e) gi[e] = sigmoidf_(gi[e] + tmp[e] + bias[j + e]);
  ldv<T, VEC>(xp + xo + H, gf); ldv<T, VEC>(rec + ro + H, tmp);
  #pragma unroll
  for (int e = 0; e < VEC; ++e) gf[e] = sigmoidf_(gf[e] + tmp[e] + bias[j + H + e]);
  ldv<T, VEC>(xp + xo + 2 * H, gg); ldv<T, VEC>(rec + ro + 2 * H, tmp);
  #pragma unroll
  for (int e = 0; e < VEC; ++e) gg[e] = tanhf(gg[e] + tmp[e] + bias[j + 2 * H + e]);
  ldv<T, VEC>(xp + xo + 3 * H, go); ldv<T, VEC>(rec + ro + 3 * H, tmp);
  #pragma unroll
  for (int e = 0; e < VEC; ++e) go[e] = sigmoidf_(go[e] + tmp[e] + bias[j + 3 * H + e]);
  ldv_f32<VEC>(c_prev + (long)b * cp_rs + j, cprev);
  float c[VEC], h[VEC];
  #pragma unroll
  for (int e = 0; e < VEC; ++e) {
    c[e] = gf[e] * cprev[e] + gi[e] * gg[e];
    h[e] = go[e] * tanhf(c[e]);
  }
  stv<T, VEC>(h_out + (long)b * h_rs + j, h);
  if (h8_out != nullptr) {
    unsigned char q[VEC];
    #pragma unroll
    for (int e = 0; e < VEC; ++e)
      q[e] = __hip_fp8_e4m3(h[e] * kH8Scale).__x;
    if constexpr (VEC == 8) {
      *reinterpret_cast<uint2*>(h8_out + (long)b * h_rs + j) =
          *reinterpret_cast<const uint2*>(q);
    } else {
      *reinterpret_cast<unsigned int*>(h8_out + (long)b * h_rs + j) =
          *reinterpret_cast<const unsigned int*>(q);
    }
  }
  stv_f32<VEC>(c_out + (long)b * c_rs + j, c);
  const long gout = (long)b * g_rs + j;
  stv<T, VEC>(gates_out + gout, gi);
  stv<T, VEC>(gates_out + gout + H, gf);
  stv<T, VEC>(gates_out + gout + 2 * H, gg);
  stv<T, VEC>(gates_out + gout + 3 * H, go);
}

// scalar tail kernel for H % VEC != 0 columns
template <typename T>
__global__ void lstm_cell_fwd_tail(
    const T* __restrict__ xp, long xp_rs, const T* __restrict__ rec, long rec_rs,
    const float* __restrict__ bias, const float* __restrict__ c_prev, long cp_rs,
    T* __restrict__ h_out, long h_rs, float* __restrict__ c_out, long c_rs,
    T* __restrict__ gates_out, long g_rs, unsigned char* __restrict__ h8_out,
    int B, int H, int j0) {
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const int ncol = H - j0;
  if (idx >= (long)B * ncol) return;
  const int b = idx / ncol, j = j0 + idx % ncol;
  const long xo = (long)b * xp_rs + j;
  const long ro = (long)b * rec_rs + j;
  float gi = ld(xp + xo) + ld(rec + ro) + bias[j];
  float gf = ld(xp + xo + H) + ld(rec + ro + H) + bias[j + H];
  float gg = ld(xp + xo + 2 * H) + ld(rec + ro + 2 * H) + bias[j + 2 * H];
  float go = ld(xp + xo + 3 * H) + ld(rec + ro + 3 * H) + bias[j + 3 * H];
  gi = sigmoidf_(gi); gf = sigmoidf_(gf); gg = tanhf(gg); go = sigmoidf_(go);
  const float c = gf * c_prev[(long)b * cp_rs + j] + gi * gg;
  const float h = go * tanhf(c);
  st(h_out + (long)b * h_rs + j, h);
  if (h8_out != nullptr)
    h8_out[(long)b * h_rs + j] = __hip_fp8_e4m3(h * kH8Scale).__x;
  c_out[(long)b * c_rs + j] = c;
  const long gout = (long)b * g_rs + j;
  st(gates_out + gout, gi);
  st(gates_out + gout + H, gf);
  st(gates_out + gout + 2 * H, gg);
  st(gates_out + gout + 3 * H, go);
}

// backward pointwise: consumes dh_ext (upstream) + dh_rec (t+1 GEMM),
// running dc (fp32, in/out), saved post-act gates, c_{t-1}, c_t.
template <typename T, bool VECTOR>
__global__ void lstm_cell_bwd(
    const T* __restrict__ dh_ext, long dhe_rs,
    const T* __restrict__ dh_rec, long dhr_rs,
    float* __restrict__ dc_buf, long dc_rs,    // in: dc_t ; out: dc_{t-1}
    const T* __restrict__ gates, long g_rs,
    const float* __restrict__ c_prev, long cp_rs,
    const float* __restrict__ c_t, long ct_rs,
    T* __restrict__ dgates, long dg_rs,
    int B, int H) {
  constexpr int VEC = VECTOR ? 16 / (int)sizeof(T) : 1;
  const int Hv = H / VEC;
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= (long)B * Hv) return;
  const int b = idx / Hv, j = (idx % Hv) * VEC;
  const long go_ = (long)b * g_rs + j;
  float gi[VEC], gf[VEC], gg[VEC], gout[VEC], dh[VEC], tmp[VEC];
  float ct[VEC], cprev[VEC], dcin[VEC];
  if (VECTOR) {
    ldv<T, VEC>(gates + go_, gi);
    ldv<T, VEC>(gates + go_ + H, gf);
    ldv<T, VEC>(gates + go_ + 2 * H, gg);
    ldv<T, VEC>(gates + go_ + 3 * H, gout);
    ldv<T, VEC>(dh_ext + (long)b * dhe_rs + j, dh);
    ldv<T, VEC>(dh_rec + (long)b * dhr_rs + j, tmp);
    ldv_f32<VEC>(c_t + (long)b * ct_rs + j, ct);
    ldv_f32<VEC>(c_prev + (long)b * cp_rs + j, cprev);
    ldv_f32<VEC>(dc_buf + (long)b * dc_rs + j, dcin);
  } else {
    gi[0] = ld(gates + go_); gf[0] = ld(gates + go_ + H);
    gg[0] = ld(gates + go_ + 2 * H); gout[0] = ld(gates + go_ + 3 * H);
    dh[0] = ld(dh_ext + (long)b * dhe_rs + j);
    tmp[0] = ld(dh_rec + (long)b * dhr_rs + j);
    ct[0] = c_t[(long)b * ct_rs + j];
    cprev[0] = c_prev[(long)b * cp_rs + j];
    dcin[0] = dc_buf[(long)b * dc_rs + j];
  }
  float dgi[VEC], dgf[VEC], dgg[VEC], dgo[VEC], dcout[VEC];
  #pragma unroll
  for (int e = 0; e < VEC; ++e) {
    const float dhe = dh[e] + tmp[e];
    const float tct = tanhf(ct[e]);
    const float do_ = dhe * tct;
    const float dct = dcin[e] + dhe * gout[e] * (1.f - tct * tct);
    dgi[e] = dct * gg[e] * gi[e] * (1.f - gi[e]);
    dgf[e] = dct * cprev[e] * gf[e] * (1.f - gf[e]);
    dgg[e] = dct * gi[e] * (1.f - gg[e] * gg[e]);
    dgo[e] = do_ * gout[e] * (1.f - gout[e]);
    dcout[e] = dct * gf[e];
  }
  const long dgo_ = (long)b * dg_rs + j;
  if (VECTOR) {
    stv_f32<VEC>(dc_buf + (long)b * dc_rs + j, dcout);
    stv<T, VEC>(dgates + dgo_, dgi);
    stv<T, VEC>(dgates + dgo_ + H, dgf);
    stv<T, VEC>(dgates + dgo_ + 2 * H, dgg);
    stv<T, VEC>(dgates + dgo_ + 3 * H, dgo);
  } else {
    dc_buf[(long)b * dc_rs + j] = dcout[0];
    st(dgates + dgo_, dgi[0]);
    st(dgates + dgo_ + H, dgf[0]);
    st(dgates + dgo_ + 2 * H, dgg[0]);
    st(dgates + dgo_ + 3 * H, dgo[0]);
  }
}

template <typename ST>
static void launch_fwd_step(const at::Tensor& xp, const at::Tensor& bias,
                            const at::Tensor& rec, const at::Tensor& c_prev,
                            long cp_off, long cp_rs, at::Tensor& hs,
                            at::Tensor& cs, at::Tensor& gates, int t, int B,
                            int T, int H, unsigned char* h8_base = nullptr) {
  // TIME-MAJOR layout: xp/hs/cs/gates are (T, B, ·) contiguous, so slice t
  // is a contiguous (B, ·) block — hipBLASLt sees contiguous operands and
  // the cell kernel gets unit row strides.
  // 64-thread blocks quadruple the workgroup count (the deployed shape
  // yields only 600 WGs at 256 threads = 2.3/CU and the kernel is
  // latency-bound; measured 432.3 vs 435.2 ms/step e2e at 64 vs 256) —
  // CI_CELL_THREADS overrides.
  static const int threads = [] {
    const char* e = getenv("CI_CELL_THREADS");
    return e ? atoi(e) : 64;
  }();
  constexpr int VEC = 16 / sizeof(ST);
  const int Hv = H / VEC;
  const ST* xpp = reinterpret_cast<const ST*>(xp.data_ptr()) + (long)t * B * 4 * H;
  const ST* recp = reinterpret_cast<const ST*>(rec.data_ptr());
  ST* hp = reinterpret_cast<ST*>(hs.data_ptr()) + (long)t * B * H;
  float* cp = cs.data_ptr<float>() + (long)t * B * H;
  ST* gp = reinterpret_cast<ST*>(gates.data_ptr()) + (long)t * B * 4 * H;
  unsigned char* h8p = h8_base ? h8_base + (long)t * B * H : nullptr;
  if (Hv > 0) {
    hipLaunchKernelGGL((lstm_cell_fwd<ST>),
        dim3(ceil_div((long)B * Hv, threads)), dim3(threads), 0, stream(),
        xpp, (long)4 * H, recp, (long)4 * H, bias.data_ptr<float>(),
        c_prev.data_ptr<float>() + cp_off, cp_rs,
        hp, (long)H, cp, (long)H, gp, (long)4 * H, h8p, B, H);
  }
  if (H % VEC) {
    const int j0 = Hv * VEC;
    hipLaunchKernelGGL((lstm_cell_fwd_tail<ST>),
        dim3(ceil_div((long)B * (H - j0), threads)), dim3(threads), 0, stream(),
        xpp, (long)4 * H, recp, (long)4 * H, bias.data_ptr<float>(),
        c_prev.data_ptr<float>() + cp_off, cp_rs,
        hp, (long)H, cp, (long)H, gp, (long)4 * H, h8p, B, H, j0);
  }
}

// hs,cs,gates are (T,B,·) preallocated; xp (T,B,4H); h0 (B,H); c0 fp32 (B,H).
void lstm_seq_forward_lib(at::Tensor xp, at::Tensor bias, at::Tensor h0,
                          at::Tensor c0, at::Tensor w_hh, at::Tensor hs,
                          at::Tensor cs, at::Tensor gates) {
  CI_CHECK_CUDA(xp); CI_CHECK_CONTIG(xp); CI_CHECK_CONTIG(hs);
  CI_CHECK_CONTIG(cs); CI_CHECK_CONTIG(gates);
  const int T = xp.size(0), B = xp.size(1);
  const int H = w_hh.size(1);
  auto w_hh_t = w_hh.t();
  auto rec = at::empty({B, 4 * H}, xp.options());
  CI_DISPATCH_FB(xp.scalar_type(), "lstm_seq_forward_lib", [&] {
    for (int t = 0; t < T; ++t) {
      auto h_prev = (t == 0) ? h0 : hs.select(0, t - 1);
      at::mm_out(rec, h_prev, w_hh_t);
      if (t == 0) {
        launch_fwd_step<scalar_t>(xp, bias, rec, c0, 0, H, hs, cs, gates, t, B, T, H);
      } else {
        launch_fwd_step<scalar_t>(xp, bias, rec, cs, (long)(t - 1) * B * H,
                                  (long)H, hs, cs, gates, t, B, T, H);
      }
    }
  });
}

// fp8 recurrent GEMM variant (CI_LSTM_FP8): the per-timestep GEMM runs
// at::_scaled_mm with an e4m3 weight (quantized once per layer-forward by
// the caller) and the e4m3 hidden state the cell kernel emitted at t-1
// with the fixed 1/448 scale (|h| < 1). Measured 28.2 vs 35.0 us/call at
// the deployed shape (scripts/lstm_fp8_probe.py). Backward is unchanged
// (bf16 saves).
void lstm_seq_forward_lib_fp8(at::Tensor xp, at::Tensor bias, at::Tensor h0,
                              at::Tensor c0, at::Tensor w8, at::Tensor wscale,
                              at::Tensor hs, at::Tensor cs, at::Tensor gates) {
  CI_CHECK_CUDA(xp); CI_CHECK_CONTIG(xp); CI_CHECK_CONTIG(hs);
  CI_CHECK_CONTIG(cs); CI_CHECK_CONTIG(gates); CI_CHECK_CONTIG(w8);
  TORCH_CHECK(xp.scalar_type() == at::ScalarType::BFloat16,
              "fp8 recurrent path is bf16-activation only");
  TORCH_CHECK(w8.scalar_type() == at::ScalarType::Float8_e4m3fn);
  const int T = xp.size(0), B = xp.size(1);
  const int H = w8.size(1);
  auto w8_t = w8.t();  // (H, 4H) column-major for _scaled_mm mat2
  auto rec = at::empty({B, 4 * H}, xp.options());
  auto hs8 = at::empty({T, B, H},
                       xp.options().dtype(at::ScalarType::Float8_e4m3fn));
  auto h8_scale = at::full({}, 1.0 / 448.0,
                           xp.options().dtype(at::ScalarType::Float));
  // t=0: quantize the incoming h0 with the same fixed scale
  auto h0_8 = at::empty({B, H},
                        xp.options().dtype(at::ScalarType::Float8_e4m3fn));
  quantize_e4m3(h0.contiguous(), h0_8, h8_scale);
  auto* h8_base = reinterpret_cast<unsigned char*>(hs8.data_ptr());
  using ST = __hip_bfloat16;
  for (int t = 0; t < T; ++t) {
    auto h8_prev = (t == 0) ? h0_8 : hs8.select(0, t - 1);
    at::_scaled_mm_out(rec, h8_prev, w8_t, h8_scale, wscale,
                       c10::nullopt, c10::nullopt,
                       at::ScalarType::BFloat16, false);
    if (t == 0) {
      launch_fwd_step<ST>(xp, bias, rec, c0, 0, H, hs, cs, gates, t, B, T, H,
                          h8_base);
    } else {
      launch_fwd_step<ST>(xp, bias, rec, cs, (long)(t - 1) * B * H, (long)H,
                          hs, cs, gates, t, B, T, H, h8_base);
    }
  }
}

// reverse loop over (T,B,·) time-major saves; dh0/dc0 are (B,H) fp32 outs.
void lstm_seq_backward(at::Tensor dhs, at::Tensor dhT, at::Tensor dcT,
                       at::Tensor gates, at::Tensor hs, at::Tensor cs,
                       at::Tensor c0, at::Tensor w_hh, at::Tensor dgates,
                       at::Tensor dh0, at::Tensor dc0) {
  CI_CHECK_CUDA(dhs); CI_CHECK_CONTIG(dhs); CI_CHECK_CONTIG(dgates);
  const int T = dhs.size(0), B = dhs.size(1);
  const int H = w_hh.size(1);
  auto dc_buf = dcT.clone();                       // (B,H) fp32 running dc
  auto dh_rec = dhT.contiguous();                  // (B,H) scalar running rec grad
  // one 46 MB transpose buys the NT GEMM layout for all T steps:
  // mm(dgates, w_hh) NN measured 304 TF vs 503 TF as mm(dgates, w_t.t())
  // (scripts/gemm_probe.py on MI355X)
  auto w_hh_tc = w_hh.t().contiguous();
  auto w_hh_nt = w_hh_tc.t();
  static const int threads = [] {
    const char* e = getenv("CI_CELL_THREADS");
    return e ? atoi(e) : 64;
  }();
  CI_DISPATCH_FB(dhs.scalar_type(), "lstm_seq_backward", [&] {
    constexpr int VEC = 16 / sizeof(scalar_t);
    const bool vec_ok = (H % VEC) == 0;
    const int blocks = ceil_div((long)B * (vec_ok ? H / VEC : H), threads);
    for (int t = T - 1; t >= 0; --t) {
      const float* cprev = (t == 0) ? c0.data_ptr<float>()
                                    : cs.data_ptr<float>() + (long)(t - 1) * B * H;
      auto kern = vec_ok ? lstm_cell_bwd<scalar_t, true>
                         : lstm_cell_bwd<scalar_t, false>;
      hipLaunchKernelGGL(kern, dim3(blocks), dim3(threads), 0, stream(),
          reinterpret_cast<const scalar_t*>(dhs.data_ptr()) + (long)t * B * H, (long)H,
          reinterpret_cast<const scalar_t*>(dh_rec.data_ptr()), (long)H,
          dc_buf.data_ptr<float>(), (long)H,
          reinterpret_cast<const scalar_t*>(gates.data_ptr()) + (long)t * B * 4 * H, (long)4 * H,
          cprev, (long)H,
          cs.data_ptr<float>() + (long)t * B * H, (long)H,
          reinterpret_cast<scalar_t*>(dgates.data_ptr()) + (long)t * B * 4 * H, (long)4 * H,
          B, H);
      // dh_{t-1} recurrent contribution: dgates_t @ w_hh (NT layout,
      // contiguous A thanks to time-major dgates)
      at::mm_out(dh_rec, dgates.select(0, t), w_hh_nt);
    }
  });
  dh0.copy_(dh_rec);
  dc0.copy_(dc_buf);
}

}  // namespace ci
