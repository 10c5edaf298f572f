// K2 serve variant: fused GEMV + cell kernel for small batches
// (SURVEY.md §2.4 "persistent-LSTM variant for small B serve path").
//
// At B<=8 (single-request serving) the recurrent step is a GEMV bound by
// streaming the 46 MB weight matrix; tiling for MFMA is pointless. One
// block per hidden unit j: its four waves compute the four gate dot
// products <h_b, W[g*H+j,:]> with lane-split K (coalesced 16-B row reads,
// wave shfl reduction), then lane 0 finishes the cell and stores
// h/c/gates — a single launch per timestep replaces hipBLASLt GEMV +
// pointwise kernel. Grid = H blocks (2400 at the deployed shape) fills
// all 256 CUs; batch rows loop inside the block (B tiny).
#include "common.h"

namespace ci {

typedef __bf16 bf16x8g __attribute__((ext_vector_type(8)));

__global__ __launch_bounds__(256) void lstm_cell_gemv(
    const __hip_bfloat16* __restrict__ h_prev, long h_rs,
    const __hip_bfloat16* __restrict__ w_hh,   // (4H, H) row-major
    const __hip_bfloat16* __restrict__ xp, long xp_rs,
    const float* __restrict__ bias,
    const float* __restrict__ c_prev, long cp_rs,
    __hip_bfloat16* __restrict__ h_out, long ho_rs,
    float* __restrict__ c_out, long co_rs,
    __hip_bfloat16* __restrict__ gates_out, long go_rs,
    int B, int H) {
  constexpr int BMAX = 8;            // dispatch guarantees B <= 8
  const int j = blockIdx.x;          // hidden unit
  const int wave = threadIdx.x >> 6; // gate g in {i,f,g,o}
  const int lane = threadIdx.x & 63;
  const __hip_bfloat16* wrow = w_hh + (long)(wave * H + j) * H;
  __shared__ float dots[BMAX][4];
  const int Hv = H / 8 * 8;
  // read each W element ONCE; accumulate all batch rows simultaneously
  // (h rows are tiny and L2-resident; W is the 46 MB stream)
  float acc[BMAX];
  #pragma unroll
  for (int b = 0; b < BMAX; ++b) acc[b] = 0.f;
  // acc[] must be register-resident: unroll over BMAX with a guard so
  // every index is compile-time (runtime-indexed arrays go to scratch —
  // CDNA guide §5.4 rule 20)
  for (int k = lane * 8; k < Hv; k += 64 * 8) {
    bf16x8g wv = *reinterpret_cast<const bf16x8g*>(wrow + k);
    #pragma unroll
    for (int b = 0; b < BMAX; ++b) {
      if (b >= B) break;
      bf16x8g hv = *reinterpret_cast<const bf16x8g*>(
          h_prev + (long)b * h_rs + k);
      #pragma unroll
      for (int e = 0; e < 8; ++e)
        acc[b] += (float)wv[e] * (float)hv[e];
    }
  }
  for (int k = Hv + lane; k < H; k += 64) {
    const float wv = ld(wrow + k);
    #pragma unroll
    for (int b = 0; b < BMAX; ++b) {
      if (b >= B) break;
      acc[b] += wv * ld(h_prev + (long)b * h_rs + k);
    }
  }
  #pragma unroll
  for (int b = 0; b < BMAX; ++b) {
    if (b >= B) break;
    float a = acc[b];
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1)
      a += __shfl_down(a, off);
    if (lane == 0) dots[b][wave] = a;
  }
  __syncthreads();
  for (int b = threadIdx.x; b < B; b += blockDim.x) {
    const long xo = (long)b * xp_rs + j;
    const float gi = sigmoidf_(dots[b][0] + ld(xp + xo) + bias[j]);
    const float gf = sigmoidf_(dots[b][1] + ld(xp + xo + H) + bias[j + H]);
    const float gg = tanhf(dots[b][2] + ld(xp + xo + 2 * H) + bias[j + 2 * H]);
    const float go = sigmoidf_(dots[b][3] + ld(xp + xo + 3 * H) + bias[j + 3 * H]);
    const float c = gf * c_prev[(long)b * cp_rs + j] + gi * gg;
    const float h = go * tanhf(c);
    st(h_out + (long)b * ho_rs + j, h);
    c_out[(long)b * co_rs + j] = c;
    const long g0 = (long)b * go_rs + j;
    st(gates_out + g0, gi);
    st(gates_out + g0 + H, gf);
    st(gates_out + g0 + 2 * H, gg);
    st(gates_out + g0 + 3 * H, go);
  }
}

// whole-sequence driver over time-major (T,B,·) saves, one launch/step.
void lstm_seq_forward_gemv(at::Tensor xp, at::Tensor bias, at::Tensor h0,
                           at::Tensor c0, at::Tensor w_hh, at::Tensor hs,
                           at::Tensor cs, at::Tensor gates) {
  CI_CHECK_CUDA(xp); CI_CHECK_CONTIG(xp); CI_CHECK_CONTIG(hs);
  CI_CHECK_CONTIG(cs); CI_CHECK_CONTIG(gates);
  TORCH_CHECK(xp.scalar_type() == at::ScalarType::BFloat16,
              "gemv cell kernel is bf16");
  const int T = xp.size(0), B = xp.size(1);
  TORCH_CHECK(B <= 8, "gemv cell kernel is for B <= 8 (got ", B, ")");
  const int H = w_hh.size(1);
  auto* hsp = reinterpret_cast<__hip_bfloat16*>(hs.data_ptr());
  auto* xpp = reinterpret_cast<const __hip_bfloat16*>(xp.data_ptr());
  auto* gp = reinterpret_cast<__hip_bfloat16*>(gates.data_ptr());
  auto* wp = reinterpret_cast<const __hip_bfloat16*>(w_hh.data_ptr());
  auto h0c = h0.contiguous();
  auto* h0p = reinterpret_cast<const __hip_bfloat16*>(h0c.data_ptr());
  for (int t = 0; t < T; ++t) {
    const __hip_bfloat16* hp = (t == 0) ? h0p : hsp + (long)(t - 1) * B * H;
    const float* cp = (t == 0) ? c0.data_ptr<float>()
                               : cs.data_ptr<float>() + (long)(t - 1) * B * H;
    hipLaunchKernelGGL(lstm_cell_gemv, dim3(H), dim3(256), 0, stream(),
        hp, (long)H, wp, xpp + (long)t * B * 4 * H, (long)4 * H,
        bias.data_ptr<float>(), cp, (long)H,
        hsp + (long)t * B * H, (long)H,
        cs.data_ptr<float>() + (long)t * B * H, (long)H,
        gp + (long)t * B * 4 * H, (long)4 * H, B, H);
  }
}


// ---- fp8 (OCP e4m3) weight variant --------------------------------------
// Serving is bound by streaming W_hh (46 MB/layer-step at the deployed
// shape); storing W as e4m3 with per-row scales halves that stream.
// Hardware unpack: __builtin_amdgcn_cvt_pk_f32_fp8 converts packed fp8
// pairs at VALU rate. Opt-in (CI_SERVE_FP8W=1), eval-path only.
// fp8x4_to_f32 unpack helper now lives in common.h (shared with ce.hip)

__global__ __launch_bounds__(256) void lstm_cell_gemv_fp8(
    const __hip_bfloat16* __restrict__ h_prev, long h_rs,
    const unsigned char* __restrict__ w8,      // (4H, H) e4m3
    const float* __restrict__ wscale,          // (4H) per-row scales
    const __hip_bfloat16* __restrict__ xp, long xp_rs,
    const float* __restrict__ bias,
    const float* __restrict__ c_prev, long cp_rs,
    __hip_bfloat16* __restrict__ h_out, long ho_rs,
    float* __restrict__ c_out, long co_rs,
    __hip_bfloat16* __restrict__ gates_out, long go_rs,
    int B, int H) {
  constexpr int BMAX = 8;
  const int j = blockIdx.x;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int row = wave * H + j;
  const unsigned char* wrow = w8 + (long)row * H;
  __shared__ float dots[BMAX][4];
  const int Hv = H / 16 * 16;
  float acc[BMAX];
  #pragma unroll
  for (int b = 0; b < BMAX; ++b) acc[b] = 0.f;
  float wf[16];
  for (int k = lane * 16; k < Hv; k += 64 * 16) {
    const uint4 wq = *reinterpret_cast<const uint4*>(wrow + k);  // 16 fp8
    fp8x4_to_f32(wq.x, wf);
    fp8x4_to_f32(wq.y, wf + 4);
    fp8x4_to_f32(wq.z, wf + 8);
    fp8x4_to_f32(wq.w, wf + 12);
    #pragma unroll
    for (int b = 0; b < BMAX; ++b) {
      if (b >= B) break;
      const bf16x8g h0v = *reinterpret_cast<const bf16x8g*>(
          h_prev + (long)b * h_rs + k);
      const bf16x8g h1v = *reinterpret_cast<const bf16x8g*>(
          h_prev + (long)b * h_rs + k + 8);
      #pragma unroll
      for (int e = 0; e < 8; ++e) acc[b] += wf[e] * (float)h0v[e];
      #pragma unroll
      for (int e = 0; e < 8; ++e) acc[b] += wf[8 + e] * (float)h1v[e];
    }
  }
  for (int k = Hv + lane; k < H; k += 64) {
    float w1[4];
    fp8x4_to_f32((unsigned int)wrow[k], w1);  // low byte -> w1[0]
    #pragma unroll
    for (int b = 0; b < BMAX; ++b) {
      if (b >= B) break;
      acc[b] += w1[0] * ld(h_prev + (long)b * h_rs + k);
    }
  }
  const float sc = wscale[row];
  #pragma unroll
  for (int b = 0; b < BMAX; ++b) {
    if (b >= B) break;
    float a = acc[b] * sc;
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1)
      a += __shfl_down(a, off);
    if (lane == 0) dots[b][wave] = a;
  }
  __syncthreads();
  for (int b = threadIdx.x; b < B; b += blockDim.x) {
    const long xo = (long)b * xp_rs + j;
    const float gi = sigmoidf_(dots[b][0] + ld(xp + xo) + bias[j]);
    const float gf = sigmoidf_(dots[b][1] + ld(xp + xo + H) + bias[j + H]);
    const float gg = tanhf(dots[b][2] + ld(xp + xo + 2 * H) + bias[j + 2 * H]);
    const float go = sigmoidf_(dots[b][3] + ld(xp + xo + 3 * H) + bias[j + 3 * H]);
    const float c = gf * c_prev[(long)b * cp_rs + j] + gi * gg;
    const float h = go * tanhf(c);
    st(h_out + (long)b * ho_rs + j, h);
    c_out[(long)b * co_rs + j] = c;
    const long g0 = (long)b * go_rs + j;
    st(gates_out + g0, gi);
    st(gates_out + g0 + H, gf);
    st(gates_out + g0 + 2 * H, gg);
    st(gates_out + g0 + 3 * H, go);
  }
}

void lstm_seq_forward_gemv_fp8(at::Tensor xp, at::Tensor bias, at::Tensor h0,
                               at::Tensor c0, at::Tensor w8, at::Tensor wscale,
                               at::Tensor hs, at::Tensor cs, at::Tensor gates) {
  CI_CHECK_CUDA(xp); CI_CHECK_CONTIG(xp); CI_CHECK_CONTIG(hs);
  CI_CHECK_CONTIG(cs); CI_CHECK_CONTIG(gates); CI_CHECK_CONTIG(w8);
  const int T = xp.size(0), B = xp.size(1);
  TORCH_CHECK(B <= 8, "fp8 gemv kernel is for B <= 8");
  const int H = w8.size(1);
  auto* hsp = reinterpret_cast<__hip_bfloat16*>(hs.data_ptr());
  auto* xpp = reinterpret_cast<const __hip_bfloat16*>(xp.data_ptr());
  auto* gp = reinterpret_cast<__hip_bfloat16*>(gates.data_ptr());
  auto h0c = h0.contiguous();
  auto* h0p = reinterpret_cast<const __hip_bfloat16*>(h0c.data_ptr());
  for (int t = 0; t < T; ++t) {
    const __hip_bfloat16* hp = (t == 0) ? h0p : hsp + (long)(t - 1) * B * H;
    const float* cp = (t == 0) ? c0.data_ptr<float>()
                               : cs.data_ptr<float>() + (long)(t - 1) * B * H;
    hipLaunchKernelGGL(lstm_cell_gemv_fp8, dim3(H), dim3(256), 0, stream(),
        hp, (long)H, w8.data_ptr<unsigned char>(), wscale.data_ptr<float>(),
        xpp + (long)t * B * 4 * H, (long)4 * H,
        bias.data_ptr<float>(), cp, (long)H,
        hsp + (long)t * B * H, (long)H,
        cs.data_ptr<float>() + (long)t * B * H, (long)H,
        gp + (long)t * B * 4 * H, (long)4 * H, B, H);
  }
}

}  // namespace ci
