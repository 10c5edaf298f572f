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

// ---- persistent whole-sequence variant (CI_SERVE_PERSISTENT) ------------
// One launch per LAYER instead of one per timestep: the grid stays
// resident and a software grid barrier separates timesteps (NOTES r1
// item 2: the cross-timestep persistent grid aiming at the ~3 ms
// weight-stream floor; ~2x T x n_layers launch overheads removed).
// Safety: the spin has a hard cap — on overflow the kernel sets a fail
// flag and EXITS instead of hanging the GPU; the host falls back to the
// per-step path when the flag is set. Grid size = occupancy-derived
// resident capacity (all blocks must be co-resident for the barrier).
//
// MEASURED NEGATIVE (kept as a documented experiment, opt-in only):
// at the deployed shape the barrier itself costs ~50 us/step — the
// generation counter bounces across all 8 XCDs' L2s — so the best
// persistent config (nb=256, 15.4 ms at T=300) is ~4.7x SLOWER than
// per-step launches (3.3 ms). gpurun_out/r2m_pers.log /
// profiles/BENCH_HISTORY.md. Also: hipOccupancyMaxActiveBlocks
// overestimates by 1 block/CU here (claimed 7, 6 resident), hence the
// safety margin below.

__device__ __forceinline__ bool grid_sync_capped(unsigned int* cnt,
                                                 unsigned int* gen,
                                                 unsigned int nb) {
  __syncthreads();
  __shared__ int ok_s;
  if (threadIdx.x == 0) {
    ok_s = 1;
    __threadfence();                       // publish this block's writes
    const unsigned int g = atomicAdd(gen, 0u);   // read generation FIRST
    if (atomicAdd(cnt, 1u) == nb - 1) {
      atomicExch(cnt, 0u);
      __threadfence();
      atomicAdd(gen, 1u);                  // release the cohort
    } else {
      long spins = 0;
      while (atomicAdd(gen, 0u) == g) {
        __builtin_amdgcn_s_sleep(32);
        if (++spins > (1 << 20)) { ok_s = 0; break; }  // bail, don't hang
      }
    }
    __threadfence();                       // acquire: invalidate caches
  }
  __syncthreads();
  return ok_s != 0;
}

__global__ __launch_bounds__(256) void lstm_seq_gemv_persistent(
    const __hip_bfloat16* __restrict__ h0, long h_rs,
    const __hip_bfloat16* __restrict__ w_hh,
    const __hip_bfloat16* __restrict__ xp,
    const float* __restrict__ bias,
    const float* __restrict__ c0, long cp_rs,
    __hip_bfloat16* __restrict__ hs,
    float* __restrict__ cs,
    __hip_bfloat16* __restrict__ gates_out,
    unsigned int* __restrict__ barrier_ws,   // [cnt, gen]
    int* __restrict__ fail_flag,
    int B, int H, int T) {
  constexpr int BMAX = 8;
  const unsigned int NB = gridDim.x;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int Hv = H / 8 * 8;
  __shared__ float dots[BMAX][4];
  for (int t = 0; t < T; ++t) {
    const __hip_bfloat16* hp = (t == 0) ? h0 : hs + (long)(t - 1) * B * H;
    const float* cp = (t == 0) ? c0 : cs + (long)(t - 1) * B * H;
    const long h_prev_rs = (t == 0) ? h_rs : (long)H;
    const long c_prev_rs = (t == 0) ? cp_rs : (long)H;
    const __hip_bfloat16* xpt = xp + (long)t * B * 4 * H;
    __hip_bfloat16* h_out = hs + (long)t * B * H;
    float* c_out = cs + (long)t * B * H;
    __hip_bfloat16* g_out = gates_out + (long)t * B * 4 * H;
    for (int j = blockIdx.x; j < H; j += NB) {
      const __hip_bfloat16* wrow = w_hh + (long)(wave * H + j) * H;
      float acc[BMAX];
      #pragma unroll
      for (int b = 0; b < BMAX; ++b) acc[b] = 0.f;
      for (int k = lane * 8; k < Hv; k += 64 * 8) {
        bf16x8g wv = *reinterpret_cast<const bf16x8g*>(wrow + k);
        #pragma unroll
        for (int b = 0; b < BMAX; ++b) {
          if (b >= B) break;
          bf16x8g hv = *reinterpret_cast<const bf16x8g*>(
              hp + (long)b * h_prev_rs + k);
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
          acc[b] += wv * ld(hp + (long)b * h_prev_rs + k);
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
        const long xo = (long)b * 4 * H + j;
        const float gi = sigmoidf_(dots[b][0] + ld(xpt + xo) + bias[j]);
        const float gf = sigmoidf_(dots[b][1] + ld(xpt + xo + H) + bias[j + H]);
        const float gg = tanhf(dots[b][2] + ld(xpt + xo + 2 * H) + bias[j + 2 * H]);
        const float go = sigmoidf_(dots[b][3] + ld(xpt + xo + 3 * H) + bias[j + 3 * H]);
        const float c = gf * cp[(long)b * c_prev_rs + j] + gi * gg;
        const float h = go * tanhf(c);
        st(h_out + (long)b * H + j, h);
        c_out[(long)b * H + j] = c;
        const long g0 = (long)b * 4 * H + j;
        st(g_out + g0, gi);
        st(g_out + g0 + H, gf);
        st(g_out + g0 + 2 * H, gg);
        st(g_out + g0 + 3 * H, go);
      }
      __syncthreads();  // dots reused by the next j of this block
    }
    if (!grid_sync_capped(barrier_ws, barrier_ws + 1, NB)) {
      if (threadIdx.x == 0) atomicExch(fail_flag, 1);
      return;
    }
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


// persistent driver: ONE launch for the whole sequence. Returns the grid
// size used, or 0 when a co-resident grid is not available (caller falls
// back to the per-step driver). ws = int32[4] zeroed workspace
// [cnt, gen, fail, pad]; caller checks ws[2] after the stream syncs.
long lstm_seq_forward_gemv_persistent(at::Tensor xp, at::Tensor bias,
                                      at::Tensor h0, at::Tensor c0,
                                      at::Tensor w_hh, at::Tensor hs,
                                      at::Tensor cs, at::Tensor gates,
                                      at::Tensor ws) {
  CI_CHECK_CUDA(xp); CI_CHECK_CONTIG(xp); CI_CHECK_CONTIG(hs);
  CI_CHECK_CONTIG(cs); CI_CHECK_CONTIG(gates); CI_CHECK_CONTIG(ws);
  TORCH_CHECK(xp.scalar_type() == at::ScalarType::BFloat16);
  TORCH_CHECK(ws.numel() >= 4 && ws.scalar_type() == at::ScalarType::Int);
  const int T = xp.size(0), B = xp.size(1);
  TORCH_CHECK(B <= 8, "gemv persistent kernel is for B <= 8");
  const int H = w_hh.size(1);
  int per_cu = 0;
  if (hipOccupancyMaxActiveBlocksPerMultiprocessor(
          &per_cu, reinterpret_cast<const void*>(&lstm_seq_gemv_persistent),
          256, 0) != hipSuccess || per_cu < 1) {
    return 0;
  }
  const int cus = at::cuda::getCurrentDeviceProperties()->multiProcessorCount;
  per_cu = std::max(1, per_cu - 1);  // measured: API claims 7/CU, 6 resident
  long nb = std::min<long>((long)per_cu * cus, H);
  if (const char* env = getenv("CI_PERS_NB")) {   // residency experiments
    nb = std::min<long>(std::max(1L, atol(env)), H);
  }
  auto h0c = h0.contiguous();
  hipLaunchKernelGGL(lstm_seq_gemv_persistent, dim3(nb), dim3(256), 0,
      stream(),
      reinterpret_cast<const __hip_bfloat16*>(h0c.data_ptr()), (long)H,
      reinterpret_cast<const __hip_bfloat16*>(w_hh.data_ptr()),
      reinterpret_cast<const __hip_bfloat16*>(xp.data_ptr()),
      bias.data_ptr<float>(), c0.data_ptr<float>(), (long)H,
      reinterpret_cast<__hip_bfloat16*>(hs.data_ptr()),
      cs.data_ptr<float>(),
      reinterpret_cast<__hip_bfloat16*>(gates.data_ptr()),
      reinterpret_cast<unsigned int*>(ws.data_ptr()),
      reinterpret_cast<int*>(ws.data_ptr()) + 2, B, H, T);
  return nb;
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
