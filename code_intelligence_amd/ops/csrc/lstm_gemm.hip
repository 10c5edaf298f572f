// K2 fused mode: hand-written CDNA4 MFMA LSTM cell kernel (v2).
//
// Per timestep, ONE kernel computes the recurrent GEMM
//   pre[b, n'] = sum_k h_prev[b,k] * W_hh[row(n'), k]
// with output columns GATE-INTERLEAVED (n' = 4*j + g -> original gate row
// g*H + j, PyTorch gate order i,f,g,o) so all four gates of hidden unit j
// land in adjacent columns of one tile and the epilogue finishes the cell
// locally: c_t = sigm(f)*c + sigm(i)*tanh(g); h_t = sigm(o)*tanh(c_t),
// storing h (bf16), c (fp32) and post-activation gates (for K7 backward)
// with no extra kernel and no HBM round-trip of the 4H-wide pre-matrix.
//
// v2 structure (cdna_hip_programming.md par.5 "step-3" class, adapted):
//  * 128x64 tile, BK=64, 4 waves of 64x32, mfma_f32_16x16x32_bf16
//    -> 600 blocks at the deployed shape (B=512, H=2400): ~2.3 blocks/CU
//    (the v1 128x128 tile gave only 300 blocks = 1.17/CU, latency-bound)
//  * interior tiles staged by __builtin_amdgcn_global_load_lds (16 B,
//    lane-linear LDS dest; the T2 XOR swizzle moves to the per-lane SOURCE
//    address), double-buffered, issued BEFORE the MFMAs of the current
//    tile so the DMA flight hides under compute
//  * edge tiles (batch/N/K tails) fall back to register staging with the
//    same swizzled LDS image
//  * XCD-aware bijective blockIdx remap (T1): the MT blocks sharing one
//    W-panel run on one XCD so the panel stays in that XCD's L2
//  * the W interleave permutation row(n') = (n'&3)*H + (n'>>2) is applied
//    on the staging source address; W_hh stays in checkpoint layout.
//
// Weight-drop (K3) masks are applied by the caller (masked W arrives
// here), matching fastai WeightDropout (train.py:70).
#include "common.h"

namespace ci {

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

constexpr int BK = 64;
constexpr int THREADS = 256;             // 4 waves: 2(M) x 2(N)

// element offset of (row, col) in the swizzled [rows][BK] bf16 LDS tile:
// 16B chunks within a 128B row are XOR'd by (row&7) (T2).
static __device__ __forceinline__ int swz(int row, int col) {
  const int chunk = col >> 3;
  return row * BK + ((chunk ^ (row & 7)) << 3) + (col & 7);
}

template <bool PERM>
static __device__ __forceinline__ long src_row(int abs_row, int H) {
  return PERM ? (long)(abs_row & 3) * H + (abs_row >> 2) : abs_row;
}

// ---- glds staging: linear LDS dest, swizzle folded into the source ----
// R rows x 64 cols bf16 = R*8 16B slots; one wave-instruction covers 64
// slots (lane-linear). slot s16 -> row = s16>>3, chunk = (s16&7) ^ (row&7).
template <int R, bool PERM>
static __device__ __forceinline__ void stage_glds(
    const __hip_bfloat16* __restrict__ src, long row_stride, int row0,
    int k0, int H, __hip_bfloat16* lds) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  #pragma unroll
  for (int i = 0; i < R / 32; ++i) {
    const int s16 = i * 256 + wave * 64 + lane;
    const int row = s16 >> 3;
    const int chunk = (s16 & 7) ^ (row & 7);
    const long grow = src_row<PERM>(row0 + row, H);
    const __hip_bfloat16* g = src + grow * row_stride + k0 + chunk * 8;
    // C-style casts switch address space (generic->global/LDS) — the
    // pattern ck_tile uses for llvm.amdgcn.*.load.lds operands
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)(g),
        (__attribute__((address_space(3))) unsigned int*)(
            lds + (long)(i * 256 + wave * 64) * 8),
        16, 0, 0);
  }
}

// ---- register staging fallback for edge tiles (zero-filled) -----------
template <int R, bool PERM>
static __device__ __forceinline__ void stage_reg(
    const __hip_bfloat16* __restrict__ src, long row_stride, int row0,
    int row_lim, int k0, int k_lim, int H, __hip_bfloat16* lds) {
  // 256 threads move R*8 slots; thread t handles slots t, t+256, ...
  #pragma unroll
  for (int i = 0; i < R / 32; ++i) {
    const int s16 = i * 256 + threadIdx.x;
    const int row = s16 >> 3;
    const int x = s16 & 7;
    const int chunk = x ^ (row & 7);
    const int abs_row = row0 + row;
    const int k = k0 + chunk * 8;
    bf16x8 v = {};
    if (abs_row < row_lim && k + 8 <= k_lim) {
      v = *reinterpret_cast<const bf16x8*>(
          src + src_row<PERM>(abs_row, H) * row_stride + k);
    }
    *reinterpret_cast<bf16x8*>(lds + (long)s16 * 8) = v;
  }
}

static __device__ __forceinline__ bf16x8 frag(const __hip_bfloat16* lds,
                                              int row_base, int ks, int lane) {
  const int row = row_base + (lane & 15);
  const int col = ks * 32 + ((lane >> 4) << 3);
  return *reinterpret_cast<const bf16x8*>(lds + swz(row, col));
}

// counted wait: vmcnt <= n outstanding VMEM ops, all LDS ops drained
// (lgkmcnt=0 keeps reg-staged edge tiles correct), expcnt unconstrained.
// CDNA s_waitcnt imm: vmcnt[3:0]+[15:14], expcnt[6:4], lgkmcnt[13:8].
template <int N>
static __device__ __forceinline__ void waitcnt_vm() {
  __builtin_amdgcn_s_waitcnt((0x7 << 4) | (N & 0xF) | (((N >> 4) & 0x3) << 14));
}

template <int BM, int BN, int STAGES = 2>
__global__ __launch_bounds__(THREADS) void lstm_cell_fused(
    const __hip_bfloat16* __restrict__ h_prev, long h_rs,
    const __hip_bfloat16* __restrict__ w_hh,   // (4H, H) checkpoint layout
    const __hip_bfloat16* __restrict__ xp, long xp_rs,  // (B,4H) block layout
    const float* __restrict__ bias,            // (4H) block layout
    const float* __restrict__ c_prev, long cp_rs,
    __hip_bfloat16* __restrict__ h_out, long ho_rs,
    float* __restrict__ c_out, long co_rs,
    __hip_bfloat16* __restrict__ gates_out, long go_rs,  // block layout
    int B, int H, int MT) {
  // T1 bijective XCD remap: blocks sharing a W panel -> one XCD's L2
  const int nwg = gridDim.x;
  const int p = blockIdx.x;
  const int xcd = p % 8, pos = p / 8;
  const int q = nwg / 8, r = nwg % 8;
  const int logical = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  const int nt = logical / MT;
  const int mt = logical % MT;

  const int N = 4 * H, K = H;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = (wave & 1) * (BM / 2);   // wave row offset
  const int wn = (wave >> 1) * (BN / 2);   // wave col offset

  constexpr int A_ELEMS = BM * BK;       // bf16 elements per A tile
  constexpr int B_ELEMS = BN * BK;
  constexpr int FM = BM / 2 / 16;        // 16x16 fragments per wave (M)
  constexpr int FN = BN / 2 / 16;        //                       (N)
  extern __shared__ __attribute__((aligned(16))) char smem[];
  __hip_bfloat16* lsb = reinterpret_cast<__hip_bfloat16*>(smem);
  // buffer b: [A_b (BMx64) | B_b (BNx64)]
#define LA(b) (lsb + (b) * (A_ELEMS + B_ELEMS))
#define LB(b) (lsb + (b) * (A_ELEMS + B_ELEMS) + A_ELEMS)

  f32x4 acc[FM][FN] = {};

  const int row0_a = mt * BM, row0_b = nt * BN;
  const int nk = (K + BK - 1) / BK;
  const bool interior_rows = (row0_a + BM <= B) && (row0_b + BN <= N);

  auto stage = [&](int kt, int buf) {
    const int k0 = kt * BK;
    if (interior_rows && k0 + BK <= K) {
      stage_glds<BM, false>(h_prev, h_rs, row0_a, k0, H, LA(buf));
      stage_glds<BN, true>(w_hh, K, row0_b, k0, H, LB(buf));
    } else {
      stage_reg<BM, false>(h_prev, h_rs, row0_a, B, k0, K, H, LA(buf));
      stage_reg<BN, true>(w_hh, K, row0_b, N, k0, K, H, LB(buf));
    }
  };

  if constexpr (STAGES == 2) {
    stage(0, 0);
    __syncthreads();
    for (int kt = 0; kt < nk; ++kt) {
      const int cur = kt & 1;
      if (kt + 1 < nk) stage(kt + 1, cur ^ 1);  // flight hides under MFMA
      #pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        bf16x8 af[4], bfr[2];
        #pragma unroll
        for (int f = 0; f < 4; ++f) af[f] = frag(LA(cur), wm + f * 16, ks, lane);
        #pragma unroll
        for (int f = 0; f < 2; ++f) bfr[f] = frag(LB(cur), wn + f * 16, ks, lane);
        #pragma unroll
        for (int fm = 0; fm < 4; ++fm)
          #pragma unroll
          for (int fn = 0; fn < 2; ++fn)
            acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                af[fm], bfr[fn], acc[fm][fn], 0, 0, 0);
      }
      __syncthreads();  // drains the glds queue (vmcnt0) + barrier
    }
  } else {
    // 3-stage pipeline: prefetch depth 2 so a tile's glds flight hides
    // under TWO MFMA phases (the 2-stage version stalls ~half the loop on
    // load latency — PMC r1 'occupancy/latency-bound'). Counted
    // s_waitcnt vmcnt(12) tolerates the 12 in-flight loads of the two
    // younger tiles ((BM+BN)/32 = 6 glds per thread per stage).
    stage(0, 0);
    if (nk > 1) stage(1, 1);
    if (nk > 2) stage(2, 2);
    for (int kt = 0; kt < nk; ++kt) {
      const int cur = kt % 3;
      waitcnt_vm<12>();        // stage(kt) retired; kt+1/kt+2 may be in flight
      __builtin_amdgcn_s_barrier();
      #pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        bf16x8 af[4], bfr[2];
        #pragma unroll
        for (int f = 0; f < 4; ++f) af[f] = frag(LA(cur), wm + f * 16, ks, lane);
        #pragma unroll
        for (int f = 0; f < 2; ++f) bfr[f] = frag(LB(cur), wn + f * 16, ks, lane);
        #pragma unroll
        for (int fm = 0; fm < 4; ++fm)
          #pragma unroll
          for (int fn = 0; fn < 2; ++fn)
            acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                af[fm], bfr[fn], acc[fm][fn], 0, 0, 0);
      }
      __builtin_amdgcn_s_barrier();  // all waves done reading buf cur
      if (kt + 3 < nk) stage(kt + 3, cur);  // reuse the freed slot
    }
    waitcnt_vm<0>();  // retire any tail-stage loads before the epilogue
  }

  // ---- epilogue: stash pre-activations in LDS, finish the cell ---------
  float* pre = reinterpret_cast<float*>(smem);   // [BM][BN] fp32
  constexpr int PRS = BN;
  #pragma unroll
  for (int fm = 0; fm < FM; ++fm) {
    #pragma unroll
    for (int fn = 0; fn < FN; ++fn) {
      const int col = wn + fn * 16 + (lane & 15);
      #pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const int row = wm + fm * 16 + ((lane >> 4) << 2) + rr;
        pre[row * PRS + col] = acc[fm][fn][rr];
      }
    }
  }
  __syncthreads();

  // BN cols = BN/4 hidden units; thread p2 handles (b, j) pairs
  const int jt = nt * (BN / 4);
  constexpr int JPT = BN / 4;
  for (int p2 = threadIdx.x; p2 < BM * JPT; p2 += THREADS) {
    const int br = p2 / JPT;          // row within tile
    const int jj = p2 % JPT;          // unit within tile
    const int b = row0_a + br;
    const int j = jt + jj;
    if (b >= B || j >= H) continue;
    const float* qp = pre + br * PRS + jj * 4;
    const long xo = (long)b * xp_rs + j;
    float gi = qp[0] + __bfloat162float(xp[xo]) + bias[j];
    float gf = qp[1] + __bfloat162float(xp[xo + H]) + bias[j + H];
    float gg = qp[2] + __bfloat162float(xp[xo + 2 * H]) + bias[j + 2 * H];
    float go = qp[3] + __bfloat162float(xp[xo + 3 * H]) + bias[j + 3 * H];
    gi = sigmoidf_(gi); gf = sigmoidf_(gf); gg = tanhf(gg); go = sigmoidf_(go);
    const float c = gf * c_prev[(long)b * cp_rs + j] + gi * gg;
    const float h = go * tanhf(c);
    h_out[(long)b * ho_rs + j] = __float2bfloat16(h);
    c_out[(long)b * co_rs + j] = c;
    const long g0 = (long)b * go_rs + j;
    gates_out[g0] = __float2bfloat16(gi);
    gates_out[g0 + H] = __float2bfloat16(gf);
    gates_out[g0 + 2 * H] = __float2bfloat16(gg);
    gates_out[g0 + 3 * H] = __float2bfloat16(go);
  }
}

// driver: whole-sequence forward, one fused launch per timestep.
template <int BM, int BN, int STAGES = 2>
static void run_fused(at::Tensor& xp, at::Tensor& bias, at::Tensor& h0,
                      at::Tensor& c0, at::Tensor& w_hh, at::Tensor& hs,
                      at::Tensor& cs, at::Tensor& gates) {
  const int T = xp.size(0), B = xp.size(1);   // TIME-MAJOR (T,B,·)
  const int H = w_hh.size(1);
  const int MT = ceil_div(B, BM);
  const dim3 grid(MT * ceil_div(4 * H, BN));
  const size_t lds = std::max(
      (size_t)STAGES * (BM + BN) * BK * sizeof(__hip_bfloat16),
      (size_t)BM * BN * sizeof(float));
  if (lds > 64 * 1024) {  // 3-stage config exceeds the default dyn-LDS cap
    static bool attr_set = false;
    if (!attr_set) {
      (void)hipFuncSetAttribute(
          reinterpret_cast<const void*>(&lstm_cell_fused<BM, BN, STAGES>),
          hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds);
      attr_set = true;
    }
  }
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
    hipLaunchKernelGGL((lstm_cell_fused<BM, BN, STAGES>), grid, dim3(THREADS),
        lds, stream(),
        hp, (long)H, wp, xpp + (long)t * B * 4 * H, (long)4 * H,
        bias.data_ptr<float>(), cp, (long)H,
        hsp + (long)t * B * H, (long)H,
        cs.data_ptr<float>() + (long)t * B * H, (long)H,
        gp + (long)t * B * 4 * H, (long)4 * H, B, H, MT);
  }
}

void lstm_seq_forward_fused(at::Tensor xp, at::Tensor bias, at::Tensor h0,
                            at::Tensor c0, at::Tensor w_hh, at::Tensor hs,
                            at::Tensor cs, at::Tensor gates) {
  CI_CHECK_CUDA(xp); CI_CHECK_CONTIG(xp); CI_CHECK_CONTIG(hs);
  CI_CHECK_CONTIG(cs); CI_CHECK_CONTIG(gates);
  TORCH_CHECK(xp.scalar_type() == at::ScalarType::BFloat16,
              "fused LSTM cell kernel is bf16; use CI_LSTM_MODE=lib for fp32");
  TORCH_CHECK(w_hh.is_contiguous(), "w_hh must be contiguous");
  TORCH_CHECK(w_hh.size(1) % 8 == 0, "H must be a multiple of 8");
  // Measured at the deployed shape (BENCH_HISTORY r2): the 3-stage
  // pipeline LOSES end-to-end (468 vs 443 ms/step) — its 74 KB LDS
  // footprint drops occupancy from 3 blocks/CU to 2, and the lost
  // block-level overlap outweighs the deeper intra-block pipeline. The
  // 2-stage schedule stays the default; CI_FUSED_PIPE=3 selects the
  // experiment.
  const char* tile = getenv("CI_FUSED_TILE");
  const char* pipe = getenv("CI_FUSED_PIPE");
  const bool p3 = pipe != nullptr && std::string(pipe) == "3";
  if (tile && std::string(tile) == "64x64") {
    if (p3) run_fused<64, 64, 3>(xp, bias, h0, c0, w_hh, hs, cs, gates);
    else    run_fused<64, 64, 2>(xp, bias, h0, c0, w_hh, hs, cs, gates);
  } else {
    if (p3) run_fused<128, 64, 3>(xp, bias, h0, c0, w_hh, hs, cs, gates);
    else    run_fused<128, 64, 2>(xp, bias, h0, c0, w_hh, hs, cs, gates);
  }
}

}  // namespace ci
