// K2 fused mode: hand-written CDNA4 MFMA LSTM cell kernel.
//
// Per timestep, ONE kernel computes
//   pre[b, n'] = sum_k h_prev[b,k] * W_hh[row(n'), k]  (+ xp + bias)
// with the output columns GATE-INTERLEAVED: n' = 4*j + g maps to the
// original gate row g*H + j (g in {i,f,g,o}, PyTorch gate order). That puts
// all four gates of hidden unit j in four adjacent columns of the same
// output tile, so the epilogue can finish the whole cell locally:
//   c_t = sigm(f)*c_{t-1} + sigm(i)*tanh(g);  h_t = sigm(o)*tanh(c_t)
// and store h (bf16), c (fp32) and the post-activation gates (for K7
// backward) without a second kernel or a round-trip of the 4H-wide
// pre-activation matrix through HBM.
//
// GEMM structure (cdna_hip_programming.md §5, "step-2/3" class):
//   128x128 tile, BK=64, 4 waves of 64x64, mfma_f32_16x16x32_bf16,
//   double-buffered LDS staging with XOR-swizzled 16B chunks
//   (T2: byte ^= (row&7)<<4 equivalent) read back as b128 fragments.
// The interleave permutation row(n') = (n'&3)*H + (n'>>2) is applied on the
// *global source address* of the W staging loads — W_hh itself stays in the
// checkpoint layout, no pre-permute pass.
//
// Reference op semantics: SURVEY.md §2.4 K2/K3; the weight-drop mask is
// applied by the caller (masked W_hh is what arrives here), matching
// fastai WeightDropout (train.py:70).
#include "common.h"

namespace ci {

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

constexpr int BM = 128, BN = 128, BK = 64;
constexpr int THREADS = 256;            // 4 waves, each owns a 64x64 subtile
constexpr int LDS_ELEMS = BM * BK;      // per operand tile (bf16 elements)

// element offset of (row, col) in a swizzled [rows][BK] bf16 LDS tile:
// 16B chunks within a 128B row are XOR'd by (row&7) — T2 swizzle.
static __device__ __forceinline__ int swz(int row, int col) {
  const int chunk = col >> 3;
  return row * BK + ((chunk ^ (row & 7)) << 3) + (col & 7);
}

// stage a BMxBK bf16 tile from global (row-major, arbitrary row stride,
// optional row permutation for W) into swizzled LDS. 256 threads, each
// moves 4 x 16B chunks. rows beyond row_lim / k beyond k_lim are zeroed.
template <bool PERM>
static __device__ __forceinline__ void stage_tile(
    const __hip_bfloat16* __restrict__ src, long row_stride, int row0,
    int row_lim, int k0, int k_lim, int H, __hip_bfloat16* lds) {
  const int tid = threadIdx.x;
  // thread t: row pair r = t>>1, half = t&1 covers 4 chunks of 8 elems
  const int r = tid >> 1;
  const int cbase = (tid & 1) * 4;  // chunk index base (of 8 per row)
  const int grow_t = row0 + r;
  long grow;
  if (PERM) {
    // output col n' = grow_t ; source row = (n'&3)*H + (n'>>2)
    grow = (long)(grow_t & 3) * H + (grow_t >> 2);
  } else {
    grow = grow_t;
  }
  const bool rok = grow_t < row_lim;
  #pragma unroll
  for (int c = 0; c < 4; ++c) {
    const int chunk = cbase + c;
    const int k = k0 + (chunk << 3);
    bf16x8 v = {};
    if (rok && k + 8 <= k_lim) {
      v = *reinterpret_cast<const bf16x8*>(src + grow * row_stride + k);
    }
    *reinterpret_cast<bf16x8*>(lds + swz(r, chunk << 3)) = v;
  }
}

// fragment loads: lane l reads rows (fr*16 + (l&15)), k ((l>>4)*8) of the
// 64-wide k-slice ks (0 or 1) — 16B contiguous = ds_read_b128.
static __device__ __forceinline__ bf16x8 frag(const __hip_bfloat16* lds,
                                              int row_base, int ks, int lane) {
  const int row = row_base + (lane & 15);
  const int col = ks * 32 + ((lane >> 4) << 3);
  return *reinterpret_cast<const bf16x8*>(lds + swz(row, col));
}

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
  const int NT = gridDim.x / MT;
  // block -> (mt, nt): consecutive ids share nt so the W panel stays hot in
  // one XCD's L2 (placement is a perf hint only).
  const int nt = blockIdx.x / MT;
  const int mt = blockIdx.x % MT;
  const int N = 4 * H, K = H;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = (wave & 1) * 64;    // wave row offset in tile
  const int wn = (wave >> 1) * 64;   // wave col offset

  extern __shared__ __attribute__((aligned(16))) char smem[];
  __hip_bfloat16* lsb = reinterpret_cast<__hip_bfloat16*>(smem);
  // buf 0: [A0 | B0], buf 1: [A1 | B1]
#define LA(buf) (lsb + (buf) * 2 * LDS_ELEMS)
#define LB(buf) (lsb + (buf) * 2 * LDS_ELEMS + LDS_ELEMS)

  f32x4 acc[4][4] = {};

  const int row0_a = mt * BM, row0_b = nt * BN;
  const int nk = (K + BK - 1) / BK;
  stage_tile<false>(h_prev, h_rs, row0_a, B, 0, K, H, LA(0));
  stage_tile<true>(w_hh, K, row0_b, N, 0, K, H, LB(0));
  __syncthreads();

  for (int kt = 0; kt < nk; ++kt) {
    const int cur = kt & 1;
    if (kt + 1 < nk) {
      stage_tile<false>(h_prev, h_rs, row0_a, B, (kt + 1) * BK, K, H, LA(cur ^ 1));
      stage_tile<true>(w_hh, K, row0_b, N, (kt + 1) * BK, K, H, LB(cur ^ 1));
    }
    #pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 af[4], bf[4];
      #pragma unroll
      for (int f = 0; f < 4; ++f) af[f] = frag(LA(cur), wm + f * 16, ks, lane);
      #pragma unroll
      for (int f = 0; f < 4; ++f) bf[f] = frag(LB(cur), wn + f * 16, ks, lane);
      #pragma unroll
      for (int fm = 0; fm < 4; ++fm)
        #pragma unroll
        for (int fn = 0; fn < 4; ++fn)
          acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fm], bf[fn], acc[fm][fn], 0, 0, 0);
    }
    __syncthreads();
  }

  // ---- epilogue: stash pre-activations in LDS, finish the cell ----------
  // fp32 tile [BM][BN] = 64 KiB (fits the default dynamic-LDS cap; the
  // 2-way b32 write conflict this leaves is epilogue-only and cheap)
  float* pre = reinterpret_cast<float*>(smem);
  constexpr int PRS = BN;
  #pragma unroll
  for (int fm = 0; fm < 4; ++fm) {
    #pragma unroll
    for (int fn = 0; fn < 4; ++fn) {
      const int col = wn + fn * 16 + (lane & 15);
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = wm + fm * 16 + ((lane >> 4) << 2) + r;
        pre[row * PRS + col] = acc[fm][fn][r];
      }
    }
  }
  __syncthreads();

  // 128 rows x 32 units per tile; thread p handles (b, j) pairs
  const int jt = nt * 32;  // first hidden unit of this tile
  for (int p = threadIdx.x; p < BM * 32; p += THREADS) {
    const int br = p >> 5;           // row within tile
    const int jj = p & 31;           // unit within tile
    const int b = row0_a + br;
    const int j = jt + jj;
    if (b >= B || j >= H) continue;
    const float* q = pre + br * PRS + jj * 4;
    const long xo = (long)b * xp_rs + j;
    float gi = q[0] + __bfloat162float(xp[xo]) + bias[j];
    float gf = q[1] + __bfloat162float(xp[xo + H]) + bias[j + H];
    float gg = q[2] + __bfloat162float(xp[xo + 2 * H]) + bias[j + 2 * H];
    float go = q[3] + __bfloat162float(xp[xo + 3 * H]) + bias[j + 3 * H];
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
void lstm_seq_forward_fused(at::Tensor xp, at::Tensor bias, at::Tensor h0,
                            at::Tensor c0, at::Tensor w_hh, at::Tensor hs,
                            at::Tensor cs, at::Tensor gates) {
  CI_CHECK_CUDA(xp); CI_CHECK_CONTIG(xp); CI_CHECK_CONTIG(hs);
  CI_CHECK_CONTIG(cs); CI_CHECK_CONTIG(gates);
  TORCH_CHECK(xp.scalar_type() == at::ScalarType::BFloat16,
              "fused LSTM cell kernel is bf16; use CI_LSTM_MODE=lib for fp32");
  TORCH_CHECK(w_hh.is_contiguous(), "w_hh must be contiguous");
  const int B = xp.size(0), T = xp.size(1);
  const int H = w_hh.size(1);
  TORCH_CHECK(H % 8 == 0, "H must be a multiple of 8");
  const int MT = ceil_div(B, BM);
  const dim3 grid(MT * ceil_div(4 * H, BN));
  const size_t lds = std::max((size_t)4 * LDS_ELEMS * sizeof(__hip_bfloat16),
                              (size_t)BM * BN * sizeof(float));
  auto* hsp = reinterpret_cast<__hip_bfloat16*>(hs.data_ptr());
  auto* xpp = reinterpret_cast<const __hip_bfloat16*>(xp.data_ptr());
  auto* gp = reinterpret_cast<__hip_bfloat16*>(gates.data_ptr());
  auto* wp = reinterpret_cast<const __hip_bfloat16*>(w_hh.data_ptr());
  auto h0c = h0.contiguous();
  auto* h0p = reinterpret_cast<const __hip_bfloat16*>(h0c.data_ptr());
  for (int t = 0; t < T; ++t) {
    const __hip_bfloat16* hp = (t == 0) ? h0p : hsp + (long)(t - 1) * H;
    const long h_rs = (t == 0) ? H : (long)T * H;
    const float* cp = (t == 0) ? c0.data_ptr<float>()
                               : cs.data_ptr<float>() + (long)(t - 1) * H;
    const long cp_rs = (t == 0) ? H : (long)T * H;
    hipLaunchKernelGGL(lstm_cell_fused, grid, dim3(THREADS), lds, stream(),
        hp, h_rs, wp, xpp + (long)t * 4 * H, (long)T * 4 * H,
        bias.data_ptr<float>(), cp, cp_rs,
        hsp + (long)t * H, (long)T * H,
        cs.data_ptr<float>() + (long)t * H, (long)T * H,
        gp + (long)t * 4 * H, (long)T * 4 * H, B, H, MT);
  }
}

}  // namespace ci
