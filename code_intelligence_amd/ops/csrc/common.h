// Common helpers for the code_intelligence_amd gfx950 HIP kernels.
// Target: AMD Instinct MI355X (CDNA4, wave64, 256 CUs / 8 XCDs).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#define CI_CHECK_CUDA(x) TORCH_CHECK((x).is_cuda(), #x " must be on device")
#define CI_CHECK_CONTIG(x) TORCH_CHECK((x).is_contiguous(), #x " must be contiguous")

namespace ci {

constexpr int kWave = 64;  // CDNA wavefront

static inline hipStream_t stream() {
  return at::hip::getCurrentHIPStream().stream();
}

// ---- scalar conversion helpers ------------------------------------------
template <typename T> struct Cvt;
template <> struct Cvt<float> {
  static __device__ __forceinline__ float load(const float* p) { return *p; }
  static __device__ __forceinline__ void store(float* p, float v) { *p = v; }
};
template <> struct Cvt<__hip_bfloat16> {
  static __device__ __forceinline__ float load(const __hip_bfloat16* p) {
    return __bfloat162float(*p);
  }
  static __device__ __forceinline__ void store(__hip_bfloat16* p, float v) {
    *p = __float2bfloat16(v);
  }
};

template <typename T>
static __device__ __forceinline__ float ld(const T* p) { return Cvt<T>::load(p); }
template <typename T>
static __device__ __forceinline__ void st(T* p, float v) { Cvt<T>::store(p, v); }

// ---- 16-B vectorized load/store of VEC scalars as fp32 lanes -------------
// (G13: scalar bf16 loads cost ~2x; VEC = 16/sizeof(T))
template <typename T, int VEC>
static __device__ __forceinline__ void ldv(const T* p, float* out) {
  T buf[VEC];
  *reinterpret_cast<int4*>(buf) = *reinterpret_cast<const int4*>(p);
  #pragma unroll
  for (int e = 0; e < VEC; ++e) out[e] = ld(buf + e);
}

template <typename T, int VEC>
static __device__ __forceinline__ void stv(T* p, const float* v) {
  T buf[VEC];
  #pragma unroll
  for (int e = 0; e < VEC; ++e) st(buf + e, v[e]);
  *reinterpret_cast<int4*>(p) = *reinterpret_cast<const int4*>(buf);
}

template <int VEC>
static __device__ __forceinline__ void ldv_f32(const float* p, float* out) {
  #pragma unroll
  for (int e = 0; e < VEC; e += 4)
    *reinterpret_cast<float4*>(out + e) = *reinterpret_cast<const float4*>(p + e);
}

template <int VEC>
static __device__ __forceinline__ void stv_f32(float* p, const float* v) {
  #pragma unroll
  for (int e = 0; e < VEC; e += 4)
    *reinterpret_cast<float4*>(p + e) = *reinterpret_cast<const float4*>(v + e);
}

static __device__ __forceinline__ float sigmoidf_(float x) {
  return 1.0f / (1.0f + __expf(-x));
}

// OCP e4m3 unpack: gfx950 hardware cvt converts packed fp8 pairs to f32
// at VALU rate (used by the fp8-weight GEMV and the fp8-resident CE).
static __device__ __forceinline__ void fp8x4_to_f32(unsigned int u,
                                                    float* out) {
  typedef float f32x2_ __attribute__((ext_vector_type(2)));
  f32x2_ lo = __builtin_amdgcn_cvt_pk_f32_fp8(u, false);
  f32x2_ hi = __builtin_amdgcn_cvt_pk_f32_fp8(u, true);
  out[0] = lo[0]; out[1] = lo[1]; out[2] = hi[0]; out[3] = hi[1];
}

// dispatch on an ATen dtype (float32 / bfloat16)
#define CI_DISPATCH_FB(DTYPE, NAME, ...)                                   \
  [&] {                                                                    \
    switch (DTYPE) {                                                       \
      case at::ScalarType::Float: {                                        \
        using scalar_t = float;                                            \
        return __VA_ARGS__();                                              \
      }                                                                    \
      case at::ScalarType::BFloat16: {                                     \
        using scalar_t = __hip_bfloat16;                                   \
        return __VA_ARGS__();                                              \
      }                                                                    \
      default:                                                             \
        TORCH_CHECK(false, NAME ": unsupported dtype ", DTYPE);            \
    }                                                                      \
  }()

static inline int ceil_div(long a, long b) { return (int)((a + b - 1) / b); }

}  // namespace ci
