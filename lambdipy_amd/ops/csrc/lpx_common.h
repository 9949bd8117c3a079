// lambdipy_amd ops — common device helpers for gfx950 (CDNA4) kernels.
//
// MI355X-first design notes (see /opt/skills/guides/cdna_hip_programming.md):
//  - wavefront = 64 lanes; block sizes are multiples of 64
//  - memory-bound kernels use 16 B/lane vectorized loads (G13)
//  - MFMA intrinsics per §3; bf16 via __builtin_amdgcn_mfma_f32_16x16x32_bf16
//  - no CUDA compatibility paths: this header is HIP/CDNA4 only.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <stdint.h>

#define LPX_WAVE 64

// ---- vector types -------------------------------------------------------
typedef __attribute__((ext_vector_type(2))) float f32x2;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(4))) short bf16x4;   // 4 bf16 in 2 VGPRs
typedef __attribute__((ext_vector_type(8))) short bf16x8;   // 8 bf16 in 4 VGPRs (MFMA A/B frag)
typedef __attribute__((ext_vector_type(4))) int i32x4;
typedef __attribute__((ext_vector_type(2))) int i32x2;

static_assert(sizeof(bf16x8) == 16, "bf16x8 must be 16 B");

// ---- bf16 <-> f32 -------------------------------------------------------
__device__ __forceinline__ float lpx_bf16_to_f32(short u) {
  union { float f; uint32_t i; } c;
  c.i = ((uint32_t)(uint16_t)u) << 16;
  return c.f;
}

__device__ __forceinline__ short lpx_f32_to_bf16(float f) {
  // round-to-nearest-even, matching PyTorch's bf16 conversion
  union { float f; uint32_t i; } c;
  c.f = f;
  uint32_t x = c.i;
  uint32_t rounding = 0x7FFF + ((x >> 16) & 1);
  x += rounding;
  return (short)(x >> 16);
}

// ---- wave-wide reduction (64 lanes) -------------------------------------
__device__ __forceinline__ float lpx_wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return v;  // valid in lane 0
}

__device__ __forceinline__ float lpx_wave_allreduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;  // valid in all lanes
}

__device__ __forceinline__ float lpx_wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_down(v, off, 64));
  return v;
}

// ---- XCD-aware blockIdx swizzle (T1, bijective form) --------------------
// MI355X: 8 XCDs, dispatcher places block b on XCD b%8. Remap so each XCD
// gets a contiguous chunk of the grid (neighbor tiles share L2).
__device__ __forceinline__ unsigned lpx_xcd_swizzle(unsigned bid, unsigned nwg) {
  const unsigned NXCD = 8u;
  if (nwg < 2 * NXCD) return bid;
  unsigned q = nwg / NXCD, r = nwg % NXCD;
  unsigned xcd = bid % NXCD, idx = bid / NXCD;
  unsigned base = (xcd < r) ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q;
  return base + idx;
}

// ---- f32 -> OCP e4m3 via the hw scaled conversion (RNE). The hw op
// returns NaN above the e4m3 max, so clamp to +-448 first (satfinite).
__device__ __forceinline__ unsigned char lpx_f32_to_e4m3(float x) {
  typedef __attribute__((ext_vector_type(2))) short lpx_s16x2;
  x = fminf(fmaxf(x, -448.f), 448.f);
  lpx_s16x2 t = {0, 0};
  t = __builtin_amdgcn_cvt_scalef32_pk_fp8_f32(t, x, x, 1.0f, false);
  return (unsigned char)(t[0] & 0xff);
}

// ---- grid helper ---------------------------------------------------------
static inline unsigned lpx_ceil_div(unsigned a, unsigned b) { return (a + b - 1) / b; }

#define LPX_HIP_CHECK(expr)                                                   \
  do {                                                                        \
    hipError_t _e = (expr);                                                   \
    if (_e != hipSuccess) {                                                   \
      TORCH_CHECK(false, "HIP error ", hipGetErrorString(_e), " at ",         \
                  __FILE__, ":", __LINE__);                                   \
    }                                                                         \
  } while (0)
