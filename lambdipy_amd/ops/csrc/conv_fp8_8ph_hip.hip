#include "hip/hip_runtime.h"
// Implicit-GEMM fp8 conv on the 8-phase glds pipeline + scaled MFMA.
//
// The register-staged implicit conv (fp8_gemm.hip) runs at v2-GEMM
// efficiency; this kernel reuses the gemm_mx_8ph structure (quadrant
// phases, counted vmcnt, raw barriers, 32x32 scaled-MFMA fragments with
// unit scales) with the A-side glds computing the NHWC patch-gather
// address per 16-B cell. glds cannot conditionally skip, so out-of-bounds
// (padding) lanes redirect their source pointer to a small device ZERO
// PAGE — the LDS slot then receives zeros, exactly like the register-staged
// kernel's bounds check.
//
// Requires: M = N*HO*WO % 256 == 0, KO % 256 == 0, K = R*S*C % 128 == 0,
// C % 16 == 0 (a 16-B cell never crosses an (r,s) tap). The launcher in
// fp8_gemm.hip falls back to the register-staged kernel otherwise.

#include "lpx_common.h"
#include "kernels.h"

namespace lpx {

namespace {

constexpr int BMC = 256, BKC = 128;  // output-pixel rows x 128 patch bytes
constexpr int THREADSC = 512;
constexpr int HALF_BYTES = 128 * 128;
constexpr int PARITY_BYTES = 4 * HALF_BYTES;

typedef __attribute__((ext_vector_type(8))) int i32x8v;

struct ConvGeomC {
  int N, H, W, C, R, S, HO, WO, stride, pad;
};

__device__ __forceinline__ int phys_cellc(int row, int cell) {
  return row * 8 + ((cell + (row >> 1)) & 7);
}
__device__ __forceinline__ int src_cellc(int q) {
  const int row = q >> 3;
  return ((q & 7) - (row >> 1)) & 7;
}

template <int N>
__device__ __forceinline__ void waitvmc() {
  if constexpr (N == 0) asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  else if constexpr (N == 4) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  else if constexpr (N == 8) asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
  else static_assert(N == 0, "unsupported vmcnt");
}

// Per-thread output-pixel context: each thread stages the SAME 4 rows
// every call (2 j-slots x 2 A-halves), so the m -> (n, ho, wo) divisions
// hoist out of the phase loop entirely.
struct RowCtx {
  long n_off;   // n * H (pre-scaled row base)
  int pho, pwo; // ho*stride - pad, wo*stride - pad
  int okm;      // m < M
};

__device__ __forceinline__ void make_row_ctx(const ConvGeomC g, int M, int bm,
                                             int w, int lane, RowCtx (&c)[4]) {
#pragma unroll
  for (int half = 0; half < 2; ++half)
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const int q = (j * 8 + w) * 64 + lane;
      const int m = bm + half * 128 + (q >> 3);
      const int wo = m % g.WO;
      const int ho = (m / g.WO) % g.HO;
      const int n = m / (g.WO * g.HO);
      c[half * 2 + j].n_off = (long)n * g.H;
      c[half * 2 + j].pho = ho * g.stride - g.pad;
      c[half * 2 + j].pwo = wo * g.stride - g.pad;
      c[half * 2 + j].okm = m < M;
    }
}

// A half: 128 output-pixel rows x 128 patch bytes, gather-addressed.
// S_T (tap count) is a template so rs/S_T folds; C is a power of two
// (shift clog2) — no runtime div/mod in the phase path.
template <int S_T>
__device__ __forceinline__ void stage_act(
    const unsigned char* __restrict__ xq, const unsigned char* __restrict__ zp,
    const ConvGeomC g, int clog2, const RowCtx (&ctx)[4], int tile, int half,
    int ksteps, char* __restrict__ smem, int w, int lane) {
  const int tc = tile < ksteps ? tile : ksteps - 1;
  const int k0 = tc * BKC;
  char* const base = smem + (size_t)(tile & 1) * PARITY_BYTES +
                     (size_t)half * HALF_BYTES;
#pragma unroll
  for (int j = 0; j < 2; ++j) {
    const int qbase = (j * 8 + w) * 64;
    const int q = qbase + lane;
    const int cell = src_cellc(q);
    const int k = k0 + cell * 16;
    const int c0 = k & ((1 << clog2) - 1);
    const int rs = k >> clog2;
    const int r_f = rs / S_T;
    const int s_f = rs - r_f * S_T;
    const RowCtx& rc = ctx[half * 2 + j];
    const int hi = rc.pho + r_f;
    const int wi = rc.pwo + s_f;
    const bool ok = rc.okm & (hi >= 0) & (hi < g.H) & (wi >= 0) & (wi < g.W);
    const unsigned char* src =
        ok ? xq + (((rc.n_off + hi) * g.W + wi) << clog2) + c0
           : zp + (lane & 3) * 16;  // zero page (padding -> zeros in LDS)
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)src,
        (__attribute__((address_space(3))) void*)(base + qbase * 16), 16, 0, 0);
  }
}

// B half: weights [KO][K], standard row-major staging
__device__ __forceinline__ void stage_wgt(
    const unsigned char* __restrict__ Wq, long K, int bn, int tile, int half,
    int ksteps, char* __restrict__ smem, int w, int lane) {
  const int tc = tile < ksteps ? tile : ksteps - 1;
  const long k0 = (long)tc * BKC;
  const int row0 = bn + half * 128;
  char* const base = smem + (size_t)(tile & 1) * PARITY_BYTES +
                     (size_t)(2 + half) * HALF_BYTES;
#pragma unroll
  for (int j = 0; j < 2; ++j) {
    const int qbase = (j * 8 + w) * 64;
    const int q = qbase + lane;
    const int r = q >> 3, cell = src_cellc(q);
    const unsigned char* g = Wq + (long)(row0 + r) * K + (k0 + cell * 16);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)g,
        (__attribute__((address_space(3))) void*)(base + qbase * 16), 16, 0, 0);
  }
}

__device__ __forceinline__ void stage_slotc(int t, int p, int* tile, int* hh) {
  switch (p) {
    case 0: *tile = t + 1; *hh = 3; break;
    case 1: *tile = t + 1; *hh = 1; break;
    case 2: *tile = t + 2; *hh = 0; break;
    default: *tile = t + 2; *hh = 2; break;
  }
}

template <int S_T>
__device__ __forceinline__ void stage_any(
    const unsigned char* __restrict__ xq, const unsigned char* __restrict__ zp,
    const unsigned char* __restrict__ Wq, const ConvGeomC g, int clog2,
    const RowCtx (&ctx)[4], int K,
    int bn, int tile, int hh, int ksteps, char* __restrict__ smem,
    int w, int lane) {
  if (hh < 2)
    stage_act<S_T>(xq, zp, g, clog2, ctx, tile, hh, ksteps, smem, w, lane);
  else
    stage_wgt(Wq, K, bn, tile, hh - 2, ksteps, smem, w, lane);
}

__device__ __forceinline__ i32x8v fragc(const char* __restrict__ smem,
                                        int parity, int hh, int row_in_half,
                                        int kg, int fkb) {
  const char* half = smem + (size_t)parity * PARITY_BYTES +
                     (size_t)hh * HALF_BYTES;
  const int c0 = kg * 4 + fkb * 2;
  const i32x4 lo = *reinterpret_cast<const i32x4*>(
      half + phys_cellc(row_in_half, c0) * 16);
  const i32x4 hi = *reinterpret_cast<const i32x4*>(
      half + phys_cellc(row_in_half, c0 + 1) * 16);
  return i32x8v{lo.x, lo.y, lo.z, lo.w, hi.x, hi.y, hi.z, hi.w};
}

template <bool BIAS, bool RELU, bool OUT8, int S_T>
__global__ __launch_bounds__(THREADSC, 1) void conv_fp8_8ph_kernel(
    const unsigned char* __restrict__ xq, const unsigned char* __restrict__ Wq,
    const unsigned char* __restrict__ zp, float* __restrict__ C,
    unsigned char* __restrict__ C8, const float* __restrict__ bias,
    ConvGeomC g, int clog2, int M, int N, int K, float descale, float out_inv,
    unsigned tiles_m, unsigned tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];

  const unsigned nwg = tiles_m * tiles_n;
  const unsigned bid = lpx_xcd_swizzle(blockIdx.x, nwg);
  const int bm = (int)(bid / tiles_n) * BMC;
  const int bn = (int)(bid % tiles_n) * BMC;

  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;
  const int wsr = w >> 2, wsc = w & 3;
  const int fr = lane & 31;
  const int fkb = lane >> 5;

  f32x16 acc[4][2];
#pragma unroll
  for (int p = 0; p < 4; ++p)
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
      for (int e = 0; e < 16; ++e) acc[p][i][e] = 0.f;

  const int ksteps = (K + BKC - 1) / BKC;

  RowCtx ctx[4];
  make_row_ctx(g, M, bm, w, lane, ctx);

  stage_any<S_T>(xq, zp, Wq, g, clog2, ctx, K, bn, 0, 0, ksteps, smem, w, lane);
  stage_any<S_T>(xq, zp, Wq, g, clog2, ctx, K, bn, 0, 2, ksteps, smem, w, lane);
  stage_any<S_T>(xq, zp, Wq, g, clog2, ctx, K, bn, 0, 3, ksteps, smem, w, lane);
  stage_any<S_T>(xq, zp, Wq, g, clog2, ctx, K, bn, 0, 1, ksteps, smem, w, lane);
  stage_any<S_T>(xq, zp, Wq, g, clog2, ctx, K, bn, 1, 0, ksteps, smem, w, lane);
  stage_any<S_T>(xq, zp, Wq, g, clog2, ctx, K, bn, 1, 2, ksteps, smem, w, lane);
  waitvmc<4>();
  __builtin_amdgcn_s_barrier();

  if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);  // T5 static young-half priority

  for (int t = 0; t < ksteps; ++t) {
    const int parity = t & 1;
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      const int rh = (p >> 1) & 1;
      const int ch = p & 1;
      i32x8v aF[2][2], bF[2];
      auto read_kg = [&](int kg) {
#pragma unroll
        for (int mf = 0; mf < 2; ++mf)
          aF[mf][kg] = fragc(smem, parity, rh, wsr * 64 + mf * 32 + fr, kg,
                             fkb);
        bF[kg] = fragc(smem, parity, 2 + ch, wsc * 32 + fr, kg, fkb);
      };
      read_kg(0);  // kg=1 deferred past the barrier (bf16-twin +5%)
      int st, sh;
      stage_slotc(t, p, &st, &sh);
      stage_any<S_T>(xq, zp, Wq, g, clog2, ctx, K, bn, st, sh, ksteps, smem,
                     w, lane);

      __builtin_amdgcn_s_barrier();
      read_kg(1);
#pragma unroll
      for (int kg = 0; kg < 2; ++kg)
#pragma unroll
        for (int mf = 0; mf < 2; ++mf)
          acc[p][mf] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
              aF[mf][kg], bF[kg], acc[p][mf], 0, 0, 0, 127, 0, 127);
      waitvmc<8>();
      __builtin_amdgcn_s_barrier();
    }
  }
  waitvmc<0>();

  const int ocol_l = lane & 31;
  const int orow_base = 4 * (lane >> 5);
#pragma unroll
  for (int p = 0; p < 4; ++p) {
    const int rh = (p >> 1) & 1, ch = p & 1;
#pragma unroll
    for (int mf = 0; mf < 2; ++mf) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int row = bm + rh * 128 + wsr * 64 + mf * 32 + (r & 3) +
                        8 * (r >> 2) + orow_base;
        const int col = bn + ch * 128 + wsc * 32 + ocol_l;
        float v = acc[p][mf][r] * descale;
        if constexpr (BIAS) v += bias[col];
        if constexpr (RELU) v = v > 0.f ? v : 0.f;
        if constexpr (OUT8)
          C8[(long)row * N + col] = lpx_f32_to_e4m3(v * out_inv);
        else
          C[(long)row * N + col] = v;
      }
    }
  }
}

// 64-B device zero page for padding redirects (per-process, lazy)
unsigned char* conv_zero_page() {
  static unsigned char* zp = nullptr;
  if (!zp) {
    void* p = nullptr;
    if (hipMalloc(&p, 64) == hipSuccess && p) {
      hipMemset(p, 0, 64);
      zp = static_cast<unsigned char*>(p);
    }
  }
  return zp;
}

}  // namespace

bool conv_fp8_8ph_supported(int M, int KO, int K, int Cin) {
  return M % BMC == 0 && KO % BMC == 0 && K % BKC == 0 && K >= BKC &&
         Cin >= 16 && (Cin & (Cin - 1)) == 0;  // pow2 C: shift addressing
}

hipError_t conv_fp8_implicit_8ph(const unsigned char* xq,
                                 const unsigned char* Wq, float* C,
                                 unsigned char* C8, const float* bias, int Nb,
                                 int H, int W, int Cin, int R, int S, int HO,
                                 int WO, int stride, int pad, int KO,
                                 float descale, bool relu, float out_inv,
                                 hipStream_t stream) {
  const int M = Nb * HO * WO;
  const int K = R * S * Cin;
  unsigned char* zp = conv_zero_page();
  if (!zp) return hipErrorOutOfMemory;
  ConvGeomC g{Nb, H, W, Cin, R, S, HO, WO, stride, pad};
  int clog2 = 0;
  while ((1 << clog2) < Cin) ++clog2;
  if (S != 1 && S != 3 && S != 7) return hipErrorInvalidValue;
  const unsigned tiles_m = (unsigned)(M / BMC), tiles_n = (unsigned)(KO / BMC);
  const size_t lds = 2 * PARITY_BYTES;  // 128 KiB
  dim3 grid(tiles_m * tiles_n), block(THREADSC);
#define LPX_C8PH_S(B_, R_, O_)                                                 \
  do {                                                                         \
    if (S == 3)                                                                \
     hipLaunchKernelGGL(( conv_fp8_8ph_kernel<B_, R_, O_, 3>), dim3(grid), dim3(block), lds, stream,         \
          xq, Wq, zp, C, C8, bias, g, clog2, M, KO, K, descale, out_inv,       \
          tiles_m, tiles_n);                                                   \
    else if (S == 1)                                                           \
     hipLaunchKernelGGL(( conv_fp8_8ph_kernel<B_, R_, O_, 1>), dim3(grid), dim3(block), lds, stream,         \
          xq, Wq, zp, C, C8, bias, g, clog2, M, KO, K, descale, out_inv,       \
          tiles_m, tiles_n);                                                   \
    else                                                                       \
     hipLaunchKernelGGL(( conv_fp8_8ph_kernel<B_, R_, O_, 7>), dim3(grid), dim3(block), lds, stream,         \
          xq, Wq, zp, C, C8, bias, g, clog2, M, KO, K, descale, out_inv,       \
          tiles_m, tiles_n);                                                   \
  } while (0)
  const bool b = bias != nullptr, o8 = C8 != nullptr;
  if (o8) {
    if (b && relu) LPX_C8PH_S(true, true, true);
    else if (b) LPX_C8PH_S(true, false, true);
    else if (relu) LPX_C8PH_S(false, true, true);
    else LPX_C8PH_S(false, false, true);
  } else {
    if (b && relu) LPX_C8PH_S(true, true, false);
    else if (b) LPX_C8PH_S(true, false, false);
    else if (relu) LPX_C8PH_S(false, true, false);
    else LPX_C8PH_S(false, false, false);
  }
#undef LPX_C8PH_S
  return hipGetLastError();
}

}  // namespace lpx
