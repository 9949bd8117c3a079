#include "hip/hip_runtime.h"
// fp8 (OCP e4m3) GEMM v3 for gfx950 — 256x256/BK128, glds 8-phase pipeline.
//
// Same schedule as gemm_bf16_8ph.hip (quadrant phases, one half-tile glds per
// phase, counted vmcnt, raw barriers) with fp8 bytes: BK = 128 ELEMENTS so a
// half-tile stays 128 rows x 128 B (byte-identical staging geometry). Per
// phase: 8 fragments x 4 k-sub-steps = 32 v_mfma_f32_16x16x32_fp8_fp8.
// LDS rotation permutes 16-B cells per row (phys16 = (cell16 + row) & 7),
// preserving the 8-B sub-cell, which is conflict-free for the ds_read_b64
// fragment groups (same-kb row pairs r/r+8 always differ in kb parity).
// Requires M,N % 256 == 0 and K % 128 == 0 (launcher falls back to v2).

#include "lpx_common.h"

namespace lpx {

namespace {

constexpr int BM8 = 256, BK8 = 128;  // 128 fp8 elements = 128 B per row
constexpr int THREADS8 = 512;
constexpr int HALF_BYTES = 128 * 128;
constexpr int PARITY_BYTES = 4 * HALF_BYTES;

typedef __attribute__((ext_vector_type(2))) int fp8frag8;  // 8 e4m3 bytes

// logical 16-B cell -> physical (rotation per row)
__device__ __forceinline__ int phys_cell16(int row, int cell16) {
  return row * 8 + ((cell16 + row) & 7);
}
__device__ __forceinline__ int src_cell16(int q) {
  const int row = q >> 3;
  return ((q & 7) - row) & 7;
}

__device__ __forceinline__ void stage_half8f(const unsigned char* __restrict__ A,
                                             const unsigned char* __restrict__ Bt,
                                             long K, int bm, int bn, int tile,
                                             int hh, int ksteps,
                                             char* __restrict__ smem, int w,
                                             int lane) {
  const int tc = tile < ksteps ? tile : ksteps - 1;
  const long k0 = (long)tc * BK8;
  const unsigned char* src = (hh < 2) ? A : Bt;
  const int row0 = ((hh < 2) ? bm : bn) + (hh & 1) * 128;
  char* const base = smem + (size_t)(tile & 1) * PARITY_BYTES +
                     (size_t)hh * HALF_BYTES;
#pragma unroll
  for (int j = 0; j < 2; ++j) {
    const int qbase = (j * 8 + w) * 64;
    const int q = qbase + lane;
    const int r = q >> 3, cell = src_cell16(q);
    const unsigned char* g = src + (long)(row0 + r) * K + (k0 + cell * 16);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)g,
        (__attribute__((address_space(3))) void*)(base + qbase * 16), 16, 0, 0);
  }
}

__device__ __forceinline__ void stage_slot8f(int t, int p, int* tile, int* hh) {
  switch (p) {
    case 0: *tile = t + 1; *hh = 3; break;
    case 1: *tile = t + 1; *hh = 1; break;
    case 2: *tile = t + 2; *hh = 0; break;
    default: *tile = t + 2; *hh = 2; break;
  }
}

// fragment (8 B) pointer: ks in [0,4), kb = lane>>4 in [0,4)
__device__ __forceinline__ const fp8frag8* frag_ptr8f(
    const char* __restrict__ smem, int parity, int hh, int row_in_half,
    int ks, int kb) {
  const int cell8 = ks * 4 + kb;                  // logical 8-B cell [0,16)
  const int c16 = phys_cell16(row_in_half, cell8 >> 1);
  const int byte = c16 * 16 + (cell8 & 1) * 8;
  return reinterpret_cast<const fp8frag8*>(
      smem + (size_t)parity * PARITY_BYTES + (size_t)hh * HALF_BYTES + byte);
}

template <bool RELU, bool BIAS>
__global__ __launch_bounds__(THREADS8, 2) void gemm_fp8_8ph_kernel(
    const unsigned char* __restrict__ A, const unsigned char* __restrict__ Bt,
    float* __restrict__ C, const float* __restrict__ bias, int M, int N,
    int K, float descale, unsigned tiles_m, unsigned tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];

  const unsigned nwg = tiles_m * tiles_n;
  const unsigned bid = lpx_xcd_swizzle(blockIdx.x, nwg);
  const int bm = (int)(bid / tiles_n) * BM8;
  const int bn = (int)(bid % tiles_n) * BM8;

  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;
  const int wsr = w >> 2, wsc = w & 3;
  const int fr = lane & 15;
  const int fkb = lane >> 4;  // 0..3

  f32x4 acc[4][4][2];
#pragma unroll
  for (int q = 0; q < 4; ++q)
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j) acc[q][i][j] = {0.f, 0.f, 0.f, 0.f};

  const int ksteps = (K + BK8 - 1) / BK8;

  stage_half8f(A, Bt, K, bm, bn, 0, 0, ksteps, smem, w, lane);
  stage_half8f(A, Bt, K, bm, bn, 0, 2, ksteps, smem, w, lane);
  stage_half8f(A, Bt, K, bm, bn, 0, 3, ksteps, smem, w, lane);
  stage_half8f(A, Bt, K, bm, bn, 0, 1, ksteps, smem, w, lane);
  stage_half8f(A, Bt, K, bm, bn, 1, 0, ksteps, smem, w, lane);
  stage_half8f(A, Bt, K, bm, bn, 1, 2, ksteps, smem, w, lane);
  asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  // T5 static form: one priority raise for the younger dispatch half, no
  // per-phase flips (within-probe +23-24% on the bf16 8-phase kernel).
  if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);

  for (int t = 0; t < ksteps; ++t) {
    const int parity = t & 1;
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      const int rh = (p >> 1) & 1;
      const int ch = p & 1;
      fp8frag8 aF[4][4], bF[2][4];
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
#pragma unroll
        for (int mf = 0; mf < 4; ++mf)
          aF[mf][ks] = *frag_ptr8f(smem, parity, rh,
                                   wsr * 64 + mf * 16 + fr, ks, fkb);
#pragma unroll
        for (int nf = 0; nf < 2; ++nf)
          bF[nf][ks] = *frag_ptr8f(smem, parity, 2 + ch,
                                   wsc * 32 + nf * 16 + fr, ks, fkb);
      }
      int st, sh;
      stage_slot8f(t, p, &st, &sh);
      stage_half8f(A, Bt, K, bm, bn, st, sh, ksteps, smem, w, lane);

      __builtin_amdgcn_s_barrier();
#pragma unroll
      for (int mf = 0; mf < 4; ++mf)
#pragma unroll
        for (int nf = 0; nf < 2; ++nf)
#pragma unroll
          for (int ks = 0; ks < 4; ++ks)
            acc[p][mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
                __builtin_bit_cast(long, aF[mf][ks]),
                __builtin_bit_cast(long, bF[nf][ks]), acc[p][mf][nf], 0, 0, 0);
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
      __builtin_amdgcn_s_barrier();
    }
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");

  const int orow_l = (lane >> 4) * 4;
  const int ocol_l = lane & 15;
#pragma unroll
  for (int p = 0; p < 4; ++p) {
    const int rh = (p >> 1) & 1, ch = p & 1;
#pragma unroll
    for (int mf = 0; mf < 4; ++mf) {
#pragma unroll
      for (int nf = 0; nf < 2; ++nf) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = bm + rh * 128 + wsr * 64 + mf * 16 + orow_l + r;
          const int col = bn + ch * 128 + wsc * 32 + nf * 16 + ocol_l;
          float v = acc[p][mf][nf][r] * descale;
          if constexpr (BIAS) v += bias[col];
          if constexpr (RELU) v = v > 0.f ? v : 0.f;
          C[(long)row * N + col] = v;
        }
      }
    }
  }
}

}  // namespace

bool gemm_fp8_8ph_supported(int M, int N, int K) {
  return M % BM8 == 0 && N % BM8 == 0 && K % BK8 == 0 && K >= BK8;
}

hipError_t gemm_fp8_nt8ph(const unsigned char* A, const unsigned char* Bt,
                          float* C, const float* bias, int M, int N, int K,
                          float descale, bool relu, hipStream_t stream) {
  const unsigned tiles_m = (unsigned)(M / BM8), tiles_n = (unsigned)(N / BM8);
  const size_t lds = 2 * PARITY_BYTES;
  dim3 grid(tiles_m * tiles_n), block(THREADS8);
  if (bias && relu)
   hipLaunchKernelGGL(( gemm_fp8_8ph_kernel<true, true>), dim3(grid), dim3(block), lds, stream, 
        A, Bt, C, bias, M, N, K, descale, tiles_m, tiles_n);
  else if (bias)
   hipLaunchKernelGGL(( gemm_fp8_8ph_kernel<false, true>), dim3(grid), dim3(block), lds, stream, 
        A, Bt, C, bias, M, N, K, descale, tiles_m, tiles_n);
  else if (relu)
   hipLaunchKernelGGL(( gemm_fp8_8ph_kernel<true, false>), dim3(grid), dim3(block), lds, stream, 
        A, Bt, C, bias, M, N, K, descale, tiles_m, tiles_n);
  else
   hipLaunchKernelGGL(( gemm_fp8_8ph_kernel<false, false>), dim3(grid), dim3(block), lds, stream, 
        A, Bt, C, bias, M, N, K, descale, tiles_m, tiles_n);
  return hipGetLastError();
}

}  // namespace lpx
