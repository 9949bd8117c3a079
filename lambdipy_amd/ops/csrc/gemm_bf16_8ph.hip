// bf16 GEMM v3 for gfx950 — 256x256/BK64, glds-staged 8-phase pipeline.
//
// The guide's fast-template class (cdna_hip_programming.md §5 'The 256^2
// 8-phase template'): 512 threads (8 waves), 2 K-tiles per 8 phases, each
// phase computes ONE 128x128 tile-quadrant (rh, ch) = ((p>>1)&1, p&1) —
// every wave contributes a 64x32 sub-tile, 16 MFMAs — while ONE half-tile
// (128 rows x 64 k of A or B) is prefetched by global_load_lds (2
// instructions per thread). Counted s_waitcnt vmcnt(6) per phase (never 0 in
// the loop) keeps 3 half-tiles in flight across raw s_barriers; staging lands
// in the slot the previous-parity tile freed one phase earlier:
//   phase 4T+0 reads {A-lo,B-lo}, +1 {A-lo,B-hi}, +2 {A-hi,B-lo}, +3 {A-hi,B-hi}
//   phase 4T+0 stages (T+1, B-hi), +1 (T+1, A-hi), +2 (T+2, A-lo), +3 (T+2, B-lo)
// LDS 128 KiB: 2 parities x (A,B) x 2 halves x 16 KiB, lane-linear glds
// image with a per-row ROTATION permutation applied to the SOURCE address
// and the ds_read address (rule 21): phys 16-B cell = row*8 + (cell+row)%8.
// Verified conflict-free for the b128 fragment-read lane groups (each
// group's 16 lanes land on 16 distinct 16-B slots of the 256-B bank row),
// and the glds write stays lane-linear (no write conflicts); the source
// still touches whole 128-B row segments, so coalescing is preserved.
// Requires M,N % 256 == 0 and K % 64 == 0 (launcher falls back to v2).

#include "lpx_common.h"

namespace lpx {

namespace {

constexpr int BM8 = 256, BK8 = 64;
constexpr int THREADS8 = 512;
constexpr int HALF_BYTES = 128 * 128;  // 128 rows x 128 B
constexpr int PARITY_BYTES = 4 * HALF_BYTES;  // A-lo A-hi B-lo B-hi

using bf16frag = bf16x8;

// logical (row, cell) -> physical 16-B cell index (rotation per row)
__device__ __forceinline__ int phys_cell(int row, int cell) {
  return row * 8 + ((cell + row) & 7);
}
// physical cell q -> logical cell within its row (staging side)
__device__ __forceinline__ int src_cell(int q) {
  const int row = q >> 3;
  return ((q & 7) - row) & 7;
}

// stage one half-tile: hh 0=A-lo 1=A-hi 2=B-lo 3=B-hi of K-tile `tile`
__device__ __forceinline__ void stage_half8(const uint16_t* __restrict__ A,
                                            const uint16_t* __restrict__ Bt,
                                            long K, int bm, int bn, int tile,
                                            int hh, int ksteps,
                                            char* __restrict__ smem, int w,
                                            int lane) {
  const int tc = tile < ksteps ? tile : ksteps - 1;  // clamp (never read)
  const long k0 = (long)tc * BK8;
  const uint16_t* src = (hh < 2) ? A : Bt;
  const int row0 = ((hh < 2) ? bm : bn) + (hh & 1) * 128;
  char* const base = smem + (size_t)(tile & 1) * PARITY_BYTES +
                     (size_t)hh * HALF_BYTES;
#pragma unroll
  for (int j = 0; j < 2; ++j) {
    const int qbase = (j * 8 + w) * 64;  // 16-B cell stripe of this wave
    const int q = qbase + lane;
    const int r = q >> 3, cell = src_cell(q);
    const uint16_t* g = src + (long)(row0 + r) * K + (k0 + cell * 8);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)g,
        (__attribute__((address_space(3))) void*)(base + qbase * 16), 16, 0, 0);
  }
}

// which (tile, half) the cyclic schedule stages at phase (t, p)
__device__ __forceinline__ void stage_slot(int t, int p, int* tile, int* hh) {
  switch (p) {
    case 0: *tile = t + 1; *hh = 3; break;  // (T+1, B-hi)
    case 1: *tile = t + 1; *hh = 1; break;  // (T+1, A-hi)
    case 2: *tile = t + 2; *hh = 0; break;  // (T+2, A-lo)
    default: *tile = t + 2; *hh = 2; break; // (T+2, B-lo)
  }
}

__device__ __forceinline__ const bf16frag* frag_ptr(const char* __restrict__ smem,
                                                    int parity, int hh,
                                                    int row_in_half, int cell) {
  const int q = phys_cell(row_in_half, cell);
  return reinterpret_cast<const bf16frag*>(
      smem + (size_t)parity * PARITY_BYTES + (size_t)hh * HALF_BYTES + q * 16);
}

// ---- BK=32 half-LDS variant (tune 17) ------------------------------------
// Same 4-phase quadrant schedule with 32-deep K-steps: half-tiles are 8 KiB
// (1 glds/thread), LDS drops 128->64 KiB so the kernel runs 2 blocks/CU
// (4 waves/SIMD) — the extra block interleaves across the other's barriers,
// trading per-phase MFMA count (8 vs 16) for occupancy-hidden latency.
constexpr int BK32 = 32;
constexpr int HALF32 = 128 * 64;            // 128 rows x 64 B
constexpr int PAR32 = 4 * HALF32;           // A-lo A-hi B-lo B-hi

__device__ __forceinline__ int phys_cell32(int row, int cell) {
  return row * 4 + ((cell + row) & 3);      // 4 16-B cells per 64-B row
}
__device__ __forceinline__ int src_cell32(int q) {
  const int row = q >> 2;
  return ((q & 3) - row) & 3;
}

__device__ __forceinline__ void stage_half32(const uint16_t* __restrict__ A,
                                             const uint16_t* __restrict__ Bt,
                                             long K, int bm, int bn, int tile,
                                             int hh, int ksteps,
                                             char* __restrict__ smem, int w,
                                             int lane) {
  const int tc = tile < ksteps ? tile : ksteps - 1;  // clamp (never read)
  const long k0 = (long)tc * BK32;
  const uint16_t* src = (hh < 2) ? A : Bt;
  const int row0 = ((hh < 2) ? bm : bn) + (hh & 1) * 128;
  char* const base = smem + (size_t)(tile & 1) * PAR32 +
                     (size_t)hh * HALF32;
  const int q = w * 64 + lane;               // one 16-B cell per thread
  const int r = q >> 2, cell = src_cell32(q);
  const uint16_t* g = src + (long)(row0 + r) * K + (k0 + cell * 8);
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) void*)g,
      (__attribute__((address_space(3))) void*)(base + q * 16), 16, 0, 0);
}

__device__ __forceinline__ const bf16frag* frag_ptr32(
    const char* __restrict__ smem, int parity, int hh, int row_in_half,
    int cell) {
  const int q = phys_cell32(row_in_half, cell);
  return reinterpret_cast<const bf16frag*>(
      smem + (size_t)parity * PAR32 + (size_t)hh * HALF32 + q * 16);
}

template <bool BF16_OUT>
__global__ __launch_bounds__(THREADS8, 2) void gemm_bf16_8ph32_kernel(
    const uint16_t* __restrict__ A, const uint16_t* __restrict__ Bt,
    void* __restrict__ Cout, int M, int N, int K, unsigned tiles_m,
    unsigned tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];

  const unsigned nwg = tiles_m * tiles_n;
  const unsigned bid = lpx_xcd_swizzle(blockIdx.x, nwg);
  const int bm = (int)(bid / tiles_n) * BM8;
  const int bn = (int)(bid % tiles_n) * BM8;

  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;
  const int wsr = w >> 2, wsc = w & 3;
  const int fr = lane & 15;
  const int fkb = lane >> 4;

  f32x4 acc[4][4][2];
#pragma unroll
  for (int q = 0; q < 4; ++q)
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j) acc[q][i][j] = {0.f, 0.f, 0.f, 0.f};

  const int ksteps = (K + BK32 - 1) / BK32;

  // prologue mirrors the BK64 schedule (tile0 all 4 halves + tile1 lo pair)
  stage_half32(A, Bt, K, bm, bn, 0, 0, ksteps, smem, w, lane);
  stage_half32(A, Bt, K, bm, bn, 0, 2, ksteps, smem, w, lane);
  stage_half32(A, Bt, K, bm, bn, 0, 3, ksteps, smem, w, lane);
  stage_half32(A, Bt, K, bm, bn, 0, 1, ksteps, smem, w, lane);
  stage_half32(A, Bt, K, bm, bn, 1, 0, ksteps, smem, w, lane);
  stage_half32(A, Bt, K, bm, bn, 1, 2, ksteps, smem, w, lane);
  asm volatile("s_waitcnt vmcnt(2)" ::: "memory");  // tile0 A-lo/B-lo landed
  __builtin_amdgcn_s_barrier();

  if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);

  for (int t = 0; t < ksteps; ++t) {
    const int parity = t & 1;
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      const int rh = (p >> 1) & 1;
      const int ch = p & 1;
      bf16frag aF[4], bF[2];
      const int cell = fkb;                  // 4 cells = the full 32-k step
#pragma unroll
      for (int mf = 0; mf < 4; ++mf)
        aF[mf] = *frag_ptr32(smem, parity, rh, wsr * 64 + mf * 16 + fr, cell);
#pragma unroll
      for (int nf = 0; nf < 2; ++nf)
        bF[nf] = *frag_ptr32(smem, parity, 2 + ch, wsc * 32 + nf * 16 + fr,
                             cell);
      int st, sh;
      stage_slot(t, p, &st, &sh);
      stage_half32(A, Bt, K, bm, bn, st, sh, ksteps, smem, w, lane);
      __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
#pragma unroll
      for (int mf = 0; mf < 4; ++mf)
#pragma unroll
        for (int nf = 0; nf < 2; ++nf)
          acc[p][mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              aF[mf], bF[nf], acc[p][mf][nf], 0, 0, 0);
      if (p & 1) asm volatile("s_waitcnt vmcnt(3)" ::: "memory");
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
          if constexpr (BF16_OUT)
            reinterpret_cast<uint16_t*>(Cout)[(long)row * N + col] =
                (uint16_t)lpx_f32_to_bf16(acc[p][mf][nf][r]);
          else
            reinterpret_cast<float*>(Cout)[(long)row * N + col] =
                acc[p][mf][nf][r];
        }
      }
    }
  }
}

template <bool BF16_OUT, int VM, bool LG0, int PRIO, int ILV = 0,
          int THR = -1, bool ASM = false>
__global__ __launch_bounds__(THREADS8, 2) void gemm_bf16_8ph_kernel(
    const uint16_t* __restrict__ A, const uint16_t* __restrict__ Bt,
    void* __restrict__ Cout, int M, int N, int K, unsigned tiles_m,
    unsigned tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];

  const unsigned nwg = tiles_m * tiles_n;
  const unsigned bid = lpx_xcd_swizzle(blockIdx.x, nwg);
  const int bm = (int)(bid / tiles_n) * BM8;
  const int bn = (int)(bid % tiles_n) * BM8;

  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;       // wave id 0..7
  const int wsr = w >> 2;               // wave sub-row within quadrant (0..1)
  const int wsc = w & 3;                // wave sub-col within quadrant (0..3)

  const int fr = lane & 15;             // fragment row
  const int fkb = lane >> 4;            // 8-wide k block within 32-deep step

  // acc[quad][mf 0..3][nf 0..1]
  f32x4 acc[4][4][2];
#pragma unroll
  for (int q = 0; q < 4; ++q)
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j) acc[q][i][j] = {0.f, 0.f, 0.f, 0.f};

  const int ksteps = (K + BK8 - 1) / BK8;

  // prologue: tile0 {A-lo, B-lo, B-hi, A-hi} + tile1 {A-lo, B-lo}
  stage_half8(A, Bt, K, bm, bn, 0, 0, ksteps, smem, w, lane);
  stage_half8(A, Bt, K, bm, bn, 0, 2, ksteps, smem, w, lane);
  stage_half8(A, Bt, K, bm, bn, 0, 3, ksteps, smem, w, lane);
  stage_half8(A, Bt, K, bm, bn, 0, 1, ksteps, smem, w, lane);
  stage_half8(A, Bt, K, bm, bn, 1, 0, ksteps, smem, w, lane);
  stage_half8(A, Bt, K, bm, bn, 1, 2, ksteps, smem, w, lane);
  asm volatile("s_waitcnt vmcnt(4)" ::: "memory");  // tile0 A-lo/B-lo landed
  __builtin_amdgcn_s_barrier();

  if constexpr (PRIO == 1) {
    // T5 static form: the second-dispatched half loses VALU arbitration on
    // every segment; one static priority raise, no per-phase flips.
    if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
      __builtin_amdgcn_s_setprio(1);
  }

  // REUSE variant (tune 24): fragments live ACROSS phases — phases 0/1
  // share the same A fragments (rh=0) and phases 0/2 the same B fragments
  // (ch=0), so each fragment is ds_read ONCE per K-tile: 24 reads/wave/
  // K-tile instead of 48 (the quadrant scheme re-reads half of them).
  // Staging/barrier/vmcnt schedule unchanged.
  bf16frag aR[4][2], aR2[4][2], bR0[2][2], bR1[2][2];
  for (int t = 0; t < ksteps; ++t) {
    const int parity = t & 1;
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      const int rh = (p >> 1) & 1;  // quadrant row half
      const int ch = p & 1;        // quadrant col half
      if constexpr (ILV == 9 || ILV == 10) {  // REUSE schedule
        // ILV 10: balance the read schedule — prefetch the rh=1 A frags
        // during the light phase 1 (reads per phase 12/12/0/0 vs 12/4/8/0)
        if (p == 0 || (ILV == 9 && p == 2)) {
#pragma unroll
          for (int ks = 0; ks < 2; ++ks)
#pragma unroll
            for (int mf = 0; mf < 4; ++mf)
              aR[mf][ks] = *frag_ptr(smem, parity, rh,
                                     wsr * 64 + mf * 16 + fr, ks * 4 + fkb);
        }
        if (ILV == 10 && p == 1) {
#pragma unroll
          for (int ks = 0; ks < 2; ++ks)
#pragma unroll
            for (int mf = 0; mf < 4; ++mf)
              aR2[mf][ks] = *frag_ptr(smem, parity, 1,
                                      wsr * 64 + mf * 16 + fr, ks * 4 + fkb);
        }
        if (p == 0) {
#pragma unroll
          for (int ks = 0; ks < 2; ++ks)
#pragma unroll
            for (int nf = 0; nf < 2; ++nf)
              bR0[nf][ks] = *frag_ptr(smem, parity, 2,
                                      wsc * 32 + nf * 16 + fr, ks * 4 + fkb);
        } else if (p == 1) {
#pragma unroll
          for (int ks = 0; ks < 2; ++ks)
#pragma unroll
            for (int nf = 0; nf < 2; ++nf)
              bR1[nf][ks] = *frag_ptr(smem, parity, 3,
                                      wsc * 32 + nf * 16 + fr, ks * 4 + fkb);
        }
        int st9, sh9;
        stage_slot(t, p, &st9, &sh9);
        stage_half8(A, Bt, K, bm, bn, st9, sh9, ksteps, smem, w, lane);
        if constexpr (THR >= 0)
          asm volatile("s_waitcnt lgkmcnt(%0)" ::"i"(THR) : "memory");
        __builtin_amdgcn_s_barrier();
        auto& bC = (ch == 0) ? bR0 : bR1;
        auto& aC = (ILV == 10 && rh == 1) ? aR2 : aR;
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
#pragma unroll
          for (int mf = 0; mf < 4; ++mf)
#pragma unroll
            for (int nf = 0; nf < 2; ++nf)
              acc[p][mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  aC[mf][ks], bC[nf][ks], acc[p][mf][nf], 0, 0, 0);
        if constexpr (VM == 60) {
          if (p & 1) asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
        } else {
          asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
        }
        __builtin_amdgcn_s_barrier();
        continue;
      }
      // ASM variant (tune 16): one inline-asm block per k-half — 6
      // ds_read_b128 issued back-to-back, then the 8 MFMAs gated by
      // DESCENDING counted lgkmcnt (4/2/1/0: wait only for the operands
      // the next MFMA pair needs; LDS ops retire in order). The compiler's
      // own schedule drains lgkmcnt(0) before each MFMA group (see
      // disassembly), serializing read latency the counted form hides.
      auto asm_ks = [&](int ks) {
        const int cell = ks * 4 + fkb;
        const size_t pbase = (size_t)parity * PARITY_BYTES;
        unsigned aA[4], aB[2];
#pragma unroll
        for (int mf = 0; mf < 4; ++mf)
          aA[mf] = (unsigned)(size_t)(
              __attribute__((address_space(3))) char*)(
              smem + pbase + (size_t)rh * HALF_BYTES +
              phys_cell(wsr * 64 + mf * 16 + fr, cell) * 16);
#pragma unroll
        for (int nf = 0; nf < 2; ++nf)
          aB[nf] = (unsigned)(size_t)(
              __attribute__((address_space(3))) char*)(
              smem + pbase + (size_t)(2 + ch) * HALF_BYTES +
              phys_cell(wsc * 32 + nf * 16 + fr, cell) * 16);
        bf16frag A0, A1, A2, A3, B0, B1;
        asm volatile(
            "ds_read_b128 %[A0], %[aA0]\n\t"
            "ds_read_b128 %[B0], %[aB0]\n\t"
            "ds_read_b128 %[A1], %[aA1]\n\t"
            "ds_read_b128 %[B1], %[aB1]\n\t"
            "ds_read_b128 %[A2], %[aA2]\n\t"
            "ds_read_b128 %[A3], %[aA3]\n\t"
            "s_waitcnt lgkmcnt(4)\n\t"
            "v_mfma_f32_16x16x32_bf16 %[C00], %[A0], %[B0], %[C00]\n\t"
            "s_waitcnt lgkmcnt(2)\n\t"
            "v_mfma_f32_16x16x32_bf16 %[C01], %[A0], %[B1], %[C01]\n\t"
            "v_mfma_f32_16x16x32_bf16 %[C10], %[A1], %[B0], %[C10]\n\t"
            "v_mfma_f32_16x16x32_bf16 %[C11], %[A1], %[B1], %[C11]\n\t"
            "s_waitcnt lgkmcnt(1)\n\t"
            "v_mfma_f32_16x16x32_bf16 %[C20], %[A2], %[B0], %[C20]\n\t"
            "v_mfma_f32_16x16x32_bf16 %[C21], %[A2], %[B1], %[C21]\n\t"
            "s_waitcnt lgkmcnt(0)\n\t"
            "v_mfma_f32_16x16x32_bf16 %[C30], %[A3], %[B0], %[C30]\n\t"
            "v_mfma_f32_16x16x32_bf16 %[C31], %[A3], %[B1], %[C31]"
            : [C00] "+v"(acc[p][0][0]), [C01] "+v"(acc[p][0][1]),
              [C10] "+v"(acc[p][1][0]), [C11] "+v"(acc[p][1][1]),
              [C20] "+v"(acc[p][2][0]), [C21] "+v"(acc[p][2][1]),
              [C30] "+v"(acc[p][3][0]), [C31] "+v"(acc[p][3][1]),
              [A0] "=&v"(A0), [A1] "=&v"(A1), [A2] "=&v"(A2),
              [A3] "=&v"(A3), [B0] "=&v"(B0), [B1] "=&v"(B1)
            : [aA0] "v"(aA[0]), [aA1] "v"(aA[1]), [aA2] "v"(aA[2]),
              [aA3] "v"(aA[3]), [aB0] "v"(aB[0]), [aB1] "v"(aB[1])
            : "memory");
      };
      // fragment reads for this quadrant (ILV=1: defer the ks=1 half past
      // the barrier so its ds_read latency hides under the ks=0 MFMAs)
      bf16frag aF[4][2], bF[2][2];
      auto read_ks = [&](int ks) {
        const int cell = ks * 4 + fkb;
#pragma unroll
        for (int mf = 0; mf < 4; ++mf)
          aF[mf][ks] = *frag_ptr(smem, parity, rh,
                                 wsr * 64 + mf * 16 + fr, cell);
#pragma unroll
        for (int nf = 0; nf < 2; ++nf)
          bF[nf][ks] = *frag_ptr(smem, parity, 2 + ch,
                                 wsc * 32 + nf * 16 + fr, cell);
      };
      if constexpr (ASM) {
        asm_ks(0);  // reads + MFMAs of the first k-half (ends lgkm-clean)
        int st0, sh0;
        stage_slot(t, p, &st0, &sh0);
        stage_half8(A, Bt, K, bm, bn, st0, sh0, ksteps, smem, w, lane);
        __builtin_amdgcn_s_barrier();
        asm_ks(1);
      } else {
      read_ks(0);
      if constexpr (ILV == 0) read_ks(1);
      // stage the slot this phase frees
      int st, sh;
      stage_slot(t, p, &st, &sh);
      stage_half8(A, Bt, K, bm, bn, st, sh, ksteps, smem, w, lane);

      // optional ds_read-queue throttle before the barrier (guide §5: when
      // 12 ds_reads are issued per phase, waiting the oldest 4 here keeps
      // the post-barrier lgkmcnt(0) short without serializing the issue)
      if constexpr (THR >= 0)
        asm volatile("s_waitcnt lgkmcnt(%0)" ::"i"(THR) : "memory");
      __builtin_amdgcn_s_barrier();
      if constexpr (LG0) asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if constexpr (PRIO == 0) __builtin_amdgcn_s_setprio(1);
      if constexpr (ILV == 1) {
        read_ks(1);
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
#pragma unroll
          for (int mf = 0; mf < 4; ++mf)
#pragma unroll
            for (int nf = 0; nf < 2; ++nf)
              acc[p][mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  aF[mf][ks], bF[nf][ks], acc[p][mf][nf], 0, 0, 0);
      } else {
#pragma unroll
        for (int mf = 0; mf < 4; ++mf)
#pragma unroll
          for (int nf = 0; nf < 2; ++nf)
#pragma unroll
            for (int ks = 0; ks < 2; ++ks)
              acc[p][mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  aF[mf][ks], bF[nf][ks], acc[p][mf][nf], 0, 0, 0);
      }
      if constexpr (PRIO == 0) __builtin_amdgcn_s_setprio(0);
      }
      if constexpr (VM == 60) {
        // guide-template placement: wait only at phases 1/3 (twice per
        // K-tile); every read is >=4 calls back, allow-last-3 covers it
        if (p & 1) asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
      } else if constexpr (VM == 8) {
        asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
      } else {
        asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
      }
      __builtin_amdgcn_s_barrier();
    }
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");

  // epilogue: quad (rh, ch); rows rh*128 + wsr*64 + mf*16 + (l>>4)*4 + r
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
          if constexpr (BF16_OUT)
            reinterpret_cast<uint16_t*>(Cout)[(long)row * N + col] =
                (uint16_t)lpx_f32_to_bf16(acc[p][mf][nf][r]);
          else
            reinterpret_cast<float*>(Cout)[(long)row * N + col] =
                acc[p][mf][nf][r];
        }
      }
    }
  }
}

}  // namespace

template <bool BF16_OUT, int VM = 60, bool LG0 = false, int PRIO = 1,
          int ILV = 1, int THR = -1, bool ASM = false>
static hipError_t launch8ph(const uint16_t* A, const uint16_t* Bt, void* C,
                            int M, int N, int K, hipStream_t stream) {
  const unsigned tiles_m = (unsigned)(M / BM8), tiles_n = (unsigned)(N / BM8);
  const size_t lds = 2 * PARITY_BYTES;  // 128 KiB
  gemm_bf16_8ph_kernel<BF16_OUT, VM, LG0, PRIO, ILV, THR, ASM>
      <<<dim3(tiles_m * tiles_n), dim3(THREADS8), lds, stream>>>(
          A, Bt, C, M, N, K, tiles_m, tiles_n);
  return hipGetLastError();
}

bool gemm_bf16_8ph_supported(int M, int N, int K) {
  return M % BM8 == 0 && N % BM8 == 0 && K % BK8 == 0 && K >= BK8;
}

hipError_t gemm_bf16_nt8ph(const uint16_t* A, const uint16_t* Bt, float* C,
                           int M, int N, int K, hipStream_t stream) {
  // tune-25 config (r2 A/B): FRAGMENT-REUSE schedule — A fragments are
  // shared across the column-half phases and B fragments across the
  // row-half phases, so each fragment is ds_read ONCE per K-tile (24
  // reads/wave/K-tile vs the quadrant scheme's 48). +6-7% within-probe
  // over the previous default (1105/1211 vs 1041/1134 on one box).
  return launch8ph<false, 60, false, 1, 9, -1>(A, Bt, C, M, N, K, stream);
}

// tuning variants for within-probe A/B (scripts/ab_gemm8.py)
hipError_t gemm_bf16_nt8ph_tune(const uint16_t* A, const uint16_t* Bt,
                                float* C, int M, int N, int K, int tune,
                                hipStream_t stream) {
  switch (tune) {
    case 1: return launch8ph<false, 8, true>(A, Bt, C, M, N, K, stream);
    case 2: return launch8ph<false, 6, false>(A, Bt, C, M, N, K, stream);
    case 3: return launch8ph<false, 8, false>(A, Bt, C, M, N, K, stream);
    case 4: return launch8ph<false, 8, false, 1>(A, Bt, C, M, N, K, stream);
    case 5: return launch8ph<false, 8, false, 2>(A, Bt, C, M, N, K, stream);
    case 6: return launch8ph<false, 60, true, 1>(A, Bt, C, M, N, K, stream);
    case 7: return launch8ph<false, 60, true, 0>(A, Bt, C, M, N, K, stream);
    case 8: return launch8ph<false, 60, false, 1>(A, Bt, C, M, N, K, stream);
    case 9: return launch8ph<false, 60, false, 1, 1>(A, Bt, C, M, N, K, stream);
    case 10: return launch8ph<false, 8, false, 1, 1>(A, Bt, C, M, N, K, stream);
    // guide-template-exact order: ALL 12 ds_reads before the barrier,
    // lgkmcnt(0) after it, per-phase setprio for every wave
    case 11: return launch8ph<false, 60, true, 0, 0>(A, Bt, C, M, N, K, stream);
    // + the lgkmcnt(8) throttle before the barrier
    case 12: return launch8ph<false, 60, true, 0, 0, 8>(A, Bt, C, M, N, K, stream);
    // throttle on the current default's deferred-read form
    case 13: return launch8ph<false, 60, true, 1, 1, 8>(A, Bt, C, M, N, K, stream);
    case 14: return launch8ph<false, 60, false, 1, 1, 8>(A, Bt, C, M, N, K, stream);
    // template-exact but with the static half-wave priority raise
    case 15: return launch8ph<false, 60, true, 1, 0, 8>(A, Bt, C, M, N, K, stream);
    // hand-asm phase body: counted descending lgkmcnt between ds_reads
    // and MFMAs (the compiler's schedule drains lgkmcnt(0) per group)
    case 16: return launch8ph<false, 60, false, 1, 1, -1, true>(A, Bt, C, M, N, K, stream);
    // ILV=0 without the forced lgkmcnt(0): all 12 ds_reads issued before
    // the barrier, MFMAs gated only by the compiler's own counted waits
    case 18: return launch8ph<false, 60, false, 1, 0>(A, Bt, C, M, N, K, stream);
    case 19: return launch8ph<false, 60, false, 1, 0, 8>(A, Bt, C, M, N, K, stream);
    // throttle sweep for the ILV0 form (12 reads pre-barrier)
    case 20: return launch8ph<false, 60, false, 1, 0, 4>(A, Bt, C, M, N, K, stream);
    case 21: return launch8ph<false, 60, false, 1, 0, 6>(A, Bt, C, M, N, K, stream);
    case 22: return launch8ph<false, 60, false, 1, 0, 10>(A, Bt, C, M, N, K, stream);
    case 23: return launch8ph<false, 8, false, 1, 0, 8>(A, Bt, C, M, N, K, stream);
    // fragment-reuse schedule: 24 ds_reads/wave/K-tile instead of 48
    case 24: return launch8ph<false, 60, false, 1, 9, 8>(A, Bt, C, M, N, K, stream);
    case 25: return launch8ph<false, 60, false, 1, 9, -1>(A, Bt, C, M, N, K, stream);
    // reuse + balanced read prefetch (rh1 A frags read during phase 1)
    case 26: return launch8ph<false, 60, false, 1, 10, -1>(A, Bt, C, M, N, K, stream);
    // BK=32 half-LDS (64 KiB -> 2 blocks/CU) variant
    case 17: {
      if (K % BK32 != 0) return hipErrorInvalidValue;
      const unsigned tm = (unsigned)(M / BM8), tn = (unsigned)(N / BM8);
      gemm_bf16_8ph32_kernel<false>
          <<<dim3(tm * tn), dim3(THREADS8), 2 * PAR32, stream>>>(
              A, Bt, C, M, N, K, tm, tn);
      return hipGetLastError();
    }
    default: return launch8ph<false, 6, true>(A, Bt, C, M, N, K, stream);
  }
}

hipError_t gemm_bf16_nt8ph_bf16out(const uint16_t* A, const uint16_t* Bt,
                                   uint16_t* C, int M, int N, int K,
                                   hipStream_t stream) {
  return launch8ph<true>(A, Bt, C, M, N, K, stream);
}

}  // namespace lpx
