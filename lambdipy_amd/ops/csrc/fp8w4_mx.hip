// Mixed-format MX GEMM: fp8 (e4m3) activations x fp4 (e2m1) weights.
//
// cbsz=0 / blgp=4 on v_mfma_scale_f32_32x32x64_f8f6f4 measures 4.96 PF —
// the FULL fp8 rate with HALF the B-operand bytes (scripts/mfma_rate.py
// probe 6). For weight-heavy inference GEMMs this halves weight traffic at
// unchanged compute rate and keeps activations at e4m3 precision.
//
// Structure = the 8-phase glds pipeline with per-side geometries:
//   A halves: 128 rows x 128 B (fp8-MX shuffled packing, quantize_mx),
//             2 glds/thread, (row>>1)&7 8-cell rotation;
//   B halves: 128 rows x 64 B (fp4-MX linear packing, quantize_mx4),
//             1 glds/thread, ((row>>1)+(row>>2))&3 4-cell rotation.
// Per-phase glds event counts are [2,2,2,1] + the phase-0 scale dword —
// every 4-call vmcnt window sums to a constant 7. Scales are tri-buffered
// tile-major blocks ([A 256 rows | B 256 rows] x 4 B) exactly as in
// fp8_mx_8ph.hip. Requires M,N % 256 == 0, K % 128 == 0.

#include "lpx_common.h"
#include "kernels.h"

namespace lpx {

namespace {

constexpr int BMW = 256, BKW = 128;  // elements per k-tile
constexpr int THREADSW = 512;
constexpr int A_HALF = 128 * 128;    // fp8 bytes
constexpr int B_HALF = 128 * 64;     // fp4 bytes
constexpr int PARITYW = 2 * A_HALF + 2 * B_HALF;  // 48 KiB
constexpr int SC_TILEW = 2048;
constexpr int SC_BASEW = 2 * PARITYW;

typedef __attribute__((ext_vector_type(8))) int i32x8v;

// A side: 8 16-B cells/row, rotation (row>>1)&7
__device__ __forceinline__ int phys_a(int row, int cell) {
  return row * 8 + ((cell + (row >> 1)) & 7);
}
__device__ __forceinline__ int src_a(int q) {
  const int row = q >> 3;
  return ((q & 7) - (row >> 1)) & 7;
}
// B side: 4 16-B cells/row, rotation ((row>>1)+(row>>2))&3
__device__ __forceinline__ int rot_b(int row) {
  return ((row >> 1) + (row >> 2)) & 3;
}
__device__ __forceinline__ int phys_b(int row, int cell) {
  return row * 4 + ((cell + rot_b(row)) & 3);
}
__device__ __forceinline__ int src_b(int q) {
  const int row = q >> 2;
  return ((q & 3) - rot_b(row)) & 3;
}

template <int N>
__device__ __forceinline__ void waitvw() {
  if constexpr (N == 0) asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  else if constexpr (N == 4) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  else if constexpr (N == 7) asm volatile("s_waitcnt vmcnt(7)" ::: "memory");
  else static_assert(N == 0, "unsupported vmcnt");
}

// hh: 0/1 = A halves (2 glds), 2/3 = B halves (1 glds)
__device__ __forceinline__ void stage_w(
    const unsigned char* __restrict__ A, const unsigned char* __restrict__ Bt,
    long K, int bm, int bn, int tile, int hh, int ksteps,
    char* __restrict__ smem, int w, int lane) {
  const int tc = tile < ksteps ? tile : ksteps - 1;
  char* const pbase = smem + (size_t)(tile & 1) * PARITYW;
  if (hh < 2) {
    const long k0 = (long)tc * BKW;  // fp8: 1 B/elem
    const int row0 = bm + hh * 128;
    char* const base = pbase + (size_t)hh * A_HALF;
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const int qbase = (j * 8 + w) * 64;
      const int q = qbase + lane;
      const int r = q >> 3, cell = src_a(q);
      const unsigned char* g = A + (long)(row0 + r) * K + (k0 + cell * 16);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)g,
          (__attribute__((address_space(3))) void*)(base + qbase * 16), 16, 0,
          0);
    }
  } else {
    const long Kb = K >> 1;          // fp4: 0.5 B/elem
    const long k0 = (long)tc * (BKW >> 1);
    const int row0 = bn + (hh - 2) * 128;
    char* const base = pbase + 2 * A_HALF + (size_t)(hh - 2) * B_HALF;
    const int t = w * 64 + lane;     // 512 cells of 16 B
    const int r = t >> 2, cell = src_b(t);
    const unsigned char* g = Bt + (long)(row0 + r) * Kb + (k0 + cell * 16);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)g,
        (__attribute__((address_space(3))) void*)(base + (t >> 6) * 1024), 16,
        0, 0);
  }
}

__device__ __forceinline__ void stage_scw(
    const unsigned char* __restrict__ sA, const unsigned char* __restrict__ sB,
    int Mtot, int Ntot, int bm, int bn, int tile, int ksteps,
    char* __restrict__ smem, int w, int lane) {
  const int tc = tile < ksteps ? tile : ksteps - 1;
  const int t = w * 64 + lane;
  const unsigned char* g =
      (t < 256) ? sA + ((long)tc * Mtot + bm + t) * 4
                : sB + ((long)tc * Ntot + bn + (t - 256)) * 4;
  char* const base = smem + SC_BASEW + (size_t)(tile % 3) * SC_TILEW + w * 256;
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) void*)g,
      (__attribute__((address_space(3))) void*)base, 4, 0, 0);
}

__device__ __forceinline__ void slot_w(int t, int p, int* tile, int* hh) {
  switch (p) {
    case 0: *tile = t + 1; *hh = 3; break;
    case 1: *tile = t + 1; *hh = 1; break;
    case 2: *tile = t + 2; *hh = 0; break;
    default: *tile = t + 2; *hh = 2; break;
  }
}

__device__ __forceinline__ i32x8v frag_af(const char* __restrict__ smem,
                                          int parity, int rh, int row, int kg,
                                          int fkb) {
  const char* half = smem + (size_t)parity * PARITYW + (size_t)rh * A_HALF;
  const int c0 = kg * 4 + fkb * 2;
  const i32x4 lo =
      *reinterpret_cast<const i32x4*>(half + phys_a(row, c0) * 16);
  const i32x4 hi =
      *reinterpret_cast<const i32x4*>(half + phys_a(row, c0 + 1) * 16);
  return i32x8v{lo.x, lo.y, lo.z, lo.w, hi.x, hi.y, hi.z, hi.w};
}

__device__ __forceinline__ i32x8v frag_bf(const char* __restrict__ smem,
                                          int parity, int ch, int row, int kg,
                                          int fkb) {
  const char* half = smem + (size_t)parity * PARITYW + 2 * A_HALF +
                     (size_t)ch * B_HALF;
  const i32x4 v = *reinterpret_cast<const i32x4*>(
      half + phys_b(row, kg * 2 + fkb) * 16);
  return i32x8v{v.x, v.y, v.z, v.w, 0, 0, 0, 0};
}

__device__ __forceinline__ unsigned scw(const char* __restrict__ smem,
                                        int slot3, int side_row) {
  return *reinterpret_cast<const unsigned*>(
      smem + SC_BASEW + (size_t)slot3 * SC_TILEW + side_row * 4);
}

__global__ __launch_bounds__(THREADSW, 1) void gemm_w4_kernel(
    const unsigned char* __restrict__ A, const unsigned char* __restrict__ sA,
    const unsigned char* __restrict__ Bt, const unsigned char* __restrict__ sB,
    float* __restrict__ C, int M, int N, int K, unsigned tiles_m,
    unsigned tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];

  const unsigned nwg = tiles_m * tiles_n;
  const unsigned bid = lpx_xcd_swizzle(blockIdx.x, nwg);
  const int bm = (int)(bid / tiles_n) * BMW;
  const int bn = (int)(bid % tiles_n) * BMW;

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

  const int ksteps = (K + BKW - 1) / BKW;

  stage_w(A, Bt, K, bm, bn, 0, 0, ksteps, smem, w, lane);  // A-lo0 (2)
  stage_w(A, Bt, K, bm, bn, 0, 2, ksteps, smem, w, lane);  // B-lo0 (1)
  stage_w(A, Bt, K, bm, bn, 0, 3, ksteps, smem, w, lane);  // B-hi0 (1)
  stage_w(A, Bt, K, bm, bn, 0, 1, ksteps, smem, w, lane);  // A-hi0 (2)
  stage_scw(sA, sB, M, N, bm, bn, 0, ksteps, smem, w, lane);  // sc0 (1)
  stage_scw(sA, sB, M, N, bm, bn, 1, ksteps, smem, w, lane);  // sc1 (1)
  stage_w(A, Bt, K, bm, bn, 1, 0, ksteps, smem, w, lane);  // A-lo1 (2)
  stage_w(A, Bt, K, bm, bn, 1, 2, ksteps, smem, w, lane);  // B-lo1 (1)
  // 11 events; tile0 data (6) + sc0 (7th) required -> allow last 4
  waitvw<4>();
  __builtin_amdgcn_s_barrier();

  if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);  // T5 static young-half priority

  for (int t = 0; t < ksteps; ++t) {
    const int parity = t & 1;
    unsigned awT[2][2], bwT[2];
    {
      const int slot3 = t % 3;
#pragma unroll
      for (int rh = 0; rh < 2; ++rh)
#pragma unroll
        for (int mf = 0; mf < 2; ++mf)
          awT[rh][mf] = scw(smem, slot3, rh * 128 + wsr * 64 + mf * 32 + fr);
#pragma unroll
      for (int ch = 0; ch < 2; ++ch)
        bwT[ch] = scw(smem, slot3, 256 + ch * 128 + wsc * 32 + fr);
    }
    // fragment REUSE (the bf16/MX schedule): each fragment ds_read once
    // per K-tile — A pairs shared across column-half phases, B across
    // row-half phases.
    i32x8v aR[2][2], bR0[2], bR1[2];
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      const int rh = (p >> 1) & 1;
      const int ch = p & 1;
      if (p == 0 || p == 2) {
#pragma unroll
        for (int kg = 0; kg < 2; ++kg)
#pragma unroll
          for (int mf = 0; mf < 2; ++mf)
            aR[mf][kg] = frag_af(smem, parity, rh,
                                 wsr * 64 + mf * 32 + fr, kg, fkb);
      }
      if (p == 0) {
#pragma unroll
        for (int kg = 0; kg < 2; ++kg)
          bR0[kg] = frag_bf(smem, parity, 0, wsc * 32 + fr, kg, fkb);
      } else if (p == 1) {
#pragma unroll
        for (int kg = 0; kg < 2; ++kg)
          bR1[kg] = frag_bf(smem, parity, 1, wsc * 32 + fr, kg, fkb);
      }
      int st, sh;
      slot_w(t, p, &st, &sh);
      stage_w(A, Bt, K, bm, bn, st, sh, ksteps, smem, w, lane);
      if (p == 0)
        stage_scw(sA, sB, M, N, bm, bn, t + 2, ksteps, smem, w, lane);

      __builtin_amdgcn_s_barrier();
      auto& bC = (ch == 0) ? bR0 : bR1;
#pragma unroll
      for (int kg = 0; kg < 2; ++kg)
#pragma unroll
        for (int mf = 0; mf < 2; ++mf) {
          const int sa = (int)((awT[rh][mf] >> (8 * (kg * 2 + fkb))) & 0xff);
          const int sb = (int)((bwT[ch] >> (8 * (kg * 2 + fkb))) & 0xff);
          acc[p][mf] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
              aR[mf][kg], bC[kg], acc[p][mf], 0, 4, 0, sa, 0, sb);
        }
      waitvw<7>();  // constant: every 4-call window sums to 7
      __builtin_amdgcn_s_barrier();
    }
  }
  waitvw<0>();

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
        C[(long)row * N + col] = acc[p][mf][r];
      }
    }
  }
}

}  // namespace

bool gemm_mx_w4_supported(int M, int N, int K) {
  return M % BMW == 0 && N % BMW == 0 && K % BKW == 0 && K >= BKW;
}

hipError_t gemm_mx_w4_nt8ph(const unsigned char* A, const unsigned char* sA,
                            const unsigned char* Bt, const unsigned char* sB,
                            float* C, int M, int N, int K,
                            hipStream_t stream) {
  const unsigned tiles_m = (unsigned)(M / BMW), tiles_n = (unsigned)(N / BMW);
  const size_t lds = 2 * PARITYW + 3 * SC_TILEW;  // 102 KiB
  gemm_w4_kernel<<<dim3(tiles_m * tiles_n), dim3(THREADSW), lds, stream>>>(
      A, sA, Bt, sB, C, M, N, K, tiles_m, tiles_n);
  return hipGetLastError();
}

}  // namespace lpx
