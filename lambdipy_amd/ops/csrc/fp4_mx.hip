// MX block-scaled fp4 (OCP e2m1) GEMM for gfx950 — the full-rate path.
//
// v_mfma_scale_f32_32x32x64_f8f6f4 with cbsz/blgp=4 (e2m1) issues at 2x the
// fp8 rate (9.67 vs 4.89 PF register-resident, scripts/mfma_rate.py; dense
// fp4 silicon peak is ~10 PF). Operands shrink to 16 B/lane (4 VGPRs — the
// disassembly shows v[16:19] operands), halving LDS and HBM traffic too.
//
// fp4 scale semantics DIFFER from fp8 (scripts/fp4_diag.py, hypothesis H1
// exact to 0.00000): lane l's scale byte applies to lane l's OWN 16 operand
// bytes = 32 e2m1 elements = exactly one MX-32 block. So operands pack
// LINEARLY (no quarter shuffle): lane kb = l>>5 holds contiguous original
// k [kb*32, +32) of its row, and sc[r][g][kb] covers that block. Nibble
// order within a block is irrelevant to the product (A and B share the
// permutation and the block scale is constant), but quantizer and tests
// use low-nibble-first (the v_cvt_scalef32_pk_fp4_f32 convention).
//
// 8-phase glds pipeline: half-tile = 128 rows x 64 B = one 16-B glds per
// thread per stage call; per-tile-cycle call events [2,1,1,1] keep the
// counted vmcnt constant at 5 (3 after p3, covering the next tile's
// scales). LDS cell rotation is ((row>>1)+(row>>2))&3: rows 4, 8 and 12
// apart land on the same 16-bank base, and this rotation separates all of
// them (plain (row>>1)&3 collides at +8). 68 KiB LDS -> 2 blocks/CU.

#include "lpx_common.h"

namespace lpx {

namespace {

constexpr int THREADS = 256;

typedef __attribute__((ext_vector_type(8))) int i32x8v;
typedef __attribute__((ext_vector_type(2))) short s16x2;

// ---- MX fp4 quantization -------------------------------------------------
// x f32 [R][K], K%64==0 -> q [R][K/2] linear nibbles + tile-major scales
// [ceil(groups/2)][R][4] (contiguous per GEMM k-tile -> coalesced staging)
__global__ __launch_bounds__(THREADS) void quantize_mx4_kernel(
    const float* __restrict__ x, unsigned char* __restrict__ q,
    unsigned char* __restrict__ sc, int R, int K) {
  const int groups = K >> 6;
  const long total = (long)R * groups * 2;  // thread per (row, group, block)
  for (long idx = (long)blockIdx.x * THREADS + threadIdx.x; idx < total;
       idx += (long)gridDim.x * THREADS) {
    const int kb = (int)(idx & 1);
    const int g = (int)((idx >> 1) % groups);
    const int r = (int)(idx / (2 * groups));
    const float* src = x + (long)r * K + g * 64 + kb * 32;
    const f32x4* src4 = reinterpret_cast<const f32x4*>(src);
    f32x4 buf[8];
    float amax = 0.f;
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      buf[i] = src4[i];
#pragma unroll
      for (int j = 0; j < 4; ++j) amax = fmaxf(amax, fabsf(buf[i][j]));
    }
    // E8M0 scale, non-clipping: amax/2^e <= 6 (e2m1 max)
    int e = 0;
    if (amax > 0.f) {
      int ex;
      (void)frexpf(amax, &ex);       // amax = m * 2^ex, m in [0.5, 1)
      e = ex - 3;                    // amax / 2^e in [4, 8)
      if (amax * exp2f((float)-e) > 6.f) ++e;  // -> (3, 6], no saturation
    }
    if (e < -127) e = -127;
    if (e > 127) e = 127;
    sc[((long)(g >> 1) * R + r) * 4 + (g & 1) * 2 + kb] =
        (unsigned char)(e + 127);  // tile-major [ceil(groups/2)][R][4]
    const float scl = exp2f((float)e);  // hw cvt divides by scale
    unsigned char* base = q + (long)r * (K >> 1) + g * 32;
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      const int pos = 2 * kb + h;  // LINEAR slot (fp4 lanes own their block)
      // quarter = 16 elements -> 8 bytes = 2 u32 (4 cvt calls each; the
      // byte-select operand must be a literal)
      unsigned int w[2] = {0, 0};
#pragma unroll
      for (int u = 0; u < 2; ++u) {
        const f32x4 v0 = buf[h * 4 + u * 2];
        const f32x4 v1 = buf[h * 4 + u * 2 + 1];
        unsigned int acc = 0;
        acc = __builtin_amdgcn_cvt_scalef32_pk_fp4_f32(acc, v0[0], v0[1], scl, 0);
        acc = __builtin_amdgcn_cvt_scalef32_pk_fp4_f32(acc, v0[2], v0[3], scl, 1);
        acc = __builtin_amdgcn_cvt_scalef32_pk_fp4_f32(acc, v1[0], v1[1], scl, 2);
        acc = __builtin_amdgcn_cvt_scalef32_pk_fp4_f32(acc, v1[2], v1[3], scl, 3);
        w[u] = acc;
      }
      unsigned int* dst =
          reinterpret_cast<unsigned int*>(base + pos * 8);
      dst[0] = w[0];
      dst[1] = w[1];
    }
  }
}

// ---- 8-phase MX fp4 GEMM -------------------------------------------------

constexpr int BM4 = 256, BK4 = 128;  // 128 elements = 64 B per row
constexpr int THREADS4 = 512;
constexpr int HALF4 = 128 * 64;      // bytes per half-tile
constexpr int PARITY4 = 4 * HALF4;   // 32 KiB
constexpr int SC_TILE4 = 2048;
constexpr int SC_BASE4 = 2 * PARITY4;

__device__ __forceinline__ int rot4(int row) {
  return ((row >> 1) + (row >> 2)) & 3;
}
__device__ __forceinline__ int phys_cell4(int row, int cell) {
  return row * 4 + ((cell + rot4(row)) & 3);
}
__device__ __forceinline__ int src_cell4(int q) {
  const int row = q >> 2;
  return ((q & 3) - rot4(row)) & 3;
}

template <int N>
__device__ __forceinline__ void waitvm4() {
  if constexpr (N == 0) asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  else if constexpr (N == 2) asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
  else if constexpr (N == 3) asm volatile("s_waitcnt vmcnt(3)" ::: "memory");
  else if constexpr (N == 5) asm volatile("s_waitcnt vmcnt(5)" ::: "memory");
  else static_assert(N == 0, "unsupported vmcnt");
}

// one 16-B glds per thread: 512 cells = 128 rows x 4
__device__ __forceinline__ void stage_half4(
    const unsigned char* __restrict__ A, const unsigned char* __restrict__ Bt,
    long Kb, int bm, int bn, int tile, int hh, int ksteps,
    char* __restrict__ smem, int t) {
  const int tc = tile < ksteps ? tile : ksteps - 1;
  const long k0 = (long)tc * 64;  // bytes per BK4 tile
  const unsigned char* src = (hh < 2) ? A : Bt;
  const int row0 = ((hh < 2) ? bm : bn) + (hh & 1) * 128;
  char* const base = smem + (size_t)(tile & 1) * PARITY4 +
                     (size_t)hh * HALF4;
  const int r = t >> 2, cell = src_cell4(t);
  const unsigned char* g = src + (long)(row0 + r) * Kb + (k0 + cell * 16);
  // LDS dest must be WAVE-UNIFORM (hw adds lane*16): wave w owns cells
  // [w*64, +64) -> base + w*1024.
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) void*)g,
      (__attribute__((address_space(3))) void*)(base + (t >> 6) * 1024),
      16, 0, 0);
}

__device__ __forceinline__ void stage_scales4(
    const unsigned char* __restrict__ sA, const unsigned char* __restrict__ sB,
    int Mtot, int Ntot, int bm, int bn, int tile, int ksteps,
    char* __restrict__ smem, int w, int lane) {
  const int tc = tile < ksteps ? tile : ksteps - 1;
  const int t = w * 64 + lane;
  // tile-major scales -> coalesced
  const unsigned char* g =
      (t < 256) ? sA + ((long)tc * Mtot + bm + t) * 4
                : sB + ((long)tc * Ntot + bn + (t - 256)) * 4;
  char* const base = smem + SC_BASE4 + (size_t)(tile % 3) * SC_TILE4 + w * 256;
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) void*)g,
      (__attribute__((address_space(3))) void*)base, 4, 0, 0);
}

__device__ __forceinline__ void stage_slot4(int t, int p, int* tile, int* hh) {
  switch (p) {
    case 0: *tile = t + 1; *hh = 3; break;
    case 1: *tile = t + 1; *hh = 1; break;
    case 2: *tile = t + 2; *hh = 0; break;
    default: *tile = t + 2; *hh = 2; break;
  }
}

// 16-B fragment: one b128; regs [0..3] carry the data (fp4 operands use 4
// VGPRs), [4..7] are ignored by the instruction.
__device__ __forceinline__ i32x8v frag4(const char* __restrict__ smem,
                                        int parity, int hh, int row_in_half,
                                        int kg, int fkb) {
  const char* half = smem + (size_t)parity * PARITY4 + (size_t)hh * HALF4;
  const i32x4 v = *reinterpret_cast<const i32x4*>(
      half + phys_cell4(row_in_half, kg * 2 + fkb) * 16);
  return i32x8v{v.x, v.y, v.z, v.w, 0, 0, 0, 0};
}

// the row's 4 scale bytes as one word (extract (w>>8*(kg*2+fkb))&0xff)
__device__ __forceinline__ unsigned scaleword4(const char* __restrict__ smem,
                                               int slot3, int hh,
                                               int row_in_half) {
  const int row = (hh < 2 ? 0 : 1024) + ((hh & 1) * 128 + row_in_half) * 4;
  return *reinterpret_cast<const unsigned*>(
      smem + SC_BASE4 + (size_t)slot3 * SC_TILE4 + row);
}

__global__ __launch_bounds__(THREADS4, 2) void gemm_mx4_8ph_kernel(
    const unsigned char* __restrict__ A, const unsigned char* __restrict__ sA,
    const unsigned char* __restrict__ Bt, const unsigned char* __restrict__ sB,
    float* __restrict__ C, int M, int N, int K, unsigned tiles_m,
    unsigned tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];

  const unsigned nwg = tiles_m * tiles_n;
  const unsigned bid = lpx_xcd_swizzle(blockIdx.x, nwg);
  const int bm = (int)(bid / tiles_n) * BM4;
  const int bn = (int)(bid % tiles_n) * BM4;

  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;
  const int wsr = w >> 2, wsc = w & 3;
  const int fr = lane & 31;
  const int fkb = lane >> 5;
  const long Kb = K >> 1;  // bytes per row
  const int groups = K >> 6;

  f32x16 acc[4][2];
#pragma unroll
  for (int p = 0; p < 4; ++p)
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
      for (int e = 0; e < 16; ++e) acc[p][i][e] = 0.f;

  const int ksteps = (K + BK4 - 1) / BK4;
  const int t512 = threadIdx.x;

  stage_half4(A, Bt, Kb, bm, bn, 0, 0, ksteps, smem, t512);
  stage_half4(A, Bt, Kb, bm, bn, 0, 2, ksteps, smem, t512);
  stage_half4(A, Bt, Kb, bm, bn, 0, 3, ksteps, smem, t512);
  stage_half4(A, Bt, Kb, bm, bn, 0, 1, ksteps, smem, t512);
  stage_scales4(sA, sB, M, N, bm, bn, 0, ksteps, smem, w, lane);
  stage_scales4(sA, sB, M, N, bm, bn, 1, ksteps, smem, w, lane);
  stage_half4(A, Bt, Kb, bm, bn, 1, 0, ksteps, smem, t512);
  stage_half4(A, Bt, Kb, bm, bn, 1, 2, ksteps, smem, t512);
  waitvm4<3>();  // 9 prologue events; tile0 data (4) + sc0 (5th) required
  __builtin_amdgcn_s_barrier();

  if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);  // T5 static young-half priority

  for (int t = 0; t < ksteps; ++t) {
    const int parity = t & 1;
    const int slot3 = t % 3;
    // scale words hoisted per tile; fragment REUSE across phases (the
    // bf16/MX schedule): each 16-B fragment ds_read ONCE per K-tile.
    unsigned awT[2][2], bwT[2];
#pragma unroll
    for (int rh = 0; rh < 2; ++rh)
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
        awT[rh][mf] = scaleword4(smem, slot3, rh, wsr * 64 + mf * 32 + fr);
#pragma unroll
    for (int ch = 0; ch < 2; ++ch)
      bwT[ch] = scaleword4(smem, slot3, 2 + ch, wsc * 32 + fr);
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
            aR[mf][kg] = frag4(smem, parity, rh, wsr * 64 + mf * 32 + fr,
                               kg, fkb);
      }
      if (p == 0) {
#pragma unroll
        for (int kg = 0; kg < 2; ++kg)
          bR0[kg] = frag4(smem, parity, 2, wsc * 32 + fr, kg, fkb);
      } else if (p == 1) {
#pragma unroll
        for (int kg = 0; kg < 2; ++kg)
          bR1[kg] = frag4(smem, parity, 3, wsc * 32 + fr, kg, fkb);
      }
      int st, sh;
      stage_slot4(t, p, &st, &sh);
      stage_half4(A, Bt, Kb, bm, bn, st, sh, ksteps, smem, t512);
      if (p == 0)  // tri-buffered slot (t+2)%3, never the one being read
        stage_scales4(sA, sB, M, N, bm, bn, t + 2, ksteps, smem, w, lane);

      __builtin_amdgcn_s_barrier();
      auto& bC = (ch == 0) ? bR0 : bR1;
      const unsigned* aw = awT[rh];
      const unsigned bw = bwT[ch];
#pragma unroll
      for (int kg = 0; kg < 2; ++kg)
#pragma unroll
        for (int mf = 0; mf < 2; ++mf) {
          const int sa = (int)((aw[mf] >> (8 * (kg * 2 + fkb))) & 0xff);
          const int sb = (int)((bw >> (8 * (kg * 2 + fkb))) & 0xff);
          acc[p][mf] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
              aR[mf][kg], bC[kg], acc[p][mf], 4, 4, 0, sa, 0, sb);
        }
      waitvm4<5>();  // constant: window [2,1,1,1]; scales 8 calls back
      __builtin_amdgcn_s_barrier();
    }
  }
  waitvm4<0>();

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

// ---- BK=256 (wide) variant -----------------------------------------------
// 128-B rows = 256 elements per k-tile: double the MFMA work per barrier
// pair (8 scaled fp4 MFMAs/phase/wave — the same compute-per-phase balance
// that takes the unit-scale fp8 kernel to 2.2 PF). LDS 136 KiB -> 1
// block/CU. Cell rotation (row>>1)&7 is the fp8-MX one (b128-quarter
// conflict-free for 128-B rows).

constexpr int BK4W = 256;              // elements; 128 B per row
constexpr int HALF4W = 128 * 128;      // bytes per half-tile
constexpr int PARITY4W = 4 * HALF4W;   // 64 KiB
constexpr int SC_TILE4W = 4096;        // [A 256x8 B | B 256x8 B]
constexpr int SC_BASE4W = 2 * PARITY4W;

__device__ __forceinline__ int phys_cellw(int row, int cell) {
  return row * 8 + ((cell + (row >> 1)) & 7);
}
__device__ __forceinline__ int src_cellw(int q) {
  const int row = q >> 3;
  return ((q & 7) - (row >> 1)) & 7;
}

template <int N>
__device__ __forceinline__ void waitvmw() {
  if constexpr (N == 0) asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  else if constexpr (N == 6) asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
  else if constexpr (N == 10) asm volatile("s_waitcnt vmcnt(10)" ::: "memory");
  else static_assert(N == 0, "unsupported vmcnt");
}

__device__ __forceinline__ void stage_halfw(
    const unsigned char* __restrict__ A, const unsigned char* __restrict__ Bt,
    long Kb, int bm, int bn, int tile, int hh, int ksteps,
    char* __restrict__ smem, int w, int lane) {
  const int tc = tile < ksteps ? tile : ksteps - 1;
  const long k0 = (long)tc * 128;  // bytes per BK4W tile
  const unsigned char* src = (hh < 2) ? A : Bt;
  const int row0 = ((hh < 2) ? bm : bn) + (hh & 1) * 128;
  char* const base = smem + (size_t)(tile & 1) * PARITY4W +
                     (size_t)hh * HALF4W;
#pragma unroll
  for (int j = 0; j < 2; ++j) {
    const int qbase = (j * 8 + w) * 64;
    const int q = qbase + lane;
    const int r = q >> 3, cell = src_cellw(q);
    const unsigned char* g = src + (long)(row0 + r) * Kb + (k0 + cell * 16);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)g,
        (__attribute__((address_space(3))) void*)(base + qbase * 16), 16, 0, 0);
  }
}

// the tile's 4-KB scale block: 2 x 4-B glds per thread
__device__ __forceinline__ void stage_scalesw(
    const unsigned char* __restrict__ sA, const unsigned char* __restrict__ sB,
    int Mtot, int Ntot, int bm, int bn, int tile, int ksteps,
    char* __restrict__ smem, int w, int lane) {
  const int tc = tile < ksteps ? tile : ksteps - 1;
  char* const base = smem + SC_BASE4W + (size_t)(tile % 3) * SC_TILE4W;
  // BK=256 tile = scale-kt pair {2*tc, 2*tc+1}. glds j stages block
  // kt = 2*tc+j for ALL threads (fully coalesced: 256 consecutive dwords
  // per side); LDS image = [dj 2 KB][A rows | B rows] x 4 B.
#pragma unroll
  for (int j = 0; j < 2; ++j) {
    const int t = w * 64 + lane;            // dword 0..511 within the block
    const int side = t >> 8, r = t & 255;
    const unsigned char* g =
        (side == 0 ? sA + ((long)(2 * tc + j) * Mtot + bm + r) * 4
                   : sB + ((long)(2 * tc + j) * Ntot + bn + r) * 4);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)g,
        (__attribute__((address_space(3))) void*)(base + j * 2048 + w * 256),
        4, 0, 0);
  }
}

__device__ __forceinline__ i32x8v fragw(const char* __restrict__ smem,
                                        int parity, int hh, int row_in_half,
                                        int kg, int fkb) {
  const char* half = smem + (size_t)parity * PARITY4W + (size_t)hh * HALF4W;
  const i32x4 v = *reinterpret_cast<const i32x4*>(
      half + phys_cellw(row_in_half, kg * 2 + fkb) * 16);
  return i32x8v{v.x, v.y, v.z, v.w, 0, 0, 0, 0};
}

// the row's 8 scale bytes ([4 groups][2 blocks]) as a dword pair
__device__ __forceinline__ void scalewordw(const char* __restrict__ smem,
                                           int slot3, int hh,
                                           int row_in_half, unsigned out[2]) {
  // LDS image: [dj 2 KB][A 1 KB | B 1 KB], one dword per row
  const int row = (hh < 2 ? 0 : 1024) + ((hh & 1) * 128 + row_in_half) * 4;
  const char* base = smem + SC_BASE4W + (size_t)slot3 * SC_TILE4W + row;
  out[0] = *reinterpret_cast<const unsigned*>(base);
  out[1] = *reinterpret_cast<const unsigned*>(base + 2048);
}

__global__ __launch_bounds__(THREADS4, 1) void gemm_mx4w_8ph_kernel(
    const unsigned char* __restrict__ A, const unsigned char* __restrict__ sA,
    const unsigned char* __restrict__ Bt, const unsigned char* __restrict__ sB,
    float* __restrict__ C, int M, int N, int K, unsigned tiles_m,
    unsigned tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];

  const unsigned nwg = tiles_m * tiles_n;
  const unsigned bid = lpx_xcd_swizzle(blockIdx.x, nwg);
  const int bm = (int)(bid / tiles_n) * BM4;
  const int bn = (int)(bid % tiles_n) * BM4;

  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;
  const int wsr = w >> 2, wsc = w & 3;
  const int fr = lane & 31;
  const int fkb = lane >> 5;
  const long Kb = K >> 1;
  const int groups = K >> 6;

  f32x16 acc[4][2];
#pragma unroll
  for (int p = 0; p < 4; ++p)
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
      for (int e = 0; e < 16; ++e) acc[p][i][e] = 0.f;

  const int ksteps = (K + BK4W - 1) / BK4W;

  stage_halfw(A, Bt, Kb, bm, bn, 0, 0, ksteps, smem, w, lane);
  stage_halfw(A, Bt, Kb, bm, bn, 0, 2, ksteps, smem, w, lane);
  stage_halfw(A, Bt, Kb, bm, bn, 0, 3, ksteps, smem, w, lane);
  stage_halfw(A, Bt, Kb, bm, bn, 0, 1, ksteps, smem, w, lane);
  stage_scalesw(sA, sB, M, N, bm, bn, 0, ksteps, smem, w, lane);
  stage_scalesw(sA, sB, M, N, bm, bn, 1, ksteps, smem, w, lane);
  stage_halfw(A, Bt, Kb, bm, bn, 1, 0, ksteps, smem, w, lane);
  stage_halfw(A, Bt, Kb, bm, bn, 1, 2, ksteps, smem, w, lane);
  waitvmw<6>();  // 16 events; tile0 data (8) + sc0 (10th) required
  __builtin_amdgcn_s_barrier();

  if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);

  for (int t = 0; t < ksteps; ++t) {
    const int parity = t & 1;
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      const int rh = (p >> 1) & 1;
      const int ch = p & 1;
      i32x8v aF[2][4], bF[4];
      int sa[2][4], sb[4];
      unsigned aw[2][2], bw[2];
      const int slot3 = t % 3;
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
        scalewordw(smem, slot3, rh, wsr * 64 + mf * 32 + fr, aw[mf]);
      scalewordw(smem, slot3, 2 + ch, wsc * 32 + fr, bw);
#pragma unroll
      for (int kg = 0; kg < 4; ++kg) {
        const int sh8 = 8 * (((kg & 1) * 2 + fkb));
#pragma unroll
        for (int mf = 0; mf < 2; ++mf) {
          const int r = wsr * 64 + mf * 32 + fr;
          aF[mf][kg] = fragw(smem, parity, rh, r, kg, fkb);
          sa[mf][kg] = (int)((aw[mf][kg >> 1] >> sh8) & 0xff);
        }
        const int rb = wsc * 32 + fr;
        bF[kg] = fragw(smem, parity, 2 + ch, rb, kg, fkb);
        sb[kg] = (int)((bw[kg >> 1] >> sh8) & 0xff);
      }
      int st, sh;
      stage_slot4(t, p, &st, &sh);
      stage_halfw(A, Bt, Kb, bm, bn, st, sh, ksteps, smem, w, lane);
      if (p == 0)  // tri-buffered slot (t+2)%3
        stage_scalesw(sA, sB, M, N, bm, bn, t + 2, ksteps, smem, w, lane);

      __builtin_amdgcn_s_barrier();
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
#pragma unroll
        for (int kg = 0; kg < 4; ++kg)
          acc[p][mf] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
              aF[mf][kg], bF[kg], acc[p][mf], 4, 4, 0, sa[mf][kg], 0, sb[kg]);
      // events per tile cycle: [4,2,2,2] -> any 4-call window = 10;
      // tri-buffered scales are 8+ calls back, no tight p3 wait
      waitvmw<10>();
      __builtin_amdgcn_s_barrier();
    }
  }
  waitvmw<0>();

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

// ---- tri-parity variant: 3 data buffers, ONE barrier per phase -----------
// The 2-parity kernel is glds/barrier latency-bound (PMC wait 0.33,
// mfma_busy 0.79): with a third data parity the staged slot is never the
// one being read, so the second (post-MFMA) barrier of every phase
// disappears (4 barriers/tile instead of 8) and the counted vmcnt windows
// deepen to 8-11 in-flight events. Costs occupancy: 102 KiB LDS -> 1
// block/CU (2 waves/SIMD) — the A/B decides.
constexpr int SC_BASE3 = 3 * PARITY4;

__device__ __forceinline__ void stage_half4_3p(
    const unsigned char* __restrict__ A, const unsigned char* __restrict__ Bt,
    long Kb, int bm, int bn, int tile, int hh, int ksteps,
    char* __restrict__ smem, int t) {
  const int tc = tile < ksteps ? tile : ksteps - 1;
  const long k0 = (long)tc * 64;
  const unsigned char* src = (hh < 2) ? A : Bt;
  const int row0 = ((hh < 2) ? bm : bn) + (hh & 1) * 128;
  char* const base = smem + (size_t)(tile % 3) * PARITY4 +
                     (size_t)hh * HALF4;
  const int r = t >> 2, cell = src_cell4(t);
  const unsigned char* g = src + (long)(row0 + r) * Kb + (k0 + cell * 16);
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) void*)g,
      (__attribute__((address_space(3))) void*)(base + (t >> 6) * 1024),
      16, 0, 0);
}

__device__ __forceinline__ void stage_scales4_3p(
    const unsigned char* __restrict__ sA, const unsigned char* __restrict__ sB,
    int Mtot, int Ntot, int bm, int bn, int tile, int ksteps,
    char* __restrict__ smem, int w, int lane) {
  const int tc = tile < ksteps ? tile : ksteps - 1;
  const int t = w * 64 + lane;
  const unsigned char* g =
      (t < 256) ? sA + ((long)tc * Mtot + bm + t) * 4
                : sB + ((long)tc * Ntot + bn + (t - 256)) * 4;
  char* const base = smem + SC_BASE3 + (size_t)(tile % 3) * SC_TILE4 + w * 256;
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) void*)g,
      (__attribute__((address_space(3))) void*)base, 4, 0, 0);
}

__device__ __forceinline__ i32x8v frag4_3p(const char* __restrict__ smem,
                                           int par3, int hh, int row_in_half,
                                           int kg, int fkb) {
  const char* half = smem + (size_t)par3 * PARITY4 + (size_t)hh * HALF4;
  const i32x4 v = *reinterpret_cast<const i32x4*>(
      half + phys_cell4(row_in_half, kg * 2 + fkb) * 16);
  return i32x8v{v.x, v.y, v.z, v.w, 0, 0, 0, 0};
}

__device__ __forceinline__ unsigned scaleword4_3p(const char* __restrict__ smem,
                                                  int slot3, int hh,
                                                  int row_in_half) {
  const int row = (hh < 2 ? 0 : 1024) + ((hh & 1) * 128 + row_in_half) * 4;
  return *reinterpret_cast<const unsigned*>(
      smem + SC_BASE3 + (size_t)slot3 * SC_TILE4 + row);
}

template <int N>
__device__ __forceinline__ void waitvm3p() {
  asm volatile("s_waitcnt vmcnt(%0)" ::"i"(N) : "memory");
}

__global__ __launch_bounds__(THREADS4, 1) void gemm_mx4_3p_kernel(
    const unsigned char* __restrict__ A, const unsigned char* __restrict__ sA,
    const unsigned char* __restrict__ Bt, const unsigned char* __restrict__ sB,
    float* __restrict__ C, int M, int N, int K, unsigned tiles_m,
    unsigned tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];

  const unsigned nwg = tiles_m * tiles_n;
  const unsigned bid = lpx_xcd_swizzle(blockIdx.x, nwg);
  const int bm = (int)(bid / tiles_n) * BM4;
  const int bn = (int)(bid % tiles_n) * BM4;

  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;
  const int wsr = w >> 2, wsc = w & 3;
  const int fr = lane & 31;
  const int fkb = lane >> 5;
  const long Kb = K >> 1;

  f32x16 acc[4][2];
#pragma unroll
  for (int p = 0; p < 4; ++p)
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
      for (int e = 0; e < 16; ++e) acc[p][i][e] = 0.f;

  const int ksteps = (K + BK4 - 1) / BK4;
  const int t512 = threadIdx.x;

  // prologue: t0 all, sc0, sc1, t1 all, t2 lo pair — windows derived so the
  // steady per-phase waits stay valid from tile 0 (see counts below)
  stage_half4_3p(A, Bt, Kb, bm, bn, 0, 0, ksteps, smem, t512);
  stage_half4_3p(A, Bt, Kb, bm, bn, 0, 2, ksteps, smem, t512);
  stage_half4_3p(A, Bt, Kb, bm, bn, 0, 3, ksteps, smem, t512);
  stage_half4_3p(A, Bt, Kb, bm, bn, 0, 1, ksteps, smem, t512);
  stage_scales4_3p(sA, sB, M, N, bm, bn, 0, ksteps, smem, w, lane);
  stage_scales4_3p(sA, sB, M, N, bm, bn, 1, ksteps, smem, w, lane);
  stage_half4_3p(A, Bt, Kb, bm, bn, 1, 0, ksteps, smem, t512);
  stage_half4_3p(A, Bt, Kb, bm, bn, 1, 2, ksteps, smem, t512);
  stage_half4_3p(A, Bt, Kb, bm, bn, 1, 3, ksteps, smem, t512);
  stage_half4_3p(A, Bt, Kb, bm, bn, 1, 1, ksteps, smem, t512);
  stage_half4_3p(A, Bt, Kb, bm, bn, 2, 0, ksteps, smem, t512);
  stage_half4_3p(A, Bt, Kb, bm, bn, 2, 2, ksteps, smem, t512);
  waitvm3p<6>();  // everything up to sc1 (t0 halves + both scale slots)
  __builtin_amdgcn_s_barrier();

  if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);

  i32x8v aR3[2][2], bR30[2], bR31[2];
  for (int t = 0; t < ksteps; ++t) {
    const int par3 = t % 3;
    const int slot3 = t % 3;
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      const int rh = (p >> 1) & 1;
      const int ch = p & 1;
      // per-phase counted waits (events/tile = 5: p0 D+S, p1..p3 D):
      //  p0 gates on sc(t)   @ (t-2,p0)-S: 8 younger events
      //  p1 gates on B-hi(t) @ (t-2,p0)-D: 11
      //  p2 gates on A-hi(t) @ (t-2,p1):   10
      //  p3 needs nothing new (barrier only)
      if (p == 0) waitvm3p<8>();
      else if (p == 1) waitvm3p<11>();
      else if (p == 2) waitvm3p<10>();
      __builtin_amdgcn_s_barrier();

      // fragment reuse (see the 2-parity kernel): A pairs shared across
      // the column-half phases, B pairs across the row-half phases
      if (p == 0 || p == 2) {
#pragma unroll
        for (int kg = 0; kg < 2; ++kg)
#pragma unroll
          for (int mf = 0; mf < 2; ++mf)
            aR3[mf][kg] = frag4_3p(smem, par3, rh,
                                   wsr * 64 + mf * 32 + fr, kg, fkb);
      }
      if (p == 0) {
#pragma unroll
        for (int kg = 0; kg < 2; ++kg)
          bR30[kg] = frag4_3p(smem, par3, 2, wsc * 32 + fr, kg, fkb);
      } else if (p == 1) {
#pragma unroll
        for (int kg = 0; kg < 2; ++kg)
          bR31[kg] = frag4_3p(smem, par3, 3, wsc * 32 + fr, kg, fkb);
      }
      unsigned aw[2], bw;
#pragma unroll
      for (int mf = 0; mf < 2; ++mf)
        aw[mf] = scaleword4_3p(smem, slot3, rh, wsr * 64 + mf * 32 + fr);
      bw = scaleword4_3p(smem, slot3, 2 + ch, wsc * 32 + fr);
      // staging schedule: p0 (t+2,B-hi)+sc(t+2), p1 (t+2,A-hi),
      // p2 (t+3,A-lo), p3 (t+3,B-lo) — (t+3) lands in parity t%3 whose
      // lo slots were last read at p1/p2 (>=1 barrier earlier)
      if (p == 0) {
        stage_half4_3p(A, Bt, Kb, bm, bn, t + 2, 3, ksteps, smem, t512);
        stage_scales4_3p(sA, sB, M, N, bm, bn, t + 2, ksteps, smem, w, lane);
      } else if (p == 1) {
        stage_half4_3p(A, Bt, Kb, bm, bn, t + 2, 1, ksteps, smem, t512);
      } else if (p == 2) {
        stage_half4_3p(A, Bt, Kb, bm, bn, t + 3, 0, ksteps, smem, t512);
      } else {
        stage_half4_3p(A, Bt, Kb, bm, bn, t + 3, 2, ksteps, smem, t512);
      }
      auto& bC3 = (ch == 0) ? bR30 : bR31;
#pragma unroll
      for (int kg = 0; kg < 2; ++kg)
#pragma unroll
        for (int mf = 0; mf < 2; ++mf) {
          const int sa = (int)((aw[mf] >> (8 * (kg * 2 + fkb))) & 0xff);
          const int sb = (int)((bw >> (8 * (kg * 2 + fkb))) & 0xff);
          acc[p][mf] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
              aR3[mf][kg], bC3[kg], acc[p][mf], 4, 4, 0, sa, 0, sb);
        }
    }
  }
  waitvm3p<0>();

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

hipError_t quantize_mx_fp4(const float* x, unsigned char* q, unsigned char* sc,
                           int R, int K, hipStream_t stream) {
  if (K % 64 != 0) return hipErrorInvalidValue;
  const long total = (long)R * (K >> 6) * 2;
  unsigned blocks = (unsigned)((total + THREADS - 1) / THREADS);
  if (blocks > 4096u) blocks = 4096u;
  quantize_mx4_kernel<<<dim3(blocks ? blocks : 1), dim3(THREADS), 0, stream>>>(
      x, q, sc, R, K);
  return hipGetLastError();
}

bool gemm_mx_fp4_supported(int M, int N, int K) {
  return M % BM4 == 0 && N % BM4 == 0 && K % BK4 == 0 && K >= BK4;
}

hipError_t gemm_mx_fp4_nt8ph(const unsigned char* A, const unsigned char* sA,
                             const unsigned char* Bt, const unsigned char* sB,
                             float* C, int M, int N, int K,
                             hipStream_t stream) {
  const unsigned tiles_m = (unsigned)(M / BM4), tiles_n = (unsigned)(N / BM4);
  // BK=128 wins at every measured shape after the coalesced tile-major
  // scale staging (2956 vs 2043 TF @8192: 2 blocks/CU beats the wide
  // kernel's per-phase work). Wide kept behind LPX_FP4_WIDE=1 for A/B.
  const char* wenv = getenv("LPX_FP4_WIDE");
  const bool use_wide = wenv && wenv[0] == '1';
  if (use_wide && K % BK4W == 0) {
    const size_t lds = 2 * PARITY4W + 3 * SC_TILE4W;  // 140 KiB
    gemm_mx4w_8ph_kernel<<<dim3(tiles_m * tiles_n), dim3(THREADS4), lds,
                           stream>>>(A, sA, Bt, sB, C, M, N, K, tiles_m,
                                     tiles_n);
    return hipGetLastError();
  }
  // tri-parity wins on long-K shapes (2692 vs 2549 TF @8192 within-probe;
  // loses @4096: 2364 vs 2569 — the deep windows need many K-steps to
  // amortize the occupancy drop). LPX_FP4_3P=0/1 force.
  const char* p3 = getenv("LPX_FP4_3P");
  const bool use3p = p3 ? (p3[0] == '1') : (K >= 6144);
  if (use3p) {
    // tri-parity: 1 barrier/phase, deep vmcnt windows, 1 block/CU (A/B)
    const size_t lds3 = 3 * PARITY4 + 3 * SC_TILE4;  // 102 KiB
    gemm_mx4_3p_kernel<<<dim3(tiles_m * tiles_n), dim3(THREADS4), lds3,
                         stream>>>(A, sA, Bt, sB, C, M, N, K, tiles_m,
                                   tiles_n);
    return hipGetLastError();
  }
  const size_t lds = 2 * PARITY4 + 3 * SC_TILE4;  // 70 KiB -> 2 blocks/CU
  gemm_mx4_8ph_kernel<<<dim3(tiles_m * tiles_n), dim3(THREADS4), lds, stream>>>(
      A, sA, Bt, sB, C, M, N, K, tiles_m, tiles_n);
  return hipGetLastError();
}

}  // namespace lpx
