// MX block-scaled fp8 GEMM for gfx950 — v_mfma_scale_f32_32x32x64_f8f6f4.
//
// The ONLY path to the ~4.6 PF fp8 ceiling (guide §3: non-scaled fp8 runs at
// the bf16 rate; MX K=64/128 doubles it). Empirically derived semantics
// (scripts/mx_sweep.py, one-lane scale perturbation with quarter-marked
// data): for the 32x32x64 shape, lane l sources row/col (l&31); lane pair
// {l, l+32} concatenates to that row's 64 k-bytes. Lane l's SINGLE scale
// byte (always byte0 of the scale VGPR; byte1 unused) applies to BYTE-HALF
// (l>>5) of BOTH lanes of its row — with plain contiguous packing the scale
// blocks would be the interleaved quarters {kb, kb+2}. We pack each 64-k
// group SHUFFLED as [q0 | q2 | q1 | q3] (quarter = 16 elements) instead,
// which makes the hardware scale blocks equal the standard contiguous OCP
// MX-32 blocks {2kb, 2kb+1}:
//   - quantize_mx_fp8 quantizes per contiguous 32-element block (amax ->
//     E8M0 2^e, e4m3 payload), writing quarters to their shuffled slots;
//   - scale sc[r][g][kb] covers original k [g*64 + kb*32, +32);
//   - the MFMA applies scales in hardware; NO epilogue descale is needed.
// C/D layout (32x32): col = l&31, row = (r&3) + 8*(r>>2) + 4*(l>>5) — probed.

#include "lpx_common.h"

namespace lpx {

namespace {

constexpr int THREADS = 256;

typedef __attribute__((ext_vector_type(8))) int i32x8v;
typedef __attribute__((ext_vector_type(2))) short s16x2;

// ---- MX quantization -----------------------------------------------------
// x f32 [R][K] row-major, K % 64 == 0. Outputs:
//   q  u8 [R][K]      shuffled e4m3 payload ([q0|q2|q1|q3] per 64-group)
//   sc E8M0 block scales, TILE-MAJOR [ceil(groups/2)][R][4]: one BK=128
//   GEMM tile's scale block is contiguous (rows x 4 B), so the GEMM's
//   scale staging is fully coalesced (row-major scales would cost ~16x
//   their bytes in L2 lines). Byte (g&1)*2 + kb of row slot r covers
//   original k [g*64 + kb*32, +32).
__global__ __launch_bounds__(THREADS) void quantize_mx_kernel(
    const float* __restrict__ x, unsigned char* __restrict__ q,
    unsigned char* __restrict__ sc, int R, int K) {
  const int groups = K >> 6;
  const long total = (long)R * groups * 2;  // one thread per (row, group, kb)
  for (long idx = (long)blockIdx.x * THREADS + threadIdx.x; idx < total;
       idx += (long)gridDim.x * THREADS) {
    const int kb = (int)(idx & 1);
    const int g = (int)((idx >> 1) % groups);
    const int r = (int)(idx / (2 * groups));
    const float* src = x + (long)r * K + g * 64 + kb * 32;
    // standard contiguous MX-32 block: original k [g*64 + kb*32, +32).
    // The MFMA applies lane-kb's scale to byte-half kb of BOTH k-lanes;
    // with the [q0|q2|q1|q3] data shuffle that is exactly this block.
    const f32x4* src4 = reinterpret_cast<const f32x4*>(src);
    f32x4 buf[8];
    float amax = 0.f;
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      buf[i] = src4[i];
#pragma unroll
      for (int j = 0; j < 4; ++j) amax = fmaxf(amax, fabsf(buf[i][j]));
    }
    // E8M0 shared scale: 2^e, non-clipping (amax/2^e <= 448, the e4m3 max)
    int e = 0;
    if (amax > 0.f) {
      int ex;
      (void)frexpf(amax, &ex);      // amax = m * 2^ex, m in [0.5, 1)
      e = ex - 9;                   // amax / 2^e in [256, 512)
      if (amax * exp2f((float)-e) > 448.f) ++e;  // avoid saturating block max
    }
    if (e < -127) e = -127;
    if (e > 127) e = 127;
    sc[((long)(g >> 1) * R + r) * 4 + (g & 1) * 2 + kb] =
        (unsigned char)(e + 127);
    const float scl = exp2f((float)e);  // hw cvt divides by the scale operand
    unsigned char* base = q + (long)r * K + g * 64;
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      const int qq = 2 * kb + h;                    // original quarter index
      const int pos = ((qq & 1) << 1) | (qq >> 1);  // shuffled slot
      // v_cvt_scalef32_pk_fp8_f32: RNE fp8(src/scale), 2 elems per call
      s16x2 pk[4];
#pragma unroll
      for (int jj = 0; jj < 4; ++jj) {
        const f32x4 v = buf[h * 4 + jj];
        s16x2 t = {0, 0};
        t = __builtin_amdgcn_cvt_scalef32_pk_fp8_f32(t, v[0], v[1], scl, false);
        t = __builtin_amdgcn_cvt_scalef32_pk_fp8_f32(t, v[2], v[3], scl, true);
        pk[jj] = t;
      }
      i32x4 out;
      out.x = (int)(unsigned short)pk[0][0] | ((int)(unsigned short)pk[0][1] << 16);
      out.y = (int)(unsigned short)pk[1][0] | ((int)(unsigned short)pk[1][1] << 16);
      out.z = (int)(unsigned short)pk[2][0] | ((int)(unsigned short)pk[2][1] << 16);
      out.w = (int)(unsigned short)pk[3][0] | ((int)(unsigned short)pk[3][1] << 16);
      *reinterpret_cast<i32x4*>(base + pos * 16) = out;
    }
  }
}

// ---- MX GEMM (v2 register-staged structure, 256x256 tile, BK=64 elems) ----

constexpr int BM = 256, BN = 256, BK = 64;  // 64 fp8 bytes per row per tile
constexpr int THREADS2 = 512;
constexpr int ROWB = BK + 16;               // 80-B padded LDS rows
constexpr int CHUNKS = BM * BK / 16 / THREADS2;  // 16-B chunks/thread = 2

__device__ __forceinline__ void stage_mx(const unsigned char* __restrict__ src,
                                         int rows_total, int K, int row0,
                                         int k0, i32x4 (&regs)[CHUNKS]) {
  const int t = threadIdx.x;
#pragma unroll
  for (int i = 0; i < CHUNKS; ++i) {
    const int c = t + i * THREADS2;
    const int r = c >> 2;
    const int cell = (c & 3) * 16;
    const int grow = row0 + r;
    const int gk = k0 + cell;
    i32x4 v = {0, 0, 0, 0};
    if (grow < rows_total && gk + 16 <= K)
      v = *reinterpret_cast<const i32x4*>(src + (long)grow * K + gk);
    regs[i] = v;
  }
}

__device__ __forceinline__ void write_mx_lds(const i32x4 (&regs)[CHUNKS],
                                             char* __restrict__ lds) {
  const int t = threadIdx.x;
#pragma unroll
  for (int i = 0; i < CHUNKS; ++i) {
    const int c = t + i * THREADS2;
    const int r = c >> 2;
    const int cell = (c & 3) * 16;
    *reinterpret_cast<i32x4*>(lds + r * ROWB + cell) = regs[i];
  }
}

// stage the tile's 2x256 scale bytes (per side) into LDS registers
__device__ __forceinline__ unsigned stage_sc(const unsigned char* __restrict__ sc,
                                             int rows_total, int groups,
                                             int row0, int ks) {
  // threads 0..127 cover one side's 256 rows x 2 scales as u16 pairs
  const int t = threadIdx.x;
  if (t >= 128) return 0;
  const int r = row0 + t * 2;
  unsigned v = 0x7f7f;  // scale 1.0 pair for out-of-range rows
  const unsigned char* base =
      sc + ((long)(ks >> 1) * rows_total) * 4 + (ks & 1) * 2;
  if (r + 1 < rows_total) {
    const unsigned short a =
        *reinterpret_cast<const unsigned short*>(base + (long)r * 4);
    const unsigned short b =
        *reinterpret_cast<const unsigned short*>(base + (long)(r + 1) * 4);
    v = (unsigned)a | ((unsigned)b << 16);
  } else if (r < rows_total) {
    v = *reinterpret_cast<const unsigned short*>(base + (long)r * 4);
  }
  return v;
}

__global__ __launch_bounds__(THREADS2, 2) void gemm_mx_kernel(
    const unsigned char* __restrict__ A, const unsigned char* __restrict__ sA,
    const unsigned char* __restrict__ Bt, const unsigned char* __restrict__ sB,
    float* __restrict__ C, int M, int N, int K, unsigned tiles_m,
    unsigned tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* const lA = smem;                      // [256][80]
  char* const lB = smem + BM * ROWB;
  unsigned char* const lsA =
      reinterpret_cast<unsigned char*>(smem + 2 * BM * ROWB);  // [256][2]
  unsigned char* const lsB = lsA + BM * 2;

  const unsigned nwg = tiles_m * tiles_n;
  const unsigned bid = lpx_xcd_swizzle(blockIdx.x, nwg);
  const int bm = (int)(bid / tiles_n) * BM;
  const int bn = (int)(bid % tiles_n) * BN;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = wid >> 2, wc = wid & 3;  // 2x4 wave grid; 128x64 per wave

  const int fr = lane & 31;   // fragment row/col (32-wide)
  const int fkb = lane >> 5;  // 0/1: which interleaved 32-byte half

  // acc: 4 m-frags x 2 n-frags of 32x32 -> 8 x f32x16 = 128 VGPR
  f32x16 acc[4][2];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int e = 0; e < 16; ++e) acc[i][j][e] = 0.f;

  const int groups = K >> 6;

  i32x4 aRegs[CHUNKS], bRegs[CHUNKS];
  unsigned scA, scB;
  stage_mx(A, M, K, bm, 0, aRegs);
  stage_mx(Bt, N, K, bn, 0, bRegs);
  scA = stage_sc(sA, M, groups, bm, 0);
  scB = stage_sc(sB, N, groups, bn, 0);
  write_mx_lds(aRegs, lA);
  write_mx_lds(bRegs, lB);
  if (threadIdx.x < 128) {
    reinterpret_cast<unsigned*>(lsA)[threadIdx.x] = scA;
    reinterpret_cast<unsigned*>(lsB)[threadIdx.x] = scB;
  }
  __syncthreads();

  if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);  // T5 static form

  const int ksteps = groups;
  for (int ks = 0; ks < ksteps; ++ks) {
    if (ks + 1 < ksteps) {
      stage_mx(A, M, K, bm, (ks + 1) * BK, aRegs);
      stage_mx(Bt, N, K, bn, (ks + 1) * BK, bRegs);
      scA = stage_sc(sA, M, groups, bm, ks + 1);
      scB = stage_sc(sB, N, groups, bn, ks + 1);
    }
    // fragments: 32 B per lane = 2 x b128 at [row][fkb*32 .. +32)
    i32x8v aF[4], bF[2];
    int sa[4], sb[2];
#pragma unroll
    for (int mf = 0; mf < 4; ++mf) {
      const int r = wr * 128 + mf * 32 + fr;
      const char* p = lA + r * ROWB + fkb * 32;
      i32x4 lo = *reinterpret_cast<const i32x4*>(p);
      i32x4 hi = *reinterpret_cast<const i32x4*>(p + 16);
      aF[mf] = i32x8v{lo.x, lo.y, lo.z, lo.w, hi.x, hi.y, hi.z, hi.w};
      sa[mf] = lsA[r * 2 + fkb];
    }
#pragma unroll
    for (int nf = 0; nf < 2; ++nf) {
      const int r = wc * 64 + nf * 32 + fr;
      const char* p = lB + r * ROWB + fkb * 32;
      i32x4 lo = *reinterpret_cast<const i32x4*>(p);
      i32x4 hi = *reinterpret_cast<const i32x4*>(p + 16);
      bF[nf] = i32x8v{lo.x, lo.y, lo.z, lo.w, hi.x, hi.y, hi.z, hi.w};
      sb[nf] = lsB[r * 2 + fkb];
    }
#pragma unroll
    for (int mf = 0; mf < 4; ++mf)
#pragma unroll
      for (int nf = 0; nf < 2; ++nf)
        acc[mf][nf] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
            aF[mf], bF[nf], acc[mf][nf], 0, 0, 0, sa[mf], 0, sb[nf]);
    __syncthreads();
    if (ks + 1 < ksteps) {
      write_mx_lds(aRegs, lA);
      write_mx_lds(bRegs, lB);
      if (threadIdx.x < 128) {
        reinterpret_cast<unsigned*>(lsA)[threadIdx.x] = scA;
        reinterpret_cast<unsigned*>(lsB)[threadIdx.x] = scB;
      }
    }
    __syncthreads();
  }

  // epilogue: 32x32 D map per fragment
  const int ocol_l = lane & 31;
  const int orow_base = 4 * (lane >> 5);
#pragma unroll
  for (int mf = 0; mf < 4; ++mf) {
#pragma unroll
    for (int nf = 0; nf < 2; ++nf) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int row = bm + wr * 128 + mf * 32 + (r & 3) + 8 * (r >> 2) +
                        orow_base;
        const int col = bn + wc * 64 + nf * 32 + ocol_l;
        if (row < M && col < N) C[(long)row * N + col] = acc[mf][nf][r];
      }
    }
  }
}

}  // namespace

hipError_t quantize_mx_fp8(const float* x, unsigned char* q, unsigned char* sc,
                           int R, int K, hipStream_t stream) {
  if (K % 64 != 0) return hipErrorInvalidValue;
  const long total = (long)R * (K >> 6) * 2;
  unsigned blocks = (unsigned)((total + THREADS - 1) / THREADS);
  if (blocks > 4096u) blocks = 4096u;
  quantize_mx_kernel<<<dim3(blocks ? blocks : 1), dim3(THREADS), 0, stream>>>(
      x, q, sc, R, K);
  return hipGetLastError();
}

hipError_t gemm_mx_fp8_nt(const unsigned char* A, const unsigned char* sA,
                          const unsigned char* Bt, const unsigned char* sB,
                          float* C, int M, int N, int K, hipStream_t stream) {
  if (K % 64 != 0) return hipErrorInvalidValue;
  const unsigned tiles_m = lpx_ceil_div(M, BM), tiles_n = lpx_ceil_div(N, BN);
  const size_t lds = (size_t)2 * BM * ROWB + 4 * BM;  // 40 KiB + scales
  gemm_mx_kernel<<<dim3(tiles_m * tiles_n), dim3(THREADS2), lds, stream>>>(
      A, sA, Bt, sB, C, M, N, K, tiles_m, tiles_n);
  return hipGetLastError();
}

}  // namespace lpx
