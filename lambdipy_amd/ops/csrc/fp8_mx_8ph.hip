// MX block-scaled fp8 GEMM v2 for gfx950 — glds 8-phase pipeline.
//
// Same quadrant-phase schedule as gemm_bf16_8ph.hip / fp8_gemm_8ph.hip
// (one half-tile glds prefetch per phase, counted vmcnt across raw
// barriers, T5 static young-half priority) but built around
// v_mfma_scale_f32_32x32x64_f8f6f4, which issues at 2x the plain-fp8 rate
// (4.90 vs 2.43 PF measured register-resident, scripts/mfma_rate.py).
//
// Differences from the plain fp8 8-phase kernel:
//   - fragments are 32 B/lane (2 consecutive 16-B cells), so the LDS cell
//     permutation is rotation by (row>>1)&7 — (row&7) would 2-way conflict
//     the 16-lane b128 quarters (rows step 2 within a bank half).
//   - wave tile is 64x32 of each 128x128 quadrant: 2x1 fragments of 32x32.
//   - per stage call each thread issues 3 glds (2x16-B data + 1x4-B scale
//     dword), keeping vmcnt counts uniform; scale dwords land in a 512-B
//     per-half LDS strip ([128 rows][4 bytes], tile-kt's 2 groups x 2
//     blocks), duplicated by wave pairs (same-source benign).
//   - operands/scales are the shuffled MX packing of quantize_mx_fp8
//     (fp8_mx.hip); scales are applied by the MFMA, no epilogue descale.
// Requires M,N % 256 == 0, K % 128 == 0 (launcher falls back to the
// register-staged MX kernel for other shapes).

#include "lpx_common.h"
#include "kernels.h"

namespace lpx {

namespace {

constexpr int BMX = 256, BKX = 128;  // 128 fp8 elements = 128 B per row
constexpr int THREADSX = 512;
constexpr int HALF_BYTES = 128 * 128;              // data bytes per half-tile
constexpr int PARITY_BYTES = 4 * HALF_BYTES;
constexpr int SC_TILE = 2048;  // per-tile scale block [A 256 rows | B 256]x4B
constexpr int SC_BASE = 2 * PARITY_BYTES;
// scales are TRI-buffered (slot = tile % 3): phase 0 of tile t stages tile
// t+2's block, so the write slot never aliases the slot being read and
// every phase keeps the same relaxed vmcnt (no tight p3 wait).

typedef __attribute__((ext_vector_type(8))) int i32x8v;

// logical 16-B cell -> physical: rotation by (row>>1) within the row
__device__ __forceinline__ int phys_cellx(int row, int cell) {
  return row * 8 + ((cell + (row >> 1)) & 7);
}
// inverse: which logical cell belongs at linear slot q = row*8 + pc
__device__ __forceinline__ int src_cellx(int q) {
  const int row = q >> 3;
  return ((q & 7) - (row >> 1)) & 7;
}

template <int N>
__device__ __forceinline__ void waitvm() {
  if constexpr (N == 0) asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  else if constexpr (N == 4) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  else if constexpr (N == 5) asm volatile("s_waitcnt vmcnt(5)" ::: "memory");
  else if constexpr (N == 6) asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
  else if constexpr (N == 8) asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
  else if constexpr (N == 9) asm volatile("s_waitcnt vmcnt(9)" ::: "memory");
  else if constexpr (N == 12) asm volatile("s_waitcnt vmcnt(12)" ::: "memory");
  else static_assert(N == 0, "unsupported vmcnt");
}

// Stage one half-tile (128 rows x 128 B data, 2 glds per thread).
// hh: 0/1 = A halves, 2/3 = B halves.
__device__ __forceinline__ void stage_halfx(
    const unsigned char* __restrict__ A, const unsigned char* __restrict__ Bt,
    long K, int bm, int bn, int tile, int hh, int ksteps,
    char* __restrict__ smem, int w, int lane) {
  const int tc = tile < ksteps ? tile : ksteps - 1;
  const long k0 = (long)tc * BKX;
  const unsigned char* src = (hh < 2) ? A : Bt;
  const int row0 = ((hh < 2) ? bm : bn) + (hh & 1) * 128;
  char* const base = smem + (size_t)(tile & 1) * PARITY_BYTES +
                     (size_t)hh * HALF_BYTES;
  // data: 1024 cells of 16 B; each thread fetches the global cell that
  // belongs at its linear LDS slot (rotation absorbed on the source side)
#pragma unroll
  for (int j = 0; j < 2; ++j) {
    const int qbase = (j * 8 + w) * 64;
    const int q = qbase + lane;
    const int r = q >> 3, cell = src_cellx(q);
    const unsigned char* g = src + (long)(row0 + r) * K + (k0 + cell * 16);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)g,
        (__attribute__((address_space(3))) void*)(base + qbase * 16), 16, 0, 0);
  }
}

// Stage the WHOLE tile's scales (2 KB: [A rows 0..255 | B rows 0..255] x
// one dword per row = this k-tile's [2 groups][2 blocks]) with exactly ONE
// 4-B glds per thread — no duplication. Issued only in phase 0, so the
// 4-call vmcnt window always contains one scale load (constant counts).
__device__ __forceinline__ void stage_scalesx(
    const unsigned char* __restrict__ sA, const unsigned char* __restrict__ sB,
    int Mtot, int Ntot, int bm, int bn, int tile, int ksteps,
    char* __restrict__ smem, int w, int lane) {
  const int tc = tile < ksteps ? tile : ksteps - 1;
  const int t = w * 64 + lane;
  // tile-major scales: the tile's block is contiguous -> coalesced glds
  const unsigned char* g =
      (t < 256) ? sA + ((long)tc * Mtot + bm + t) * 4
                : sB + ((long)tc * Ntot + bn + (t - 256)) * 4;
  char* const base = smem + SC_BASE + (size_t)(tile % 3) * SC_TILE + w * 256;
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) void*)g,
      (__attribute__((address_space(3))) void*)base, 4, 0, 0);
}

__device__ __forceinline__ void stage_slotx(int t, int p, int* tile, int* hh) {
  switch (p) {
    case 0: *tile = t + 1; *hh = 3; break;
    case 1: *tile = t + 1; *hh = 1; break;
    case 2: *tile = t + 2; *hh = 0; break;
    default: *tile = t + 2; *hh = 2; break;
  }
}

// 32-B fragment (2 x b128) for row_in_half, k-group kg, block half fkb
__device__ __forceinline__ i32x8v fragx(const char* __restrict__ smem,
                                        int parity, int hh, int row_in_half,
                                        int kg, int fkb) {
  const char* half = smem + (size_t)parity * PARITY_BYTES +
                     (size_t)hh * HALF_BYTES;
  const int c0 = kg * 4 + fkb * 2;  // logical cells c0, c0+1
  const i32x4 lo = *reinterpret_cast<const i32x4*>(
      half + phys_cellx(row_in_half, c0) * 16);
  const i32x4 hi = *reinterpret_cast<const i32x4*>(
      half + phys_cellx(row_in_half, c0 + 1) * 16);
  return i32x8v{lo.x, lo.y, lo.z, lo.w, hi.x, hi.y, hi.z, hi.w};
}

// the row's 4 scale bytes ([2 groups][2 blocks]) as one word; extract with
// (w >> 8*(kg*2+fkb)) & 0xff — one LDS read instead of 2+ byte reads.
__device__ __forceinline__ unsigned scaleword(const char* __restrict__ smem,
                                              int slot3, int hh,
                                              int row_in_half) {
  const int row = (hh < 2 ? 0 : 1024) + ((hh & 1) * 128 + row_in_half) * 4;
  return *reinterpret_cast<const unsigned*>(
      smem + SC_BASE + (size_t)slot3 * SC_TILE + row);
}

// SC=true: per-block MX scales from LDS (sA/sB used; descale/bias ignored).
// SC=false: unit scales (literal 127) — plain per-tensor fp8 GEMM on the
// 2x-rate scaled MFMA (the k-permutation is self-consistent between A and
// B, so unshuffled operands contract correctly) with fused descale/bias/relu.
template <bool SC, bool BIAS, bool RELU, bool OUT8 = false,
          bool RES = false>
__global__ __launch_bounds__(THREADSX, 1) void gemm_mx_8ph_kernel(
    const unsigned char* __restrict__ A, const unsigned char* __restrict__ sA,
    const unsigned char* __restrict__ Bt, const unsigned char* __restrict__ sB,
    float* __restrict__ C, unsigned char* __restrict__ C8,
    const float* __restrict__ bias, const float* __restrict__ resid,
    float descale, float out_inv,
    int M, int N, int K, unsigned tiles_m, unsigned tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];

  const unsigned nwg = tiles_m * tiles_n;
  const unsigned bid = lpx_xcd_swizzle(blockIdx.x, nwg);
  const int bm = (int)(bid / tiles_n) * BMX;
  const int bn = (int)(bid % tiles_n) * BMX;

  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;
  const int wsr = w >> 2, wsc = w & 3;  // 2x4 waves; 64x32 of each quadrant
  const int fr = lane & 31;
  const int fkb = lane >> 5;

  f32x16 acc[4][2];  // [quadrant][m-frag], 1 n-frag per wave
#pragma unroll
  for (int p = 0; p < 4; ++p)
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
      for (int e = 0; e < 16; ++e) acc[p][i][e] = 0.f;

  const int ksteps = (K + BKX - 1) / BKX;

  stage_halfx(A, Bt, K, bm, bn, 0, 0, ksteps, smem, w, lane);
  stage_halfx(A, Bt, K, bm, bn, 0, 2, ksteps, smem, w, lane);
  stage_halfx(A, Bt, K, bm, bn, 0, 3, ksteps, smem, w, lane);
  stage_halfx(A, Bt, K, bm, bn, 0, 1, ksteps, smem, w, lane);
  if constexpr (SC) {
    stage_scalesx(sA, sB, M, N, bm, bn, 0, ksteps, smem, w, lane);
    stage_scalesx(sA, sB, M, N, bm, bn, 1, ksteps, smem, w, lane);
  }
  stage_halfx(A, Bt, K, bm, bn, 1, 0, ksteps, smem, w, lane);
  stage_halfx(A, Bt, K, bm, bn, 1, 2, ksteps, smem, w, lane);
  waitvm<SC ? 5 : 4>();
  __builtin_amdgcn_s_barrier();

  if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);  // T5 static young-half priority

  for (int t = 0; t < ksteps; ++t) {
    const int parity = t & 1;
    // hoist the tile's 6 scale words out of the phase loop (they are
    // constant across the 4 phases; 6 LDS reads/tile instead of 12)
    unsigned awT[2][2] = {{0, 0}, {0, 0}}, bwT[2] = {0, 0};
    if constexpr (SC) {
      const int slot3 = t % 3;
#pragma unroll
      for (int rh = 0; rh < 2; ++rh)
#pragma unroll
        for (int mf = 0; mf < 2; ++mf)
          awT[rh][mf] = scaleword(smem, slot3, rh, wsr * 64 + mf * 32 + fr);
#pragma unroll
      for (int ch = 0; ch < 2; ++ch)
        bwT[ch] = scaleword(smem, slot3, 2 + ch, wsc * 32 + fr);
    }
    // fragment REUSE (the bf16 tune-25 schedule, +6-7% there): A fragment
    // pairs are shared across the column-half phases and B pairs across
    // the row-half phases — each 32-B fragment is ds_read ONCE per K-tile
    // (12 fragment reads/wave/K-tile instead of 24). Staging overlap is
    // unchanged and still safe: A-lo/B-lo's LAST LDS reads move EARLIER
    // (phase 0), further from their p2/p3 restaging.
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
            aR[mf][kg] = fragx(smem, parity, rh, wsr * 64 + mf * 32 + fr,
                               kg, fkb);
      }
      if (p == 0) {
#pragma unroll
        for (int kg = 0; kg < 2; ++kg)
          bR0[kg] = fragx(smem, parity, 2, wsc * 32 + fr, kg, fkb);
      } else if (p == 1) {
#pragma unroll
        for (int kg = 0; kg < 2; ++kg)
          bR1[kg] = fragx(smem, parity, 3, wsc * 32 + fr, kg, fkb);
      }
      const unsigned* aw = awT[rh];
      const unsigned bw = bwT[ch];
      int st, sh;
      stage_slotx(t, p, &st, &sh);
      stage_halfx(A, Bt, K, bm, bn, st, sh, ksteps, smem, w, lane);
      if constexpr (SC)
        if (p == 0)  // tile t+2's scale block into slot (t+2)%3 (never the
                     // slot being read), 1 glds/thread, no duplication
          stage_scalesx(sA, sB, M, N, bm, bn, t + 2, ksteps, smem, w, lane);

      __builtin_amdgcn_s_barrier();
      auto& bC = (ch == 0) ? bR0 : bR1;
#pragma unroll
      for (int kg = 0; kg < 2; ++kg)
#pragma unroll
        for (int mf = 0; mf < 2; ++mf) {
          const int sa = SC ? (int)((aw[mf] >> (8 * (kg * 2 + fkb))) & 0xff)
                            : 127;
          const int sb = SC ? (int)((bw >> (8 * (kg * 2 + fkb))) & 0xff)
                            : 127;
          acc[p][mf] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
              aR[mf][kg], bC[kg], acc[p][mf], 0, 0, 0, sa, 0, sb);
        }
      // SC: per-tile-cycle call events are [3,2,2,2]; any 4-call window
      // holds one phase-0, so 9 is constant. Tile t+1's scales were staged
      // at p0 of t-1 (8 calls back, outside the window) — tri-buffered
      // slots make that prefetch distance safe.
      if constexpr (SC) waitvm<9>();
      else waitvm<8>();
      __builtin_amdgcn_s_barrier();
    }
  }
  waitvm<0>();

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
        float v = acc[p][mf][r];
        if constexpr (!SC) v *= descale;
        if constexpr (BIAS) v += bias[col];
        if constexpr (RES) v += resid[(long)row * N + col];  // identity add
        if constexpr (RELU) v = v > 0.f ? v : 0.f;
        if constexpr (OUT8)
          C8[(long)row * N + col] = lpx_f32_to_e4m3(v * out_inv);
        else
          C[(long)row * N + col] = v;
      }
    }
  }
}

}  // namespace

bool gemm_mx_8ph_supported(int M, int N, int K) {
  return M % BMX == 0 && N % BMX == 0 && K % BKX == 0 && K >= BKX;
}

hipError_t gemm_mx_fp8_nt8ph(const unsigned char* A, const unsigned char* sA,
                             const unsigned char* Bt, const unsigned char* sB,
                             float* C, int M, int N, int K,
                             hipStream_t stream) {
  const unsigned tiles_m = (unsigned)(M / BMX), tiles_n = (unsigned)(N / BMX);
  const size_t lds = 2 * PARITY_BYTES + 3 * SC_TILE;  // 134 KiB
  gemm_mx_8ph_kernel<true, false, false>
      <<<dim3(tiles_m * tiles_n), dim3(THREADSX), lds, stream>>>(
          A, sA, Bt, sB, C, nullptr, nullptr, nullptr, 1.f, 1.f, M, N, K, tiles_m,
          tiles_n);
  return hipGetLastError();
}

// Per-tensor-scale fp8 GEMM on the 2x-rate scaled MFMA (unit block scales).
hipError_t gemm_fp8_nt8phx(const unsigned char* A, const unsigned char* Bt,
                           float* C, const float* bias, int M, int N, int K,
                           float descale, bool relu, hipStream_t stream) {
  return gemm_fp8_nt8phx_res(A, Bt, C, bias, nullptr, M, N, K, descale, relu,
                             stream);
}

hipError_t gemm_fp8_nt8phx_res(const unsigned char* A, const unsigned char* Bt,
                               float* C, const float* bias, const float* resid,
                               int M, int N, int K, float descale, bool relu,
                               hipStream_t stream) {
  const unsigned tiles_m = (unsigned)(M / BMX), tiles_n = (unsigned)(N / BMX);
  const size_t lds = 2 * PARITY_BYTES;  // 128 KiB, no scale region
  dim3 grid(tiles_m * tiles_n), block(THREADSX);
  if (resid) {
    // RES is a template param: a runtime per-element branch in the
    // unrolled epilogue generated conditional loads and cost ~4% e2e
    if (bias && relu)
      gemm_mx_8ph_kernel<false, true, true, false, true>
          <<<grid, block, lds, stream>>>(A, nullptr, Bt, nullptr, C, nullptr,
                                         bias, resid, descale, 1.f, M, N, K,
                                         tiles_m, tiles_n);
    else if (bias)
      gemm_mx_8ph_kernel<false, true, false, false, true>
          <<<grid, block, lds, stream>>>(A, nullptr, Bt, nullptr, C, nullptr,
                                         bias, resid, descale, 1.f, M, N, K,
                                         tiles_m, tiles_n);
    else if (relu)
      gemm_mx_8ph_kernel<false, false, true, false, true>
          <<<grid, block, lds, stream>>>(A, nullptr, Bt, nullptr, C, nullptr,
                                         nullptr, resid, descale, 1.f, M, N,
                                         K, tiles_m, tiles_n);
    else
      gemm_mx_8ph_kernel<false, false, false, false, true>
          <<<grid, block, lds, stream>>>(A, nullptr, Bt, nullptr, C, nullptr,
                                         nullptr, resid, descale, 1.f, M, N,
                                         K, tiles_m, tiles_n);
    return hipGetLastError();
  }
  if (bias && relu)
    gemm_mx_8ph_kernel<false, true, true><<<grid, block, lds, stream>>>(
        A, nullptr, Bt, nullptr, C, nullptr, bias, resid, descale, 1.f, M, N, K,
        tiles_m, tiles_n);
  else if (bias)
    gemm_mx_8ph_kernel<false, true, false><<<grid, block, lds, stream>>>(
        A, nullptr, Bt, nullptr, C, nullptr, bias, resid, descale, 1.f, M, N, K,
        tiles_m, tiles_n);
  else if (relu)
    gemm_mx_8ph_kernel<false, false, true><<<grid, block, lds, stream>>>(
        A, nullptr, Bt, nullptr, C, nullptr, nullptr, resid, descale, 1.f, M, N, K,
        tiles_m, tiles_n);
  else
    gemm_mx_8ph_kernel<false, false, false><<<grid, block, lds, stream>>>(
        A, nullptr, Bt, nullptr, C, nullptr, nullptr, resid, descale, 1.f, M, N, K,
        tiles_m, tiles_n);
  return hipGetLastError();
}

// Same with fused epilogue quantization (out bytes = e4m3(v * out_inv)).
hipError_t gemm_fp8_nt8phx_q8(const unsigned char* A, const unsigned char* Bt,
                              unsigned char* C8, const float* bias, int M,
                              int N, int K, float descale, bool relu,
                              float out_inv, hipStream_t stream) {
  const unsigned tiles_m = (unsigned)(M / BMX), tiles_n = (unsigned)(N / BMX);
  const size_t lds = 2 * PARITY_BYTES;
  dim3 grid(tiles_m * tiles_n), block(THREADSX);
  if (bias && relu)
    gemm_mx_8ph_kernel<false, true, true, true><<<grid, block, lds, stream>>>(
        A, nullptr, Bt, nullptr, nullptr, C8, bias, nullptr, descale, out_inv, M, N, K,
        tiles_m, tiles_n);
  else if (bias)
    gemm_mx_8ph_kernel<false, true, false, true><<<grid, block, lds, stream>>>(
        A, nullptr, Bt, nullptr, nullptr, C8, bias, nullptr, descale, out_inv, M, N, K,
        tiles_m, tiles_n);
  else if (relu)
    gemm_mx_8ph_kernel<false, false, true, true><<<grid, block, lds, stream>>>(
        A, nullptr, Bt, nullptr, nullptr, C8, nullptr, nullptr, descale, out_inv, M, N,
        K, tiles_m, tiles_n);
  else
    gemm_mx_8ph_kernel<false, false, false, true><<<grid, block, lds, stream>>>(
        A, nullptr, Bt, nullptr, nullptr, C8, nullptr, nullptr, descale, out_inv, M, N,
        K, tiles_m, tiles_n);
  return hipGetLastError();
}

}  // namespace lpx
