#include "hip/hip_runtime.h"
// bf16 GEMM v2 for gfx950 — 256x256 tile, BK=64, 8 waves, register-staged
// T14 (write-after-barrier) pipeline.
//
// Structure (cdna_hip_programming.md §5 'glds vs register staging' table +
// T14): one 256x256 output tile per 512-thread workgroup (8 waves as 2Mx4N,
// 128x64 per wave), K stepped by 64. Staging is global->register->LDS with
// the T14 split: next-tile loads ISSUE before the compute phase (HBM latency
// hides under 64 MFMAs/wave), the ds_write pass runs AFTER the barrier that
// ends the compute phase, into the single LDS buffer (A 36 KiB + B 36 KiB,
// 144-B padded rows -> <=2-way ds_read_b128 conflicts). Per the guide's
// measured ladder this register-staged 256^2/BK64 shape lands within ~6% of
// the glds variant (~1.1 PF/s class at 4096^3) without glds' lane-linear
// LDS-image constraint.
//
// Fragment maps are identical to gemm_bf16.hip (v_mfma_f32_16x16x32_bf16).

#include "lpx_common.h"

namespace lpx {

namespace {

constexpr int BM = 256, BN = 256, BK = 64;
constexpr int THREADS = 512;            // 8 waves, 2(M) x 4(N)
constexpr int ROW_SHORTS = BK + 8;      // 144-B padded LDS row
constexpr int CHUNKS = BM * BK / 8 / THREADS;  // 16-B chunks per thread = 4

using bf16frag = bf16x8;

// global -> registers (coalesced 16 B/lane; zero-filled out of range)
__device__ __forceinline__ void load_tile_regs(const uint16_t* __restrict__ src,
                                               int rows_total, int K, int row0,
                                               int k0, bf16x8 (&regs)[CHUNKS]) {
  const int t = threadIdx.x;
#pragma unroll
  for (int i = 0; i < CHUNKS; ++i) {
    const int c = t + i * THREADS;
    const int r = c >> 3;          // row within tile (8 chunks per 128-B row)
    const int cell = c & 7;        // 16-B cell within row
    const int grow = row0 + r;
    const int gk = k0 + cell * 8;
    if (grow < rows_total && gk + 8 <= K) {
      regs[i] = *reinterpret_cast<const bf16x8*>(src + (long)grow * K + gk);
    } else {
      short z[8] = {0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
      for (int j = 0; j < 8; ++j)
        if (grow < rows_total && gk + j < K)
          z[j] = (short)src[(long)grow * K + gk + j];
      regs[i] = *reinterpret_cast<const bf16x8*>(z);
    }
  }
}

__device__ __forceinline__ void write_tile_lds(const bf16x8 (&regs)[CHUNKS],
                                               short* __restrict__ lds) {
  const int t = threadIdx.x;
#pragma unroll
  for (int i = 0; i < CHUNKS; ++i) {
    const int c = t + i * THREADS;
    const int r = c >> 3;
    const int cell = c & 7;
    *reinterpret_cast<bf16x8*>(lds + r * ROW_SHORTS + cell * 8) = regs[i];
  }
}

template <bool BF16_OUT>
__global__ __launch_bounds__(THREADS, 2) void gemm_bf16_nt256_kernel(
    const uint16_t* __restrict__ A, const uint16_t* __restrict__ Bt,
    void* __restrict__ Cout, int M, int N, int K, unsigned tiles_m,
    unsigned tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  short* const lA = reinterpret_cast<short*>(smem);
  short* const lB = lA + BM * ROW_SHORTS;

  const unsigned nwg = tiles_m * tiles_n;
  const unsigned bid = lpx_xcd_swizzle(blockIdx.x, nwg);
  const int bm = (int)(bid / tiles_n) * BM;
  const int bn = (int)(bid % tiles_n) * BN;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = wid >> 2, wc = wid & 3;  // 2x4 wave grid; 128x64 per wave

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int frag_row = lane & 15;
  const int frag_kb = lane >> 4;  // 8-wide k block within a 32-deep k-step

  bf16x8 aRegs[CHUNKS], bRegs[CHUNKS];
  load_tile_regs(A, M, K, bm, 0, aRegs);
  load_tile_regs(Bt, N, K, bn, 0, bRegs);
  write_tile_lds(aRegs, lA);
  write_tile_lds(bRegs, lB);
  __syncthreads();

  const int ksteps = (K + BK - 1) / BK;
  for (int ks = 0; ks < ksteps; ++ks) {
    // T14 STAGE_LOAD: issue next tile's global loads before compute
    if (ks + 1 < ksteps) {
      load_tile_regs(A, M, K, bm, (ks + 1) * BK, aRegs);
      load_tile_regs(Bt, N, K, bn, (ks + 1) * BK, bRegs);
    }
    // compute: 2 k-sub-steps of 32, 32 MFMAs each
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      const int cell = kk * 4 + frag_kb;  // 16-B cell holding this k block
      bf16frag aF[8], bF[4];
#pragma unroll
      for (int mf = 0; mf < 8; ++mf) {
        const int r = wr * 128 + mf * 16 + frag_row;
        aF[mf] = *reinterpret_cast<const bf16frag*>(lA + r * ROW_SHORTS + cell * 8);
      }
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        const int r = wc * 64 + nf * 16 + frag_row;
        bF[nf] = *reinterpret_cast<const bf16frag*>(lB + r * ROW_SHORTS + cell * 8);
      }
#pragma unroll
      for (int mf = 0; mf < 8; ++mf)
#pragma unroll
        for (int nf = 0; nf < 4; ++nf)
          acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              aF[mf], bF[nf], acc[mf][nf], 0, 0, 0);
    }
    __syncthreads();  // all waves done reading this tile
    if (ks + 1 < ksteps) {
      // T14 STAGE_WRITE: after the barrier, re-using the same registers
      write_tile_lds(aRegs, lA);
      write_tile_lds(bRegs, lB);
    }
    __syncthreads();
  }

  const int orow0 = bm + wr * 128 + (lane >> 4) * 4;
  const int ocol0 = bn + wc * 64 + (lane & 15);
#pragma unroll
  for (int mf = 0; mf < 8; ++mf) {
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = orow0 + mf * 16 + r;
        const int col = ocol0 + nf * 16;
        if (row < M && col < N) {
          if constexpr (BF16_OUT)
            reinterpret_cast<uint16_t*>(Cout)[(long)row * N + col] =
                (uint16_t)lpx_f32_to_bf16(acc[mf][nf][r]);
          else
            reinterpret_cast<float*>(Cout)[(long)row * N + col] = acc[mf][nf][r];
        }
      }
    }
  }
}

}  // namespace

template <bool BF16_OUT>
static hipError_t launch256(const uint16_t* A, const uint16_t* Bt, void* C,
                            int M, int N, int K, hipStream_t stream) {
  const unsigned tiles_m = lpx_ceil_div(M, BM), tiles_n = lpx_ceil_div(N, BN);
  const size_t lds = (size_t)2 * BM * ROW_SHORTS * sizeof(short);  // 72 KiB
 hipLaunchKernelGGL(( gemm_bf16_nt256_kernel<BF16_OUT>), dim3(dim3(tiles_m * tiles_n)), dim3(dim3(THREADS)),
                                     lds, stream, A, Bt, C, M, N, K, tiles_m,
                                                    tiles_n);
  return hipGetLastError();
}

hipError_t gemm_bf16_nt256(const uint16_t* A, const uint16_t* Bt, float* C,
                           int M, int N, int K, hipStream_t stream) {
  return launch256<false>(A, Bt, C, M, N, K, stream);
}

hipError_t gemm_bf16_nt256_bf16out(const uint16_t* A, const uint16_t* Bt,
                                   uint16_t* C, int M, int N, int K,
                                   hipStream_t stream) {
  return launch256<true>(A, Bt, C, M, N, K, stream);
}

}  // namespace lpx
