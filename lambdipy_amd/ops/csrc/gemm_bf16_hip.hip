#include "hip/hip_runtime.h"
// bf16 GEMM for gfx950 (CDNA4) — MFMA 16x16x32, LDS-tiled.
//
// v1 structure (the optimization ladder's step-2 shape, register-staged with
// +16B row padding instead of an XOR swizzle — cdna_hip_programming.md §5):
//   - 128x128 output tile per 256-thread block (4 waves, 2x2), BK=32
//   - NT layout: A[M][K], Bt[N][K], both row-major K-contiguous, so A and B
//     staging are symmetric 16 B/lane coalesced loads
//   - LDS tiles [128][BK] bf16 padded to 80 B rows (conflict-free b128 reads)
//   - per K-step each wave issues 4+4 ds_read_b128 and 16 MFMAs
//   - XCD-aware block swizzle (T1, bijective form) for L2 locality
//
// MFMA fragment maps (v_mfma_f32_16x16x32_bf16):
//   A: lane l holds A[row = l&15][k = (l>>4)*8 + 0..7]   (16 B, k-contiguous)
//   B: lane l holds B[k = (l>>4)*8 + 0..7][col = l&15]   (same map on Bt rows)
//   D: lane l, reg r -> row = (l>>4)*4 + r, col = l&15
// (guide §3 'Fragment layout'; verified on-GPU by tests/test_ops_gpu.py's
//  asymmetric-operand checks).

#include "lpx_common.h"

namespace lpx {

namespace {

constexpr int BM = 128, BN = 128, BK = 32;
constexpr int THREADS = 256;            // 4 waves, 2x2 wave grid
constexpr int ROW_BYTES = BK * 2 + 16;  // 80 B padded LDS row
constexpr int ROW_SHORTS = ROW_BYTES / 2;

using bf16frag = bf16x8;

__device__ __forceinline__ void stage_tile_nt(const uint16_t* __restrict__ src,
                                              int rows_total, int K,
                                              int row0, int k0,
                                              short* __restrict__ lds) {
  // 128 rows x 32 bf16 (64 B) per tile; 256 threads x 2 chunks of 16 B.
  // chunk c (0..511): row = c / 4, 16-B block = c % 4.
  const int t = threadIdx.x;
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int c = t + i * THREADS;
    const int r = c >> 2;
    const int kb = c & 3;  // 16-B block within the row (8 bf16)
    const int grow = row0 + r;
    const int gk = k0 + kb * 8;
    bf16x8 v;
    if (grow < rows_total && gk + 8 <= K) {
      v = *reinterpret_cast<const bf16x8*>(src + (long)grow * K + gk);
    } else {
      short z[8] = {0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
      for (int j = 0; j < 8; ++j)
        if (grow < rows_total && gk + j < K) z[j] = (short)src[(long)grow * K + gk + j];
      v = *reinterpret_cast<const bf16x8*>(z);
    }
    *reinterpret_cast<bf16x8*>(lds + r * ROW_SHORTS + kb * 8) = v;
  }
}

template <bool BF16_OUT>
__global__ __launch_bounds__(THREADS) void gemm_bf16_nt_kernel(
    const uint16_t* __restrict__ A, const uint16_t* __restrict__ Bt,
    void* __restrict__ Cout, int M, int N, int K, unsigned tiles_m,
    unsigned tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  short* const sbase = reinterpret_cast<short*>(smem);
  // buffer layout: [A0 | A1 | B0 | B1], each BM*ROW_SHORTS shorts
  const int BUF = BM * ROW_SHORTS;

  const unsigned nwg = tiles_m * tiles_n;
  const unsigned bid = lpx_xcd_swizzle(blockIdx.x, nwg);
  const int bm = (int)(bid / tiles_n) * BM;
  const int bn = (int)(bid % tiles_n) * BN;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = wid >> 1, wc = wid & 1;  // wave sub-tile (64x64)

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int frag_row = lane & 15;      // row within 16x16 fragment
  const int frag_kb = lane >> 4;       // which 8-wide k block (0..3)

  int cur = 0;
  stage_tile_nt(A, M, K, bm, 0, sbase);
  stage_tile_nt(Bt, N, K, bn, 0, sbase + 2 * BUF);
  __syncthreads();

  const int ksteps = (K + BK - 1) / BK;
  for (int ks = 0; ks < ksteps; ++ks) {
    short* const aCur = sbase + cur * BUF;
    short* const bCur = sbase + 2 * BUF + cur * BUF;
    if (ks + 1 < ksteps) {
      stage_tile_nt(A, M, K, bm, (ks + 1) * BK, sbase + (cur ^ 1) * BUF);
      stage_tile_nt(Bt, N, K, bn, (ks + 1) * BK, sbase + 2 * BUF + (cur ^ 1) * BUF);
    }
    // fragment loads: lane reads its 16 B at row (sub-tile row) x k-block
    bf16frag aF[4], bF[4];
#pragma unroll
    for (int mf = 0; mf < 4; ++mf) {
      const int r = wr * 64 + mf * 16 + frag_row;
      aF[mf] = *reinterpret_cast<const bf16frag*>(aCur + r * ROW_SHORTS + frag_kb * 8);
    }
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      const int r = wc * 64 + nf * 16 + frag_row;
      bF[nf] = *reinterpret_cast<const bf16frag*>(bCur + r * ROW_SHORTS + frag_kb * 8);
    }
#pragma unroll
    for (int mf = 0; mf < 4; ++mf)
#pragma unroll
      for (int nf = 0; nf < 4; ++nf)
        acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            aF[mf], bF[nf], acc[mf][nf], 0, 0, 0);
    __syncthreads();
    cur ^= 1;
  }

  // epilogue: D lane map row=(l>>4)*4+r, col=l&15
  const int orow0 = bm + wr * 64 + (lane >> 4) * 4;
  const int ocol0 = bn + wc * 64 + (lane & 15);
#pragma unroll
  for (int mf = 0; mf < 4; ++mf) {
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

template <bool BF16_OUT>
hipError_t launch(const uint16_t* A, const uint16_t* Bt, void* C, int M, int N,
                  int K, hipStream_t stream) {
  const unsigned tiles_m = lpx_ceil_div(M, BM), tiles_n = lpx_ceil_div(N, BN);
  const size_t lds = 4u * BM * ROW_BYTES;  // 2 dbuf x (A,B) x 10 KiB
 hipLaunchKernelGGL(( gemm_bf16_nt_kernel<BF16_OUT>), dim3(dim3(tiles_m * tiles_n)), dim3(dim3(THREADS)), lds,
                                  stream, A, Bt, C, M, N, K, tiles_m, tiles_n);
  return hipGetLastError();
}

}  // namespace

hipError_t gemm_bf16_nt(const uint16_t* A, const uint16_t* Bt, float* C, int M,
                        int N, int K, hipStream_t stream) {
  return launch<false>(A, Bt, C, M, N, K, stream);
}

hipError_t gemm_bf16_nt_bf16out(const uint16_t* A, const uint16_t* Bt,
                                uint16_t* C, int M, int N, int K,
                                hipStream_t stream) {
  return launch<true>(A, Bt, C, M, N, K, stream);
}

}  // namespace lpx
