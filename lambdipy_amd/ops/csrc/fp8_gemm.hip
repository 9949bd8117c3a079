// fp8 (OCP e4m3) GEMM + quantize + NHWC im2col for gfx950 — config 5.
//
// ResNet-50 inference path (BASELINE.json config 5 'fp8 MFMA'):
//   - quantize_fp8: f32/bf16 -> e4m3 with a per-tensor scale (elementwise,
//     vectorized 4 B/lane out). gfx950 fp8 is OCP e4m3fn — NOT MI300X fnuz
//     (guide §4).
//   - im2col_fp8: NHWC activations -> [M = N*HO*WO, K = R*S*C] e4m3 patch
//     matrix, quantization fused (coalesced along C).
//   - gemm_fp8_nt: C[M,N] f32 = A[M,K] @ Bt[N,K]^T on
//     v_mfma_f32_16x16x32_fp8_fp8, same 128x128 LDS-tiled structure as the
//     bf16 GEMM (gemm_bf16.hip), BK=32, padded 48-B LDS rows. The f32
//     result is descaled by (scale_a * scale_b) in the epilogue, with
//     optional bias + relu fused (conv epilogue).
//
// Fragment map (v_mfma_f32_16x16x32_fp8_fp8): lane l holds 8 e4m3 bytes of
// A at [row = l&15][k = (l>>4)*8 + 0..7]; same map for Bt rows; D as bf16
// variant (col = l&15, row = (l>>4)*4 + reg).

#include "lpx_common.h"
#include "kernels.h"

namespace lpx {

namespace {

constexpr int THREADS = 256;

// ---- f32 -> e4m3 (OCP, saturating) --------------------------------------
__device__ __forceinline__ unsigned char f32_to_e4m3(float x) {
  // OCP e4m3fn: 1s4e3m, bias 7, max finite 448, no inf, nan = 0x7f/0xff.
  union { float f; uint32_t u; } c;
  c.f = x;
  const uint32_t sign = (c.u >> 24) & 0x80;
  float ax = fabsf(x);
  if (!(ax == ax)) return (unsigned char)(sign | 0x7f);  // nan
  if (ax > 448.f) ax = 448.f;                            // satfinite
  c.f = ax;
  int32_t e = (int32_t)((c.u >> 23) & 0xff) - 127;
  uint32_t m = c.u & 0x7fffff;
  uint32_t out;
  if (ax == 0.f) {
    out = 0;
  } else if (e < -6) {
    // subnormal: value = q * 2^-9, q in [0,8), round-to-nearest-even
    float scaled = ax * 512.f;  // 2^9
    uint32_t q = (uint32_t)scaled;
    float frac = scaled - (float)q;
    if (frac > 0.5f || (frac == 0.5f && (q & 1))) ++q;
    out = (q >= 8) ? (1u << 3) : q;  // q==8 rounds up into normal range
  } else {
    // normal: exponent field e+7 in [1,15], 3-bit mantissa, RNE
    uint32_t mant = m >> 20;           // top 3 bits
    uint32_t rest = m & 0xfffff;
    uint32_t half = 0x80000;
    if (rest > half || (rest == half && (mant & 1))) {
      ++mant;
      if (mant == 8) { mant = 0; ++e; }
    }
    if (e > 8) { e = 8; mant = 7; }    // saturate at 448 (e=8 -> field 15)
    out = ((uint32_t)(e + 7) << 3) | mant;
  }
  return (unsigned char)(sign | out);
}

__global__ __launch_bounds__(THREADS) void quantize_fp8_kernel(
    const float* __restrict__ x, unsigned char* __restrict__ out, long n,
    float inv_scale) {
  for (long i = (long)blockIdx.x * THREADS + threadIdx.x; i < n;
       i += (long)gridDim.x * THREADS)
    out[i] = f32_to_e4m3(x[i] * inv_scale);
}

// ---- NHWC im2col (fused quantize) ---------------------------------------
// x: [N, H, W, C] f32. out: [N*HO*WO, R*S*C] e4m3 rows (patch matrix).
__global__ __launch_bounds__(THREADS) void im2col_fp8_kernel(
    const float* __restrict__ x, unsigned char* __restrict__ out, int N,
    int H, int W, int C, int R, int S, int HO, int WO, int stride, int pad,
    float inv_scale) {
  // one thread per (m, 16-byte k-chunk): one index decomposition per chunk
  // with incremental (c, s, r) carries — ~16x fewer div chains than the
  // per-byte form (the stem's 1.6M x 147 patch matrix was div-bound).
  const long M = (long)N * HO * WO;
  const int K = R * S * C;
  const int KC = (K + 15) >> 4;  // chunks per row
  for (long idx = (long)blockIdx.x * THREADS + threadIdx.x; idx < M * KC;
       idx += (long)gridDim.x * THREADS) {
    const long m = idx / KC;
    int k = (int)(idx - m * KC) * 16;
    const int wo = (int)(m % WO);
    const int ho = (int)((m / WO) % HO);
    const int n = (int)(m / ((long)WO * HO));
    const int hbase = ho * stride - pad;
    const int wbase = wo * stride - pad;
    int c = k % C;
    int srs = k / C;
    int sf = srs % S;
    int rf = srs / S;
    unsigned char buf[16];
    const int kend = (k + 16 <= K) ? k + 16 : K;
#pragma unroll 4
    for (int j = 0; j < 16; ++j) {
      if (k + j < kend) {
        const int hi = hbase + rf;
        const int wi = wbase + sf;
        float v = 0.f;
        if (hi >= 0 && hi < H && wi >= 0 && wi < W)
          v = x[(((long)n * H + hi) * W + wi) * C + c];
        buf[j] = f32_to_e4m3(v * inv_scale);
        if (++c == C) { c = 0; if (++sf == S) { sf = 0; ++rf; } }
      } else {
        buf[j] = 0;
      }
    }
    unsigned char* dst = out + m * K + k;
    if (kend - k == 16 && (((long)(m * K + k)) & 15) == 0) {
      *reinterpret_cast<i32x4*>(dst) = *reinterpret_cast<const i32x4*>(buf);
    } else {  // K=147 rows start unaligned; byte stores stay coalesced-ish
      for (int j = 0; j < kend - k; ++j) dst[j] = buf[j];
    }
  }
}

// ---- fp8 GEMM ------------------------------------------------------------

constexpr int BM = 128, BN = 128, BK = 32;
constexpr int ROW_BYTES = BK + 16;  // 48-B padded LDS row (32 fp8 + pad)

typedef __attribute__((ext_vector_type(2))) int fp8frag;  // 8 e4m3 bytes

__device__ __forceinline__ void stage_fp8_nt(const unsigned char* __restrict__ src,
                                             int rows_total, int K, int row0,
                                             int k0, char* __restrict__ lds) {
  // 128 rows x 32 B per tile = 4 KiB; 256 threads x 1 chunk of 16 B.
  const int t = threadIdx.x;
  const int r = t >> 1;
  const int kb = (t & 1) * 16;
  const int grow = row0 + r;
  const int gk = k0 + kb;
  i32x4 v = {0, 0, 0, 0};
  if (grow < rows_total) {
    if (gk + 16 <= K) {
      v = *reinterpret_cast<const i32x4*>(src + (long)grow * K + gk);
    } else {
      unsigned char z[16] = {0};
      for (int j = 0; j < 16; ++j)
        if (gk + j < K) z[j] = src[(long)grow * K + gk + j];
      v = *reinterpret_cast<const i32x4*>(z);
    }
  }
  *reinterpret_cast<i32x4*>(lds + r * ROW_BYTES + kb) = v;
}

template <bool RELU, bool BIAS>
__global__ __launch_bounds__(THREADS) void gemm_fp8_nt_kernel(
    const unsigned char* __restrict__ A, const unsigned char* __restrict__ Bt,
    float* __restrict__ C, const float* __restrict__ bias,
    const float* __restrict__ resid, int M, int N,
    int K, float descale, unsigned tiles_m, unsigned tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* const sbase = smem;
  const int BUF = BM * ROW_BYTES;

  const unsigned nwg = tiles_m * tiles_n;
  const unsigned bid = lpx_xcd_swizzle(blockIdx.x, nwg);
  const int bm = (int)(bid / tiles_n) * BM;
  const int bn = (int)(bid % tiles_n) * BN;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = wid >> 1, wc = wid & 1;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int frag_row = lane & 15;
  const int frag_kb = lane >> 4;  // 8-byte k block (0..3)

  int cur = 0;
  stage_fp8_nt(A, M, K, bm, 0, sbase);
  stage_fp8_nt(Bt, N, K, bn, 0, sbase + 2 * BUF);
  __syncthreads();

  const int ksteps = (K + BK - 1) / BK;
  for (int ks = 0; ks < ksteps; ++ks) {
    char* const aCur = sbase + cur * BUF;
    char* const bCur = sbase + 2 * BUF + cur * BUF;
    if (ks + 1 < ksteps) {
      stage_fp8_nt(A, M, K, bm, (ks + 1) * BK, sbase + (cur ^ 1) * BUF);
      stage_fp8_nt(Bt, N, K, bn, (ks + 1) * BK, sbase + 2 * BUF + (cur ^ 1) * BUF);
    }
    fp8frag aF[4], bF[4];
#pragma unroll
    for (int mf = 0; mf < 4; ++mf) {
      const int r = wr * 64 + mf * 16 + frag_row;
      aF[mf] = *reinterpret_cast<const fp8frag*>(aCur + r * ROW_BYTES + frag_kb * 8);
    }
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      const int r = wc * 64 + nf * 16 + frag_row;
      bF[nf] = *reinterpret_cast<const fp8frag*>(bCur + r * ROW_BYTES + frag_kb * 8);
    }
#pragma unroll
    for (int mf = 0; mf < 4; ++mf)
#pragma unroll
      for (int nf = 0; nf < 4; ++nf)
        acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
            __builtin_bit_cast(long, aF[mf]), __builtin_bit_cast(long, bF[nf]),
            acc[mf][nf], 0, 0, 0);
    __syncthreads();
    cur ^= 1;
  }

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
          float v = acc[mf][nf][r] * descale;
          if constexpr (BIAS) v += bias[col];
          if (resid) v += resid[(long)row * N + col];  // fused residual add
          if constexpr (RELU) v = v > 0.f ? v : 0.f;
          C[(long)row * N + col] = v;
        }
      }
    }
  }
}

// ---- fp8 GEMM v2: 256x256/BK64, 8 waves, register-staged T14 pipeline ----
// Same structure as gemm_bf16_256.hip; fp8 rows are 64 B (padded to 80 B,
// conflict-free ds_read_b64 fragment reads). ~2x the MFMA density per LDS
// byte of the bf16 version.

constexpr int BM2 = 256, BN2 = 256, BK2 = 64;
constexpr int THREADS2 = 512;
constexpr int ROWB2 = BK2 + 16;                       // 80-B padded row
constexpr int CHUNKS2 = BM2 * BK2 / 16 / THREADS2;    // 16-B chunks/thread = 2

__device__ __forceinline__ void load_fp8_regs(const unsigned char* __restrict__ src,
                                              int rows_total, int K, int row0,
                                              int k0, i32x4 (&regs)[CHUNKS2]) {
  const int t = threadIdx.x;
#pragma unroll
  for (int i = 0; i < CHUNKS2; ++i) {
    const int c = t + i * THREADS2;
    const int r = c >> 2;
    const int cell = (c & 3) * 16;
    const int grow = row0 + r;
    const int gk = k0 + cell;
    if (grow < rows_total && gk + 16 <= K) {
      regs[i] = *reinterpret_cast<const i32x4*>(src + (long)grow * K + gk);
    } else {
      unsigned char z[16] = {0};
      for (int j = 0; j < 16; ++j)
        if (grow < rows_total && gk + j < K) z[j] = src[(long)grow * K + gk + j];
      regs[i] = *reinterpret_cast<const i32x4*>(z);
    }
  }
}

__device__ __forceinline__ void write_fp8_lds(const i32x4 (&regs)[CHUNKS2],
                                              char* __restrict__ lds) {
  const int t = threadIdx.x;
#pragma unroll
  for (int i = 0; i < CHUNKS2; ++i) {
    const int c = t + i * THREADS2;
    const int r = c >> 2;
    const int cell = (c & 3) * 16;
    *reinterpret_cast<i32x4*>(lds + r * ROWB2 + cell) = regs[i];
  }
}

template <bool RELU, bool BIAS>
__global__ __launch_bounds__(THREADS2, 2) void gemm_fp8_nt256_kernel(
    const unsigned char* __restrict__ A, const unsigned char* __restrict__ Bt,
    float* __restrict__ C, const float* __restrict__ bias,
    const float* __restrict__ resid, int M, int N,
    int K, float descale, unsigned tiles_m, unsigned tiles_n) {
  // LDS double-buffered by K-step parity: one barrier per step (see the
  // conv kernel's note); 80 KiB still fits 2 blocks/CU.
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int SLOT = 2 * BM2 * ROWB2;

  const unsigned nwg = tiles_m * tiles_n;
  const unsigned bid = lpx_xcd_swizzle(blockIdx.x, nwg);
  const int bm = (int)(bid / tiles_n) * BM2;
  const int bn = (int)(bid % tiles_n) * BN2;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = wid >> 2, wc = wid & 3;  // 2x4 wave grid; 128x64 per wave

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int frag_row = lane & 15;
  const int frag_kb = lane >> 4;

  i32x4 aRegs[CHUNKS2], bRegs[CHUNKS2];
  load_fp8_regs(A, M, K, bm, 0, aRegs);
  load_fp8_regs(Bt, N, K, bn, 0, bRegs);
  write_fp8_lds(aRegs, smem);
  write_fp8_lds(bRegs, smem + BM2 * ROWB2);
  __syncthreads();

  const int ksteps = (K + BK2 - 1) / BK2;
  for (int ks = 0; ks < ksteps; ++ks) {
    char* const lA = smem + (ks & 1) * SLOT;
    char* const lB = lA + BM2 * ROWB2;
    if (ks + 1 < ksteps) {
      load_fp8_regs(A, M, K, bm, (ks + 1) * BK2, aRegs);
      load_fp8_regs(Bt, N, K, bn, (ks + 1) * BK2, bRegs);
    }
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      const int off = kk * 32 + frag_kb * 8;  // byte offset of this k block
      fp8frag aF[8], bF[4];
#pragma unroll
      for (int mf = 0; mf < 8; ++mf) {
        const int r = wr * 128 + mf * 16 + frag_row;
        aF[mf] = *reinterpret_cast<const fp8frag*>(lA + r * ROWB2 + off);
      }
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        const int r = wc * 64 + nf * 16 + frag_row;
        bF[nf] = *reinterpret_cast<const fp8frag*>(lB + r * ROWB2 + off);
      }
#pragma unroll
      for (int mf = 0; mf < 8; ++mf)
#pragma unroll
        for (int nf = 0; nf < 4; ++nf)
          acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
              __builtin_bit_cast(long, aF[mf]), __builtin_bit_cast(long, bF[nf]),
              acc[mf][nf], 0, 0, 0);
    }
    if (ks + 1 < ksteps) {
      char* const nA = smem + ((ks + 1) & 1) * SLOT;
      write_fp8_lds(aRegs, nA);
      write_fp8_lds(bRegs, nA + BM2 * ROWB2);
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
          float v = acc[mf][nf][r] * descale;
          if constexpr (BIAS) v += bias[col];
          if (resid) v += resid[(long)row * N + col];  // fused residual add
          if constexpr (RELU) v = v > 0.f ? v : 0.f;
          C[(long)row * N + col] = v;
        }
      }
    }
  }
}

// ---- tall-skinny fp8 GEMM (N <= 64) on the scaled MFMA -------------------
// The 1x1 convs with tiny KO (ResNet layer1: N=64, M~400k, K=64..147) fall
// off every 256-wide tile; the 128^2 v1 kernel wastes half its tile and
// runs the slow 16x16x32 op. This kernel: 256x64 tile, 8 waves of 32x64
// (1x2 fragments of 32x32 on the scaled MFMA, unit scales), register-staged
// double buffer, 80-B padded LDS rows (b128-quarter conflict-free), 50 KiB
// LDS -> 3 blocks/CU. Handles arbitrary K and N <= 64 tails by zero-fill.

constexpr int BMT = 256, BNT = 64, BKT = 64;   // elements; 64 B rows
constexpr int ROWT = BKT + 16;                 // 80-B padded
constexpr int THREADST = 512;
constexpr int CHT = BMT * BKT / 16 / THREADST; // 2 chunks/thread (A)

typedef __attribute__((ext_vector_type(8))) int fp8ts_frag;

__device__ __forceinline__ void ts_stage_a(const unsigned char* __restrict__ A,
                                           int M, int K, int bm, int k0,
                                           i32x4 (&regs)[CHT]) {
  const int t = threadIdx.x;
#pragma unroll
  for (int i = 0; i < CHT; ++i) {
    const int c = t + i * THREADST;
    const int r = c >> 2, cell = (c & 3) * 16;
    const int grow = bm + r, gk = k0 + cell;
    i32x4 v = {0, 0, 0, 0};
    if (grow < M) {
      if (gk + 16 <= K) {
        v = *reinterpret_cast<const i32x4*>(A + (long)grow * K + gk);
      } else if (gk < K) {
        unsigned char z[16] = {0};
        for (int j = 0; j < 16 && gk + j < K; ++j)
          z[j] = A[(long)grow * K + gk + j];
        v = *reinterpret_cast<const i32x4*>(z);
      }
    }
    regs[i] = v;
  }
}

__device__ __forceinline__ i32x4 ts_stage_b(const unsigned char* __restrict__ Bt,
                                            int N, int K, int bn, int k0) {
  const int t = threadIdx.x;
  i32x4 v = {0, 0, 0, 0};
  if (t < 256) {
    const int r = t >> 2, cell = (t & 3) * 16;
    const int grow = bn + r, gk = k0 + cell;
    if (grow < N) {
      if (gk + 16 <= K) {
        v = *reinterpret_cast<const i32x4*>(Bt + (long)grow * K + gk);
      } else if (gk < K) {
        unsigned char z[16] = {0};
        for (int j = 0; j < 16 && gk + j < K; ++j)
          z[j] = Bt[(long)grow * K + gk + j];
        v = *reinterpret_cast<const i32x4*>(z);
      }
    }
  }
  return v;
}

template <bool RELU, bool BIAS, bool EARLY = false>
__global__ __launch_bounds__(THREADST, 3) void gemm_fp8_ts_kernel(
    const unsigned char* __restrict__ A, const unsigned char* __restrict__ Bt,
    float* __restrict__ C, const float* __restrict__ bias, int M, int N,
    int K, float descale, unsigned tiles_m, unsigned tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int ABUF = BMT * ROWT;       // 20 KiB
  const int BBUF = BNT * ROWT;       // 5 KiB
  char* const lA = smem;             // [2][256][80]
  char* const lB = smem + 2 * ABUF;  // [2][64][80]

  const unsigned nwg = tiles_m * tiles_n;
  const unsigned bid = lpx_xcd_swizzle(blockIdx.x, nwg);
  const int bm = (int)(bid / tiles_n) * BMT;
  const int bn = (int)(bid % tiles_n) * BNT;

  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;    // 8 waves, 32 rows each
  const int fr = lane & 31;
  const int fkb = lane >> 5;

  f32x16 acc[2];
#pragma unroll
  for (int j = 0; j < 2; ++j)
#pragma unroll
    for (int e = 0; e < 16; ++e) acc[j][e] = 0.f;

  i32x4 aR[CHT];
  i32x4 bR;
  ts_stage_a(A, M, K, bm, 0, aR);
  bR = ts_stage_b(Bt, N, K, bn, 0);
  {
    const int t = threadIdx.x;
#pragma unroll
    for (int i = 0; i < CHT; ++i) {
      const int c = t + i * THREADST;
      *reinterpret_cast<i32x4*>(lA + (c >> 2) * ROWT + (c & 3) * 16) = aR[i];
    }
    if (t < 256)
      *reinterpret_cast<i32x4*>(lB + (t >> 2) * ROWT + (t & 3) * 16) = bR;
  }
  __syncthreads();

  const int ksteps = (K + BKT - 1) / BKT;
  int cur = 0;
  for (int ks = 0; ks < ksteps; ++ks) {
    if (ks + 1 < ksteps) {
      ts_stage_a(A, M, K, bm, (ks + 1) * BKT, aR);
      bR = ts_stage_b(Bt, N, K, bn, (ks + 1) * BKT);
      // EARLY (pairwise-v9 schedule): the cur^1 slot was consumed before
      // the previous barrier, so commit NOW — aR/bR die before the MFMA
      // block instead of living through it, and the pre-barrier tail is
      // just the MFMAs; 6 waves/SIMD hide the exposed vmcnt wait.
      if constexpr (EARLY) {
        const int t = threadIdx.x;
        char* aN = lA + (cur ^ 1) * ABUF;
#pragma unroll
        for (int i = 0; i < CHT; ++i) {
          const int c = t + i * THREADST;
          *reinterpret_cast<i32x4*>(aN + (c >> 2) * ROWT + (c & 3) * 16) =
              aR[i];
        }
        if (t < 256)
          *reinterpret_cast<i32x4*>(lB + (cur ^ 1) * BBUF + (t >> 2) * ROWT +
                                    (t & 3) * 16) = bR;
      }
    }
    const char* aCur = lA + cur * ABUF;
    const char* bCur = lB + cur * BBUF;
    fp8ts_frag aF, bF[2];
    {
      const char* p = aCur + (w * 32 + fr) * ROWT + fkb * 32;
      const i32x4 lo = *reinterpret_cast<const i32x4*>(p);
      const i32x4 hi = *reinterpret_cast<const i32x4*>(p + 16);
      aF = fp8ts_frag{lo.x, lo.y, lo.z, lo.w, hi.x, hi.y, hi.z, hi.w};
    }
#pragma unroll
    for (int nf = 0; nf < 2; ++nf) {
      const char* p = bCur + (nf * 32 + fr) * ROWT + fkb * 32;
      const i32x4 lo = *reinterpret_cast<const i32x4*>(p);
      const i32x4 hi = *reinterpret_cast<const i32x4*>(p + 16);
      bF[nf] = fp8ts_frag{lo.x, lo.y, lo.z, lo.w, hi.x, hi.y, hi.z, hi.w};
    }
#pragma unroll
    for (int nf = 0; nf < 2; ++nf)
      acc[nf] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
          aF, bF[nf], acc[nf], 0, 0, 0, 127, 0, 127);
    // single barrier per step: the cur^1 slot's reads finished at the
    // PREVIOUS step's barrier, so writing it during this step's MFMAs is
    // safe (same argument as conv_fp8_impx)
    if (!EARLY && ks + 1 < ksteps) {
      const int t = threadIdx.x;
      char* aN = lA + (cur ^ 1) * ABUF;
#pragma unroll
      for (int i = 0; i < CHT; ++i) {
        const int c = t + i * THREADST;
        *reinterpret_cast<i32x4*>(aN + (c >> 2) * ROWT + (c & 3) * 16) = aR[i];
      }
      if (t < 256)
        *reinterpret_cast<i32x4*>(lB + (cur ^ 1) * BBUF + (t >> 2) * ROWT +
                                  (t & 3) * 16) = bR;
    }
    __syncthreads();
    cur ^= 1;
  }

  const int ocol_l = lane & 31;
  const int orow_base = 4 * (lane >> 5);
#pragma unroll
  for (int nf = 0; nf < 2; ++nf) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int row = bm + w * 32 + (r & 3) + 8 * (r >> 2) + orow_base;
      const int col = bn + nf * 32 + ocol_l;
      if (row < M && col < N) {
        float v = acc[nf][r] * descale;
        if constexpr (BIAS) v += bias[col];
        if constexpr (RELU) v = v > 0.f ? v : 0.f;
        C[(long)row * N + col] = v;
      }
    }
  }
}

// ---- implicit-GEMM fp8 conv (v2 structure) -------------------------------
// out[m = (n,ho,wo)][ko] = sum_{r,s,c} xq[n][ho*st+r-p][wo*st+s-p][c] * w[ko][r,s,c]
// The A staging gathers 16-B chunks of pre-quantized NHWC activations
// directly (C % 16 == 0 so a chunk never crosses an (r,s) cell; spatial
// padding and row tails are zero-filled) — no materialized patch matrix,
// which saves writing+reading R*S x the activation bytes of im2col.

struct ConvGeom {
  int N, H, W, C, R, S, HO, WO, stride, pad;
};

// Incremental gather cursor for the implicit-conv A staging: the m-part
// (n, ho, wo -> hbase/wbase/base address) is LOOP-INVARIANT per thread and
// the k-part (r, s, c0) advances by BK2 per K-step with cheap carries —
// replaces two integer divides per chunk per step (the kernel is only 34%
// MFMA-busy; address VALU was a main non-MFMA cost, cf. im2col's same fix).
struct ImpCursor {
  long addrN;        // (long)n * H * W * C
  int hbase, wbase;  // ho*stride - pad, wo*stride - pad
  int r, s, c0;      // tap decomposition of the current k
  int klin;          // current k (tail checks)
  bool mok;          // m < M
};

__device__ __forceinline__ void imp_init(ImpCursor (&cur)[CHUNKS2],
                                         const ConvGeom g, int M, int row0) {
  const int t = threadIdx.x;
#pragma unroll
  for (int i = 0; i < CHUNKS2; ++i) {
    const int c = t + i * THREADS2;
    const int rt = c >> 2;
    const int cell = (c & 3) * 16;
    const int m = row0 + rt;
    cur[i].mok = m < M;
    const int mm = cur[i].mok ? m : 0;
    const int wo = mm % g.WO;
    const int ho = (mm / g.WO) % g.HO;
    const int n = mm / (g.WO * g.HO);
    cur[i].hbase = ho * g.stride - g.pad;
    cur[i].wbase = wo * g.stride - g.pad;
    cur[i].addrN = (long)n * g.H * g.W * g.C;
    cur[i].klin = cell;
    cur[i].c0 = cell % g.C;
    const int rs = cell / g.C;
    cur[i].s = rs % g.S;
    cur[i].r = rs / g.S;
  }
}

__device__ __forceinline__ void imp_advance(ImpCursor (&cur)[CHUNKS2],
                                            const ConvGeom g) {
#pragma unroll
  for (int i = 0; i < CHUNKS2; ++i) {
    cur[i].klin += BK2;
    cur[i].c0 += BK2;
    while (cur[i].c0 >= g.C) {
      cur[i].c0 -= g.C;
      if (++cur[i].s == g.S) {
        cur[i].s = 0;
        ++cur[i].r;
      }
    }
  }
}

__device__ __forceinline__ void imp_gather(const unsigned char* __restrict__ xq,
                                           const ImpCursor (&cur)[CHUNKS2],
                                           const ConvGeom g, int K,
                                           i32x4 (&regs)[CHUNKS2]) {
#pragma unroll
  for (int i = 0; i < CHUNKS2; ++i) {
    i32x4 v = {0, 0, 0, 0};
    if (cur[i].mok && cur[i].klin + 16 <= K) {
      const int hi = cur[i].hbase + cur[i].r;
      const int wi = cur[i].wbase + cur[i].s;
      if (hi >= 0 && hi < g.H && wi >= 0 && wi < g.W)
        v = *reinterpret_cast<const i32x4*>(
            xq + cur[i].addrN + ((long)hi * g.W + wi) * g.C + cur[i].c0);
    }
    regs[i] = v;
  }
}

__device__ __forceinline__ void stage_imp(const unsigned char* __restrict__ xq,
                                          const ConvGeom g, int M, int K,
                                          int row0, int k0,
                                          i32x4 (&regs)[CHUNKS2]) {
  const int t = threadIdx.x;
#pragma unroll
  for (int i = 0; i < CHUNKS2; ++i) {
    const int c = t + i * THREADS2;
    const int rt = c >> 2;               // tile row
    const int cell = (c & 3) * 16;       // 16-B cell within BK
    const int m = row0 + rt;
    const int k = k0 + cell;
    i32x4 v = {0, 0, 0, 0};
    if (m < M && k + 16 <= K) {
      const int c0 = k % g.C;
      const int rs = k / g.C;
      const int s = rs % g.S;
      const int r = rs / g.S;
      const int wo = m % g.WO;
      const int ho = (m / g.WO) % g.HO;
      const int n = m / (g.WO * g.HO);
      const int hi = ho * g.stride + r - g.pad;
      const int wi = wo * g.stride + s - g.pad;
      if (hi >= 0 && hi < g.H && wi >= 0 && wi < g.W)
        v = *reinterpret_cast<const i32x4*>(
            xq + (((long)n * g.H + hi) * g.W + wi) * g.C + c0);
    }
    regs[i] = v;
  }
}

template <bool RELU, bool BIAS>
__global__ __launch_bounds__(THREADS2, 2) void conv_fp8_imp_kernel(
    const unsigned char* __restrict__ xq, const unsigned char* __restrict__ Wq,
    float* __restrict__ C, const float* __restrict__ bias, ConvGeom g, int M,
    int N, int K, float descale, unsigned tiles_m, unsigned tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* const lA = smem;
  char* const lB = smem + BM2 * ROWB2;

  const unsigned nwg = tiles_m * tiles_n;
  const unsigned bid = lpx_xcd_swizzle(blockIdx.x, nwg);
  const int bm = (int)(bid / tiles_n) * BM2;
  const int bn = (int)(bid % tiles_n) * BN2;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = wid >> 2, wc = wid & 3;

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int frag_row = lane & 15;
  const int frag_kb = lane >> 4;

  i32x4 aRegs[CHUNKS2], bRegs[CHUNKS2];
  stage_imp(xq, g, M, K, bm, 0, aRegs);
  load_fp8_regs(Wq, N, K, bn, 0, bRegs);
  write_fp8_lds(aRegs, lA);
  write_fp8_lds(bRegs, lB);
  __syncthreads();

  const int ksteps = (K + BK2 - 1) / BK2;
  for (int ks = 0; ks < ksteps; ++ks) {
    if (ks + 1 < ksteps) {
      stage_imp(xq, g, M, K, bm, (ks + 1) * BK2, aRegs);
      load_fp8_regs(Wq, N, K, bn, (ks + 1) * BK2, bRegs);
    }
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      const int off = kk * 32 + frag_kb * 8;
      fp8frag aF[8], bF[4];
#pragma unroll
      for (int mf = 0; mf < 8; ++mf) {
        const int r = wr * 128 + mf * 16 + frag_row;
        aF[mf] = *reinterpret_cast<const fp8frag*>(lA + r * ROWB2 + off);
      }
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        const int r = wc * 64 + nf * 16 + frag_row;
        bF[nf] = *reinterpret_cast<const fp8frag*>(lB + r * ROWB2 + off);
      }
#pragma unroll
      for (int mf = 0; mf < 8; ++mf)
#pragma unroll
        for (int nf = 0; nf < 4; ++nf)
          acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
              __builtin_bit_cast(long, aF[mf]), __builtin_bit_cast(long, bF[nf]),
              acc[mf][nf], 0, 0, 0);
    }
    __syncthreads();
    if (ks + 1 < ksteps) {
      write_fp8_lds(aRegs, lA);
      write_fp8_lds(bRegs, lB);
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
          float v = acc[mf][nf][r] * descale;
          if constexpr (BIAS) v += bias[col];
          if constexpr (RELU) v = v > 0.f ? v : 0.f;
          C[(long)row * N + col] = v;
        }
      }
    }
  }
}

// ---- implicit conv on the scaled MFMA (unit block scales) ----------------
// Same staging/padded-LDS geometry, but v_mfma_scale_f32_32x32x64_f8f6f4 at
// 2x the 16x16x32 rate: wave tile 128x64 = 4x2 fragments of 32x32, one
// MFMA per fragment pair per 64-B k-tile (the operand k-permutation is
// self-consistent between A and B). 80-B padded rows are conflict-free for
// the 16-lane b128 quarters (bank starts 20r mod 64 tile perfectly).

typedef __attribute__((ext_vector_type(8))) int fp8frag32;

__device__ __forceinline__ fp8frag32 frag32(const char* __restrict__ lds,
                                            int row, int kb) {
  const char* p = lds + row * ROWB2 + kb * 32;
  const i32x4 lo = *reinterpret_cast<const i32x4*>(p);
  const i32x4 hi = *reinterpret_cast<const i32x4*>(p + 16);
  return fp8frag32{lo.x, lo.y, lo.z, lo.w, hi.x, hi.y, hi.z, hi.w};
}

template <bool RELU, bool BIAS, bool OUT8 = false>
__global__ __launch_bounds__(THREADS2, 2) void conv_fp8_impx_kernel(
    const unsigned char* __restrict__ xq, const unsigned char* __restrict__ Wq,
    float* __restrict__ C, unsigned char* __restrict__ C8,
    const float* __restrict__ bias, ConvGeom g, int M,
    int N, int K, float descale, float out_inv, unsigned tiles_m,
    unsigned tiles_n) {
  // LDS double-buffered by K-step parity: the write of tile ks+1 goes to
  // the slot whose reads finished at the PREVIOUS step's barrier, so each
  // step needs ONE barrier (after the writes) instead of two — the
  // write-after-read stall between the MFMA block and the LDS fill
  // disappears.  NOTE: at 228 VGPRs this kernel runs 2 waves/SIMD
  // (register-limited — the 80-KiB LDS would have allowed 2 blocks/CU);
  // conv_fp8_impx2 below fixes that and is the default (LPX_CONV_V2=0
  // selects this one).
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int SLOT = 2 * BM2 * ROWB2;

  const unsigned nwg = tiles_m * tiles_n;
  const unsigned bid = lpx_xcd_swizzle(blockIdx.x, nwg);
  const int bm = (int)(bid / tiles_n) * BM2;
  const int bn = (int)(bid % tiles_n) * BN2;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = wid >> 2, wc = wid & 3;  // wave tile 128x64
  const int fr = lane & 31;
  const int fkb = lane >> 5;

  f32x16 acc[4][2];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int e = 0; e < 16; ++e) acc[i][j][e] = 0.f;

  i32x4 aRegs[CHUNKS2], bRegs[CHUNKS2];
  ImpCursor cur[CHUNKS2];
  imp_init(cur, g, M, bm);
  imp_gather(xq, cur, g, K, aRegs);
  load_fp8_regs(Wq, N, K, bn, 0, bRegs);
  write_fp8_lds(aRegs, smem);
  write_fp8_lds(bRegs, smem + BM2 * ROWB2);
  __syncthreads();

  const int ksteps = (K + BK2 - 1) / BK2;
  for (int ks = 0; ks < ksteps; ++ks) {
    char* const lA = smem + (ks & 1) * SLOT;
    char* const lB = lA + BM2 * ROWB2;
    if (ks + 1 < ksteps) {
      imp_advance(cur, g);
      imp_gather(xq, cur, g, K, aRegs);
      load_fp8_regs(Wq, N, K, bn, (ks + 1) * BK2, bRegs);
    }
    fp8frag32 aF[4], bF[2];
#pragma unroll
    for (int mf = 0; mf < 4; ++mf)
      aF[mf] = frag32(lA, wr * 128 + mf * 32 + fr, fkb);
#pragma unroll
    for (int nf = 0; nf < 2; ++nf)
      bF[nf] = frag32(lB, wc * 64 + nf * 32 + fr, fkb);
#pragma unroll
    for (int mf = 0; mf < 4; ++mf)
#pragma unroll
      for (int nf = 0; nf < 2; ++nf)
        acc[mf][nf] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
            aF[mf], bF[nf], acc[mf][nf], 0, 0, 0, 127, 0, 127);
    if (ks + 1 < ksteps) {
      char* const nA = smem + ((ks + 1) & 1) * SLOT;
      write_fp8_lds(aRegs, nA);
      write_fp8_lds(bRegs, nA + BM2 * ROWB2);
    }
    __syncthreads();
  }

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
        if (row < M && col < N) {
          float v = acc[mf][nf][r] * descale;
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
}

// ---- impx2: the conv at 4 waves/SIMD ------------------------------------
// conv_fp8_impx allocates 228 VGPRs (acc[4][2] f32x16 = 128 alone), so it
// runs 2 waves/SIMD (512-entry file, MI355X_MICROARCH 'Register files') —
// NOT the 2 blocks/CU its LDS would allow.  Same 256^2 tile with 1024
// threads (16 waves, 4x4; 64x64 per wave) quarters the accumulator file
// (acc[2][2] = 64) to fit 4 waves/SIMD — the same restructure that took
// the bf16-split pairwise from 153 to 206 TF.
__device__ __forceinline__ void imp_init2(ImpCursor& cu, const ConvGeom g,
                                          int M, int row0) {
  const int c = threadIdx.x;  // 1024 threads x 1 16-B chunk
  const int rt = c >> 2;
  const int cell = (c & 3) * 16;
  const int m = row0 + rt;
  cu.mok = m < M;
  const int mm = cu.mok ? m : 0;
  const int wo = mm % g.WO;
  const int ho = (mm / g.WO) % g.HO;
  const int n = mm / (g.WO * g.HO);
  cu.hbase = ho * g.stride - g.pad;
  cu.wbase = wo * g.stride - g.pad;
  cu.addrN = (long)n * g.H * g.W * g.C;
  cu.klin = cell;
  cu.c0 = cell % g.C;
  const int rs = cell / g.C;
  cu.s = rs % g.S;
  cu.r = rs / g.S;
}

__device__ __forceinline__ void imp_advance2(ImpCursor& cu, const ConvGeom g) {
  cu.klin += BK2;
  cu.c0 += BK2;
  while (cu.c0 >= g.C) {
    cu.c0 -= g.C;
    if (++cu.s == g.S) {
      cu.s = 0;
      ++cu.r;
    }
  }
}

__device__ __forceinline__ i32x4 imp_gather2(const unsigned char* __restrict__ xq,
                                             const ImpCursor& cu,
                                             const ConvGeom g, int K) {
  i32x4 v = {0, 0, 0, 0};
  if (cu.mok && cu.klin + 16 <= K) {
    const int hi = cu.hbase + cu.r;
    const int wi = cu.wbase + cu.s;
    if (hi >= 0 && hi < g.H && wi >= 0 && wi < g.W)
      v = *reinterpret_cast<const i32x4*>(
          xq + cu.addrN + ((long)hi * g.W + wi) * g.C + cu.c0);
  }
  return v;
}

__device__ __forceinline__ i32x4 load_fp8_regs2(
    const unsigned char* __restrict__ src, int rows_total, int K, int row0,
    int k0) {
  const int c = threadIdx.x;
  const int r = c >> 2;
  const int cell = (c & 3) * 16;
  const int grow = row0 + r;
  const int gk = k0 + cell;
  if (grow < rows_total && gk + 16 <= K)
    return *reinterpret_cast<const i32x4*>(src + (long)grow * K + gk);
  unsigned char z[16] = {0};
  for (int j = 0; j < 16; ++j)
    if (grow < rows_total && gk + j < K) z[j] = src[(long)grow * K + gk + j];
  return *reinterpret_cast<const i32x4*>(z);
}

__device__ __forceinline__ void write_fp8_lds2(const i32x4 v,
                                               char* __restrict__ lds) {
  const int c = threadIdx.x;
  *reinterpret_cast<i32x4*>(lds + (c >> 2) * ROWB2 + (c & 3) * 16) = v;
}

template <bool RELU, bool BIAS, bool OUT8 = false, bool EARLY = true>
__global__ __launch_bounds__(1024, 1) void conv_fp8_impx2_kernel(
    const unsigned char* __restrict__ xq, const unsigned char* __restrict__ Wq,
    float* __restrict__ C, unsigned char* __restrict__ C8,
    const float* __restrict__ bias, ConvGeom g, int M, int N, int K,
    float descale, float out_inv, unsigned tiles_m, unsigned tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int SLOT = 2 * BM2 * ROWB2;

  const unsigned nwg = tiles_m * tiles_n;
  const unsigned bid = lpx_xcd_swizzle(blockIdx.x, nwg);
  const int bm = (int)(bid / tiles_n) * BM2;
  const int bn = (int)(bid % tiles_n) * BN2;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = wid >> 2, wc = wid & 3;  // 4x4 waves; 64x64 per wave
  const int fr = lane & 31;
  const int fkb = lane >> 5;

  f32x16 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int e = 0; e < 16; ++e) acc[i][j][e] = 0.f;

  ImpCursor cu;
  imp_init2(cu, g, M, bm);
  i32x4 aReg = imp_gather2(xq, cu, g, K);
  i32x4 bReg = load_fp8_regs2(Wq, N, K, bn, 0);
  write_fp8_lds2(aReg, smem);
  write_fp8_lds2(bReg, smem + BM2 * ROWB2);
  __syncthreads();

  const int ksteps = (K + BK2 - 1) / BK2;
  for (int ks = 0; ks < ksteps; ++ks) {
    char* const lA = smem + (ks & 1) * SLOT;
    char* const lB = lA + BM2 * ROWB2;
    if (ks + 1 < ksteps) {
      imp_advance2(cu, g);
      aReg = imp_gather2(xq, cu, g, K);
      bReg = load_fp8_regs2(Wq, N, K, bn, (ks + 1) * BK2);
      if constexpr (EARLY) {  // pairwise-v9: free slot, shed the live range
        char* const nA = smem + ((ks + 1) & 1) * SLOT;
        write_fp8_lds2(aReg, nA);
        write_fp8_lds2(bReg, nA + BM2 * ROWB2);
      }
    }
    fp8frag32 aF[2], bF[2];
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
      aF[mf] = frag32(lA, wr * 64 + mf * 32 + fr, fkb);
#pragma unroll
    for (int nf = 0; nf < 2; ++nf)
      bF[nf] = frag32(lB, wc * 64 + nf * 32 + fr, fkb);
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
      for (int nf = 0; nf < 2; ++nf)
        acc[mf][nf] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
            aF[mf], bF[nf], acc[mf][nf], 0, 0, 0, 127, 0, 127);
    if (!EARLY && ks + 1 < ksteps) {
      char* const nA = smem + ((ks + 1) & 1) * SLOT;
      write_fp8_lds2(aReg, nA);
      write_fp8_lds2(bReg, nA + BM2 * ROWB2);
    }
    __syncthreads();
  }

  const int ocol_l = lane & 31;
  const int orow_base = 4 * (lane >> 5);
#pragma unroll
  for (int mf = 0; mf < 2; ++mf) {
#pragma unroll
    for (int nf = 0; nf < 2; ++nf) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int row = bm + wr * 64 + mf * 32 + (r & 3) + 8 * (r >> 2) +
                        orow_base;
        const int col = bn + wc * 64 + nf * 32 + ocol_l;
        if (row < M && col < N) {
          float v = acc[mf][nf][r] * descale;
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
}

}  // namespace

hipError_t conv_fp8_implicit(const unsigned char* xq, const unsigned char* Wq,
                             float* C, const float* bias, int Nb, int H, int W,
                             int Cin, int R, int S, int HO, int WO, int stride,
                             int pad, int KO, float descale, bool relu,
                             hipStream_t stream) {
  if (Cin % 16 != 0) return hipErrorInvalidValue;  // chunk alignment
  ConvGeom g{Nb, H, W, Cin, R, S, HO, WO, stride, pad};
  const int M = Nb * HO * WO;
  const int K = R * S * Cin;
  // 8-phase glds conv measured SLOWER than this register-staged kernel
  // (13.6k vs 15.0k img/s ResNet bs128): the gather-bound staging needs the
  // 2-blocks/CU occupancy more than the pipeline. Kept behind LPX_CONV_8PH.
  const char* c8e = getenv("LPX_CONV_8PH");
  if (c8e && c8e[0] == '1' && conv_fp8_8ph_supported(M, KO, K, Cin))
    return conv_fp8_implicit_8ph(xq, Wq, C, nullptr, bias, Nb, H, W, Cin, R,
                                 S, HO, WO, stride, pad, KO, descale, relu,
                                 1.f, stream);
  const unsigned tiles_m = lpx_ceil_div(M, BM2), tiles_n = lpx_ceil_div(KO, BN2);
  const size_t lds = (size_t)4 * BM2 * ROWB2;  // 2 LDS slots (see kernel)
  dim3 grid(tiles_m * tiles_n), block(THREADS2);
  // impx2 (1024 threads / 4 waves per SIMD, early-commit) is the DEFAULT:
  // +9-11% ResNet e2e within-probe over the 228-VGPR 2-waves/SIMD impx
  // (18.8k vs 17.3k img/s bs128).  LPX_CONV_V2=0 forces legacy impx,
  // '2' the commit-after-MFMA form (128 VGPRs + 2 spills).
  const char* cv2 = getenv("LPX_CONV_V2");
  const char cvm = cv2 ? cv2[0] : '1';
  if (cvm != '0' && bias) {
    dim3 blk2(1024);
    if (cvm == '1') {
      if (relu)
        conv_fp8_impx2_kernel<true, true, false, true><<<grid, blk2, lds,
                                                         stream>>>(
            xq, Wq, C, nullptr, bias, g, M, KO, K, descale, 1.f, tiles_m,
            tiles_n);
      else
        conv_fp8_impx2_kernel<false, true, false, true><<<grid, blk2, lds,
                                                          stream>>>(
            xq, Wq, C, nullptr, bias, g, M, KO, K, descale, 1.f, tiles_m,
            tiles_n);
    } else {
      if (relu)
        conv_fp8_impx2_kernel<true, true, false, false><<<grid, blk2, lds,
                                                          stream>>>(
            xq, Wq, C, nullptr, bias, g, M, KO, K, descale, 1.f, tiles_m,
            tiles_n);
      else
        conv_fp8_impx2_kernel<false, true, false, false><<<grid, blk2, lds,
                                                           stream>>>(
            xq, Wq, C, nullptr, bias, g, M, KO, K, descale, 1.f, tiles_m,
            tiles_n);
    }
    return hipGetLastError();
  }
  if (bias && relu)
    conv_fp8_impx_kernel<true, true><<<grid, block, lds, stream>>>(
        xq, Wq, C, nullptr, bias, g, M, KO, K, descale, 1.f, tiles_m, tiles_n);
  else if (bias)
    conv_fp8_impx_kernel<false, true><<<grid, block, lds, stream>>>(
        xq, Wq, C, nullptr, bias, g, M, KO, K, descale, 1.f, tiles_m, tiles_n);
  else if (relu)
    conv_fp8_impx_kernel<true, false><<<grid, block, lds, stream>>>(
        xq, Wq, C, nullptr, bias, g, M, KO, K, descale, 1.f, tiles_m, tiles_n);
  else
    conv_fp8_impx_kernel<false, false><<<grid, block, lds, stream>>>(
        xq, Wq, C, nullptr, bias, g, M, KO, K, descale, 1.f, tiles_m, tiles_n);
  return hipGetLastError();
}

// Fused epilogue quantization: out bytes = e4m3(conv_out * out_inv) — the
// next layer's pre-quantized input (skips a full f32 write + read + u8
// write per layer pair on the ResNet path).
hipError_t conv_fp8_implicit_q8(const unsigned char* xq,
                                const unsigned char* Wq, unsigned char* C8,
                                const float* bias, int Nb, int H, int W,
                                int Cin, int R, int S, int HO, int WO,
                                int stride, int pad, int KO, float descale,
                                bool relu, float out_inv, hipStream_t stream) {
  if (Cin % 16 != 0) return hipErrorInvalidValue;
  ConvGeom g{Nb, H, W, Cin, R, S, HO, WO, stride, pad};
  const int M = Nb * HO * WO;
  const int K = R * S * Cin;
  const char* c8e2 = getenv("LPX_CONV_8PH");
  if (c8e2 && c8e2[0] == '1' && conv_fp8_8ph_supported(M, KO, K, Cin))
    return conv_fp8_implicit_8ph(xq, Wq, nullptr, C8, bias, Nb, H, W, Cin, R,
                                 S, HO, WO, stride, pad, KO, descale, relu,
                                 out_inv, stream);
  const unsigned tiles_m = lpx_ceil_div(M, BM2), tiles_n = lpx_ceil_div(KO, BN2);
  const size_t lds = (size_t)4 * BM2 * ROWB2;  // 2 LDS slots (see kernel)
  dim3 grid(tiles_m * tiles_n), block(THREADS2);
  const char* cv2q = getenv("LPX_CONV_V2");
  const char cvmq = cv2q ? cv2q[0] : '1';  // impx2 default (see above)
  if (cvmq != '0' && bias) {
    dim3 blk2(1024);
    if (cvmq == '1') {
      if (relu)
        conv_fp8_impx2_kernel<true, true, true, true><<<grid, blk2, lds,
                                                        stream>>>(
            xq, Wq, nullptr, C8, bias, g, M, KO, K, descale, out_inv,
            tiles_m, tiles_n);
      else
        conv_fp8_impx2_kernel<false, true, true, true><<<grid, blk2, lds,
                                                         stream>>>(
            xq, Wq, nullptr, C8, bias, g, M, KO, K, descale, out_inv,
            tiles_m, tiles_n);
    } else {
      if (relu)
        conv_fp8_impx2_kernel<true, true, true, false><<<grid, blk2, lds,
                                                         stream>>>(
            xq, Wq, nullptr, C8, bias, g, M, KO, K, descale, out_inv,
            tiles_m, tiles_n);
      else
        conv_fp8_impx2_kernel<false, true, true, false><<<grid, blk2, lds,
                                                          stream>>>(
            xq, Wq, nullptr, C8, bias, g, M, KO, K, descale, out_inv,
            tiles_m, tiles_n);
    }
    return hipGetLastError();
  }
  if (bias && relu)
    conv_fp8_impx_kernel<true, true, true><<<grid, block, lds, stream>>>(
        xq, Wq, nullptr, C8, bias, g, M, KO, K, descale, out_inv, tiles_m,
        tiles_n);
  else if (bias)
    conv_fp8_impx_kernel<false, true, true><<<grid, block, lds, stream>>>(
        xq, Wq, nullptr, C8, bias, g, M, KO, K, descale, out_inv, tiles_m,
        tiles_n);
  else if (relu)
    conv_fp8_impx_kernel<true, false, true><<<grid, block, lds, stream>>>(
        xq, Wq, nullptr, C8, bias, g, M, KO, K, descale, out_inv, tiles_m,
        tiles_n);
  else
    conv_fp8_impx_kernel<false, false, true><<<grid, block, lds, stream>>>(
        xq, Wq, nullptr, C8, bias, g, M, KO, K, descale, out_inv, tiles_m,
        tiles_n);
  return hipGetLastError();
}

hipError_t quantize_fp8(const float* x, unsigned char* out, long n,
                        float inv_scale, hipStream_t stream) {
  unsigned blocks = (unsigned)((n + THREADS - 1) / THREADS);
  if (blocks > 2048u) blocks = 2048u;
  quantize_fp8_kernel<<<dim3(blocks ? blocks : 1), dim3(THREADS), 0, stream>>>(
      x, out, n, inv_scale);
  return hipGetLastError();
}

hipError_t im2col_fp8(const float* x, unsigned char* out, int N, int H, int W,
                      int C, int R, int S, int HO, int WO, int stride, int pad,
                      float inv_scale, hipStream_t stream) {
  const long total = (long)N * HO * WO * R * S * C;
  unsigned blocks = (unsigned)((total + THREADS - 1) / THREADS);
  if (blocks > 8192u) blocks = 8192u;
  im2col_fp8_kernel<<<dim3(blocks ? blocks : 1), dim3(THREADS), 0, stream>>>(
      x, out, N, H, W, C, R, S, HO, WO, stride, pad, inv_scale);
  return hipGetLastError();
}

hipError_t gemm_fp8_nt(const unsigned char* A, const unsigned char* Bt,
                       float* C, const float* bias, int M, int N, int K,
                       float descale, bool relu, hipStream_t stream) {
  return gemm_fp8_nt_res(A, Bt, C, bias, nullptr, M, N, K, descale, relu,
                         stream);
}

// resid != nullptr fuses `C += resid` (pre-relu) into the epilogue — the
// ResNet bottleneck's identity add (saves a full elementwise read+write
// pass over the block output).
hipError_t gemm_fp8_nt_res(const unsigned char* A, const unsigned char* Bt,
                           float* C, const float* bias, const float* resid,
                           int M, int N, int K, float descale, bool relu,
                           hipStream_t stream) {
  // LPX_FP8_ROUTE forces a tier for A/B: t=tall-skinny  b=nt256  n=nt
  // (and widens that tier's shape gate to take the forced shapes).
  const char* route = getenv("LPX_FP8_ROUTE");
  const char rc = route ? route[0] : 0;
  // scaled-MFMA path first: 2x the 16x16x32 instruction rate (unit scales)
  if (!rc && M >= 512 && N >= 512 && M % 256 == 0 && N % 256 == 0 &&
      K % 128 == 0 && K >= 128)
    return gemm_fp8_nt8phx_res(A, Bt, C, bias, resid, M, N, K, descale, relu,
                               stream);
  if (!rc && !resid && M >= 1024 && N >= 1024 &&
      gemm_fp8_8ph_supported(M, N, K))
    return gemm_fp8_nt8ph(A, Bt, C, bias, M, N, K, descale, relu, stream);
  // M-split ('s' forces; default at chip-filling grids): M%256!=0 blocks
  // the scaled-MFMA pipeline (glds staging has no row clamp) — run it on
  // the aligned M0 rows and recurse for the <=255-row tail (lands on nt).
  // Measured (ab_fp8_route.py): 815 vs 672 (nt256) vs 593 (nt) TF at
  // 25216x1024x1024 (~396 tiles); LOSES below ~hundred tiles (195 vs 263
  // at 4928x1024) — underfilled 8phx grids don't pay, so the default
  // requires >=256 256^2-tiles.
  const long t256 = (long)lpx_ceil_div(M, 256) * lpx_ceil_div(N, 256);
  if ((rc == 's' || (!rc && t256 >= 256)) && N >= 512 && N % 256 == 0 &&
      K % 128 == 0 && K >= 128 && M >= 768 && M % 256 != 0) {
    const int M0 = (M / 256) * 256;
    hipError_t err = gemm_fp8_nt8phx_res(A, Bt, C, bias, resid, M0, N, K,
                                         descale, relu, stream);
    if (err != hipSuccess) return err;
    return gemm_fp8_nt_res(A + (long)M0 * K, Bt, C + (long)M0 * N, bias,
                           resid ? resid + (long)M0 * N : nullptr, M - M0, N,
                           K, descale, relu, stream);
  }
  // tall-skinny: N<=256 measured faster than every wider-tile route for
  // the ResNet 1x1 shapes (324 vs 219 TF @50176x128x256, 486 vs 263
  // @12544x256x1024 — scripts/ab_fp8_route.py)
  if (!resid && (rc == 't' || !rc) && N <= 256 && M >= 2048) {
    // tall-skinny (ResNet small-KO 1x1 convs): 256x64 scaled-MFMA tile
    const unsigned tm = lpx_ceil_div(M, BMT), tn = lpx_ceil_div(N, BNT);
    const size_t tlds = (size_t)2 * (BMT + BNT) * ROWT;  // 50 KiB
    dim3 tg(tm * tn), tb(THREADST);
    const char* ee = getenv("LPX_FP8_EARLY");
    if (ee && ee[0] == '1') {  // pairwise-v9 early-commit A/B
      if (bias && relu)
        gemm_fp8_ts_kernel<true, true, true><<<tg, tb, tlds, stream>>>(
            A, Bt, C, bias, M, N, K, descale, tm, tn);
      else if (bias)
        gemm_fp8_ts_kernel<false, true, true><<<tg, tb, tlds, stream>>>(
            A, Bt, C, bias, M, N, K, descale, tm, tn);
      else if (relu)
        gemm_fp8_ts_kernel<true, false, true><<<tg, tb, tlds, stream>>>(
            A, Bt, C, bias, M, N, K, descale, tm, tn);
      else
        gemm_fp8_ts_kernel<false, false, true><<<tg, tb, tlds, stream>>>(
            A, Bt, C, bias, M, N, K, descale, tm, tn);
      return hipGetLastError();
    }
    if (bias && relu)
      gemm_fp8_ts_kernel<true, true><<<tg, tb, tlds, stream>>>(
          A, Bt, C, bias, M, N, K, descale, tm, tn);
    else if (bias)
      gemm_fp8_ts_kernel<false, true><<<tg, tb, tlds, stream>>>(
          A, Bt, C, bias, M, N, K, descale, tm, tn);
    else if (relu)
      gemm_fp8_ts_kernel<true, false><<<tg, tb, tlds, stream>>>(
          A, Bt, C, bias, M, N, K, descale, tm, tn);
    else
      gemm_fp8_ts_kernel<false, false><<<tg, tb, tlds, stream>>>(
          A, Bt, C, bias, M, N, K, descale, tm, tn);
    return hipGetLastError();
  }
  // nt256 only when its 256^2 grid fills the chip: at 26-104 tiles the
  // 128^2 nt kernel wins (136 vs 83 TF @3136x512, 316 vs 260 @3136x2048 —
  // ResNet's stage-4 shapes); at ~400 tiles nt256 wins (370 vs 343
  // @12608x2048).  LPX_FP8_ROUTE: b=force nt256, n=force nt.
  const bool big = rc == 'n' ? false
                 : rc == 'b' ? (M >= 512 && N >= 128 && K >= 64)
                             : (M >= 512 && N >= 512 && K >= 64 &&
                                t256 >= 256);
  if (big) {
    const unsigned tiles_m = lpx_ceil_div(M, BM2), tiles_n = lpx_ceil_div(N, BN2);
    const size_t lds = (size_t)4 * BM2 * ROWB2;  // 80 KiB (2 LDS slots)
    dim3 grid(tiles_m * tiles_n), block(THREADS2);
    if (bias && relu)
      gemm_fp8_nt256_kernel<true, true><<<grid, block, lds, stream>>>(
          A, Bt, C, bias, resid, M, N, K, descale, tiles_m, tiles_n);
    else if (bias)
      gemm_fp8_nt256_kernel<false, true><<<grid, block, lds, stream>>>(
          A, Bt, C, bias, resid, M, N, K, descale, tiles_m, tiles_n);
    else if (relu)
      gemm_fp8_nt256_kernel<true, false><<<grid, block, lds, stream>>>(
          A, Bt, C, bias, resid, M, N, K, descale, tiles_m, tiles_n);
    else
      gemm_fp8_nt256_kernel<false, false><<<grid, block, lds, stream>>>(
          A, Bt, C, bias, resid, M, N, K, descale, tiles_m, tiles_n);
    return hipGetLastError();
  }
  const unsigned tiles_m = lpx_ceil_div(M, BM), tiles_n = lpx_ceil_div(N, BN);
  const size_t lds = 4u * BM * ROW_BYTES;
  dim3 grid(tiles_m * tiles_n), block(THREADS);
  if (bias && relu)
    gemm_fp8_nt_kernel<true, true><<<grid, block, lds, stream>>>(
        A, Bt, C, bias, resid, M, N, K, descale, tiles_m, tiles_n);
  else if (bias)
    gemm_fp8_nt_kernel<false, true><<<grid, block, lds, stream>>>(
        A, Bt, C, bias, resid, M, N, K, descale, tiles_m, tiles_n);
  else if (relu)
    gemm_fp8_nt_kernel<true, false><<<grid, block, lds, stream>>>(
        A, Bt, C, bias, resid, M, N, K, descale, tiles_m, tiles_n);
  else
    gemm_fp8_nt_kernel<false, false><<<grid, block, lds, stream>>>(
        A, Bt, C, bias, resid, M, N, K, descale, tiles_m, tiles_n);
  return hipGetLastError();
}

}  // namespace lpx
