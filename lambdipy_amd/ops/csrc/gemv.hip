// Batched GEMV for gfx950 — memory-bound, one wave per output row.
//
// y[b,m] = sum_k A[b,m,k] * x[b,k]
//
// Design (guide Appendix B 'Reduction' + G13): the weight matrix is streamed
// once, so the kernel is HBM-bound; the lever is vectorized 16 B/lane loads
// (short8 for bf16, float4 for f32) and enough waves to saturate (grid-stride
// over rows, 4 waves per block). Target >= 2/3 of the 6.29 TB/s measured BW.

#include "lpx_common.h"

namespace lpx {

namespace {

constexpr int THREADS = 256;  // 4 waves per block

__global__ __launch_bounds__(THREADS) void gemv_bf16_kernel(
    const uint16_t* __restrict__ A, const uint16_t* __restrict__ x,
    float* __restrict__ y, int B, int M, long K) {
  const int lane = threadIdx.x & 63;
  const int wave = (blockIdx.x * (THREADS / 64)) + (threadIdx.x >> 6);
  const int total_rows = B * M;
  const int nwaves = gridDim.x * (THREADS / 64);

  for (int row = wave; row < total_rows; row += nwaves) {
    const int b = row / M;
    const uint16_t* arow = A + (long)row * K;
    const uint16_t* xrow = x + (long)b * K;
    float sum = 0.f;
    // vector main loop: 8 bf16 per lane per iteration -> 512 elems/wave
    long k = (long)lane * 8;
    const long kvec_end = K & ~7L;
    for (; k + 8 <= kvec_end; k += 64 * 8) {
      const bf16x8 av = *reinterpret_cast<const bf16x8*>(arow + k);
      const bf16x8 xv = *reinterpret_cast<const bf16x8*>(xrow + k);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        sum += lpx_bf16_to_f32(av[j]) * lpx_bf16_to_f32(xv[j]);
    }
    // tail
    for (long kt = kvec_end + lane; kt < K; kt += 64)
      sum += lpx_bf16_to_f32((short)arow[kt]) * lpx_bf16_to_f32((short)xrow[kt]);
    sum = lpx_wave_allreduce_sum(sum);
    if (lane == 0) y[row] = sum;
  }
}

__global__ __launch_bounds__(THREADS) void gemv_f32_kernel(
    const float* __restrict__ A, const float* __restrict__ x,
    float* __restrict__ y, int B, int M, long K) {
  const int lane = threadIdx.x & 63;
  const int wave = (blockIdx.x * (THREADS / 64)) + (threadIdx.x >> 6);
  const int total_rows = B * M;
  const int nwaves = gridDim.x * (THREADS / 64);

  for (int row = wave; row < total_rows; row += nwaves) {
    const int b = row / M;
    const float* arow = A + (long)row * K;
    const float* xrow = x + (long)b * K;
    float sum = 0.f;
    long k = (long)lane * 4;
    const long kvec_end = K & ~3L;
    for (; k + 4 <= kvec_end; k += 64 * 4) {
      const f32x4 av = *reinterpret_cast<const f32x4*>(arow + k);
      const f32x4 xv = *reinterpret_cast<const f32x4*>(xrow + k);
#pragma unroll
      for (int j = 0; j < 4; ++j) sum += av[j] * xv[j];
    }
    for (long kt = kvec_end + lane; kt < K; kt += 64)
      sum += arow[kt] * xrow[kt];
    sum = lpx_wave_allreduce_sum(sum);
    if (lane == 0) y[row] = sum;
  }
}

unsigned pick_grid(int total_rows) {
  // 4 waves per block; cap at ~2048 blocks and grid-stride (G11)
  unsigned want = lpx_ceil_div(total_rows, THREADS / 64);
  return want < 2048u ? (want ? want : 1u) : 2048u;
}

}  // namespace

hipError_t gemv_bf16(const uint16_t* A, const uint16_t* x, float* y, int B,
                     int M, int K, hipStream_t stream) {
  gemv_bf16_kernel<<<dim3(pick_grid(B * M)), dim3(THREADS), 0, stream>>>(
      A, x, y, B, M, (long)K);
  return hipGetLastError();
}

hipError_t gemv_f32(const float* A, const float* x, float* y, int B, int M,
                    int K, hipStream_t stream) {
  gemv_f32_kernel<<<dim3(pick_grid(B * M)), dim3(THREADS), 0, stream>>>(
      A, x, y, B, M, (long)K);
  return hipGetLastError();
}

}  // namespace lpx
