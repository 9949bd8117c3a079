// CSR SpMV for gfx950 — adaptive rows-per-wave / wave-per-row.
//
// y = A @ x, A in CSR (int32 indptr/indices, f32 data).
//
// Design: memory-bound, irregular. Two variants chosen by
// mean row length on the host:
//   - short rows (mean nnz < 48): 8-lane segments, 8 rows per wave — keeps
//     all 64 lanes busy on short rows and coalesces across neighbor rows;
//   - long rows: one wave per row, f32x4-vectorized nnz sweep, shuffle reduce.
// x gathers are served by L2/L3 (x is <= a few MB for sklearn/scipy-sized
// problems; 256 MiB L3 absorbs the re-reads).

#include "lpx_common.h"

namespace lpx {

namespace {

constexpr int THREADS = 256;

// --- one wave per row (long rows) ---------------------------------------
__global__ __launch_bounds__(THREADS) void spmv_wave_row(
    const int* __restrict__ indptr, const int* __restrict__ indices,
    const float* __restrict__ data, const float* __restrict__ x,
    float* __restrict__ y, int n_rows) {
  const int lane = threadIdx.x & 63;
  const int wave = blockIdx.x * (THREADS / 64) + (threadIdx.x >> 6);
  const int nwaves = gridDim.x * (THREADS / 64);
  for (int row = wave; row < n_rows; row += nwaves) {
    const int lo = indptr[row], hi = indptr[row + 1];
    float sum = 0.f;
    if (((lo & 3) == 0) && (hi - lo) >= 1024) {
      const int vend = lo + ((hi - lo) & ~3);
      for (int j = lo + lane * 4; j < vend; j += 64 * 4) {
        const i32x4 idx = *reinterpret_cast<const i32x4*>(indices + j);
        const f32x4 d = *reinterpret_cast<const f32x4*>(data + j);
        sum += d.x * x[idx.x] + d.y * x[idx.y] + d.z * x[idx.z] + d.w * x[idx.w];
      }
      for (int j = vend + lane; j < hi; j += 64) sum += data[j] * x[indices[j]];
    } else {
      for (int j = lo + lane; j < hi; j += 64) sum += data[j] * x[indices[j]];
    }
    sum = lpx_wave_allreduce_sum(sum);
    if (lane == 0) y[row] = sum;
  }
}

// --- 8-lane segments, 8 rows per wave (short rows) ------------------------
// Rows whose nnz range is 16-B aligned and a multiple of 4 take the
// vectorized path: each slot loads int4 indices + float4 data (16 B/lane,
// G13) — 4x fewer load instructions on the streaming data/indices arrays.
__global__ __launch_bounds__(THREADS) void spmv_seg8(
    const int* __restrict__ indptr, const int* __restrict__ indices,
    const float* __restrict__ data, const float* __restrict__ x,
    float* __restrict__ y, int n_rows) {
  const int lane = threadIdx.x & 63;
  const int seg = lane >> 3;        // 0..7: which row of this wave's 8
  const int slot = lane & 7;        // 0..7: position within segment
  const int wave = blockIdx.x * (THREADS / 64) + (threadIdx.x >> 6);
  const int nwaves = gridDim.x * (THREADS / 64);
  for (int row0 = wave * 8; row0 < n_rows; row0 += nwaves * 8) {
    const int row = row0 + seg;
    float sum = 0.f;
    if (row < n_rows) {
      const int lo = indptr[row], hi = indptr[row + 1];
      if (((lo & 3) == 0) && (((hi - lo) & 3) == 0)) {
        for (int j = lo + slot * 4; j < hi; j += 32) {
          const i32x4 idx = *reinterpret_cast<const i32x4*>(indices + j);
          const f32x4 d = *reinterpret_cast<const f32x4*>(data + j);
          sum += d.x * x[idx.x] + d.y * x[idx.y] + d.z * x[idx.z] + d.w * x[idx.w];
        }
      } else {
        for (int j = lo + slot; j < hi; j += 8) sum += data[j] * x[indices[j]];
      }
    }
    // reduce within the 8-lane segment
#pragma unroll
    for (int off = 4; off > 0; off >>= 1) sum += __shfl_down(sum, off, 64);
    if (slot == 0 && row < n_rows) y[row] = sum;
  }
}

}  // namespace

hipError_t spmv_csr_f32(const int* indptr, const int* indices,
                        const float* data, const float* x, float* y,
                        int n_rows, long nnz, hipStream_t stream) {
  const double mean = n_rows ? (double)nnz / n_rows : 0.0;
  if (mean < 48.0) {
    unsigned waves = lpx_ceil_div(n_rows, 8);
    unsigned blocks = lpx_ceil_div(waves, THREADS / 64);
    if (blocks > 2048u) blocks = 2048u;
    spmv_seg8<<<dim3(blocks ? blocks : 1), dim3(THREADS), 0, stream>>>(
        indptr, indices, data, x, y, n_rows);
  } else {
    unsigned blocks = lpx_ceil_div(n_rows, THREADS / 64);
    if (blocks > 2048u) blocks = 2048u;
    spmv_wave_row<<<dim3(blocks ? blocks : 1), dim3(THREADS), 0, stream>>>(
        indptr, indices, data, x, y, n_rows);
  }
  return hipGetLastError();
}

}  // namespace lpx
