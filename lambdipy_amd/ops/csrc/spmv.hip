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

// --- x-tile-in-LDS (column-sorted rows, x too big to stay cache-hot) -----
// The seg8 kernel is gather-bound when x outgrows L2: every random 4-B
// gather pulls a 64-B sector from L2/MALL (16x amplification — 1M x 32
// measures 1.3 TB/s eff vs the 6.3 TB/s stream roofline). This kernel
// streams x through LDS in column tiles instead: one persistent block per
// CU owns a contiguous row chunk and keeps a register cursor per row;
// tiles are loaded cooperatively (coalesced f32x4), gathers hit LDS, and
// every idx/data byte is still read exactly once (sorted columns => each
// row's elements for tile t are one contiguous sub-range). Requires
// CSR-canonical sorted column indices per row (scipy guarantees this).
constexpr int XT_FLOATS = 32768;          // 128 KiB tile -> 1 block/CU
constexpr int XT_ROWS_PER_THREAD = 16;    // register cursors + sums

__global__ __launch_bounds__(THREADS, 1) void spmv_xtile(
    const int* __restrict__ indptr, const int* __restrict__ indices,
    const float* __restrict__ data, const float* __restrict__ x,
    float* __restrict__ y, int n_rows, int n_cols) {
  __shared__ __attribute__((aligned(16))) float xl[XT_FLOATS];
  const int t = threadIdx.x;
  // this block's contiguous row chunk; thread r-strided rows keep the
  // per-iteration row streams of a wave on adjacent 128-B blocks
  const int chunk = (n_rows + gridDim.x - 1) / gridDim.x;
  const int r0 = blockIdx.x * chunk;
  const int r1 = min(r0 + chunk, n_rows);

  int cur[XT_ROWS_PER_THREAD], hi[XT_ROWS_PER_THREAD];
  float sum[XT_ROWS_PER_THREAD];
#pragma unroll
  for (int i = 0; i < XT_ROWS_PER_THREAD; ++i) {
    const int row = r0 + t + i * THREADS;
    const bool ok = row < r1;
    cur[i] = ok ? indptr[row] : 0;
    hi[i] = ok ? indptr[row + 1] : 0;
    sum[i] = 0.f;
  }

  const int ntiles = (n_cols + XT_FLOATS - 1) / XT_FLOATS;
  for (int tile = 0; tile < ntiles; ++tile) {
    const int c0 = tile * XT_FLOATS;
    const int cend = min(c0 + XT_FLOATS, n_cols);
    {  // cooperative tile load, f32x4
      const int nt = cend - c0;
      const int nv = nt >> 2;
      const f32x4* src = reinterpret_cast<const f32x4*>(x + c0);
      f32x4* dst = reinterpret_cast<f32x4*>(xl);
      for (int j = t; j < nv; j += THREADS) dst[j] = src[j];
      for (int j = nv * 4 + t; j < nt; j += THREADS) xl[j] = x[c0 + j];
    }
    __syncthreads();
#pragma unroll
    for (int i = 0; i < XT_ROWS_PER_THREAD; ++i) {
      int j = cur[i];
      const int jh = hi[i];
      float s = sum[i];
      while (j < jh) {
        const int c = indices[j];
        if (c >= cend) break;
        s += data[j] * xl[c - c0];
        ++j;
      }
      sum[i] = s;
      cur[i] = j;
    }
    __syncthreads();
  }
#pragma unroll
  for (int i = 0; i < XT_ROWS_PER_THREAD; ++i) {
    const int row = r0 + t + i * THREADS;
    if (row < r1) y[row] = sum[i];
  }
}

}  // namespace

hipError_t spmv_csr_xtile_f32(const int* indptr, const int* indices,
                              const float* data, const float* x, float* y,
                              int n_rows, int n_cols, hipStream_t stream) {
  // grid sized so every block's row chunk fits its register cursors
  // (chunk = ceil(n_rows/blocks) <= XT_ROWS_PER_THREAD * THREADS)
  unsigned blocks = lpx_ceil_div(n_rows, XT_ROWS_PER_THREAD * THREADS);
  spmv_xtile<<<dim3(blocks ? blocks : 1), dim3(THREADS), 0, stream>>>(
      indptr, indices, data, x, y, n_rows, n_cols);
  return hipGetLastError();
}

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
