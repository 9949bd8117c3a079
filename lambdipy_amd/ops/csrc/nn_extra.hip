// Utility kernels for the torch-free serving runtime (liblpx_nn).
//
// Everything the lite handlers did with torch ops before: device-side
// random generation (weights, batches, synthetic CSR), reductions (amax,
// checksum), NHWC global average pool, elementwise add+relu, per-row
// argmax, f32->bf16 conversion, FFT twiddle generation. All bandwidth-bound
// passes: 64-wide waves, grid-stride loops, vectorized where layout allows.
// (SURVEY.md §2.2 N11/N14 — cold start must not pay numpy/torch imports for
// any of this.)

#include "lpx_common.h"
#include "kernels.h"

namespace lpx {

namespace {

constexpr int THREADS = 256;

// xorshift128 step shared by the generators
__device__ __forceinline__ unsigned xs_step(unsigned& s0, unsigned& s1) {
  unsigned x = s0;
  s0 = s1;
  x ^= x << 23;
  s1 = x ^ s0 ^ (x >> 17) ^ (s0 >> 26);
  return s1 + s0;
}

__device__ __forceinline__ void xs_seed(long tid, unsigned seed, unsigned& s0,
                                        unsigned& s1) {
  s0 = 0x9E3779B9u ^ (unsigned)tid ^ seed;
  s1 = 0x85EBCA6Bu + (unsigned)(tid * 2654435761u);
  for (int k = 0; k < 3; ++k) (void)xs_step(s0, s1);
}

// standard normal pair via Box-Muller
__device__ __forceinline__ f32x2 xs_randn2(unsigned& s0, unsigned& s1) {
  const unsigned u1 = xs_step(s0, s1);
  const unsigned u2 = xs_step(s0, s1);
  const float f1 = ((u1 >> 8) + 1.0f) * (1.0f / 16777217.0f);  // (0,1]
  const float f2 = (u2 >> 8) * (1.0f / 16777216.0f);           // [0,1)
  const float r = sqrtf(-2.0f * logf(f1));
  float sn, cs;
  __sincosf(6.2831853f * f2, &sn, &cs);
  return f32x2{r * cs, r * sn};
}

__global__ __launch_bounds__(THREADS) void randn_kernel(
    float* __restrict__ out, long n, unsigned seed, float scale) {
  const long tid = (long)blockIdx.x * THREADS + threadIdx.x;
  unsigned s0, s1;
  xs_seed(tid, seed, s0, s1);
  for (long i = tid * 2; i < n; i += (long)gridDim.x * THREADS * 2) {
    const f32x2 v = xs_randn2(s0, s1);
    out[i] = v.x * scale;
    if (i + 1 < n) out[i + 1] = v.y * scale;
  }
}

// interleaved complex64: re ~ N(0,1), im = 0 (the fft handler's input)
__global__ __launch_bounds__(THREADS) void randn_c64_kernel(
    float* __restrict__ out, long n_complex, unsigned seed) {
  const long tid = (long)blockIdx.x * THREADS + threadIdx.x;
  unsigned s0, s1;
  xs_seed(tid, seed, s0, s1);
  for (long i = tid * 2; i < n_complex; i += (long)gridDim.x * THREADS * 2) {
    const f32x2 v = xs_randn2(s0, s1);
    *reinterpret_cast<f32x2*>(out + 2 * i) = f32x2{v.x, 0.f};
    if (i + 1 < n_complex)
      *reinterpret_cast<f32x2*>(out + 2 * (i + 1)) = f32x2{v.y, 0.f};
  }
}

// abs-max via atomicMax on the bit pattern (valid for non-negative floats);
// caller zeroes out_bits first. One atomic per wave.
__global__ __launch_bounds__(THREADS) void amax_kernel(
    const float* __restrict__ x, unsigned* __restrict__ out_bits, long n) {
  float m = 0.f;
  for (long i = (long)blockIdx.x * THREADS + threadIdx.x; i < n;
       i += (long)gridDim.x * THREADS)
    m = fmaxf(m, fabsf(x[i]));
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    m = fmaxf(m, __shfl_down(m, off, 64));
  if ((threadIdx.x & 63) == 0)
    atomicMax(out_bits, __float_as_uint(m));
}

__global__ __launch_bounds__(THREADS) void checksum_f32_kernel(
    const float* __restrict__ x, double* __restrict__ out, long n) {
  double local = 0.0;
  for (long i = (long)blockIdx.x * THREADS + threadIdx.x; i < n;
       i += (long)gridDim.x * THREADS)
    local += (double)x[i];
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    local += __shfl_down(local, off, 64);
  if ((threadIdx.x & 63) == 0) atomicAdd(out, local);
}

// NHWC global average pool: x [N, HW, C] -> out [N, C]. Threads stride over
// (n, c4) pairs; the HW loop reads C-contiguous f32x4 (coalesced).
__global__ __launch_bounds__(THREADS) void gap_nhwc_kernel(
    const float* __restrict__ x, float* __restrict__ out, int N, int HW,
    int C) {
  const int c4 = C >> 2;
  const long total = (long)N * c4;
  const float inv = 1.0f / (float)HW;
  for (long i = (long)blockIdx.x * THREADS + threadIdx.x; i < total;
       i += (long)gridDim.x * THREADS) {
    const int cv = (int)(i % c4);
    const int n = (int)(i / c4);
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    const float* base = x + ((long)n * HW) * C + cv * 4;
    for (int p = 0; p < HW; ++p) {
      const f32x4 v = *reinterpret_cast<const f32x4*>(base + (long)p * C);
      acc.x += v.x; acc.y += v.y; acc.z += v.z; acc.w += v.w;
    }
    acc.x *= inv; acc.y *= inv; acc.z *= inv; acc.w *= inv;
    *reinterpret_cast<f32x4*>(out + (long)n * C + cv * 4) = acc;
  }
}

__global__ __launch_bounds__(THREADS) void add_relu_kernel(
    const float* __restrict__ a, const float* __restrict__ b,
    float* __restrict__ out, long n) {
  const long n4 = n >> 2;
  for (long i = (long)blockIdx.x * THREADS + threadIdx.x; i < n4;
       i += (long)gridDim.x * THREADS) {
    const f32x4 va = reinterpret_cast<const f32x4*>(a)[i];
    const f32x4 vb = reinterpret_cast<const f32x4*>(b)[i];
    f32x4 v;
    v.x = fmaxf(va.x + vb.x, 0.f);
    v.y = fmaxf(va.y + vb.y, 0.f);
    v.z = fmaxf(va.z + vb.z, 0.f);
    v.w = fmaxf(va.w + vb.w, 0.f);
    reinterpret_cast<f32x4*>(out)[i] = v;
  }
  // tail
  for (long i = n4 * 4 + (long)blockIdx.x * THREADS + threadIdx.x; i < n;
       i += (long)gridDim.x * THREADS)
    out[i] = fmaxf(a[i] + b[i], 0.f);
}

__global__ __launch_bounds__(THREADS) void argmax_rows_kernel(
    const float* __restrict__ x, int* __restrict__ out, long rows, int cols) {
  for (long r = (long)blockIdx.x * THREADS + threadIdx.x; r < rows;
       r += (long)gridDim.x * THREADS) {
    const float* p = x + r * cols;
    int best = 0;
    float bv = p[0];
    for (int c = 1; c < cols; ++c) {
      const float v = p[c];
      if (v > bv) { bv = v; best = c; }
    }
    out[r] = best;
  }
}

__global__ __launch_bounds__(THREADS) void f32_to_bf16_kernel(
    const float* __restrict__ x, uint16_t* __restrict__ out, long n) {
  for (long i = (long)blockIdx.x * THREADS + threadIdx.x; i < n;
       i += (long)gridDim.x * THREADS)
    out[i] = (uint16_t)lpx_f32_to_bf16(x[i]);
}

// FFT twiddles: out[k] = (cos, sin)(-2*pi*k/n), k < n/2 — computed in f64
// on device to match the host (torch/numpy) twiddle tables bit-for-bit
// within f32 rounding.
__global__ __launch_bounds__(THREADS) void twiddles_kernel(
    float* __restrict__ out, int n) {
  const int half = n >> 1;
  for (int k = blockIdx.x * THREADS + threadIdx.x; k < half;
       k += gridDim.x * THREADS) {
    const double ang = -2.0 * 3.14159265358979323846 * (double)k / (double)n;
    out[2 * k] = (float)cos(ang);
    out[2 * k + 1] = (float)sin(ang);
  }
}

// Synthetic fixed-nnz CSR: indptr[i] = i*nnz; per row, nnz random columns
// sorted ascending (insertion sort in registers, nnz <= 64); data ~ N(0,1).
__global__ __launch_bounds__(THREADS) void csrgen_kernel(
    int* __restrict__ indptr, int* __restrict__ indices,
    float* __restrict__ data, int rows, int nnz, int ncols, unsigned seed) {
  constexpr int NNZ_MAX = 64;
  const long tid = (long)blockIdx.x * THREADS + threadIdx.x;
  for (long r = tid; r < rows; r += (long)gridDim.x * THREADS) {
    unsigned s0, s1;
    xs_seed(r, seed, s0, s1);
    int cols[NNZ_MAX];
    for (int j = 0; j < nnz; ++j) {
      const int c = (int)(xs_step(s0, s1) % (unsigned)ncols);
      // insertion sort as we go
      int k = j;
      while (k > 0 && cols[k - 1] > c) { cols[k] = cols[k - 1]; --k; }
      cols[k] = c;
    }
    int* ip = indices + r * nnz;
    float* dp = data + r * nnz;
    for (int j = 0; j < nnz; ++j) {
      ip[j] = cols[j];
      const f32x2 v = xs_randn2(s0, s1);
      dp[j] = v.x;
    }
    indptr[r] = (int)(r * nnz);
    if (r == rows - 1) indptr[rows] = rows * nnz;
  }
}

unsigned grid_for(long n, int per_thread = 1) {
  long blocks = (n / per_thread + THREADS - 1) / THREADS;
  if (blocks > 8192) blocks = 8192;
  return (unsigned)(blocks > 0 ? blocks : 1);
}

}  // namespace

hipError_t randn_f32(float* out, long n, unsigned seed, float scale,
                     hipStream_t stream) {
  randn_kernel<<<dim3(grid_for(n, 2)), dim3(THREADS), 0, stream>>>(out, n,
                                                                   seed, scale);
  return hipGetLastError();
}

hipError_t randn_c64(float* out, long n_complex, unsigned seed,
                     hipStream_t stream) {
  randn_c64_kernel<<<dim3(grid_for(n_complex, 2)), dim3(THREADS), 0, stream>>>(
      out, n_complex, seed);
  return hipGetLastError();
}

hipError_t amax_f32(const float* x, unsigned* out_bits, long n,
                    hipStream_t stream) {
  amax_kernel<<<dim3(grid_for(n, 8)), dim3(THREADS), 0, stream>>>(x, out_bits,
                                                                  n);
  return hipGetLastError();
}

hipError_t checksum_f32(const float* x, double* out, long n,
                        hipStream_t stream) {
  checksum_f32_kernel<<<dim3(grid_for(n, 8)), dim3(THREADS), 0, stream>>>(
      x, out, n);
  return hipGetLastError();
}

hipError_t gap_nhwc_f32(const float* x, float* out, int N, int HW, int C,
                        hipStream_t stream) {
  if (C % 4 != 0) return hipErrorInvalidValue;
  gap_nhwc_kernel<<<dim3(grid_for((long)N * (C / 4))), dim3(THREADS), 0,
                    stream>>>(x, out, N, HW, C);
  return hipGetLastError();
}

hipError_t add_relu_f32(const float* a, const float* b, float* out, long n,
                        hipStream_t stream) {
  add_relu_kernel<<<dim3(grid_for(n, 4)), dim3(THREADS), 0, stream>>>(a, b,
                                                                      out, n);
  return hipGetLastError();
}

hipError_t argmax_rows_f32(const float* x, int* out, long rows, int cols,
                           hipStream_t stream) {
  argmax_rows_kernel<<<dim3(grid_for(rows)), dim3(THREADS), 0, stream>>>(
      x, out, rows, cols);
  return hipGetLastError();
}

hipError_t f32_to_bf16(const float* x, uint16_t* out, long n,
                       hipStream_t stream) {
  f32_to_bf16_kernel<<<dim3(grid_for(n)), dim3(THREADS), 0, stream>>>(x, out,
                                                                      n);
  return hipGetLastError();
}

hipError_t fft_twiddles(float* out, int n, hipStream_t stream) {
  twiddles_kernel<<<dim3(grid_for(n / 2)), dim3(THREADS), 0, stream>>>(out, n);
  return hipGetLastError();
}

hipError_t csrgen_fixed(int* indptr, int* indices, float* data, int rows,
                        int nnz, int ncols, unsigned seed,
                        hipStream_t stream) {
  if (nnz > 64 || nnz < 1) return hipErrorInvalidValue;
  csrgen_kernel<<<dim3(grid_for(rows)), dim3(THREADS), 0, stream>>>(
      indptr, indices, data, rows, nnz, ncols, seed);
  return hipGetLastError();
}

}  // namespace lpx
