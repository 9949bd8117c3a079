// Batched 1-D complex64 FFT for gfx950 — Stockham radix-2 in LDS.
//
// Design: one workgroup per batch element; the whole
// sequence (N <= 4096 complex64) lives in two LDS ping-pong buffers
// (2 x 32 KiB at N=4096, within the 160 KiB/CU budget). Stockham autosort
// needs no bit-reversal pass; log2(N) stages, each stage N/2 butterflies
// spread over 256 threads. Twiddles are a host-precomputed [N/2] complex
// table (exp(-2*pi*i*k/N)) read through L2/L3 (broadcast across blocks) —
// on-device sincos would turn this memory-bound kernel VALU-bound
// (guide Appendix B, trig-heavy ops).
//
// Larger N falls back to a global-memory Stockham (same recurrence, tensors
// ping-pong in HBM) — still bandwidth-bound, log2(N) passes.

#include "lpx_common.h"

namespace lpx {

namespace {

constexpr int THREADS = 256;

__device__ __forceinline__ f32x2 cmul(f32x2 a, f32x2 b) {
  return f32x2{a.x * b.x - a.y * b.y, a.x * b.y + a.y * b.x};
}

// Stockham in LDS, radix-4 stages (+ one radix-2 stage when log2(n) is odd).
// Half the stage count / barrier count of pure radix-2 (OTFFT-style
// recurrence; twiddles w1 = tw[p*s], w2 = tw[2*p*s], w3 = w1*w2).
template <bool INV>
__global__ __launch_bounds__(THREADS) void fft_lds_kernel(
    const float* __restrict__ xg, float* __restrict__ outg,
    const float* __restrict__ tw, int batch, int n, int log2n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  f32x2* buf0 = reinterpret_cast<f32x2*>(smem);
  f32x2* buf1 = buf0 + n;

  const int t = threadIdx.x;
  const int half = n >> 1;
  const int quarter = n >> 2;
  const f32x2* twc = reinterpret_cast<const f32x2*>(tw);

  for (int b = blockIdx.x; b < batch; b += gridDim.x) {
    // 16-B vectorized load (two complex64 per lane — G13)
    const f32x4* src4 = reinterpret_cast<const f32x4*>(xg) + (long)b * (n >> 1);
    f32x4* b04 = reinterpret_cast<f32x4*>(buf0);
    for (int i = t; i < (n >> 1); i += THREADS) b04[i] = src4[i];
    __syncthreads();

    f32x2* x = buf0;
    f32x2* y = buf1;
    int s = 1;
    int n_rem = n;
    while (n_rem >= 4) {
      const int m = n_rem >> 2;
      for (int u = t; u < quarter; u += THREADS) {
        const int p = u / s;
        const int q = u - p * s;
        const f32x2 a = x[q + s * p];
        const f32x2 bv = x[q + s * (p + m)];
        const f32x2 c = x[q + s * (p + 2 * m)];
        const f32x2 d = x[q + s * (p + 3 * m)];
        f32x2 w1 = twc[(long)p * s];
        f32x2 w2 = twc[(long)2 * p * s];
        if (INV) { w1.y = -w1.y; w2.y = -w2.y; }
        const f32x2 w3 = cmul(w1, w2);
        const f32x2 apc = {a.x + c.x, a.y + c.y};
        const f32x2 amc = {a.x - c.x, a.y - c.y};
        const f32x2 bpd = {bv.x + d.x, bv.y + d.y};
        // jbmd = j*(b-d) for forward, -j*(b-d) for inverse
        f32x2 jbmd = {-(bv.y - d.y), bv.x - d.x};
        if (INV) jbmd = f32x2{-jbmd.x, -jbmd.y};
        y[q + s * (4 * p + 0)] = f32x2{apc.x + bpd.x, apc.y + bpd.y};
        y[q + s * (4 * p + 1)] =
            cmul(f32x2{amc.x - jbmd.x, amc.y - jbmd.y}, w1);
        y[q + s * (4 * p + 2)] = cmul(f32x2{apc.x - bpd.x, apc.y - bpd.y}, w2);
        y[q + s * (4 * p + 3)] =
            cmul(f32x2{amc.x + jbmd.x, amc.y + jbmd.y}, w3);
      }
      __syncthreads();
      f32x2* tmp = x; x = y; y = tmp;
      s <<= 2;
      n_rem >>= 2;
    }
    if (n_rem == 2) {  // final radix-2 stage (odd log2n); m = 1
      for (int u = t; u < half; u += THREADS) {
        const int p = u / s;
        const int q = u - p * s;
        const f32x2 a = x[q + s * p];
        const f32x2 bv = x[q + s * (p + 1)];
        f32x2 w = twc[(long)p * s];
        if (INV) w.y = -w.y;
        y[q + s * (2 * p)] = f32x2{a.x + bv.x, a.y + bv.y};
        y[q + s * (2 * p + 1)] = cmul(f32x2{a.x - bv.x, a.y - bv.y}, w);
      }
      __syncthreads();
      f32x2* tmp = x; x = y; y = tmp;
    }

    f32x2* dst_lds = x;  // result lives in x after the final swap
    f32x4* dst4 = reinterpret_cast<f32x4*>(outg) + (long)b * (n >> 1);
    const f32x4* src_lds4 = reinterpret_cast<const f32x4*>(dst_lds);
    if (INV) {
      const float inv_n = 1.0f / (float)n;
      for (int i = t; i < (n >> 1); i += THREADS) {
        f32x4 v = src_lds4[i];
        dst4[i] = f32x4{v.x * inv_n, v.y * inv_n, v.z * inv_n, v.w * inv_n};
      }
    } else {
      for (int i = t; i < (n >> 1); i += THREADS) dst4[i] = src_lds4[i];
    }
    __syncthreads();
  }
}

// global-memory fallback for n > 4096: one stage per kernel launch
template <bool INV>
__global__ __launch_bounds__(THREADS) void fft_global_stage(
    const float* __restrict__ xg, float* __restrict__ yg,
    const float* __restrict__ tw, int batch, int n, int s, int m) {
  const int half = n >> 1;
  const long total = (long)batch * half;
  for (long idx = (long)blockIdx.x * THREADS + threadIdx.x; idx < total;
       idx += (long)gridDim.x * THREADS) {
    const int b = (int)(idx / half);
    const int u = (int)(idx - (long)b * half);
    const int p = u / s;
    const int q = u - p * s;
    const f32x2* x = reinterpret_cast<const f32x2*>(xg) + (long)b * n;
    f32x2* y = reinterpret_cast<f32x2*>(yg) + (long)b * n;
    const f32x2 a = x[q + s * p];
    const f32x2 bv = x[q + s * (p + m)];
    f32x2 w = reinterpret_cast<const f32x2*>(tw)[(long)p * s];
    if (INV) w.y = -w.y;
    y[q + s * (2 * p)] = f32x2{a.x + bv.x, a.y + bv.y};
    y[q + s * (2 * p + 1)] = cmul(f32x2{a.x - bv.x, a.y - bv.y}, w);
  }
}

__global__ __launch_bounds__(THREADS) void scale_kernel(float* __restrict__ d,
                                                        long total,
                                                        float scale) {
  for (long i = (long)blockIdx.x * THREADS + threadIdx.x; i < total;
       i += (long)gridDim.x * THREADS)
    d[i] *= scale;
}

}  // namespace

hipError_t fft_c64(const float* x, float* out, const float* twiddles,
                   int batch, int n, bool inverse, hipStream_t stream) {
  if (n <= 0 || (n & (n - 1)) != 0) return hipErrorInvalidValue;
  int log2n = 0;
  while ((1 << log2n) < n) ++log2n;
  if (n == 1) {
    return hipMemcpyAsync(out, x, (size_t)batch * 8, hipMemcpyDeviceToDevice,
                          stream);
  }
  if (n <= 4096) {
    const size_t lds = (size_t)2 * n * sizeof(f32x2);
    unsigned blocks = batch < 8192 ? (unsigned)batch : 8192u;
    if (inverse)
      fft_lds_kernel<true><<<dim3(blocks), dim3(THREADS), lds, stream>>>(
          x, out, twiddles, batch, n, log2n);
    else
      fft_lds_kernel<false><<<dim3(blocks), dim3(THREADS), lds, stream>>>(
          x, out, twiddles, batch, n, log2n);
    return hipGetLastError();
  }
  // global fallback: ping-pong between out and a scratch half... we use out
  // and x alternately; log2n stages. Requires log2n passes; final must land
  // in out. We bounce out <-> (float*)x is not allowed (x const). Caller
  // provides out only, so allocate scratch here (acceptable: n>4096 path is
  // rare and not cold-start critical).
  float* scratch = nullptr;
  hipError_t e = hipMallocAsync(&scratch, (size_t)batch * n * 8, stream);
  if (e != hipSuccess) return e;
  const float* src = x;
  float* dsts[2] = {(log2n % 2 == 1) ? out : scratch,
                    (log2n % 2 == 1) ? scratch : out};
  long halfw = (long)batch * (n >> 1);
  unsigned blocks = (unsigned)((halfw + THREADS - 1) / THREADS);
  if (blocks > 4096u) blocks = 4096u;
  int s = 1;
  for (int stage = 0; stage < log2n; ++stage) {
    const int m = n >> (stage + 1);
    float* dst = dsts[stage % 2];
    if (inverse)
      fft_global_stage<true><<<dim3(blocks), dim3(THREADS), 0, stream>>>(
          src, dst, twiddles, batch, n, s, m);
    else
      fft_global_stage<false><<<dim3(blocks), dim3(THREADS), 0, stream>>>(
          src, dst, twiddles, batch, n, s, m);
    src = dst;
    s <<= 1;
  }
  if (inverse) {
    long total = (long)batch * n * 2;
    scale_kernel<<<dim3(blocks), dim3(THREADS), 0, stream>>>(out, total,
                                                             1.0f / n);
  }
  hipFreeAsync(scratch, stream);
  return hipGetLastError();
}

}  // namespace lpx
