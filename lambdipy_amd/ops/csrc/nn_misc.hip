// Misc NHWC NN kernels for the fp8 ResNet path.
//
// maxpool2d_nhwc removes the NCHW permute round-trip the torch fallback
// needs (two full-tensor transposes around F.max_pool2d); reads are
// C-contiguous f32x4 in NHWC so the kernel is a straight bandwidth pass.

#include "lpx_common.h"

namespace lpx {

namespace {

constexpr int THREADS = 256;

__global__ __launch_bounds__(THREADS) void maxpool_nhwc_kernel(
    const float* __restrict__ x, float* __restrict__ out, int N, int H, int W,
    int C, int k, int stride, int pad, int HO, int WO) {
  const int c4 = C >> 2;  // C % 4 == 0
  const long total = (long)N * HO * WO * c4;
  for (long i = (long)blockIdx.x * THREADS + threadIdx.x; i < total;
       i += (long)gridDim.x * THREADS) {
    const int cv = (int)(i % c4);
    long t = i / c4;
    const int wo = (int)(t % WO);
    t /= WO;
    const int ho = (int)(t % HO);
    const int n = (int)(t / HO);
    f32x4 best = {-3.4e38f, -3.4e38f, -3.4e38f, -3.4e38f};
    const int h0 = ho * stride - pad, w0 = wo * stride - pad;
#pragma unroll 3
    for (int r = 0; r < 8; ++r) {
      if (r >= k) break;
      const int hi = h0 + r;
      if (hi < 0 || hi >= H) continue;
#pragma unroll 3
      for (int s = 0; s < 8; ++s) {
        if (s >= k) break;
        const int wi = w0 + s;
        if (wi < 0 || wi >= W) continue;
        const f32x4 v = *reinterpret_cast<const f32x4*>(
            x + (((long)n * H + hi) * W + wi) * C + cv * 4);
        best.x = fmaxf(best.x, v.x);
        best.y = fmaxf(best.y, v.y);
        best.z = fmaxf(best.z, v.z);
        best.w = fmaxf(best.w, v.w);
      }
    }
    *reinterpret_cast<f32x4*>(out + (((long)n * HO + ho) * WO + wo) * C +
                              cv * 4) = best;
  }
}

}  // namespace

hipError_t maxpool2d_nhwc_f32(const float* x, float* out, int N, int H, int W,
                              int C, int k, int stride, int pad,
                              hipStream_t stream) {
  if (C % 4 != 0 || k > 8) return hipErrorInvalidValue;
  const int HO = (H + 2 * pad - k) / stride + 1;
  const int WO = (W + 2 * pad - k) / stride + 1;
  const long total = (long)N * HO * WO * (C / 4);
  unsigned blocks = (unsigned)((total + THREADS - 1) / THREADS);
  if (blocks > 65535u) blocks = 65535u;
  maxpool_nhwc_kernel<<<dim3(blocks ? blocks : 1), dim3(THREADS), 0, stream>>>(
      x, out, N, H, W, C, k, stride, pad, HO, WO);
  return hipGetLastError();
}

}  // namespace lpx
