// Raw MFMA issue-rate probe: scaled 32x32x64 f8f6f4 vs plain 16x16x32 fp8.
// Register-resident operands, 8 independent accumulator chains per thread.
// Build (on a GPU box): hipcc --offload-arch=gfx950 -O3 -shared -fPIC
//   mfma_rate.hip -o mfma_rate.so
#include <hip/hip_runtime.h>

typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(8))) int i32x8v;

__global__ __launch_bounds__(512) void rate_scaled(float* out, int iters) {
  i32x8v a = {0, 0, 0, 0, 0, 0, 0, 0}, b = a;
  f32x16 acc[8];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int e = 0; e < 16; ++e) acc[i][e] = 0.f;
  for (int it = 0; it < iters; ++it) {
#pragma unroll
    for (int i = 0; i < 8; ++i)
      acc[i] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
          a, b, acc[i], 0, 0, 0, 127, 0, 127);
  }
  float s = 0.f;
#pragma unroll
  for (int i = 0; i < 8; ++i) s += acc[i][0];
  if (s != 0.f) out[0] = s;  // never taken; keeps the loop alive
}

__global__ __launch_bounds__(512) void rate_scaled_fp4(float* out, int iters) {
  i32x8v a = {0, 0, 0, 0, 0, 0, 0, 0}, b = a;
  f32x16 acc[8];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int e = 0; e < 16; ++e) acc[i][e] = 0.f;
  for (int it = 0; it < iters; ++it) {
#pragma unroll
    for (int i = 0; i < 8; ++i)
      acc[i] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
          a, b, acc[i], 4, 4, 0, 127, 0, 127);  // cbsz/blgp 4 = e2m1 fp4
  }
  float s = 0.f;
#pragma unroll
  for (int i = 0; i < 8; ++i) s += acc[i][0];
  if (s != 0.f) out[0] = s;
}

typedef __attribute__((ext_vector_type(4))) float f32x4v;

__global__ __launch_bounds__(512) void rate_scaled_16x128(float* out, int iters) {
  i32x8v a = {0, 0, 0, 0, 0, 0, 0, 0}, b = a;
  f32x4v acc[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};
  for (int it = 0; it < iters; ++it) {
#pragma unroll
    for (int i = 0; i < 8; ++i)
      acc[i] = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
          a, b, acc[i], 0, 0, 0, 127, 0, 127);
  }
  float s = 0.f;
#pragma unroll
  for (int i = 0; i < 8; ++i) s += acc[i][0];
  if (s != 0.f) out[0] = s;
}

__global__ __launch_bounds__(512) void rate_mixed84(float* out, int iters) {
  i32x8v a = {0, 0, 0, 0, 0, 0, 0, 0}, b = a;
  f32x16 acc[8];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int e = 0; e < 16; ++e) acc[i][e] = 0.f;
  for (int it = 0; it < iters; ++it) {
#pragma unroll
    for (int i = 0; i < 8; ++i)
      acc[i] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
          a, b, acc[i], 0, 4, 0, 127, 0, 127);  // A=e4m3, B=e2m1
  }
  float s = 0.f;
#pragma unroll
  for (int i = 0; i < 8; ++i) s += acc[i][0];
  if (s != 0.f) out[0] = s;
}

__global__ __launch_bounds__(512) void rate_plain(float* out, int iters) {
  long a = 0, b = 0;
  f32x4 acc[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};
  for (int it = 0; it < iters; ++it) {
#pragma unroll
    for (int i = 0; i < 8; ++i)
      acc[i] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(a, b, acc[i], 0, 0, 0);
  }
  float s = 0.f;
#pragma unroll
  for (int i = 0; i < 8; ++i) s += acc[i][0];
  if (s != 0.f) out[0] = s;
}

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

__global__ __launch_bounds__(512) void rate_bf16_16(float* out, int iters) {
  bf16x8 a = {0, 0, 0, 0, 0, 0, 0, 0}, b = a;
  f32x4v acc[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};
  for (int it = 0; it < iters; ++it) {
#pragma unroll
    for (int i = 0; i < 8; ++i)
      acc[i] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[i], 0, 0, 0);
  }
  float s = 0.f;
#pragma unroll
  for (int i = 0; i < 8; ++i) s += acc[i][0];
  if (s != 0.f) out[0] = s;
}

__global__ __launch_bounds__(512) void rate_bf16_32(float* out, int iters) {
  bf16x8 a = {0, 0, 0, 0, 0, 0, 0, 0}, b = a;
  f32x16 acc[8];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int e = 0; e < 16; ++e) acc[i][e] = 0.f;
  for (int it = 0; it < iters; ++it) {
#pragma unroll
    for (int i = 0; i < 8; ++i)
      acc[i] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc[i], 0, 0, 0);
  }
  float s = 0.f;
#pragma unroll
  for (int i = 0; i < 8; ++i) s += acc[i][0];
  if (s != 0.f) out[0] = s;
}

extern "C" int run_rate(int which, int blocks, int iters, float* ms_out) {
  float* d;
  if (hipMalloc(&d, 4) != hipSuccess) return 1;
  hipEvent_t e0, e1;
  hipEventCreate(&e0);
  hipEventCreate(&e1);
  auto launch = [&](int n) {
    if (which == 0) rate_scaled<<<blocks, 512>>>(d, n);
    else if (which == 1) rate_plain<<<blocks, 512>>>(d, n);
    else if (which == 2) rate_scaled_fp4<<<blocks, 512>>>(d, n);
    else if (which == 3) rate_scaled_16x128<<<blocks, 512>>>(d, n);
    else if (which == 4) rate_bf16_16<<<blocks, 512>>>(d, n);
    else if (which == 5) rate_bf16_32<<<blocks, 512>>>(d, n);
    else rate_mixed84<<<blocks, 512>>>(d, n);
  };
  launch(iters / 10 + 1);
  hipDeviceSynchronize();
  hipEventRecord(e0);
  launch(iters);
  hipEventRecord(e1);
  if (hipDeviceSynchronize() != hipSuccess) return 2;
  hipEventElapsedTime(ms_out, e0, e1);
  hipFree(d);
  return 0;
}
