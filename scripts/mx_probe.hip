// MX-scaled fp8 MFMA layout probe (gfx950): one v_mfma_scale_f32_32x32x64
// with assumed fragment map A[row=l&31][k=(l>>5)*32+0..31] (32 B/lane),
// per-lane E8M0 scales. Verifies against host reference via unit scales
// (127 -> 2^0) and a non-trivial scale pattern.
#include <hip/hip_runtime.h>
#include <stdint.h>

typedef __attribute__((ext_vector_type(8))) int i32x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;

extern "C" __global__ void mx_probe_kernel(const uint8_t* __restrict__ A,
                                           const uint8_t* __restrict__ B,
                                           const uint8_t* __restrict__ scaleA,
                                           const uint8_t* __restrict__ scaleB,
                                           float* __restrict__ D) {
  // A: [32][64] fp8 row-major; B: [64][32] -> we pass Bt[32][64] (col j holds
  // B[k][j] as Bt[j][k]); scaleA: [32][2] per (row, kblock); scaleB: [32][2]
  // per (col, kblock). One wave.
  const int l = threadIdx.x;
  const int rc = l & 31;          // row of A / col of B
  const int kb = l >> 5;          // k-block (0..1), 32 elems each
  i32x8 a = *reinterpret_cast<const i32x8*>(A + rc * 64 + kb * 32);
  i32x8 b = *reinterpret_cast<const i32x8*>(B + rc * 64 + kb * 32);
  f32x16 c;
#pragma unroll
  for (int i = 0; i < 16; ++i) c[i] = 0.f;
  const int sa = scaleA[rc * 2 + kb];
  const int sb = scaleB[rc * 2 + kb];
  c = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(a, b, c, 0, 0, 0, sa, 0, sb);
  // D map (32x32): col = l&31, row = (reg&3) + 8*(reg>>2) + 4*(l>>5)
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * (l >> 5);
    const int col = l & 31;
    D[row * 32 + col] = c[r];
  }
}

extern "C" int run_mx_probe(const uint8_t* hA, const uint8_t* hB,
                            const uint8_t* hSA, const uint8_t* hSB,
                            float* hD) {
  uint8_t *dA, *dB, *dSA, *dSB;
  float* dD;
  (void)hipMalloc(&dA, 32 * 64);
  (void)hipMalloc(&dB, 32 * 64);
  (void)hipMalloc(&dSA, 64);
  (void)hipMalloc(&dSB, 64);
  (void)hipMalloc(&dD, 32 * 32 * 4);
  (void)hipMemcpy(dA, hA, 32 * 64, hipMemcpyHostToDevice);
  (void)hipMemcpy(dB, hB, 32 * 64, hipMemcpyHostToDevice);
  (void)hipMemcpy(dSA, hSA, 64, hipMemcpyHostToDevice);
  (void)hipMemcpy(dSB, hSB, 64, hipMemcpyHostToDevice);
  hipLaunchKernelGGL(mx_probe_kernel, dim3(1), dim3(64), 0, 0, dA, dB, dSA, dSB, dD);
  hipError_t e = hipDeviceSynchronize();
  (void)hipMemcpy(hD, dD, 32 * 32 * 4, hipMemcpyDeviceToHost);
  (void)hipFree(dA); (void)hipFree(dB); (void)hipFree(dSA); (void)hipFree(dSB); (void)hipFree(dD);
  return (int)e;
}
