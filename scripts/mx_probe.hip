// MX-scaled fp8 MFMA scale-mapping probe (gfx950).
// Fragment map verified (unit scales pass): A[row=l&31][k=(l>>5)*32+0..31].
// This probe sweeps per-lane scale sourcing hypotheses:
//   mode 0: sa = S[l&31][l>>5]                  (scale of lane's own block)
//   mode 1: sa = S[l>>1][l&1]                   (row-major pairs over lanes)
//   mode 2: sa = S[l&31][0] | S[l&31][1] << 8   (both blocks in one reg)
//   mode 3: mode 2 with opsel = 1
#include <hip/hip_runtime.h>
#include <stdint.h>

typedef __attribute__((ext_vector_type(8))) int i32x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;

__device__ int pick_scale(const uint8_t* S, int l, int mode) {
  switch (mode & 3) {
    case 0: return S[(l & 31) * 2 + (l >> 5)];
    case 1: return S[(l >> 1) * 2 + (l & 1)];
    default: return (int)S[(l & 31) * 2] | ((int)S[(l & 31) * 2 + 1] << 8);
  }
}

// lane-id-encoding modes for mapping derivation (scripts/mx_probe.py):
//  4: sa = 127 + (l&7),        sb = 127
//  5: sa = 127 + ((l>>3)&7),   sb = 127
//  6: sa = 127,                sb = 127 + (l&7)
//  7: sa = 127,                sb = 127 + ((l>>3)&7)
__device__ void probe_scales(int l, int mode, int* sa, int* sb) {
  const int lo = 127 + (l & 7);
  const int hi = 127 + ((l >> 3) & 7);
  *sa = 127; *sb = 127;
  if (mode == 4) *sa = lo;
  else if (mode == 5) *sa = hi;
  else if (mode == 6) *sb = lo;
  else if (mode == 7) *sb = hi;
}

// mode 8: sweep — iteration L perturbs ONLY lane L's A-scale (+1 exponent);
// D for each L lands in Dsweep[L][32][32]. Host decodes which (row, 16-elem
// sub-block) lane L sources from the delta vs the unit-scale baseline.
extern "C" __global__ void mx_sweep_kernel(const uint8_t* __restrict__ A,
                                           const uint8_t* __restrict__ B,
                                           float* __restrict__ Dsweep) {
  const int l = threadIdx.x;
  const int rc = l & 31;
  const int kb = l >> 5;
  i32x8 a = *reinterpret_cast<const i32x8*>(A + rc * 64 + kb * 32);
  i32x8 b = *reinterpret_cast<const i32x8*>(B + rc * 64 + kb * 32);
  for (int L = 0; L < 64; ++L) {
    f32x16 c;
#pragma unroll
    for (int i = 0; i < 16; ++i) c[i] = 0.f;
    const int sa = (l == L) ? 128 : 127;
    c = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(a, b, c, 0, 0, 0, sa,
                                                        0, 127);
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int row = (r & 3) + 8 * (r >> 2) + 4 * (l >> 5);
      const int col = l & 31;
      Dsweep[L * 1024 + row * 32 + col] = c[r];
    }
  }
}

// byte-1 variant of the sweep (perturbs scale byte1 instead of byte0)
extern "C" __global__ void mx_sweep_kernel_b1(const uint8_t* __restrict__ A,
                                              const uint8_t* __restrict__ B,
                                              float* __restrict__ Dsweep) {
  const int l = threadIdx.x;
  const int rc = l & 31;
  const int kb = l >> 5;
  i32x8 a = *reinterpret_cast<const i32x8*>(A + rc * 64 + kb * 32);
  i32x8 b = *reinterpret_cast<const i32x8*>(B + rc * 64 + kb * 32);
  for (int L = 0; L < 64; ++L) {
    f32x16 c;
#pragma unroll
    for (int i = 0; i < 16; ++i) c[i] = 0.f;
    const int sa = 127 | (((l == L) ? 128 : 127) << 8);
    c = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(a, b, c, 0, 0, 0, sa,
                                                        0, 127);
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int row = (r & 3) + 8 * (r >> 2) + 4 * (l >> 5);
      const int col = l & 31;
      Dsweep[L * 1024 + row * 32 + col] = c[r];
    }
  }
}

extern "C" int run_mx_sweep_b1(const uint8_t* hA, const uint8_t* hB, float* hD) {
  uint8_t *dA, *dB;
  float* dD;
  (void)hipMalloc(&dA, 32 * 64);
  (void)hipMalloc(&dB, 32 * 64);
  (void)hipMalloc(&dD, 64 * 1024 * 4);
  (void)hipMemcpy(dA, hA, 32 * 64, hipMemcpyHostToDevice);
  (void)hipMemcpy(dB, hB, 32 * 64, hipMemcpyHostToDevice);
  hipLaunchKernelGGL(mx_sweep_kernel_b1, dim3(1), dim3(64), 0, 0, dA, dB, dD);
  hipError_t e = hipDeviceSynchronize();
  (void)hipMemcpy(hD, dD, 64 * 1024 * 4, hipMemcpyDeviceToHost);
  (void)hipFree(dA); (void)hipFree(dB); (void)hipFree(dD);
  return (int)e;
}

extern "C" int run_mx_sweep(const uint8_t* hA, const uint8_t* hB, float* hD) {
  uint8_t *dA, *dB;
  float* dD;
  (void)hipMalloc(&dA, 32 * 64);
  (void)hipMalloc(&dB, 32 * 64);
  (void)hipMalloc(&dD, 64 * 1024 * 4);
  (void)hipMemcpy(dA, hA, 32 * 64, hipMemcpyHostToDevice);
  (void)hipMemcpy(dB, hB, 32 * 64, hipMemcpyHostToDevice);
  hipLaunchKernelGGL(mx_sweep_kernel, dim3(1), dim3(64), 0, 0, dA, dB, dD);
  hipError_t e = hipDeviceSynchronize();
  (void)hipMemcpy(hD, dD, 64 * 1024 * 4, hipMemcpyDeviceToHost);
  (void)hipFree(dA); (void)hipFree(dB); (void)hipFree(dD);
  return (int)e;
}

extern "C" __global__ void mx_probe_kernel(const uint8_t* __restrict__ A,
                                           const uint8_t* __restrict__ B,
                                           const uint8_t* __restrict__ scaleA,
                                           const uint8_t* __restrict__ scaleB,
                                           float* __restrict__ D, int mode) {
  const int l = threadIdx.x;
  const int rc = l & 31;
  const int kb = l >> 5;
  i32x8 a = *reinterpret_cast<const i32x8*>(A + rc * 64 + kb * 32);
  i32x8 b = *reinterpret_cast<const i32x8*>(B + rc * 64 + kb * 32);
  f32x16 c;
#pragma unroll
  for (int i = 0; i < 16; ++i) c[i] = 0.f;
  int sa, sb;
  if (mode >= 4) {
    probe_scales(l, mode, &sa, &sb);
  } else {
    sa = pick_scale(scaleA, l, mode);
    sb = pick_scale(scaleB, l, mode);
  }
  if (mode == 3)
    c = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(a, b, c, 0, 0, 1, sa, 1, sb);
  else
    c = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(a, b, c, 0, 0, 0, sa, 0, sb);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * (l >> 5);
    const int col = l & 31;
    D[row * 32 + col] = c[r];
  }
}

extern "C" int run_mx_probe(const uint8_t* hA, const uint8_t* hB,
                            const uint8_t* hSA, const uint8_t* hSB,
                            float* hD, int mode) {
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
  hipLaunchKernelGGL(mx_probe_kernel, dim3(1), dim3(64), 0, 0, dA, dB, dSA,
                     dSB, dD, mode);
  hipError_t e = hipDeviceSynchronize();
  (void)hipMemcpy(hD, dD, 32 * 32 * 4, hipMemcpyDeviceToHost);
  (void)hipFree(dA); (void)hipFree(dB); (void)hipFree(dSA); (void)hipFree(dSB); (void)hipFree(dD);
  return (int)e;
}
