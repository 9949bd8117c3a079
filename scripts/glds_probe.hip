// Probe: LDS lane addressing of global_load_lds for sizes 1/2/4.
// Fills LDS from a known byte pattern and dumps it: reveals whether lane i's
// data lands at lds_base + i*size (tight) or i*4 (dword slots).
// Build on box: hipcc --offload-arch=gfx950 -O2 -shared -fPIC glds_probe.hip -o glds_probe.so
#include <hip/hip_runtime.h>

template <int SIZE>
__global__ __launch_bounds__(64) void probe(const unsigned char* __restrict__ src,
                                            unsigned char* __restrict__ out) {
  __shared__ unsigned char lds[1024];
  for (int i = threadIdx.x; i < 1024; i += 64) lds[i] = 0xEE;
  __syncthreads();
  const unsigned char* g = src + threadIdx.x * SIZE;
  const auto gp = (const __attribute__((address_space(1))) void*)g;
  const auto lp = (__attribute__((address_space(3))) void*)&lds[0];
  if constexpr (SIZE == 1) __builtin_amdgcn_global_load_lds(gp, lp, 1, 0, 0);
  else if constexpr (SIZE == 2) __builtin_amdgcn_global_load_lds(gp, lp, 2, 0, 0);
  else __builtin_amdgcn_global_load_lds(gp, lp, 4, 0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  for (int i = threadIdx.x; i < 1024; i += 64) out[i] = lds[i];
}

extern "C" int run_glds_probe(const unsigned char* src, unsigned char* out,
                              int size) {
  if (size == 1) probe<1><<<1, 64>>>(src, out);
  else if (size == 2) probe<2><<<1, 64>>>(src, out);
  else probe<4><<<1, 64>>>(src, out);
  if (hipDeviceSynchronize() != hipSuccess) return 2;
  return (int)hipGetLastError();
}
