// Standalone code objects for the torch-free cold-start runtime (lpx_rt).
//
// Compiled with `hipcc --genco --offload-arch=gfx950` into
// standalone_gfx950.hsaco and loaded at cold start via hipModuleLoad by the
// ctypes shim (shims/lpx_rt.py) — no JIT, no libtorch in the critical path
// ("loads precompiled .hsaco at import; cold-start ms is
// the metric"). Kernels mirror the torch-extension versions (rf.hip):
//   - lpx_rf_heap:   heap-layout RandomForest traversal (complete forests)
//   - lpx_argmax_rows: per-row argmax of [S, C] f32 -> int32 labels
//   - lpx_xgen:      synthetic ~N(0,1) feature matrix (xorshift + Box-Muller)

#include <hip/hip_fp16.h>
#include <hip/hip_runtime.h>
#include <stdint.h>

#define THREADS 256
#define CMAX 8
#define F_LDS_MAX 63

typedef __attribute__((ext_vector_type(2))) int i32x2;

extern "C" __global__ __launch_bounds__(THREADS) void lpx_rf_heap(
    const i32x2* __restrict__ nodes, const float* __restrict__ leaf_values,
    int n_trees, int depth, const float* __restrict__ X,
    float* __restrict__ out, int S, int F, int C) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* xl = reinterpret_cast<float*>(smem);
  const int t = threadIdx.x;
  const int rowf = F + 1;
  const int tree_stride = (1 << depth) - 1;
  const int n_leaves = 1 << depth;
  const bool stage = F <= F_LDS_MAX;

  for (int tile = blockIdx.x * THREADS; tile < S; tile += gridDim.x * THREADS) {
    const int s = tile + t;
    const int tile_n = min(THREADS, S - tile);
    if (stage) {
      const long total = (long)tile_n * F;
      for (long i = t; i < total; i += THREADS) {
        const int r = (int)(i / F), c = (int)(i - (long)r * F);
        xl[r * rowf + c] = X[((long)tile + r) * F + c];
      }
      __syncthreads();
    }
    if (s < S) {
      const float* xrow = stage ? (xl + t * rowf) : (X + (long)s * F);
      float acc[CMAX];
#pragma unroll
      for (int c = 0; c < CMAX; ++c) acc[c] = 0.f;
      for (int tr = 0; tr < n_trees; ++tr) {
        const i32x2* tb = nodes + (long)tr * tree_stride;
        int nd = 0;
        for (int d = 0; d < depth; ++d) {
          const i32x2 n = tb[nd];
          const float xv = xrow[(unsigned)n.x];
          nd = 2 * nd + 1 + (xv <= __int_as_float(n.y) ? 0 : 1);
        }
        const float* lv =
            leaf_values + ((long)tr * n_leaves + nd - tree_stride) * C;
#pragma unroll
        for (int c = 0; c < CMAX; ++c)
          if (c < C) acc[c] += lv[c];
      }
#pragma unroll
      for (int c = 0; c < CMAX; ++c)
        if (c < C) out[(long)s * C + c] = acc[c];
    }
    if (stage) __syncthreads();
  }
}

extern "C" __global__ __launch_bounds__(THREADS) void lpx_argmax_rows(
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

// xorshift128 + Box-Muller synthetic standard-normal features
extern "C" __global__ __launch_bounds__(THREADS) void lpx_xgen(
    float* __restrict__ out, long n, unsigned seed) {
  const long tid = (long)blockIdx.x * THREADS + threadIdx.x;
  unsigned s0 = 0x9E3779B9u ^ (unsigned)tid ^ seed;
  unsigned s1 = 0x85EBCA6Bu + (unsigned)(tid * 2654435761u);
  for (long i = tid * 2; i < n; i += (long)gridDim.x * THREADS * 2) {
    // two xorshift steps -> two uniforms
    unsigned x = s0;
    const unsigned y = s1;
    s0 = y;
    x ^= x << 23;
    s1 = x ^ y ^ (x >> 17) ^ (y >> 26);
    const unsigned u1 = s1 + y;
    x = s0; s0 = s1; unsigned xx = x ^ (x << 23);
    s1 = xx ^ s0 ^ (xx >> 17) ^ (s0 >> 26);
    const unsigned u2 = s1 + s0;
    const float f1 = ((u1 >> 8) + 1.0f) * (1.0f / 16777217.0f);  // (0,1]
    const float f2 = (u2 >> 8) * (1.0f / 16777216.0f);           // [0,1)
    const float r = sqrtf(-2.0f * logf(f1));
    float sn, cs;
    __sincosf(6.2831853f * f2, &sn, &cs);
    out[i] = r * cs;
    if (i + 1 < n) out[i + 1] = r * sn;
  }
}

// device-side synthetic forest generation (heap layout) — removes the numpy
// import + model H2D upload from the cold-start critical path.
extern "C" __global__ __launch_bounds__(THREADS) void lpx_forestgen_nodes(
    i32x2* __restrict__ nodes, long n_internal_total, int n_features,
    unsigned seed) {
  const long tid = (long)blockIdx.x * THREADS + threadIdx.x;
  for (long i = tid; i < n_internal_total; i += (long)gridDim.x * THREADS) {
    unsigned s0 = 0x9E3779B9u ^ (unsigned)i ^ seed;
    unsigned s1 = 0x85EBCA6Bu + (unsigned)(i * 2654435761u);
    // advance a few steps to decorrelate
    for (int k = 0; k < 3; ++k) {
      unsigned x = s0; s0 = s1; x ^= x << 23;
      s1 = x ^ s0 ^ (x >> 17) ^ (s0 >> 26);
    }
    const unsigned u1 = s1 + s0;
    unsigned x = s0; s0 = s1; unsigned xx = x ^ (x << 23);
    s1 = xx ^ s0 ^ (xx >> 17) ^ (s0 >> 26);
    const unsigned u2 = s1 + s0;
    const float f1 = ((u1 >> 8) + 1.0f) * (1.0f / 16777217.0f);
    const float f2 = (u2 >> 8) * (1.0f / 16777216.0f);
    float sn, cs;
    __sincosf(6.2831853f * f2, &sn, &cs);
    // rounded to fp16-representable f32 (matches lpx_rt.random_heap_forest_np:
    // the compact 4-B-node RF kernel traverses such forests bitwise-exactly)
    const float thresh = __half2float(__float2half(
        sqrtf(-2.0f * logf(f1)) * cs));
    i32x2 nd;
    nd.x = (int)(u2 % (unsigned)n_features);
    nd.y = __float_as_int(thresh);
    nodes[i] = nd;
  }
}

extern "C" __global__ __launch_bounds__(THREADS) void lpx_forestgen_leaves(
    float* __restrict__ leaves, long n_rows, int C, unsigned seed) {
  const long tid = (long)blockIdx.x * THREADS + threadIdx.x;
  for (long r = tid; r < n_rows; r += (long)gridDim.x * THREADS) {
    unsigned s0 = 0xDEADBEEFu ^ (unsigned)r ^ seed;
    unsigned s1 = 0xC2B2AE35u + (unsigned)(r * 2246822519u);
    float v[CMAX];
    float sum = 0.f;
    for (int c = 0; c < C && c < CMAX; ++c) {
      unsigned x = s0; s0 = s1; x ^= x << 23;
      s1 = x ^ s0 ^ (x >> 17) ^ (s0 >> 26);
      const float u = ((s1 + s0) >> 8) * (1.0f / 16777216.0f) + 1e-3f;
      v[c] = u;
      sum += u;
    }
    for (int c = 0; c < C && c < CMAX; ++c)
      // fp16-representable (matches random_heap_forest_np; exact for the
      // fp16-leaf-table compact kernel)
      leaves[r * C + c] = __half2float(__float2half(v[c] / sum));
  }
}

// ---- compact-forest fast path (mirrors rf.hip v8 + fp16 leaves + TILP4) --
// Node value for heap index i of the synthetic forest: same xorshift hash
// as lpx_forestgen_nodes, packed compact: (u16 feat << 16) | fp16(thresh).
__device__ __forceinline__ unsigned lpx_node_compact(long gi, int n_features,
                                                     unsigned seed) {
  unsigned s0 = 0x9E3779B9u ^ (unsigned)gi ^ seed;
  unsigned s1 = 0x85EBCA6Bu + (unsigned)(gi * 2654435761u);
  for (int k = 0; k < 3; ++k) {
    unsigned x = s0; s0 = s1; x ^= x << 23;
    s1 = x ^ s0 ^ (x >> 17) ^ (s0 >> 26);
  }
  const unsigned u1 = s1 + s0;
  unsigned x = s0; s0 = s1; unsigned xx = x ^ (x << 23);
  s1 = xx ^ s0 ^ (xx >> 17) ^ (s0 >> 26);
  const unsigned u2 = s1 + s0;
  const float f1 = ((u1 >> 8) + 1.0f) * (1.0f / 16777217.0f);
  const float f2 = (u2 >> 8) * (1.0f / 16777216.0f);
  float sn, cs;
  __sincosf(6.2831853f * f2, &sn, &cs);
  const __half h = __float2half(sqrtf(-2.0f * logf(f1)) * cs);
  const unsigned feat = u2 % (unsigned)n_features;
  return (feat << 16) | (unsigned)(unsigned short)__half_as_ushort(h);
}

// cpairs[tr*ppt + n] = {compact(2n+1), compact(2n+2)}; croots[tr] = node 0.
extern "C" __global__ __launch_bounds__(THREADS) void lpx_forestgen_cpairs(
    i32x2* __restrict__ cpairs, unsigned* __restrict__ croots, int n_trees,
    int depth, int n_features, unsigned seed) {
  const int stride = (1 << depth) - 1;
  const int ppt = (1 << (depth - 1)) - 1;
  const long total = (long)n_trees * ppt;
  const long tid = (long)blockIdx.x * THREADS + threadIdx.x;
  for (long i = tid; i < total; i += (long)gridDim.x * THREADS) {
    const int tr = (int)(i / ppt);
    const int n = (int)(i - (long)tr * ppt);
    const long base = (long)tr * stride;
    i32x2 rec;
    rec.x = (int)lpx_node_compact(base + 2 * n + 1, n_features, seed);
    rec.y = (int)lpx_node_compact(base + 2 * n + 2, n_features, seed);
    cpairs[i] = rec;
    if (n == 0) croots[tr] = lpx_node_compact(base, n_features, seed);
  }
}

// fp16 leaf table with the same values as lpx_forestgen_leaves (which
// already rounds to fp16-representable f32) — bitwise-equal accumulation.
extern "C" __global__ __launch_bounds__(THREADS) void lpx_forestgen_leaves16(
    unsigned short* __restrict__ leaves, long n_rows, int C, unsigned seed) {
  const long tid = (long)blockIdx.x * THREADS + threadIdx.x;
  for (long r = tid; r < n_rows; r += (long)gridDim.x * THREADS) {
    unsigned s0 = 0xDEADBEEFu ^ (unsigned)r ^ seed;
    unsigned s1 = 0xC2B2AE35u + (unsigned)(r * 2246822519u);
    float v[CMAX];
    float sum = 0.f;
    for (int c = 0; c < C && c < CMAX; ++c) {
      unsigned x = s0; s0 = s1; x ^= x << 23;
      s1 = x ^ s0 ^ (x >> 17) ^ (s0 >> 26);
      const float u = ((s1 + s0) >> 8) * (1.0f / 16777216.0f) + 1e-3f;
      v[c] = u;
      sum += u;
    }
    for (int c = 0; c < C && c < CMAX; ++c)
      leaves[r * C + c] = __half_as_ushort(__float2half(v[c] / sum));
  }
}

// compact child-pair traversal, fp16 leaves, tree-ILP 4 (rf.hip v8 port)
extern "C" __global__ __launch_bounds__(THREADS) void lpx_rf_cpair16(
    const i32x2* __restrict__ cpairs, const unsigned* __restrict__ croots,
    const unsigned short* __restrict__ leaves16, int n_trees, int depth,
    const float* __restrict__ X, float* __restrict__ out, int S, int F,
    int C) {
  constexpr int TILP = 4;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* xl = reinterpret_cast<float*>(smem);
  const int t = threadIdx.x;
  const int rowf = F + 1;
  const int tree_stride = (1 << depth) - 1;
  const int ppt = (1 << (depth - 1)) - 1;
  const int n_leaves = 1 << depth;
  const bool stage = F <= F_LDS_MAX;

  for (int tile = blockIdx.x * THREADS; tile < S; tile += gridDim.x * THREADS) {
    const int s = tile + t;
    const int tile_n = min(THREADS, S - tile);
    if (stage) {
      const long total = (long)tile_n * F;
      for (long i = t; i < total; i += THREADS) {
        const int r = (int)(i / F), c = (int)(i - (long)r * F);
        xl[r * rowf + c] = X[((long)tile + r) * F + c];
      }
      __syncthreads();
    }
    if (s < S) {
      const float* xrow = stage ? (xl + t * rowf) : (X + (long)s * F);
      float acc[CMAX];
#pragma unroll
      for (int c = 0; c < CMAX; ++c) acc[c] = 0.f;
      const int tr_main = n_trees - (n_trees % TILP);
      for (int tr = 0; tr < n_trees; tr += (tr < tr_main ? TILP : 1)) {
        const int nt = (tr < tr_main) ? TILP : 1;
        int nd[TILP];
        unsigned cf[TILP];
        float ct[TILP];
#pragma unroll
        for (int ti = 0; ti < TILP; ++ti) {
          const unsigned root = croots[tr + (ti < nt ? ti : 0)];
          nd[ti] = 0;
          cf[ti] = root >> 16;
          ct[ti] = __half2float(__ushort_as_half(
              (unsigned short)(root & 0xffffu)));
        }
        for (int d = 0; d < depth - 1; ++d) {
          i32x2 pr[TILP];
#pragma unroll
          for (int ti = 0; ti < TILP; ++ti)
            if (ti < nt) pr[ti] = cpairs[(long)(tr + ti) * ppt + nd[ti]];
#pragma unroll
          for (int ti = 0; ti < TILP; ++ti) {
            if (ti < nt) {
              const int go = (xrow[cf[ti]] <= ct[ti]) ? 0 : 1;
              nd[ti] = 2 * nd[ti] + 1 + go;
              const unsigned c = (unsigned)(go ? pr[ti].y : pr[ti].x);
              cf[ti] = c >> 16;
              ct[ti] = __half2float(__ushort_as_half(
                  (unsigned short)(c & 0xffffu)));
            }
          }
        }
#pragma unroll
        for (int ti = 0; ti < TILP; ++ti) {
          if (ti < nt) {
            const int go = (xrow[cf[ti]] <= ct[ti]) ? 0 : 1;
            const int leaf = 2 * nd[ti] + 1 + go - tree_stride;
            const unsigned short* lv =
                leaves16 + ((long)(tr + ti) * n_leaves + leaf) * C;
#pragma unroll
            for (int c = 0; c < CMAX; ++c)
              if (c < C)
                acc[c] += __half2float(__ushort_as_half(lv[c]));
          }
        }
      }
#pragma unroll
      for (int c = 0; c < CMAX; ++c)
        if (c < C) out[(long)s * C + c] = acc[c];
    }
    if (stage) __syncthreads();
  }
}

extern "C" __global__ __launch_bounds__(THREADS) void lpx_checksum_i32(
    const int* __restrict__ x, unsigned long long* __restrict__ out, long n) {
  long long local = 0;
  for (long i = (long)blockIdx.x * THREADS + threadIdx.x; i < n;
       i += (long)gridDim.x * THREADS)
    local += x[i];
  // wave reduce then one atomic per wave
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    local += __shfl_down(local, off, 64);
  if ((threadIdx.x & 63) == 0)
    atomicAdd(out, (unsigned long long)local);
}
