// RandomForest inference for gfx950 — SoA forest traversal.
//
// THE metric workload (BASELINE.json config 4). Design (SURVEY.md §2.2 N9 +
// §7 hard-part 4 'RF traversal on SIMD-64'):
//   - forest in SoA: one int4 per node {feat, thresh-bits, left, right};
//     feat < 0 marks a leaf and `left` indexes the leaf-value table row.
//     16-B nodes -> one global_load_dwordx4 gather per step; the whole forest
//     (a few MB) is L2/L3-resident, so gathers are cache-served.
//   - one lane = one sample; a 256-thread block owns a 256-sample tile whose
//     feature rows are staged in LDS ([256][F+1] f32, +1 pad to spread banks)
//     when F <= 63, so the per-step feature gather never re-touches HBM.
//   - ILP trees walked concurrently per lane (independent pointer-chase
//     chains -> ILP node loads in flight, hides L2 latency). ILP=2 measured best
//     on the bench forest (256 trees, depth 10; the kernel is L2-BW-bound).
//   - per-class accumulators live in registers (compile-time CMAX=8 unroll;
//     the Python wrapper chunks wider classifiers over class groups).
//   - divergence: lanes exit at different depths; finished chains idle until
//     the wave's while-condition clears — depth-synchronous enough for
//     similar-depth CART trees.

#include "lpx_common.h"
#include "kernels.h"

namespace lpx {

namespace {

constexpr int THREADS = 256;
constexpr int CMAX = 8;
constexpr int F_LDS_MAX = 63;  // 256*(63+1)*4 = 64 KiB LDS

template <bool STAGE_X, int ILP>
__global__ __launch_bounds__(THREADS) void rf_kernel(
    const i32x4* __restrict__ nodes, const float* __restrict__ leaf_values,
    const int* __restrict__ roots, int n_trees, const float* __restrict__ X,
    float* __restrict__ out, int S, int F, int C) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* xl = reinterpret_cast<float*>(smem);  // [256][F+1] when STAGE_X
  const int t = threadIdx.x;
  const int rowf = F + 1;

  for (int tile = blockIdx.x * THREADS; tile < S; tile += gridDim.x * THREADS) {
    const int s = tile + t;
    const int tile_n = min(THREADS, S - tile);
    if (STAGE_X) {
      const long total = (long)tile_n * F;
      for (long i = t; i < total; i += THREADS) {
        const int r = (int)(i / F), c = (int)(i - (long)r * F);
        xl[r * rowf + c] = X[((long)tile + r) * F + c];
      }
      __syncthreads();
    }

    if (s < S) {
      const float* xrow = STAGE_X ? (xl + t * rowf) : (X + (long)s * F);
      float acc[CMAX];
#pragma unroll
      for (int c = 0; c < CMAX; ++c) acc[c] = 0.f;

      for (int tr = 0; tr < n_trees; tr += ILP) {
        int nd[ILP];
#pragma unroll
        for (int i = 0; i < ILP; ++i)
          nd[i] = (tr + i < n_trees) ? roots[tr + i] : -1;
        bool any = true;
        while (any) {
          any = false;
#pragma unroll
          for (int i = 0; i < ILP; ++i) {
            if (nd[i] >= 0) {
              const i32x4 n = nodes[nd[i]];
              if (n.x >= 0) {
                const float xv = xrow[n.x];
                nd[i] = (xv <= __int_as_float(n.y)) ? n.z : n.w;
                any = true;
              } else {
                const float* lv = leaf_values + (long)n.z * C;
#pragma unroll
                for (int c = 0; c < CMAX; ++c)
                  if (c < C) acc[c] += lv[c];
                nd[i] = -1;
              }
            }
          }
        }
      }
#pragma unroll
      for (int c = 0; c < CMAX; ++c)
        if (c < C) out[(long)s * C + c] = acc[c];
    }
    if (STAGE_X) __syncthreads();
  }
}

// ---- packed 8-B node kernel (v2) ----------------------------------------
// Node = {u32 meta, u32 val}: meta = feat(11b) << 21 | right_delta(21b);
// depth-first layout makes the left child implicit (nd+1) and co-locates the
// '<=' chain in the same cachelines; right child = nd + delta. Leaf:
// feat == 0x7FF and val = leaf-value row. Exact f32 thresholds (val bits).
// Halves the L2 line traffic of the 16-B SoA (the v1 kernel measured
// L2-BW-bound at ~35 TB/s of 64-B lines for 16-B nodes).

constexpr int LEAF_FEAT = 0x7FF;

template <bool STAGE_X, int ILP>
__global__ __launch_bounds__(THREADS) void rf8_kernel(
    const i32x2* __restrict__ nodes, const float* __restrict__ leaf_values,
    const int* __restrict__ roots, int n_trees, const float* __restrict__ X,
    float* __restrict__ out, int S, int F, int C) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* xl = reinterpret_cast<float*>(smem);
  const int t = threadIdx.x;
  const int rowf = F + 1;

  for (int tile = blockIdx.x * THREADS; tile < S; tile += gridDim.x * THREADS) {
    const int s = tile + t;
    const int tile_n = min(THREADS, S - tile);
    if (STAGE_X) {
      const long total = (long)tile_n * F;
      for (long i = t; i < total; i += THREADS) {
        const int r = (int)(i / F), c = (int)(i - (long)r * F);
        xl[r * rowf + c] = X[((long)tile + r) * F + c];
      }
      __syncthreads();
    }

    if (s < S) {
      const float* xrow = STAGE_X ? (xl + t * rowf) : (X + (long)s * F);
      float acc[CMAX];
#pragma unroll
      for (int c = 0; c < CMAX; ++c) acc[c] = 0.f;

      for (int tr = 0; tr < n_trees; tr += ILP) {
        int nd[ILP];
#pragma unroll
        for (int i = 0; i < ILP; ++i)
          nd[i] = (tr + i < n_trees) ? roots[tr + i] : -1;
        bool any = true;
        while (any) {
          any = false;
#pragma unroll
          for (int i = 0; i < ILP; ++i) {
            if (nd[i] >= 0) {
              const i32x2 n = nodes[nd[i]];
              const unsigned meta = (unsigned)n.x;
              const int feat = (int)(meta >> 21);
              if (feat != LEAF_FEAT) {
                const float xv = xrow[feat];
                nd[i] = (xv <= __int_as_float(n.y))
                            ? nd[i] + 1
                            : nd[i] + (int)(meta & 0x1FFFFF);
                any = true;
              } else {
                const float* lv = leaf_values + (long)n.y * C;
#pragma unroll
                for (int c = 0; c < CMAX; ++c)
                  if (c < C) acc[c] += lv[c];
                nd[i] = -1;
              }
            }
          }
        }
      }
#pragma unroll
      for (int c = 0; c < CMAX; ++c)
        if (c < C) out[(long)s * C + c] = acc[c];
    }
    if (STAGE_X) __syncthreads();
  }
}

// ---- forest-in-LDS chunked kernel (v3) ----------------------------------
// The v1/v2 kernels are L2 *line-throughput* bound: every node step pulls a
// 64-B line for 8-16 B of node, and 4 blocks/CU thrash the 32-KiB L1 so the
// depth-first locality is lost. v3 stages a CHUNK of whole trees (packed 8-B
// nodes) into LDS and traverses from there: node fetches become ds_read_b64
// (no line over-fetch, ~75 TB/s aggregate), forest bytes stream on-chip once
// per block per chunk. One block per CU; per-class partial sums accumulate
// into out[] across chunks (block owns its sample tiles exclusively).
//
// Host splits trees into chunks fitting: LDS = chunk nodes * 8 + X tile
// (256 x (F+1) x 4 when F <= F_LDS_MAX) <= 160 KiB.

template <bool STAGE_X, int ILP>
__global__ __launch_bounds__(THREADS) void rf_lds_kernel(
    const i32x2* __restrict__ nodes, const float* __restrict__ leaf_values,
    const int* __restrict__ roots, const int* __restrict__ chunk_tree_begin,
    int n_chunks, int n_trees, int total_nodes, const float* __restrict__ X,
    float* __restrict__ out, int S, int F, int C, int max_chunk_nodes) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  i32x2* lnodes = reinterpret_cast<i32x2*>(smem);                // chunk nodes
  float* xl = reinterpret_cast<float*>(smem + (size_t)max_chunk_nodes * 8);
  const int t = threadIdx.x;
  const int rowf = F + 1;

  for (int ch = 0; ch < n_chunks; ++ch) {
    const int tree0 = chunk_tree_begin[ch];
    const int tree1 = chunk_tree_begin[ch + 1];
    const int node0 = roots[tree0];
    const int node1 = (tree1 < n_trees) ? roots[tree1] : total_nodes;
    const int count = node1 - node0;  // host guarantees <= max_chunk_nodes
    __syncthreads();  // previous iteration done with LDS
    for (int i = t; i < count; i += THREADS) lnodes[i] = nodes[node0 + i];
    __syncthreads();

    for (int tile = blockIdx.x * THREADS; tile < S;
         tile += gridDim.x * THREADS) {
      const int s = tile + t;
      const int tile_n = min(THREADS, S - tile);
      if (STAGE_X) {
        const long total = (long)tile_n * F;
        for (long i = t; i < total; i += THREADS) {
          const int r = (int)(i / F), c = (int)(i - (long)r * F);
          xl[r * rowf + c] = X[((long)tile + r) * F + c];
        }
        __syncthreads();
      }
      if (s < S) {
        const float* xrow = STAGE_X ? (xl + t * rowf) : (X + (long)s * F);
        float acc[CMAX];
#pragma unroll
        for (int c = 0; c < CMAX; ++c) acc[c] = 0.f;
        for (int tr = tree0; tr < tree1; tr += ILP) {
          int nd[ILP];
#pragma unroll
          for (int i = 0; i < ILP; ++i)
            nd[i] = (tr + i < tree1) ? roots[tr + i] - node0 : -1;
          bool any = true;
          while (any) {
            any = false;
#pragma unroll
            for (int i = 0; i < ILP; ++i) {
              if (nd[i] >= 0) {
                const i32x2 n = lnodes[nd[i]];
                const unsigned meta = (unsigned)n.x;
                const int feat = (int)(meta >> 21);
                if (feat != LEAF_FEAT) {
                  const float xv = xrow[feat];
                  nd[i] = (xv <= __int_as_float(n.y))
                              ? nd[i] + 1
                              : nd[i] + (int)(meta & 0x1FFFFF);
                  any = true;
                } else {
                  const float* lv = leaf_values + (long)n.y * C;
#pragma unroll
                  for (int c = 0; c < CMAX; ++c)
                    if (c < C) acc[c] += lv[c];
                  nd[i] = -1;
                }
              }
            }
          }
        }
        float* orow = out + (long)s * C;
        if (ch == 0) {
#pragma unroll
          for (int c = 0; c < CMAX; ++c)
            if (c < C) orow[c] = acc[c];
        } else {
#pragma unroll
          for (int c = 0; c < CMAX; ++c)
            if (c < C) orow[c] += acc[c];
        }
      }
      if (STAGE_X) __syncthreads();
    }
  }
}

// ---- heap-layout kernel (v4, complete equal-depth forests) ---------------
// Level-order (binary-heap) node layout: tree t's INTERNAL nodes only,
// contiguous, node i's children at 2i+1 / 2i+2; a sample exits after exactly
// `depth` steps at leaf row = i - (2^depth - 1). Wins:
//   - branchless fixed-depth loop (no divergence at all);
//   - all lanes walk the SAME tree at the SAME depth, and each level is a
//     contiguous block, so the wave's 64 node loads coalesce into the few
//     lines of that level (top levels: 1-8 lines for the whole wave) — the
//     general kernel pays ~1 L2 line per lane per step and is line-bound.
// node = {u32 feat, f32 thresh} (8 B); leaf values in leaf_values rows
// [t * 2^depth + leaf_idx]. Used for ForestSoA.random forests (and heapified
// sklearn trees when padding stays cheap); ragged forests use rf8_kernel.

template <bool STAGE_X, int SILP, int BLK = THREADS>
__global__ __launch_bounds__(BLK) void rf_heap_kernel(
    const i32x2* __restrict__ nodes, const float* __restrict__ leaf_values,
    int n_trees, int depth, const float* __restrict__ X,
    float* __restrict__ out, int S, int F, int C) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* xl = reinterpret_cast<float*>(smem);
  const int t = threadIdx.x;
  const int rowf = F + 1;
  const int tree_stride = (1 << depth) - 1;  // internal nodes per tree
  const int n_leaves = 1 << depth;

  for (int tile = blockIdx.x * (BLK * SILP); tile < S;
       tile += gridDim.x * (BLK * SILP)) {
    const int tile_n = min(BLK * SILP, S - tile);
    if (STAGE_X) {
      const long total = (long)tile_n * F;
      for (long i = t; i < total; i += BLK) {
        const int r = (int)(i / F), c = (int)(i - (long)r * F);
        xl[r * rowf + c] = X[((long)tile + r) * F + c];
      }
      __syncthreads();
    }

    int s[SILP];
    const float* xrow[SILP];
#pragma unroll
    for (int j = 0; j < SILP; ++j) {
      s[j] = tile + t + j * BLK;
      xrow[j] = STAGE_X ? (xl + (t + j * BLK) * rowf)
                        : (X + (long)s[j] * F);
    }
    float acc[SILP][CMAX];
#pragma unroll
    for (int j = 0; j < SILP; ++j)
#pragma unroll
      for (int c = 0; c < CMAX; ++c) acc[j][c] = 0.f;

    for (int tr = 0; tr < n_trees; ++tr) {
      const i32x2* tb = nodes + (long)tr * tree_stride;
      int nd[SILP];
#pragma unroll
      for (int j = 0; j < SILP; ++j) nd[j] = 0;
      for (int d = 0; d < depth; ++d) {
#pragma unroll
        for (int j = 0; j < SILP; ++j) {
          if (s[j] < S) {
            const i32x2 n = tb[nd[j]];
            const float xv = xrow[j][(unsigned)n.x];
            nd[j] = 2 * nd[j] + 1 + (xv <= __int_as_float(n.y) ? 0 : 1);
          }
        }
      }
#pragma unroll
      for (int j = 0; j < SILP; ++j) {
        if (s[j] < S) {
          const int leaf = nd[j] - tree_stride;  // - (2^depth - 1)
          const float* lv = leaf_values + ((long)tr * n_leaves + leaf) * C;
#pragma unroll
          for (int c = 0; c < CMAX; ++c)
            if (c < C) acc[j][c] += lv[c];
        }
      }
    }
#pragma unroll
    for (int j = 0; j < SILP; ++j)
      if (s[j] < S)
#pragma unroll
        for (int c = 0; c < CMAX; ++c)
          if (c < C) out[(long)s[j] * C + c] = acc[j][c];
    if (STAGE_X) __syncthreads();
  }
}

// v6: child-PAIR heap traversal — nodes repacked so node n's two children
// (heap 2n+1, 2n+2) sit in ONE 16-B record pairs[tr][n] (n < 2^(depth-1)-1;
// roots in a separate [n_trees] i32x2 array). A level step loads the child
// pair of the CURRENT node while comparing it, then selects the next node
// from registers: depth-1 random 16-B L2 requests per tree instead of
// depth 8-B requests. The v4 kernel is L2 REQUEST-bound (0.895 wait frac,
// invariant to block size / occupancy — profiles/README), and the L2 moves
// a full line per random request either way, so ~10% fewer requests plus a
// root served from a 2-KB cached array is a direct request-count cut.
template <bool STAGE_X, int SILP, int TILP = 1>
__global__ __launch_bounds__(THREADS) void rf_heap_pair_kernel(
    const i32x4* __restrict__ pairs, const i32x2* __restrict__ roots,
    const float* __restrict__ leaf_values, int n_trees, int depth,
    const float* __restrict__ X, float* __restrict__ out, int S, int F,
    int C) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* xl = reinterpret_cast<float*>(smem);
  const int t = threadIdx.x;
  const int rowf = F + 1;
  const int tree_stride = (1 << depth) - 1;
  const int ppt = (1 << (depth - 1)) - 1;  // pairs per tree
  const int n_leaves = 1 << depth;

  for (int tile = blockIdx.x * (THREADS * SILP); tile < S;
       tile += gridDim.x * (THREADS * SILP)) {
    const int tile_n = min(THREADS * SILP, S - tile);
    if (STAGE_X) {
      const long total = (long)tile_n * F;
      for (long i = t; i < total; i += THREADS) {
        const int r = (int)(i / F), c = (int)(i - (long)r * F);
        xl[r * rowf + c] = X[((long)tile + r) * F + c];
      }
      __syncthreads();
    }

    int s[SILP];
    const float* xrow[SILP];
#pragma unroll
    for (int j = 0; j < SILP; ++j) {
      s[j] = tile + t + j * THREADS;
      xrow[j] = STAGE_X ? (xl + (t + j * THREADS) * rowf)
                        : (X + (long)s[j] * F);
    }
    float acc[SILP][CMAX];
#pragma unroll
    for (int j = 0; j < SILP; ++j)
#pragma unroll
      for (int c = 0; c < CMAX; ++c) acc[j][c] = 0.f;

    // TILP independent tree chains per lane (latency-bound kernel; see the
    // compact variant): no extra LDS, registers only.
    const int tr_main = n_trees - (n_trees % TILP);
    for (int tr = 0; tr < n_trees; tr += (tr < tr_main ? TILP : 1)) {
      const int nt = (tr < tr_main) ? TILP : 1;
      int nd[SILP][TILP];
      int cf[SILP][TILP];
      float ct[SILP][TILP];
#pragma unroll
      for (int ti = 0; ti < TILP; ++ti) {
        const i32x2 root = roots[tr + (ti < nt ? ti : 0)];
#pragma unroll
        for (int j = 0; j < SILP; ++j) {
          nd[j][ti] = 0;
          cf[j][ti] = root.x;
          ct[j][ti] = __int_as_float(root.y);
        }
      }
      for (int d = 0; d < depth - 1; ++d) {
#pragma unroll
        for (int j = 0; j < SILP; ++j) {
          if (s[j] < S) {
            i32x4 pr[TILP];
#pragma unroll
            for (int ti = 0; ti < TILP; ++ti)
              if (ti < nt)
                pr[ti] = pairs[(long)(tr + ti) * ppt + nd[j][ti]];
#pragma unroll
            for (int ti = 0; ti < TILP; ++ti) {
              if (ti < nt) {
                const int go =
                    (xrow[j][(unsigned)cf[j][ti]] <= ct[j][ti]) ? 0 : 1;
                nd[j][ti] = 2 * nd[j][ti] + 1 + go;
                cf[j][ti] = go ? pr[ti].z : pr[ti].x;
                ct[j][ti] = __int_as_float(go ? pr[ti].w : pr[ti].y);
              }
            }
          }
        }
      }
#pragma unroll
      for (int j = 0; j < SILP; ++j) {
        if (s[j] < S) {
#pragma unroll
          for (int ti = 0; ti < TILP; ++ti) {
            if (ti < nt) {
              const int go =
                  (xrow[j][(unsigned)cf[j][ti]] <= ct[j][ti]) ? 0 : 1;
              const int leaf = 2 * nd[j][ti] + 1 + go - tree_stride;
              const float* lv =
                  leaf_values + ((long)(tr + ti) * n_leaves + leaf) * C;
#pragma unroll
              for (int c = 0; c < CMAX; ++c)
                if (c < C) acc[j][c] += lv[c];
            }
          }
        }
      }
    }
#pragma unroll
    for (int j = 0; j < SILP; ++j)
      if (s[j] < S)
#pragma unroll
        for (int c = 0; c < CMAX; ++c)
          if (c < C) out[(long)s[j] * C + c] = acc[j][c];
    if (STAGE_X) __syncthreads();
  }
}

// v8: COMPACT child-pair traversal — 4-B nodes (u16 feature id, fp16
// threshold: node = feat<<16 | f16bits) so a child-pair record is 8 B and
// the roots array 4 B/tree. The pair kernel is L2 REQUEST-bound but the
// quad A/B showed bytes/latency also matter (NOTES item 4): halving the
// bytes of every random request keeps the request count of v6 while
// halving its line footprint — two pair records now share a 16-B sector.
// Numerics: thresholds are rounded f32->fp16 (RNE) at pack time; a
// decision flips only for x strictly between the f32 and fp16 threshold
// (|dt| <= 2^-11 relative), quantified in tests. For forests whose
// thresholds are already fp16-representable the traversal is EXACT.
// Feature ids must fit u16 (F <= 65535 — always true here).
__device__ __forceinline__ float lpx_h2f(unsigned short u) {
  __half_raw r;
  r.x = u;
  return __half2float(__half(r));
}

template <bool STAGE_X, int SILP, bool LV16 = false, int TILP = 1,
          int NT = THREADS>
__global__ __launch_bounds__(NT) void rf_heap_cpair_kernel(
    const i32x2* __restrict__ pairs, const int* __restrict__ roots,
    const float* __restrict__ leaf_values, int n_trees, int depth,
    const float* __restrict__ X, float* __restrict__ out, int S, int F,
    int C) {
  // LV16: leaf_values is fp16 [rows, C] (C <= 8): the leaf row is ONE 16-B
  // request instead of 32 B — exact when leaves are fp16-representable
  // (the synthetic generators quantize; accumulation stays f32).
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* xl = reinterpret_cast<float*>(smem);
  const int t = threadIdx.x;
  const int rowf = F + 1;
  const int tree_stride = (1 << depth) - 1;
  const int ppt = (1 << (depth - 1)) - 1;  // pairs per tree
  const int n_leaves = 1 << depth;

  for (int tile = blockIdx.x * (NT * SILP); tile < S;
       tile += gridDim.x * (NT * SILP)) {
    const int tile_n = min(NT * SILP, S - tile);
    if (STAGE_X) {
      const long total = (long)tile_n * F;
      for (long i = t; i < total; i += NT) {
        const int r = (int)(i / F), c = (int)(i - (long)r * F);
        xl[r * rowf + c] = X[((long)tile + r) * F + c];
      }
      __syncthreads();
    }

    int s[SILP];
    const float* xrow[SILP];
#pragma unroll
    for (int j = 0; j < SILP; ++j) {
      s[j] = tile + t + j * NT;
      xrow[j] = STAGE_X ? (xl + (t + j * NT) * rowf)
                        : (X + (long)s[j] * F);
    }
    float acc[SILP][CMAX];
#pragma unroll
    for (int j = 0; j < SILP; ++j)
#pragma unroll
      for (int c = 0; c < CMAX; ++c) acc[j][c] = 0.f;

    // TILP independent tree chains per lane fill the fetch latency the
    // serial per-tree walk exposes (the kernel is latency-bound: PMC
    // wait_frac 0.85 with ONE outstanding request per lane). No extra LDS
    // (same sample row serves every chain), unlike sample-ILP.
    const int tr_main = n_trees - (n_trees % TILP);
    for (int tr = 0; tr < n_trees; tr += (tr < tr_main ? TILP : 1)) {
      const int nt = (tr < tr_main) ? TILP : 1;
      int nd[SILP][TILP];
      unsigned cf[SILP][TILP];
      float ct[SILP][TILP];
#pragma unroll
      for (int ti = 0; ti < TILP; ++ti) {
        const unsigned root =
            (unsigned)roots[tr + (ti < nt ? ti : 0)];
#pragma unroll
        for (int j = 0; j < SILP; ++j) {
          nd[j][ti] = 0;
          cf[j][ti] = root >> 16;
          ct[j][ti] = lpx_h2f((unsigned short)(root & 0xffffu));
        }
      }
      for (int d = 0; d < depth - 1; ++d) {
#pragma unroll
        for (int j = 0; j < SILP; ++j) {
          if (s[j] < S) {
            i32x2 pr[TILP];
#pragma unroll
            for (int ti = 0; ti < TILP; ++ti)
              if (ti < nt)
                pr[ti] = pairs[(long)(tr + ti) * ppt + nd[j][ti]];
#pragma unroll
            for (int ti = 0; ti < TILP; ++ti) {
              if (ti < nt) {
                const int go = (xrow[j][cf[j][ti]] <= ct[j][ti]) ? 0 : 1;
                nd[j][ti] = 2 * nd[j][ti] + 1 + go;
                const unsigned c = (unsigned)(go ? pr[ti].y : pr[ti].x);
                cf[j][ti] = c >> 16;
                ct[j][ti] = lpx_h2f((unsigned short)(c & 0xffffu));
              }
            }
          }
        }
      }
#pragma unroll
      for (int j = 0; j < SILP; ++j) {
        if (s[j] < S) {
#pragma unroll
          for (int ti = 0; ti < TILP; ++ti) {
            if (ti < nt) {
              const int go = (xrow[j][cf[j][ti]] <= ct[j][ti]) ? 0 : 1;
              const int leaf = 2 * nd[j][ti] + 1 + go - tree_stride;
              if constexpr (LV16) {
                const unsigned short* lv16 =
                    reinterpret_cast<const unsigned short*>(leaf_values) +
                    ((long)(tr + ti) * n_leaves + leaf) * C;
#pragma unroll
                for (int c = 0; c < CMAX; ++c)
                  if (c < C) acc[j][c] += lpx_h2f(lv16[c]);
              } else {
                const float* lv =
                    leaf_values + ((long)(tr + ti) * n_leaves + leaf) * C;
#pragma unroll
                for (int c = 0; c < CMAX; ++c)
                  if (c < C) acc[j][c] += lv[c];
              }
            }
          }
        }
      }
    }
#pragma unroll
    for (int j = 0; j < SILP; ++j)
      if (s[j] < S)
#pragma unroll
        for (int c = 0; c < CMAX; ++c)
          if (c < C) out[(long)s[j] * C + c] = acc[j][c];
    if (STAGE_X) __syncthreads();
  }
}

// v7: QUAD records — two tree levels per random fetch. Record r for an
// even-level node n (level d=2k, within-level offset o, record slot
// (4^k-1)/3 + o) is 64 B: {children(n) [2 nodes], grandchildren(n)
// [4 nodes], 16 B pad}. One 64-B-aligned record read = 3 dwordx4 loads of
// the SAME line (first misses L1 -> one L2 request, the rest hit L1), and
// serves TWO compares: depth/2 L2 requests per tree instead of depth-1.
// Slot recurrence mirrors the heap with branching factor 4:
// slot' = 4*slot + 1 + (2*b1 + b2). Requires even depth (launcher
// falls back to the pair kernel otherwise). Memory: 2x the flat heap
// (each node appears in its parent's and grandparent's records).
template <bool STAGE_X, int SILP>
__global__ __launch_bounds__(THREADS) void rf_heap_quad_kernel(
    const i32x4* __restrict__ quads, const i32x2* __restrict__ roots,
    const float* __restrict__ leaf_values, int n_trees, int depth,
    const float* __restrict__ X, float* __restrict__ out, int S, int F,
    int C) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* xl = reinterpret_cast<float*>(smem);
  const int t = threadIdx.x;
  const int rowf = F + 1;
  const int tree_stride = (1 << depth) - 1;
  const int qpt = ((1 << depth) - 1) / 3;  // (4^(depth/2) - 1) / 3 records
  const int n_leaves = 1 << depth;

  for (int tile = blockIdx.x * (THREADS * SILP); tile < S;
       tile += gridDim.x * (THREADS * SILP)) {
    const int tile_n = min(THREADS * SILP, S - tile);
    if (STAGE_X) {
      const long total = (long)tile_n * F;
      for (long i = t; i < total; i += THREADS) {
        const int r = (int)(i / F), c = (int)(i - (long)r * F);
        xl[r * rowf + c] = X[((long)tile + r) * F + c];
      }
      __syncthreads();
    }

    int s[SILP];
    const float* xrow[SILP];
#pragma unroll
    for (int j = 0; j < SILP; ++j) {
      s[j] = tile + t + j * THREADS;
      xrow[j] = STAGE_X ? (xl + (t + j * THREADS) * rowf)
                        : (X + (long)s[j] * F);
    }
    float acc[SILP][CMAX];
#pragma unroll
    for (int j = 0; j < SILP; ++j)
#pragma unroll
      for (int c = 0; c < CMAX; ++c) acc[j][c] = 0.f;

    for (int tr = 0; tr < n_trees; ++tr) {
      const i32x4* tq = quads + (long)tr * qpt * 4;
      const i32x2 root = roots[tr];
      int slot[SILP], nn[SILP], cf[SILP];
      float ct[SILP];
#pragma unroll
      for (int j = 0; j < SILP; ++j) {
        slot[j] = 0;
        nn[j] = 0;
        cf[j] = root.x;
        ct[j] = __int_as_float(root.y);
      }
      for (int k = 0; k < (depth >> 1) - 1; ++k) {  // full two-level steps
#pragma unroll
        for (int j = 0; j < SILP; ++j) {
          if (s[j] < S) {
            const i32x4* rec = tq + slot[j] * 4;
            const i32x4 ch = rec[0];
            const i32x4 g01 = rec[1];
            const i32x4 g23 = rec[2];
            const int b1 = (xrow[j][(unsigned)cf[j]] <= ct[j]) ? 0 : 1;
            const int chf = b1 ? ch.z : ch.x;
            const float cht = __int_as_float(b1 ? ch.w : ch.y);
            const int b2 = (xrow[j][(unsigned)chf] <= cht) ? 0 : 1;
            const int q = 2 * b1 + b2;
            const i32x4 gg = (q < 2) ? g01 : g23;
            cf[j] = (q & 1) ? gg.z : gg.x;
            ct[j] = __int_as_float((q & 1) ? gg.w : gg.y);
            slot[j] = 4 * slot[j] + 1 + q;
            nn[j] = 4 * nn[j] + 3 + q;
          }
        }
      }
#pragma unroll
      for (int j = 0; j < SILP; ++j) {  // partial step: children then leaves
        if (s[j] < S) {
          const i32x4 ch = tq[slot[j] * 4];
          const int b1 = (xrow[j][(unsigned)cf[j]] <= ct[j]) ? 0 : 1;
          const int chf = b1 ? ch.z : ch.x;
          const float cht = __int_as_float(b1 ? ch.w : ch.y);
          const int b2 = (xrow[j][(unsigned)chf] <= cht) ? 0 : 1;
          const int leaf = 4 * nn[j] + 3 + 2 * b1 + b2 - tree_stride;
          const float* lv = leaf_values + ((long)tr * n_leaves + leaf) * C;
#pragma unroll
          for (int c = 0; c < CMAX; ++c)
            if (c < C) acc[j][c] += lv[c];
        }
      }
    }
#pragma unroll
    for (int j = 0; j < SILP; ++j)
      if (s[j] < S)
#pragma unroll
        for (int c = 0; c < CMAX; ++c)
          if (c < C) out[(long)s[j] * C + c] = acc[j][c];
    if (STAGE_X) __syncthreads();
  }
}

// v5: two-level heap traversal — the top `top` levels of every tree are
// cached in LDS (loaded once per block), the remaining levels read from
// L2/HBM. The kernel is L2-request-bound (243 Ms/s = ~5 TB/s of 8-B node
// fetches at the bench shape), so serving the hottest, perfectly-shared
// top nodes from LDS removes top/depth of that traffic.
template <bool STAGE_X, int SILP>
__global__ __launch_bounds__(THREADS) void rf_heap2_kernel(
    const i32x2* __restrict__ nodes, const float* __restrict__ leaf_values,
    int n_trees, int depth, int top, const float* __restrict__ X,
    float* __restrict__ out, int S, int F, int C) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* xl = reinterpret_cast<float*>(smem);
  const int rowf = F + 1;
  const size_t xbytes = STAGE_X ? (size_t)THREADS * SILP * rowf * 4 : 0;
  i32x2* ltop = reinterpret_cast<i32x2*>(smem + xbytes);
  const int t = threadIdx.x;
  const int tree_stride = (1 << depth) - 1;
  const int top_stride = (1 << top) - 1;
  const int n_leaves = 1 << depth;

  {  // cooperative top-level load (once per block; heap prefix per tree)
    const long total = (long)n_trees * top_stride;
    for (long i = t; i < total; i += THREADS) {
      const int tr = (int)(i / top_stride), nd = (int)(i % top_stride);
      ltop[i] = nodes[(long)tr * tree_stride + nd];
    }
    __syncthreads();
  }

  for (int tile = blockIdx.x * (THREADS * SILP); tile < S;
       tile += gridDim.x * (THREADS * SILP)) {
    const int tile_n = min(THREADS * SILP, S - tile);
    if (STAGE_X) {
      const long total = (long)tile_n * F;
      for (long i = t; i < total; i += THREADS) {
        const int r = (int)(i / F), c = (int)(i - (long)r * F);
        xl[r * rowf + c] = X[((long)tile + r) * F + c];
      }
      __syncthreads();
    }

    int s[SILP];
    const float* xrow[SILP];
#pragma unroll
    for (int j = 0; j < SILP; ++j) {
      s[j] = tile + t + j * THREADS;
      xrow[j] = STAGE_X ? (xl + (t + j * THREADS) * rowf)
                        : (X + (long)s[j] * F);
    }
    float acc[SILP][CMAX];
#pragma unroll
    for (int j = 0; j < SILP; ++j)
#pragma unroll
      for (int c = 0; c < CMAX; ++c) acc[j][c] = 0.f;

    for (int tr = 0; tr < n_trees; ++tr) {
      const i32x2* ttop = ltop + (long)tr * top_stride;
      const i32x2* tb = nodes + (long)tr * tree_stride;
      int nd[SILP];
#pragma unroll
      for (int j = 0; j < SILP; ++j) nd[j] = 0;
      for (int d = 0; d < top; ++d) {
#pragma unroll
        for (int j = 0; j < SILP; ++j) {
          if (s[j] < S) {
            const i32x2 n = ttop[nd[j]];
            const float xv = xrow[j][(unsigned)n.x];
            nd[j] = 2 * nd[j] + 1 + (xv <= __int_as_float(n.y) ? 0 : 1);
          }
        }
      }
      for (int d = top; d < depth; ++d) {
#pragma unroll
        for (int j = 0; j < SILP; ++j) {
          if (s[j] < S) {
            const i32x2 n = tb[nd[j]];
            const float xv = xrow[j][(unsigned)n.x];
            nd[j] = 2 * nd[j] + 1 + (xv <= __int_as_float(n.y) ? 0 : 1);
          }
        }
      }
#pragma unroll
      for (int j = 0; j < SILP; ++j) {
        if (s[j] < S) {
          const int leaf = nd[j] - tree_stride;
          const float* lv = leaf_values + ((long)tr * n_leaves + leaf) * C;
#pragma unroll
          for (int c = 0; c < CMAX; ++c)
            if (c < C) acc[j][c] += lv[c];
        }
      }
    }
#pragma unroll
    for (int j = 0; j < SILP; ++j)
      if (s[j] < S)
#pragma unroll
        for (int c = 0; c < CMAX; ++c)
          if (c < C) out[(long)s[j] * C + c] = acc[j][c];
    if (STAGE_X) __syncthreads();
  }
}

}  // namespace

hipError_t rf_predict_heap_quad_f32(const int* quads, const int* roots,
                                    const float* leaf_values, int n_trees,
                                    int depth, const float* X, float* out,
                                    int S, int F, int C, int silp,
                                    hipStream_t stream) {
  if (C > CMAX || depth < 4 || depth > 26 || (depth & 1))
    return hipErrorInvalidValue;
  const int SILP = (silp == 2) ? 2 : (silp == 4 ? 4 : 1);
  const bool stage = F <= F_LDS_MAX;
  const size_t lds = stage ? (size_t)THREADS * SILP * (F + 1) * 4 : 0;
  if (lds > 160 * 1024) return hipErrorInvalidValue;
  unsigned blocks = lpx_ceil_div(S, THREADS * SILP);
  if (blocks > 4096u) blocks = 4096u;
  if (blocks == 0) blocks = 1;
  const auto* qq = reinterpret_cast<const i32x4*>(quads);
  const auto* rr = reinterpret_cast<const i32x2*>(roots);
#define LPX_RFHQ_CASE(I)                                                      \
  if (SILP == I) {                                                            \
    if (stage)                                                                \
      rf_heap_quad_kernel<true, I>                                            \
          <<<dim3(blocks), dim3(THREADS), lds, stream>>>(                     \
              qq, rr, leaf_values, n_trees, depth, X, out, S, F, C);          \
    else                                                                      \
      rf_heap_quad_kernel<false, I>                                           \
          <<<dim3(blocks), dim3(THREADS), 0, stream>>>(                       \
              qq, rr, leaf_values, n_trees, depth, X, out, S, F, C);          \
    return hipGetLastError();                                                 \
  }
  LPX_RFHQ_CASE(1)
  LPX_RFHQ_CASE(2)
  LPX_RFHQ_CASE(4)
#undef LPX_RFHQ_CASE
  return hipErrorInvalidValue;
}

hipError_t rf_predict_heap_pair_f32(const int* pairs, const int* roots,
                                    const float* leaf_values, int n_trees,
                                    int depth, const float* X, float* out,
                                    int S, int F, int C, int silp,
                                    hipStream_t stream) {
  if (C > CMAX || depth < 2 || depth > 26) return hipErrorInvalidValue;
  const int TILP = (silp == 21) ? 2 : (silp == 41 ? 4 : (silp == 81 ? 8 : 1));
  const int SILP = (silp == 2) ? 2 : (silp == 4 ? 4 : 1);
  const bool stage = F <= F_LDS_MAX;
  const size_t lds = stage ? (size_t)THREADS * SILP * (F + 1) * 4 : 0;
  if (lds > 160 * 1024) return hipErrorInvalidValue;
  unsigned blocks = lpx_ceil_div(S, THREADS * SILP);
  if (blocks > 4096u) blocks = 4096u;
  if (blocks == 0) blocks = 1;
  const auto* pp = reinterpret_cast<const i32x4*>(pairs);
  const auto* rr = reinterpret_cast<const i32x2*>(roots);
#define LPX_RFHP_CASE(I, T)                                                   \
  if (SILP == I && TILP == T) {                                               \
    if (stage)                                                                \
      rf_heap_pair_kernel<true, I, T>                                         \
          <<<dim3(blocks), dim3(THREADS), lds, stream>>>(                     \
              pp, rr, leaf_values, n_trees, depth, X, out, S, F, C);          \
    else                                                                      \
      rf_heap_pair_kernel<false, I, T>                                        \
          <<<dim3(blocks), dim3(THREADS), 0, stream>>>(                       \
              pp, rr, leaf_values, n_trees, depth, X, out, S, F, C);          \
    return hipGetLastError();                                                 \
  }
  LPX_RFHP_CASE(1, 1)
  LPX_RFHP_CASE(2, 1)
  LPX_RFHP_CASE(4, 1)
  LPX_RFHP_CASE(1, 2)
  LPX_RFHP_CASE(1, 4)
  LPX_RFHP_CASE(1, 8)
#undef LPX_RFHP_CASE
  return hipErrorInvalidValue;
}

hipError_t rf_predict_heap_cpair_f32(const int* cpairs, const int* roots,
                                     const float* leaf_values, int n_trees,
                                     int depth, const float* X, float* out,
                                     int S, int F, int C, int silp,
                                     hipStream_t stream) {
  if (C > CMAX || depth < 2 || depth > 26 || F > 0xffff)
    return hipErrorInvalidValue;
  // silp 21/41 encode tree-ILP 2/4 (SILP stays 1)
  const int TILP = (silp == 21) ? 2 : (silp == 41 ? 4 : (silp == 81 ? 8 : 1));
  const int SILP = (silp == 2) ? 2 : (silp == 4 ? 4 : 1);
  const bool stage = F <= F_LDS_MAX;
  if (silp == 141 && stage) {
    // TILP4 at 128-thread blocks: halves the X-tile LDS (17 KiB) so 8
    // blocks fit per CU -> 8 waves/SIMD (the 43-VGPR kernel is LDS-
    // occupancy-limited at 256 threads: 34 KiB -> 4 blocks -> 4 waves)
    const size_t l128 = (size_t)128 * (F + 1) * 4;
    unsigned b128 = lpx_ceil_div(S, 128);
    if (b128 > 8192u) b128 = 8192u;
    rf_heap_cpair_kernel<true, 1, false, 4, 128>
        <<<dim3(b128), dim3(128), l128, stream>>>(
            reinterpret_cast<const i32x2*>(cpairs), roots, leaf_values,
            n_trees, depth, X, out, S, F, C);
    return hipGetLastError();
  }
  const size_t lds = stage ? (size_t)THREADS * SILP * (F + 1) * 4 : 0;
  if (lds > 160 * 1024) return hipErrorInvalidValue;
  unsigned blocks = lpx_ceil_div(S, THREADS * SILP);
  if (blocks > 4096u) blocks = 4096u;
  if (blocks == 0) blocks = 1;
  const auto* pp = reinterpret_cast<const i32x2*>(cpairs);
#define LPX_RFHC_CASE(I, T)                                                   \
  if (SILP == I && TILP == T) {                                               \
    if (stage)                                                                \
      rf_heap_cpair_kernel<true, I, false, T>                                 \
          <<<dim3(blocks), dim3(THREADS), lds, stream>>>(                     \
              pp, roots, leaf_values, n_trees, depth, X, out, S, F, C);       \
    else                                                                      \
      rf_heap_cpair_kernel<false, I, false, T>                                \
          <<<dim3(blocks), dim3(THREADS), 0, stream>>>(                       \
              pp, roots, leaf_values, n_trees, depth, X, out, S, F, C);       \
    return hipGetLastError();                                                 \
  }
  LPX_RFHC_CASE(1, 1)
  LPX_RFHC_CASE(2, 1)
  LPX_RFHC_CASE(4, 1)
  LPX_RFHC_CASE(1, 2)
  LPX_RFHC_CASE(1, 4)
  LPX_RFHC_CASE(1, 8)
#undef LPX_RFHC_CASE
  return hipErrorInvalidValue;
}

hipError_t rf_predict_heap_cpair_lv16_f32(const int* cpairs, const int* roots,
                                          const unsigned short* lv16,
                                          int n_trees, int depth,
                                          const float* X, float* out, int S,
                                          int F, int C, int silp,
                                          hipStream_t stream) {
  if (C > CMAX || depth < 2 || depth > 26 || F > 0xffff)
    return hipErrorInvalidValue;
  // silp 21/41 encode tree-ILP 2/4 (SILP stays 1)
  const int TILP = (silp == 21) ? 2 : (silp == 41 ? 4 : (silp == 81 ? 8 : 1));
  const int SILP = (silp == 2) ? 2 : (silp == 4 ? 4 : 1);
  const bool stage = F <= F_LDS_MAX;
  if (silp == 141 && stage) {  // 128-thread TILP4 (see cpair_f32 launcher)
    const size_t l128 = (size_t)128 * (F + 1) * 4;
    unsigned b128 = lpx_ceil_div(S, 128);
    if (b128 > 8192u) b128 = 8192u;
    rf_heap_cpair_kernel<true, 1, true, 4, 128>
        <<<dim3(b128), dim3(128), l128, stream>>>(
            reinterpret_cast<const i32x2*>(cpairs), roots,
            reinterpret_cast<const float*>(lv16), n_trees, depth, X, out, S,
            F, C);
    return hipGetLastError();
  }
  const size_t lds = stage ? (size_t)THREADS * SILP * (F + 1) * 4 : 0;
  if (lds > 160 * 1024) return hipErrorInvalidValue;
  unsigned blocks = lpx_ceil_div(S, THREADS * SILP);
  if (blocks > 4096u) blocks = 4096u;
  if (blocks == 0) blocks = 1;
  const auto* pp = reinterpret_cast<const i32x2*>(cpairs);
  const auto* lv = reinterpret_cast<const float*>(lv16);
#define LPX_RFHC16_CASE(I, T)                                                 \
  if (SILP == I && TILP == T) {                                               \
    if (stage)                                                                \
      rf_heap_cpair_kernel<true, I, true, T>                                  \
          <<<dim3(blocks), dim3(THREADS), lds, stream>>>(                     \
              pp, roots, lv, n_trees, depth, X, out, S, F, C);                \
    else                                                                      \
      rf_heap_cpair_kernel<false, I, true, T>                                 \
          <<<dim3(blocks), dim3(THREADS), 0, stream>>>(                       \
              pp, roots, lv, n_trees, depth, X, out, S, F, C);                \
    return hipGetLastError();                                                 \
  }
  LPX_RFHC16_CASE(1, 1)
  LPX_RFHC16_CASE(2, 1)
  LPX_RFHC16_CASE(4, 1)
  LPX_RFHC16_CASE(1, 2)
  LPX_RFHC16_CASE(1, 4)
  LPX_RFHC16_CASE(1, 8)
#undef LPX_RFHC16_CASE
  return hipErrorInvalidValue;
}

hipError_t rf_predict_heap2_f32(const int* nodes_heap, const float* leaf_values,
                                int n_trees, int depth, const float* X,
                                float* out, int S, int F, int C, int silp,
                                int top, hipStream_t stream) {
  if (C > CMAX || depth < 1 || depth > 26) return hipErrorInvalidValue;
  const int SILP = (silp == 2) ? 2 : (silp == 4 ? 4 : 1);
  const bool stage = F <= F_LDS_MAX;
  const size_t xbytes = stage ? (size_t)THREADS * SILP * (F + 1) * 4 : 0;
  if (top < 0) {
    // largest top with total LDS <= 78 KiB (2 blocks/CU)
    top = 0;
    while (top < depth &&
           xbytes + (size_t)n_trees * ((2u << top) - 1) * 8 <= 78 * 1024)
      ++top;
  }
  if (top > depth) top = depth;
  const size_t lds = xbytes + (size_t)n_trees * ((1u << top) - 1) * 8;
  if (lds > 160 * 1024 || top < 1)
    return rf_predict_heap_f32(nodes_heap, leaf_values, n_trees, depth, X, out,
                               S, F, C, silp, stream);
  unsigned blocks = lpx_ceil_div(S, THREADS * SILP);
  if (blocks > 2048u) blocks = 2048u;
  if (blocks == 0) blocks = 1;
  const auto* nh = reinterpret_cast<const i32x2*>(nodes_heap);
#define LPX_RFH2_CASE(I)                                                       \
  if (SILP == I) {                                                             \
    if (stage)                                                                 \
      rf_heap2_kernel<true, I><<<dim3(blocks), dim3(THREADS), lds, stream>>>(  \
          nh, leaf_values, n_trees, depth, top, X, out, S, F, C);              \
    else                                                                       \
      rf_heap2_kernel<false, I><<<dim3(blocks), dim3(THREADS), lds, stream>>>( \
          nh, leaf_values, n_trees, depth, top, X, out, S, F, C);              \
    return hipGetLastError();                                                  \
  }
  LPX_RFH2_CASE(1)
  LPX_RFH2_CASE(2)
  LPX_RFH2_CASE(4)
#undef LPX_RFH2_CASE
  return hipErrorInvalidValue;
}

hipError_t rf_predict_heap_f32(const int* nodes_heap, const float* leaf_values,
                               int n_trees, int depth, const float* X,
                               float* out, int S, int F, int C, int silp,
                               hipStream_t stream) {
  if (C > CMAX || depth < 1 || depth > 26) return hipErrorInvalidValue;
  const int SILP = (silp == 2) ? 2 : (silp == 4 ? 4 : 1);
  const bool stage = F <= F_LDS_MAX;
  // X tile covers THREADS*SILP samples when staged
  const size_t lds = stage ? (size_t)THREADS * SILP * (F + 1) * sizeof(float) : 0;
  if (lds > 160 * 1024) return hipErrorInvalidValue;
  unsigned blocks = lpx_ceil_div(S, THREADS * SILP);
  if (blocks > 4096u) blocks = 4096u;
  if (blocks == 0) blocks = 1;
  const auto* nh = reinterpret_cast<const i32x2*>(nodes_heap);
#define LPX_RFH_CASE(I)                                                       \
  if (SILP == I) {                                                            \
    if (stage)                                                                \
      rf_heap_kernel<true, I><<<dim3(blocks), dim3(THREADS), lds, stream>>>(  \
          nh, leaf_values, n_trees, depth, X, out, S, F, C);                  \
    else                                                                      \
      rf_heap_kernel<false, I><<<dim3(blocks), dim3(THREADS), 0, stream>>>(   \
          nh, leaf_values, n_trees, depth, X, out, S, F, C);                  \
    return hipGetLastError();                                                 \
  }
  LPX_RFH_CASE(1)
  LPX_RFH_CASE(2)
  LPX_RFH_CASE(4)
#undef LPX_RFH_CASE
  if (silp == 101 || silp == 102) {
    // A/B: 128-thread blocks (half the X-stage LDS -> 9-10 blocks/CU of
    // resident waves vs 4 at 256 threads)
    const size_t l2 = stage ? (size_t)128 * (F + 1) * sizeof(float) : 0;
    unsigned b2 = lpx_ceil_div(S, 128);
    if (b2 > 8192u) b2 = 8192u;
    if (silp == 101) {
      if (stage)
        rf_heap_kernel<true, 1, 128><<<dim3(b2), dim3(128), l2, stream>>>(
            nh, leaf_values, n_trees, depth, X, out, S, F, C);
      else
        rf_heap_kernel<false, 1, 128><<<dim3(b2), dim3(128), 0, stream>>>(
            nh, leaf_values, n_trees, depth, X, out, S, F, C);
    } else {
      b2 = lpx_ceil_div(S, 64);
      if (b2 > 16384u) b2 = 16384u;
      const size_t l3 = stage ? (size_t)64 * (F + 1) * sizeof(float) : 0;
      if (stage)
        rf_heap_kernel<true, 1, 64><<<dim3(b2), dim3(64), l3, stream>>>(
            nh, leaf_values, n_trees, depth, X, out, S, F, C);
      else
        rf_heap_kernel<false, 1, 64><<<dim3(b2), dim3(64), 0, stream>>>(
            nh, leaf_values, n_trees, depth, X, out, S, F, C);
    }
    return hipGetLastError();
  }
  return hipErrorInvalidValue;
}

hipError_t rf_predict8_lds_f32(const int* nodes8, const float* leaf_values,
                               const int* roots, const int* chunk_tree_begin,
                               int n_chunks, int max_chunk_nodes, int n_trees,
                               int total_nodes, const float* X, float* out,
                               int S, int F, int C, hipStream_t stream) {
  if (C > CMAX) return hipErrorInvalidValue;
  const bool stage = F <= F_LDS_MAX;
  const size_t xbytes = stage ? (size_t)THREADS * (F + 1) * sizeof(float) : 0;
  const size_t lds = (size_t)max_chunk_nodes * 8 + xbytes;
  if (lds > 160 * 1024) return hipErrorInvalidValue;
  unsigned blocks = lpx_ceil_div(S, THREADS);
  if (blocks > 256u) blocks = 256u;  // one block per CU (full-LDS kernel)
  if (blocks == 0) blocks = 1;
  const auto* n8 = reinterpret_cast<const i32x2*>(nodes8);
  if (stage)
    rf_lds_kernel<true, 2><<<dim3(blocks), dim3(THREADS), lds, stream>>>(
        n8, leaf_values, roots, chunk_tree_begin, n_chunks, n_trees,
        total_nodes, X, out, S, F, C, max_chunk_nodes);
  else
    rf_lds_kernel<false, 2><<<dim3(blocks), dim3(THREADS), lds, stream>>>(
        n8, leaf_values, roots, chunk_tree_begin, n_chunks, n_trees,
        total_nodes, X, out, S, F, C, max_chunk_nodes);
  return hipGetLastError();
}

hipError_t rf_predict8_f32(const int* nodes8, const float* leaf_values,
                           const int* roots, int n_trees, const float* X,
                           float* out, int S, int F, int C, int ilp,
                           hipStream_t stream) {
  if (C > CMAX) return hipErrorInvalidValue;
  unsigned blocks = lpx_ceil_div(S, THREADS);
  if (blocks > 4096u) blocks = 4096u;
  if (blocks == 0) blocks = 1;
  const bool stage = F <= F_LDS_MAX;
  const size_t lds = stage ? (size_t)THREADS * (F + 1) * sizeof(float) : 0;
  const auto* n8 = reinterpret_cast<const i32x2*>(nodes8);
#define LPX_RF8_CASE(I)                                                       \
  if (ilp == I) {                                                             \
    if (stage)                                                                \
      rf8_kernel<true, I><<<dim3(blocks), dim3(THREADS), lds, stream>>>(      \
          n8, leaf_values, roots, n_trees, X, out, S, F, C);                  \
    else                                                                      \
      rf8_kernel<false, I><<<dim3(blocks), dim3(THREADS), 0, stream>>>(       \
          n8, leaf_values, roots, n_trees, X, out, S, F, C);                  \
    return hipGetLastError();                                                 \
  }
  LPX_RF8_CASE(1)
  LPX_RF8_CASE(2)
  LPX_RF8_CASE(4)
#undef LPX_RF8_CASE
  return hipErrorInvalidValue;
}

hipError_t rf_predict_f32(const int* nodes4, const float* leaf_values,
                          const int* roots, int n_trees, const float* X,
                          float* out, int S, int F, int C,
                          hipStream_t stream) {
  if (C > CMAX) return hipErrorInvalidValue;  // Python wrapper chunks classes
  unsigned blocks = lpx_ceil_div(S, THREADS);
  if (blocks > 4096u) blocks = 4096u;
  if (blocks == 0) blocks = 1;
  const bool stage = F <= F_LDS_MAX;
  const size_t lds = stage ? (size_t)THREADS * (F + 1) * sizeof(float) : 0;
  const auto* n4 = reinterpret_cast<const i32x4*>(nodes4);
  if (stage)
    rf_kernel<true, 2><<<dim3(blocks), dim3(THREADS), lds, stream>>>(
        n4, leaf_values, roots, n_trees, X, out, S, F, C);
  else
    rf_kernel<false, 2><<<dim3(blocks), dim3(THREADS), 0, stream>>>(
        n4, leaf_values, roots, n_trees, X, out, S, F, C);
  return hipGetLastError();
}

// A/B entry for the ILP sweep (scripts/ab_rf.py)
hipError_t rf_predict_f32_ilp(const int* nodes4, const float* leaf_values,
                              const int* roots, int n_trees, const float* X,
                              float* out, int S, int F, int C, int ilp,
                              hipStream_t stream) {
  if (C > CMAX) return hipErrorInvalidValue;
  unsigned blocks = lpx_ceil_div(S, THREADS);
  if (blocks > 4096u) blocks = 4096u;
  if (blocks == 0) blocks = 1;
  const bool stage = F <= F_LDS_MAX;
  const size_t lds = stage ? (size_t)THREADS * (F + 1) * sizeof(float) : 0;
  const auto* n4 = reinterpret_cast<const i32x4*>(nodes4);
#define LPX_RF_CASE(I)                                                        \
  if (ilp == I) {                                                             \
    if (stage)                                                                \
      rf_kernel<true, I><<<dim3(blocks), dim3(THREADS), lds, stream>>>(       \
          n4, leaf_values, roots, n_trees, X, out, S, F, C);                  \
    else                                                                      \
      rf_kernel<false, I><<<dim3(blocks), dim3(THREADS), 0, stream>>>(        \
          n4, leaf_values, roots, n_trees, X, out, S, F, C);                  \
    return hipGetLastError();                                                 \
  }
  LPX_RF_CASE(1)
  LPX_RF_CASE(2)
  LPX_RF_CASE(4)
  LPX_RF_CASE(8)
#undef LPX_RF_CASE
  return hipErrorInvalidValue;
}

}  // namespace lpx
