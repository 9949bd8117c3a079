// lambdipy_amd ops — plain C++ launcher API (no torch headers here).
// Each launcher enqueues on the given hipStream_t and returns hipError_t.
#pragma once

#include <hip/hip_runtime.h>
#include <stdint.h>

namespace lpx {

// C[M,N] (f32) = A[M,K] (bf16, row-major) @ Bt[N,K]^T (bf16, row-major).
// NT layout: both operands K-contiguous (torch Linear weight layout).
hipError_t gemm_bf16_nt(const uint16_t* A, const uint16_t* Bt, float* C,
                        int M, int N, int K, hipStream_t stream);
// Same but C stored as bf16.
hipError_t gemm_bf16_nt_bf16out(const uint16_t* A, const uint16_t* Bt, uint16_t* C,
                                int M, int N, int K, hipStream_t stream);
// v3: glds-staged 8-phase pipeline (M,N%256==0, K%64==0).
bool gemm_bf16_8ph_supported(int M, int N, int K);
hipError_t gemm_bf16_nt8ph(const uint16_t* A, const uint16_t* Bt, float* C,
                           int M, int N, int K, hipStream_t stream);
hipError_t gemm_bf16_nt8ph_bf16out(const uint16_t* A, const uint16_t* Bt,
                                   uint16_t* C, int M, int N, int K,
                                   hipStream_t stream);
hipError_t gemm_bf16_nt8ph_tune(const uint16_t* A, const uint16_t* Bt,
                                float* C, int M, int N, int K, int tune,
                                hipStream_t stream);
// v2: 256x256/BK64 8-wave register-staged pipeline (large shapes).
hipError_t gemm_bf16_nt256(const uint16_t* A, const uint16_t* Bt, float* C,
                           int M, int N, int K, hipStream_t stream);
hipError_t gemm_bf16_nt256_bf16out(const uint16_t* A, const uint16_t* Bt,
                                   uint16_t* C, int M, int N, int K,
                                   hipStream_t stream);

// Batched GEMV: y[b,m] = sum_k A[b,m,k] * x[b,k].  bf16 in, f32 out.
hipError_t gemv_bf16(const uint16_t* A, const uint16_t* x, float* y,
                     int B, int M, int K, hipStream_t stream);
// f32 variant.
hipError_t gemv_f32(const float* A, const float* x, float* y,
                    int B, int M, int K, hipStream_t stream);

// CSR SpMV: y = A @ x (f32, int32 indices). mean_nnz picks the variant.
hipError_t spmv_csr_f32(const int* indptr, const int* indices, const float* data,
                        const float* x, float* y, int n_rows, long nnz,
                        hipStream_t stream);
// x-tile-in-LDS variant for column-SORTED rows with x too big to stay
// cache-hot (random 4-B gathers pull 64-B sectors; this streams x through
// LDS in 128-KiB column tiles instead). WRONG RESULTS on unsorted rows.
hipError_t spmv_csr_xtile_f32(const int* indptr, const int* indices,
                              const float* data, const float* x, float* y,
                              int n_rows, int n_cols, hipStream_t stream);

// Batched 1-D complex64 FFT, N = power of two (<= 4096), Stockham in LDS.
// x/out: [batch][N] interleaved (re,im) f32. twiddles: [N/2] complex
// (host-precomputed, exp(-2*pi*i*k/N)). inverse: conjugate + 1/N scale.
hipError_t fft_c64(const float* x, float* out, const float* twiddles,
                   int batch, int n, bool inverse, hipStream_t stream);

// Pairwise squared-euclidean: D[i,j] = ||X[i]-Y[j]||^2, f32 (exact-f32 MFMA).
// xsq/ysq: precomputed row norms (f32). X [M][K], Y [N][K], both row-major.
hipError_t pairwise_sqeuclidean_f32(const float* X, const float* Y,
                                    const float* xsq, const float* ysq,
                                    float* D, int M, int N, int K,
                                    hipStream_t stream);
// Row-wise squared norms: out[i] = sum_k X[i,k]^2.
hipError_t row_sqnorm_f32(const float* X, float* out, int M, int K,
                          hipStream_t stream);

// RandomForest inference over a SoA forest (16-B nodes packed in `nodes4`:
// {int feat; float thresh; int left; int right}, feat<0 => leaf and `left`
// indexes leaf_values[C]). X [S][F] f32 row-major. out [S][C] f32: per-class
// sums over trees (caller argmaxes / divides by n_trees).
hipError_t rf_predict_f32(const int* nodes4, const float* leaf_values,
                          const int* roots, int n_trees, const float* X,
                          float* out, int S, int F, int C, hipStream_t stream);
// Packed 8-B depth-first nodes (v2): meta = feat<<21 | right_delta; left
// child implicit (nd+1); leaf feat=0x7FF, val = leaf row. Halves L2 traffic.
hipError_t rf_predict8_f32(const int* nodes8, const float* leaf_values,
                           const int* roots, int n_trees, const float* X,
                           float* out, int S, int F, int C, int ilp,
                           hipStream_t stream);
// Heap-layout traversal (v4): complete equal-depth forests, branchless
// fixed-depth loop; nodes_heap = internal nodes only, level-order per tree.
hipError_t rf_predict_heap_f32(const int* nodes_heap, const float* leaf_values,
                               int n_trees, int depth, const float* X,
                               float* out, int S, int F, int C, int silp,
                               hipStream_t stream);
// v7: quad-record heap — quads = int32 [n_trees*(2^depth-1)/3, 16]
// (64-B records: children + grandchildren of each even-level node); two
// levels per random fetch. Even depth only.
hipError_t rf_predict_heap_quad_f32(const int* quads, const int* roots,
                                    const float* leaf_values, int n_trees,
                                    int depth, const float* X, float* out,
                                    int S, int F, int C, int silp,
                                    hipStream_t stream);
// v6: child-pair heap — pairs = int32 [n_trees*(2^(depth-1)-1), 4] (node n's
// two children as one 16-B record), roots = int32 [n_trees, 2]; one random
// 16-B request per level instead of one 8-B (the v4 kernel is L2
// request-bound).
hipError_t rf_predict_heap_pair_f32(const int* pairs, const int* roots,
                                    const float* leaf_values, int n_trees,
                                    int depth, const float* X, float* out,
                                    int S, int F, int C, int silp,
                                    hipStream_t stream);
// v8: compact child-pair heap — 4-B nodes (u16 feat << 16 | fp16-thr bits),
// cpairs int32 [n_trees*(2^(depth-1)-1), 2], roots int32 [n_trees]; halves
// the bytes of every random L2 request vs v6. fp16 thresholds: exact for
// fp16-representable forests, else RNE-rounded decisions (see rf.hip).
hipError_t rf_predict_heap_cpair_f32(const int* cpairs, const int* roots,
                                     const float* leaf_values, int n_trees,
                                     int depth, const float* X, float* out,
                                     int S, int F, int C, int silp,
                                     hipStream_t stream);
// fp16 leaf table variant (lv16 = fp16 bits [rows, C], C <= 8): one 16-B
// leaf request; exact when leaves are fp16-representable.
hipError_t rf_predict_heap_cpair_lv16_f32(const int* cpairs, const int* roots,
                                          const unsigned short* lv16,
                                          int n_trees, int depth,
                                          const float* X, float* out, int S,
                                          int F, int C, int silp,
                                          hipStream_t stream);
// v5: two-level heap — top `top` levels served from LDS (top<0: auto-pick
// largest fitting 2-blocks/CU; falls back to v4 when nothing fits).
hipError_t rf_predict_heap2_f32(const int* nodes_heap, const float* leaf_values,
                                int n_trees, int depth, const float* X,
                                float* out, int S, int F, int C, int silp,
                                int top, hipStream_t stream);
// Forest-in-LDS chunked traversal (v3): whole-tree chunks staged in LDS
// (host-computed chunk_tree_begin[n_chunks+1]); out accumulates across
// chunks. One block per CU.
hipError_t rf_predict8_lds_f32(const int* nodes8, const float* leaf_values,
                               const int* roots, const int* chunk_tree_begin,
                               int n_chunks, int max_chunk_nodes, int n_trees,
                               int total_nodes, const float* X, float* out,
                               int S, int F, int C, hipStream_t stream);
// ILP-sweep entry (ilp in {1,2,4,8}) for perf A/B.
hipError_t rf_predict_f32_ilp(const int* nodes4, const float* leaf_values,
                              const int* roots, int n_trees, const float* X,
                              float* out, int S, int F, int C, int ilp,
                              hipStream_t stream);

// ---- fp8 (OCP e4m3) path — config 5 -------------------------------------
// f32 -> e4m3 with inv_scale premultiply (per-tensor scaling).
hipError_t quantize_fp8(const float* x, unsigned char* out, long n,
                        float inv_scale, hipStream_t stream);
// NHWC [N,H,W,C] f32 -> [N*HO*WO, R*S*C] e4m3 patch matrix (im2col+quantize).
hipError_t im2col_fp8(const float* x, unsigned char* out, int N, int H, int W,
                      int C, int R, int S, int HO, int WO, int stride, int pad,
                      float inv_scale, hipStream_t stream);
// C[M,N] f32 = descale * (A[M,K] @ Bt[N,K]^T) (+bias) (relu) — fp8 MFMA.
hipError_t gemm_fp8_nt(const unsigned char* A, const unsigned char* Bt,
                       float* C, const float* bias, int M, int N, int K,
                       float descale, bool relu, hipStream_t stream);
// resid != nullptr: fused `C += resid` before relu (bottleneck identity add)
hipError_t gemm_fp8_nt_res(const unsigned char* A, const unsigned char* Bt,
                           float* C, const float* bias, const float* resid,
                           int M, int N, int K, float descale, bool relu,
                           hipStream_t stream);
// Implicit-GEMM fp8 conv: xq NHWC e4m3 (pre-quantized), Wq [KO, R*S*C];
// out f32 [N*HO*WO, KO]; requires C % 16 == 0.
hipError_t conv_fp8_implicit(const unsigned char* xq, const unsigned char* Wq,
                             float* C, const float* bias, int Nb, int H, int W,
                             int Cin, int R, int S, int HO, int WO, int stride,
                             int pad, int KO, float descale, bool relu,
                             hipStream_t stream);
// fp8 v3: glds 8-phase pipeline (M,N%256==0, K%128==0).
bool gemm_fp8_8ph_supported(int M, int N, int K);
hipError_t gemm_fp8_nt8ph(const unsigned char* A, const unsigned char* Bt,
                          float* C, const float* bias, int M, int N, int K,
                          float descale, bool relu, hipStream_t stream);

// ---- MX block-scaled fp8 (v_mfma_scale_f32_32x32x64_f8f6f4) --------------
// x f32 [R,K] (K%64==0) -> q e4m3 bytes [R,K] in MFMA-shuffled order
// ([q0|q2|q1|q3] per 64-group) + sc E8M0 [R, K/64, 2] block scales.
hipError_t quantize_mx_fp8(const float* x, unsigned char* q, unsigned char* sc,
                           int R, int K, hipStream_t stream);
// C f32 [M,N] = dequant(A) @ dequant(Bt)^T — scales applied in hardware.
hipError_t gemm_mx_fp8_nt(const unsigned char* A, const unsigned char* sA,
                          const unsigned char* Bt, const unsigned char* sB,
                          float* C, int M, int N, int K, hipStream_t stream);
// v2: glds 8-phase pipeline (M,N%256==0, K%128==0), 2x MFMA rate.
bool gemm_mx_8ph_supported(int M, int N, int K);
hipError_t gemm_mx_fp8_nt8ph(const unsigned char* A, const unsigned char* sA,
                             const unsigned char* Bt, const unsigned char* sB,
                             float* C, int M, int N, int K,
                             hipStream_t stream);
// per-tensor-scale fp8 GEMM on the scaled MFMA with unit block scales
// (same shape constraints as gemm_mx_8ph_supported).
hipError_t gemm_fp8_nt8phx(const unsigned char* A, const unsigned char* Bt,
                           float* C, const float* bias, int M, int N, int K,
                           float descale, bool relu, hipStream_t stream);
hipError_t gemm_fp8_nt8phx_res(const unsigned char* A, const unsigned char* Bt,
                               float* C, const float* bias, const float* resid,
                               int M, int N, int K, float descale, bool relu,
                               hipStream_t stream);
// fused epilogue quantize: out = e4m3((descale*acc + bias, relu) * out_inv)
hipError_t gemm_fp8_nt8phx_q8(const unsigned char* A, const unsigned char* Bt,
                              unsigned char* C8, const float* bias, int M,
                              int N, int K, float descale, bool relu,
                              float out_inv, hipStream_t stream);
// 8-phase glds implicit conv (scaled MFMA, zero-page padding redirect);
// M%256, KO%256, K=R*S*C%128, C%16. C8 != nullptr => fused e4m3 out.
bool conv_fp8_8ph_supported(int M, int KO, int K, int Cin);
hipError_t conv_fp8_implicit_8ph(const unsigned char* xq,
                                 const unsigned char* Wq, float* C,
                                 unsigned char* C8, const float* bias, int Nb,
                                 int H, int W, int Cin, int R, int S, int HO,
                                 int WO, int stride, int pad, int KO,
                                 float descale, bool relu, float out_inv,
                                 hipStream_t stream);
hipError_t conv_fp8_implicit_q8(const unsigned char* xq,
                                const unsigned char* Wq, unsigned char* C8,
                                const float* bias, int Nb, int H, int W,
                                int Cin, int R, int S, int HO, int WO,
                                int stride, int pad, int KO, float descale,
                                bool relu, float out_inv, hipStream_t stream);

// NHWC f32 max-pool (k<=8, C%4==0) — avoids the NCHW permute round trip.
hipError_t maxpool2d_nhwc_f32(const float* x, float* out, int N, int H, int W,
                              int C, int k, int stride, int pad,
                              hipStream_t stream);

// ---- MX block-scaled fp4 (e2m1, 2x the fp8 MFMA rate) --------------------
// x f32 [R,K] (K%64==0) -> q nibbles [R,K/2] (shuffled) + sc [R,K/64,2].
hipError_t quantize_mx_fp4(const float* x, unsigned char* q, unsigned char* sc,
                           int R, int K, hipStream_t stream);
bool gemm_mx_fp4_supported(int M, int N, int K);
// mixed: fp8-MX activations x fp4-MX weights at the full fp8 rate with
// half the weight bytes (probe: 4.96 PF).
bool gemm_mx_w4_supported(int M, int N, int K);
hipError_t gemm_mx_w4_nt8ph(const unsigned char* A, const unsigned char* sA,
                            const unsigned char* Bt, const unsigned char* sB,
                            float* C, int M, int N, int K,
                            hipStream_t stream);
hipError_t gemm_mx_fp4_nt8ph(const unsigned char* A, const unsigned char* sA,
                             const unsigned char* Bt, const unsigned char* sB,
                             float* C, int M, int N, int K,
                             hipStream_t stream);

// ---- torch-free serving utilities (nn_extra.hip; liblpx_nn C API) -------
// Device randn (xorshift128 + Box-Muller), scaled: out[i] ~ N(0, scale^2).
hipError_t randn_f32(float* out, long n, unsigned seed, float scale,
                     hipStream_t stream);
// Interleaved complex64 with re ~ N(0,1), im = 0.
hipError_t randn_c64(float* out, long n_complex, unsigned seed,
                     hipStream_t stream);
// Abs-max via atomicMax on float bits; caller zeroes out_bits first.
hipError_t amax_f32(const float* x, unsigned* out_bits, long n,
                    hipStream_t stream);
hipError_t checksum_f32(const float* x, double* out, long n,
                        hipStream_t stream);
// NHWC global average pool: [N, HW, C] -> [N, C], C % 4 == 0.
hipError_t gap_nhwc_f32(const float* x, float* out, int N, int HW, int C,
                        hipStream_t stream);
hipError_t add_relu_f32(const float* a, const float* b, float* out, long n,
                        hipStream_t stream);
hipError_t argmax_rows_f32(const float* x, int* out, long rows, int cols,
                           hipStream_t stream);
hipError_t f32_to_bf16(const float* x, uint16_t* out, long n,
                       hipStream_t stream);
// out[k] = (cos, sin)(-2 pi k / n) for k < n/2 (fft_c64's twiddle table).
hipError_t fft_twiddles(float* out, int n, hipStream_t stream);
// Synthetic fixed-nnz CSR (sorted random columns, N(0,1) data), nnz <= 64.
hipError_t csrgen_fixed(int* indptr, int* indices, float* data, int rows,
                        int nnz, int ncols, unsigned seed, hipStream_t stream);

}  // namespace lpx
