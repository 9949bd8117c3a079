// liblpx_nn C API — the torch-free serving runtime's entire host surface.
//
// Built by builder/nnlib.py with hipcc into liblpx_nn_gfx950.so (AOT, no
// JIT at cold start) and loaded via ctypes (shims/lpx_nn.py). Wraps the
// same lpx:: launchers the torch extension uses (kernels.h) — one kernel
// code base, two front ends (config 5's runtime
// with "no multi-backend dispatch"; this is a binding, not a dispatcher).
// Every function returns hipError_t as int; pointers are device pointers.

#include <hip/hip_runtime.h>

#include "kernels.h"

#define API extern "C" __attribute__((visibility("default")))

// ---- streams / graphs ----------------------------------------------------

API int lpxn_stream_create(void** s) {
  return (int)hipStreamCreateWithFlags(reinterpret_cast<hipStream_t*>(s),
                                       hipStreamNonBlocking);
}

API int lpxn_stream_destroy(void* s) {
  return (int)hipStreamDestroy((hipStream_t)s);
}

API int lpxn_stream_sync(void* s) {
  return (int)hipStreamSynchronize((hipStream_t)s);
}

API int lpxn_memset(void* p, int value, size_t n, void* s) {
  return (int)hipMemsetAsync(p, value, n, (hipStream_t)s);
}

// hipGraph capture of everything launched on `s` between begin/end; end
// instantiates and returns the executable graph (the ~120-launch ResNet
// forward replays as ONE launch afterwards).
API int lpxn_graph_begin(void* s) {
  return (int)hipStreamBeginCapture((hipStream_t)s,
                                    hipStreamCaptureModeThreadLocal);
}

API int lpxn_graph_end(void* s, void** exec_out) {
  hipGraph_t graph = nullptr;
  hipError_t e = hipStreamEndCapture((hipStream_t)s, &graph);
  if (e != hipSuccess) return (int)e;
  hipGraphExec_t exec = nullptr;
  e = hipGraphInstantiate(&exec, graph, nullptr, nullptr, 0);
  hipGraphDestroy(graph);
  if (e != hipSuccess) return (int)e;
  *exec_out = (void*)exec;
  return 0;
}

API int lpxn_graph_launch(void* exec, void* s) {
  return (int)hipGraphLaunch((hipGraphExec_t)exec, (hipStream_t)s);
}

API int lpxn_graph_destroy(void* exec) {
  return (int)hipGraphExecDestroy((hipGraphExec_t)exec);
}

// ---- generators / reductions / elementwise -------------------------------

API int lpxn_randn(void* out, long n, unsigned seed, float scale, void* s) {
  return (int)lpx::randn_f32((float*)out, n, seed, scale, (hipStream_t)s);
}

API int lpxn_randn_c64(void* out, long n_complex, unsigned seed, void* s) {
  return (int)lpx::randn_c64((float*)out, n_complex, seed, (hipStream_t)s);
}

API int lpxn_amax(const void* x, void* out_bits, long n, void* s) {
  return (int)lpx::amax_f32((const float*)x, (unsigned*)out_bits, n,
                            (hipStream_t)s);
}

API int lpxn_checksum_f32(const void* x, void* out, long n, void* s) {
  return (int)lpx::checksum_f32((const float*)x, (double*)out, n,
                                (hipStream_t)s);
}

API int lpxn_gap_nhwc(const void* x, void* out, int N, int HW, int C,
                      void* s) {
  return (int)lpx::gap_nhwc_f32((const float*)x, (float*)out, N, HW, C,
                                (hipStream_t)s);
}

API int lpxn_add_relu(const void* a, const void* b, void* out, long n,
                      void* s) {
  return (int)lpx::add_relu_f32((const float*)a, (const float*)b, (float*)out,
                                n, (hipStream_t)s);
}

API int lpxn_argmax_rows(const void* x, void* out, long rows, int cols,
                         void* s) {
  return (int)lpx::argmax_rows_f32((const float*)x, (int*)out, rows, cols,
                                   (hipStream_t)s);
}

API int lpxn_f32_to_bf16(const void* x, void* out, long n, void* s) {
  return (int)lpx::f32_to_bf16((const float*)x, (uint16_t*)out, n,
                               (hipStream_t)s);
}

// ---- fp8 path (config 5) -------------------------------------------------

API int lpxn_quantize_fp8(const void* x, void* out, long n, float inv_scale,
                          void* s) {
  return (int)lpx::quantize_fp8((const float*)x, (unsigned char*)out, n,
                                inv_scale, (hipStream_t)s);
}

API int lpxn_im2col_fp8(const void* x, void* out, int N, int H, int W, int C,
                        int R, int S, int HO, int WO, int stride, int pad,
                        float inv_scale, void* s) {
  return (int)lpx::im2col_fp8((const float*)x, (unsigned char*)out, N, H, W,
                              C, R, S, HO, WO, stride, pad, inv_scale,
                              (hipStream_t)s);
}

API int lpxn_gemm_fp8(const void* A, const void* Bt, void* C,
                      const void* bias, const void* resid, int M, int N,
                      int K, float descale, int relu, void* s) {
  return (int)lpx::gemm_fp8_nt_res(
      (const unsigned char*)A, (const unsigned char*)Bt, (float*)C,
      (const float*)bias, (const float*)resid, M, N, K, descale, relu != 0,
      (hipStream_t)s);
}

// fused e4m3 output when the 8-phase shape holds; returns
// hipErrorInvalidValue otherwise (caller falls back to f32 + quantize).
API int lpxn_gemm_fp8_q8(const void* A, const void* Bt, void* C8,
                         const void* bias, int M, int N, int K, float descale,
                         int relu, float out_inv, void* s) {
  if (!(M % 256 == 0 && N % 256 == 0 && K % 128 == 0 && K >= 128))
    return (int)hipErrorInvalidValue;
  return (int)lpx::gemm_fp8_nt8phx_q8(
      (const unsigned char*)A, (const unsigned char*)Bt, (unsigned char*)C8,
      (const float*)bias, M, N, K, descale, relu != 0, out_inv,
      (hipStream_t)s);
}

API int lpxn_conv_fp8(const void* xq, const void* Wq, void* C,
                      const void* bias, int Nb, int H, int W, int Cin, int R,
                      int S, int HO, int WO, int stride, int pad, int KO,
                      float descale, int relu, void* s) {
  return (int)lpx::conv_fp8_implicit(
      (const unsigned char*)xq, (const unsigned char*)Wq, (float*)C,
      (const float*)bias, Nb, H, W, Cin, R, S, HO, WO, stride, pad, KO,
      descale, relu != 0, (hipStream_t)s);
}

API int lpxn_conv_fp8_q8(const void* xq, const void* Wq, void* C8,
                         const void* bias, int Nb, int H, int W, int Cin,
                         int R, int S, int HO, int WO, int stride, int pad,
                         int KO, float descale, int relu, float out_inv,
                         void* s) {
  return (int)lpx::conv_fp8_implicit_q8(
      (const unsigned char*)xq, (const unsigned char*)Wq, (unsigned char*)C8,
      (const float*)bias, Nb, H, W, Cin, R, S, HO, WO, stride, pad, KO,
      descale, relu != 0, out_inv, (hipStream_t)s);
}

API int lpxn_maxpool_nhwc(const void* x, void* out, int N, int H, int W,
                          int C, int k, int stride, int pad, void* s) {
  return (int)lpx::maxpool2d_nhwc_f32((const float*)x, (float*)out, N, H, W,
                                      C, k, stride, pad, (hipStream_t)s);
}

// ---- bf16 GEMM (config 2) ------------------------------------------------

API int lpxn_gemm_bf16(const void* A, const void* Bt, void* C, int M, int N,
                       int K, void* s) {
  return (int)lpx::gemm_bf16_nt((const uint16_t*)A, (const uint16_t*)Bt,
                                (float*)C, M, N, K, (hipStream_t)s);
}

API int lpxn_gemm_bf16_bf16out(const void* A, const void* Bt, void* C, int M,
                               int N, int K, void* s) {
  return (int)lpx::gemm_bf16_nt_bf16out((const uint16_t*)A,
                                        (const uint16_t*)Bt, (uint16_t*)C, M,
                                        N, K, (hipStream_t)s);
}

// ---- FFT + SpMV (config 3) -----------------------------------------------

API int lpxn_fft_twiddles(void* out, int n, void* s) {
  return (int)lpx::fft_twiddles((float*)out, n, (hipStream_t)s);
}

API int lpxn_fft_c64(const void* x, void* out, const void* twiddles,
                     int batch, int n, int inverse, void* s) {
  return (int)lpx::fft_c64((const float*)x, (float*)out,
                           (const float*)twiddles, batch, n, inverse != 0,
                           (hipStream_t)s);
}

API int lpxn_csrgen(void* indptr, void* indices, void* data, int rows,
                    int nnz, int ncols, unsigned seed, void* s) {
  return (int)lpx::csrgen_fixed((int*)indptr, (int*)indices, (float*)data,
                                rows, nnz, ncols, seed, (hipStream_t)s);
}

API int lpxn_spmv_csr(const void* indptr, const void* indices,
                      const void* data, const void* x, void* y, int n_rows,
                      long nnz, void* s) {
  return (int)lpx::spmv_csr_f32((const int*)indptr, (const int*)indices,
                                (const float*)data, (const float*)x,
                                (float*)y, n_rows, nnz, (hipStream_t)s);
}

// ---- GEMV / pairwise (sklearn aux) ---------------------------------------

API int lpxn_gemv_f32(const void* A, const void* x, void* y, int B, int M,
                      int K, void* s) {
  return (int)lpx::gemv_f32((const float*)A, (const float*)x, (float*)y, B,
                            M, K, (hipStream_t)s);
}

API int lpxn_pairwise(const void* X, const void* Y, const void* xsq,
                      const void* ysq, void* D, int M, int N, int K,
                      void* s) {
  return (int)lpx::pairwise_sqeuclidean_f32(
      (const float*)X, (const float*)Y, (const float*)xsq, (const float*)ysq,
      (float*)D, M, N, K, (hipStream_t)s);
}

API int lpxn_row_sqnorm(const void* X, void* out, int M, int K, void* s) {
  return (int)lpx::row_sqnorm_f32((const float*)X, (float*)out, M, K,
                                  (hipStream_t)s);
}

// error-string passthrough for the ctypes side
API const char* lpxn_error_string(int e) {
  return hipGetErrorString((hipError_t)e);
}
