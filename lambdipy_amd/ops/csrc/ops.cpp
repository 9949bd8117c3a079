// lambdipy_amd ops — PyTorch-ROCm bindings for the gfx950 kernel library.
//
// ROCm-native: streams come from c10::hip (no CUDA-named API, nothing for
// hipify to rewrite — BASELINE.json forbids CUDA-compat shims/dual paths).
// Tensors live on the ROCm device (torch reports it as "cuda").

#include <torch/extension.h>

#include <ATen/hip/impl/HIPGuardImplMasqueradingAsCUDA.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

#include "kernels.h"

namespace {

#define CHECK_DEV(x)                                                        \
  TORCH_CHECK((x).is_cuda(), #x " must be on the ROCm device (got CPU)");   \
  TORCH_CHECK((x).is_contiguous(), #x " must be contiguous")

hipStream_t current_stream(const at::Tensor& t) {
  return c10::hip::getCurrentHIPStreamMasqueradingAsCUDA(t.device().index()).stream();
}

void check_hip(hipError_t e, const char* what) {
  TORCH_CHECK(e == hipSuccess, what, " failed: ", hipGetErrorString(e));
}

// ---- GEMM ----------------------------------------------------------------

at::Tensor gemm_bf16(const at::Tensor& A, const at::Tensor& Bt, bool bf16_out,
                     int64_t variant) {
  CHECK_DEV(A);
  CHECK_DEV(Bt);
  TORCH_CHECK(A.scalar_type() == at::kBFloat16 && Bt.scalar_type() == at::kBFloat16,
              "gemm_bf16 expects bf16 operands");
  TORCH_CHECK(A.dim() == 2 && Bt.dim() == 2, "gemm_bf16 expects 2-D operands");
  const int64_t M = A.size(0), K = A.size(1), N = Bt.size(0);
  TORCH_CHECK(Bt.size(1) == K, "gemm_bf16: A is [M,K], Bt must be [N,K] (NT layout)");
  c10::hip::HIPGuardMasqueradingAsCUDA guard(A.device());
  auto C = at::empty({M, N}, A.options().dtype(bf16_out ? at::kBFloat16 : at::kFloat));
  const auto* a = reinterpret_cast<const uint16_t*>(A.data_ptr());
  const auto* b = reinterpret_cast<const uint16_t*>(Bt.data_ptr());
  // variant: -1 auto, 1 = 128^2, 2 = 256^2 reg-staged, 3 = 256^2 8-phase glds
  auto s = current_stream(A);
  const bool want8 = (variant == 3) ||
                     (variant < 0 && M >= 1024 && N >= 1024 &&
                      lpx::gemm_bf16_8ph_supported((int)M, (int)N, (int)K));
  if (want8 && lpx::gemm_bf16_8ph_supported((int)M, (int)N, (int)K)) {
    if (bf16_out)
      check_hip(lpx::gemm_bf16_nt8ph_bf16out(a, b, reinterpret_cast<uint16_t*>(C.data_ptr()),
                                             (int)M, (int)N, (int)K, s),
                "gemm_bf16_nt8ph");
    else
      check_hip(lpx::gemm_bf16_nt8ph(a, b, C.data_ptr<float>(), (int)M, (int)N,
                                     (int)K, s),
                "gemm_bf16_nt8ph");
    return C;
  }
  const bool big = (variant == 2) ||
                   (variant < 0 && M >= 512 && N >= 512 && K >= 64);
  if (big) {
    if (bf16_out)
      check_hip(lpx::gemm_bf16_nt256_bf16out(a, b, reinterpret_cast<uint16_t*>(C.data_ptr()),
                                             (int)M, (int)N, (int)K, s),
                "gemm_bf16_nt256");
    else
      check_hip(lpx::gemm_bf16_nt256(a, b, C.data_ptr<float>(), (int)M, (int)N,
                                     (int)K, s),
                "gemm_bf16_nt256");
  } else if (bf16_out) {
    check_hip(lpx::gemm_bf16_nt_bf16out(a, b, reinterpret_cast<uint16_t*>(C.data_ptr()),
                                        (int)M, (int)N, (int)K, s),
              "gemm_bf16_nt");
  } else {
    check_hip(lpx::gemm_bf16_nt(a, b, C.data_ptr<float>(), (int)M, (int)N, (int)K, s),
              "gemm_bf16_nt");
  }
  return C;
}

// ---- GEMV ----------------------------------------------------------------

at::Tensor gemv(const at::Tensor& A_in, const at::Tensor& x_in) {
  auto A = A_in, x = x_in;
  CHECK_DEV(A);
  CHECK_DEV(x);
  bool squeeze = false;
  if (A.dim() == 2) {  // [M,K] @ [K]
    A = A.unsqueeze(0);
    x = x.unsqueeze(0);
    squeeze = true;
  }
  TORCH_CHECK(A.dim() == 3 && x.dim() == 2, "gemv expects A [B,M,K], x [B,K]");
  const int64_t B = A.size(0), M = A.size(1), K = A.size(2);
  TORCH_CHECK(x.size(0) == B && x.size(1) == K, "gemv shape mismatch");
  c10::hip::HIPGuardMasqueradingAsCUDA guard(A.device());
  auto y = at::empty({B, M}, A.options().dtype(at::kFloat));
  if (A.scalar_type() == at::kBFloat16) {
    TORCH_CHECK(x.scalar_type() == at::kBFloat16, "gemv: dtype mismatch");
    check_hip(lpx::gemv_bf16(reinterpret_cast<const uint16_t*>(A.data_ptr()),
                             reinterpret_cast<const uint16_t*>(x.data_ptr()),
                             y.data_ptr<float>(), (int)B, (int)M, (int)K,
                             current_stream(A)),
              "gemv_bf16");
  } else if (A.scalar_type() == at::kFloat) {
    TORCH_CHECK(x.scalar_type() == at::kFloat, "gemv: dtype mismatch");
    check_hip(lpx::gemv_f32(A.data_ptr<float>(), x.data_ptr<float>(),
                            y.data_ptr<float>(), (int)B, (int)M, (int)K,
                            current_stream(A)),
              "gemv_f32");
  } else {
    TORCH_CHECK(false, "gemv supports bf16 and f32");
  }
  return squeeze ? y.squeeze(0) : y;
}

// ---- CSR SpMV ------------------------------------------------------------

at::Tensor spmv_csr(const at::Tensor& indptr, const at::Tensor& indices,
                    const at::Tensor& data, const at::Tensor& x,
                    int64_t variant) {
  CHECK_DEV(indptr);
  CHECK_DEV(indices);
  CHECK_DEV(data);
  CHECK_DEV(x);
  TORCH_CHECK(indptr.scalar_type() == at::kInt && indices.scalar_type() == at::kInt,
              "spmv_csr expects int32 indptr/indices");
  TORCH_CHECK(data.scalar_type() == at::kFloat && x.scalar_type() == at::kFloat,
              "spmv_csr expects f32 data/x");
  const int64_t n_rows = indptr.numel() - 1;
  c10::hip::HIPGuardMasqueradingAsCUDA guard(x.device());
  auto y = at::empty({n_rows}, x.options());
  // variant 2: x-tile-in-LDS (requires CSR-canonical sorted columns —
  // scipy guarantees this; the caller opts in because sortedness is not
  // checked here). -1/0: adaptive seg8/wave-row.
  if (variant == 2) {
    check_hip(lpx::spmv_csr_xtile_f32(
                  indptr.data_ptr<int>(), indices.data_ptr<int>(),
                  data.data_ptr<float>(), x.data_ptr<float>(),
                  y.data_ptr<float>(), (int)n_rows, (int)x.numel(),
                  current_stream(x)),
              "spmv_csr_xtile_f32");
    return y;
  }
  check_hip(lpx::spmv_csr_f32(indptr.data_ptr<int>(), indices.data_ptr<int>(),
                              data.data_ptr<float>(), x.data_ptr<float>(),
                              y.data_ptr<float>(), (int)n_rows,
                              (long)data.numel(), current_stream(x)),
            "spmv_csr_f32");
  return y;
}

// ---- FFT -----------------------------------------------------------------

at::Tensor fft_c64(const at::Tensor& x, const at::Tensor& twiddles, bool inverse) {
  CHECK_DEV(x);
  CHECK_DEV(twiddles);
  TORCH_CHECK(x.scalar_type() == at::kComplexFloat, "fft_c64 expects complex64");
  TORCH_CHECK(twiddles.scalar_type() == at::kComplexFloat, "twiddles must be complex64");
  const int64_t n = x.size(-1);
  TORCH_CHECK(n >= 1 && (n & (n - 1)) == 0, "fft_c64: N must be a power of two");
  TORCH_CHECK(twiddles.numel() >= n / 2, "twiddle table too small");
  const int64_t batch = x.numel() / n;
  c10::hip::HIPGuardMasqueradingAsCUDA guard(x.device());
  auto out = at::empty_like(x);
  check_hip(lpx::fft_c64(reinterpret_cast<const float*>(x.data_ptr()),
                         reinterpret_cast<float*>(out.data_ptr()),
                         reinterpret_cast<const float*>(twiddles.data_ptr()),
                         (int)batch, (int)n, inverse, current_stream(x)),
            "fft_c64");
  return out;
}

// ---- pairwise ------------------------------------------------------------

at::Tensor pairwise_sqeuclidean(const at::Tensor& X, const at::Tensor& Y) {
  CHECK_DEV(X);
  CHECK_DEV(Y);
  TORCH_CHECK(X.scalar_type() == at::kFloat && Y.scalar_type() == at::kFloat,
              "pairwise_sqeuclidean expects f32");
  TORCH_CHECK(X.dim() == 2 && Y.dim() == 2 && X.size(1) == Y.size(1),
              "pairwise_sqeuclidean expects [M,K], [N,K]");
  const int64_t M = X.size(0), N = Y.size(0), K = X.size(1);
  c10::hip::HIPGuardMasqueradingAsCUDA guard(X.device());
  auto xsq = at::empty({M}, X.options());
  auto ysq = at::empty({N}, Y.options());
  auto D = at::empty({M, N}, X.options());
  auto s = current_stream(X);
  check_hip(lpx::row_sqnorm_f32(X.data_ptr<float>(), xsq.data_ptr<float>(), (int)M,
                                (int)K, s),
            "row_sqnorm_f32");
  check_hip(lpx::row_sqnorm_f32(Y.data_ptr<float>(), ysq.data_ptr<float>(), (int)N,
                                (int)K, s),
            "row_sqnorm_f32");
  check_hip(lpx::pairwise_sqeuclidean_f32(X.data_ptr<float>(), Y.data_ptr<float>(),
                                          xsq.data_ptr<float>(), ysq.data_ptr<float>(),
                                          D.data_ptr<float>(), (int)M, (int)N, (int)K, s),
            "pairwise_sqeuclidean_f32");
  return D;
}

// ---- RandomForest --------------------------------------------------------

at::Tensor rf_predict(const at::Tensor& nodes, const at::Tensor& leaf_values,
                      const at::Tensor& roots, const at::Tensor& X,
                      int64_t ilp) {
  CHECK_DEV(nodes);
  CHECK_DEV(leaf_values);
  CHECK_DEV(roots);
  CHECK_DEV(X);
  TORCH_CHECK(nodes.scalar_type() == at::kInt && nodes.dim() == 2 && nodes.size(1) == 4,
              "nodes must be int32 [n_nodes, 4]");
  TORCH_CHECK(leaf_values.scalar_type() == at::kFloat && leaf_values.dim() == 2,
              "leaf_values must be f32 [n_leaf_rows, C]");
  TORCH_CHECK(roots.scalar_type() == at::kInt, "roots must be int32");
  TORCH_CHECK(X.scalar_type() == at::kFloat && X.dim() == 2, "X must be f32 [S,F]");
  const int64_t S = X.size(0), F = X.size(1), C = leaf_values.size(1);
  TORCH_CHECK(C <= 8, "rf_predict kernel handles C<=8 per call (wrapper chunks)");
  c10::hip::HIPGuardMasqueradingAsCUDA guard(X.device());
  auto out = at::empty({S, C}, X.options());
  if (ilp <= 0) {
    check_hip(lpx::rf_predict_f32(nodes.data_ptr<int>(), leaf_values.data_ptr<float>(),
                                  roots.data_ptr<int>(), (int)roots.numel(),
                                  X.data_ptr<float>(), out.data_ptr<float>(), (int)S,
                                  (int)F, (int)C, current_stream(X)),
              "rf_predict_f32");
  } else {
    check_hip(lpx::rf_predict_f32_ilp(nodes.data_ptr<int>(), leaf_values.data_ptr<float>(),
                                      roots.data_ptr<int>(), (int)roots.numel(),
                                      X.data_ptr<float>(), out.data_ptr<float>(), (int)S,
                                      (int)F, (int)C, (int)ilp, current_stream(X)),
              "rf_predict_f32_ilp");
  }
  return out;
}

// ---- fp8 path (config 5) -------------------------------------------------

at::Tensor quantize_fp8(const at::Tensor& x, double scale) {
  CHECK_DEV(x);
  TORCH_CHECK(x.scalar_type() == at::kFloat, "quantize_fp8 expects f32");
  c10::hip::HIPGuardMasqueradingAsCUDA guard(x.device());
  auto out = at::empty(x.sizes(), x.options().dtype(at::kByte));
  check_hip(lpx::quantize_fp8(x.data_ptr<float>(), out.data_ptr<uint8_t>(),
                              x.numel(), (float)(1.0 / scale), current_stream(x)),
            "quantize_fp8");
  return out;
}

at::Tensor im2col_fp8(const at::Tensor& x, int64_t R, int64_t S, int64_t stride,
                      int64_t pad, double scale) {
  CHECK_DEV(x);
  TORCH_CHECK(x.scalar_type() == at::kFloat && x.dim() == 4,
              "im2col_fp8 expects NHWC f32 [N,H,W,C]");
  const int64_t N = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  const int64_t HO = (H + 2 * pad - R) / stride + 1;
  const int64_t WO = (W + 2 * pad - S) / stride + 1;
  c10::hip::HIPGuardMasqueradingAsCUDA guard(x.device());
  auto out = at::empty({N * HO * WO, R * S * C}, x.options().dtype(at::kByte));
  check_hip(lpx::im2col_fp8(x.data_ptr<float>(), out.data_ptr<uint8_t>(), (int)N,
                            (int)H, (int)W, (int)C, (int)R, (int)S, (int)HO,
                            (int)WO, (int)stride, (int)pad, (float)(1.0 / scale),
                            current_stream(x)),
            "im2col_fp8");
  return out;
}

at::Tensor gemm_fp8(const at::Tensor& A, const at::Tensor& Bt, double descale,
                    const c10::optional<at::Tensor>& bias, bool relu,
                    const c10::optional<at::Tensor>& resid) {
  CHECK_DEV(A);
  CHECK_DEV(Bt);
  TORCH_CHECK(A.scalar_type() == at::kByte && Bt.scalar_type() == at::kByte,
              "gemm_fp8 expects e4m3 bytes");
  TORCH_CHECK(A.dim() == 2 && Bt.dim() == 2 && A.size(1) == Bt.size(1),
              "gemm_fp8 expects A [M,K], Bt [N,K]");
  const int64_t M = A.size(0), K = A.size(1), N = Bt.size(0);
  c10::hip::HIPGuardMasqueradingAsCUDA guard(A.device());
  auto C = at::empty({M, N}, A.options().dtype(at::kFloat));
  const float* bptr = nullptr;
  if (bias.has_value()) {
    CHECK_DEV(bias.value());
    TORCH_CHECK(bias->numel() == N, "bias must be [N]");
    bptr = bias->data_ptr<float>();
  }
  const float* rptr = nullptr;
  if (resid.has_value()) {
    CHECK_DEV(resid.value());
    TORCH_CHECK(resid->numel() == M * N && resid->is_contiguous(),
                "resid must be contiguous [M, N]");
    rptr = resid->data_ptr<float>();
  }
  check_hip(lpx::gemm_fp8_nt_res(A.data_ptr<uint8_t>(), Bt.data_ptr<uint8_t>(),
                                 C.data_ptr<float>(), bptr, rptr, (int)M,
                                 (int)N, (int)K, (float)descale, relu,
                                 current_stream(A)),
            "gemm_fp8_nt_res");
  return C;
}

at::Tensor maxpool2d_nhwc(const at::Tensor& x, int64_t k, int64_t stride,
                          int64_t pad) {
  CHECK_DEV(x);
  TORCH_CHECK(x.scalar_type() == at::kFloat && x.dim() == 4 &&
              x.size(3) % 4 == 0, "maxpool2d_nhwc expects f32 NHWC, C%4==0");
  const int64_t N = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  const int64_t HO = (H + 2 * pad - k) / stride + 1;
  const int64_t WO = (W + 2 * pad - k) / stride + 1;
  c10::hip::HIPGuardMasqueradingAsCUDA guard(x.device());
  auto out = at::empty({N, HO, WO, C}, x.options());
  check_hip(lpx::maxpool2d_nhwc_f32(x.data_ptr<float>(), out.data_ptr<float>(),
                                    (int)N, (int)H, (int)W, (int)C, (int)k,
                                    (int)stride, (int)pad, current_stream(x)),
            "maxpool2d_nhwc_f32");
  return out;
}

at::Tensor gemm_fp8_q8(const at::Tensor& A, const at::Tensor& Bt,
                       double descale, const c10::optional<at::Tensor>& bias,
                       bool relu, double out_scale) {
  CHECK_DEV(A);
  CHECK_DEV(Bt);
  TORCH_CHECK(A.scalar_type() == at::kByte && Bt.scalar_type() == at::kByte,
              "gemm_fp8_q8 expects e4m3 bytes");
  TORCH_CHECK(A.dim() == 2 && Bt.dim() == 2 && A.size(1) == Bt.size(1),
              "gemm_fp8_q8 expects A [M,K], Bt [N,K]");
  const int64_t M = A.size(0), K = A.size(1), N = Bt.size(0);
  c10::hip::HIPGuardMasqueradingAsCUDA guard(A.device());
  const float* bptr = nullptr;
  if (bias.has_value()) {
    CHECK_DEV(bias.value());
    bptr = bias->data_ptr<float>();
  }
  auto C8 = at::empty({M, N}, A.options().dtype(at::kByte));
  const float oinv = (float)(1.0 / out_scale);
  if (M % 256 == 0 && N % 256 == 0 && K % 128 == 0 && K >= 128) {
    check_hip(lpx::gemm_fp8_nt8phx_q8(A.data_ptr<uint8_t>(),
                                      Bt.data_ptr<uint8_t>(),
                                      C8.data_ptr<uint8_t>(), bptr, (int)M,
                                      (int)N, (int)K, (float)descale, relu,
                                      oinv, current_stream(A)),
              "gemm_fp8_nt8phx_q8");
  } else {
    // unfused fallback for unaligned shapes: f32 GEMM + quantize pass
    auto C = at::empty({M, N}, A.options().dtype(at::kFloat));
    check_hip(lpx::gemm_fp8_nt(A.data_ptr<uint8_t>(), Bt.data_ptr<uint8_t>(),
                               C.data_ptr<float>(), bptr, (int)M, (int)N,
                               (int)K, (float)descale, relu,
                               current_stream(A)),
              "gemm_fp8_nt");
    check_hip(lpx::quantize_fp8(C.data_ptr<float>(), C8.data_ptr<uint8_t>(),
                                C.numel(), oinv, current_stream(A)),
              "quantize_fp8");
  }
  return C8;
}

at::Tensor conv_fp8_q8(const at::Tensor& xq, const at::Tensor& Wq, int64_t H,
                       int64_t W, int64_t Cin, int64_t R, int64_t S,
                       int64_t stride, int64_t pad, double descale,
                       const c10::optional<at::Tensor>& bias, bool relu,
                       double out_scale) {
  CHECK_DEV(xq);
  CHECK_DEV(Wq);
  TORCH_CHECK(xq.scalar_type() == at::kByte && Wq.scalar_type() == at::kByte,
              "conv_fp8_q8 expects e4m3 bytes");
  const int64_t numel = xq.numel();
  const int64_t Nb = numel / (H * W * Cin);
  const int64_t KO = Wq.size(0);
  const int64_t HO = (H + 2 * pad - R) / stride + 1;
  const int64_t WO = (W + 2 * pad - S) / stride + 1;
  c10::hip::HIPGuardMasqueradingAsCUDA guard(xq.device());
  const float* bptr = nullptr;
  if (bias.has_value()) bptr = bias->data_ptr<float>();
  auto C8 = at::empty({Nb * HO * WO, KO}, xq.options());
  check_hip(lpx::conv_fp8_implicit_q8(
                xq.data_ptr<uint8_t>(), Wq.data_ptr<uint8_t>(),
                C8.data_ptr<uint8_t>(), bptr, (int)Nb, (int)H, (int)W,
                (int)Cin, (int)R, (int)S, (int)HO, (int)WO, (int)stride,
                (int)pad, (int)KO, (float)descale, relu,
                (float)(1.0 / out_scale), current_stream(xq)),
            "conv_fp8_implicit_q8");
  return C8;
}

std::vector<at::Tensor> quantize_mx(const at::Tensor& x) {
  CHECK_DEV(x);
  TORCH_CHECK(x.scalar_type() == at::kFloat && x.dim() == 2 &&
              x.size(1) % 64 == 0, "quantize_mx expects f32 [R, K%64==0]");
  const int64_t R = x.size(0), K = x.size(1);
  c10::hip::HIPGuardMasqueradingAsCUDA guard(x.device());
  auto q = at::empty({R, K}, x.options().dtype(at::kByte));
  // tile-major scales: [ceil(groups/2)][R][4] (coalesced GEMM staging)
  auto sc = at::empty({(K / 64 + 1) / 2, R, 4}, x.options().dtype(at::kByte));
  check_hip(lpx::quantize_mx_fp8(x.data_ptr<float>(), q.data_ptr<uint8_t>(),
                                 sc.data_ptr<uint8_t>(), (int)R, (int)K,
                                 current_stream(x)),
            "quantize_mx_fp8");
  return {q, sc};
}

at::Tensor gemm_mx(const at::Tensor& A, const at::Tensor& sA,
                   const at::Tensor& Bt, const at::Tensor& sB) {
  CHECK_DEV(A);
  CHECK_DEV(Bt);
  CHECK_DEV(sA);
  CHECK_DEV(sB);
  TORCH_CHECK(A.scalar_type() == at::kByte && Bt.scalar_type() == at::kByte &&
              sA.scalar_type() == at::kByte && sB.scalar_type() == at::kByte,
              "gemm_mx expects byte tensors");
  TORCH_CHECK(A.dim() == 2 && Bt.dim() == 2 && A.size(1) == Bt.size(1) &&
              A.size(1) % 64 == 0, "gemm_mx expects A [M,K], Bt [N,K], K%64==0");
  const int64_t M = A.size(0), K = A.size(1), N = Bt.size(0);
  const int64_t kt = (K / 64 + 1) / 2;
  TORCH_CHECK(sA.numel() == kt * M * 4 && sB.numel() == kt * N * 4,
              "scale shapes must be [ceil(K/128), rows, 4] (tile-major)");
  c10::hip::HIPGuardMasqueradingAsCUDA guard(A.device());
  auto C = at::empty({M, N}, A.options().dtype(at::kFloat));
  if (lpx::gemm_mx_8ph_supported((int)M, (int)N, (int)K))
    check_hip(lpx::gemm_mx_fp8_nt8ph(
                  A.data_ptr<uint8_t>(), sA.data_ptr<uint8_t>(),
                  Bt.data_ptr<uint8_t>(), sB.data_ptr<uint8_t>(),
                  C.data_ptr<float>(), (int)M, (int)N, (int)K,
                  current_stream(A)),
              "gemm_mx_fp8_nt8ph");
  else
    check_hip(lpx::gemm_mx_fp8_nt(A.data_ptr<uint8_t>(), sA.data_ptr<uint8_t>(),
                                  Bt.data_ptr<uint8_t>(), sB.data_ptr<uint8_t>(),
                                  C.data_ptr<float>(), (int)M, (int)N, (int)K,
                                  current_stream(A)),
              "gemm_mx_fp8_nt");
  return C;
}

std::vector<at::Tensor> quantize_mx4(const at::Tensor& x) {
  CHECK_DEV(x);
  TORCH_CHECK(x.scalar_type() == at::kFloat && x.dim() == 2 &&
              x.size(1) % 64 == 0, "quantize_mx4 expects f32 [R, K%64==0]");
  const int64_t R = x.size(0), K = x.size(1);
  c10::hip::HIPGuardMasqueradingAsCUDA guard(x.device());
  auto q = at::empty({R, K / 2}, x.options().dtype(at::kByte));
  // tile-major scales: [ceil(groups/2)][R][4]
  auto sc = at::empty({(K / 64 + 1) / 2, R, 4}, x.options().dtype(at::kByte));
  check_hip(lpx::quantize_mx_fp4(x.data_ptr<float>(), q.data_ptr<uint8_t>(),
                                 sc.data_ptr<uint8_t>(), (int)R, (int)K,
                                 current_stream(x)),
            "quantize_mx_fp4");
  return {q, sc};
}

at::Tensor gemm_mx4(const at::Tensor& A, const at::Tensor& sA,
                    const at::Tensor& Bt, const at::Tensor& sB) {
  CHECK_DEV(A);
  CHECK_DEV(Bt);
  CHECK_DEV(sA);
  CHECK_DEV(sB);
  TORCH_CHECK(A.scalar_type() == at::kByte && Bt.scalar_type() == at::kByte &&
              sA.scalar_type() == at::kByte && sB.scalar_type() == at::kByte,
              "gemm_mx4 expects byte tensors");
  TORCH_CHECK(A.dim() == 2 && Bt.dim() == 2 && A.size(1) == Bt.size(1),
              "gemm_mx4 expects A [M,K/2], Bt [N,K/2]");
  const int64_t M = A.size(0), K = A.size(1) * 2, N = Bt.size(0);
  TORCH_CHECK(lpx::gemm_mx_fp4_supported((int)M, (int)N, (int)K),
              "gemm_mx4 requires M,N%256==0 and K%128==0");
  const int64_t kt = (K / 64 + 1) / 2;
  TORCH_CHECK(sA.numel() == kt * M * 4 && sB.numel() == kt * N * 4,
              "scale shapes must be [ceil(K/128), rows, 4] (tile-major)");
  c10::hip::HIPGuardMasqueradingAsCUDA guard(A.device());
  auto C = at::empty({M, N}, A.options().dtype(at::kFloat));
  check_hip(lpx::gemm_mx_fp4_nt8ph(A.data_ptr<uint8_t>(), sA.data_ptr<uint8_t>(),
                                   Bt.data_ptr<uint8_t>(), sB.data_ptr<uint8_t>(),
                                   C.data_ptr<float>(), (int)M, (int)N, (int)K,
                                   current_stream(A)),
            "gemm_mx_fp4_nt8ph");
  return C;
}

at::Tensor gemm_mx_w4(const at::Tensor& A, const at::Tensor& sA,
                      const at::Tensor& Bt4, const at::Tensor& sB) {
  CHECK_DEV(A);
  CHECK_DEV(Bt4);
  CHECK_DEV(sA);
  CHECK_DEV(sB);
  TORCH_CHECK(A.scalar_type() == at::kByte && Bt4.scalar_type() == at::kByte,
              "gemm_mx_w4 expects byte tensors");
  const int64_t M = A.size(0), K = A.size(1), N = Bt4.size(0);
  TORCH_CHECK(Bt4.size(1) * 2 == K,
              "gemm_mx_w4: Bt must be fp4 nibbles [N, K/2]");
  TORCH_CHECK(lpx::gemm_mx_w4_supported((int)M, (int)N, (int)K),
              "gemm_mx_w4 requires M,N%256==0 and K%128==0");
  const int64_t kt = (K / 64 + 1) / 2;
  TORCH_CHECK(sA.numel() == kt * M * 4 && sB.numel() == kt * N * 4,
              "scale shapes must be [ceil(K/128), rows, 4] (tile-major)");
  c10::hip::HIPGuardMasqueradingAsCUDA guard(A.device());
  auto C = at::empty({M, N}, A.options().dtype(at::kFloat));
  check_hip(lpx::gemm_mx_w4_nt8ph(A.data_ptr<uint8_t>(), sA.data_ptr<uint8_t>(),
                                  Bt4.data_ptr<uint8_t>(),
                                  sB.data_ptr<uint8_t>(), C.data_ptr<float>(),
                                  (int)M, (int)N, (int)K, current_stream(A)),
            "gemm_mx_w4_nt8ph");
  return C;
}

at::Tensor rf_predict8(const at::Tensor& nodes8, const at::Tensor& leaf_values,
                       const at::Tensor& roots, const at::Tensor& X,
                       int64_t ilp) {
  CHECK_DEV(nodes8);
  CHECK_DEV(leaf_values);
  CHECK_DEV(roots);
  CHECK_DEV(X);
  TORCH_CHECK(nodes8.scalar_type() == at::kInt && nodes8.dim() == 2 &&
              nodes8.size(1) == 2, "nodes8 must be int32 [n_nodes, 2]");
  TORCH_CHECK(leaf_values.scalar_type() == at::kFloat && leaf_values.dim() == 2,
              "leaf_values must be f32 [n_leaf_rows, C]");
  TORCH_CHECK(roots.scalar_type() == at::kInt, "roots must be int32");
  TORCH_CHECK(X.scalar_type() == at::kFloat && X.dim() == 2, "X must be f32 [S,F]");
  const int64_t S = X.size(0), F = X.size(1), C = leaf_values.size(1);
  TORCH_CHECK(C <= 8, "rf_predict8 kernel handles C<=8 per call");
  TORCH_CHECK(F <= 2046, "rf_predict8 packs feat in 11 bits (F<=2046)");
  c10::hip::HIPGuardMasqueradingAsCUDA guard(X.device());
  auto out = at::empty({S, C}, X.options());
  check_hip(lpx::rf_predict8_f32(nodes8.data_ptr<int>(), leaf_values.data_ptr<float>(),
                                 roots.data_ptr<int>(), (int)roots.numel(),
                                 X.data_ptr<float>(), out.data_ptr<float>(), (int)S,
                                 (int)F, (int)C, (int)(ilp > 0 ? ilp : 2),
                                 current_stream(X)),
            "rf_predict8_f32");
  return out;
}

at::Tensor rf_predict8_lds(const at::Tensor& nodes8, const at::Tensor& leaf_values,
                           const at::Tensor& roots, const at::Tensor& chunk_tree_begin,
                           int64_t max_chunk_nodes, const at::Tensor& X) {
  CHECK_DEV(nodes8);
  CHECK_DEV(leaf_values);
  CHECK_DEV(roots);
  CHECK_DEV(chunk_tree_begin);
  CHECK_DEV(X);
  TORCH_CHECK(nodes8.scalar_type() == at::kInt && nodes8.dim() == 2 &&
              nodes8.size(1) == 2, "nodes8 must be int32 [n_nodes, 2]");
  TORCH_CHECK(chunk_tree_begin.scalar_type() == at::kInt,
              "chunk_tree_begin must be int32 [n_chunks+1]");
  const int64_t S = X.size(0), F = X.size(1), C = leaf_values.size(1);
  TORCH_CHECK(C <= 8 && F <= 2046, "rf_predict8_lds: C<=8, F<=2046");
  const int64_t n_chunks = chunk_tree_begin.numel() - 1;
  c10::hip::HIPGuardMasqueradingAsCUDA guard(X.device());
  auto out = at::empty({S, C}, X.options());
  check_hip(lpx::rf_predict8_lds_f32(
                nodes8.data_ptr<int>(), leaf_values.data_ptr<float>(),
                roots.data_ptr<int>(), chunk_tree_begin.data_ptr<int>(),
                (int)n_chunks, (int)max_chunk_nodes, (int)roots.numel(),
                (int)nodes8.size(0), X.data_ptr<float>(), out.data_ptr<float>(),
                (int)S, (int)F, (int)C, current_stream(X)),
            "rf_predict8_lds_f32");
  return out;
}

at::Tensor gemm_bf16_tune8(const at::Tensor& A, const at::Tensor& Bt, int64_t tune) {
  CHECK_DEV(A);
  CHECK_DEV(Bt);
  const int64_t M = A.size(0), K = A.size(1), N = Bt.size(0);
  c10::hip::HIPGuardMasqueradingAsCUDA guard(A.device());
  auto C = at::empty({M, N}, A.options().dtype(at::kFloat));
  check_hip(lpx::gemm_bf16_nt8ph_tune(
                reinterpret_cast<const uint16_t*>(A.data_ptr()),
                reinterpret_cast<const uint16_t*>(Bt.data_ptr()),
                C.data_ptr<float>(), (int)M, (int)N, (int)K, (int)tune,
                current_stream(A)),
            "gemm_bf16_nt8ph_tune");
  return C;
}

at::Tensor rf_predict_heap(const at::Tensor& nodes_heap,
                           const at::Tensor& leaf_values, int64_t depth,
                           const at::Tensor& X, int64_t silp, int64_t top) {
  CHECK_DEV(nodes_heap);
  CHECK_DEV(leaf_values);
  CHECK_DEV(X);
  TORCH_CHECK(nodes_heap.scalar_type() == at::kInt && nodes_heap.dim() == 2 &&
              nodes_heap.size(1) == 2, "nodes_heap must be int32 [n, 2]");
  TORCH_CHECK(X.scalar_type() == at::kFloat && X.dim() == 2, "X must be f32 [S,F]");
  const int64_t S = X.size(0), F = X.size(1), C = leaf_values.size(1);
  TORCH_CHECK(C <= 8, "rf_predict_heap: C<=8 per call");
  const int64_t tree_stride = (1LL << depth) - 1;
  TORCH_CHECK(nodes_heap.size(0) % tree_stride == 0, "nodes/depth mismatch");
  const int64_t n_trees = nodes_heap.size(0) / tree_stride;
  TORCH_CHECK(leaf_values.size(0) == n_trees * (1LL << depth),
              "leaf_values rows != n_trees * 2^depth");
  c10::hip::HIPGuardMasqueradingAsCUDA guard(X.device());
  auto out = at::empty({S, C}, X.options());
  if (top == -2)
    check_hip(lpx::rf_predict_heap_f32(
                  nodes_heap.data_ptr<int>(), leaf_values.data_ptr<float>(),
                  (int)n_trees, (int)depth, X.data_ptr<float>(),
                  out.data_ptr<float>(), (int)S, (int)F, (int)C, (int)silp,
                  current_stream(X)),
              "rf_predict_heap_f32");
  else
    check_hip(lpx::rf_predict_heap2_f32(
                  nodes_heap.data_ptr<int>(), leaf_values.data_ptr<float>(),
                  (int)n_trees, (int)depth, X.data_ptr<float>(),
                  out.data_ptr<float>(), (int)S, (int)F, (int)C, (int)silp,
                  (int)top, current_stream(X)),
              "rf_predict_heap2_f32");
  return out;
}

at::Tensor rf_predict_heap_quad(const at::Tensor& quads,
                                const at::Tensor& roots,
                                const at::Tensor& leaf_values, int64_t depth,
                                const at::Tensor& X, int64_t silp) {
  CHECK_DEV(quads);
  CHECK_DEV(roots);
  CHECK_DEV(leaf_values);
  CHECK_DEV(X);
  TORCH_CHECK(quads.scalar_type() == at::kInt && quads.dim() == 2 &&
              quads.size(1) == 16, "quads must be int32 [n, 16]");
  TORCH_CHECK(roots.scalar_type() == at::kInt && roots.dim() == 2 &&
              roots.size(1) == 2, "roots must be int32 [n_trees, 2]");
  TORCH_CHECK(X.scalar_type() == at::kFloat && X.dim() == 2,
              "X must be f32 [S,F]");
  const int64_t S = X.size(0), F = X.size(1), C = leaf_values.size(1);
  TORCH_CHECK(C <= 8, "rf_predict_heap_quad: C<=8 per call");
  const int64_t n_trees = roots.size(0);
  const int64_t qpt = ((1LL << depth) - 1) / 3;
  TORCH_CHECK(quads.size(0) == n_trees * qpt, "quads/depth mismatch");
  TORCH_CHECK(leaf_values.size(0) == n_trees * (1LL << depth),
              "leaf_values rows != n_trees * 2^depth");
  c10::hip::HIPGuardMasqueradingAsCUDA guard(X.device());
  auto out = at::empty({S, C}, X.options());
  check_hip(lpx::rf_predict_heap_quad_f32(
                quads.data_ptr<int>(), roots.data_ptr<int>(),
                leaf_values.data_ptr<float>(), (int)n_trees, (int)depth,
                X.data_ptr<float>(), out.data_ptr<float>(), (int)S, (int)F,
                (int)C, (int)silp, current_stream(X)),
            "rf_predict_heap_quad_f32");
  return out;
}

at::Tensor rf_predict_heap_pair(const at::Tensor& pairs,
                                const at::Tensor& roots,
                                const at::Tensor& leaf_values, int64_t depth,
                                const at::Tensor& X, int64_t silp) {
  CHECK_DEV(pairs);
  CHECK_DEV(roots);
  CHECK_DEV(leaf_values);
  CHECK_DEV(X);
  TORCH_CHECK(pairs.scalar_type() == at::kInt && pairs.dim() == 2 &&
              pairs.size(1) == 4, "pairs must be int32 [n, 4]");
  TORCH_CHECK(roots.scalar_type() == at::kInt && roots.dim() == 2 &&
              roots.size(1) == 2, "roots must be int32 [n_trees, 2]");
  TORCH_CHECK(X.scalar_type() == at::kFloat && X.dim() == 2,
              "X must be f32 [S,F]");
  const int64_t S = X.size(0), F = X.size(1), C = leaf_values.size(1);
  TORCH_CHECK(C <= 8, "rf_predict_heap_pair: C<=8 per call");
  const int64_t n_trees = roots.size(0);
  const int64_t ppt = (1LL << (depth - 1)) - 1;
  TORCH_CHECK(pairs.size(0) == n_trees * ppt, "pairs/depth mismatch");
  TORCH_CHECK(leaf_values.size(0) == n_trees * (1LL << depth),
              "leaf_values rows != n_trees * 2^depth");
  c10::hip::HIPGuardMasqueradingAsCUDA guard(X.device());
  auto out = at::empty({S, C}, X.options());
  check_hip(lpx::rf_predict_heap_pair_f32(
                pairs.data_ptr<int>(), roots.data_ptr<int>(),
                leaf_values.data_ptr<float>(), (int)n_trees, (int)depth,
                X.data_ptr<float>(), out.data_ptr<float>(), (int)S, (int)F,
                (int)C, (int)silp, current_stream(X)),
            "rf_predict_heap_pair_f32");
  return out;
}

at::Tensor rf_predict_heap_cpair(const at::Tensor& cpairs,
                                 const at::Tensor& roots,
                                 const at::Tensor& leaf_values, int64_t depth,
                                 const at::Tensor& X, int64_t silp) {
  // v8 compact layout: node = (u16 feat << 16) | fp16-threshold-bits.
  CHECK_DEV(cpairs);
  CHECK_DEV(roots);
  CHECK_DEV(leaf_values);
  CHECK_DEV(X);
  TORCH_CHECK(cpairs.scalar_type() == at::kInt && cpairs.dim() == 2 &&
              cpairs.size(1) == 2, "cpairs must be int32 [n, 2]");
  TORCH_CHECK(roots.scalar_type() == at::kInt && roots.dim() == 1,
              "roots must be int32 [n_trees]");
  TORCH_CHECK(X.scalar_type() == at::kFloat && X.dim() == 2,
              "X must be f32 [S,F]");
  const int64_t S = X.size(0), F = X.size(1), C = leaf_values.size(1);
  TORCH_CHECK(C <= 8, "rf_predict_heap_cpair: C<=8 per call");
  TORCH_CHECK(F <= 0xffff, "rf_predict_heap_cpair: F must fit u16");
  const int64_t n_trees = roots.size(0);
  const int64_t ppt = (1LL << (depth - 1)) - 1;
  TORCH_CHECK(cpairs.size(0) == n_trees * ppt, "cpairs/depth mismatch");
  TORCH_CHECK(leaf_values.size(0) == n_trees * (1LL << depth),
              "leaf_values rows != n_trees * 2^depth");
  c10::hip::HIPGuardMasqueradingAsCUDA guard(X.device());
  auto out = at::empty({S, C}, X.options());
  check_hip(lpx::rf_predict_heap_cpair_f32(
                cpairs.data_ptr<int>(), roots.data_ptr<int>(),
                leaf_values.data_ptr<float>(), (int)n_trees, (int)depth,
                X.data_ptr<float>(), out.data_ptr<float>(), (int)S, (int)F,
                (int)C, (int)silp, current_stream(X)),
            "rf_predict_heap_cpair_f32");
  return out;
}

at::Tensor rf_predict_heap_cpair16(const at::Tensor& cpairs,
                                   const at::Tensor& roots,
                                   const at::Tensor& lv16, int64_t depth,
                                   const at::Tensor& X, int64_t silp) {
  // v8 + fp16 leaf table (one 16-B leaf request; exact for
  // fp16-representable leaf values).
  CHECK_DEV(cpairs);
  CHECK_DEV(roots);
  CHECK_DEV(lv16);
  CHECK_DEV(X);
  TORCH_CHECK(cpairs.scalar_type() == at::kInt && cpairs.dim() == 2 &&
              cpairs.size(1) == 2, "cpairs must be int32 [n, 2]");
  TORCH_CHECK(roots.scalar_type() == at::kInt && roots.dim() == 1,
              "roots must be int32 [n_trees]");
  TORCH_CHECK(lv16.scalar_type() == at::kHalf && lv16.dim() == 2,
              "lv16 must be fp16 [rows, C]");
  TORCH_CHECK(X.scalar_type() == at::kFloat && X.dim() == 2,
              "X must be f32 [S,F]");
  const int64_t S = X.size(0), F = X.size(1), C = lv16.size(1);
  TORCH_CHECK(C <= 8 && F <= 0xffff, "rf_predict_heap_cpair16: C<=8, F<=u16");
  const int64_t n_trees = roots.size(0);
  const int64_t ppt = (1LL << (depth - 1)) - 1;
  TORCH_CHECK(cpairs.size(0) == n_trees * ppt, "cpairs/depth mismatch");
  TORCH_CHECK(lv16.size(0) == n_trees * (1LL << depth),
              "lv16 rows != n_trees * 2^depth");
  c10::hip::HIPGuardMasqueradingAsCUDA guard(X.device());
  auto out = at::empty({S, C}, X.options());
  check_hip(lpx::rf_predict_heap_cpair_lv16_f32(
                cpairs.data_ptr<int>(), roots.data_ptr<int>(),
                reinterpret_cast<const unsigned short*>(lv16.data_ptr()),
                (int)n_trees, (int)depth, X.data_ptr<float>(),
                out.data_ptr<float>(), (int)S, (int)F, (int)C, (int)silp,
                current_stream(X)),
            "rf_predict_heap_cpair_lv16_f32");
  return out;
}

at::Tensor conv_fp8(const at::Tensor& xq, const at::Tensor& Wq,
                    int64_t H, int64_t W, int64_t Cin, int64_t R, int64_t S,
                    int64_t stride, int64_t pad, double descale,
                    const c10::optional<at::Tensor>& bias, bool relu) {
  CHECK_DEV(xq);
  CHECK_DEV(Wq);
  TORCH_CHECK(xq.scalar_type() == at::kByte && Wq.scalar_type() == at::kByte,
              "conv_fp8 expects e4m3 bytes");
  TORCH_CHECK(Cin % 16 == 0, "conv_fp8 requires C % 16 == 0");
  const int64_t Nb = xq.numel() / (H * W * Cin);
  const int64_t KO = Wq.size(0);
  TORCH_CHECK(Wq.size(1) == R * S * Cin, "Wq must be [KO, R*S*C]");
  const int64_t HO = (H + 2 * pad - R) / stride + 1;
  const int64_t WO = (W + 2 * pad - S) / stride + 1;
  c10::hip::HIPGuardMasqueradingAsCUDA guard(xq.device());
  auto out = at::empty({Nb * HO * WO, KO}, xq.options().dtype(at::kFloat));
  const float* bptr = nullptr;
  if (bias.has_value()) {
    CHECK_DEV(bias.value());
    bptr = bias->data_ptr<float>();
  }
  check_hip(lpx::conv_fp8_implicit(xq.data_ptr<uint8_t>(), Wq.data_ptr<uint8_t>(),
                                   out.data_ptr<float>(), bptr, (int)Nb, (int)H,
                                   (int)W, (int)Cin, (int)R, (int)S, (int)HO,
                                   (int)WO, (int)stride, (int)pad, (int)KO,
                                   (float)descale, relu, current_stream(xq)),
            "conv_fp8_implicit");
  return out;
}

std::string build_info() {
  return std::string("lambdipy_amd ops: gfx950 (CDNA4), HIP ") +
         std::to_string(HIP_VERSION_MAJOR) + "." + std::to_string(HIP_VERSION_MINOR);
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "lambdipy_amd gfx950 kernel library (MFMA/LDS-tiled HIP)";
  m.def("gemm_bf16", &gemm_bf16, "C = A[M,K] @ Bt[N,K]^T (bf16 MFMA)",
        py::arg("A"), py::arg("Bt"), py::arg("bf16_out") = false,
        py::arg("variant") = -1);
  m.def("gemv", &gemv, "y[b,m] = A[b,m,:] . x[b,:]", py::arg("A"), py::arg("x"));
  m.def("spmv_csr", &spmv_csr, "y = CSR(A) @ x", py::arg("indptr"),
        py::arg("indices"), py::arg("data"), py::arg("x"),
        py::arg("variant") = -1);
  m.def("fft_c64", &fft_c64, "batched complex64 FFT (Stockham)", py::arg("x"),
        py::arg("twiddles"), py::arg("inverse") = false);
  m.def("pairwise_sqeuclidean", &pairwise_sqeuclidean,
        "D[i,j] = ||X_i - Y_j||^2 (f32 MFMA)", py::arg("X"), py::arg("Y"));
  m.def("rf_predict", &rf_predict, "RandomForest per-class sums over trees",
        py::arg("nodes"), py::arg("leaf_values"), py::arg("roots"), py::arg("X"),
        py::arg("ilp") = 0);
  m.def("rf_predict_heap", &rf_predict_heap,
        "RandomForest per-class sums (heap layout, complete forests)",
        py::arg("nodes_heap"), py::arg("leaf_values"), py::arg("depth"),
        py::arg("X"), py::arg("silp") = 1, py::arg("top") = -2);
  m.def("rf_predict_heap_quad", &rf_predict_heap_quad,
        "RandomForest per-class sums (quad-record heap layout)",
        py::arg("quads"), py::arg("roots"), py::arg("leaf_values"),
        py::arg("depth"), py::arg("X"), py::arg("silp") = 1);
  m.def("rf_predict_heap_pair", &rf_predict_heap_pair,
        "RandomForest per-class sums (child-pair heap layout)",
        py::arg("pairs"), py::arg("roots"), py::arg("leaf_values"),
        py::arg("depth"), py::arg("X"), py::arg("silp") = 1);
  m.def("rf_predict_heap_cpair", &rf_predict_heap_cpair,
        "v8 compact child-pair heap traversal (u16 feat + fp16 thr)",
        py::arg("cpairs"), py::arg("roots"), py::arg("leaf_values"),
        py::arg("depth"), py::arg("X"), py::arg("silp") = 1);
  m.def("rf_predict_heap_cpair16", &rf_predict_heap_cpair16,
        "v8 compact traversal with fp16 leaf table",
        py::arg("cpairs"), py::arg("roots"), py::arg("lv16"),
        py::arg("depth"), py::arg("X"), py::arg("silp") = 1);
  m.def("rf_predict8_lds", &rf_predict8_lds,
        "RandomForest per-class sums (forest chunks staged in LDS)",
        py::arg("nodes8"), py::arg("leaf_values"), py::arg("roots"),
        py::arg("chunk_tree_begin"), py::arg("max_chunk_nodes"), py::arg("X"));
  m.def("rf_predict8", &rf_predict8,
        "RandomForest per-class sums (packed 8-B depth-first nodes)",
        py::arg("nodes8"), py::arg("leaf_values"), py::arg("roots"),
        py::arg("X"), py::arg("ilp") = 0);
  m.def("quantize_fp8", &quantize_fp8, "f32 -> OCP e4m3 bytes (x / scale)",
        py::arg("x"), py::arg("scale") = 1.0);
  m.def("im2col_fp8", &im2col_fp8, "NHWC f32 -> [M, R*S*C] e4m3 patches",
        py::arg("x"), py::arg("R"), py::arg("S"), py::arg("stride"),
        py::arg("pad"), py::arg("scale") = 1.0);
  m.def("conv_fp8", &conv_fp8, "implicit-GEMM fp8 conv (NHWC e4m3)",
        py::arg("xq"), py::arg("Wq"), py::arg("H"), py::arg("W"), py::arg("Cin"),
        py::arg("R"), py::arg("S"), py::arg("stride"), py::arg("pad"),
        py::arg("descale") = 1.0, py::arg("bias") = py::none(),
        py::arg("relu") = false);
  m.def("gemm_fp8", &gemm_fp8,
        "fp8 e4m3 MFMA GEMM with fused descale/bias/resid/relu",
        py::arg("A"), py::arg("Bt"), py::arg("descale") = 1.0,
        py::arg("bias") = py::none(), py::arg("relu") = false,
        py::arg("resid") = py::none());
  m.def("maxpool2d_nhwc", &maxpool2d_nhwc, "NHWC f32 max pool",
        py::arg("x"), py::arg("k"), py::arg("stride"), py::arg("pad"));
  m.def("gemm_fp8_q8", &gemm_fp8_q8,
        "fp8 GEMM with fused epilogue quantization (e4m3 out)", py::arg("A"),
        py::arg("Bt"), py::arg("descale"), py::arg("bias") = c10::nullopt,
        py::arg("relu") = false, py::arg("out_scale") = 1.0);
  m.def("conv_fp8_q8", &conv_fp8_q8,
        "implicit fp8 conv with fused epilogue quantization", py::arg("xq"),
        py::arg("Wq"), py::arg("H"), py::arg("W"), py::arg("Cin"),
        py::arg("R"), py::arg("S"), py::arg("stride"), py::arg("pad"),
        py::arg("descale"), py::arg("bias") = c10::nullopt,
        py::arg("relu") = false, py::arg("out_scale") = 1.0);
  m.def("quantize_mx", &quantize_mx,
        "f32 [R,K] -> (e4m3 bytes shuffled, E8M0 [R,K/64,2] block scales)",
        py::arg("x"));
  m.def("gemm_mx", &gemm_mx,
        "MX block-scaled fp8 GEMM (scales applied in hardware)", py::arg("A"),
        py::arg("sA"), py::arg("Bt"), py::arg("sB"));
  m.def("quantize_mx4", &quantize_mx4,
        "f32 [R,K] -> (e2m1 nibbles shuffled [R,K/2], E8M0 scales)",
        py::arg("x"));
  m.def("gemm_mx_w4", &gemm_mx_w4,
        "mixed MX GEMM: fp8 activations x fp4 weights (full fp8 rate, half "
        "weight bytes)", py::arg("A"), py::arg("sA"), py::arg("Bt4"),
        py::arg("sB"));
  m.def("gemm_mx4", &gemm_mx4,
        "MX block-scaled fp4 GEMM (2x fp8 MFMA rate)", py::arg("A"),
        py::arg("sA"), py::arg("Bt"), py::arg("sB"));
  m.def("gemm_bf16_tune8", &gemm_bf16_tune8, "8-phase GEMM tuning variants",
        py::arg("A"), py::arg("Bt"), py::arg("tune") = 0);
  m.def("build_info", &build_info);
}
