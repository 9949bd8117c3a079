"""GPU op library: hand-written gfx950 HIP kernels behind torch tensors.

This package is designed to be bundled into recipe wheels as ``lpx_ops``
(the packager copies it verbatim — relative imports only). The extension
``_lpx_ops`` is AOT-compiled for gfx950; there is no JIT fallback, no
multi-backend dispatch and no CPU execution path in here — on a GPU machine
a missing extension raises immediately (fail-loud contract), while CPU-only
dev machines may import the package for packaging/CLI work.
"""

from __future__ import annotations

from .api import (
    ext,
    extension_available,
    extension_error,
    fft,
    gemm_bf16,
    gemm_mx,
    gemm_mx_fp4,
    gemm_mx_w4,
    gemv,
    ifft,
    matmul_bf16,
    matmul_mx,
    matmul_mx_fp4,
    matmul_mx_w4,
    quantize_mx,
    quantize_mx_fp4,
    pairwise_sqeuclidean,
    require_extension,
    rf_predict_proba_sum,
    spmv_csr,
)

__all__ = [
    "ext",
    "extension_available",
    "extension_error",
    "require_extension",
    "gemm_bf16",
    "matmul_bf16",
    "quantize_mx",
    "gemm_mx",
    "matmul_mx",
    "quantize_mx_fp4",
    "gemm_mx_fp4",
    "gemm_mx_w4",
    "matmul_mx_fp4",
    "matmul_mx_w4",
    "gemv",
    "spmv_csr",
    "fft",
    "ifft",
    "pairwise_sqeuclidean",
    "rf_predict_proba_sum",
]
