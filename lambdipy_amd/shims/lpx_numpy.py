"""lpx_numpy — numpy dense hot ops on gfx950 MFMA (recipe: numpy).

Bundled into the numpy recipe layer. Replaces the
vendored-OpenBLAS role for the handler's hot path: matmul / matvec run on the
hand-written bf16 MFMA GEMM and batched-GEMV HIP kernels instead of host BLAS.
Not a dispatcher: inputs are moved to the ROCm device and the gfx950 kernel is
the only execution path (a missing extension raises).
"""

from __future__ import annotations

import numpy as np
import torch

try:  # bundled layer form
    import lpx_ops as _ops
except ImportError:  # in-repo dev form
    from lambdipy_amd import ops as _ops

_DEV = "cuda"


def _to_bf16(t: np.ndarray) -> torch.Tensor:
    return torch.from_numpy(np.ascontiguousarray(t)).to(_DEV).to(torch.bfloat16)


def matmul(a: np.ndarray, b: np.ndarray) -> np.ndarray:
    """C = a @ b on the bf16 MFMA GEMM kernel (f32 accumulate/output)."""
    ta, tb = _to_bf16(a), _to_bf16(b)
    c = _ops.matmul_bf16(ta, tb)
    return c.float().cpu().numpy()


def matmul_t(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """Device-resident matmul (bf16 in, f32 out) — no host round trip."""
    return _ops.matmul_bf16(a, b)


def matvec(a: np.ndarray, x: np.ndarray) -> np.ndarray:
    ta = _to_bf16(a)
    tx = _to_bf16(x)
    y = _ops.gemv(ta, tx)
    return y.cpu().numpy()


def device_ready() -> bool:
    return torch.cuda.is_available() and _ops.extension_available()
