"""lpx_scipy — scipy FFT + CSR SpMV on gfx950 (recipe: scipy).

Backs the scipy layer's hot ops: batched complex64
Stockham FFT in LDS and adaptive CSR SpMV, both hand-written HIP kernels.
"""

from __future__ import annotations

import numpy as np
import torch

try:
    import lpx_ops as _ops
except ImportError:
    from lambdipy_amd import ops as _ops

_DEV = "cuda"


def fft(x: np.ndarray) -> np.ndarray:
    """1-D FFT over the last axis (complex64, power-of-two N)."""
    t = torch.from_numpy(np.ascontiguousarray(x.astype(np.complex64))).to(_DEV)
    return _ops.fft(t).cpu().numpy()


def ifft(x: np.ndarray) -> np.ndarray:
    t = torch.from_numpy(np.ascontiguousarray(x.astype(np.complex64))).to(_DEV)
    return _ops.ifft(t).cpu().numpy()


def fft_t(x: torch.Tensor) -> torch.Tensor:
    return _ops.fft(x)


def ifft_t(x: torch.Tensor) -> torch.Tensor:
    return _ops.ifft(x)


def csr_matvec(indptr, indices, data, x) -> np.ndarray:
    """y = A @ x for scipy.sparse.csr_matrix pieces (f32, int32 indices)."""
    ti = torch.from_numpy(np.ascontiguousarray(indptr, dtype=np.int32)).to(_DEV)
    tj = torch.from_numpy(np.ascontiguousarray(indices, dtype=np.int32)).to(_DEV)
    td = torch.from_numpy(np.ascontiguousarray(data, dtype=np.float32)).to(_DEV)
    tx = torch.from_numpy(np.ascontiguousarray(x, dtype=np.float32)).to(_DEV)
    return _ops.spmv_csr(ti, tj, td, tx).cpu().numpy()


def csr_matvec_t(indptr: torch.Tensor, indices: torch.Tensor,
                 data: torch.Tensor, x: torch.Tensor) -> torch.Tensor:
    return _ops.spmv_csr(indptr, indices, data, x)
