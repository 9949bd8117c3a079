"""Python API over the gfx950 extension.

Loading contract (fail-loud): the compiled ``_lpx_ops``
extension lives NEXT TO this file (in-tree / in-wheel, AOT gfx950). If torch
sees a GPU and the extension is missing or broken, every op raises — there is
deliberately no eager/CPU fallback on GPU machines, so a silently-slow path
cannot masquerade as the native one.
"""

from __future__ import annotations

import importlib
import math
import os
from typing import Optional

import torch

_ext = None
_ext_error: Optional[BaseException] = None

try:
    _ext = importlib.import_module("._lpx_ops", __package__)
except Exception as e:  # noqa: BLE001 — record and re-raise at use time
    _ext = None
    _ext_error = e


def extension_available() -> bool:
    return _ext is not None


def extension_error() -> Optional[BaseException]:
    return _ext_error


def require_extension():
    if _ext is None:
        raise ImportError(
            "lambdipy_amd ops extension (_lpx_ops, gfx950) is not built/loadable. "
            "Build it in-tree with `PYTORCH_ROCM_ARCH=gfx950 python setup.py "
            f"build_ext --inplace`. Original error: {_ext_error!r}"
        )
    return _ext


def ext():
    """The raw extension module (fail-loud)."""
    return require_extension()


# ---- dense ---------------------------------------------------------------

def gemm_bf16(A: torch.Tensor, Bt: torch.Tensor, bf16_out: bool = False,
              variant: int = -1) -> torch.Tensor:
    """C = A[M,K] @ Bt[N,K]^T — NT layout, bf16 MFMA kernel.

    variant: -1 auto, 1 = 128^2 tile, 2 = 256^2/BK64 register-staged pipeline."""
    return require_extension().gemm_bf16(A.contiguous(), Bt.contiguous(), bf16_out, variant)


def matmul_bf16(A: torch.Tensor, B: torch.Tensor, bf16_out: bool = False) -> torch.Tensor:
    """C = A @ B for row-major B (transposes B on device into NT layout)."""
    return gemm_bf16(A, B.t().contiguous(), bf16_out)


def gemv(A: torch.Tensor, x: torch.Tensor) -> torch.Tensor:
    return require_extension().gemv(A.contiguous(), x.contiguous())


def quantize_mx(x: torch.Tensor):
    """f32 [R, K%64==0] -> (e4m3 bytes in MFMA-shuffled order, E8M0 scales
    TILE-MAJOR [ceil(K/128), R, 4] so GEMM scale staging is coalesced).
    One shared 2^e scale per contiguous 32-element MX block (hardware scale
    coverage of v_mfma_scale_f32_32x32x64_f8f6f4 decoded in
    scripts/mx_sweep.py)."""
    q, sc = require_extension().quantize_mx(x.contiguous())
    return q, sc


def gemm_mx(A: torch.Tensor, sA: torch.Tensor, Bt: torch.Tensor,
            sB: torch.Tensor) -> torch.Tensor:
    """MX block-scaled fp8 GEMM: C = dequant(A) @ dequant(Bt)^T, f32 out.
    Operands/scales come from quantize_mx; scales are applied in hardware."""
    return require_extension().gemm_mx(A, sA, Bt, sB)


def matmul_mx(A: torch.Tensor, B: torch.Tensor) -> torch.Tensor:
    """Convenience: quantize both f32 operands to MX fp8 and multiply."""
    qa, sa = quantize_mx(A)
    qb, sb = quantize_mx(B.t().contiguous())
    return gemm_mx(qa, sa, qb, sb)


def quantize_mx_fp4(x: torch.Tensor):
    """f32 [R, K%64==0] -> (e2m1 nibbles [R, K/2], linear order, E8M0
    scales TILE-MAJOR [ceil(K/128), R, 4]). 2x the fp8 MFMA rate."""
    q, sc = require_extension().quantize_mx4(x.contiguous())
    return q, sc


def gemm_mx_fp4(A: torch.Tensor, sA: torch.Tensor, Bt: torch.Tensor,
                sB: torch.Tensor) -> torch.Tensor:
    """MX block-scaled fp4 GEMM (M,N%256, K%128); f32 out, hw scales."""
    return require_extension().gemm_mx4(A, sA, Bt, sB)


def matmul_mx_fp4(A: torch.Tensor, B: torch.Tensor) -> torch.Tensor:
    """Convenience: quantize both f32 operands to MX fp4 and multiply."""
    qa, sa = quantize_mx_fp4(A)
    qb, sb = quantize_mx_fp4(B.t().contiguous())
    return gemm_mx_fp4(qa, sa, qb, sb)


def gemm_mx_w4(A: torch.Tensor, sA: torch.Tensor, Bt4: torch.Tensor,
               sB: torch.Tensor) -> torch.Tensor:
    """Mixed MX GEMM: fp8-MX activations x fp4-MX weights — the full fp8
    MFMA rate with half the weight bytes (weight-heavy inference GEMMs)."""
    return require_extension().gemm_mx_w4(A, sA, Bt4, sB)


def matmul_mx_w4(A: torch.Tensor, B: torch.Tensor) -> torch.Tensor:
    """Convenience: activations to fp8-MX, weights to fp4-MX, multiply."""
    qa, sa = quantize_mx(A)
    qb, sb = quantize_mx_fp4(B.t().contiguous())
    return gemm_mx_w4(qa, sa, qb, sb)


# ---- sparse --------------------------------------------------------------

def spmv_csr(indptr: torch.Tensor, indices: torch.Tensor, data: torch.Tensor,
             x: torch.Tensor, variant: int = -1) -> torch.Tensor:
    """y = A @ x. variant: -1 adaptive seg8/wave-row; 2 = x-tile-in-LDS
    (caller asserts CSR-canonical sorted columns — scipy's invariant)."""
    return require_extension().spmv_csr(
        indptr.contiguous(), indices.contiguous(), data.contiguous(),
        x.contiguous(), variant
    )


# ---- FFT -----------------------------------------------------------------

_twiddle_cache = {}


def _twiddles(n: int, device: torch.device) -> torch.Tensor:
    key = (n, str(device))
    t = _twiddle_cache.get(key)
    if t is None:
        k = torch.arange(n // 2, dtype=torch.float64)
        ang = -2.0 * math.pi * k / n
        t = torch.complex(torch.cos(ang), torch.sin(ang)).to(torch.complex64).to(device)
        _twiddle_cache[key] = t
    return t


def fft(x: torch.Tensor) -> torch.Tensor:
    """Batched 1-D FFT over the last axis. Power-of-two N runs the LDS
    Stockham kernel directly; any other N runs Bluestein's chirp-z over the
    same kernel (scipy.fft parity for arbitrary sizes — SURVEY.md §2.2 N6).
    """
    x = x.contiguous()
    n = x.shape[-1]
    if n & (n - 1) == 0:
        return require_extension().fft_c64(x, _twiddles(n, x.device), False)
    return _bluestein(x, inverse=False)


def ifft(x: torch.Tensor) -> torch.Tensor:
    x = x.contiguous()
    n = x.shape[-1]
    if n & (n - 1) == 0:
        return require_extension().fft_c64(x, _twiddles(n, x.device), True)
    return _bluestein(x, inverse=True)


_bluestein_cache = {}


def _bluestein_plan(n: int, device: torch.device):
    """(chirp a_k = e^{-i pi k^2 / n}, FFT(b) for the padded conv kernel,
    m): Bluestein expresses DFT_n as a circular convolution that runs on
    the power-of-two kernel. Chirp computed in f64 (k^2/n up to ~n, f32
    phase would lose the FFT's accuracy)."""
    key = (n, str(device))
    plan = _bluestein_cache.get(key)
    if plan is None:
        m = 1
        while m < 2 * n - 1:
            m <<= 1
        k = torch.arange(n, dtype=torch.float64)
        ang = math.pi * (k * k % (2 * n)) / n  # mod keeps f64 phase exact
        a = torch.complex(torch.cos(ang), -torch.sin(ang))  # e^{-i pi k^2/n}
        b = torch.zeros(m, dtype=torch.complex128)
        conj = torch.complex(torch.cos(ang), torch.sin(ang))
        b[:n] = conj
        b[m - n + 1:] = conj[1:].flip(0)
        a32 = a.to(torch.complex64).to(device)
        b32 = b.to(torch.complex64).to(device)
        e = require_extension()
        fb = e.fft_c64(b32.unsqueeze(0).contiguous(), _twiddles(m, device),
                       False)[0]
        plan = (a32, fb, m)
        _bluestein_cache[key] = plan
    return plan


def _bluestein(x: torch.Tensor, inverse: bool) -> torch.Tensor:
    n = x.shape[-1]
    shape = x.shape
    x2 = x.reshape(-1, n)
    if inverse:  # IDFT_n(x) = conj(DFT_n(conj(x))) / n
        return (_bluestein(x2.conj(), False).conj() / n).reshape(shape)
    a, fb, m = _bluestein_plan(n, x.device)
    e = require_extension()
    tw = _twiddles(m, x.device)
    ap = torch.zeros(x2.shape[0], m, dtype=torch.complex64, device=x.device)
    ap[:, :n] = x2 * a
    # circular convolution via the pow-2 kernel: conv = IFFT(FFT(ap) * fb)
    fa = e.fft_c64(ap.contiguous(), tw, False)
    prod = (fa * fb).contiguous()
    conv = e.fft_c64(prod, tw, True)
    return (conv[:, :n] * a).reshape(shape)


# ---- pairwise ------------------------------------------------------------

def pairwise_sqeuclidean(X: torch.Tensor, Y: torch.Tensor) -> torch.Tensor:
    return require_extension().pairwise_sqeuclidean(X.contiguous(), Y.contiguous())


# ---- RandomForest --------------------------------------------------------

def _thresholds_fp16_exact(nodes: torch.Tensor) -> bool:
    """True iff every heap-node threshold is exactly fp16-representable
    (then the compact 4-B-node kernel is bitwise-exact). One pass over the
    node table, cached on the tensor keyed by its mutation counter."""
    cached = getattr(nodes, "_lpx_f16exact_cache", None)
    if cached is not None and cached[0] == nodes._version:
        return cached[1]
    thr = nodes.view(-1, 2)[:, 1].contiguous().view(torch.float32)
    ok = bool(thr.to(torch.float16).to(torch.float32).eq(thr).all().item())
    nodes._lpx_f16exact_cache = (nodes._version, ok)
    return ok

def rf_predict_proba_sum(nodes: torch.Tensor, leaf_values: torch.Tensor,
                         roots: torch.Tensor, X: torch.Tensor,
                         packed: bool = False, chunks=None,
                         layout: str = None, depth: int = None) -> torch.Tensor:
    """Per-class sums over trees: out [S, C]. Chunks classes by 8 (kernel CMAX).

    packed: nodes are the depth-first 8-B layout ([n,2] int32).
    layout="heap": nodes are level-order internal nodes (complete forests).
    chunks: optional (chunk_tree_begin, max_chunk_nodes) — routes to the
    forest-in-LDS kernel (needs packed nodes)."""
    e = require_extension()
    if layout == "heap":
        assert depth is not None
        # v6 child-pair layout by default (250 vs 242 Ms/s at the bench
        # shape: one random 16-B L2 request per level instead of one 8-B —
        # the kernel is L2 REQUEST-bound, profiles/README). The repack is a
        # one-time gather cached ON the nodes tensor, keyed by torch's
        # mutation counter (_version): any in-place write (e.g. a new
        # broadcast of model bytes into the same tensor) invalidates the
        # cache and the pair layout is rebuilt on the next predict.
        # LPX_RF_PAIR=0 falls back to the v4 single-node kernel.
        # The two-level LDS variant (top>=0) measured SLOWER (146 Ms/s):
        # top nodes are already L1-hot and the extra LDS halves occupancy.
        use_pair = depth >= 2 and os.environ.get("LPX_RF_PAIR", "1") == "1"
        # v8 compact nodes (u16 feat + fp16 threshold, 8-B pair records):
        # halves the bytes of every random request (+6% measured at the
        # bench shape). Routing preserves exact-match semantics:
        #   LPX_RF_COMPACT=auto (default): compact ONLY when every
        #     threshold is exactly fp16-representable — then the compact
        #     traversal is bitwise-identical to the f32 kernels (tested).
        #     sklearn-fitted f32 forests fail the check and keep the exact
        #     pair kernel.
        #   =1 forces compact (fp16-rounded decisions), =0 disables.
        cmode = os.environ.get("LPX_RF_COMPACT", "auto")
        use_compact = use_pair and cmode != "0" and (
            cmode == "1" or _thresholds_fp16_exact(nodes))

        def fn(n, lv, r, x):
            if use_compact and x.shape[1] <= 0xFFFF:
                cached = getattr(n, "_lpx_cpair_cache", None)
                if cached is None or cached[0] != n._version:
                    stride = (1 << depth) - 1
                    T = n.shape[0] // stride
                    nh = n.view(T, stride, 2)
                    thr16 = (nh[..., 1].contiguous().view(torch.float32)
                             .to(torch.float16).view(torch.int16)
                             .to(torch.int32) & 0xFFFF)
                    cn = (nh[..., 0] << 16) | thr16  # [T, stride] i32
                    ppt = (1 << (depth - 1)) - 1
                    idx = torch.arange(ppt, device=n.device)
                    cp = (torch.stack([cn[:, 2 * idx + 1], cn[:, 2 * idx + 2]],
                                      dim=2).reshape(T * ppt, 2).contiguous())
                    cached = (n._version, cp, cn[:, 0].contiguous())
                    n._lpx_cpair_cache = cached
                _, cp, rts = cached
                # fp16 leaf table when exact (one 16-B leaf request): same
                # auto-exact rule as the node thresholds
                lvc = getattr(lv, "_lpx_lv16_cache", None)
                if lvc is None or lvc[0] != lv._version:
                    lv16 = lv.to(torch.float16)
                    ok = bool(lv16.to(torch.float32).eq(lv).all().item())
                    lvc = (lv._version, lv16.contiguous() if ok else None)
                    lv._lpx_lv16_cache = lvc
                # tree-ILP 4: four independent chains per lane fill the
                # fetch latency (A/B: 280 -> 314 Ms/s at the bench shape)
                if lvc[1] is not None:
                    return e.rf_predict_heap_cpair16(cp, rts, lvc[1], depth,
                                                     x, 41)
                return e.rf_predict_heap_cpair(cp, rts, lv, depth, x, 41)
            if use_pair:
                cached = getattr(n, "_lpx_pair_cache", None)
                if cached is None or cached[0] != n._version:
                    stride = (1 << depth) - 1
                    T = n.shape[0] // stride
                    nh = n.view(T, stride, 2)
                    ppt = (1 << (depth - 1)) - 1
                    idx = torch.arange(ppt, device=n.device)
                    pairs = (torch.cat([nh[:, 2 * idx + 1, :],
                                        nh[:, 2 * idx + 2, :]], dim=2)
                             .reshape(T * ppt, 4).contiguous())
                    cached = (n._version, pairs, nh[:, 0, :].contiguous())
                    n._lpx_pair_cache = cached
                _, pairs, rts = cached
                return e.rf_predict_heap_pair(pairs, rts, lv, depth, x, 1)
            return e.rf_predict_heap(n, lv, depth, x, 1, -2)
    elif packed and chunks is not None:
        ctb, max_nodes = chunks

        def fn(n, lv, r, x):
            return e.rf_predict8_lds(n, lv, r, ctb, max_nodes, x)
    elif packed:
        fn = e.rf_predict8
    else:
        fn = e.rf_predict
    C = leaf_values.shape[1]
    if C <= 8:
        return fn(nodes, leaf_values, roots, X.contiguous())
    outs = []
    for c0 in range(0, C, 8):
        lv = leaf_values[:, c0 : c0 + 8].contiguous()
        outs.append(fn(nodes, lv, roots, X.contiguous()))
    return torch.cat(outs, dim=1)
