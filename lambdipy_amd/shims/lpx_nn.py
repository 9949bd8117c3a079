"""lpx_nn — torch-free NN serving runtime over liblpx_nn_gfx950.so (ctypes).

The config-2/3/5 counterpart of lpx_rt's RF path:
a handler imports this shim (~ms), dlopens the AOT gfx950 library (no JIT)
and serves bf16 matmul / FFT+SpMV / fp8 ResNet-50 with zero PyTorch import —
the ~1.3 s libtorch import leaves the cold-start metric entirely, and the
config-5 layer shrinks from "needs a multi-GB torch runtime" to a few MB
(liblpx_nn + this file + the handler).

The ResNet-50 executor mirrors shims/lpx_torch.py's fp8 inference graph
(folded-BN conv+bias+relu chains, frozen activation scales after one
calibration batch, fused e4m3 epilogues between aligned convs, fused
residual adds) and can replay the whole ~120-kernel forward as ONE
hipGraph launch (capture()).
"""

from __future__ import annotations

import ctypes
import math
import os
import struct
import sys
from typing import Dict, List, Optional, Tuple

try:
    import lpx_rt
except ImportError:  # in-repo use (layer shims are flat modules)
    from lambdipy_amd.shims import lpx_rt

E4M3_MAX = 448.0
CAL_HEADROOM = 2.0  # frozen activation scales get 2x headroom (lpx_torch.py)

_lib = None


class NNError(RuntimeError):
    pass


def _find_lib() -> str:
    name = "liblpx_nn_gfx950.so"
    here = os.path.dirname(os.path.abspath(__file__))
    cands = [os.path.join(here, name),
             os.path.join(here, "lpx_ops", name),
             os.path.join(os.path.dirname(here), "ops", name)]
    for sp in sys.path:
        cands.append(os.path.join(sp, "lpx_ops", name))
        cands.append(os.path.join(sp, name))
    for c in cands:
        if os.path.exists(c):
            return c
    raise NNError(f"{name} not found (searched {len(cands)} locations); "
                  "build it with lambdipy_amd.builder.ensure_nn_lib()")


def lib() -> ctypes.CDLL:
    """liblpx_nn with argtypes declared on every entry point (fail-loud:
    a GPU box without the library raises, never falls back)."""
    global _lib
    if _lib is not None:
        return _lib
    L = ctypes.CDLL(_find_lib(), mode=ctypes.RTLD_GLOBAL)
    vp, i32, u32, i64, f32, sz = (ctypes.c_void_p, ctypes.c_int,
                                  ctypes.c_uint, ctypes.c_long,
                                  ctypes.c_float, ctypes.c_size_t)
    pp = ctypes.POINTER(vp)
    sigs = {
        "lpxn_stream_create": [pp],
        "lpxn_stream_destroy": [vp],
        "lpxn_stream_sync": [vp],
        "lpxn_memset": [vp, i32, sz, vp],
        "lpxn_graph_begin": [vp],
        "lpxn_graph_end": [vp, pp],
        "lpxn_graph_launch": [vp, vp],
        "lpxn_graph_destroy": [vp],
        "lpxn_randn": [vp, i64, u32, f32, vp],
        "lpxn_randn_c64": [vp, i64, u32, vp],
        "lpxn_amax": [vp, vp, i64, vp],
        "lpxn_checksum_f32": [vp, vp, i64, vp],
        "lpxn_gap_nhwc": [vp, vp, i32, i32, i32, vp],
        "lpxn_add_relu": [vp, vp, vp, i64, vp],
        "lpxn_argmax_rows": [vp, vp, i64, i32, vp],
        "lpxn_f32_to_bf16": [vp, vp, i64, vp],
        "lpxn_quantize_fp8": [vp, vp, i64, f32, vp],
        "lpxn_im2col_fp8": [vp, vp, i32, i32, i32, i32, i32, i32, i32, i32,
                            i32, i32, f32, vp],
        "lpxn_gemm_fp8": [vp, vp, vp, vp, vp, i32, i32, i32, f32, i32, vp],
        "lpxn_gemm_fp8_q8": [vp, vp, vp, vp, i32, i32, i32, f32, i32, f32, vp],
        "lpxn_conv_fp8": [vp, vp, vp, vp, i32, i32, i32, i32, i32, i32, i32,
                          i32, i32, i32, i32, f32, i32, vp],
        "lpxn_conv_fp8_q8": [vp, vp, vp, vp, i32, i32, i32, i32, i32, i32,
                             i32, i32, i32, i32, i32, f32, i32, f32, vp],
        "lpxn_maxpool_nhwc": [vp, vp, i32, i32, i32, i32, i32, i32, i32, vp],
        "lpxn_gemm_bf16": [vp, vp, vp, i32, i32, i32, vp],
        "lpxn_gemm_bf16_bf16out": [vp, vp, vp, i32, i32, i32, vp],
        "lpxn_fft_twiddles": [vp, i32, vp],
        "lpxn_fft_c64": [vp, vp, vp, i32, i32, i32, vp],
        "lpxn_csrgen": [vp, vp, vp, i32, i32, i32, u32, vp],
        "lpxn_spmv_csr": [vp, vp, vp, vp, vp, i32, i64, vp],
        "lpxn_gemv_f32": [vp, vp, vp, i32, i32, i32, vp],
        "lpxn_pairwise": [vp, vp, vp, vp, vp, i32, i32, i32, vp],
        "lpxn_row_sqnorm": [vp, vp, i32, i32, vp],
    }
    for fname, args in sigs.items():
        f = getattr(L, fname)
        f.argtypes = args
        f.restype = i32
    L.lpxn_error_string.argtypes = [i32]
    L.lpxn_error_string.restype = ctypes.c_char_p
    _lib = L
    return L


def _ck(rc: int, what: str) -> None:
    if rc != 0:
        try:
            msg = lib().lpxn_error_string(rc)
            msg = msg.decode() if msg else str(rc)
        except Exception:  # noqa: BLE001 — error path best effort
            msg = str(rc)
        raise NNError(f"{what} failed: hip error {rc} ({msg})")


class Stream:
    def __init__(self):
        p = ctypes.c_void_p(0)
        _ck(lib().lpxn_stream_create(ctypes.byref(p)), "stream_create")
        self.h = p.value

    def sync(self) -> None:
        _ck(lib().lpxn_stream_sync(self.h), "stream_sync")


class Pool:
    """Caching device allocator (free-list keyed by 256-B-rounded size).

    Deterministic reuse: the same alloc/free sequence returns the same
    pointers, which is what makes hipGraph capture of a pooled forward
    safe (buffer addresses recorded in the graph stay valid)."""

    def __init__(self):
        self.free_lists: Dict[int, List[int]] = {}
        self.live: Dict[int, int] = {}  # ptr -> rounded size

    @staticmethod
    def _round(n: int) -> int:
        return (max(n, 1) + 255) & ~255

    def alloc(self, nbytes: int) -> int:
        r = self._round(nbytes)
        fl = self.free_lists.get(r)
        if fl:
            p = fl.pop()
        else:
            p = lpx_rt.malloc(r)
        self.live[p] = r
        return p

    def free(self, ptr: int) -> None:
        r = self.live.pop(ptr)
        self.free_lists.setdefault(r, []).append(ptr)


def amax(x: int, n: int, pool: Pool, stream: Stream) -> float:
    """Device abs-max -> host float (syncs the stream; calibration only)."""
    L = lib()
    buf = pool.alloc(4)
    _ck(L.lpxn_memset(buf, 0, 4, stream.h), "memset(amax)")
    _ck(L.lpxn_amax(x, buf, n, stream.h), "amax")
    stream.sync()
    raw = ctypes.c_uint(0)
    lpx_rt.memcpy_d2h(ctypes.addressof(raw), buf, 4)
    pool.free(buf)
    return struct.unpack("f", struct.pack("I", raw.value))[0]


def checksum(x: int, n: int, pool: Pool, stream: Stream) -> float:
    L = lib()
    buf = pool.alloc(8)
    _ck(L.lpxn_memset(buf, 0, 8, stream.h), "memset(checksum)")
    _ck(L.lpxn_checksum_f32(x, buf, n, stream.h), "checksum_f32")
    stream.sync()
    raw = ctypes.c_double(0)
    lpx_rt.memcpy_d2h(ctypes.addressof(raw), buf, 8)
    pool.free(buf)
    return raw.value


# ---- fp8 ResNet-50 executor ----------------------------------------------

class _Conv:
    """One folded-BN fp8 conv layer: quantized weights + f32 bias living in
    the model's weight slab, per-tensor weight scale, frozen activation
    scale after calibration (mirrors lpx_torch.Conv2dFP8)."""

    __slots__ = ("cin", "cout", "k", "stride", "pad", "relu",
                 "wq", "bias", "wscale", "sa", "wbytes")

    def __init__(self, cin, cout, k, stride, pad, relu):
        self.cin, self.cout, self.k = cin, cout, k
        self.stride, self.pad, self.relu = stride, pad, relu
        self.wq = 0        # device ptr (e4m3 [cout, k*k*cin]) in the slab
        self.bias = 0      # device ptr (f32 [cout]) in the slab
        self.wscale = 1.0
        self.sa: Optional[float] = None
        self.wbytes = cout * k * k * cin

    def out_shape(self, H: int, W: int) -> Tuple[int, int]:
        ho = (H + 2 * self.pad - self.k) // self.stride + 1
        wo = (W + 2 * self.pad - self.k) // self.stride + 1
        return ho, wo


RESNET50_LAYERS = [(64, 256, 3, 1), (128, 512, 4, 2), (256, 1024, 6, 2),
                   (512, 2048, 3, 2)]


def resnet50_spec(n_classes: int = 1000):
    """The ResNet-50 v1 conv table (no device work — CPU-testable).

    Returns (convs, blocks, conv1, fc): convs is every conv incl. the FC
    head in slab order; blocks is [(c1, c2, c3, down-or-None)]."""
    convs: List[_Conv] = []
    blocks: List[Tuple[_Conv, _Conv, _Conv, Optional[_Conv]]] = []
    conv1 = _Conv(3, 64, 7, 2, 3, True)
    convs.append(conv1)
    cin = 64
    for mid, cout, n, stride in RESNET50_LAYERS:
        for i in range(n):
            st = stride if i == 0 else 1
            c1 = _Conv(cin, mid, 1, 1, 0, True)
            c2 = _Conv(mid, mid, 3, st, 1, True)
            c3 = _Conv(mid, cout, 1, 1, 0, False)
            dn = _Conv(cin, cout, 1, st, 0, False) \
                if (st != 1 or cin != cout) else None
            for c in (c1, c2, c3, dn):
                if c:
                    convs.append(c)
            blocks.append((c1, c2, c3, dn))
            cin = cout
    fc = _Conv(2048, n_classes, 1, 1, 0, False)  # reuses the conv fields
    convs.append(fc)
    return convs, blocks, conv1, fc


class ResNet50Lite:
    """ResNet-50 v1 fp8 inference without torch (random-init weights,
    BASELINE.json config 5). Weight slab layout (single hipMalloc so a
    multi-GPU broadcast is ONE RCCL call): all e4m3 weights, then all f32
    biases, 256-B aligned each."""

    def __init__(self, n_classes: int = 1000, seed: int = 3,
                 pool: Optional[Pool] = None, stream: Optional[Stream] = None,
                 alloc_only: bool = False):
        self.n_classes = n_classes
        self.pool = pool or Pool()
        self.stream = stream or Stream()
        self.convs, self.blocks, self.conv1, self.fc = resnet50_spec(n_classes)
        self._graph = None
        self._graph_key = None
        self._logits = 0
        self._owned: List[int] = []
        self._lay_out_slab()
        if not alloc_only:
            self._init_weights(seed)

    # -- weights ------------------------------------------------------------

    def _lay_out_slab(self) -> None:
        off = 0
        offs_w, offs_b = [], []
        for c in self.convs:
            offs_w.append(off)
            off += (c.wbytes + 255) & ~255
        for c in self.convs:
            offs_b.append(off)
            nb = (self.n_classes if c is self.fc else c.cout) * 4
            off += (nb + 255) & ~255
        self.slab_bytes = off
        self.slab = lpx_rt.malloc(off)
        for c, ow, ob in zip(self.convs, offs_w, offs_b):
            c.wq = self.slab + ow
            c.bias = self.slab + ob

    def _init_weights(self, seed: int) -> None:
        """Device-side random init + per-tensor e4m3 quantization (the
        lpx_torch flow: w ~ N(0, 1/fan_in), bias ~ 0.01*N(0,1), scale =
        amax/448). No host weight bytes at all."""
        L = lib()
        st = self.stream
        scratch = self.pool.alloc(max(c.wbytes for c in self.convs) * 4)
        for i, c in enumerate(self.convs):
            n = c.wbytes
            fan_in = c.cin * c.k * c.k if c is not self.fc else 2048
            _ck(L.lpxn_randn(scratch, n, (seed * 1000003 + i * 97) & 0xFFFFFFFF,
                             1.0 / math.sqrt(fan_in), st.h), "randn(w)")
            c.wscale = max(amax(scratch, n, self.pool, st), 1e-8) / E4M3_MAX
            _ck(L.lpxn_quantize_fp8(scratch, c.wq, n, 1.0 / c.wscale, st.h),
                "quantize(w)")
            nb = self.n_classes if c is self.fc else c.cout
            if c is self.fc:
                _ck(L.lpxn_memset(c.bias, 0, nb * 4, st.h), "memset(fc bias)")
            else:
                _ck(L.lpxn_randn(c.bias, nb,
                                 (seed * 2000003 + i * 193) & 0xFFFFFFFF,
                                 0.01, st.h), "randn(b)")
        st.sync()
        self.pool.free(scratch)

    def export_scales(self) -> dict:
        return {"w": [c.wscale for c in self.convs],
                "sa": [c.sa for c in self.convs]}

    def import_scales(self, s: dict) -> None:
        for c, w, sa in zip(self.convs, s["w"], s["sa"]):
            c.wscale, c.sa = w, sa

    # -- checkpointing -------------------------------------------------------

    def save_weights(self, path: str) -> None:
        """One file: JSON header (scales incl. frozen activation scales if
        calibrated) + the raw weight slab (e4m3 weights + f32 biases)."""
        import json

        buf = (ctypes.c_ubyte * self.slab_bytes)()
        self.stream.sync()
        lpx_rt.memcpy_d2h(ctypes.addressof(buf), self.slab, self.slab_bytes)
        hdr = json.dumps({"format": 1, "slab_bytes": self.slab_bytes,
                          "n_classes": self.n_classes,
                          "scales": self.export_scales()}).encode()
        with open(path, "wb") as f:
            f.write(len(hdr).to_bytes(8, "little"))
            f.write(hdr)
            f.write(bytes(buf))

    def load_weights(self, path: str) -> None:
        """Restore slab + scales (use with alloc_only=True for a cold start
        that serves REAL weights: one sequential read + one H2D copy)."""
        import json

        with open(path, "rb") as f:
            hlen = int.from_bytes(f.read(8), "little")
            hdr = json.loads(f.read(hlen).decode())
            assert hdr["format"] == 1 and hdr["n_classes"] == self.n_classes
            assert hdr["slab_bytes"] == self.slab_bytes, "architecture mismatch"
            blob = f.read(self.slab_bytes)
        arr = (ctypes.c_ubyte * self.slab_bytes).from_buffer_copy(blob)
        lpx_rt.memcpy_h2d(self.slab, ctypes.addressof(arr), self.slab_bytes)
        self.import_scales(hdr["scales"])

    # -- layers -------------------------------------------------------------

    def _conv_f32(self, c: _Conv, x: int, N: int, H: int, W: int,
                  calibrate: bool) -> Tuple[int, int, int]:
        """Unfused conv: f32 in -> f32 out [N*HO*WO, cout] (bias+relu fused
        in the GEMM epilogue). Used by the calibration pass and any conv
        whose producer could not emit e4m3 directly."""
        L, st, pool = lib(), self.stream, self.pool
        n_act = N * H * W * c.cin
        if calibrate and c.sa is None:
            c.sa = max(amax(x, n_act, pool, st), 1e-8) / E4M3_MAX * CAL_HEADROOM
        sa = c.sa
        ho, wo = c.out_shape(H, W)
        M = N * ho * wo
        out = pool.alloc(M * c.cout * 4)
        desc = sa * c.wscale
        if c.k == 1 and c.stride == 1 and c.pad == 0:
            xq = pool.alloc(n_act)
            _ck(L.lpxn_quantize_fp8(x, xq, n_act, 1.0 / sa, st.h), "quant(a)")
            _ck(L.lpxn_gemm_fp8(xq, c.wq, out, c.bias, None, M, c.cout,
                                c.cin, desc, int(c.relu), st.h), "gemm_fp8")
            pool.free(xq)
        elif c.cin % 16 == 0:
            xq = pool.alloc(n_act)
            _ck(L.lpxn_quantize_fp8(x, xq, n_act, 1.0 / sa, st.h), "quant(a)")
            _ck(L.lpxn_conv_fp8(xq, c.wq, out, c.bias, N, H, W, c.cin, c.k,
                                c.k, ho, wo, c.stride, c.pad, c.cout, desc,
                                int(c.relu), st.h), "conv_fp8")
            pool.free(xq)
        else:  # stem (C=3): materialized im2col patch matrix
            K = c.k * c.k * c.cin
            aq = pool.alloc(M * K)
            _ck(L.lpxn_im2col_fp8(x, aq, N, H, W, c.cin, c.k, c.k, ho, wo,
                                  c.stride, c.pad, 1.0 / sa, st.h), "im2col")
            _ck(L.lpxn_gemm_fp8(aq, c.wq, out, c.bias, None, M, c.cout, K,
                                desc, int(c.relu), st.h), "gemm_fp8(stem)")
            pool.free(aq)
        return out, ho, wo

    def _conv_q8(self, c: _Conv, xq: int, N: int, H: int, W: int,
                 out_scale: Optional[float], resid: Optional[int] = None,
                 relu: Optional[bool] = None) -> Tuple[int, int, int]:
        """Fused-chain conv: e4m3 in; out e4m3 at out_scale, or f32 when
        out_scale is None (mirrors lpx_torch.Conv2dFP8.run_q)."""
        L, st, pool = lib(), self.stream, self.pool
        desc = c.sa * c.wscale
        ho, wo = c.out_shape(H, W)
        M = N * ho * wo
        if relu is None:
            relu = c.relu
        if c.k == 1 and c.stride == 1 and c.pad == 0:
            if out_scale is None:
                out = pool.alloc(M * c.cout * 4)
                _ck(L.lpxn_gemm_fp8(xq, c.wq, out, c.bias, resid, M, c.cout,
                                    c.cin, desc, int(relu), st.h), "gemm_fp8")
            else:
                out = pool.alloc(M * c.cout)
                rc = L.lpxn_gemm_fp8_q8(xq, c.wq, out, c.bias, M, c.cout,
                                        c.cin, desc, int(relu),
                                        1.0 / out_scale, st.h)
                if rc != 0:  # unaligned shape: f32 GEMM + quantize pass
                    tmp = pool.alloc(M * c.cout * 4)
                    _ck(L.lpxn_gemm_fp8(xq, c.wq, tmp, c.bias, None, M,
                                        c.cout, c.cin, desc, int(relu), st.h),
                        "gemm_fp8")
                    _ck(L.lpxn_quantize_fp8(tmp, out, M * c.cout,
                                            1.0 / out_scale, st.h), "quant")
                    pool.free(tmp)
        else:
            if out_scale is None:
                out = pool.alloc(M * c.cout * 4)
                _ck(L.lpxn_conv_fp8(xq, c.wq, out, c.bias, N, H, W, c.cin,
                                    c.k, c.k, ho, wo, c.stride, c.pad, c.cout,
                                    desc, int(relu), st.h), "conv_fp8")
            else:
                out = pool.alloc(M * c.cout)
                _ck(L.lpxn_conv_fp8_q8(xq, c.wq, out, c.bias, N, H, W, c.cin,
                                       c.k, c.k, ho, wo, c.stride, c.pad,
                                       c.cout, desc, int(relu),
                                       1.0 / out_scale, st.h), "conv_fp8_q8")
        return out, ho, wo

    # -- forward ------------------------------------------------------------

    @property
    def calibrated(self) -> bool:
        return self.fc.sa is not None and all(
            c.sa is not None for c in self.convs)

    def forward(self, x: int, N: int, H: int = 224, W: int = 224) -> int:
        """x: device ptr, NHWC f32 [N,H,W,3]. Returns logits ptr (f32
        [N, n_classes], pool-owned — freed on the next forward)."""
        if self._graph is not None and self._graph_key == (x, N, H, W):
            _ck(lib().lpxn_graph_launch(self._graph, self.stream.h),
                "graph_launch")
            return self._logits
        calibrate = not self.calibrated
        out = self._forward(x, N, H, W, calibrate)
        return out

    def _free_prev(self):
        for p in self._owned:
            self.pool.free(p)
        self._owned = []

    def _forward(self, x: int, N: int, H: int, W: int, calibrate: bool) -> int:
        L, st, pool = lib(), self.stream, self.pool
        self._free_prev()
        owned = []
        y, h, w = self._conv_f32(self.conv1, x, N, H, W, calibrate)
        # maxpool 3x3 s2 p1 (native NHWC)
        ho, wo = (h + 2 * 1 - 3) // 2 + 1, (w + 2 * 1 - 3) // 2 + 1
        mp = pool.alloc(N * ho * wo * 64 * 4)
        _ck(L.lpxn_maxpool_nhwc(y, mp, N, h, w, 64, 3, 2, 1, st.h), "maxpool")
        pool.free(y)
        cur, h, w, ccur = mp, ho, wo, 64
        for (c1, c2, c3, dn) in self.blocks:
            if calibrate:
                idn = cur
                if dn is not None:
                    idn, _, _ = self._conv_f32(dn, cur, N, h, w, calibrate)
                y1, h1, w1 = self._conv_f32(c1, cur, N, h, w, calibrate)
                y2, h2, w2 = self._conv_f32(c2, y1, N, h1, w1, calibrate)
                pool.free(y1)
                y3, h3, w3 = self._conv_f32(c3, y2, N, h2, w2, calibrate)
                pool.free(y2)
                nout = N * h3 * w3 * c3.cout
                _ck(L.lpxn_add_relu(y3, idn, y3, nout, st.h), "add_relu")
                if idn is not cur:
                    pool.free(idn)
                pool.free(cur)
                cur, h, w, ccur = y3, h3, w3, c3.cout
            else:
                # fused chain (lpx_torch.Bottleneck fused branch): one
                # quantize of the block input; conv1/conv2 emit e4m3 at the
                # NEXT conv's frozen scale; conv3 fuses the identity add
                # when the 8-phase shape holds.
                n_act = N * h * w * ccur
                xq = pool.alloc(n_act)
                _ck(L.lpxn_quantize_fp8(cur, xq, n_act, 1.0 / c1.sa, st.h),
                    "quant(block)")
                if dn is None:
                    idn = cur
                else:
                    # dn calibrated on the same tensor as c1 -> same sa
                    idn, _, _ = self._conv_q8(dn, xq, N, h, w, None)
                q1, h1, w1 = self._conv_q8(c1, xq, N, h, w, c2.sa)
                q2, h2, w2 = self._conv_q8(c2, q1, N, h1, w1, c3.sa)
                pool.free(q1)
                M3 = N * h2 * w2
                fuse = (M3 % 256 == 0 and c3.cout % 256 == 0
                        and c3.cin % 128 == 0)
                if fuse:
                    y3, h3, w3 = self._conv_q8(c3, q2, N, h2, w2, None,
                                               resid=idn, relu=True)
                else:
                    y3, h3, w3 = self._conv_q8(c3, q2, N, h2, w2, None)
                    _ck(L.lpxn_add_relu(y3, idn, y3,
                                        N * h3 * w3 * c3.cout, st.h),
                        "add_relu")
                pool.free(q2)
                pool.free(xq)
                if idn is not cur:
                    pool.free(idn)
                pool.free(cur)
                cur, h, w, ccur = y3, h3, w3, c3.cout
        # global average pool -> [N, 2048]
        gap = pool.alloc(N * 2048 * 4)
        _ck(L.lpxn_gap_nhwc(cur, gap, N, h * w, 2048, st.h), "gap")
        pool.free(cur)
        fc = self.fc
        if calibrate and fc.sa is None:
            fc.sa = max(amax(gap, N * 2048, pool, st), 1e-8) \
                / E4M3_MAX * CAL_HEADROOM
        gq = pool.alloc(N * 2048)
        _ck(L.lpxn_quantize_fp8(gap, gq, N * 2048, 1.0 / fc.sa, st.h),
            "quant(fc)")
        pool.free(gap)
        logits = pool.alloc(N * self.n_classes * 4)
        _ck(L.lpxn_gemm_fp8(gq, fc.wq, logits, fc.bias, None, N,
                            self.n_classes, 2048, fc.sa * fc.wscale, 0, st.h),
            "gemm_fp8(fc)")
        pool.free(gq)
        owned.append(logits)
        self._owned = owned
        self._logits = logits
        return logits

    def capture(self, x: int, N: int, H: int = 224, W: int = 224) -> bool:
        """Record the fused forward as a hipGraph (one launch per batch
        afterwards). Requires calibration to have run (sync-free path).
        Returns False (and stays eager) if capture fails."""
        assert self.calibrated, "run one calibration batch first"
        L = lib()
        # warm the pool so capture-time allocations hit the free lists
        self._forward(x, N, H, W, False)
        self.stream.sync()
        try:
            _ck(L.lpxn_graph_begin(self.stream.h), "graph_begin")
            logits = self._forward(x, N, H, W, False)
            g = ctypes.c_void_p(0)
            _ck(L.lpxn_graph_end(self.stream.h, ctypes.byref(g)), "graph_end")
        except NNError:
            return False
        self._graph = g.value
        self._graph_key = (x, N, H, W)
        self._logits = logits
        return True


# ---- config-2: bf16 matmul ------------------------------------------------

class MatmulLite:
    """4096^2 bf16 MFMA matmul serving (config 2), torch-free."""

    def __init__(self, n: int = 4096, seed: int = 0,
                 pool: Optional[Pool] = None, stream: Optional[Stream] = None):
        self.n = n
        self.pool = pool or Pool()
        self.stream = stream or Stream()
        L, st = lib(), self.stream
        scratch = self.pool.alloc(n * n * 4)
        self.A = self.pool.alloc(n * n * 2)
        self.Bt = self.pool.alloc(n * n * 2)
        _ck(L.lpxn_randn(scratch, n * n, seed * 7919 + 11, 1.0, st.h), "randn")
        _ck(L.lpxn_f32_to_bf16(scratch, self.A, n * n, st.h), "to_bf16")
        _ck(L.lpxn_randn(scratch, n * n, seed * 7919 + 23, 1.0, st.h), "randn")
        _ck(L.lpxn_f32_to_bf16(scratch, self.Bt, n * n, st.h), "to_bf16")
        self.pool.free(scratch)
        self.C = self.pool.alloc(n * n * 4)

    def run(self) -> int:
        _ck(lib().lpxn_gemm_bf16(self.A, self.Bt, self.C, self.n, self.n,
                                 self.n, self.stream.h), "gemm_bf16")
        return self.C

    def checksum(self) -> float:
        return checksum(self.C, self.n * self.n, self.pool, self.stream)


# ---- config-3: FFT + CSR SpMV ---------------------------------------------

class FftSpmvLite:
    """Batched c64 FFT + CSR SpMV serving (config 3), torch-free; synthetic
    inputs generated on device (no numpy import on the cold path)."""

    def __init__(self, batch: int = 1024, n: int = 1024, rows: int = 262144,
                 nnz: int = 32, seed: int = 0, pool: Optional[Pool] = None,
                 stream: Optional[Stream] = None):
        self.batch, self.n, self.rows, self.nnz = batch, n, rows, nnz
        self.pool = pool or Pool()
        self.stream = stream or Stream()
        L, st, p = lib(), self.stream, self.pool
        self.x = p.alloc(batch * n * 8)
        self.fout = p.alloc(batch * n * 8)
        self.tw = p.alloc(n * 4)  # n/2 complex = n floats
        _ck(L.lpxn_randn_c64(self.x, batch * n, seed * 31 + 7, st.h), "randn")
        _ck(L.lpxn_fft_twiddles(self.tw, n, st.h), "twiddles")
        self.indptr = p.alloc((rows + 1) * 4)
        self.indices = p.alloc(rows * nnz * 4)
        self.data = p.alloc(rows * nnz * 4)
        self.vec = p.alloc(rows * 4)
        self.y = p.alloc(rows * 4)
        _ck(L.lpxn_csrgen(self.indptr, self.indices, self.data, rows, nnz,
                          rows, seed * 131 + 3, st.h), "csrgen")
        _ck(L.lpxn_randn(self.vec, rows, seed * 131 + 17, 1.0, st.h), "randn")

    def run(self) -> Tuple[int, int]:
        L, st = lib(), self.stream
        _ck(L.lpxn_fft_c64(self.x, self.fout, self.tw, self.batch, self.n, 0,
                           st.h), "fft_c64")
        _ck(L.lpxn_spmv_csr(self.indptr, self.indices, self.data, self.vec,
                            self.y, self.rows, self.rows * self.nnz, st.h),
            "spmv_csr")
        return self.fout, self.y

    def checksum(self) -> float:
        a = checksum(self.fout, self.batch * self.n * 2, self.pool,
                     self.stream)
        b = checksum(self.y, self.rows, self.pool, self.stream)
        return a + b
