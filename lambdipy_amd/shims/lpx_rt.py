"""lpx_rt — torch-free gfx950 runtime (ctypes over libamdhip64 + librccl).

The cold-start-critical path of SURVEY.md §2.2 N11/N15: a handler imports
this shim (~ms), hipModuleLoads the AOT standalone_gfx950.hsaco (no JIT) and
launches kernels directly; multi-GPU fan-out uses RCCL's single-process
ncclCommInitAll + broadcast/allGather over xGMI. No PyTorch import anywhere
on this path — the ~0.8 s libtorch import leaves the metric.
"""

from __future__ import annotations

import ctypes
import os
from typing import Optional, Sequence

_hip = None
_rccl = None


def hip():
    global _hip
    if _hip is None:
        _hip = ctypes.CDLL("libamdhip64.so", mode=ctypes.RTLD_GLOBAL)
        vp, i32, u32, sz = (ctypes.c_void_p, ctypes.c_int, ctypes.c_uint,
                            ctypes.c_size_t)
        _hip.hipGetErrorString.restype = ctypes.c_char_p
        _hip.hipGetErrorString.argtypes = [i32]
        _hip.hipGetDeviceCount.argtypes = [ctypes.POINTER(i32)]
        _hip.hipSetDevice.argtypes = [i32]
        _hip.hipMalloc.argtypes = [ctypes.POINTER(vp), sz]
        _hip.hipMemcpy.argtypes = [vp, vp, sz, i32]
        _hip.hipDeviceSynchronize.argtypes = []
        _hip.hipModuleLoad.argtypes = [ctypes.POINTER(vp), ctypes.c_char_p]
        _hip.hipModuleGetFunction.argtypes = [ctypes.POINTER(vp), vp,
                                              ctypes.c_char_p]
        _hip.hipModuleLaunchKernel.argtypes = [
            vp, u32, u32, u32, u32, u32, u32, u32, vp,
            ctypes.POINTER(vp), ctypes.POINTER(vp)]
        for f in (_hip.hipGetDeviceCount, _hip.hipSetDevice, _hip.hipMalloc,
                  _hip.hipMemcpy, _hip.hipDeviceSynchronize,
                  _hip.hipModuleLoad, _hip.hipModuleGetFunction,
                  _hip.hipModuleLaunchKernel):
            f.restype = i32
    return _hip


def rccl():
    global _rccl
    if _rccl is None:
        _rccl = ctypes.CDLL("librccl.so", mode=ctypes.RTLD_GLOBAL)
    return _rccl


class HipError(RuntimeError):
    pass


def _ck(code: int, what: str = "hip call") -> None:
    if code != 0:
        msg = hip().hipGetErrorString(code)
        raise HipError(f"{what} failed: {msg.decode() if msg else code}")


# ---- device/memory -------------------------------------------------------

def device_count() -> int:
    n = ctypes.c_int(0)
    _ck(hip().hipGetDeviceCount(ctypes.byref(n)), "hipGetDeviceCount")
    return n.value


def set_device(d: int) -> None:
    _ck(hip().hipSetDevice(d), "hipSetDevice")


def malloc(nbytes: int) -> int:
    p = ctypes.c_void_p(0)
    _ck(hip().hipMalloc(ctypes.byref(p), ctypes.c_size_t(nbytes)), "hipMalloc")
    return p.value


def _vp(p):
    return p if isinstance(p, ctypes.c_void_p) else ctypes.c_void_p(p)


def memcpy_h2d(dst: int, src, nbytes: int) -> None:
    _ck(hip().hipMemcpy(_vp(dst), _vp(src), ctypes.c_size_t(nbytes), 1),
        "hipMemcpy H2D")


def memcpy_d2h(dst, src: int, nbytes: int) -> None:
    _ck(hip().hipMemcpy(_vp(dst), _vp(src), ctypes.c_size_t(nbytes), 2),
        "hipMemcpy D2H")


def sync() -> None:
    _ck(hip().hipDeviceSynchronize(), "hipDeviceSynchronize")


# ---- module / kernels ----------------------------------------------------

class Module:
    """hipModule wrapper over the AOT .hsaco (found next to this file or in
    lpx_ops/, falling back to the in-repo ops dir)."""

    SEARCH = ("standalone_gfx950.hsaco",
              os.path.join("lpx_ops", "standalone_gfx950.hsaco"))

    def __init__(self, path: Optional[str] = None):
        if path is None:
            here = os.path.dirname(os.path.abspath(__file__))
            candidates = [os.path.join(here, s) for s in self.SEARCH]
            candidates += [os.path.join(os.path.dirname(here), "ops",
                                        "standalone_gfx950.hsaco")]
            for sp in __import__("sys").path:
                candidates.append(os.path.join(sp, "lpx_ops",
                                               "standalone_gfx950.hsaco"))
            path = next((c for c in candidates if os.path.exists(c)), None)
            if path is None:
                raise HipError("standalone_gfx950.hsaco not found (AOT code "
                               "object missing from the layer)")
        self.path = path
        self._mod = ctypes.c_void_p(0)
        _ck(hip().hipModuleLoad(ctypes.byref(self._mod), path.encode()),
            "hipModuleLoad")
        self._fns = {}

    def fn(self, name: str):
        f = self._fns.get(name)
        if f is None:
            f = ctypes.c_void_p(0)
            _ck(hip().hipModuleGetFunction(ctypes.byref(f), self._mod,
                                           name.encode()),
                f"hipModuleGetFunction({name})")
            self._fns[name] = f
        return f

    def launch(self, name: str, grid: int, block: int, shared: int,
               arg_struct: ctypes.Structure) -> None:
        buf = ctypes.byref(arg_struct)
        size = ctypes.c_size_t(ctypes.sizeof(arg_struct))
        # HIP_LAUNCH_PARAM_BUFFER_POINTER=1, _BUFFER_SIZE=2, _END=3
        extra = (ctypes.c_void_p * 5)(
            ctypes.c_void_p(1), ctypes.cast(buf, ctypes.c_void_p),
            ctypes.c_void_p(2), ctypes.cast(ctypes.byref(size), ctypes.c_void_p),
            ctypes.c_void_p(3))
        _ck(hip().hipModuleLaunchKernel(self.fn(name), grid, 1, 1, block, 1, 1,
                                        shared, None, None, extra),
            f"hipModuleLaunchKernel({name})")


# ---- kernel arg structs (match standalone.hip signatures) ----------------

class RfHeapArgs(ctypes.Structure):
    _fields_ = [("nodes", ctypes.c_void_p), ("leaf_values", ctypes.c_void_p),
                ("n_trees", ctypes.c_int), ("depth", ctypes.c_int),
                ("X", ctypes.c_void_p), ("out", ctypes.c_void_p),
                ("S", ctypes.c_int), ("F", ctypes.c_int), ("C", ctypes.c_int)]


class ArgmaxArgs(ctypes.Structure):
    _fields_ = [("x", ctypes.c_void_p), ("out", ctypes.c_void_p),
                ("rows", ctypes.c_long), ("cols", ctypes.c_int)]


class XgenArgs(ctypes.Structure):
    _fields_ = [("out", ctypes.c_void_p), ("n", ctypes.c_long),
                ("seed", ctypes.c_uint)]


def rf_heap(mod: Module, nodes: int, leaves: int, n_trees: int, depth: int,
            X: int, out: int, S: int, F: int, C: int) -> None:
    blocks = min((S + 255) // 256, 4096) or 1
    shared = 256 * (F + 1) * 4 if F <= 63 else 0
    mod.launch("lpx_rf_heap", blocks, 256, shared,
               RfHeapArgs(nodes, leaves, n_trees, depth, X, out, S, F, C))


def argmax_rows(mod: Module, x: int, out: int, rows: int, cols: int) -> None:
    blocks = min((rows + 255) // 256, 4096) or 1
    mod.launch("lpx_argmax_rows", blocks, 256, 0, ArgmaxArgs(x, out, rows, cols))


def xgen(mod: Module, out: int, n: int, seed: int) -> None:
    blocks = min((n // 2 + 255) // 256, 4096) or 1
    mod.launch("lpx_xgen", blocks, 256, 0, XgenArgs(out, n, seed))


class ForestGenNodesArgs(ctypes.Structure):
    _fields_ = [("nodes", ctypes.c_void_p), ("n", ctypes.c_long),
                ("n_features", ctypes.c_int), ("seed", ctypes.c_uint)]


class ForestGenLeavesArgs(ctypes.Structure):
    _fields_ = [("leaves", ctypes.c_void_p), ("n_rows", ctypes.c_long),
                ("C", ctypes.c_int), ("seed", ctypes.c_uint)]


class ForestGenCPairsArgs(ctypes.Structure):
    _fields_ = [("cpairs", ctypes.c_void_p), ("croots", ctypes.c_void_p),
                ("n_trees", ctypes.c_int), ("depth", ctypes.c_int),
                ("n_features", ctypes.c_int), ("seed", ctypes.c_uint)]


class ForestGenLeaves16Args(ctypes.Structure):
    _fields_ = [("leaves", ctypes.c_void_p), ("n_rows", ctypes.c_long),
                ("C", ctypes.c_int), ("seed", ctypes.c_uint)]


class RfCpair16Args(ctypes.Structure):
    _fields_ = [("cpairs", ctypes.c_void_p), ("croots", ctypes.c_void_p),
                ("leaves16", ctypes.c_void_p), ("n_trees", ctypes.c_int),
                ("depth", ctypes.c_int), ("X", ctypes.c_void_p),
                ("out", ctypes.c_void_p), ("S", ctypes.c_int),
                ("F", ctypes.c_int), ("C", ctypes.c_int)]


class ChecksumArgs(ctypes.Structure):
    _fields_ = [("x", ctypes.c_void_p), ("out", ctypes.c_void_p),
                ("n", ctypes.c_long)]


def forestgen(mod: Module, nodes: int, n_internal_total: int, n_features: int,
              leaves: int, n_leaf_rows: int, C: int, seed: int) -> None:
    """Device-side heap-forest generation (no host model bytes at all)."""
    b1 = min((n_internal_total + 255) // 256, 4096) or 1
    mod.launch("lpx_forestgen_nodes", b1, 256, 0,
               ForestGenNodesArgs(nodes, n_internal_total, n_features, seed))
    b2 = min((n_leaf_rows + 255) // 256, 4096) or 1
    mod.launch("lpx_forestgen_leaves", b2, 256, 0,
               ForestGenLeavesArgs(leaves, n_leaf_rows, C, seed ^ 0x5bd1e995))


def forestgen_compact(mod: Module, cpairs: int, croots: int, n_trees: int,
                      depth: int, n_features: int, leaves16: int,
                      n_leaf_rows: int, C: int, seed: int) -> None:
    """Device-side synthetic forest DIRECTLY in the compact layout the v8
    traversal wants (u16 feat | fp16 thr pair records, fp16 leaf table) —
    same node/leaf values as forestgen (shared per-index hash)."""
    ppt = (1 << (depth - 1)) - 1
    total = n_trees * ppt
    b1 = min((total + 255) // 256, 4096) or 1
    mod.launch("lpx_forestgen_cpairs", b1, 256, 0,
               ForestGenCPairsArgs(cpairs, croots, n_trees, depth,
                                   n_features, seed))
    b2 = min((n_leaf_rows + 255) // 256, 4096) or 1
    mod.launch("lpx_forestgen_leaves16", b2, 256, 0,
               ForestGenLeaves16Args(leaves16, n_leaf_rows, C,
                                     seed ^ 0x5bd1e995))


def rf_cpair16(mod: Module, cpairs: int, croots: int, leaves16: int,
               n_trees: int, depth: int, X: int, out: int, S: int, F: int,
               C: int) -> None:
    """Compact child-pair traversal with fp16 leaves + tree-ILP 4 (the
    serving-rate kernel; mirrors the torch extension's default)."""
    blocks = min((S + 255) // 256, 4096) or 1
    shared = 256 * (F + 1) * 4 if F <= 63 else 0
    mod.launch("lpx_rf_cpair16", blocks, 256, shared,
               RfCpair16Args(cpairs, croots, leaves16, n_trees, depth, X,
                             out, S, F, C))


def checksum_i32(mod: Module, x: int, out: int, n: int) -> None:
    blocks = min((n + 255) // 256, 2048) or 1
    mod.launch("lpx_checksum_i32", blocks, 256, 0, ChecksumArgs(x, out, n))


# ---- torch-free synthetic forest (heap layout) ---------------------------

def random_heap_forest_np(n_trees: int, depth: int, n_features: int,
                          n_classes: int, seed: int = 0):
    """Complete forest in level-order (heap) layout, numpy only.

    Returns (nodes int32 [n_trees*(2^d-1), 2] {feat, thresh-bits},
    leaf_values f32 [n_trees*2^d, C] row-normalized). Identical arrays to
    lpx_sklearn.ForestSoA.random(...) for the same arguments (that path
    delegates here)."""
    import numpy as np

    rng = np.random.default_rng(seed)
    n_internal = (1 << depth) - 1
    n_leaves = 1 << depth
    total = n_trees * n_internal
    feat = rng.integers(0, n_features, total).astype(np.uint32)
    # thresholds rounded to fp16-REPRESENTABLE f32 values: an equally
    # random model (same traversal work), but the compact 4-B-node kernel
    # (ops/csrc/rf.hip v8) traverses it bitwise-identically to the f32
    # kernels — exact-match semantics preserved while halving node bytes.
    # sklearn-fitted forests keep full f32 thresholds (and the f32 kernel).
    thr = (rng.standard_normal(total).astype(np.float32)
           .astype(np.float16).astype(np.float32))
    nodes = np.stack([feat, thr.view(np.uint32)], axis=1).view(np.int32)
    lv = rng.random((n_trees * n_leaves, n_classes)).astype(np.float32)
    lv /= lv.sum(axis=1, keepdims=True)
    # fp16-representable leaf values (same reasoning as thresholds: the
    # fp16-leaf-table kernel then accumulates bitwise-identically to f32;
    # rows still sum to 1 within ~5e-4)
    lv = lv.astype(np.float16).astype(np.float32)
    return np.ascontiguousarray(nodes), np.ascontiguousarray(lv)


# ---- RCCL (single-process multi-GPU) -------------------------------------

NCCL_FLOAT32 = 7
NCCL_INT32 = 2

_rccl_typed = False


def _rccl_typed_lib():
    """librccl with explicit argtypes/restypes on every entry point we call.

    ctypes' default int-promotion happens to work on x86-64 for most of
    these, but one wrong-width argument (e.g. count as c_int instead of
    c_size_t) would corrupt silently — declare everything."""
    global _rccl_typed
    rc = rccl()
    if not _rccl_typed:
        vp, i32, sz = ctypes.c_void_p, ctypes.c_int, ctypes.c_size_t
        rc.ncclGetErrorString.restype = ctypes.c_char_p
        rc.ncclGetErrorString.argtypes = [i32]
        rc.ncclGetVersion.restype = i32
        rc.ncclGetVersion.argtypes = [ctypes.POINTER(i32)]
        rc.ncclCommInitAll.restype = i32
        rc.ncclCommInitAll.argtypes = [ctypes.POINTER(vp), i32,
                                       ctypes.POINTER(i32)]
        rc.ncclCommDestroy.restype = i32
        rc.ncclCommDestroy.argtypes = [vp]
        rc.ncclGroupStart.restype = i32
        rc.ncclGroupStart.argtypes = []
        rc.ncclGroupEnd.restype = i32
        rc.ncclGroupEnd.argtypes = []
        # (send, recv, count, dtype, [root,] comm, stream)
        rc.ncclBroadcast.restype = i32
        rc.ncclBroadcast.argtypes = [vp, vp, sz, i32, i32, vp, vp]
        rc.ncclAllGather.restype = i32
        rc.ncclAllGather.argtypes = [vp, vp, sz, i32, vp, vp]
        _rccl_typed = True
    return rc


def _nck(r: int, what: str) -> None:
    if r != 0:
        try:
            msg = _rccl_typed_lib().ncclGetErrorString(r)
            msg = msg.decode() if msg else str(r)
        except Exception:  # noqa: BLE001 — error-path best effort
            msg = str(r)
        raise HipError(f"{what} failed: nccl error {r} ({msg})")


def nccl_version() -> int:
    """RCCL version code (e.g. 22807); 0 if the library is unusable."""
    try:
        v = ctypes.c_int(0)
        _nck(_rccl_typed_lib().ncclGetVersion(ctypes.byref(v)), "ncclGetVersion")
        return v.value
    except Exception:  # noqa: BLE001 — probe, not a hard dependency
        return 0


class Comms:
    """ncclCommInitAll over g devices in ONE process (xGMI fan-out without a
    process-per-GPU rendezvous — the cold-start-cheap topology).

    init_timeout_s: watchdog on comm init — ncclCommInitAll can hang
    indefinitely when a device is wedged; a serving process must die loudly
    (the cold-start harness sees EOF and reports) rather than hang the box.
    """

    def __init__(self, ndev: int, init_timeout_s: float = 120.0):
        import threading

        self.ndev = ndev
        self.comms = (ctypes.c_void_p * ndev)()
        devs = (ctypes.c_int * ndev)(*range(ndev))
        rc = _rccl_typed_lib()
        done = threading.Event()

        def _watchdog():
            if not done.wait(init_timeout_s):
                import sys as _sys
                print(f"lpx_rt.Comms: ncclCommInitAll({ndev}) exceeded "
                      f"{init_timeout_s}s watchdog — aborting process",
                      file=_sys.stderr, flush=True)
                os._exit(124)

        wd = threading.Thread(target=_watchdog, daemon=True)
        wd.start()
        try:
            _nck(rc.ncclCommInitAll(self.comms, ndev, devs),
                 f"ncclCommInitAll({ndev})")
        finally:
            done.set()

    def broadcast(self, bufs: Sequence[int], count: int, dtype: int,
                  root: int = 0) -> None:
        rc = _rccl_typed_lib()
        _nck(rc.ncclGroupStart(), "ncclGroupStart")
        try:
            for d in range(self.ndev):
                set_device(d)
                # each rank passes device-local pointers; only root's send
                # matters
                _nck(rc.ncclBroadcast(bufs[d], bufs[d], count, dtype, root,
                                      self.comms[d], None),
                     f"ncclBroadcast(dev {d})")
        except Exception:
            rc.ncclGroupEnd()  # never leave the group open on error
            raise
        _nck(rc.ncclGroupEnd(), "ncclGroupEnd")

    def all_gather(self, send: Sequence[int], recv: Sequence[int],
                   count: int, dtype: int) -> None:
        rc = _rccl_typed_lib()
        _nck(rc.ncclGroupStart(), "ncclGroupStart")
        try:
            for d in range(self.ndev):
                set_device(d)
                _nck(rc.ncclAllGather(send[d], recv[d], count, dtype,
                                      self.comms[d], None),
                     f"ncclAllGather(dev {d})")
        except Exception:
            rc.ncclGroupEnd()
            raise
        _nck(rc.ncclGroupEnd(), "ncclGroupEnd")

    def destroy(self) -> None:
        for d in range(self.ndev):
            _nck(_rccl_typed_lib().ncclCommDestroy(self.comms[d]),
                 f"ncclCommDestroy(dev {d})")
