"""Tests for the torch-free NN serving runtime (lpx_nn / liblpx_nn).

CPU half: spec/layout invariants, handler wiring, layer assembly.
GPU half (-m gpu): lite executors vs numpy/torch references, ResNet-50
determinism + hipGraph replay equality, cold-start end-to-end.
"""

import ctypes
import os

import numpy as np
import pytest

from lambdipy_amd.shims.lpx_nn import resnet50_spec

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


# ---- CPU: spec + wiring ---------------------------------------------------

def test_resnet50_spec_shape():
    convs, blocks, conv1, fc = resnet50_spec(1000)
    # 1 stem + 16 blocks x 3 convs + 4 downsamples + fc = 54
    assert len(blocks) == 16
    assert len(convs) == 1 + 16 * 3 + 4 + 1
    n_down = sum(1 for b in blocks if b[3] is not None)
    assert n_down == 4
    params = sum(c.wbytes for c in convs)
    # ResNet-50 conv+fc weights ~25.5M params (BN folded, biases excluded)
    assert 24e6 < params < 27e6
    assert conv1.k == 7 and conv1.stride == 2 and conv1.cin == 3
    assert fc.cin == 2048 and fc.cout == 1000
    # stage transitions: block outputs are 256/512/1024/2048
    assert blocks[2][2].cout == 256 and blocks[-1][2].cout == 2048


def test_lite_handlers_registered():
    from lambdipy_amd.runtime.coldstart import (HANDLERS, LITE_GPUS_ENV,
                                                LITE_HANDLERS)

    hdir = os.path.join(REPO, "lambdipy_amd", "runtime", "handlers")
    for h in ("matmul_lite", "fftspmv_lite", "resnet50_lite", "rf_lite"):
        assert h in HANDLERS and h in LITE_HANDLERS
        assert os.path.exists(os.path.join(hdir, HANDLERS[h] + ".py"))
    assert LITE_GPUS_ENV["resnet50_lite"] == "LPX_RESNET_GPUS"


def test_install_handlers_ships_nn_runtime(tmp_path):
    from lambdipy_amd.cli import _install_handlers

    d = str(tmp_path / "layer")
    os.makedirs(d)
    _install_handlers(d, ["resnet50_lite", "matmul_lite"])
    assert os.path.exists(os.path.join(d, "lpx_nn.py"))
    assert os.path.exists(os.path.join(d, "lpx_rt.py"))
    assert os.path.exists(os.path.join(d, "liblpx_nn_gfx950.so"))
    # a handlers-only rf_lite layer must be self-contained too
    d2 = str(tmp_path / "layer2")
    os.makedirs(d2)
    _install_handlers(d2, ["rf_lite"])
    assert os.path.exists(os.path.join(d2, "standalone_gfx950.hsaco"))
    assert os.path.exists(
        os.path.join(d, "lpx_handlers", "resnet50_lite_handler.py"))


def test_nn_lib_exports():
    """The AOT library exposes the full C API (cross-compiled; symbol check
    only — no GPU needed to dlopen and resolve)."""
    from lambdipy_amd.builder.wheel_builder import ensure_nn_lib

    lib = ctypes.CDLL(ensure_nn_lib())
    for sym in ("lpxn_gemm_bf16", "lpxn_gemm_fp8", "lpxn_conv_fp8_q8",
                "lpxn_fft_c64", "lpxn_spmv_csr", "lpxn_randn", "lpxn_amax",
                "lpxn_gap_nhwc", "lpxn_maxpool_nhwc", "lpxn_graph_begin",
                "lpxn_csrgen", "lpxn_argmax_rows"):
        assert hasattr(lib, sym), f"missing symbol {sym}"


def test_early_extract_covers_nn_files():
    """The lite cold-start path must eagerly extract everything the lite
    handlers import (a deferred liblpx_nn would deadlock the first dlopen)."""
    # replicate the predicate from cold_start's lite branch
    def early(n):
        base = os.path.basename(n)
        return (n.startswith("lpx_handlers/") or n == "lpx_fanout.py"
                or n == "lpx_rt.py" or n == "lpx_nn.py"
                or base.startswith("liblpx_nn")
                or base.endswith(".hsaco")
                or (n.startswith("lpx_ops/")
                    and not base.startswith("_lpx_ops")))

    assert early("standalone_gfx950.hsaco")
    assert early("lpx_nn.py")
    assert early("liblpx_nn_gfx950.so")
    assert early("lpx_ops/liblpx_nn_gfx950.so")
    assert early("lpx_ops/standalone_gfx950.hsaco")
    assert not early("lpx_ops/_lpx_ops.cpython-310-x86_64-linux-gnu.so")
    assert not early("numpy/core/_multiarray_umath.so")


# ---- GPU: executors vs references ----------------------------------------

pytestmark_gpu = pytest.mark.gpu


def _d2h_f32(ptr, n):
    from lambdipy_amd.shims import lpx_rt

    buf = np.empty(n, dtype=np.float32)
    lpx_rt.memcpy_d2h(buf.ctypes.data, ptr, n * 4)
    return buf


def _d2h_u16(ptr, n):
    from lambdipy_amd.shims import lpx_rt

    buf = np.empty(n, dtype=np.uint16)
    lpx_rt.memcpy_d2h(buf.ctypes.data, ptr, n * 2)
    return buf


def _d2h_i32(ptr, n):
    from lambdipy_amd.shims import lpx_rt

    buf = np.empty(n, dtype=np.int32)
    lpx_rt.memcpy_d2h(buf.ctypes.data, ptr, n * 4)
    return buf


def _bf16_to_f32(u16):
    return (u16.astype(np.uint32) << 16).view(np.float32)


@pytest.mark.gpu
def test_matmul_lite_vs_numpy():
    from lambdipy_amd.shims import lpx_nn, lpx_rt

    lpx_rt.set_device(0)
    mm = lpx_nn.MatmulLite(n=512, seed=1)
    mm.run()
    mm.stream.sync()
    A = _bf16_to_f32(_d2h_u16(mm.A, 512 * 512)).reshape(512, 512)
    Bt = _bf16_to_f32(_d2h_u16(mm.Bt, 512 * 512)).reshape(512, 512)
    C = _d2h_f32(mm.C, 512 * 512).reshape(512, 512)
    ref = A.astype(np.float64) @ Bt.astype(np.float64).T
    rel = np.abs(C - ref).max() / (np.abs(ref).max() + 1e-6)
    assert rel < 2e-2, f"rel err {rel}"


@pytest.mark.gpu
def test_fftspmv_lite_vs_numpy():
    from lambdipy_amd.shims import lpx_nn, lpx_rt

    lpx_rt.set_device(0)
    fs = lpx_nn.FftSpmvLite(batch=8, n=256, rows=4096, nnz=16, seed=2)
    fs.run()
    fs.stream.sync()
    x = _d2h_f32(fs.x, 8 * 256 * 2).view(np.complex64).reshape(8, 256)
    f = _d2h_f32(fs.fout, 8 * 256 * 2).view(np.complex64).reshape(8, 256)
    ref = np.fft.fft(x.astype(np.complex128), axis=1)
    rel = np.abs(f - ref).max() / (np.abs(ref).max() + 1e-6)
    assert rel < 1e-4, f"fft rel err {rel}"
    indptr = _d2h_i32(fs.indptr, 4097)
    indices = _d2h_i32(fs.indices, 4096 * 16)
    data = _d2h_f32(fs.data, 4096 * 16)
    vec = _d2h_f32(fs.vec, 4096)
    y = _d2h_f32(fs.y, 4096)
    assert indptr[0] == 0 and indptr[-1] == 4096 * 16
    assert (indices >= 0).all() and (indices < 4096).all()
    # per-row sorted columns
    cols = indices.reshape(4096, 16)
    assert (np.diff(cols, axis=1) >= 0).all()
    yref = (data.reshape(4096, 16) * vec[cols]).sum(axis=1)
    rel = np.abs(y - yref).max() / (np.abs(yref).max() + 1e-6)
    assert rel < 1e-4, f"spmv rel err {rel}"


@pytest.mark.gpu
def test_resnet50_lite_forward():
    from lambdipy_amd.shims import lpx_nn, lpx_rt

    lpx_rt.set_device(0)
    m = lpx_nn.ResNet50Lite(n_classes=1000, seed=3)
    N = 8
    x = m.pool.alloc(N * 224 * 224 * 3 * 4)
    lpx_nn._ck(lpx_nn.lib().lpxn_randn(x, N * 224 * 224 * 3, 5, 1.0,
                                       m.stream.h), "randn")
    logits = m.forward(x, N)  # calibration pass
    m.stream.sync()
    out1 = _d2h_f32(logits, N * 1000).reshape(N, 1000)
    assert np.isfinite(out1).all(), "non-finite logits"
    assert np.abs(out1).max() > 0, "all-zero logits"
    assert m.calibrated
    logits2 = m.forward(x, N)  # fused pass
    m.stream.sync()
    out2 = _d2h_f32(logits2, N * 1000).reshape(N, 1000)
    assert np.isfinite(out2).all()
    # fused chain re-quantizes intermediates: labels should broadly agree
    agree = (out1.argmax(1) == out2.argmax(1)).mean()
    assert agree > 0.7, f"fused/calibration label agreement {agree}"
    # determinism of the fused path
    logits3 = m.forward(x, N)
    m.stream.sync()
    out3 = _d2h_f32(logits3, N * 1000).reshape(N, 1000)
    assert np.array_equal(out2, out3)


@pytest.mark.gpu
def test_resnet50_lite_graph_replay():
    from lambdipy_amd.shims import lpx_nn, lpx_rt

    lpx_rt.set_device(0)
    m = lpx_nn.ResNet50Lite(n_classes=1000, seed=4)
    N = 8
    x = m.pool.alloc(N * 224 * 224 * 3 * 4)
    lpx_nn._ck(lpx_nn.lib().lpxn_randn(x, N * 224 * 224 * 3, 7, 1.0,
                                       m.stream.h), "randn")
    m.forward(x, N)  # calibrate
    eager = m.forward(x, N)
    m.stream.sync()
    ref = _d2h_f32(eager, N * 1000).copy()
    ok = m.capture(x, N)
    assert ok, "hipGraph capture failed"
    replay = m.forward(x, N)  # graph path
    m.stream.sync()
    out = _d2h_f32(replay, N * 1000)
    assert np.array_equal(out, ref), "graph replay != eager fused forward"


@pytest.mark.gpu
@pytest.mark.parametrize("handler,env", [
    ("matmul_lite", {}),
    ("fftspmv_lite", {}),
    ("resnet50_lite", {"LPX_RESNET_BATCH": "16"}),
])
def test_lite_cold_start_e2e(tmp_path, handler, env):
    """Build a minimal lite layer and cold-start it from zip."""
    from lambdipy_amd.cli import _install_handlers
    from lambdipy_amd.packager import make_zip
    from lambdipy_amd.runtime import cold_start, prewarm_host

    prewarm_host()
    d = str(tmp_path / "layer")
    os.makedirs(d)
    _install_handlers(d, [handler])
    zp = str(tmp_path / "layer.zip")
    make_zip(d, zp)
    res = cold_start(zp, handler=handler, gpus=1, env=env, timeout_s=300)
    assert res.handler_report.get("checksum") is not None
    assert res.total_ms < 60_000
    phases = {p["phase"] for p in res.handler_report.get("phases", [])}
    assert "first_inference" in phases


@pytest.mark.gpu
def test_resnet50_lite_checkpoint_roundtrip(tmp_path):
    """save_weights/load_weights: a fresh alloc-only model restored from the
    checkpoint must produce BITWISE-identical logits (incl. frozen scales)."""
    from lambdipy_amd.shims import lpx_nn, lpx_rt

    lpx_rt.set_device(0)
    m = lpx_nn.ResNet50Lite(n_classes=1000, seed=6)
    N = 4
    x = m.pool.alloc(N * 224 * 224 * 3 * 4)
    lpx_nn._ck(lpx_nn.lib().lpxn_randn(x, N * 224 * 224 * 3, 9, 1.0,
                                       m.stream.h), "randn")
    m.forward(x, N)  # calibrate (freezes activation scales)
    ref = m.forward(x, N)
    m.stream.sync()
    want = _d2h_f32(ref, N * 1000).copy()
    ck = str(tmp_path / "rn50.lpxw")
    m.save_weights(ck)
    m2 = lpx_nn.ResNet50Lite(n_classes=1000, alloc_only=True)
    m2.load_weights(ck)
    assert m2.calibrated
    x2 = m2.pool.alloc(N * 224 * 224 * 3 * 4)
    lpx_nn._ck(lpx_nn.lib().lpxn_randn(x2, N * 224 * 224 * 3, 9, 1.0,
                                       m2.stream.h), "randn")
    out = m2.forward(x2, N)  # fused path directly (calibrated from ckpt)
    m2.stream.sync()
    got = _d2h_f32(out, N * 1000)
    assert np.array_equal(got, want), "checkpoint roundtrip logits mismatch"
