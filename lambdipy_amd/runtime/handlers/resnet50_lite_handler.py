"""Config-5 handler, torch-free fast path: fp8 ResNet-50 inference over the
ctypes HIP runtime (lpx_nn over liblpx_nn_gfx950.so).

No PyTorch import anywhere: the AOT library is dlopened at start (no JIT),
the 25M random-init weights are generated ON DEVICE and quantized to e4m3 by
the hw conversions, and the whole forward runs the hand-written scaled-MFMA
implicit-conv/GEMM kernels (BASELINE.json config 5: "PyTorch-ROCm ResNet-50
layer stripped <250 MB, fp8 MFMA" — the runtime IS the stripped layer:
liblpx_nn + shim + handler, a few MB total; VERDICT round-1 item 1).

Multi-GPU (LPX_RESNET_GPUS > 1): single-process RCCL fan-out like rf_lite —
broadcast the one weight slab, broadcast+slice the image batch, all_gather
the predicted labels over xGMI.
"""

from __future__ import annotations

import json
import os
import time


def main() -> dict:
    t0 = time.perf_counter()
    phases = []

    def mark(name, start):
        phases.append({"phase": name, "dur_ms": (time.perf_counter() - start) * 1e3})

    ready = os.environ.get("LPX_READY_FILE")
    if ready:  # overlapped cold start
        t = time.perf_counter()
        while not os.path.exists(ready):
            time.sleep(0.002)
        mark("wait_layer_extract", t)

    t = time.perf_counter()
    try:
        import lpx_rt
        import lpx_nn
    except ImportError:
        from lambdipy_amd.shims import lpx_rt, lpx_nn
    mark("import_shim", t)

    batch = int(os.environ.get("LPX_RESNET_BATCH", "64"))  # per GPU
    res = int(os.environ.get("LPX_RESNET_RES", "224"))
    gpus = int(os.environ.get("LPX_RESNET_GPUS",
                              os.environ.get("WORLD_SIZE", "1")))

    t = time.perf_counter()
    avail = lpx_rt.device_count()
    g = min(gpus, avail)
    while g & (g - 1):
        g &= g - 1  # degrade to a power of two (SURVEY.md §5 failure row)
    lpx_rt.set_device(0)
    lpx_nn.lib()  # dlopen + first hip call
    mark("hip_init_lib_load", t)

    t = time.perf_counter()
    # LPX_RESNET_WEIGHTS: serve a real checkpoint (saved by
    # ResNet50Lite.save_weights) instead of device random-init
    wpath = os.environ.get("LPX_RESNET_WEIGHTS")
    models = []
    for d in range(g):
        lpx_rt.set_device(d)
        models.append(lpx_nn.ResNet50Lite(
            n_classes=1000, seed=3, alloc_only=(d > 0 or wpath is not None)))
    lpx_rt.set_device(0)
    if wpath:
        models[0].load_weights(wpath)
    models[0].stream.sync()
    mark("model_build", t)

    comms = None
    t = time.perf_counter()
    # LPX_RF_FORCE_COMMS=1: init RCCL + run every collective even at g=1
    # (world-1 collectives are valid) so a 1-GPU box covers the full
    # multi-GPU code path.
    if g > 1 or os.environ.get("LPX_RF_FORCE_COMMS") == "1":
        comms = lpx_rt.Comms(g)
    mark("comm_init", t)

    t = time.perf_counter()
    if comms:
        # ONE broadcast: the whole weight slab (e4m3 weights + f32 biases)
        comms.broadcast([m.slab for m in models],
                        models[0].slab_bytes // 4, lpx_rt.NCCL_INT32, root=0)
    mark("model_broadcast", t)

    t = time.perf_counter()
    # synthetic batch NHWC f32, generated on GPU 0, broadcast + local slice
    shard_el = batch * res * res * 3
    total_el = shard_el * g
    X = []
    for d in range(g):
        lpx_rt.set_device(d)
        X.append(lpx_rt.malloc(total_el * 4))
    lpx_rt.set_device(0)
    L = lpx_nn.lib()
    lpx_nn._ck(L.lpxn_randn(X[0], total_el, 5, 1.0, models[0].stream.h),
               "randn(batch)")
    models[0].stream.sync()
    if comms:
        comms.broadcast(X, total_el, lpx_rt.NCCL_FLOAT32, root=0)
    mark("batch_data", t)

    t = time.perf_counter()
    # calibration runs once on GPU 0; frozen scales are shared (the weights
    # are identical on every device after the broadcast)
    lpx_rt.set_device(0)
    logits0 = models[0].forward(X[0], batch, res, res)
    scales = models[0].export_scales()
    labels, labels_all = [], []
    for d in range(g):
        lpx_rt.set_device(d)
        labels.append(lpx_rt.malloc(batch * 4))
        labels_all.append(lpx_rt.malloc(batch * g * 4))
        m = models[d]
        if d > 0:
            m.import_scales(scales)
            lg = m.forward(X[d] + d * shard_el * 4, batch, res, res)
        else:
            lg = logits0
        lpx_nn._ck(L.lpxn_argmax_rows(lg, labels[d], batch, 1000, m.stream.h),
                   "argmax")
    if comms:
        for d in range(g):
            lpx_rt.set_device(d)
            models[d].stream.sync()
        comms.all_gather(labels, labels_all, batch, lpx_rt.NCCL_INT32)
    # checksum: sum of predicted labels on the host (tiny: g*batch int32)
    import ctypes
    src = labels_all[0] if comms else labels[0]
    n_lab = batch * g if comms else batch
    for d in range(g):
        lpx_rt.set_device(d)
        models[d].stream.sync()
    lpx_rt.set_device(0)
    buf = (ctypes.c_int * n_lab)()
    lpx_rt.memcpy_d2h(ctypes.addressof(buf), src, n_lab * 4)
    ck = float(sum(buf))
    mark("first_inference", t)

    out = {
        "handler": "resnet50_lite",
        "world_size": g,
        "batch": batch * g,
        "res": res,
        "checksum": ck,
        "total_ms": (time.perf_counter() - t0) * 1e3,
        "phases": phases,
    }
    print("LPX_RESULT " + json.dumps(out), flush=True)
    if comms:
        comms.destroy()
    return out


if __name__ == "__main__":
    main()
