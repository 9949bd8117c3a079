"""Config-2 handler, torch-free fast path: 4096^2 bf16 MFMA matmul over the
ctypes HIP runtime (lpx_nn). Operands are generated on device; the only
host work is the ctypes launches — no torch/numpy import on the cold path
(VERDICT round-1 item 6: kill the ~1.3 s torch-import floor)."""

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
    if ready:
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

    n = int(os.environ.get("LPX_MATMUL_N", "4096"))

    t = time.perf_counter()
    lpx_rt.set_device(0)
    lpx_nn.lib()
    mark("hip_init_lib_load", t)

    t = time.perf_counter()
    mm = lpx_nn.MatmulLite(n=n, seed=0)
    mm.stream.sync()
    mark("model_and_data", t)

    t = time.perf_counter()
    mm.run()
    mm.stream.sync()
    mark("first_inference", t)

    ck = mm.checksum()
    out = {
        "handler": "matmul_lite",
        "n": n,
        "checksum": ck,
        "total_ms": (time.perf_counter() - t0) * 1e3,
        "phases": phases,
    }
    print("LPX_RESULT " + json.dumps(out), flush=True)
    return out


if __name__ == "__main__":
    main()
