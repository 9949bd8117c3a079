"""Config-2 handler: 4096x4096 bf16 matmul on the gfx950 MFMA GEMM kernel.

Cold-start -> first-inference is the metric (BASELINE.json:2); phases are
timed per SURVEY.md §3.4 (import / hip-init / model-build / first-launch).
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

    # Wait for the layer BEFORE importing torch: torch itself imports numpy,
    # and a half-extracted layer numpy on PYTHONPATH would poison the import
    # (the torch-free rf_lite handler keeps the full unzip/import overlap).
    ready = os.environ.get("LPX_READY_FILE")
    if ready:
        t = time.perf_counter()
        while not os.path.exists(ready):
            time.sleep(0.002)
        mark("wait_layer_extract", t)

    t = time.perf_counter()
    import torch
    mark("import_torch", t)

    t = time.perf_counter()
    try:
        import lpx_numpy
    except ImportError:
        from lambdipy_amd.shims import lpx_numpy
    mark("import_shim", t)

    n = int(os.environ.get("LPX_MATMUL_N", "4096"))
    t = time.perf_counter()
    torch.manual_seed(0)
    dev = torch.device("cuda", 0)
    a = torch.randn(n, n, device=dev, dtype=torch.bfloat16)
    b = torch.randn(n, n, device=dev, dtype=torch.bfloat16)
    torch.cuda.synchronize()
    mark("hip_init_and_data", t)

    t = time.perf_counter()
    c = lpx_numpy.matmul_t(a, b)
    torch.cuda.synchronize()
    mark("first_inference", t)

    checksum = float(c.float().sum())
    out = {
        "handler": "matmul",
        "n": n,
        "checksum": checksum,
        "total_ms": (time.perf_counter() - t0) * 1e3,
        "phases": phases,
    }
    print("LPX_RESULT " + json.dumps(out), flush=True)
    return out


if __name__ == "__main__":
    main()
