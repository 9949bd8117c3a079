"""Decode global_load_lds lane->LDS addressing for sizes 1/2/4."""
import ctypes
import os
import subprocess

import numpy as np
import torch  # noqa: F401  (hip runtime init)

HERE = os.path.dirname(os.path.abspath(__file__))
SO = os.path.join(HERE, "glds_probe.so")
if not os.path.exists(SO):
    subprocess.run(["/opt/rocm/bin/hipcc", "--offload-arch=gfx950", "-O2",
                    "-shared", "-fPIC", os.path.join(HERE, "glds_probe.hip"),
                    "-o", SO], check=True)
lib = ctypes.CDLL(SO)

hip = ctypes.CDLL("libamdhip64.so")
src = ctypes.c_void_p()
out = ctypes.c_void_p()
hip.hipMalloc(ctypes.byref(src), 4096)
hip.hipMalloc(ctypes.byref(out), 1024)
pat = np.arange(256, dtype=np.uint8).tobytes() * 16
hip.hipMemcpy(src, pat, 4096, 1)  # H2D

for size in (1, 2, 4):
    rc = lib.run_glds_probe(src, out, size)
    buf = np.zeros(1024, dtype=np.uint8)
    hip.hipMemcpy(buf.ctypes.data_as(ctypes.c_void_p), out, 1024, 2)  # D2H
    assert rc == 0, rc
    got = buf[:64 * max(size, 1)]
    # expectation A: lane i -> lds[i*size]  (tight packing)
    tight = np.concatenate([np.arange(i * size, i * size + size, dtype=np.uint8)
                            for i in range(64)])
    # expectation B: lane i -> lds[i*4] (dword slots, sub-dword in low bytes)
    ok_tight = np.array_equal(buf[:64 * size], tight)
    print(f"size={size}: tight={ok_tight} first32={buf[:32].tolist()}")
