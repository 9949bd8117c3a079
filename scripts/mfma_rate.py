"""Measure raw MFMA issue rates (PFLOP/s): scaled 32x32x64 vs 16x16x32 fp8."""
import ctypes
import os
import subprocess

HERE = os.path.dirname(os.path.abspath(__file__))
SO = os.path.join(HERE, "mfma_rate.so")

if not os.path.exists(SO):
    subprocess.run(["/opt/rocm/bin/hipcc", "--offload-arch=gfx950", "-O3",
                    "-shared", "-fPIC", os.path.join(HERE, "mfma_rate.hip"),
                    "-o", SO], check=True)

lib = ctypes.CDLL(SO)
BLOCKS, ITERS = 512, 20000
for which, name, flops_per in ((0, "scaled 32x32x64 fp8", 8 * 131072),
                               (1, "plain  16x16x32 fp8", 8 * 16384),
                               (2, "scaled 32x32x64 fp4", 8 * 131072),
                               (3, "scaled 16x16x128 fp8", 8 * 65536),
                               (4, "plain 16x16x32 bf16", 8 * 16384),
                               (5, "plain 32x32x16 bf16", 8 * 32768),
                               (6, "scaled 32x32x64 fp8xfp4", 8 * 131072)):
    ms = ctypes.c_float()
    rc = lib.run_rate(which, BLOCKS, ITERS, ctypes.byref(ms))
    assert rc == 0, rc
    waves = BLOCKS * 512 // 64
    fl = float(waves) * ITERS * flops_per
    print(f"{name}: {fl / (ms.value / 1e3) / 1e15:6.2f} PF  ({ms.value:.2f} ms)")
