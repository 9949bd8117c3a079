"""Run the MX-scaled MFMA layout probe and check against a host reference."""
import ctypes
import os
import sys

import numpy as np
import torch  # noqa: F401  (ensures HIP runtime libs are resolvable)

HERE = os.path.dirname(os.path.abspath(__file__))


def e4m3_to_f32(b):
    return torch.tensor(b, dtype=torch.uint8).view(torch.float8_e4m3fn).float().numpy()


def main():
    lib = ctypes.CDLL(os.path.join(HERE, "mx_probe.so"))
    rng = np.random.default_rng(0)
    # random e4m3 bit patterns, avoiding nan (0x7f/0xff)
    A = rng.integers(0, 256, (32, 64), dtype=np.uint8)
    B = rng.integers(0, 256, (32, 64), dtype=np.uint8)
    A[(A & 0x7F) == 0x7F] = 0x30
    B[(B & 0x7F) == 0x7F] = 0x30
    SA = (127 + rng.integers(-3, 4, (32, 2))).astype(np.uint8)
    SB = (127 + rng.integers(-3, 4, (32, 2))).astype(np.uint8)
    Af = e4m3_to_f32(A).reshape(32, 64)
    Bf = e4m3_to_f32(B).reshape(32, 64)
    sA = 2.0 ** (SA.astype(np.float64) - 127)
    sB = 2.0 ** (SB.astype(np.float64) - 127)
    ref = np.zeros((32, 32))
    for kb in range(2):
        ref += (Af[:, kb*32:(kb+1)*32] * sA[:, kb:kb+1]) @ \
               (Bf[:, kb*32:(kb+1)*32] * sB[:, kb:kb+1]).T
    for mode in range(4):
        D = np.zeros((32, 32), dtype=np.float32)
        rc = lib.run_mx_probe(
            A.ctypes.data_as(ctypes.c_void_p), B.ctypes.data_as(ctypes.c_void_p),
            SA.ctypes.data_as(ctypes.c_void_p), SB.ctypes.data_as(ctypes.c_void_p),
            D.ctypes.data_as(ctypes.c_void_p), ctypes.c_int(mode))
        assert rc == 0, f"hip error {rc}"
        err = np.abs(D - ref).max()
        scale = np.abs(ref).max() + 1e-9
        print(f"mode {mode}: max err {err:.4f} rel {err/scale:.2e} "
              f"{'OK' if err/scale < 1e-3 else 'MISMATCH'}")


if __name__ == "__main__":
    main()
