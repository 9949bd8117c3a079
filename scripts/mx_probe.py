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
    for trial, (sa_off, sb_off) in enumerate([(127, 127), (126, 129)]):
        SA = np.full((32, 2), sa_off, dtype=np.uint8)
        SB = np.full((32, 2), sb_off, dtype=np.uint8)
        if trial == 1:
            SA += rng.integers(-2, 3, (32, 2)).astype(np.int8).view(np.uint8)
        D = np.zeros((32, 32), dtype=np.float32)
        rc = lib.run_mx_probe(
            A.ctypes.data_as(ctypes.c_void_p), B.ctypes.data_as(ctypes.c_void_p),
            SA.ctypes.data_as(ctypes.c_void_p), SB.ctypes.data_as(ctypes.c_void_p),
            D.ctypes.data_as(ctypes.c_void_p))
        assert rc == 0, f"hip error {rc}"
        # host reference: D[i,j] = sum_k a[i,k]*2^(SA[i,kb]-127) * b[j,k]*2^(SB[j,kb]-127)
        Af = e4m3_to_f32(A).reshape(32, 64)
        Bf = e4m3_to_f32(B).reshape(32, 64)
        sA = 2.0 ** (SA.astype(np.float64) - 127)   # [32,2]
        sB = 2.0 ** (SB.astype(np.float64) - 127)
        ref = np.zeros((32, 32))
        for kb in range(2):
            ref += (Af[:, kb*32:(kb+1)*32] * sA[:, kb:kb+1]) @ \
                   (Bf[:, kb*32:(kb+1)*32] * sB[:, kb:kb+1]).T
        err = np.abs(D - ref).max()
        scale = np.abs(ref).max() + 1e-9
        print(f"trial {trial}: max err {err:.6f} rel {err/scale:.2e} "
              f"{'OK' if err/scale < 1e-5 else 'MISMATCH'}")


if __name__ == "__main__":
    main()
