"""A/B: fp4 MX GEMM 2-parity (default) vs tri-parity (LPX_FP4_3P=1).
New sync structure -> race screen (multi-run, multi-size refcheck vs the
2-parity kernel) before timing."""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from lambdipy_amd import ops  # noqa: E402

e = ops.ext()


def timeit(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def quant(x):
    return e.quantize_mx4(x.contiguous())


def main():
    # race screen at small sizes, 5 runs each
    for sz in (256, 512, 1024):
        A = torch.randn(sz, sz, device="cuda")
        qa, sa = quant(A)
        qb, sb = quant(torch.randn(sz, sz, device="cuda"))
        os.environ["LPX_FP4_3P"] = "0"
        ref = e.gemm_mx4(qa, sa, qb, sb)
        os.environ["LPX_FP4_3P"] = "1"
        for rep in range(5):
            out = e.gemm_mx4(qa, sa, qb, sb)
            assert torch.equal(out, ref), f"3p race-screen FAIL sz={sz} rep={rep}"
        os.environ.pop("LPX_FP4_3P", None)
    print("3p race-screen OK (256/512/1024 x5)")

    for n in (4096, 8192):
        A = torch.randn(n, n, device="cuda")
        B = torch.randn(n, n, device="cuda")
        qa, sa = quant(A)
        qb, sb = quant(B)
        flops = 2.0 * n ** 3
        os.environ["LPX_FP4_3P"] = "0"
        ref = e.gemm_mx4(qa, sa, qb, sb)
        dt = timeit(lambda: e.gemm_mx4(qa, sa, qb, sb))
        print(f"2p @ {n}: {flops/dt/1e12:6.0f} TF")
        os.environ["LPX_FP4_3P"] = "1"
        out = e.gemm_mx4(qa, sa, qb, sb)
        ok = torch.equal(out, ref)
        dt = timeit(lambda: e.gemm_mx4(qa, sa, qb, sb))
        print(f"3p @ {n}: {flops/dt/1e12:6.0f} TF  exact={ok}")
        os.environ.pop("LPX_FP4_3P", None)


if __name__ == "__main__":
    main()
