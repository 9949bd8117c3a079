"""A/B: MX block-scaled fp8 GEMM vs plain fp8 v2/v3 (TFLOP/s)."""
import sys
import time

import torch

sys.path.insert(0, "/root/repo")
from lambdipy_amd import ops  # noqa: E402

e = ops.ext()


def bench(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    for n in (4096, 8192):
        a = torch.randn(n, n, device="cuda")
        b = torch.randn(n, n, device="cuda")
        qa, sa = e.quantize_mx(a)
        qb, sb = e.quantize_mx(b)
        aq8 = e.quantize_fp8(a, 1.0)
        bq8 = e.quantize_fp8(b, 1.0)
        fl = 2 * n**3 / 1e12

        dt = bench(lambda: e.gemm_mx(qa, sa, qb, sb))
        print(f"n={n} gemm_mx      {fl/dt:8.0f} TF  {dt*1e3:.3f} ms")
        dt = bench(lambda: e.gemm_fp8(aq8, bq8, 1.0))
        print(f"n={n} gemm_fp8(v3) {fl/dt:8.0f} TF  {dt*1e3:.3f} ms")
        dt = bench(lambda: e.quantize_mx(a))
        print(f"n={n} quantize_mx  {2*n*n*4/dt/1e12:8.2f} TB/s rd+wr  {dt*1e3:.3f} ms")
        qa4, sa4 = e.quantize_mx4(a)
        qb4, sb4 = e.quantize_mx4(b)
        dt = bench(lambda: e.gemm_mx4(qa4, sa4, qb4, sb4))
        print(f"n={n} gemm_mx4     {fl/dt:8.0f} TF  {dt*1e3:.3f} ms")
        dt = bench(lambda: e.gemm_mx_w4(qa, sa, qb4, sb4))
        print(f"n={n} gemm_mx_w4   {fl/dt:8.0f} TF  {dt*1e3:.3f} ms")


if __name__ == "__main__":
    main()
