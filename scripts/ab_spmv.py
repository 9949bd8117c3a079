"""A/B: seg8 vs x-tile-in-LDS SpMV at the profile shape (1M x 32, sorted
columns) + a PMC-friendly single-variant mode (V=0|2 env)."""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from lambdipy_amd import ops  # noqa: E402


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def build(rows, nnz, dev="cuda"):
    torch.manual_seed(0)
    idx = torch.randint(0, rows, (rows, nnz), device=dev, dtype=torch.int32)
    idx, _ = torch.sort(idx, dim=1)
    indices = idx.flatten().contiguous()
    indptr = torch.arange(0, (rows + 1) * nnz, nnz, device=dev,
                          dtype=torch.int32)
    data = torch.randn(rows * nnz, dtype=torch.float32, device=dev)
    x = torch.randn(rows, dtype=torch.float32, device=dev)
    return indptr, indices, data, x


def main():
    only = os.environ.get("V")
    for rows, nnz in ((1 << 20, 32), (1 << 18, 32), (1 << 20, 64),
                      (1 << 22, 16)):
        indptr, indices, data, x = build(rows, nnz)
        ref = ops.spmv_csr(indptr, indices, data, x, -1)
        gb = rows * nnz * 12 / 1e9  # idx+data+gather-payload accounting
        for v in ((int(only),) if only else (-1, 2)):
            y = ops.spmv_csr(indptr, indices, data, x, v)
            ok = torch.allclose(y, ref, rtol=1e-4, atol=1e-3)
            dt = timeit(lambda: ops.spmv_csr(indptr, indices, data, x, v))
            print(f"rows=2^{rows.bit_length()-1} nnz={nnz} v={v:2d}: "
                  f"{dt*1e6:8.1f} us  {gb/dt/1e3:.2f} TB/s eff  ok={ok}")


if __name__ == "__main__":
    main()
