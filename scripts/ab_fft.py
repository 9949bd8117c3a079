"""FFT throughput: LDS-swizzle A/B (effective TB/s = 2*bytes/time)."""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from lambdipy_amd import ops  # noqa: E402


def t(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    for batch, n in ((4096, 1024), (1024, 4096), (16384, 256), (8192, 2048)):
        g = torch.Generator().manual_seed(0)
        x = (torch.randn(batch, n, generator=g) +
             1j * torch.randn(batch, n, generator=g)).to(torch.complex64).to("cuda")
        f = ops.fft(x)
        ref = torch.fft.fft(x)
        rel = (f - ref).abs().max().item() / (ref.abs().max().item() + 1e-6)
        dt = t(lambda: ops.fft(x))
        tbps = 2 * x.numel() * 8 / dt / 1e12
        print(f"batch={batch:6d} n={n:5d}: {dt*1e3:7.3f} ms  {tbps:5.2f} TB/s eff  rel={rel:.2e}")


if __name__ == "__main__":
    main()
