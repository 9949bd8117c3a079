"""A/B the GEMM variants on the GPU box (correctness + TF)."""
import sys
import time

import torch

sys.path.insert(0, "/root/repo")
from lambdipy_amd import ops  # noqa: E402


def check(m, n, k, variant):
    g = torch.Generator().manual_seed(m + n + k)
    A = torch.randn(m, k, generator=g).to("cuda").bfloat16()
    Bt = torch.randn(n, k, generator=g).to("cuda").bfloat16()
    C = ops.gemm_bf16(A, Bt, variant=variant)
    ref = A.float() @ Bt.float().t()
    rel = (C - ref).abs().max().item() / (ref.abs().max().item() + 1e-6)
    ok = "OK" if rel < 2e-2 else "FAIL"
    print(f"v{variant} {m}x{n}x{k}: rel={rel:.4f} {ok}")


def perf(m, n, k, variant, iters=20):
    A = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
    Bt = torch.randn(n, k, device="cuda", dtype=torch.bfloat16)
    for _ in range(3):
        ops.gemm_bf16(A, Bt, variant=variant)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        ops.gemm_bf16(A, Bt, variant=variant)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    print(f"PERF v{variant} {m}x{n}x{k}: {dt*1e3:.3f} ms {2.0*m*n*k/dt/1e12:.0f} TF")


if __name__ == "__main__":
    import sys as _sys
    variants = [int(x) for x in (_sys.argv[1:] or ["2", "3"])]
    for v in variants:
        for shape in [(256, 256, 256), (512, 512, 64), (512, 768, 128),
                      (1024, 1024, 1024), (4096, 4096, 4096)]:
            if v == 3 and any(s % 256 for s in shape[:2]) or (v == 3 and shape[2] % 64):
                continue
            check(*shape, v)
    for v in variants:
        perf(4096, 4096, 4096, v)
        perf(8192, 8192, 8192, v, iters=8)
