"""A/B sweep: fp8 GEMM v2, RF tree-ILP, RF grid size. Run on the GPU box."""
import sys
import time

import torch

sys.path.insert(0, "/root/repo")
from lambdipy_amd import ops  # noqa: E402
from lambdipy_amd.shims.lpx_sklearn import ForestSoA  # noqa: E402

e = ops.ext()


def timeit(fn, iters=20, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    # ---- fp8 GEMM v2 correctness + perf
    g = torch.Generator().manual_seed(0)
    for (m, n, k) in [(512, 512, 64), (600, 700, 100), (1024, 1024, 512)]:
        A = torch.randn(m, k, generator=g).to("cuda")
        Bt = torch.randn(n, k, generator=g).to("cuda")
        Aq = e.quantize_fp8(A, 0.25)
        Bq = e.quantize_fp8(Bt, 0.25)
        C = e.gemm_fp8(Aq, Bq, 0.0625)
        ref = (Aq.view(torch.float8_e4m3fn).float() * 0.25) @ \
              (Bq.view(torch.float8_e4m3fn).float() * 0.25).t()
        err = (C - ref).abs().max().item() / (ref.abs().max().item() + 1e-6)
        ok = "OK" if err < 1e-3 else "FAIL"
        print(f"fp8v2 {m}x{n}x{k}: rel={err:.5f} {ok}")

    for sz in (4096, 8192):
        A = torch.randint(0, 255, (sz, sz), device="cuda", dtype=torch.uint8)
        B = torch.randint(0, 255, (sz, sz), device="cuda", dtype=torch.uint8)
        dt = timeit(lambda: e.gemm_fp8(A, B, 1.0), iters=10)
        print(f"PERF fp8 {sz}^3: {dt*1e3:.3f} ms {2.0*sz**3/dt/1e12:.0f} TF")

    # ---- RF ILP sweep
    f = ForestSoA.random(256, 10, 32, 8, seed=17, device="cuda")
    X = torch.randn(1 << 20, 32, device="cuda")
    base = None
    for ilp in (1, 2, 4, 8):
        out = e.rf_predict(f.nodes, f.leaf_values, f.roots, X, ilp)
        if base is None:
            base = out
        assert torch.equal(out, base), f"ILP {ilp} mismatch"
        dt = timeit(lambda: e.rf_predict(f.nodes, f.leaf_values, f.roots, X, ilp))
        print(f"PERF rf ilp={ilp}: {dt*1e3:.3f} ms {(1<<20)/dt/1e6:.0f} Msamples/s")

    # deeper forest shape (512 trees, depth 12)
    f2 = ForestSoA.random(512, 12, 32, 8, seed=3, device="cuda")
    for ilp in (2, 4, 8):
        dt = timeit(lambda: e.rf_predict(f2.nodes, f2.leaf_values, f2.roots, X, ilp),
                    iters=10)
        print(f"PERF rf512d12 ilp={ilp}: {dt*1e3:.3f} ms {(1<<20)/dt/1e6:.0f} Msamples/s")


def rf_packed_ab():
    """Packed vs classic RF traversal A/B (appended after the v2 kernel)."""
    from lambdipy_amd.shims.lpx_sklearn import ForestSoA
    import numpy as np

    for trees, depth in [(256, 10), (512, 12)]:
        fp = ForestSoA.random(trees, depth, 32, 8, seed=17, device="cuda")
        assert fp.packed
        X = torch.randn(1 << 20, 32, device="cuda")
        dt = timeit(lambda: fp.predict_proba_t(X), iters=15)
        print(f"PERF rf-packed {trees}x{depth}: {dt*1e3:.3f} ms "
              f"{(1<<20)/dt/1e6:.0f} Msamples/s")


if __name__ == "__main__":
    main()
    rf_packed_ab()
