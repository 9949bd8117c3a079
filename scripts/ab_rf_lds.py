"""Forest-in-LDS RF kernel A/B vs global-memory packed kernel."""
import sys
import time

import torch

sys.path.insert(0, "/root/repo")
from lambdipy_amd import ops  # noqa: E402
from lambdipy_amd.shims.lpx_sklearn import ForestSoA  # noqa: E402

e = ops.ext()


def timeit(fn, iters=15, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    X = torch.randn(1 << 20, 32, device="cuda")
    for trees, depth in [(256, 10), (512, 12), (128, 8)]:
        f = ForestSoA._random_classic(trees, depth, 32, 8, seed=17, device=None).pack().to("cuda")
        plan = f._plan_chunks(32)
        assert plan is not None, "no chunk plan"
        ctb, mx = plan
        p_glob = e.rf_predict8(f.nodes, f.leaf_values, f.roots, X, 2)
        p_lds = e.rf_predict8_lds(f.nodes, f.leaf_values, f.roots, ctb, mx, X)
        ok = torch.allclose(p_glob, p_lds, atol=1e-5)
        print(f"rf {trees}x{depth}: chunks={ctb.numel()-1} max_nodes={mx} "
              f"match={ok}")
        assert ok
        dtg = timeit(lambda: e.rf_predict8(f.nodes, f.leaf_values, f.roots, X, 2))
        dtl = timeit(lambda: e.rf_predict8_lds(f.nodes, f.leaf_values, f.roots,
                                               ctb, mx, X))
        print(f"  global {dtg*1e3:.3f} ms ({(1<<20)/dtg/1e6:.0f} Ms/s)  "
              f"lds {dtl*1e3:.3f} ms ({(1<<20)/dtl/1e6:.0f} Ms/s)  "
              f"speedup {dtg/dtl:.2f}x")


if __name__ == "__main__":
    main()
