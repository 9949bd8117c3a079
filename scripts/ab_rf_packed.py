"""Packed vs classic RF traversal A/B + ILP sweep on the packed kernel."""
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


def unpack_classic(fp: ForestSoA) -> ForestSoA:
    """Rebuild the classic [n,4] layout from a packed forest (for A/B)."""
    import numpy as np

    nodes = fp.nodes.cpu().numpy()
    meta = np.ascontiguousarray(nodes[:, 0]).view(np.uint32)
    n = nodes.shape[0]
    feat = (meta >> 21).astype(np.int32)
    is_leaf = feat == 0x7FF
    cl = np.empty((n, 4), dtype=np.int32)
    cl[:, 0] = np.where(is_leaf, -1, feat)
    cl[:, 1] = nodes[:, 1]
    idx = np.arange(n, dtype=np.int32)
    cl[:, 2] = np.where(is_leaf, nodes[:, 1], idx + 1)
    cl[:, 3] = np.where(is_leaf, 0, idx + (meta & 0x1FFFFF).astype(np.int32))
    return ForestSoA(torch.from_numpy(cl).to(fp.nodes.device),
                     fp.leaf_values, fp.roots, fp.n_trees, fp.n_classes,
                     fp.classes, packed=False)


def main():
    X = torch.randn(1 << 20, 32, device="cuda")
    for trees, depth in [(256, 10), (512, 12)]:
        fp = ForestSoA._random_classic(trees, depth, 32, 8, seed=17,
                                       device=None).pack().to("cuda")
        assert fp.packed
        fc = unpack_classic(fp)
        pc = fc.predict_proba_t(X)
        pp = fp.predict_proba_t(X)
        assert torch.allclose(pc, pp, atol=1e-6), "packed != classic!"
        dtc = timeit(lambda: fc.predict_proba_t(X))
        dtp = timeit(lambda: fp.predict_proba_t(X))
        print(f"PERF rf {trees}x{depth}: classic {dtc*1e3:.3f} ms "
              f"({(1<<20)/dtc/1e6:.0f} Ms/s)  packed {dtp*1e3:.3f} ms "
              f"({(1<<20)/dtp/1e6:.0f} Ms/s)  speedup {dtc/dtp:.2f}x")
        for ilp in (1, 2, 4):
            dt = timeit(lambda: e.rf_predict8(fp.nodes, fp.leaf_values,
                                              fp.roots, X, ilp))
            print(f"  packed ilp={ilp}: {dt*1e3:.3f} ms {(1<<20)/dt/1e6:.0f} Ms/s")


def heap_ab():
    """Heap-layout kernel vs packed global kernel (same logical forest not
    required — same shape/size workload, plus CPU-reference correctness)."""
    import numpy as np
    from tests.test_forest import soa_predict_proba_cpu

    X = torch.randn(1 << 20, 32, device="cuda")
    for trees, depth in [(256, 10), (512, 12), (128, 8)]:
        fh = ForestSoA.random(trees, depth, 32, 8, seed=17, device=None)
        assert fh.layout == "heap"
        Xs = np.random.default_rng(4).standard_normal((64, 32)).astype(np.float32)
        ref = soa_predict_proba_cpu(fh, Xs)
        fg = fh.to("cuda")
        got = fg.predict_proba_t(torch.from_numpy(Xs).to("cuda")).cpu().numpy()
        np.testing.assert_allclose(got, ref, rtol=1e-4, atol=1e-5)
        for silp in (1, 2, 4):
            dt = timeit(lambda: ops.ext().rf_predict_heap(
                fg.nodes, fg.leaf_values, depth, X, silp))
            print(f"PERF rf-heap {trees}x{depth} silp={silp}: {dt*1e3:.3f} ms "
                  f"{(1<<20)/dt/1e6:.0f} Ms/s")


if __name__ == "__main__":
    main()
    heap_ab()
