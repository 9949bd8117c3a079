"""A/B: two-level heap RF kernel (top levels in LDS) vs v4, silp x top sweep."""
import sys
import time

import torch

sys.path.insert(0, "/root/repo")
from lambdipy_amd import ops  # noqa: E402
from lambdipy_amd.shims.lpx_sklearn import ForestSoA  # noqa: E402

e = ops.ext()


def bench(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    f = ForestSoA.random(256, 10, 32, 8, seed=17, device="cuda")
    assert f.layout == "heap", f.layout
    X = torch.randn(1 << 20, 32, device="cuda")
    S = X.shape[0]
    n, lv = f.nodes, f.leaf_values
    ref = e.rf_predict_heap(n, lv, 10, X, 1, -2)
    for silp in (1, 101, 102):
        for top in (-2,):
            try:
                out = e.rf_predict_heap(n, lv, 10, X, silp, top)
                ok = torch.allclose(out, ref, atol=1e-4)
                dt = bench(lambda: e.rf_predict_heap(n, lv, 10, X, silp, top))
                print(f"silp={silp} top={top:2d}: {S/dt/1e6:7.1f} Ms/s  ok={ok}")
            except RuntimeError as ex:
                print(f"silp={silp} top={top:2d}: ERR {ex}")

    # v6 child-pair layout: repack heap nodes -> (pairs, roots)
    depth = 10
    stride = (1 << depth) - 1
    T = n.shape[0] // stride
    nh = n.view(T, stride, 2)
    ppt = (1 << (depth - 1)) - 1
    idx = torch.arange(ppt, device=n.device)
    pairs = torch.cat([nh[:, 2 * idx + 1, :], nh[:, 2 * idx + 2, :]],
                      dim=2).reshape(T * ppt, 4).contiguous()
    roots = nh[:, 0, :].contiguous()
    for silp in (1, 2, 4):
        try:
            out = e.rf_predict_heap_pair(pairs, roots, lv, depth, X, silp)
            ok = torch.allclose(out, ref, atol=1e-4)
            dt = bench(lambda: e.rf_predict_heap_pair(pairs, roots, lv, depth,
                                                      X, silp))
            print(f"pair silp={silp}: {S/dt/1e6:7.1f} Ms/s  ok={ok}")
        except RuntimeError as ex:
            print(f"pair silp={silp}: ERR {ex}")

    # v7 quad records: 64-B records {children, grandchildren} per even-level
    # node; record slot recurrence slot' = 4*slot + 1 + (2*b1+b2)
    qpt = ((1 << depth) - 1) // 3
    quads = torch.zeros(T, qpt, 16, dtype=torch.int32, device=n.device)
    for k in range(depth // 2):
        base = (4**k - 1) // 3
        o = torch.arange(4**k, device=n.device)
        nn = (1 << (2 * k)) - 1 + o  # heap index of the even-level node
        quads[:, base + o, 0:2] = nh[:, 2 * nn + 1, :]
        quads[:, base + o, 2:4] = nh[:, 2 * nn + 2, :]
        if 2 * k + 2 < depth:
            for j in range(4):
                quads[:, base + o, 4 + 2 * j:6 + 2 * j] = nh[:, 4 * nn + 3 + j, :]
    quads = quads.reshape(T * qpt, 16).contiguous()
    for silp in (1, 2, 4):
        try:
            out = e.rf_predict_heap_quad(quads, roots, lv, depth, X, silp)
            ok = torch.allclose(out, ref, atol=1e-4)
            dt = bench(lambda: e.rf_predict_heap_quad(quads, roots, lv, depth,
                                                      X, silp))
            print(f"quad silp={silp}: {S/dt/1e6:7.1f} Ms/s  ok={ok}")
        except RuntimeError as ex:
            print(f"quad silp={silp}: ERR {ex}")


if __name__ == "__main__":
    main()
