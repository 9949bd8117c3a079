"""A/B: v8 compact child-pair RF kernel (4-B nodes, 8-B records) vs the
v6 f32 pair kernel at the bench shape (256 trees x d10, 1M x 32, C8).
Also quantifies the fp16-threshold decision-flip rate on an f32 forest."""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
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


def pack(n, depth):
    stride = (1 << depth) - 1
    T = n.shape[0] // stride
    nh = n.view(T, stride, 2)
    ppt = (1 << (depth - 1)) - 1
    idx = torch.arange(ppt, device=n.device)
    pairs = torch.cat([nh[:, 2 * idx + 1, :], nh[:, 2 * idx + 2, :]],
                      dim=2).reshape(T * ppt, 4).contiguous()
    roots = nh[:, 0, :].contiguous()
    thr16 = (nh[..., 1].contiguous().view(torch.float32).to(torch.float16)
             .view(torch.int16).to(torch.int32) & 0xFFFF)
    cn = (nh[..., 0] << 16) | thr16
    cpairs = (torch.stack([cn[:, 2 * idx + 1], cn[:, 2 * idx + 2]], dim=2)
              .reshape(T * ppt, 2).contiguous())
    croots = cn[:, 0].contiguous()
    return pairs, roots, cpairs, croots


def main():
    depth = int(os.environ.get("D", "10"))
    f = ForestSoA.random(256, depth, 32, 8, seed=17, device="cuda")
    X = torch.randn(1 << 20, 32, device="cuda")
    S = X.shape[0]
    n, lv = f.nodes, f.leaf_values
    pairs, roots, cpairs, croots = pack(n, depth)

    ref = e.rf_predict_heap_pair(pairs, roots, lv, depth, X, 1)
    lv16 = lv.to(torch.float16).contiguous()
    lv16_exact = bool(lv16.to(torch.float32).eq(lv).all().item())
    for silp in (41, 81, 141):
        dt = bench(lambda: e.rf_predict_heap_pair(pairs, roots, lv, depth, X, silp))
        print(f"pair    silp={silp}: {S/dt/1e6:7.1f} Ms/s")
        dt = bench(lambda: e.rf_predict_heap_cpair(cpairs, croots, lv, depth, X, silp))
        out = e.rf_predict_heap_cpair(cpairs, croots, lv, depth, X, silp)
        lab_ref = ref.argmax(1)
        lab = out.argmax(1)
        flip = (lab != lab_ref).float().mean().item()
        print(f"cpair   silp={silp}: {S/dt/1e6:7.1f} Ms/s  label-flips={flip:.2e}")
        dt = bench(lambda: e.rf_predict_heap_cpair16(cpairs, croots, lv16, depth, X, silp))
        o16 = e.rf_predict_heap_cpair16(cpairs, croots, lv16, depth, X, silp)
        exact = bool(torch.equal(o16, out)) if lv16_exact else None
        print(f"cpair16 silp={silp}: {S/dt/1e6:7.1f} Ms/s  exact_vs_cpair={exact}")

    # exactness on an fp16-representable forest: round thresholds first
    nh = n.view(-1, 2).clone()
    thr_r = (nh[:, 1].view(torch.float32).to(torch.float16)
             .to(torch.float32).view(torch.int32))
    nh[:, 1] = thr_r
    n_r = nh.view_as(n).contiguous()
    pr, rr, cpr, crr = pack(n_r, depth)
    a = e.rf_predict_heap_pair(pr, rr, lv, depth, X, 1)
    b = e.rf_predict_heap_cpair(cpr, crr, lv, depth, X, 1)
    print("exact on fp16-representable forest:", bool(torch.equal(a, b)))


if __name__ == "__main__":
    main()
