"""A/B: pairwise variants — v1 (16x16x4 64^2), v3/v4 (256^2 f32 MFMA),
v5/v6 (bf16-split emulation, 4/6 cross terms on the bf16 pipe)."""
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


def main():
    for m, n, k in ((4096, 4096, 128), (8192, 8192, 256), (4096, 4096, 32),
                    (16384, 16384, 64), (2048, 2048, 64), (2048, 2048, 256),
                    (4096, 4096, 64)):
        torch.manual_seed(0)
        X = torch.randn(m, k, device="cuda")
        Y = torch.randn(n, k, device="cuda")
        ref = None
        import os as _os
        _vl = _os.environ.get("PW_VARIANTS", "d,4,5,6")
        pairs = [((None if v == "d" else v), ("default" if v == "d" else "v" + v)) for v in _vl.split(",")]
        for env, name in pairs:
            if env is None:
                os.environ.pop("LPX_PW_V2", None)
            else:
                os.environ["LPX_PW_V2"] = env
            D = ops.pairwise_sqeuclidean(X, Y)
            if ref is None:
                ref = torch.cdist(X.double(), Y.double()) ** 2
            err = (D.double() - ref).abs().max().item()
            ok = err < 1e-2 * (ref.max().item() + 1)
            dt = timeit(lambda: ops.pairwise_sqeuclidean(X, Y))
            tf = 2.0 * m * n * k / dt / 1e12
            print(f"{name} m={m} n={n} k={k}: {dt*1e3:7.3f} ms {tf:6.1f} TF "
                  f"ok={ok} maxerr={err:.3e}")
        os.environ.pop("LPX_PW_V2", None)


if __name__ == "__main__":
    main()
