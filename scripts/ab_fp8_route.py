"""A/B the fp8 GEMM dispatch for ResNet's fallback shapes.

The r02-final trace shows gemm_fp8_nt (mfma_busy 0.133, wait 0.756)
serving the stage-2/3 1x1 convs (N=128/256 misses every fast gate) and
the FC layer.  Candidate routes per shape: nt (current), nt256 ('b'),
tall-skinny ('t', gate widened to N<=256).  Correctness: every route vs
the dequantized-operand f32 reference.
"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
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
    # (M, N, K): stage2 conv1, stage3 conv1, stage4 conv1 (ctrl: hits nt256
    # today), giant stage1 1x1 (ctrl: ts today), FC
    import json as _json
    env_shapes = os.environ.get("AB_SHAPES")
    if env_shapes:
        shapes = [tuple(t) for t in _json.loads(env_shapes)]
    else:
        shapes = [(50176, 128, 256), (12544, 256, 1024), (3136, 512, 1024),
                  (3136, 2048, 512), (200704, 64, 256), (64, 1000, 2048)]
    for (m, n, k) in shapes:
        torch.manual_seed(0)
        A = torch.randn(m, k, device="cuda")
        B = torch.randn(n, k, device="cuda")
        sa = float(A.abs().amax()) / 448.0 + 1e-8
        sb = float(B.abs().amax()) / 448.0 + 1e-8
        aq = e.quantize_fp8(A, sa)
        bq = e.quantize_fp8(B, sb)
        ref = (aq.view(torch.float8_e4m3fn).float() * sa) @ \
              (bq.view(torch.float8_e4m3fn).float() * sb).t()
        routes = [(None, "auto"), ("t", "ts"), ("b", "nt256"),
                  ("n", "nt"), ("s", "msplit")]
        if os.environ.get("AB_EARLY"):
            routes = [("t", "ts"), ("t+e", "ts-early")]
        for rc, name in routes:
            os.environ.pop("LPX_FP8_EARLY", None)
            if rc is None:
                os.environ.pop("LPX_FP8_ROUTE", None)
            else:
                if rc.endswith("+e"):
                    os.environ["LPX_FP8_EARLY"] = "1"
                    rc = rc[:-2]
                os.environ["LPX_FP8_ROUTE"] = rc
            if rc == "t" and not (n <= 256 and m >= 2048):
                continue
            if rc == "s" and not (n >= 512 and n % 256 == 0 and m % 256):
                continue
            C = e.gemm_fp8(aq, bq, sa * sb, None, False)
            err = (C - ref).abs().max().item() / (ref.abs().max().item() + 1e-6)
            dt = bench(lambda: e.gemm_fp8(aq, bq, sa * sb, None, False))
            tf = 2.0 * m * n * k / dt / 1e12
            print(f"{name:6s} m={m:6d} n={n:4d} k={k:4d}: {dt*1e3:7.3f} ms "
                  f"{tf:7.1f} TF relerr={err:.2e}")
        os.environ.pop("LPX_FP8_ROUTE", None)
        print()


if __name__ == "__main__":
    main()
