"""Race screen for the single-buffered bf16-split pairwise kernels.

v6/v8 reuse ONE LDS buffer per K-step (consume -> barrier -> commit ->
barrier); a misplaced commit would corrupt fragments mid-consume.  The
double-buffered v5/v7 overlap commit(cur^1) with consume(cur) and would
corrupt on a missing end barrier.  Screen: many repetitions across sizes
that stress barrier timing (short K = few steps; long K = many), each run
compared bitwise against itself and against the f64 reference tolerance.
"""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from lambdipy_amd import ops  # noqa: E402


def main():
    shapes = [(512, 512, 33), (1024, 768, 96), (2048, 2048, 257),
              (4096, 1280, 512)]
    for variant in ("5", "6", "7", "8", "9"):
        os.environ["LPX_PW_V2"] = variant
        for (m, n, k) in shapes:
            torch.manual_seed(m + k)
            X = torch.randn(m, k, device="cuda")
            Y = torch.randn(n, k, device="cuda")
            ref = ops.pairwise_sqeuclidean(X, Y)
            refd = torch.cdist(X.double(), Y.double()) ** 2
            bound = (3e-4 if variant in "579" else 1e-6) * (refd.max().item() + 1)
            for rep in range(8):
                out = ops.pairwise_sqeuclidean(X, Y)
                assert torch.equal(out, ref), \
                    f"v{variant} race-screen NONDETERMINISM {m}x{n}x{k} rep={rep}"
            err = (ref.double() - refd).abs().max().item()
            assert err < bound, f"v{variant} {m}x{n}x{k} err={err:.3e} > {bound:.3e}"
        print(f"v{variant} race-screen OK ({len(shapes)} shapes x8)")
    os.environ.pop("LPX_PW_V2", None)


if __name__ == "__main__":
    main()
