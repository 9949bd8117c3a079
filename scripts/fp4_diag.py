"""Determine the fp4 scaled-MFMA per-lane scale coverage empirically.

Compares gemm_mx4 output against dequant references under competing
hypotheses about which E8M0 scale the hardware applies to each shuffled
8-byte quarter position [q0|q2|q1|q3]:
  H0 (fp8-like, current): byte-half of BOTH lanes -> positions get [s0,s1,s0,s1]
  H1 (lane-owns-own-bytes): lane kb's 16 B -> positions get [s0,s0,s1,s1]
"""
import sys

import torch

sys.path.insert(0, "/root/repo")
from lambdipy_amd import ops  # noqa: E402

e = ops.ext()

E2M1 = torch.tensor([0.0, 0.5, 1.0, 1.5, 2.0, 3.0, 4.0, 6.0,
                     -0.0, -0.5, -1.0, -1.5, -2.0, -3.0, -4.0, -6.0],
                    device="cuda")


def dequant(q, sc, K, smap):
    R = q.shape[0]
    g = K // 64
    lo = E2M1[(q & 0xF).long()]
    hi = E2M1[(q >> 4).long()]
    d = torch.stack([lo, hi], dim=-1).reshape(R, g, 4, 16)
    # tile-major sc [KT, R, 4] -> [R, g, 2]
    s = sc.permute(1, 0, 2).reshape(R, -1, 2, 2).reshape(R, -1, 2)[:, :g]
    scales = torch.pow(2.0, s.float() - 127.0)
    d = d * scales[:, :, smap].unsqueeze(-1)
    return d[:, :, [0, 2, 1, 3], :].reshape(R, K)


def main():
    gtor = torch.Generator(device="cpu").manual_seed(13)
    M, N, K = 256, 256, 128
    A = (torch.randn(M, K, generator=gtor) * 2).to("cuda")
    Bt = torch.randn(N, K, generator=gtor).to("cuda")
    qa, sa = e.quantize_mx4(A)
    qb, sb = e.quantize_mx4(Bt)
    C = e.gemm_mx4(qa, sa, qb, sb)
    for name, smap in (("H0 [0,1,0,1]", [0, 1, 0, 1]),
                       ("H1 [0,0,1,1]", [0, 0, 1, 1]),
                       ("H2 [1,0,1,0]", [1, 0, 1, 0]),
                       ("H3 [1,1,0,0]", [1, 1, 0, 0])):
        ref = dequant(qa, sa, K, smap) @ dequant(qb, sb, K, smap).t()
        err = (C - ref).abs().max().item() / (ref.abs().max().item() + 1e-6)
        print(f"{name}: rel max err {err:.5f}")


if __name__ == "__main__":
    main()
