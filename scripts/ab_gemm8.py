"""Within-probe A/B of 8-phase GEMM tuning variants (guide rule 24)."""
import sys, time
import torch
sys.path.insert(0, "/root/repo")
from lambdipy_amd import ops

e = ops.ext()
A = torch.randn(4096, 4096, device="cuda", dtype=torch.bfloat16)
B = torch.randn(4096, 4096, device="cuda", dtype=torch.bfloat16)
A8 = torch.randn(8192, 8192, device="cuda", dtype=torch.bfloat16)
B8 = torch.randn(8192, 8192, device="cuda", dtype=torch.bfloat16)

# correctness once per variant (+ race screen for new sync structures:
# tune 17 is a NEW template -> multi-run multi-size refcheck)
for sz in (256, 512, 1024):
    As = torch.randn(sz, sz, device="cuda", dtype=torch.bfloat16)
    Bs = torch.randn(sz, sz, device="cuda", dtype=torch.bfloat16)
    refs = As.float() @ Bs.float().t()
    for rep in range(5):
        Cs = e.gemm_bf16_tune8(As, Bs, 17)
        rel = (Cs - refs).abs().max().item() / refs.abs().max().item()
        assert rel < 2e-2, f"tune17 race-screen FAIL sz={sz} rep={rep} rel={rel}"
print("tune17 race-screen OK (256/512/1024 x5)")
ref = A.float() @ B.float().t()
for t in (25, 26):
    C = e.gemm_bf16_tune8(A, B, t)
    rel = (C - ref).abs().max().item() / ref.abs().max().item()
    print(f"tune {t}: rel {rel:.4f} {'OK' if rel < 2e-2 else 'FAIL'}")

# interleaved rounds
import collections
times = collections.defaultdict(list)
for _ in range(3):
    for t in (25, 26):
        e.gemm_bf16_tune8(A, B, t)
torch.cuda.synchronize()
for rnd in range(8):
    for t in (25, 26):
        torch.cuda.synchronize(); t0 = time.perf_counter()
        for _ in range(5):
            e.gemm_bf16_tune8(A, B, t)
        torch.cuda.synchronize()
        times[t].append((time.perf_counter() - t0) / 5)
for t in (25, 26):
    med = sorted(times[t])[len(times[t]) // 2]
    print(f"tune {t} @4096: median {med*1e3:.3f} ms {2*4096**3/med/1e12:.0f} TF "
          f"(min {2*4096**3/max(times[t])/1e12:.0f})")
# zero-filled vs random operands (guide §5.4 rule 25: quote random; the
# template's 1563-zero vs 1320-random spread is data-dependent clocking)
Z = torch.zeros(4096, 4096, device="cuda", dtype=torch.bfloat16)
for _ in range(3):
    e.gemm_bf16_tune8(Z, Z, 14)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(10):
    e.gemm_bf16_tune8(Z, Z, 14)
torch.cuda.synchronize()
dz = (time.perf_counter() - t0) / 10
print(f"tune 14 @4096 ZERO-filled: {2*4096**3/dz/1e12:.0f} TF")

times = collections.defaultdict(list)
for rnd in range(6):
    for t in (25, 26):
        torch.cuda.synchronize(); t0 = time.perf_counter()
        for _ in range(3):
            e.gemm_bf16_tune8(A8, B8, t)
        torch.cuda.synchronize()
        times[t].append((time.perf_counter() - t0) / 3)
for t in (25, 26):
    med = sorted(times[t])[len(times[t]) // 2]
    print(f"tune {t} @8192: median {med*1e3:.3f} ms {2*8192**3/med/1e12:.0f} TF")
