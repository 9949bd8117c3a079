"""fp8 8-phase GEMM correctness + within-probe A/B vs v2."""
import sys, time, collections
import torch
sys.path.insert(0, "/root/repo")
from lambdipy_amd import ops
from lambdipy_amd.shims.lpx_sklearn import ForestSoA

e = ops.ext()
g = torch.Generator().manual_seed(0)
for (m, n, k) in [(1024, 1024, 128), (1024, 1280, 384), (2048, 2048, 1024)]:
    A = torch.randn(m, k, generator=g).to("cuda")
    Bt = torch.randn(n, k, generator=g).to("cuda")
    Aq = e.quantize_fp8(A, 0.25)
    Bq = e.quantize_fp8(Bt, 0.25)
    C = e.gemm_fp8(Aq, Bq, 0.0625)  # routed to v3 (M,N>=1024, K%128==0)
    ref = (Aq.view(torch.float8_e4m3fn).float() * 0.25) @ \
          (Bq.view(torch.float8_e4m3fn).float() * 0.25).t()
    rel = (C - ref).abs().max().item() / (ref.abs().max().item() + 1e-6)
    print(f"fp8v3 {m}x{n}x{k}: rel={rel:.5f} {'OK' if rel < 1e-3 else 'FAIL'}")
# bias+relu through v3
bias = torch.randn(1024, generator=g).to("cuda")
A = torch.randn(1024, 128, generator=g).to("cuda")
Bt = torch.randn(1024, 128, generator=g).to("cuda")
Aq, Bq = e.quantize_fp8(A, 1.0), e.quantize_fp8(Bt, 1.0)
C = e.gemm_fp8(Aq, Bq, 1.0, bias, True)
ref = torch.relu(Aq.view(torch.float8_e4m3fn).float() @ Bq.view(torch.float8_e4m3fn).float().t() + bias)
print("fp8v3 bias+relu:", "OK" if (C - ref).abs().max().item() < 1e-2 else "FAIL")

times = collections.defaultdict(list)
for sz in (4096, 8192):
    A = torch.randint(0, 255, (sz, sz), device="cuda", dtype=torch.uint8)
    B = torch.randint(0, 255, (sz, sz), device="cuda", dtype=torch.uint8)
    for _ in range(3):
        e.gemm_fp8(A, B, 1.0)
    torch.cuda.synchronize()
    for rnd in range(6):
        torch.cuda.synchronize(); t0 = time.perf_counter()
        for _ in range(5):
            e.gemm_fp8(A, B, 1.0)
        torch.cuda.synchronize()
        times[sz].append((time.perf_counter() - t0) / 5)
    med = sorted(times[sz])[len(times[sz]) // 2]
    print(f"PERF fp8v3 {sz}^3: median {med*1e3:.3f} ms {2*sz**3/med/1e12:.0f} TF")

# model_build timing with vectorized packed gen
t0 = time.perf_counter()
f = ForestSoA.random(256, 10, 32, 8, seed=17, device="cuda")
torch.cuda.synchronize()
print(f"forest build+upload: {(time.perf_counter()-t0)*1e3:.0f} ms")
X = torch.randn(1 << 20, 32, device="cuda")
for _ in range(3):
    f.predict_proba_t(X)
torch.cuda.synchronize(); t0 = time.perf_counter()
for _ in range(10):
    f.predict_proba_t(X)
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / 10
print(f"rf: {dt*1e3:.3f} ms {(1<<20)/dt/1e6:.0f} Ms/s")
