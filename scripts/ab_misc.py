"""ResNet-50 fp8 throughput + SpMV vec4 A/B + FFT baseline."""
import sys, time
import torch
sys.path.insert(0, "/root/repo")
from lambdipy_amd import ops
from lambdipy_amd.shims.lpx_torch import ResNet50FP8

e = ops.ext()

def timeit(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters

# ResNet-50 fp8 inference
model = ResNet50FP8(n_classes=1000, device="cuda", seed=3)
for bs in (32, 64, 128):
    x = torch.randn(bs, 3, 224, 224, device="cuda")
    dt = timeit(lambda: model(x), iters=5, warmup=2)
    print(f"PERF resnet50-fp8 bs={bs}: {dt*1e3:.2f} ms  {bs/dt:.0f} img/s")

# SpMV (vec4 path: uniform 32 nnz rows)
rows, nnz = 1 << 20, 32
idx = torch.randint(0, rows, (rows, nnz), device="cuda", dtype=torch.int32)
idx, _ = torch.sort(idx, dim=1)
indices = idx.flatten().contiguous()
indptr = torch.arange(0, (rows + 1) * nnz, nnz, device="cuda", dtype=torch.int32)
data = torch.randn(rows * nnz, device="cuda")
v = torch.randn(rows, device="cuda")
dt = timeit(lambda: ops.spmv_csr(indptr, indices, data, v), iters=20)
gb = (rows * nnz * 8 + rows * 8) / 1e9
print(f"PERF spmv 1Mx32: {dt*1e3:.3f} ms  {gb/dt:.2f} GB/s effective")

# FFT baseline (radix-2)
for n in (1024, 4096):
    xc = (torch.randn(4096, n, device="cuda") +
          1j * torch.randn(4096, n, device="cuda")).to(torch.complex64)
    dt = timeit(lambda: ops.fft(xc), iters=10)
    tb = 2 * xc.numel() * 8 / dt / 1e12
    print(f"PERF fft batch=4096 n={n}: {dt*1e3:.3f} ms  {tb:.2f} TB/s eff")

# hipGraph-captured ResNet (single graph launch per forward)
try:
    from lambdipy_amd.shims.lpx_torch import ResNet50FP8 as _R
    for bs in (32, 64, 128):
        m2 = _R(device="cuda", seed=0)
        x2 = torch.randn(bs, 3, 224, 224, device="cuda")
        m2(x2)  # calibration
        m2.capture_graph(bs)
        for _ in range(3):
            m2(x2)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(10):
            m2(x2)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / 10
        print(f"PERF resnet50-fp8-graph bs={bs}: {dt*1e3:.2f} ms  {bs/dt:.0f} img/s")
        del m2
        torch.cuda.empty_cache()
except Exception as ex:
    print("graph capture failed:", ex)
