"""Profiling workload: runs each hot kernel a few times under rocprofv3.

Used with --kernel-trace --stats (timings) and with --pmc (MFMA/LDS counters)
in SEPARATE runs (the two must not be combined on this pool)."""
import sys
import time

import torch

sys.path.insert(0, "/root/repo")
from lambdipy_amd import ops  # noqa: E402
from lambdipy_amd.shims.lpx_sklearn import ForestSoA  # noqa: E402


def main():
    dev = "cuda"
    # bf16 GEMM v2 (config 2 hot op)
    a = torch.randn(4096, 4096, device=dev, dtype=torch.bfloat16)
    b = torch.randn(4096, 4096, device=dev, dtype=torch.bfloat16)
    for _ in range(10):
        ops.gemm_bf16(a, b)
    torch.cuda.synchronize()

    # RF traversal (config 4, the metric workload; heap-layout kernel)
    f = ForestSoA.random(256, 10, 32, 8, seed=17, device=dev)
    X = torch.randn(1 << 20, 32, device=dev)
    for _ in range(5):
        f.predict_proba_t(X)
    torch.cuda.synchronize()

    # implicit-GEMM fp8 conv (config 5 hot op)
    e0 = ops.ext()
    xq = e0.quantize_fp8(torch.randn(64, 28, 28, 256, device=dev), 0.05)
    wq = e0.quantize_fp8(torch.randn(256, 3 * 3 * 256, device=dev), 0.05)
    for _ in range(5):
        e0.conv_fp8(xq, wq, 28, 28, 256, 3, 3, 1, 1, 0.0025)
    torch.cuda.synchronize()

    # fp8 GEMM (config 5)
    e = ops.ext()
    A8 = e.quantize_fp8(torch.randn(4096, 4096, device=dev), 0.1)
    B8 = e.quantize_fp8(torch.randn(4096, 4096, device=dev), 0.1)
    for _ in range(10):
        e.gemm_fp8(A8, B8, 0.01)
    torch.cuda.synchronize()

    # GEMV (N5)
    W = torch.randn(8192, 8192, device=dev, dtype=torch.bfloat16)
    x = torch.randn(8192, device=dev, dtype=torch.bfloat16)
    for _ in range(10):
        ops.gemv(W, x)
    torch.cuda.synchronize()

    # FFT (N6)
    xc = (torch.randn(4096, 1024, device=dev) +
          1j * torch.randn(4096, 1024, device=dev)).to(torch.complex64)
    for _ in range(5):
        ops.fft(xc)
    torch.cuda.synchronize()

    # SpMV (N7)
    rows, nnz = 1 << 20, 32
    idx = torch.randint(0, rows, (rows, nnz), device=dev, dtype=torch.int32)
    idx, _ = torch.sort(idx, dim=1)
    indptr = torch.arange(0, (rows + 1) * nnz, nnz, device=dev, dtype=torch.int32)
    data = torch.randn(rows * nnz, device=dev)
    v = torch.randn(rows, device=dev)
    for _ in range(5):
        ops.spmv_csr(indptr, idx.flatten().contiguous(), data, v)
    torch.cuda.synchronize()

    # pairwise (N8)
    P = torch.randn(4096, 128, device=dev)
    Q = torch.randn(4096, 128, device=dev)
    for _ in range(5):
        ops.pairwise_sqeuclidean(P, Q)
    torch.cuda.synchronize()

    # MX fp8 + fp4 GEMMs (scaled MFMA) and the hw quantizers
    Af = torch.randn(4096, 4096, device=dev)
    qa, sa = e.quantize_mx(Af)
    for _ in range(10):
        e.gemm_mx(qa, sa, qa, sa)
    qa4, sa4 = e.quantize_mx4(Af)
    for _ in range(10):
        e.gemm_mx4(qa4, sa4, qa4, sa4)
    torch.cuda.synchronize()

    # ResNet-50 fp8 end-to-end (fused quantize chains + NHWC maxpool)
    from lambdipy_amd.shims.lpx_torch import ResNet50FP8
    m = ResNet50FP8(device=dev, seed=0)
    xr = torch.randn(64, 3, 224, 224, device=dev)
    for _ in range(3):
        m(xr)
    torch.cuda.synchronize()
    print("profile workload done")


if __name__ == "__main__":
    main()
