#!/bin/bash
# Round-1 GPU validation: resnet fp8 test, bench.py (cold start + serving step),
# rocprofv3 stats for the RF + GEMM kernels. Run via gpurun from repo root.
set -x
cd /root/repo

echo "=== resnet50 fp8 test ===" > gpurun_out/validate1.log
timeout 400 python -m pytest tests/test_ops_gpu.py -q -k "resnet or fp8" -m gpu \
  >> gpurun_out/validate1.log 2>&1
echo "resnet rc=$?" >> gpurun_out/validate1.log

echo "=== bench.py 1 GPU (incl. layer build + cold start) ===" >> gpurun_out/validate1.log
timeout 900 python bench.py --gpus 1 --steps 10 --warmup 3 \
  >> gpurun_out/validate1.log 2>&1
echo "bench rc=$?" >> gpurun_out/validate1.log

echo "=== handler cold starts (matmul, fftspmv) ===" >> gpurun_out/validate1.log
timeout 300 python - >> gpurun_out/validate1.log 2>&1 <<'EOF'
import json
from lambdipy_amd.resolver import parse_requirements, resolve
from lambdipy_amd.packager import assemble_layer
from lambdipy_amd.builder import build_recipe_artifacts
from lambdipy_amd.cli import _install_handlers
from lambdipy_amd.runtime import cold_start

res = resolve(parse_requirements("numpy\n"))
assemble_layer(res, "/tmp/l2", wheel_builder=build_recipe_artifacts)
_install_handlers("/tmp/l2", ["matmul", "fftspmv"])
for h in ("matmul", "fftspmv"):
    r = cold_start("/tmp/l2", handler=h, gpus=1, timeout_s=240)
    print("COLDSTART", h, json.dumps(r.to_dict()))
EOF
echo "handlers rc=$?" >> gpurun_out/validate1.log

echo "=== rocprofv3 kernel stats (rf + gemm bf16 + fp8) ===" >> gpurun_out/validate1.log
cd /tmp && export TMPDIR=/tmp
timeout 400 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof1 -o rfgemm -- \
  python - <<'EOF' >> /root/repo/gpurun_out/validate1.log 2>&1
import torch, sys
sys.path.insert(0, "/root/repo")
from lambdipy_amd import ops
from lambdipy_amd.shims.lpx_sklearn import ForestSoA

f = ForestSoA.random(256, 10, 32, 8, seed=17, device="cuda")
X = torch.randn(1048576, 32, device="cuda")
for _ in range(5):
    p = f.predict_proba_t(X)
torch.cuda.synchronize()

a = torch.randn(4096, 4096, device="cuda", dtype=torch.bfloat16)
b = torch.randn(4096, 4096, device="cuda", dtype=torch.bfloat16)
for _ in range(5):
    c = ops.gemm_bf16(a, b)
torch.cuda.synchronize()

import time
for name, fn in [("rf", lambda: f.predict_proba_t(X)), ("gemm4096", lambda: ops.gemm_bf16(a, b))]:
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(10): fn()
    torch.cuda.synchronize(); dt = (time.perf_counter() - t0) / 10
    if name == "gemm4096":
        print(f"TIMING {name}: {dt*1e3:.3f} ms  {2*4096**3/dt/1e12:.1f} TF")
    else:
        print(f"TIMING {name}: {dt*1e3:.3f} ms  {1048576/dt/1e6:.2f} Msamples/s")
EOF
echo "rocprof rc=$?" >> /root/repo/gpurun_out/validate1.log
ls /root/repo/gpurun_out/prof1 >> /root/repo/gpurun_out/validate1.log 2>&1
tail -c 12000 /root/repo/gpurun_out/validate1.log
