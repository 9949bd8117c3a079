"""gfx950 fat-binary stripping (packager/fatbin.py).

Builds a tiny 4-arch HIP DSO with hipcc (cross-compile, no GPU needed),
strips it to gfx950-only, and checks the result still dlopens (which
exercises the HIP fat-binary registration path) and kept the gfx950 entry.
"""

import ctypes
import os
import shutil
import struct
import subprocess

import pytest

HIPCC = "/opt/rocm/bin/hipcc"
BUNDLER = "/opt/rocm/lib/llvm/bin/clang-offload-bundler"

pytestmark = pytest.mark.skipif(
    not (os.path.exists(HIPCC) and os.path.exists(BUNDLER)),
    reason="ROCm toolchain not available")

SRC = r"""
#include <hip/hip_runtime.h>
extern "C" __global__ void lpx_fat_test_kernel(float* x) {
  x[threadIdx.x] *= 2.0f;
}
extern "C" int lpx_fat_test_launch(float* x) {
  lpx_fat_test_kernel<<<1, 64>>>(x);
  return (int)hipGetLastError();
}
"""


@pytest.fixture(scope="module")
def fat_so(tmp_path_factory):
    d = tmp_path_factory.mktemp("fatbin")
    src = d / "fat_test.hip"
    src.write_text(SRC)
    so = d / "libfat_test.so"
    subprocess.run(
        [HIPCC, "-fPIC", "-shared", str(src), "-o", str(so), "-O1",
         "--offload-arch=gfx950", "--offload-arch=gfx942",
         "--offload-arch=gfx90a", "--offload-arch=gfx908"],
        check=True, capture_output=True)
    return so


def test_strip_keeps_gfx950_and_dlopens(fat_so, tmp_path):
    from lambdipy_amd.packager.fatbin import strip_fatbin, _parse_elf

    out = tmp_path / "stripped.so"
    st = strip_fatbin(str(fat_so), str(out))
    assert st["bundles"] >= 1 and st["stripped"] >= 1
    assert out.stat().st_size <= fat_so.stat().st_size

    # wrapper magic intact and ptrs point inside the new section
    data = bytearray(out.read_bytes())
    e = _parse_elf(data)
    fat = e.section(".hip_fatbin")
    seg = e.section(".hipFatBinSegment")
    assert fat is not None and seg is not None

    lib = ctypes.CDLL(str(out))  # registers the fat binary via HIP runtime
    assert lib.lpx_fat_test_launch is not None


def test_strip_result_lists_only_kept_archs(fat_so, tmp_path):
    from lambdipy_amd.packager.fatbin import (_bundle_extent, _parse_elf,
                                              _vaddr_to_off, strip_fatbin)

    out = tmp_path / "stripped2.so"
    strip_fatbin(str(fat_so), str(out))
    data = bytearray(out.read_bytes())
    e = _parse_elf(data)
    fat = e.section(".hip_fatbin")
    off = _vaddr_to_off(e, fat.addr)
    ext = _bundle_extent(bytes(data), off)
    bundle = tmp_path / "b.ccob"
    bundle.write_bytes(bytes(data[off:off + ext]))
    targets = subprocess.run(
        [BUNDLER, "--type=o", "--list", f"--input={bundle}"],
        capture_output=True, text=True, check=True).stdout.split()
    gpu = [t for t in targets if not t.startswith("host-")]
    assert gpu and all(t.endswith("gfx950") for t in gpu), targets


def test_strip_noop_on_plain_elf(tmp_path):
    from lambdipy_amd.packager.fatbin import strip_fatbin

    src = shutil.which("true")
    out = tmp_path / "true_copy"
    st = strip_fatbin(src, str(out))
    assert st["bundles"] == 0
    assert out.read_bytes() == open(src, "rb").read()
