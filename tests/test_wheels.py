"""ROCm wheel emission (recipes emit ROCm wheels)."""

import os
import subprocess
import sys
import zipfile

import pytest

from lambdipy_amd.builder.wheel_builder import emit_ops_wheel, emit_shim_wheel
from lambdipy_amd.recipes import get_registry


@pytest.fixture()
def dist(tmp_path):
    return str(tmp_path / "dist")


def test_ops_wheel_contents(dist):
    whl = emit_ops_wheel(dist)
    assert os.path.basename(whl).startswith("lpx_ops-")
    assert "linux_x86_64" in whl
    with zipfile.ZipFile(whl) as zf:
        names = zf.namelist()
    assert any(n.endswith(".so") for n in names)
    assert "lpx_ops/__init__.py" in names
    assert any(n.endswith("METADATA") for n in names)
    assert any(n.endswith("RECORD") for n in names)


def test_ops_wheel_content_addressed(dist):
    w1 = emit_ops_wheel(dist)
    w2 = emit_ops_wheel(dist)
    assert w1 == w2  # cache hit, same content key


def test_shim_wheels_for_all_native_recipes(dist):
    reg = get_registry()
    for name in ("numpy", "scipy", "scikit-learn", "torch"):
        whl = emit_shim_wheel(reg.get(name), dist)
        with zipfile.ZipFile(whl) as zf:
            meta = next(n for n in zf.namelist() if n.endswith("METADATA"))
            txt = zf.read(meta).decode()
        assert "Requires-Dist: lpx_ops" in txt


def test_wheel_pip_installable(dist, tmp_path):
    whl = emit_shim_wheel(get_registry().get("numpy"), dist)
    target = str(tmp_path / "site")
    r = subprocess.run(
        [sys.executable, "-m", "pip", "install", "--no-index", "--no-deps",
         "--target", target, whl],
        capture_output=True, text=True, timeout=180,
    )
    assert r.returncode == 0, r.stderr
    assert os.path.exists(os.path.join(target, "lpx_numpy.py"))
