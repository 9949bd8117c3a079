import json
import os
import zipfile

from lambdipy_amd.packager import assemble_layer, dir_size_bytes, make_zip
from lambdipy_amd.packager.prune import prune_tree
from lambdipy_amd.resolver import parse_requirements, resolve


def _mk(root, rel, content=b"x"):
    p = os.path.join(root, rel)
    os.makedirs(os.path.dirname(p), exist_ok=True)
    with open(p, "wb") as f:
        f.write(content)
    return p


def test_prune_globs(tmp_path):
    root = str(tmp_path)
    _mk(root, "pkg/mod.py")
    _mk(root, "pkg/tests/test_a.py")
    _mk(root, "pkg/sub/tests/test_b.py")
    _mk(root, "pkg/data.pyc")
    _mk(root, "pkg/keep.txt")
    n, b = prune_tree(root, ["**/tests/**", "**/*.pyc"])
    assert n == 3 and b == 3
    assert os.path.exists(os.path.join(root, "pkg/mod.py"))
    assert os.path.exists(os.path.join(root, "pkg/keep.txt"))
    assert not os.path.exists(os.path.join(root, "pkg/tests"))
    assert not os.path.exists(os.path.join(root, "pkg/sub/tests"))


def test_prune_whole_dir_pattern(tmp_path):
    root = str(tmp_path)
    _mk(root, "numpy/f2py/a.py")
    _mk(root, "numpy/core.py")
    prune_tree(root, ["numpy/f2py/**"])
    assert not os.path.exists(os.path.join(root, "numpy/f2py"))
    assert os.path.exists(os.path.join(root, "numpy/core.py"))


def test_zip_deterministic(tmp_path):
    root = str(tmp_path / "r")
    _mk(root, "a/b.txt", b"hello")
    _mk(root, "c.txt", b"world")
    z1 = str(tmp_path / "1.zip")
    z2 = str(tmp_path / "2.zip")
    make_zip(root, z1)
    make_zip(root, z2)
    assert open(z1, "rb").read() == open(z2, "rb").read()
    with zipfile.ZipFile(z1) as zf:
        assert sorted(zf.namelist()) == ["a/b.txt", "c.txt"]


def test_assemble_requests_layer(tmp_path):
    """BASELINE.json config 1: pure-Python requests-only build on CPU."""
    res = resolve(parse_requirements("requests\n"))
    build_dir = str(tmp_path / "build")
    zip_path = str(tmp_path / "layer.zip")
    report = assemble_layer(res, build_dir, zip_path=zip_path)
    assert os.path.isdir(os.path.join(build_dir, "requests"))
    assert os.path.isdir(os.path.join(build_dir, "urllib3"))
    man = json.load(open(os.path.join(build_dir, "lpx-manifest.json")))
    assert man["gpu_arch"] == "gfx950"
    assert report.sizes.unzipped_mb < 50
    assert report.sizes.zipped_mb < 10
    assert report.sizes.within_unzipped_limit
    # the layer imports standalone
    import subprocess, sys
    r = subprocess.run(
        [sys.executable, "-c", "import requests; print(requests.__version__)"],
        env={**os.environ, "PYTHONPATH": build_dir},
        capture_output=True, text=True, timeout=120,
    )
    assert r.returncode == 0, r.stderr


def test_dir_size(tmp_path):
    root = str(tmp_path)
    _mk(root, "a.bin", b"\0" * 1000)
    assert dir_size_bytes(root) == 1000
