"""CPU tests: Pipfile(.lock) front-end, keyed artifact store, pandas recipe
(VERDICT round-1 item 5 — packager parity gaps)."""

import json
import os

import pytest

from lambdipy_amd import store
from lambdipy_amd.resolver import (parse_pipfile, parse_pipfile_lock,
                                   parse_project_file)

PIPFILE = """\
[[source]]
url = "https://pypi.org/simple"
verify_ssl = true
name = "pypi"

[packages]
requests = "*"
numpy = ">=1.24"
scipy = {version = ">=1.10", markers = "python_version >= '3.8'"}

[dev-packages]
pytest = "*"
"""

PIPFILE_LOCK = {
    "_meta": {"hash": {"sha256": "x"}, "pipfile-spec": 6},
    "default": {
        "requests": {"version": "==2.31.0", "hashes": []},
        "numpy": {"version": "==1.26.4", "hashes": [],
                  "markers": "python_version >= '3.9'"},
    },
    "develop": {"pytest": {"version": "==8.0.0"}},
}


def test_parse_pipfile(tmp_path):
    p = tmp_path / "Pipfile"
    p.write_text(PIPFILE)
    reqs = parse_pipfile(str(p))
    by = {r.name: r for r in reqs}
    assert set(by) == {"requests", "numpy", "scipy"}  # dev excluded
    assert by["requests"].specifier == ""
    assert by["numpy"].specifier == ">=1.24"
    assert by["scipy"].marker == "python_version >= '3.8'"
    dev = parse_pipfile(str(p), dev=True)
    assert "pytest" in {r.name for r in dev}


def test_parse_pipfile_lock(tmp_path):
    p = tmp_path / "Pipfile.lock"
    p.write_text(json.dumps(PIPFILE_LOCK))
    reqs = parse_pipfile_lock(str(p))
    by = {r.name: r for r in reqs}
    assert set(by) == {"requests", "numpy"}
    assert by["requests"].pinned_version == "2.31.0"
    assert by["numpy"].pinned_version == "1.26.4"
    assert by["numpy"].marker == "python_version >= '3.9'"


def test_parse_project_file_prefers_lock(tmp_path):
    (tmp_path / "Pipfile").write_text(PIPFILE)
    (tmp_path / "Pipfile.lock").write_text(json.dumps(PIPFILE_LOCK))
    reqs = parse_project_file(str(tmp_path / "Pipfile"))
    assert {r.name for r in reqs} == {"requests", "numpy"}  # lock wins
    # and requirements.txt dispatch still works
    (tmp_path / "requirements.txt").write_text("einops\n")
    reqs = parse_project_file(str(tmp_path / "requirements.txt"))
    assert [r.name for r in reqs] == ["einops"]


def test_pipfile_lock_resolves(tmp_path):
    """The lock front-end feeds the normal resolver (recipe split etc.)."""
    from lambdipy_amd.resolver import resolve

    p = tmp_path / "Pipfile.lock"
    p.write_text(json.dumps({
        "_meta": {}, "default": {"scikit-learn": {"version": "*"}},
        "develop": {}}))
    res = resolve(parse_pipfile_lock(str(p)))
    names = {d.name for d in res.deps}
    assert "scikit-learn" in names
    assert any(d.recipe == "scikit-learn" for d in res.recipe_hits)


# ---- keyed artifact store -------------------------------------------------

def _touch(path, content=b"payload"):
    with open(path, "wb") as f:
        f.write(content)
    return str(path)


def test_store_publish_fetch_layer(tmp_path):
    rd = str(tmp_path / "rel")
    src = _touch(tmp_path / "mylayer.zip", b"zipbytes")
    a = store.publish_file(src, rd)
    assert a.kind == "layer" and a.sha256
    out = store.fetch_asset("mylayer", str(tmp_path / "out.zip"), rd)
    with open(out, "rb") as f:
        assert f.read() == b"zipbytes"


def test_store_wheel_keying(tmp_path):
    rd = str(tmp_path / "rel")
    w = _touch(tmp_path / "lpx_ops-0.1.0+gfx950.abc123def456-cp310-cp310-linux_x86_64.whl",
               b"wheel1")
    a = store.publish_file(w, rd)
    assert a.name == "lpx-ops"
    assert a.version == "0.1.0+gfx950.abc123def456"
    assert a.python == "cp310"
    assert a.arch == "gfx950"
    # keyed fetch: right keys hit, wrong keys miss
    out = store.fetch_asset("lpx-ops", str(tmp_path / "o.whl"), rd,
                            version="0.1.0", python="cp310", arch="gfx950")
    assert os.path.exists(out)
    with pytest.raises(store.StoreError):
        store.fetch_asset("lpx-ops", str(tmp_path / "o2.whl"), rd,
                          python="cp311")
    with pytest.raises(store.StoreError):
        store.fetch_asset("lpx-ops", str(tmp_path / "o3.whl"), rd,
                          arch="gfx90a")


def test_store_hash_verification(tmp_path):
    rd = str(tmp_path / "rel")
    src = _touch(tmp_path / "corrupt.zip", b"good")
    a = store.publish_file(src, rd)
    with open(a.path, "wb") as f:  # corrupt the stored asset
        f.write(b"evil")
    with pytest.raises(store.StoreError, match="hash mismatch"):
        store.fetch_asset("corrupt", str(tmp_path / "o.zip"), rd)


def test_cli_publish_fetch_roundtrip(tmp_path):
    from click.testing import CliRunner

    from lambdipy_amd.cli import main as cli_main

    src = _touch(tmp_path / "layer.zip", b"abc")
    rd = str(tmp_path / "rel")
    r = CliRunner().invoke(cli_main, ["publish", str(src),
                                      "--release-dir", rd])
    assert r.exit_code == 0, r.output
    out = str(tmp_path / "fetched.zip")
    r = CliRunner().invoke(cli_main, ["fetch", "layer", "-o", out,
                                      "--release-dir", rd])
    assert r.exit_code == 0, r.output
    assert os.path.exists(out)


# ---- pandas recipe --------------------------------------------------------

def test_pandas_recipe_registered():
    from lambdipy_amd.recipes import get_registry

    r = get_registry().get("pandas")
    assert r is not None and r.kind == "native"
    assert "pandas" in r.provides
    assert r.strip_native
    assert any("tests" in p for p in r.prune)


def test_pandas_resolves_to_recipe():
    from lambdipy_amd.resolver import parse_requirements, resolve

    res = resolve(parse_requirements("pandas\n"))
    assert any(d.recipe == "pandas" for d in res.recipe_hits)


def test_pandas_layer_assembles(tmp_path):
    """pandas vendors + prunes into a layer (no hipcc needed: wheel_builder
    omitted, matching the packager's unit-test contract)."""
    from lambdipy_amd.packager import assemble_layer
    from lambdipy_amd.resolver import parse_requirements, resolve

    res = resolve(parse_requirements("pandas\n"), follow_transitive=False)
    d = str(tmp_path / "build")
    report = assemble_layer(res, d)
    assert os.path.isdir(os.path.join(d, "pandas"))
    assert not os.path.isdir(os.path.join(d, "pandas", "tests"))
    assert any(x["name"] == "pandas" for x in report.deps)


def test_grpcio_recipe(tmp_path):
    from lambdipy_amd.packager import assemble_layer
    from lambdipy_amd.recipes import get_registry
    from lambdipy_amd.resolver import parse_requirements, resolve

    r = get_registry().get("grpcio")
    assert r is not None and r.kind == "native" and r.strip_native
    res = resolve(parse_requirements("grpcio\n"), follow_transitive=False)
    assert any(d.recipe == "grpcio" for d in res.recipe_hits)
    d = str(tmp_path / "build")
    assemble_layer(res, d)
    assert os.path.isdir(os.path.join(d, "grpc"))
