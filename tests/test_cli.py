import json
import os

import pytest
from click.testing import CliRunner

from lambdipy_amd.cli import main as cli_main


@pytest.fixture()
def runner():
    return CliRunner()


def test_recipes_listing(runner):
    r = runner.invoke(cli_main, ["recipes"])
    assert r.exit_code == 0, r.output
    assert "numpy" in r.output and "scikit-learn" in r.output


def test_build_and_run_requests_layer(runner, tmp_path):
    """Config 1 end-to-end: build -> zip -> cold-start (CPU only)."""
    req = tmp_path / "requirements.txt"
    req.write_text("requests\n")
    out = str(tmp_path / "build")
    zp = str(tmp_path / "layer.zip")
    r = runner.invoke(
        cli_main,
        ["build", "-r", str(req), "-o", out, "-z", zp,
         "--handler", "requests", "--json"],
    )
    assert r.exit_code == 0, r.output
    rep = json.loads(r.output[r.output.index("{"):])
    assert rep["within_250mb_unzipped"]
    assert os.path.exists(os.path.join(out, "lpx_handlers", "requests_handler.py"))
    assert os.path.exists(os.path.join(out, "lpx_fanout.py"))

    from lambdipy_amd.runtime import cold_start

    res = cold_start(zp, handler="requests", gpus=1, timeout_s=300)
    assert res.handler_report["handler"] == "requests"
    assert res.total_ms > 0
    assert res.layer_mb is not None and res.layer_mb < 10


def test_publish(runner, tmp_path):
    zp = tmp_path / "x.zip"
    import zipfile

    with zipfile.ZipFile(zp, "w") as zf:
        zf.writestr("a.txt", "hi")
    rel = tmp_path / "releases"
    r = runner.invoke(cli_main, ["publish", str(zp), "--release-dir", str(rel)])
    assert r.exit_code == 0, r.output
    # store keys assets as <name>.zip.<sha16> + sidecar json
    assert any(".zip." in f and not f.endswith(".json")
               for f in os.listdir(rel))


def test_publish_fetch_roundtrip(runner, tmp_path):
    import zipfile

    zp = tmp_path / "mylayer.zip"
    with zipfile.ZipFile(zp, "w") as zf:
        zf.writestr("m.txt", "content")
    rel = str(tmp_path / "rel")
    r = runner.invoke(cli_main, ["publish", str(zp), "--release-dir", rel])
    assert r.exit_code == 0, r.output
    out = str(tmp_path / "fetched.zip")
    r = runner.invoke(cli_main, ["fetch", "mylayer", "-o", out,
                                 "--release-dir", rel])
    assert r.exit_code == 0, r.output
    with zipfile.ZipFile(out) as zf:
        assert zf.read("m.txt") == b"content"


def test_fetch_missing(runner, tmp_path):
    r = runner.invoke(cli_main, ["fetch", "nope", "--release-dir",
                                 str(tmp_path / "empty")])
    assert r.exit_code != 0


def test_cold_start_t1_at_result_not_exit(tmp_path):
    """t1 must be taken when LPX_RESULT arrives, not at process exit
    (coldstart.py streams stdout; a handler that lingers after printing
    must not inflate total_ms), and bulk extraction must overlap."""
    import json as _json
    import zipfile

    from lambdipy_amd.runtime import cold_start

    zp = tmp_path / "layer.zip"
    handler = (
        "import json, time\n"
        "def main():\n"
        "    print('LPX_RESULT ' + json.dumps({'handler': 'lingerer'}),"
        " flush=True)\n"
        "    time.sleep(1.5)\n"  # teardown time that must NOT be timed
    )
    with zipfile.ZipFile(zp, "w") as zf:
        zf.writestr("lpx_handlers/__init__.py", "")
        zf.writestr("lpx_handlers/linger_handler.py", handler)
        zf.writestr("bulk/pad.bin", b"\0" * (1 << 20))  # deferred-set file
    res = cold_start(str(zp), handler="linger_handler", gpus=1, timeout_s=60)
    assert res.handler_report == {"handler": "lingerer"}
    # the 1.5 s linger must not be inside total_ms
    assert res.total_ms < 1200, res.total_ms
