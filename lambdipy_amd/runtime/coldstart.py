"""Cold-start harness — the metric path.

Times unzip -> fresh-interpreter spawn -> import -> hipInit -> model ->
first inference, per GPU count. t0 is taken BEFORE the unzip; t1 when rank 0
prints its LPX_RESULT line. For g > 1 the harness spawns g processes itself
(one per GPU, RANK/WORLD_SIZE/MASTER_* env) — no torchrun wrapper overhead
in the timed region.
"""

from __future__ import annotations

import json
import os
import shutil
import socket
import subprocess
import sys
import tempfile
import time
import zipfile
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from ..utils.log import get_logger

log = get_logger("runtime")

HANDLERS = {
    "requests": "requests_handler",
    "matmul": "matmul_handler",
    "matmul_lite": "matmul_lite_handler",
    "fftspmv": "fftspmv_handler",
    "fftspmv_lite": "fftspmv_lite_handler",
    "rf": "rf_handler",
    "rf_lite": "rf_lite_handler",
    "resnet50": "resnet50_handler",
    "resnet50_lite": "resnet50_lite_handler",
}

# torch-free handlers (ctypes over AOT gfx950 artifacts): all their layer
# bytes are extracted eagerly and they skip the ready-marker wait — the bulk
# unzip overlaps the whole GPU cold start.
LITE_HANDLERS = {"rf_lite", "matmul_lite", "fftspmv_lite", "resnet50_lite"}
# single-process multi-GPU (ncclCommInitAll) env knob per lite handler
LITE_GPUS_ENV = {"rf_lite": "LPX_RF_GPUS", "resnet50_lite": "LPX_RESNET_GPUS"}


@dataclass
class ColdStartResult:
    handler: str
    gpus: int
    total_ms: float  # t0 (pre-unzip) -> rank-0 LPX_RESULT
    unzip_ms: float
    handler_report: dict
    layer_mb: Optional[float] = None
    returncodes: List[int] = field(default_factory=list)

    def to_dict(self) -> dict:
        return {
            "handler": self.handler,
            "gpus": self.gpus,
            "coldstart_ms": round(self.total_ms, 1),
            "unzip_ms": round(self.unzip_ms, 1),
            "layer_mb": self.layer_mb,
            "handler_report": self.handler_report,
        }


def _parallel_extract(zip_path: str, dest: str, workers: int = 8,
                      only=None) -> None:
    """Multi-threaded zip extraction (sharded, one ZipFile per worker).

    only: optional predicate on archive names (partial extraction — the
    cold-start harness extracts handler entry files first)."""
    from concurrent.futures import ThreadPoolExecutor

    with zipfile.ZipFile(zip_path) as zf:
        infos = [i for i in zf.infolist() if not i.filename.endswith("/")]
    if only is not None:
        infos = [i for i in infos if only(i.filename)]
    dirs = {os.path.dirname(i.filename) for i in infos}
    for d in sorted(dirs):
        if d:
            os.makedirs(os.path.join(dest, d), exist_ok=True)
    # largest-first round-robin shards; each worker opens its OWN ZipFile
    # (a shared handle serializes on its lock and is slower than extractall)
    infos.sort(key=lambda i: i.file_size, reverse=True)
    shards = [infos[w::workers] for w in range(workers)]

    def _shard(shard):
        with zipfile.ZipFile(zip_path) as z:
            for info in shard:
                out = os.path.join(dest, info.filename)
                with z.open(info) as src, open(out, "wb") as dst:
                    shutil.copyfileobj(src, dst, 1 << 20)
                mode = (info.external_attr >> 16) & 0o777
                if mode:
                    os.chmod(out, mode)

    with ThreadPoolExecutor(max_workers=workers) as ex:
        list(ex.map(_shard, shards))


def prewarm_host(timeout_s: float = 120.0) -> float:
    """Warm the host's HIP module-launch path before serving requests.

    Measured on MI355X (scripts/coldstart_diag.py, profiles/README.md): the
    FIRST process on a box to go through hipModuleLoad + hipModuleLaunchKernel
    pays a one-time ~150 ms KFD/queue bring-up; every later process pays
    ~14 ms, even after the first one exits, and a resident torch process does
    NOT absorb it (different init path). That cost is host state, not
    function cold start — serving platforms absorb it by pre-warming workers
    before any request arrives, which is what this does: one subprocess that
    loads the runtime's own AOT .hsaco (no user layer bytes) and launches a
    tiny kernel. Returns elapsed ms; 0.0 if there is no usable GPU.
    """
    code = (
        "from lambdipy_amd.shims import lpx_rt\n"
        "lpx_rt.set_device(0)\n"
        "m = lpx_rt.Module()\n"
        "buf = lpx_rt.malloc(4096)\n"
        "lpx_rt.xgen(m, buf, 1024, 1)\n"
        "lpx_rt.sync()\n"
    )
    t0 = time.perf_counter()
    env = dict(os.environ)
    env["PYTHONPATH"] = (
        os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
        + (os.pathsep + env["PYTHONPATH"] if env.get("PYTHONPATH") else "")
    )
    try:
        r = subprocess.run([sys.executable, "-c", code], env=env,
                           capture_output=True, timeout=timeout_s)
        if r.returncode != 0:
            log.debug("prewarm_host: no GPU / failed (%s)", r.stderr[-200:])
            return 0.0
    except (subprocess.TimeoutExpired, OSError):
        return 0.0
    return (time.perf_counter() - t0) * 1e3


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def cold_start(
    layer: str,
    handler: str = "rf",
    gpus: int = 1,
    *,
    env: Optional[Dict[str, str]] = None,
    timeout_s: float = 600.0,
    keep_dir: bool = False,
    wrap_cmd: Optional[List[str]] = None,
) -> ColdStartResult:
    """Run one cold start of `handler` from `layer` (zip path or dir)."""
    mod = HANDLERS.get(handler, handler)
    layer_mb = None

    # failure handling: degrade to the largest available GPU
    # count (8 -> 4 -> 2 -> 1) instead of hanging in comm init
    if gpus > 1:
        try:
            import torch

            avail = torch.cuda.device_count() if torch.cuda.is_available() else 0
        except Exception:
            avail = 0
        if avail and gpus > avail:
            g = avail
            while g & (g - 1):
                g &= g - 1  # largest power of two <= avail
            log.warning("requested %d GPUs but only %d available; degrading to %d",
                        gpus, avail, g)
            gpus = max(g, 1)

    t0 = time.perf_counter()

    # Overlapped cold start: extract only the handler entry files, SPAWN the
    # interpreter immediately (its ~0.8 s torch import runs concurrently with
    # the bulk unzip), then finish extraction and release the ready marker
    # the handler waits on before importing the layer's shims.
    tmpdir = None
    ready_file = None
    deferred_extract = None
    if os.path.isfile(layer) and layer.endswith(".zip"):
        layer_mb = os.path.getsize(layer) / 2**20
        tmpdir = tempfile.mkdtemp(prefix="lpx-layer-")
        ready_file = os.path.join(tmpdir, ".lpx_ready")
        # the torch-free lite handlers import ONLY lpx_rt/lpx_nn + their AOT
        # artifacts (.hsaco / liblpx_nn): extract those eagerly and skip the
        # ready-marker wait entirely (the bulk unzip overlaps the whole GPU
        # cold start). The torch extension .so in lpx_ops/ (~14 MB) is NOT
        # part of any lite path — extracting it eagerly would serialize
        # ~0.1 s of inflate before the spawn, so it stays in the deferred
        # bulk pass; liblpx_nn (~2 MB) IS needed before the first launch.
        if handler in LITE_HANDLERS:
            # rf_lite needs only lpx_rt + the .hsaco; the NN lite handlers
            # also need lpx_nn + liblpx_nn (~2 MB — don't inflate it ahead
            # of a handler that never dlopens it)
            needs_nn = handler != "rf_lite"

            def early(n):
                base = os.path.basename(n)
                return (n.startswith("lpx_handlers/") or n == "lpx_fanout.py"
                        or n == "lpx_rt.py"
                        or (needs_nn and n == "lpx_nn.py")
                        or (needs_nn and base.startswith("liblpx_nn"))
                        or base.endswith(".hsaco")
                        or (n.startswith("lpx_ops/")
                            and not base.startswith("_lpx_ops")
                            and not (not needs_nn
                                     and base.startswith("liblpx_nn"))))
        else:
            def early(n):
                return n.startswith("lpx_handlers/") or n == "lpx_fanout.py"
        _parallel_extract(layer, tmpdir, only=early)
        layer_dir = tmpdir

        def deferred_extract():
            _parallel_extract(layer, tmpdir, only=lambda n: not early(n))
            with open(ready_file, "w") as f:
                f.write("ready")
    else:
        layer_dir = layer
    unzip_ms = (time.perf_counter() - t0) * 1e3

    base_env = dict(os.environ)
    if env:
        base_env.update(env)
    # the layer provides its packages; the image provides python + torch
    base_env["PYTHONPATH"] = layer_dir + (
        os.pathsep + base_env["PYTHONPATH"] if base_env.get("PYTHONPATH") else ""
    )
    base_env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    if ready_file and handler not in LITE_HANDLERS:  # lite files are all early
        base_env["LPX_READY_FILE"] = ready_file

    code = f"import lpx_handlers.{mod} as h; h.main()"
    argv = [sys.executable, "-c", code]
    if wrap_cmd:  # e.g. rocprofv3 --kernel-trace --stats -d DIR --
        argv = list(wrap_cmd) + argv
    procs = []
    if handler in LITE_GPUS_ENV and gpus > 1:
        # single-process multi-GPU (ncclCommInitAll) — no per-rank spawn
        e = dict(base_env, **{LITE_GPUS_ENV[handler]: str(gpus)})
        procs.append(
            subprocess.Popen(argv, env=e, stdout=subprocess.PIPE,
                             stderr=subprocess.PIPE, text=True)
        )
    elif gpus <= 1:
        procs.append(
            subprocess.Popen(argv, env=base_env,
                             stdout=subprocess.PIPE, stderr=subprocess.PIPE,
                             text=True)
        )
    else:
        port = _free_port()
        for r in range(gpus):
            e = dict(base_env)
            e.update(
                RANK=str(r), LOCAL_RANK=str(r), WORLD_SIZE=str(gpus),
                MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
            )
            procs.append(
                subprocess.Popen(argv if r == 0 else [sys.executable, "-c", code],
                                 env=e,
                                 stdout=subprocess.PIPE, stderr=subprocess.PIPE,
                                 text=True)
            )

    # bulk extraction runs CONCURRENTLY with the handler (zlib releases the
    # GIL): first-inference latency must not wait on layer bytes the handler
    # does not read — the rf_lite path needs only its eager set, and the
    # torch handlers gate themselves on the ready marker this thread writes.
    extract_thread = None
    extract_err: List[BaseException] = []
    tex0 = time.perf_counter()
    if deferred_extract is not None:
        import threading

        def _bg_extract():
            try:
                deferred_extract()
            except BaseException as e:  # surfaced after the result read
                extract_err.append(e)

        extract_thread = threading.Thread(target=_bg_extract, daemon=True)
        extract_thread.start()

    # Drain every pipe we are not actively reading (rank-0 stderr, all other
    # ranks' stdout/stderr) on background threads: a handler that emits >64 KB
    # of warnings/RCCL debug before LPX_RESULT must not deadlock the harness
    # on a full pipe.
    import threading

    def _drain(stream, buf):
        try:
            for ln in stream:
                buf.append(ln)
        except ValueError:  # closed under us at teardown
            pass

    drain_threads: List[threading.Thread] = []
    err0_buf: List[str] = []
    for i, p in enumerate(procs):
        for stream, buf in (((p.stderr, err0_buf),) if i == 0
                            else ((p.stdout, []), (p.stderr, []))):
            th = threading.Thread(target=_drain, args=(stream, buf), daemon=True)
            th.start()
            drain_threads.append(th)

    report = {}
    total_ms = None
    try:
        # t1 is taken when the LPX_RESULT line ARRIVES (the docstring
        # contract), not when the process exits — exit adds ~50 ms of
        # Python + HIP runtime teardown that is not cold-start latency.
        import selectors

        sel = selectors.DefaultSelector()
        sel.register(procs[0].stdout, selectors.EVENT_READ)
        deadline = time.perf_counter() + timeout_s
        lines0 = []
        while True:
            if not sel.select(timeout=max(0.0, deadline - time.perf_counter())):
                raise subprocess.TimeoutExpired(argv, timeout_s)
            line = procs[0].stdout.readline()
            if line == "":  # EOF before LPX_RESULT — collect stderr below
                break
            lines0.append(line)
            if line.startswith("LPX_RESULT "):
                total_ms = (time.perf_counter() - t0) * 1e3
                report = json.loads(line[len("LPX_RESULT "):])
                break
        sel.close()
        # hand rank-0's remaining stdout to a drain thread too
        th = threading.Thread(target=_drain, args=(procs[0].stdout, lines0),
                              daemon=True)
        th.start()
        drain_threads.append(th)
        if extract_thread is not None:
            extract_thread.join(timeout=timeout_s)
            unzip_ms += (time.perf_counter() - tex0) * 1e3
            if extract_err:
                raise extract_err[0]
        procs[0].wait(timeout=timeout_s)
        if not report:
            for th in drain_threads:
                th.join(timeout=5)
            out0 = "".join(lines0)
            err0 = "".join(err0_buf)
            raise RuntimeError(
                f"handler produced no LPX_RESULT (rc={procs[0].returncode})\n"
                f"stdout:\n{out0[-4000:]}\nstderr:\n{err0[-4000:]}"
            )
        rcs = [procs[0].returncode]
        for p in procs[1:]:
            try:
                p.wait(timeout=60)
            except subprocess.TimeoutExpired:
                p.kill()
            rcs.append(p.returncode)
        for th in drain_threads:
            th.join(timeout=5)
    finally:
        for p in procs:
            if p.poll() is None:
                p.kill()
        if extract_thread is not None and extract_thread.is_alive():
            extract_thread.join(timeout=60)  # never rmtree under the writer
        if tmpdir and not keep_dir:
            shutil.rmtree(tmpdir, ignore_errors=True)

    if procs[0].returncode != 0:
        raise RuntimeError(f"handler rank 0 exited {procs[0].returncode}")
    return ColdStartResult(
        handler=handler,
        gpus=gpus,
        total_ms=total_ms,
        unzip_ms=unzip_ms,
        handler_report=report,
        layer_mb=round(layer_mb, 2) if layer_mb is not None else None,
        returncodes=rcs,
    )
