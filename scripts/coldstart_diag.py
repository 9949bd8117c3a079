"""Cold-start micro-diagnostic: where do hip_init (70 ms) and the first
launch (151 ms) go, and do runtime knobs move them?

Run one variant per fresh process (HIP init is once-per-process):
    python scripts/coldstart_diag.py            # run all variants
    python scripts/coldstart_diag.py --one      # run the probe in-process

Each probe times: dlopen libamdhip64 -> device_count -> set_device ->
hipModuleLoad -> hipModuleGetFunction -> hipMalloc -> first launch (xgen,
tiny) -> sync -> second launch -> sync.
"""

from __future__ import annotations

import json
import os
import subprocess
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

VARIANTS = {
    "baseline": {},
    "sdma_off": {"HSA_ENABLE_SDMA": "0"},
    "one_queue": {"GPU_MAX_HW_QUEUES": "1"},
    "no_direct_dispatch": {"AMD_DIRECT_DISPATCH": "0"},
    "cpu_affinity": {"HIP_CPU_AFFINITY": "1"},
    "visible0": {"HIP_VISIBLE_DEVICES": "0"},
}


def probe() -> dict:
    marks = {}

    def t(name, fn):
        s = time.perf_counter()
        r = fn()
        marks[name] = round((time.perf_counter() - s) * 1e3, 2)
        return r

    from lambdipy_amd.shims import lpx_rt

    t("dlopen_hip", lpx_rt.hip)
    t("device_count", lpx_rt.device_count)
    t("set_device", lambda: lpx_rt.set_device(0))
    mod = t("module_load", lpx_rt.Module)
    t("get_function", lambda: mod.fn("lpx_xgen"))
    buf = t("hipMalloc", lambda: lpx_rt.malloc(1 << 20))
    t("first_launch", lambda: lpx_rt.xgen(mod, buf, 1024, 7))
    t("first_sync", lpx_rt.sync)
    t("second_launch", lambda: lpx_rt.xgen(mod, buf, 1024, 7))
    t("second_sync", lpx_rt.sync)
    marks["total"] = round(sum(marks.values()), 1)
    return marks


def main():
    if "--one" in sys.argv:
        print("DIAG " + json.dumps(probe()), flush=True)
        return
    # interleaved repeats: the FIRST process on a box absorbs one-time
    # driver/KFD warm-up, so run each variant several times alternating
    order = [(n, VARIANTS[n]) for n in VARIANTS]
    reps = int(os.environ.get("LPX_DIAG_REPS", "1"))
    for r in range(reps):
        for name, env in order:
            _run(name + f"#{r}", env)


def _run(name, env):
    if True:
        e = dict(os.environ, **env)
        out = subprocess.run(
            [sys.executable, os.path.abspath(__file__), "--one"],
            env=e, capture_output=True, text=True, timeout=120)
        line = next((ln for ln in out.stdout.splitlines()
                     if ln.startswith("DIAG ")), None)
        if line:
            print(f"{name:20s} {line[5:]}")
        else:
            print(f"{name:20s} FAILED rc={out.returncode} "
                  f"{out.stderr[-300:]}")


if __name__ == "__main__":
    main()
