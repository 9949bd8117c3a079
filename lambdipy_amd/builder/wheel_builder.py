"""ROCm wheel builder — installs recipe artifacts into the layer.

Reference analog: the docker builder that compiled native
packages inside Lambda-like images. Here there is no docker and *we* are the
runtime, so a native recipe's artifact is built directly:

- the gfx950 ops extension (``lambdipy_amd/ops/_lpx_ops*.so``) is AOT-compiled
  in-tree (``setup.py build_ext --inplace`` drives hipcc with
  PYTORCH_ROCM_ARCH=gfx950) and bundled into the layer as the ``lpx_ops``
  package (loader + .so, no JIT at cold start — SURVEY.md §2.2 N11);
- the recipe's shim module (lpx_numpy / lpx_scipy / lpx_sklearn / lpx_torch)
  is copied next to it.

Artifact reuse: the in-tree .so acts as the content-addressed cache entry
(rebuilt only when sources change — setup.py/ninja dependency tracking), the
analog of lambdipy's prebuilt-artifact reuse (SURVEY.md §5 checkpoint row).
"""

from __future__ import annotations

import glob
import os
import subprocess
import sys
from typing import Optional

from ..recipes import BuildRecipe
from ..resolver import ResolvedDep
from ..utils.log import get_logger

log = get_logger("builder")

PKG_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))  # lambdipy_amd/
REPO_ROOT = os.path.dirname(PKG_ROOT)
OPS_DIR = os.path.join(PKG_ROOT, "ops")
SHIMS_DIR = os.path.join(PKG_ROOT, "shims")


def ops_extension_path() -> Optional[str]:
    hits = sorted(glob.glob(os.path.join(OPS_DIR, "_lpx_ops*.so")))
    return hits[0] if hits else None


HSACO_PATH = os.path.join(OPS_DIR, "standalone_gfx950.hsaco")
HSACO_SRC = os.path.join(OPS_DIR, "csrc", "standalone.hip")


def ensure_standalone_hsaco(allow_build: bool = True) -> str:
    """AOT code object for the torch-free runtime (hipcc --genco, gfx950).

    Rebuilt when missing or older than its source; ships in-tree (the .hsaco
    is git-ignored but travels with snapshots / into recipe wheels)."""
    if os.path.exists(HSACO_PATH) and (
        os.path.getmtime(HSACO_PATH) >= os.path.getmtime(HSACO_SRC)
    ):
        return HSACO_PATH
    if not allow_build:
        raise RuntimeError("standalone hsaco missing and allow_build=False")
    log.info("building standalone gfx950 code object (hipcc --genco)...")
    subprocess.run(
        ["hipcc", "--genco", "--offload-arch=gfx950", "-O3", HSACO_SRC,
         "-o", HSACO_PATH],
        check=True,
    )
    return HSACO_PATH


NN_LIB_PATH = os.path.join(OPS_DIR, "liblpx_nn_gfx950.so")


def ensure_nn_lib(allow_build: bool = True) -> str:
    """The torch-free serving runtime library (liblpx_nn_gfx950.so).

    Compiled with hipcc straight from the SAME kernel sources the torch
    extension uses (ops/csrc/*.hip) plus the C API (nn_capi.cpp) — one
    kernel code base, two front ends. AOT gfx950, ships in-tree (git-ignored
    .so travels with snapshots / into recipe wheels). Parallel per-file
    compiles with mtime-based reuse; objects cached next to the sources in
    ops/csrc/.nnobj/."""
    csrc = os.path.join(OPS_DIR, "csrc")
    sources = sorted(
        os.path.join(csrc, f)
        for f in os.listdir(csrc)
        if (f.endswith(".hip") or f == "nn_capi.cpp")
        and not f.endswith("_hip.hip") and f != "standalone.hip"
    )
    headers = [os.path.join(csrc, h) for h in ("kernels.h", "lpx_common.h")]
    hdr_mtime = max(os.path.getmtime(h) for h in headers)
    if os.path.exists(NN_LIB_PATH):
        lib_m = os.path.getmtime(NN_LIB_PATH)
        if lib_m >= hdr_mtime and all(
            lib_m >= os.path.getmtime(s) for s in sources
        ):
            return NN_LIB_PATH
    if not allow_build:
        raise RuntimeError("liblpx_nn missing and allow_build=False")
    log.info("building liblpx_nn_gfx950.so (hipcc, %d sources)...", len(sources))
    objdir = os.path.join(csrc, ".nnobj")
    os.makedirs(objdir, exist_ok=True)
    flags = ["--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC",
             "-fvisibility=hidden", "-I", csrc]
    procs = []
    objs = []
    for s in sources:
        o = os.path.join(objdir, os.path.basename(s) + ".o")
        objs.append(o)
        if os.path.exists(o) and os.path.getmtime(o) >= max(
            os.path.getmtime(s), hdr_mtime
        ):
            continue
        procs.append((s, subprocess.Popen(
            ["hipcc", "-c", s, "-o", o] + flags,
            stdout=subprocess.PIPE, stderr=subprocess.PIPE)))
    for s, p in procs:
        out, err = p.communicate()
        if p.returncode != 0:
            raise RuntimeError(
                f"hipcc failed on {s}:\n{err.decode()[-4000:]}")
    subprocess.run(
        ["hipcc", "-shared", "-fPIC", "--offload-arch=gfx950",
         "-o", NN_LIB_PATH] + objs,
        check=True,
    )
    return NN_LIB_PATH


def ensure_ops_extension(allow_build: bool = True) -> str:
    """Return the built extension path, building it in-tree if needed."""
    p = ops_extension_path()
    if p:
        return p
    if not allow_build:
        raise RuntimeError("ops extension not built and allow_build=False")
    log.info("building gfx950 ops extension (hipcc AOT)...")
    env = dict(os.environ, PYTORCH_ROCM_ARCH="gfx950")
    subprocess.run(
        [sys.executable, "setup.py", "build_ext", "--inplace"],
        cwd=REPO_ROOT,
        env=env,
        check=True,
    )
    p = ops_extension_path()
    if not p:
        raise RuntimeError("extension build reported success but no .so found")
    return p


def _read(path: str) -> bytes:
    with open(path, "rb") as f:
        return f.read()


def emit_ops_wheel(dist_dir: str) -> str:
    """Emit (or reuse) the lpx_ops ROCm wheel: loader + AOT gfx950 .so +
    the standalone .hsaco for the torch-free cold-start runtime."""
    from .wheelgen import write_wheel

    so = ensure_ops_extension()
    hsaco = ensure_standalone_hsaco()
    nnlib = ensure_nn_lib()
    files = {
        "lpx_ops/__init__.py": _read(os.path.join(OPS_DIR, "__init__.py")),
        "lpx_ops/api.py": _read(os.path.join(OPS_DIR, "api.py")),
        f"lpx_ops/{os.path.basename(so)}": _read(so),
        f"lpx_ops/{os.path.basename(hsaco)}": _read(hsaco),
        f"lpx_ops/{os.path.basename(nnlib)}": _read(nnlib),
    }
    return write_wheel(
        dist_dir, "lpx_ops", files, platform_specific=True,
        summary="lambdipy_amd gfx950 kernel library (MFMA/LDS HIP, AOT)",
    )


def emit_shim_wheel(recipe: BuildRecipe, dist_dir: str) -> str:
    """Emit (or reuse) the per-recipe shim wheel (depends on lpx_ops)."""
    from .wheelgen import write_wheel

    shim_src = os.path.join(SHIMS_DIR, recipe.shim + ".py")
    if not os.path.exists(shim_src):
        raise FileNotFoundError(f"recipe {recipe.name}: shim {recipe.shim} not found")
    return write_wheel(
        dist_dir, recipe.shim, {recipe.shim + ".py": _read(shim_src)},
        requires=["lpx_ops"],
        summary=f"{recipe.name} GPU shim (kernels: {', '.join(recipe.kernels)})",
    )


def _unpack_wheel(wheel_path: str, build_dir: str) -> None:
    from ..packager.assemble import _vendor_wheel

    _vendor_wheel(wheel_path, build_dir)


def build_recipe_artifacts(recipe: BuildRecipe, dep: ResolvedDep, build_dir: str,
                           dist_dir: Optional[str] = None) -> dict:
    """The wheel_builder hook for assemble_layer (native recipes only).

    Emits real ROCm wheels into the wheel cache, then unpacks them into the
    layer (the reference's prebuilt-else-build flow)."""
    from .wheelgen import WHEEL_CACHE

    dist = dist_dir or WHEEL_CACHE
    ops_whl = emit_ops_wheel(dist)
    _unpack_wheel(ops_whl, build_dir)
    info = {
        "wheels": [os.path.basename(ops_whl)],
        "so_mb": round(os.path.getsize(ensure_ops_extension()) / 2**20, 2),
    }
    if recipe.shim:
        shim_whl = emit_shim_wheel(recipe, dist)
        _unpack_wheel(shim_whl, build_dir)
        info["wheels"].append(os.path.basename(shim_whl))
        info["shim"] = recipe.shim
    info["kernels"] = recipe.kernels
    return info
