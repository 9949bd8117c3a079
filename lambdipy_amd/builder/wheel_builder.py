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
import shutil
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


def _install_lpx_ops(build_dir: str) -> dict:
    """Bundle the ops package into the layer as ``lpx_ops``."""
    so = ensure_ops_extension()
    dest = os.path.join(build_dir, "lpx_ops")
    os.makedirs(dest, exist_ok=True)
    for fn in ("__init__.py", "api.py"):
        shutil.copy2(os.path.join(OPS_DIR, fn), os.path.join(dest, fn))
    shutil.copy2(so, os.path.join(dest, os.path.basename(so)))
    return {
        "so": os.path.basename(so),
        "so_mb": round(os.path.getsize(so) / 2**20, 2),
    }


def build_recipe_artifacts(recipe: BuildRecipe, dep: ResolvedDep, build_dir: str) -> dict:
    """The wheel_builder hook for assemble_layer (native recipes only)."""
    info = _install_lpx_ops(build_dir)
    if recipe.shim:
        shim_src = os.path.join(SHIMS_DIR, recipe.shim + ".py")
        if not os.path.exists(shim_src):
            raise FileNotFoundError(f"recipe {recipe.name}: shim {recipe.shim} not found")
        shutil.copy2(shim_src, os.path.join(build_dir, recipe.shim + ".py"))
        info["shim"] = recipe.shim
    info["kernels"] = recipe.kernels
    return info
