"""Layer assembly: vendored deps + recipe artifacts -> ./build/ -> layer.zip.

Reference analog: the output assembler — copies prebuilt /
built / pip-installed trees into ./build/, dedupes packages already provided
by another artifact (provides/excludes), then the user zips. Here the zip is
part of the build and sizes are reported against the Lambda-style limits.
"""

from __future__ import annotations

import json
import os
import shutil
import sys
import zipfile
from dataclasses import dataclass, field
from importlib import metadata as importlib_metadata
from typing import Dict, List, Optional, Set

from ..recipes import BuildRecipe, get_registry
from ..resolver import Resolution, ResolvedDep
from ..utils.log import get_logger
from .prune import prune_tree
from .strip import strip_native_libs
from .zipper import ZipReport, make_zip, zip_report

log = get_logger("packager")


@dataclass
class BuildReport:
    build_dir: str
    zip_path: Optional[str]
    deps: List[dict] = field(default_factory=list)
    pruned_entries: int = 0
    pruned_bytes: int = 0
    stripped_files: int = 0
    stripped_bytes: int = 0
    sizes: Optional[ZipReport] = None

    def to_dict(self) -> dict:
        d = {
            "build_dir": self.build_dir,
            "zip_path": self.zip_path,
            "deps": self.deps,
            "pruned_entries": self.pruned_entries,
            "pruned_mb": round(self.pruned_bytes / 2**20, 2),
            "stripped_files": self.stripped_files,
            "stripped_mb": round(self.stripped_bytes / 2**20, 2),
        }
        if self.sizes:
            d["unzipped_mb"] = round(self.sizes.unzipped_mb, 2)
            d["zipped_mb"] = None if self.sizes.zipped_mb is None else round(self.sizes.zipped_mb, 2)
            d["within_250mb_unzipped"] = self.sizes.within_unzipped_limit
            d["within_50mb_zipped"] = self.sizes.within_zipped_limit
        return d


def _vendor_wheel(wheel_path: str, dest: str) -> None:
    """Unpack a wheel (zip of site-packages layout) into dest, skipping
    .data scripts (a Lambda-style layer has no console entry points)."""
    with zipfile.ZipFile(wheel_path) as zf:
        for info in zf.infolist():
            name = info.filename
            if name.endswith("/"):
                continue
            parts = name.split("/")
            if parts[0].endswith(".data"):
                # purelib/platlib payloads still belong in the root
                if len(parts) > 2 and parts[1] in ("purelib", "platlib"):
                    parts = parts[2:]
                else:
                    continue  # scripts/, data/: skipped
            out = os.path.join(dest, *parts)
            os.makedirs(os.path.dirname(out), exist_ok=True)
            with zf.open(info) as src, open(out, "wb") as dst:
                shutil.copyfileobj(src, dst)
            mode = (info.external_attr >> 16) & 0xFFFF
            if mode:
                os.chmod(out, mode & 0o777 or 0o644)


def _vendor_installed(name: str, dest: str) -> None:
    """Copy an installed distribution's files from site-packages into dest."""
    dist = importlib_metadata.distribution(name)
    files = dist.files or []
    copied_roots: Set[str] = set()
    for f in files:
        rel = str(f)
        if rel.startswith(".."):
            continue  # scripts outside site-packages
        if rel.endswith((".pyc",)) or "__pycache__" in rel:
            continue
        src = str(dist.locate_file(f))
        if not os.path.isfile(src):
            continue
        out = os.path.join(dest, rel)
        os.makedirs(os.path.dirname(out), exist_ok=True)
        shutil.copy2(src, out)
        copied_roots.add(rel.split("/", 1)[0])
    if not copied_roots:
        raise RuntimeError(f"installed distribution {name} has no copyable files")


def vendor_dep(dep: ResolvedDep, dest: str) -> None:
    if dep.wheel_path:
        _vendor_wheel(dep.wheel_path, dest)
    else:
        _vendor_installed(dep.name, dest)


def assemble_layer(
    resolution: Resolution,
    build_dir: str,
    *,
    zip_path: Optional[str] = None,
    wheel_builder=None,
    handler_files: Optional[Dict[str, str]] = None,
    clean: bool = True,
) -> BuildReport:
    """Assemble the layer from a resolution.

    wheel_builder: callable(recipe, dep, build_dir) -> dict — installs the
    recipe's ROCm wheel artifacts into build_dir (ops .so + shim); supplied by
    lambdipy_amd.builder (kept injectable so the packager is unit-testable
    without hipcc).
    handler_files: mapping of archive-relative path -> local file to include
    (the inference handler entry point).
    """
    registry = get_registry()
    if clean and os.path.isdir(build_dir):
        shutil.rmtree(build_dir)
    os.makedirs(build_dir, exist_ok=True)

    report = BuildReport(build_dir=build_dir, zip_path=zip_path)

    # provides/excludes dedupe: skip a plain dep that a recipe artifact provides
    provided: Set[str] = set()
    excluded: Set[str] = set()
    for d in resolution.recipe_hits:
        r = registry.get(d.recipe)
        if r:
            provided.update(r.provides)
            excluded.update(r.excludes)

    for dep in resolution.deps:
        recipe: Optional[BuildRecipe] = registry.get(dep.recipe) if dep.recipe else None
        if recipe is None and dep.name in (provided | excluded):
            log.info("skip %s==%s (provided by a recipe artifact)", dep.name, dep.version)
            report.deps.append(
                {"name": dep.name, "version": dep.version, "action": "deduped"}
            )
            continue
        if recipe is not None and not recipe.vendor_upstream:
            # runtime-provided upstream (e.g. torch): ship only the recipe's
            # ROCm artifacts; record the runtime requirement in the manifest
            log.info("runtime-provided %s==%s (recipe %s ships shim+kernels only)",
                     dep.name, dep.version, recipe.name)
            report.deps.append(
                {"name": dep.name, "version": dep.version, "source": "runtime",
                 "action": "runtime-provided",
                 "runtime_requires": recipe.runtime_requires}
            )
            if wheel_builder is not None and recipe.kind == "native":
                info = wheel_builder(recipe, dep, build_dir)
                if info:
                    report.deps[-1]["rocm_wheel"] = info
            continue
        log.info(
            "vendor %s==%s [%s%s]",
            dep.name,
            dep.version,
            dep.source,
            f", recipe={dep.recipe}" if dep.recipe else "",
        )
        vendor_dep(dep, build_dir)
        action = "recipe" if recipe else "vendored"
        report.deps.append(
            {"name": dep.name, "version": dep.version, "source": dep.source, "action": action}
        )
        if recipe:
            n, b = prune_tree(build_dir, recipe.prune)
            report.pruned_entries += n
            report.pruned_bytes += b
            if wheel_builder is not None and recipe.kind == "native":
                info = wheel_builder(recipe, dep, build_dir)
                if info:
                    report.deps[-1]["rocm_wheel"] = info

    # global hygiene prune
    n, b = prune_tree(build_dir, ["**/__pycache__/**", "**/*.pyc"])
    report.pruned_entries += n
    report.pruned_bytes += b

    # strip native libs wherever any recipe asked for it
    if any(registry.get(d.recipe).strip_native for d in resolution.recipe_hits if registry.get(d.recipe)):
        sn, sb = strip_native_libs(build_dir)
        report.stripped_files += sn
        report.stripped_bytes += sb

    # stub out never-called native dependencies (e.g. libmagma in a torch
    # layer: 1.3 GB -> 4 MB; the loader only needs the symbol set)
    stub_names = set()
    for d in resolution.recipe_hits:
        r = registry.get(d.recipe)
        if r:
            stub_names.update(r.stub_native)
    if stub_names:
        from .stubs import DEFAULT_BENIGN, make_stub

        for root, _dirs, files in os.walk(build_dir):
            for fn in files:
                if fn in stub_names:
                    p = os.path.join(root, fn)
                    try:
                        st = make_stub(p, p, benign=DEFAULT_BENIGN.get(fn, ()))
                    except Exception as exc:  # noqa: BLE001
                        log.warning("stub skipped %s: %s", p, exc)
                        continue
                    report.stripped_files += 1
                    report.stripped_bytes += st["old_size"] - st["new_size"]

    # strip non-gfx950 fat-binary entries (HIP multi-arch DSOs)
    if any(registry.get(d.recipe).strip_archs for d in resolution.recipe_hits if registry.get(d.recipe)):
        from .fatbin import strip_fatbin

        for root, _dirs, files in os.walk(build_dir):
            for fn in files:
                if ".so" not in fn:
                    continue
                p = os.path.join(root, fn)
                try:
                    st = strip_fatbin(p)
                except Exception as exc:  # noqa: BLE001 - leave lib intact
                    log.warning("strip-archs skipped %s: %s", p, exc)
                    continue
                if st["new_size"] < st["old_size"]:
                    report.stripped_files += 1
                    report.stripped_bytes += st["old_size"] - st["new_size"]

    if handler_files:
        for rel, local in handler_files.items():
            out = os.path.join(build_dir, rel)
            os.makedirs(os.path.dirname(out), exist_ok=True)
            shutil.copy2(local, out)

    manifest = {
        "format": 1,
        "tool": "lambdipy_amd",
        "python": f"{sys.version_info.major}.{sys.version_info.minor}",
        "gpu_arch": "gfx950",
        "deps": report.deps,
    }
    with open(os.path.join(build_dir, "lpx-manifest.json"), "w") as f:
        json.dump(manifest, f, indent=1, sort_keys=True)

    # lockfile: the fully pinned set this layer was built from (reference
    # analog: the pinned dep list the resolver produced — SURVEY.md §2.1 R2)
    lock = {
        "lock_format": 1,
        "deps": [
            {"name": d.name, "version": d.version, "source": d.source,
             "recipe": d.recipe}
            for d in resolution.deps
        ],
    }
    with open(os.path.join(build_dir, "lpx-lock.json"), "w") as f:
        json.dump(lock, f, indent=1, sort_keys=True)

    if zip_path:
        make_zip(build_dir, zip_path)
    report.sizes = zip_report(build_dir, zip_path)
    return report
