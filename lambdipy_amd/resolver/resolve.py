"""Offline dependency resolution.

The reference delegates pin resolution to pip/pipenv against PyPI
(SURVEY.md §2.1 R2). This environment has no index (SURVEY.md §7 "Offline
builds"), so we resolve against the two local sources of truth:

1. the offline wheelhouse (``/opt/wheelhouse`` by default, overridable with
   ``LPX_WHEELHOUSE``) — wheel filenames give (name, version), wheel METADATA
   gives Requires-Dist for the transitive closure;
2. the running interpreter's installed distributions (importlib.metadata) —
   used when no wheel exists (e.g. numpy/scipy/torch ship in the image).

The output is a fully pinned set split into recipe-matched deps (built as
ROCm wheels by the wheel builder) and plain deps (vendored as-is), mirroring
the reference's recipe-hits vs pip-installed split.
"""

from __future__ import annotations

import os
import re
import zipfile
from dataclasses import dataclass, field
from email.parser import Parser
from importlib import metadata as importlib_metadata
from typing import Dict, List, Optional, Set, Tuple

from packaging.markers import default_environment
from packaging.requirements import Requirement as _PkgRequirement
from packaging.specifiers import SpecifierSet
from packaging.utils import canonicalize_name, parse_wheel_filename
from packaging.version import InvalidVersion, Version

from .requirements import Requirement

DEFAULT_WHEELHOUSE = os.environ.get("LPX_WHEELHOUSE", "/opt/wheelhouse")


@dataclass
class Candidate:
    name: str
    version: Version
    source: str  # "wheelhouse" | "installed"
    wheel_path: Optional[str] = None  # set for wheelhouse candidates


@dataclass
class ResolvedDep:
    name: str
    version: str
    source: str  # "wheelhouse" | "installed"
    wheel_path: Optional[str]
    recipe: Optional[str] = None  # recipe name when recipe-matched
    requested_by: Tuple[str, ...] = ()


@dataclass
class Resolution:
    deps: List[ResolvedDep]
    recipe_hits: List[ResolvedDep] = field(default_factory=list)
    plain: List[ResolvedDep] = field(default_factory=list)

    def get(self, name: str) -> Optional[ResolvedDep]:
        cn = canonicalize_name(name)
        for d in self.deps:
            if d.name == cn:
                return d
        return None


class ResolutionError(RuntimeError):
    pass


def _wheelhouse_index(wheelhouse: str) -> Dict[str, List[Candidate]]:
    idx: Dict[str, List[Candidate]] = {}
    if not os.path.isdir(wheelhouse):
        return idx
    for fn in os.listdir(wheelhouse):
        if not fn.endswith(".whl"):
            continue
        try:
            name, ver, _build, _tags = parse_wheel_filename(fn)
        except Exception:
            continue
        idx.setdefault(canonicalize_name(name), []).append(
            Candidate(canonicalize_name(name), ver, "wheelhouse", os.path.join(wheelhouse, fn))
        )
    for v in idx.values():
        v.sort(key=lambda c: c.version, reverse=True)
    return idx


def _installed_version(name: str) -> Optional[Version]:
    try:
        return Version(importlib_metadata.version(name))
    except (importlib_metadata.PackageNotFoundError, InvalidVersion):
        return None


def _wheel_requires(wheel_path: str) -> List[str]:
    """Requires-Dist lines from a wheel's METADATA."""
    try:
        with zipfile.ZipFile(wheel_path) as zf:
            meta_name = next(
                (n for n in zf.namelist() if n.endswith(".dist-info/METADATA") and n.count("/") == 1),
                None,
            )
            if meta_name is None:
                return []
            msg = Parser().parsestr(zf.read(meta_name).decode("utf-8", "replace"))
            return msg.get_all("Requires-Dist") or []
    except Exception:
        return []


def _installed_requires(name: str) -> List[str]:
    try:
        return importlib_metadata.requires(name) or []
    except importlib_metadata.PackageNotFoundError:
        return []


def _marker_env(python_version: Optional[str] = None) -> dict:
    env = dict(default_environment())
    if python_version:
        env["python_version"] = python_version
        env["python_full_version"] = python_version + ".0"
    return env


def resolve(
    requirements: List[Requirement],
    *,
    wheelhouse: str = DEFAULT_WHEELHOUSE,
    recipe_registry=None,
    follow_transitive: bool = True,
    extras_env: Optional[dict] = None,
) -> Resolution:
    """Resolve requirements to a pinned set against offline sources.

    Version selection: newest candidate satisfying the intersected specifier
    (wheelhouse preferred; installed distribution used as fallback).
    Transitive deps come from wheel METADATA / installed metadata, with
    environment markers evaluated for this interpreter.
    """
    from ..recipes import get_registry

    registry = recipe_registry if recipe_registry is not None else get_registry()
    idx = _wheelhouse_index(wheelhouse)
    env = extras_env or _marker_env()

    pinned: Dict[str, ResolvedDep] = {}
    specs: Dict[str, SpecifierSet] = {}
    parents: Dict[str, Set[str]] = {}
    queue: List[Tuple[Requirement, Optional[str]]] = [(r, None) for r in requirements]
    seen_edges: Set[Tuple[str, Optional[str]]] = set()

    while queue:
        req, parent = queue.pop(0)
        if (req.name, parent) in seen_edges:
            continue
        seen_edges.add((req.name, parent))
        if req.marker:
            try:
                from packaging.markers import Marker

                if not Marker(req.marker).evaluate(env):
                    continue
            except Exception:
                pass
        name = req.name
        specs[name] = specs.get(name, SpecifierSet()) & SpecifierSet(req.specifier)
        if parent:
            parents.setdefault(name, set()).add(parent)

        spec = specs[name]
        chosen: Optional[Candidate] = None
        for c in idx.get(name, []):
            if spec.contains(c.version, prereleases=True):
                chosen = c
                break
        if chosen is None:
            iv = _installed_version(name)
            if iv is not None and spec.contains(iv, prereleases=True):
                chosen = Candidate(name, iv, "installed")
        if chosen is None:
            hint = ""
            if idx.get(name):
                hint = f" (wheelhouse has {', '.join(str(c.version) for c in idx[name][:4])})"
            elif _installed_version(name) is not None:
                hint = f" (installed: {_installed_version(name)})"
            raise ResolutionError(
                f"cannot satisfy {name}{spec or ' (any)'} from offline sources{hint}"
            )

        prev = pinned.get(name)
        if prev is not None and prev.version == str(chosen.version):
            continue
        pinned[name] = ResolvedDep(
            name=name,
            version=str(chosen.version),
            source=chosen.source,
            wheel_path=chosen.wheel_path,
            recipe=(registry.match(name, str(chosen.version)).name
                    if registry and registry.match(name, str(chosen.version)) else None),
            requested_by=tuple(sorted(parents.get(name, set()))),
        )

        # A runtime-provided package (recipe vendor_upstream=false, e.g.
        # torch) brings its own runtime deps with it — do NOT vendor its
        # transitive closure (torch would otherwise drag 650 MB of
        # triton-rocm + sympy into the layer).
        runtime_provided = False
        if pinned[name].recipe and registry:
            rec = registry.get(pinned[name].recipe)
            runtime_provided = bool(rec) and not rec.vendor_upstream
        if follow_transitive and not runtime_provided:
            raw_reqs = (
                _wheel_requires(chosen.wheel_path)
                if chosen.wheel_path
                else _installed_requires(name)
            )
            for line in raw_reqs:
                try:
                    pr = _PkgRequirement(line)
                except Exception:
                    continue
                if pr.marker and not pr.marker.evaluate(env):
                    continue  # extras deps are skipped unless their marker holds
                queue.append(
                    (
                        Requirement(
                            name=canonicalize_name(pr.name),
                            raw=line,
                            specifier=str(pr.specifier),
                            marker=None,
                        ),
                        name,
                    )
                )

    deps = sorted(pinned.values(), key=lambda d: d.name)
    for d in deps:
        d.requested_by = tuple(sorted(parents.get(d.name, set())))
    res = Resolution(deps=deps)
    res.recipe_hits = [d for d in deps if d.recipe]
    res.plain = [d for d in deps if not d.recipe]
    return res
