"""Build-recipe registry.

Reproduces the reference's per-package recipe capability (SURVEY.md §2.1
R3/R7: one recipe per supported native package — supported version range,
build steps, prune/strip lists, provides/excludes). Our recipe format
additionally names the gfx950 HIP kernels to AOT-compile into the emitted
ROCm wheel and the shim module that exposes them.

Recipe JSON schema (``recipe_format`` 1)::

    {
      "recipe_format": 1,
      "name": "numpy",                  # canonical package name
      "versions": ">=1.24,<3",          # PEP 440 specifier of supported versions
      "kind": "native" | "pure",        # native => emits a ROCm wheel
      "kernels": ["gemm", "gemv"],      # HIP kernel groups compiled into the wheel
      "shim": "lpx_numpy",              # GPU shim module bundled with the wheel
      "provides": ["numpy"],            # deps satisfied by this artifact (dedupe)
      "excludes": [],                   # deps to drop when this recipe is present
      "prune": ["**/tests/**", ...],    # glob patterns removed by the packager
      "strip_native": true,             # run llvm-strip on bundled .so files
      "description": "..."
    }
"""

from __future__ import annotations

import glob
import json
import os
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from packaging.specifiers import SpecifierSet
from packaging.utils import canonicalize_name
from packaging.version import InvalidVersion, Version

RECIPE_FORMAT = 1
DATA_DIR = os.path.join(os.path.dirname(__file__), "data")

_REQUIRED_KEYS = {"recipe_format", "name", "versions", "kind"}
_ALLOWED_KINDS = {"native", "pure"}


class RecipeError(ValueError):
    pass


@dataclass
class BuildRecipe:
    """One per-package build recipe (reference analog: BuildRecipe,
    SURVEY.md §2.1 R3)."""

    name: str
    versions: str
    kind: str
    kernels: List[str] = field(default_factory=list)
    shim: Optional[str] = None
    provides: List[str] = field(default_factory=list)
    excludes: List[str] = field(default_factory=list)
    prune: List[str] = field(default_factory=list)
    strip_native: bool = True
    strip_archs: bool = False  # strip non-gfx950 HIP fat-binary entries
    stub_native: List[str] = field(default_factory=list)  # deps to stub out
    vendor_upstream: bool = True  # False: upstream pkg is runtime-provided
    runtime_requires: List[str] = field(default_factory=list)
    description: str = ""
    path: Optional[str] = None

    @classmethod
    def from_dict(cls, d: dict, path: Optional[str] = None) -> "BuildRecipe":
        missing = _REQUIRED_KEYS - set(d)
        if missing:
            raise RecipeError(f"recipe {path or d.get('name')}: missing keys {sorted(missing)}")
        if d["recipe_format"] != RECIPE_FORMAT:
            raise RecipeError(
                f"recipe {d['name']}: unsupported recipe_format {d['recipe_format']} "
                f"(this build supports {RECIPE_FORMAT})"
            )
        if d["kind"] not in _ALLOWED_KINDS:
            raise RecipeError(f"recipe {d['name']}: kind must be one of {sorted(_ALLOWED_KINDS)}")
        SpecifierSet(d["versions"])  # validate
        r = cls(
            name=canonicalize_name(d["name"]),
            versions=d["versions"],
            kind=d["kind"],
            kernels=list(d.get("kernels", [])),
            shim=d.get("shim"),
            provides=[canonicalize_name(p) for p in d.get("provides", [d["name"]])],
            excludes=[canonicalize_name(e) for e in d.get("excludes", [])],
            prune=list(d.get("prune", [])),
            strip_native=bool(d.get("strip_native", True)),
            strip_archs=bool(d.get("strip_archs", False)),
            stub_native=list(d.get("stub_native", [])),
            vendor_upstream=bool(d.get("vendor_upstream", True)),
            runtime_requires=list(d.get("runtime_requires", [])),
            description=d.get("description", ""),
            path=path,
        )
        # native + shim => kernels must be declared (the shim routes hot ops
        # to them); native WITHOUT a shim is legal — a strip/prune-only
        # recipe (e.g. pandas, whose numerics ride the numpy shim).
        if r.kind == "native" and r.shim and not r.kernels:
            raise RecipeError(
                f"recipe {r.name}: native recipes with a shim must list kernels")
        return r

    def supports(self, version: str) -> bool:
        try:
            return Version(version) in SpecifierSet(self.versions)
        except InvalidVersion:
            return False


class RecipeRegistry:
    def __init__(self, recipes: Dict[str, BuildRecipe]):
        self._recipes = recipes

    @classmethod
    def load(cls, data_dir: str = DATA_DIR) -> "RecipeRegistry":
        recipes: Dict[str, BuildRecipe] = {}
        for p in sorted(glob.glob(os.path.join(data_dir, "*.json"))):
            with open(p) as f:
                d = json.load(f)
            r = BuildRecipe.from_dict(d, path=p)
            recipes[r.name] = r
        return cls(recipes)

    def match(self, name: str, version: str) -> Optional[BuildRecipe]:
        cn = canonicalize_name(name)
        r = self._recipes.get(cn)
        if r is None:
            # provides aliases: e.g. a psycopg2-binary pin resolves to the
            # psycopg2 recipe (SURVEY.md §2.1 R3 provides/excludes info)
            for cand in self._recipes.values():
                if cn in (canonicalize_name(p) for p in cand.provides):
                    r = cand
                    break
        if r is not None and r.supports(version):
            return r
        return None

    def get(self, name: str) -> Optional[BuildRecipe]:
        return self._recipes.get(canonicalize_name(name))

    def names(self) -> List[str]:
        return sorted(self._recipes)

    def __len__(self) -> int:
        return len(self._recipes)


_registry: Optional[RecipeRegistry] = None


def get_registry() -> RecipeRegistry:
    global _registry
    if _registry is None:
        _registry = RecipeRegistry.load()
    return _registry
