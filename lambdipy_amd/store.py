"""Local artifact store — the offline analog of the reference's
GitHub-Releases prebuilt provider.

The reference keys prebuilt artifacts by (package, version, python-version)
and downloads + verifies per release asset; here assets live in a local
release directory, keyed by (name, version, python tag, gfx arch), each with
a sidecar JSON carrying the full sha256 that publish writes and fetch
verifies. Assets are either finished layer zips or per-recipe ROCm wheels.
"""

from __future__ import annotations

import hashlib
import json
import os
import re
import shutil
import sys
from dataclasses import dataclass
from typing import List, Optional

DEFAULT_RELEASE_DIR = os.environ.get(
    "LPX_RELEASES", os.path.expanduser("~/.lpx/releases")
)


def _sha256(path: str) -> str:
    h = hashlib.sha256()
    with open(path, "rb") as f:
        for chunk in iter(lambda: f.read(1 << 20), b""):
            h.update(chunk)
    return h.hexdigest()


def _py_tag() -> str:
    return f"cp{sys.version_info.major}{sys.version_info.minor}"


@dataclass
class Asset:
    path: str
    name: str
    version: Optional[str]
    python: Optional[str]
    arch: Optional[str]
    sha256: str
    kind: str  # "wheel" | "layer"

    def matches(self, name: str, version: Optional[str] = None,
                python: Optional[str] = None,
                arch: Optional[str] = None) -> bool:
        if self.name != name:
            return False
        if version and (self.version or "").split("+")[0] != version:
            return False
        # python/arch constraints only bind when the asset declares them
        # (pure-python layer zips carry neither)
        if python and self.python and self.python != python:
            return False
        if arch and self.arch and self.arch != arch:
            return False
        return True


_WHEEL_RE = re.compile(
    r"^(?P<name>[A-Za-z0-9_.]+)-(?P<version>[^-]+)-(?P<py>[a-z0-9]+)-"
    r"[a-z0-9]+-(?P<plat>[A-Za-z0-9_.]+)\.whl$"
)


def _wheel_key(filename: str) -> Optional[dict]:
    m = _WHEEL_RE.match(filename)
    if not m:
        return None
    version = m.group("version")
    arch = None
    if "+" in version:  # local tag carries the gfx arch (wheelgen naming)
        local = version.split("+", 1)[1]
        arch = local.split(".")[0]
    return {"name": m.group("name").replace("_", "-").lower(),
            "version": version,
            "python": m.group("py") if m.group("py") != "py3" else None,
            "arch": arch}


def publish_file(path: str, release_dir: str = DEFAULT_RELEASE_DIR, *,
                 name: Optional[str] = None, version: Optional[str] = None,
                 python: Optional[str] = None,
                 arch: Optional[str] = None) -> Asset:
    """Store one asset with a verification sidecar.

    Wheels are keyed from their PEP-427 filename automatically; explicit
    kwargs override. Layer zips are keyed by (name or filename)."""
    os.makedirs(release_dir, exist_ok=True)
    digest = _sha256(path)
    base = os.path.basename(path)
    kind = "wheel" if base.endswith(".whl") else "layer"
    key = _wheel_key(base) or {}
    meta = {
        "name": name or key.get("name") or base.rsplit(".", 1)[0],
        "version": version or key.get("version"),
        "python": python or key.get("python") or (
            _py_tag() if kind == "wheel" and key.get("python") else None),
        "arch": arch or key.get("arch"),
        "sha256": digest,
        "kind": kind,
        "bytes": os.path.getsize(path),
    }
    dest = os.path.join(release_dir, f"{base}.{digest[:16]}" +
                        ("" if base.endswith((".whl", ".zip")) else ".bin"))
    shutil.copy2(path, dest)
    with open(dest + ".json", "w") as f:
        json.dump(meta, f)
    return Asset(path=dest, sha256=digest, kind=kind,
                 name=meta["name"], version=meta["version"],
                 python=meta["python"], arch=meta["arch"])


def list_assets(release_dir: str = DEFAULT_RELEASE_DIR) -> List[Asset]:
    out: List[Asset] = []
    if not os.path.isdir(release_dir):
        return out
    for f in os.listdir(release_dir):
        if f.endswith(".json"):
            continue
        meta_path = os.path.join(release_dir, f + ".json")
        if not os.path.exists(meta_path):
            continue
        with open(meta_path) as mf:
            meta = json.load(mf)
        out.append(Asset(path=os.path.join(release_dir, f),
                         name=meta.get("name", f),
                         version=meta.get("version"),
                         python=meta.get("python"),
                         arch=meta.get("arch"),
                         sha256=meta.get("sha256", ""),
                         kind=meta.get("kind", "layer")))
    return out


class StoreError(RuntimeError):
    pass


def fetch_asset(name: str, dest: Optional[str] = None,
                release_dir: str = DEFAULT_RELEASE_DIR, *,
                version: Optional[str] = None, python: Optional[str] = None,
                arch: Optional[str] = None) -> str:
    """Find the best-matching asset, VERIFY its sha256, copy to dest.

    Best = newest mtime among key matches (versions are local-tagged
    content keys, not orderable). Raises StoreError on no match or on a
    hash mismatch (corrupt store)."""
    cands = [a for a in list_assets(release_dir)
             if a.matches(name, version, python, arch)]
    # fall back to prefix matching for layer zips published by filename
    if not cands:
        cands = [a for a in list_assets(release_dir)
                 if a.kind == "layer" and a.name.startswith(name)]
    if not cands:
        raise StoreError(
            f"no published artifact matches name={name!r} version={version!r} "
            f"python={python!r} arch={arch!r}")
    best = max(cands, key=lambda a: os.path.getmtime(a.path))
    actual = _sha256(best.path)
    if actual != best.sha256:
        raise StoreError(f"hash mismatch for {best.path}: store metadata says "
                         f"{best.sha256[:16]}, file is {actual[:16]} "
                         "(corrupt store)")
    base = os.path.basename(best.path)
    # strip the digest suffix we appended at publish
    base = re.sub(r"\.[0-9a-f]{16}(\.bin)?$", "", base) or base
    out = dest or base
    if os.path.isdir(out):
        out = os.path.join(out, base)
    shutil.copy2(best.path, out)
    return out
