"""Minimal PEP-427 wheel emission for recipe artifacts.

The reference's recipes produce prebuilt artifacts consumed on later builds
(SURVEY.md §2.1 R4/R5); ours emit real ROCm wheels: ``lpx_ops`` (the gfx950
kernel library: loader + AOT .so) and one shim wheel per native recipe
(lpx_numpy, lpx_scipy, lpx_sklearn, lpx_torch). Wheels are pip-installable
(`pip install --no-index lpx_ops-....whl`) and content-addressed in the local
wheel cache so unchanged kernels are never rebuilt or re-packed
(SURVEY.md §5 checkpoint/artifact-cache row).
"""

from __future__ import annotations

import base64
import hashlib
import os
import sys
import zipfile
from typing import Dict, List, Optional, Tuple

WHEEL_CACHE = os.environ.get("LPX_WHEEL_CACHE", os.path.expanduser("~/.lpx/wheels"))
VERSION = "0.1.0"


def _py_tag() -> str:
    return f"cp{sys.version_info.major}{sys.version_info.minor}"


def _record_hash(data: bytes) -> str:
    h = hashlib.sha256(data).digest()
    return "sha256=" + base64.urlsafe_b64encode(h).rstrip(b"=").decode()


def _content_key(files: Dict[str, bytes]) -> str:
    h = hashlib.sha256()
    for name in sorted(files):
        h.update(name.encode())
        h.update(files[name])
    return h.hexdigest()[:12]


def write_wheel(
    dist_dir: str,
    name: str,
    files: Dict[str, bytes],
    *,
    version: str = VERSION,
    platform_specific: bool = False,
    requires: Optional[List[str]] = None,
    summary: str = "",
) -> str:
    """Write a wheel named {name}-{version}+gfx950.{hash} and return its path.

    files: archive-relative path -> content (package payload, no dist-info).
    """
    key = _content_key(files)
    local = f"{version}+gfx950.{key}"
    if platform_specific:
        tag = f"{_py_tag()}-{_py_tag()}-linux_x86_64"
    else:
        tag = "py3-none-any"
    fname = f"{name}-{local}-{tag}.whl"
    os.makedirs(dist_dir, exist_ok=True)
    out = os.path.join(dist_dir, fname)
    if os.path.exists(out):
        return out  # content-addressed: same key => same wheel

    di = f"{name}-{local}.dist-info"
    metadata = (
        "Metadata-Version: 2.1\n"
        f"Name: {name}\n"
        f"Version: {local}\n"
        f"Summary: {summary or name + ' (lambdipy_amd ROCm artifact, gfx950)'}\n"
        "License: MIT\n"
        + "".join(f"Requires-Dist: {r}\n" for r in (requires or []))
    ).encode()
    wheel_meta = (
        "Wheel-Version: 1.0\n"
        "Generator: lambdipy_amd\n"
        f"Root-Is-Purelib: {'false' if platform_specific else 'true'}\n"
        f"Tag: {tag}\n"
    ).encode()

    entries: List[Tuple[str, bytes]] = sorted(files.items())
    entries.append((f"{di}/METADATA", metadata))
    entries.append((f"{di}/WHEEL", wheel_meta))
    record_rows = [
        f"{p},{_record_hash(c)},{len(c)}" for p, c in entries
    ] + [f"{di}/RECORD,,"]
    entries.append((f"{di}/RECORD", ("\n".join(record_rows) + "\n").encode()))

    tmp = out + ".tmp"
    with zipfile.ZipFile(tmp, "w", zipfile.ZIP_DEFLATED, compresslevel=6) as zf:
        for path, content in entries:
            zi = zipfile.ZipInfo(path, date_time=(1980, 1, 1, 0, 0, 0))
            zi.external_attr = (0o755 if path.endswith(".so") else 0o644) << 16
            zf.writestr(zi, content)
    os.replace(tmp, out)
    return out
