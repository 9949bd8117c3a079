"""Strip native shared objects to shrink the layer.

Reference analog: ``strip`` run on bundled .so files inside the docker build
(SURVEY.md §2.1 R5). We use ROCm's llvm-strip (works on host .so files and on
HIP fat binaries' host side); falls back to binutils strip if present.
"""

from __future__ import annotations

import os
import shutil
import subprocess
from typing import Tuple

_STRIP_CANDIDATES = [
    "/opt/rocm/lib/llvm/bin/llvm-strip",
    shutil.which("llvm-strip") or "",
    shutil.which("strip") or "",
]


def find_strip() -> str:
    for c in _STRIP_CANDIDATES:
        if c and os.path.exists(c):
            return c
    return ""


def strip_native_libs(root: str, *, keep_symbols: bool = False) -> Tuple[int, int]:
    """Run strip --strip-unneeded on every .so under root.

    Returns (files_stripped, bytes_saved). Silently skips files strip rejects.
    """
    tool = find_strip()
    if not tool:
        return (0, 0)
    n = 0
    saved = 0
    for dirpath, _dirnames, filenames in os.walk(root):
        for fn in filenames:
            if ".so" not in fn:
                continue
            full = os.path.join(dirpath, fn)
            try:
                before = os.path.getsize(full)
            except OSError:
                continue
            args = [tool, "--strip-debug" if keep_symbols else "--strip-unneeded", full]
            try:
                r = subprocess.run(args, capture_output=True, timeout=120)
            except (OSError, subprocess.TimeoutExpired):
                continue
            if r.returncode == 0:
                try:
                    after = os.path.getsize(full)
                except OSError:
                    continue
                if after < before:
                    n += 1
                    saved += before - after
    return (n, saved)
