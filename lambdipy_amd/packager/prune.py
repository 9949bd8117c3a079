"""Prune a package tree per recipe glob patterns.

Reference analog: the docker builder's prune step (tests/, docs, *.pyc —
SURVEY.md §2.1 R5). Patterns are ``fnmatch``-style globs evaluated against
paths relative to the tree root; ``**`` matches across separators.
"""

from __future__ import annotations

import os
import re
import shutil
from typing import Iterable, Tuple


def _compile(pattern: str) -> "re.Pattern[str]":
    """Translate a glob with ** into a regex over '/'-separated rel paths."""
    out = []
    i = 0
    while i < len(pattern):
        c = pattern[i]
        if c == "*":
            if pattern[i : i + 3] == "**/":
                out.append("(?:.*/)?")
                i += 3
                continue
            if pattern[i : i + 2] == "**":
                out.append(".*")
                i += 2
                continue
            out.append("[^/]*")
        elif c == "?":
            out.append("[^/]")
        else:
            out.append(re.escape(c))
        i += 1
    return re.compile("^" + "".join(out) + "$")


def prune_tree(root: str, patterns: Iterable[str]) -> Tuple[int, int]:
    """Remove files/dirs matching any pattern. Returns (entries_removed, bytes_freed)."""
    regs = [_compile(p.rstrip("/")) for p in patterns]
    if not regs:
        return (0, 0)
    removed = 0
    freed = 0

    def matches(rel: str) -> bool:
        return any(r.match(rel) for r in regs)

    # walk bottom-up so directory removals see already-pruned children
    for dirpath, dirnames, filenames in os.walk(root, topdown=True):
        rel_dir = os.path.relpath(dirpath, root)
        rel_dir = "" if rel_dir == "." else rel_dir.replace(os.sep, "/")
        # prune whole directories first (stops the walk descending)
        keep = []
        for d in dirnames:
            rel = f"{rel_dir}/{d}" if rel_dir else d
            if matches(rel) or matches(rel + "/**"):
                full = os.path.join(dirpath, d)
                for dp, _, fns in os.walk(full):
                    for fn in fns:
                        try:
                            freed += os.path.getsize(os.path.join(dp, fn))
                            removed += 1
                        except OSError:
                            pass
                shutil.rmtree(full, ignore_errors=True)
            else:
                keep.append(d)
        dirnames[:] = keep
        for fn in filenames:
            rel = f"{rel_dir}/{fn}" if rel_dir else fn
            if matches(rel):
                full = os.path.join(dirpath, fn)
                try:
                    freed += os.path.getsize(full)
                    os.unlink(full)
                    removed += 1
                except OSError:
                    pass
    return (removed, freed)
