"""Zip the assembled layer and report sizes against the Lambda-style limits.

The limits the reference engineers against: 250 MB unzipped,
50 MB zipped. We report both and flag violations; BASELINE.json config 5 makes
<250 MB unzipped a hard target for the torch layer.
"""

from __future__ import annotations

import os
import zipfile
from dataclasses import dataclass
from typing import Optional

LIMIT_UNZIPPED = 250 * 1024 * 1024
LIMIT_ZIPPED = 50 * 1024 * 1024


def dir_size_bytes(root: str) -> int:
    total = 0
    for dirpath, _d, filenames in os.walk(root):
        for fn in filenames:
            try:
                total += os.path.getsize(os.path.join(dirpath, fn))
            except OSError:
                pass
    return total


def make_zip(root: str, out_path: str, *, compresslevel: int = 9) -> int:
    """Deterministic zip of root (sorted entries, fixed timestamps).

    Returns the zipped size in bytes.
    """
    entries = []
    for dirpath, dirnames, filenames in os.walk(root):
        dirnames.sort()
        for fn in sorted(filenames):
            full = os.path.join(dirpath, fn)
            entries.append((os.path.relpath(full, root), full))
    entries.sort()
    os.makedirs(os.path.dirname(os.path.abspath(out_path)), exist_ok=True)
    with zipfile.ZipFile(out_path, "w", zipfile.ZIP_DEFLATED, compresslevel=compresslevel) as zf:
        for rel, full in entries:
            zi = zipfile.ZipInfo(rel.replace(os.sep, "/"), date_time=(1980, 1, 1, 0, 0, 0))
            zi.external_attr = (os.stat(full).st_mode & 0xFFFF) << 16
            zi.compress_type = zipfile.ZIP_DEFLATED
            with open(full, "rb") as f:
                zf.writestr(zi, f.read(), compresslevel=compresslevel)
    return os.path.getsize(out_path)


@dataclass
class ZipReport:
    unzipped_bytes: int
    zipped_bytes: Optional[int]

    @property
    def unzipped_mb(self) -> float:
        return self.unzipped_bytes / (1024 * 1024)

    @property
    def zipped_mb(self) -> Optional[float]:
        return None if self.zipped_bytes is None else self.zipped_bytes / (1024 * 1024)

    @property
    def within_unzipped_limit(self) -> bool:
        return self.unzipped_bytes <= LIMIT_UNZIPPED

    @property
    def within_zipped_limit(self) -> Optional[bool]:
        return None if self.zipped_bytes is None else self.zipped_bytes <= LIMIT_ZIPPED


def zip_report(root: str, zip_path: Optional[str] = None) -> ZipReport:
    return ZipReport(
        unzipped_bytes=dir_size_bytes(root),
        zipped_bytes=os.path.getsize(zip_path) if zip_path and os.path.exists(zip_path) else None,
    )
