"""Structured stdout logging (stdlib only — SURVEY.md §5 observability row)."""

from __future__ import annotations

import logging
import os
import sys

_FMT = "[lpx %(name)s] %(levelname).1s %(message)s"
_configured = False


def _configure() -> None:
    global _configured
    if _configured:
        return
    h = logging.StreamHandler(sys.stderr)
    h.setFormatter(logging.Formatter(_FMT))
    root = logging.getLogger("lpx")
    root.addHandler(h)
    root.setLevel(os.environ.get("LPX_LOG", "INFO").upper())
    root.propagate = False
    _configured = True


def get_logger(name: str) -> logging.Logger:
    _configure()
    return logging.getLogger(f"lpx.{name}" if not name.startswith("lpx") else name)
