"""Phase timing for the cold-start trace (SURVEY.md §5 tracing row:
per-phase cold-start timestamps — unzip / import / hipInit / moduleLoad /
comm-init / first-launch)."""

from __future__ import annotations

import json
import time
from contextlib import contextmanager
from typing import Dict, List, Optional


class Timer:
    """Monotonic phase timer; records named phase durations in ms."""

    def __init__(self) -> None:
        self.t0 = time.perf_counter()
        self.phases: List[Dict] = []

    @contextmanager
    def phase(self, name: str):
        start = time.perf_counter()
        try:
            yield
        finally:
            end = time.perf_counter()
            self.phases.append(
                {
                    "phase": name,
                    "start_ms": (start - self.t0) * 1e3,
                    "dur_ms": (end - start) * 1e3,
                }
            )

    def mark(self, name: str) -> None:
        now = time.perf_counter()
        self.phases.append({"phase": name, "start_ms": (now - self.t0) * 1e3, "dur_ms": 0.0})

    @property
    def total_ms(self) -> float:
        return (time.perf_counter() - self.t0) * 1e3

    def to_json(self) -> str:
        return json.dumps({"total_ms": self.total_ms, "phases": self.phases})


@contextmanager
def phase_trace(timer: Optional[Timer], name: str):
    if timer is None:
        yield
    else:
        with timer.phase(name):
            yield
