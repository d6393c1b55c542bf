"""Host-side phase tracing -> chrome://tracing JSON (about:tracing).

Complements rocprofv3 (device kernels) with coarse host phases
(collect/update/sync). Near-zero overhead when disabled.
"""
from __future__ import annotations

import json
import time
from contextlib import contextmanager
from pathlib import Path


class Tracer:
    def __init__(self, enabled: bool = True):
        self.enabled = enabled
        self.events: list[dict] = []
        self._t0 = time.perf_counter()

    @contextmanager
    def span(self, name: str, **args):
        if not self.enabled:
            yield
            return
        ts = (time.perf_counter() - self._t0) * 1e6
        yield
        dur = (time.perf_counter() - self._t0) * 1e6 - ts
        self.events.append({"name": name, "ph": "X", "ts": ts, "dur": dur,
                            "pid": 0, "tid": 0, "args": args})

    def save(self, path: str | Path) -> None:
        Path(path).write_text(json.dumps(
            {"traceEvents": self.events, "displayTimeUnit": "ms"}))
