"""Training metrics: JSONL logger + throughput/JCT tracking + NaN guard."""
from __future__ import annotations

import json
import math
import time
from pathlib import Path


class MetricsLogger:
    """Appends one JSON object per iteration to <dir>/metrics.jsonl and
    keeps rolling throughput; optional stdout echo."""

    def __init__(self, out_dir: str | Path | None = None, echo: bool = True,
                 env_steps_per_iter: int = 0):
        self.path = None
        if out_dir is not None:
            out = Path(out_dir)
            out.mkdir(parents=True, exist_ok=True)
            self.path = out / "metrics.jsonl"
            self._fh = open(self.path, "a")
        self.echo = echo
        self.env_steps_per_iter = env_steps_per_iter
        self._t_last = time.perf_counter()
        self.iteration = 0

    def log(self, stats: dict, iteration: int | None = None) -> dict:
        now = time.perf_counter()
        dt = now - self._t_last
        self._t_last = now
        it = self.iteration if iteration is None else iteration
        rec = {"iter": it, "wall_s": round(dt, 5)}
        if self.env_steps_per_iter:
            rec["env_steps_per_sec"] = round(self.env_steps_per_iter / dt, 1)
        rec.update({k: (round(v, 6) if isinstance(v, float) else v)
                    for k, v in stats.items() if not isinstance(v, dict)})
        if "episode_stats" in stats:
            rec["episode_stats"] = stats["episode_stats"]
        if self.path is not None:
            self._fh.write(json.dumps(rec) + "\n")
            self._fh.flush()
        if self.echo:
            print(" ".join(f"{k}={v}" for k, v in rec.items()
                           if not isinstance(v, dict)))
        self.iteration = it + 1
        return rec

    def close(self):
        if self.path is not None:
            self._fh.close()


def check_finite(stats: dict, where: str = "") -> None:
    """Failure detection: raise on NaN/Inf in training stats (a diverged
    policy silently corrupts every later iteration)."""
    for k, v in stats.items():
        if isinstance(v, float) and not math.isfinite(v):
            raise FloatingPointError(
                f"non-finite metric {k}={v} {('at ' + where) if where else ''}"
                " — training diverged (lower lr / raise ent_coef?)")
