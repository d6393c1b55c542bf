"""Synthetic job-trace generators.

Two families, mirroring BASELINE.json's configs:

* ``philly_like`` — heavy-tailed GPU demands and log-normal durations with
  the qualitative shape of the published Microsoft Philly trace (mostly
  1-GPU jobs, a heavy multi-GPU tail, minutes-to-days durations). The real
  trace is not redistributable/fetchable in this offline image, so this
  generator stands in for it; the bench reports ``data: synthetic``.
* ``poisson`` — memoryless arrivals with uniform small demands (config 5).

A trace is a struct-of-arrays sorted by arrival time; all times are int64
microseconds (docs/ENV_SPEC.md).
"""
from __future__ import annotations

from dataclasses import dataclass

import numpy as np

_US = 1_000_000


@dataclass
class JobTrace:
    arrival_us: np.ndarray  # int64, sorted ascending
    gpus: np.ndarray        # int32 (max demand)
    work_us: np.ndarray     # int64 (service time at max demand)
    min_gpus: np.ndarray | None = None  # int32; None -> == gpus (rigid)

    def __post_init__(self):
        if self.min_gpus is None:
            self.min_gpus = self.gpus.copy()

    def __len__(self) -> int:
        return len(self.arrival_us)

    def validate(self, max_job_gpus: int, lo: int = 0, hi: int | None = None) -> None:
        """Validate window [lo, hi) — a pool is only sorted per-window."""
        hi = len(self) if hi is None else hi
        a, g, w = self.arrival_us[lo:hi], self.gpus[lo:hi], self.work_us[lo:hi]
        assert self.arrival_us.dtype == np.int64
        assert self.gpus.dtype == np.int32
        assert self.work_us.dtype == np.int64
        assert (np.diff(a) >= 0).all()
        assert (g >= 1).all() and (g <= max_job_gpus).all()
        assert (w >= 1).all()


_PHILLY_SIZES = np.array([1, 2, 4, 8, 16, 32, 64], dtype=np.int32)
_PHILLY_PROBS = np.array([0.68, 0.10, 0.10, 0.07, 0.03, 0.015, 0.005])


def philly_like(
    num_jobs: int,
    seed: int,
    max_job_gpus: int,
    mean_interarrival_s: float = 60.0,
    median_duration_s: float = 600.0,
    sigma: float = 1.4,
    fail_frac: float = 0.0,
    elastic: bool = False,
) -> JobTrace:
    """fail_frac: fraction of jobs that FAIL early (the published Philly
    trace contains failed jobs): their realized service time is cut to
    1-20%% of the sampled duration. elastic: multi-GPU jobs accept down to
    a quarter of their max demand (ENV_SPEC v3)."""
    rng = np.random.default_rng(seed)
    inter = rng.exponential(mean_interarrival_s, size=num_jobs)
    arrival = np.cumsum(inter)
    arrival[0] = 0.0
    mask = _PHILLY_SIZES <= max_job_gpus
    probs = _PHILLY_PROBS[mask] / _PHILLY_PROBS[mask].sum()
    gpus = rng.choice(_PHILLY_SIZES[mask], size=num_jobs, p=probs)
    dur = rng.lognormal(mean=np.log(median_duration_s), sigma=sigma, size=num_jobs)
    dur = np.clip(dur, 30.0, 7 * 24 * 3600.0)
    if fail_frac > 0:
        failed = rng.random(num_jobs) < fail_frac
        cut = rng.uniform(0.01, 0.2, size=num_jobs)
        dur = np.where(failed, np.maximum(dur * cut, 1.0), dur)
    ming = np.maximum(1, gpus // 4).astype(np.int32) if elastic \
        else gpus.astype(np.int32)
    return JobTrace(
        arrival_us=(arrival * _US).astype(np.int64),
        gpus=gpus.astype(np.int32),
        work_us=(dur * _US).astype(np.int64),
        min_gpus=ming,
    )


def poisson(
    num_jobs: int,
    seed: int,
    max_job_gpus: int,
    rate_per_s: float = 0.05,
    mean_duration_s: float = 300.0,
    max_gpus: int = 8,
) -> JobTrace:
    rng = np.random.default_rng(seed)
    inter = rng.exponential(1.0 / rate_per_s, size=num_jobs)
    arrival = np.cumsum(inter)
    arrival[0] = 0.0
    hi = min(max_gpus, max_job_gpus)
    gpus = rng.integers(1, hi + 1, size=num_jobs)
    dur = rng.exponential(mean_duration_s, size=num_jobs)
    dur = np.clip(dur, 1.0, 24 * 3600.0)
    return JobTrace(
        arrival_us=(arrival * _US).astype(np.int64),
        gpus=gpus.astype(np.int32),
        work_us=(dur * _US).astype(np.int64),
    )


def make_pool(kind: str, num_envs: int, jobs_per_env: int, seed: int,
              max_job_gpus: int, **kw) -> tuple[JobTrace, np.ndarray, np.ndarray]:
    """A shared trace pool: per-env windows [job_lo, job_hi) of one big SoA.

    Each env's window is generated with its own sub-seed so envs decorrelate,
    then concatenated (each window is itself arrival-sorted; windows are
    independent, arrivals are re-based per window by the env).
    Returns (pool, job_lo[num_envs], job_hi[num_envs]).

    kind="file": replay a REAL trace file (CSV schema in traces/philly.py)
    — kw must carry path=...; optional oversize="clip"|"drop".
    """
    if kind == "file":
        from .philly import load_trace_csv, windows_from_trace
        path = kw.pop("path")
        tr = load_trace_csv(path, max_job_gpus, **kw)
        return windows_from_trace(tr, num_envs, jobs_per_env, seed)
    gen = philly_like if kind == "philly" else poisson
    arrs, gps, wks, mgs, lo, hi = [], [], [], [], [], []
    off = 0
    for e in range(num_envs):
        t = gen(jobs_per_env, seed * 1_000_003 + e, max_job_gpus, **kw)
        arrs.append(t.arrival_us)
        gps.append(t.gpus)
        wks.append(t.work_us)
        mgs.append(t.min_gpus)
        lo.append(off)
        off += len(t)
        hi.append(off)
    pool = JobTrace(np.concatenate(arrs), np.concatenate(gps),
                    np.concatenate(wks), np.concatenate(mgs))
    return pool, np.asarray(lo, np.int64), np.asarray(hi, np.int64)
