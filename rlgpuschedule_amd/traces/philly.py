"""Real job-trace ingestion (Philly-style CSV) behind the JobTrace
interface.

The headline metric is defined on the Microsoft Philly trace
(msr-fiddle/philly-traces). That file is not redistributable/fetchable in
this offline image, so this module defines a documented CSV schema any
Philly-derived (or other cluster) trace can be distilled into, loads it
into the same ``JobTrace`` struct-of-arrays the synthetic generators
produce, and windows it per-env. A committed sample fixture
(tests/data/philly_sample.csv) + generator (tools/make_philly_fixture.py)
stand in for the real file.

CSV schema (header row required; extra columns ignored):

  column        aliases                      meaning
  ------------  ---------------------------  -------------------------------
  submit_time   arrival_s, timestamp         job arrival, seconds (float ok;
                                             any epoch — re-based to min)
  gpus          num_gpus, gpu_num            max GPU demand (int >= 1)
  duration_s    duration, run_time           REALIZED service seconds (for
                                             failed/killed jobs this is the
                                             truncated runtime, as in the
                                             published Philly data)
  min_gpus      (optional)                   elastic minimum demand;
                                             absent -> rigid (= gpus)
  status        (optional)                   Pass/Killed/Failed —
                                             informational only (realized
                                             duration already encodes
                                             early failure)

Rows are sorted by submit_time on load; per-env windows are CONSECUTIVE
slices at seeded random offsets, preserving the real arrival process /
burstiness inside each window (each env re-bases arrivals to its window
start, as with synthetic pools).
"""
from __future__ import annotations

import csv
from pathlib import Path

import numpy as np

from .synthetic import JobTrace

_US = 1_000_000

_ALIASES = {
    "submit_time": ("submit_time", "arrival_s", "timestamp"),
    "gpus": ("gpus", "num_gpus", "gpu_num"),
    "duration_s": ("duration_s", "duration", "run_time"),
    "min_gpus": ("min_gpus",),
}


def _pick(header: list[str], field: str) -> int | None:
    for alias in _ALIASES[field]:
        if alias in header:
            return header.index(alias)
    return None


def load_trace_csv(path: str | Path, max_job_gpus: int,
                   oversize: str = "clip") -> JobTrace:
    """Load a Philly-style CSV into a JobTrace (int64-µs, arrival-sorted).

    oversize: jobs demanding more than max_job_gpus (the simulated
    cluster size) are "clip"ped to it (load-preserving, default) or
    "drop"ped.
    """
    assert oversize in ("clip", "drop")
    path = Path(path)
    with path.open(newline="") as f:
        rd = csv.reader(f)
        header = [h.strip().lower() for h in next(rd)]
        ia = _pick(header, "submit_time")
        ig = _pick(header, "gpus")
        id_ = _pick(header, "duration_s")
        im = _pick(header, "min_gpus")
        missing = [n for n, i in (("submit_time", ia), ("gpus", ig),
                                  ("duration_s", id_)) if i is None]
        if missing:
            raise ValueError(
                f"{path}: missing required column(s) {missing}; header was "
                f"{header} (see traces/philly.py for schema + aliases)")
        arr, gp, du, mg = [], [], [], []
        for ln, row in enumerate(rd, start=2):
            if not row or not "".join(row).strip():
                continue
            try:
                a = float(row[ia])
                g = int(float(row[ig]))
                d = float(row[id_])
            except (ValueError, IndexError) as e:
                raise ValueError(f"{path}:{ln}: bad row {row!r}: {e}") from e
            if g < 1 or d <= 0:
                continue  # zero-GPU / zero-runtime records (Philly has them)
            m = g
            if im is not None and len(row) > im and row[im].strip():
                m = max(1, int(float(row[im])))
            arr.append(a)
            gp.append(g)
            du.append(d)
            mg.append(min(m, g))
    if not arr:
        raise ValueError(f"{path}: no usable job rows")
    a = np.asarray(arr, np.float64)
    g = np.asarray(gp, np.int64)
    d = np.asarray(du, np.float64)
    m = np.asarray(mg, np.int64)
    if oversize == "drop":
        keep = g <= max_job_gpus
        a, g, d, m = a[keep], g[keep], d[keep], m[keep]
        if len(a) == 0:
            raise ValueError(f"{path}: every job exceeds {max_job_gpus} GPUs")
    else:
        g = np.minimum(g, max_job_gpus)
        m = np.minimum(m, g)
    order = np.argsort(a, kind="stable")
    a, g, d, m = a[order], g[order], d[order], m[order]
    a -= a[0]  # re-base epoch
    return JobTrace(
        arrival_us=np.round(a * _US).astype(np.int64),
        gpus=g.astype(np.int32),
        work_us=np.maximum(np.round(d * _US), 1).astype(np.int64),
        min_gpus=m.astype(np.int32),
    )


def windows_from_trace(trace: JobTrace, num_envs: int, jobs_per_env: int,
                       seed: int) -> tuple[JobTrace, np.ndarray, np.ndarray]:
    """Per-env windows over a loaded trace: consecutive jobs_per_env-job
    slices at seeded random offsets (consecutive slices of a sorted trace
    stay sorted, and keep the real arrival process). Returns
    (pool, job_lo, job_hi) exactly like synthetic.make_pool."""
    n = len(trace)
    if n < jobs_per_env:
        raise ValueError(
            f"trace has {n} jobs < jobs_per_env={jobs_per_env}; lower "
            "--jobs-per-env or supply a bigger trace file")
    rng = np.random.default_rng(seed)
    starts = (rng.integers(0, n - jobs_per_env + 1, size=num_envs)
              if n > jobs_per_env else np.zeros(num_envs, np.int64))
    arrs, gps, wks, mgs, lo, hi = [], [], [], [], [], []
    off = 0
    for s in starts:
        sl = slice(int(s), int(s) + jobs_per_env)
        arrs.append(trace.arrival_us[sl])
        gps.append(trace.gpus[sl])
        wks.append(trace.work_us[sl])
        mgs.append(trace.min_gpus[sl])
        lo.append(off)
        off += jobs_per_env
        hi.append(off)
    pool = JobTrace(np.concatenate(arrs), np.concatenate(gps),
                    np.concatenate(wks), np.concatenate(mgs))
    return pool, np.asarray(lo, np.int64), np.asarray(hi, np.int64)
