"""Generate the committed Philly-style CSV fixture
(tests/data/philly_sample.csv) in the schema traces/philly.py documents.

The real Philly trace is not redistributable in this offline image; this
fixture has the same column layout, realistic marginals (heavy-tail GPU
sizes, lognormal durations, failed/killed jobs with truncated realized
runtimes) and a few deliberately-messy rows (zero-GPU records, blank
lines) that the loader must skip — so swapping in a real distilled trace
file is purely a data change.

Deterministic: python tools/make_philly_fixture.py [out] regenerates the
identical file.
"""
import sys
from pathlib import Path

import numpy as np

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

from rlgpuschedule_amd.traces.synthetic import philly_like  # noqa: E402

N = 800
SEED = 20260914


def main(out: Path):
    tr = philly_like(N, SEED, max_job_gpus=64, fail_frac=0.15, elastic=True)
    rng = np.random.default_rng(SEED + 1)
    status = np.where(tr.work_us < np.int64(60_000_000),
                      rng.choice(["Failed", "Killed"], size=N), "Pass")
    lines = ["submit_time,gpus,duration_s,min_gpus,status"]
    for i in range(N):
        lines.append(
            f"{tr.arrival_us[i] / 1e6:.3f},{tr.gpus[i]},"
            f"{tr.work_us[i] / 1e6:.3f},{tr.min_gpus[i]},{status[i]}")
        if i == 100:
            lines.append("")                       # blank line (skipped)
        if i == 200:
            lines.append("123.0,0,55.0,0,Killed")  # zero-GPU record (skipped)
    out.parent.mkdir(parents=True, exist_ok=True)
    out.write_text("\n".join(lines) + "\n")
    print(f"wrote {out} ({N} jobs + 2 junk rows)")


if __name__ == "__main__":
    main(Path(sys.argv[1]) if len(sys.argv) > 1
         else REPO / "tests" / "data" / "philly_sample.csv")
