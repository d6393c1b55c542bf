"""Multi-seed JCT quality sweep with eval-based checkpoint selection.

The honest statistical claim for docs/PERF.md: distribution of final
(held-out, seed-991) greedy/sampled JCT over several training seeds,
each run selecting its best checkpoint on a DISJOINT selection pool
(seed 555). Baselines evaluated once on the same held-out pool.
"""
import json
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
sys.path.insert(0, str(Path(__file__).resolve().parent))

from train_eval_jct import run  # noqa: E402

if __name__ == "__main__":
    seeds = [int(s) for s in (sys.argv[1:] or ["7", "21", "42", "1234"])]
    out = {"runs": []}
    for i, sd in enumerate(seeds):
        import os
        kl = float(os.environ.get("SWEEP_KL", "0.0"))
        r = run(iters=3000, envs=2048, hidden=512, seed=sd,
                select_every=200, dtype="bf16", kl_coef=kl)
        rec = {"seed": sd, "train_s": r["train_s"],
               "selected_at": r.get("selected_at"),
               "selection_jct_us": r.get("selection_jct_us"),
               "greedy_jct_s": round(r["ppo_greedy"]["mean_jct_us"] / 1e6, 1),
               "sample_jct_s": round(r["ppo_sample"]["mean_jct_us"] / 1e6, 1)}
        if i == 0:
            for b in ("sjf", "fifo", "random"):
                out[b + "_jct_s"] = round(r[b]["mean_jct_us"] / 1e6, 1)
        out["runs"].append(rec)
        print("RUN", json.dumps(rec), flush=True)
    print("SWEEP", json.dumps(out))
