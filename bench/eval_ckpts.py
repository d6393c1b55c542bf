"""Re-evaluate saved sweep checkpoints on FRESH held-out pools.

Robustness check for the round-2 quality headline: the 8-seed median
was selected/evaluated on pools 555/991; this replays the exact saved
policies on additional never-seen 512-window pools and reports the
median against SJF/FIFO on the same pools. Usage:

  python bench/eval_ckpts.py --ckpt-dir gpurun_out/r2_ckpts \
      --pools 1777 2888 --out gpurun_out/r2_ckpt_reeval.json
"""
import argparse
import json
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

import torch  # noqa: E402

from rlgpuschedule_amd.agents.evaluate import (evaluate_baseline,  # noqa: E402
                                               evaluate_policy)
from rlgpuschedule_amd.ckpt.io import load_policy  # noqa: E402
from rlgpuschedule_amd.env.spec import EnvSpec  # noqa: E402
from rlgpuschedule_amd.models.policy import MLPPolicy  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--ckpt-dir", default="gpurun_out/r2_ckpts")
    ap.add_argument("--pools", type=int, nargs="+", default=[1777, 2888])
    ap.add_argument("--envs", type=int, default=512)
    ap.add_argument("--jobs", type=int, default=96)
    ap.add_argument("--out", default="gpurun_out/r2_ckpt_reeval.json")
    a = ap.parse_args()
    tkw = dict(mean_interarrival_s=90.0, median_duration_s=600.0)
    cks = sorted(Path(a.ckpt_dir).glob("seed*.pt"))
    assert cks, a.ckpt_dir
    out = {"pools": {}}
    for pool in a.pools:
        kw = dict(num_envs=a.envs, min_episodes=a.envs,
                  jobs_per_env=a.jobs, seed=pool, max_steps=30_000, **tkw)
        rec = {"baselines": {}, "ckpts": {}}
        spec = None
        for ck in cks:
            meta = load_policy(ck)
            if spec is None:
                spec = EnvSpec(**meta["env_spec"])
                for b in ("sjf", "fifo"):
                    st = evaluate_baseline(spec, b, **kw)
                    rec["baselines"][b] = {
                        "jct_s": round(st["mean_jct_us"] / 1e6, 1),
                        "episodes": st["episodes"],
                        "completed": st["completed"]}
            pol = MLPPolicy(spec.obs_dim, spec.num_actions,
                            meta["meta"]["hidden"])
            load_policy(ck, pol)
            st = evaluate_policy(spec, pol, greedy=True, device="cuda",
                                 **kw)
            rec["ckpts"][ck.stem] = {
                "jct_s": round(st["mean_jct_us"] / 1e6, 1),
                "episodes": st["episodes"], "completed": st["completed"]}
        js = sorted(v["jct_s"] for v in rec["ckpts"].values())
        n = len(js)
        rec["median_jct_s"] = (js[n // 2] if n % 2 else
                               0.5 * (js[n // 2 - 1] + js[n // 2]))
        out["pools"][str(pool)] = rec
        print(json.dumps({("pool"): pool,
                          "median": rec["median_jct_s"],
                          "sjf": rec["baselines"]["sjf"]["jct_s"],
                          "fifo": rec["baselines"]["fifo"]["jct_s"]}),
              flush=True)
    Path(a.out).parent.mkdir(parents=True, exist_ok=True)
    Path(a.out).write_text(json.dumps(out, indent=1))
    print("wrote", a.out)


if __name__ == "__main__":
    main()
