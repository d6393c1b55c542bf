"""IMPALA staleness ablation: does the V-trace
correction earn its keep?

Grid: behavior_sync (staleness) x use_vtrace (correction on/off), same
seed/env pool. With sync=1 the rollout is near-on-policy and V-trace
should be a no-op; as sync grows the behavior policy lags the learner by
up to `sync` updates and plain GAE targets become biased — V-trace's
rho/c clipping should keep learning stable where GAE degrades.

Writes a JSON with per-config mean_reward curves + final episode stats to
--out (commit the GPU run's output under profiles/).

Runs on CPU (tiny) or GPU (c5-scale); the torch VtraceLearner is used for
both arms so the ONLY difference is the target computation.
"""
from __future__ import annotations

import argparse
import dataclasses
import json
import sys
import tempfile
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

import torch  # noqa: E402

from rlgpuschedule_amd.agents.impala import ImpalaConfig, train_impala  # noqa: E402
from rlgpuschedule_amd.env.spec import cluster16  # noqa: E402
from rlgpuschedule_amd.trainer import CONFIGS  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iterations", type=int, default=300)
    ap.add_argument("--envs", type=int, default=1024)
    ap.add_argument("--syncs", type=int, nargs="+", default=[1, 4, 16])
    ap.add_argument("--seed", type=int, default=7)
    ap.add_argument("--out", default="gpurun_out/impala_staleness.json")
    args = ap.parse_args()

    gpu = torch.cuda.is_available()
    base = CONFIGS["c5_impala"]
    if gpu:
        cfg = dataclasses.replace(base, num_envs=args.envs, seed=args.seed)
    else:
        cfg = dataclasses.replace(
            base, spec=cluster16(), num_envs=min(args.envs, 16),
            jobs_per_env=16, rollout_T=16, use_hip_env=False, device="cpu",
            hidden=32, seed=args.seed)

    results = []
    for sync in args.syncs:
        for vtrace in (True, False):
            # force_torch_learner: both arms share ONE learner
            # implementation, so target computation is the only
            # difference (env/seed/schedule are shared).
            icfg = ImpalaConfig(behavior_sync=sync, use_vtrace=vtrace,
                                force_torch_learner=True)
            with tempfile.TemporaryDirectory() as td:
                out = train_impala(cfg, iterations=args.iterations,
                                   log_fn=None, out_dir=td, icfg=icfg)
                lines = [json.loads(ln) for ln in
                         (Path(td) / "metrics.jsonl").read_text()
                         .splitlines()]
            curve = [(ln["iter"], ln.get("mean_reward")) for ln in lines
                     if "mean_reward" in ln]
            tail = [v for _, v in curve[-20:]]
            head = [v for _, v in curve[:20]]
            rec = {
                "behavior_sync": sync,
                "use_vtrace": vtrace,
                "iterations": args.iterations,
                "envs": cfg.num_envs,
                "device": "gpu" if gpu else "cpu",
                "mean_reward_first20": sum(head) / max(len(head), 1),
                "mean_reward_last20": sum(tail) / max(len(tail), 1),
                "final_stats": {k: v for k, v in
                                out.get("episode_stats", {}).items()},
                "curve_every10": curve[::10],
            }
            results.append(rec)
            print(json.dumps({k: rec[k] for k in
                              ("behavior_sync", "use_vtrace",
                               "mean_reward_first20",
                               "mean_reward_last20")}))
    outp = Path(args.out)
    outp.parent.mkdir(parents=True, exist_ok=True)
    outp.write_text(json.dumps({"grid": results}, indent=1))
    print("wrote", outp)


if __name__ == "__main__":
    main()
