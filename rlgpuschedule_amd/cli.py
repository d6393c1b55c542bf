"""Command-line interface: `python -m rlgpuschedule_amd.cli train ...`

Maps the five BASELINE.json configs to the trainer; supports checkpoint
save/resume, metrics logging, and env/rollout overrides.
"""
from __future__ import annotations

import argparse
import dataclasses

from .trainer import CONFIGS, train


def main(argv=None):
    ap = argparse.ArgumentParser(prog="rlsched")
    sub = ap.add_subparsers(dest="cmd", required=True)
    tr = sub.add_parser("train", help="run a training config")
    tr.add_argument("--config", choices=sorted(CONFIGS), default="c2_ppo_mlp")
    tr.add_argument("--iterations", type=int, default=None)
    tr.add_argument("--envs", type=int, default=None)
    tr.add_argument("--rollout", type=int, default=None)
    tr.add_argument("--seed", type=int, default=None)
    tr.add_argument("--device", default=None)
    tr.add_argument("--jobs-per-env", type=int, default=None)
    tr.add_argument("--out", default=None, help="metrics/ckpt directory")
    tr.add_argument("--save-every", type=int, default=0)
    tr.add_argument("--resume", default=None, help="checkpoint path")
    ls = sub.add_parser("configs", help="list configs")
    args = ap.parse_args(argv)

    if args.cmd == "configs":
        for name, cfg in sorted(CONFIGS.items()):
            print(f"{name}: algo={cfg.algo} policy={cfg.policy} "
                  f"cluster={cfg.spec.num_nodes}x{cfg.spec.gpus_per_node} "
                  f"envs={cfg.num_envs} T={cfg.rollout_T}")
        return 0

    cfg = CONFIGS[args.config]
    over = {}
    if args.envs is not None:
        over["num_envs"] = args.envs
    if args.rollout is not None:
        over["rollout_T"] = args.rollout
    if args.seed is not None:
        over["seed"] = args.seed
    if args.device is not None:
        over["device"] = args.device
    if args.jobs_per_env is not None:
        over["jobs_per_env"] = args.jobs_per_env
    if over:
        cfg = dataclasses.replace(cfg, **over)
    out = train(cfg, iterations=args.iterations, out_dir=args.out,
                save_every=args.save_every, resume=args.resume)
    if "episode_stats" in out:
        print("episode stats:", out["episode_stats"])
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
