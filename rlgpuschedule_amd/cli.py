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
    tr.add_argument("--actor-dtype", choices=["fp32", "bf16"], default=None)
    tr.add_argument("--update-dtype", choices=["fp32", "bf16"], default=None)
    tr.add_argument("--trace-file", default=None,
                    help="replay a real Philly-style CSV trace "
                         "(schema: traces/philly.py)")
    tr.add_argument("--obs-version", type=int, choices=[1, 2], default=None,
                    help="2 = richer observations (docs/ENV_SPEC.md v4)")
    ev = sub.add_parser("eval", help="evaluate a checkpoint or baseline")
    ev.add_argument("--config", choices=sorted(CONFIGS), default="c2_ppo_mlp")
    ev.add_argument("--ckpt", default=None, help="policy checkpoint (.pt/.safetensors)")
    ev.add_argument("--baseline", choices=["fifo", "sjf", "random"],
                    default=None)
    ev.add_argument("--envs", type=int, default=32)
    ev.add_argument("--episodes", type=int, default=32)
    ev.add_argument("--jobs-per-env", type=int, default=64)
    ev.add_argument("--seed", type=int, default=1234)
    ev.add_argument("--trace-file", default=None,
                    help="evaluate on a real Philly-style CSV trace")
    ev.add_argument("--sample", action="store_true",
                    help="sample instead of greedy argmax")
    sw = sub.add_parser("train-sweep", help="multi-seed training with "
                        "eval-based checkpoint selection (agents/selection)")
    sw.add_argument("--config", choices=sorted(CONFIGS), default="c2_ppo_mlp")
    sw.add_argument("--seeds", type=int, nargs="+", default=[7, 21, 42])
    sw.add_argument("--iterations", type=int, default=None)
    sw.add_argument("--envs", type=int, default=None)
    sw.add_argument("--jobs-per-env", type=int, default=None)
    sw.add_argument("--select-every", type=int, default=200)
    sw.add_argument("--kl-coef", type=float, default=0.2,
                    help="KL-to-behavior penalty (prevents collapse)")
    sw.add_argument("--bc-rollouts", type=int, default=0,
                    help=">0: SJF behavior-cloning warm start per seed")
    sw.add_argument("--bc-ent", type=float, default=0.03)
    sw.add_argument("--save-best", default=None, help="checkpoint path for "
                    "the winning seed's selected policy")
    ls = sub.add_parser("configs", help="list configs")
    args = ap.parse_args(argv)

    if args.cmd == "configs":
        for name, cfg in sorted(CONFIGS.items()):
            print(f"{name}: algo={cfg.algo} policy={cfg.policy} "
                  f"cluster={cfg.spec.num_nodes}x{cfg.spec.gpus_per_node} "
                  f"envs={cfg.num_envs} T={cfg.rollout_T}")
        return 0

    if args.cmd == "eval":
        import json
        from .agents.evaluate import evaluate_baseline, evaluate_policy
        cfg = CONFIGS[args.config]
        if args.trace_file:
            cfg = dataclasses.replace(
                cfg, trace_kind="file",
                trace_kw={"path": args.trace_file})
        if args.baseline:
            st = evaluate_baseline(cfg.spec, args.baseline, args.envs,
                                   args.episodes,
                                   trace_kind=cfg.trace_kind,
                                   jobs_per_env=args.jobs_per_env,
                                   seed=args.seed, **cfg.trace_kw)
        else:
            from .ckpt.io import load_policy
            from .trainer import make_policy, resolve_device
            policy = make_policy(cfg, resolve_device(cfg))
            if args.ckpt:
                load_policy(args.ckpt, policy)
            st = evaluate_policy(cfg.spec, policy, args.envs, args.episodes,
                                 trace_kind=cfg.trace_kind,
                                 jobs_per_env=args.jobs_per_env,
                                 seed=args.seed, greedy=not args.sample,
                                 **cfg.trace_kw)
        print(json.dumps(st))
        return 0

    if args.cmd == "train-sweep":
        import json
        from .agents.selection import train_multi_seed
        cfg = CONFIGS[args.config]
        over = {}
        if args.envs is not None:
            over["num_envs"] = args.envs
        if args.jobs_per_env is not None:
            over["jobs_per_env"] = args.jobs_per_env
        if over:
            cfg = dataclasses.replace(cfg, **over)
        cfg = dataclasses.replace(
            cfg, ppo=dataclasses.replace(cfg.ppo, kl_coef=args.kl_coef),
            bc_rollouts=args.bc_rollouts, bc_ent=args.bc_ent)
        res = train_multi_seed(cfg, args.seeds, iterations=args.iterations,
                               select_every=args.select_every,
                               log_fn=print, save_best=args.save_best)
        res.pop("best_state", None)
        print(json.dumps(res))
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
    if args.actor_dtype is not None:
        over["actor_dtype"] = args.actor_dtype
    if args.update_dtype is not None:
        over["update_dtype"] = args.update_dtype
    if args.obs_version is not None:
        over["spec"] = dataclasses.replace(cfg.spec,
                                           obs_version=args.obs_version)
    if args.trace_file is not None:
        over["trace_kind"] = "file"
        over["trace_kw"] = {"path": args.trace_file}
    if over:
        cfg = dataclasses.replace(cfg, **over)
    out = train(cfg, iterations=args.iterations, out_dir=args.out,
                save_every=args.save_every, resume=args.resume)
    if "episode_stats" in out:
        print("episode stats:", out["episode_stats"])
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
