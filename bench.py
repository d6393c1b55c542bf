#!/usr/bin/env python3
"""Headline benchmark: env-steps/sec of the flagship training loop.

One bench "step" = one full PPO iteration: a rollout of T env transitions
across E on-device envs (batched HIP env step + policy forward + sampling)
PLUS the PPO update (GAE + minibatched ratio-clip SGD). Nothing inside the
timed region is skipped.

Driver contract: `python bench.py --gpus N --steps K --warmup W`; for N>1
it is launched under torch.distributed.run with one rank per GPU (RCCL
over xGMI). Rank 0 prints ONE JSON line.

Metric (BASELINE.json): env-steps/sec (node) on the Philly-trace 64-GPU
sim; mean JCT over finished episodes is reported alongside. Traces are
synthetic Philly-like (no network for the real trace; see
rlgpuschedule_amd/traces/synthetic.py) and weights are random-init.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from rlgpuschedule_amd.agents.ppo import PPOConfig, PPOLearner
from rlgpuschedule_amd.agents.rollout import Collector, TorchEnvAdapter
from rlgpuschedule_amd.env.gym_api import CpuVectorEnv
from rlgpuschedule_amd.env.spec import cluster64
from rlgpuschedule_amd.models.policy import GRUPolicy, MLPPolicy
from rlgpuschedule_amd.parallel import dist as pdist



def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=12)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--envs", type=int, default=16384, help="envs per GPU")
    ap.add_argument("--rollout", type=int, default=64)
    ap.add_argument("--hidden", type=int, default=256)
    ap.add_argument("--policy", choices=["mlp", "gru"], default="mlp")
    ap.add_argument("--actor-dtype", choices=["fp32", "bf16"], default="bf16")
    ap.add_argument("--update-dtype", choices=["fp32", "bf16"],
                    default="bf16")
    ap.add_argument("--minibatches", type=int, default=2)
    ap.add_argument("--shards", type=int, default=2,
                    help="env shards on separate HIP streams (overlaps "
                         "actor and env kernels across shards)")
    ap.add_argument("--learner", choices=["hip", "fused"], default="hip",
                    help="hip: fully hand-written update kernels; fused: "
                         "torch GEMMs + fused HIP loss")
    args = ap.parse_args()

    rank, world, local = pdist.init_from_env()
    has_gpu = torch.cuda.is_available()
    device = torch.device(f"cuda:{local}" if has_gpu else "cpu")
    torch.manual_seed(1234 + rank)

    spec = cluster64(horizon=1_000_000)
    E, T = args.envs, args.rollout
    if has_gpu:
        from rlgpuschedule_amd.env.hip_env import HipVectorEnv
        ns = max(1, args.shards)
        assert E % ns == 0
        env = [HipVectorEnv(spec, E // ns, trace_kind="philly",
                            jobs_per_env=256, seed=97 + rank * 31 + s)
               for s in range(ns)]
        if ns == 1:
            env = env[0]
    else:  # CPU fallback so the script runs anywhere (tiny)
        E = min(E, 8)
        env = TorchEnvAdapter(
            CpuVectorEnv(spec, E, trace_kind="philly", jobs_per_env=64,
                         seed=97 + rank), device)

    pcls = GRUPolicy if args.policy == "gru" else MLPPolicy
    policy = pcls(spec.obs_dim, spec.num_actions, args.hidden).to(device)
    pdist.broadcast_params(policy)
    grad_sync = pdist.FlatGradAllreduce(policy) if world > 1 else None
    # Throughput-oriented PPO: 1 epoch x 8 minibatches (documented in config)
    cfg = PPOConfig(epochs=1, minibatches=args.minibatches)
    if has_gpu:
        from rlgpuschedule_amd.agents.ppo_gpu import FusedPPOLearner
        from rlgpuschedule_amd.agents.ppo_hip import HipPPOLearner
        from rlgpuschedule_amd.agents.ppo_gru_hip import HipGRUPPOLearner
        from rlgpuschedule_amd.agents.rollout_gpu import GpuRolloutEngine
        if policy.is_recurrent:
            learner = HipGRUPPOLearner(policy, cfg, grad_sync,
                                       dtype=args.update_dtype)
        elif args.learner == "hip":
            learner = HipPPOLearner(policy, cfg, grad_sync,
                                    dtype=args.update_dtype)
        else:
            learner = FusedPPOLearner(policy, cfg, grad_sync)
        collector = GpuRolloutEngine(env, policy, T, seed=555 + rank,
                                     actor_dtype=args.actor_dtype)
    else:
        learner = PPOLearner(policy, cfg, grad_sync)
        collector = Collector(env, policy, T, device, seed=555 + rank)

    def one_iter():
        buf = collector.collect()
        learner.update(buf)

    for _ in range(args.warmup):
        one_iter()

    pdist.barrier()
    if has_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_iter()
    if has_gpu:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    # MAX over ranks
    elapsed = pdist.allreduce_scalar(elapsed, "max")
    pdist.barrier()

    n_gpus = world if has_gpu else args.gpus
    env_steps = args.steps * T * E * n_gpus  # whole-job aggregate
    value = env_steps / elapsed
    if isinstance(env, list):
        sts = [e.episode_stats() for e in env]
        comp = sum(s["completed"] for s in sts)
        st = {
            "episodes": sum(s["episodes"] for s in sts),
            "completed": comp,
            "mean_jct_us": (sum(s["mean_jct_us"] * s["completed"]
                                for s in sts) / comp) if comp else 0.0,
        }
    else:
        st = env.episode_stats() if hasattr(env, "episode_stats") else {}
    if rank == 0:
        out = {
            "metric": "env_steps_per_sec",
            "value": round(value, 1),
            "unit": "env-steps/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1e3, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # no published reference numbers
            "dtype": (f"mixed (actor {args.actor_dtype}, "
                      f"update {args.update_dtype})"
                      if has_gpu and "bf16" in (args.actor_dtype,
                                                args.update_dtype)
                      else "fp32"),   # CPU fallback is always torch fp32
            "data": "synthetic",
            "mean_jct_us": round(st.get("mean_jct_us", 0.0), 1),
            "episodes": st.get("episodes", 0),
            "config": {
                "model": ("ppo_gru_256" if args.policy == "gru"
                          else "ppo_mlp_256x2"),
                "sim": "philly-like 64-GPU cluster (8 nodes x 8)",
                "global_batch": T * E * n_gpus,
                "seq_len": T,
                "envs_per_gpu": E,
                "ppo": f"1 epoch x {args.minibatches} minibatches, "
                       "clip 0.2, GAE(0.99, 0.95), reward-norm",
                "update_path": args.learner if has_gpu else "torch-cpu",
                "parallelism": f"dp{n_gpus}",
                "env_shards": args.shards if has_gpu else 1,
            },
        }
        print(json.dumps(out))


if __name__ == "__main__":
    main()
