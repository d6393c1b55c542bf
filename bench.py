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
sim; mean JCT over finished episodes is reported alongside as
mean_jct_us_randpolicy (weights are random-init per the driver contract —
the QUALITY evidence lives in profiles/r*_jct_eval*.json, not here).
Traces are synthetic Philly-like by default (no network for the real
trace; see rlgpuschedule_amd/traces/); --trace-file replays a real
Philly-style CSV instead.

Transparency: single-GPU runs also measure and embed
two aux figures in the same JSON line — the all-fp32 full-loop number and
the literal BASELINE config-2 number (256 envs, T=64) — so the eventual
reference comparison cannot be accused of config inflation.
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


def _make_env(spec, E, shards, seed, trace_kind, trace_kw, jobs_per_env):
    from rlgpuschedule_amd.env.hip_env import HipVectorEnv
    ns = max(1, shards)
    assert E % ns == 0
    env = [HipVectorEnv(spec, E // ns, trace_kind=trace_kind,
                        jobs_per_env=jobs_per_env, seed=seed + s, **trace_kw)
           for s in range(ns)]
    return env[0] if ns == 1 else env


def _episode_stats(env):
    if isinstance(env, list):
        sts = [e.episode_stats() for e in env]
        comp = sum(s["completed"] for s in sts)
        return {
            "episodes": sum(s["episodes"] for s in sts),
            "completed": comp,
            "mean_jct_us": (sum(s["mean_jct_us"] * s["completed"]
                                for s in sts) / comp) if comp else 0.0,
        }
    return env.episode_stats() if hasattr(env, "episode_stats") else {}


def _gpu_bench(spec, E, T, policy_kind, hidden, actor_dtype, update_dtype,
               minibatches, shards, learner_kind, steps, warmup, seed,
               trace_kind, trace_kw, jobs_per_env, sync_dist=False,
               rollout="twokernel"):
    """Build a fresh env/policy/learner and time `steps` full iterations.
    Returns (env_steps_per_sec_1gpu, ms_per_step, episode_stats)."""
    from rlgpuschedule_amd.agents.ppo_gpu import FusedPPOLearner
    from rlgpuschedule_amd.agents.ppo_gru_hip import HipGRUPPOLearner
    from rlgpuschedule_amd.agents.ppo_hip import HipPPOLearner
    from rlgpuschedule_amd.agents.rollout_gpu import GpuRolloutEngine
    env = _make_env(spec, E, shards, seed, trace_kind, trace_kw,
                    jobs_per_env)
    pcls = GRUPolicy if policy_kind == "gru" else MLPPolicy
    policy = pcls(spec.obs_dim, spec.num_actions, hidden).to("cuda")
    grad_sync = None
    if sync_dist:
        pdist.broadcast_params(policy)
        if pdist.world_size() > 1:
            grad_sync = pdist.FlatGradAllreduce(policy)
    cfg = PPOConfig(epochs=1, minibatches=minibatches)
    if policy.is_recurrent:
        learner = HipGRUPPOLearner(policy, cfg, grad_sync,
                                   dtype=update_dtype)
    elif learner_kind == "hip":
        learner = HipPPOLearner(policy, cfg, grad_sync, dtype=update_dtype)
    else:
        learner = FusedPPOLearner(policy, cfg, grad_sync)
    collector = GpuRolloutEngine(env, policy, T, seed=seed + 1000,
                                 actor_dtype=actor_dtype, rollout=rollout)

    def one_iter():
        learner.update(collector.collect())

    for _ in range(warmup):
        one_iter()
    if sync_dist:
        pdist.barrier()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        one_iter()
    torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    if sync_dist:
        elapsed = pdist.allreduce_scalar(elapsed, "max")  # MAX over ranks
        pdist.barrier()
    return (steps * T * E / elapsed, elapsed / steps * 1e3,
            _episode_stats(env))


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
    ap.add_argument("--rollout-mode", choices=["twokernel", "mega"],
                    default="twokernel",
                    help="mega: fused whole-rollout kernel (fp32 actor, "
                         "1 shard)")
    ap.add_argument("--trace-file", default=None,
                    help="replay a real Philly-style CSV trace "
                         "(schema: rlgpuschedule_amd/traces/philly.py) "
                         "instead of the synthetic generator")
    ap.add_argument("--no-aux", action="store_true",
                    help="skip the fp32 + literal-config-2 aux "
                         "measurements (they run on 1-GPU runs only)")
    args = ap.parse_args()

    rank, world, local = pdist.init_from_env()
    has_gpu = torch.cuda.is_available()
    device = torch.device(f"cuda:{local}" if has_gpu else "cpu")
    torch.manual_seed(1234 + rank)
    if has_gpu:
        torch.cuda.set_device(local)

    trace_kind = "philly"
    trace_kw: dict = {}
    if args.trace_file:
        trace_kind = "file"
        trace_kw = {"path": args.trace_file}

    spec = cluster64(horizon=1_000_000)
    E, T = args.envs, args.rollout
    JPE = 256
    if not has_gpu:  # CPU fallback so the script runs anywhere (tiny)
        E = min(E, 8)
        env = TorchEnvAdapter(
            CpuVectorEnv(spec, E, trace_kind=trace_kind, jobs_per_env=64,
                         seed=97 + rank, **trace_kw), device)

    if has_gpu:
        if args.rollout_mode == "mega":
            args.actor_dtype = "fp32"
            args.shards = 1
        value_1gpu, ms_per_step, st = _gpu_bench(
            spec, E, T, args.policy, args.hidden, args.actor_dtype,
            args.update_dtype, args.minibatches, args.shards, args.learner,
            args.steps, args.warmup, seed=97 + rank * 31,
            trace_kind=trace_kind, trace_kw=trace_kw, jobs_per_env=JPE,
            sync_dist=True, rollout=args.rollout_mode)
        n_gpus = world
        value = value_1gpu * n_gpus  # whole-job aggregate (weak scaling)
        elapsed_ms = ms_per_step
    else:
        policy = (GRUPolicy if args.policy == "gru" else MLPPolicy)(
            spec.obs_dim, spec.num_actions, args.hidden).to(device)
        pdist.broadcast_params(policy)
        grad_sync = pdist.FlatGradAllreduce(policy) if world > 1 else None
        cfg = PPOConfig(epochs=1, minibatches=args.minibatches)
        learner = PPOLearner(policy, cfg, grad_sync)
        collector = Collector(env, policy, T, device, seed=555 + rank)
        for _ in range(args.warmup):
            learner.update(collector.collect())
        pdist.barrier()
        t0 = time.perf_counter()
        for _ in range(args.steps):
            learner.update(collector.collect())
        elapsed = time.perf_counter() - t0
        elapsed = pdist.allreduce_scalar(elapsed, "max")
        pdist.barrier()
        n_gpus = args.gpus
        value = args.steps * T * E * n_gpus / elapsed
        elapsed_ms = elapsed / args.steps * 1e3
        st = _episode_stats(env)

    # Aux transparency figures (1-GPU runs only; each is a fresh
    # env+policy+learner so nothing leaks from the headline run)
    aux = {}
    if has_gpu and world == 1 and not args.no_aux and args.policy == "mlp":
        v_fp32, ms_fp32, _ = _gpu_bench(
            spec, E, T, "mlp", args.hidden, "fp32", "fp32",
            args.minibatches, args.shards, args.learner,
            max(5, args.steps // 2), max(2, args.warmup // 2), seed=4242,
            trace_kind=trace_kind, trace_kw=trace_kw, jobs_per_env=JPE)
        v_256, ms_256, _ = _gpu_bench(
            spec, 256, 64, "mlp", args.hidden, args.actor_dtype,
            args.update_dtype, args.minibatches, 1, args.learner,
            max(50, args.steps), 10, seed=777,
            trace_kind=trace_kind, trace_kw=trace_kw, jobs_per_env=JPE)
        aux = {
            "fp32_env_steps_per_sec": round(v_fp32, 1),
            "fp32_ms_per_step": round(ms_fp32, 3),
            "cfg2_256env_env_steps_per_sec": round(v_256, 1),
            "cfg2_256env_ms_per_step": round(ms_256, 3),
        }

    if rank == 0:
        out = {
            "metric": "env_steps_per_sec",
            "value": round(value, 1),
            "unit": "env-steps/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed_ms, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # no published reference numbers
            "dtype": (f"mixed (actor {args.actor_dtype}, "
                      f"update {args.update_dtype})"
                      if has_gpu and "bf16" in (args.actor_dtype,
                                                args.update_dtype)
                      else "fp32"),   # CPU fallback is always torch fp32
            "data": (f"trace-file:{os.path.basename(args.trace_file)}"
                     if args.trace_file else "synthetic"),
            # random-init weights per the driver contract — NOT the
            # quality headline (that is profiles/r*_jct_eval*.json)
            "mean_jct_us_randpolicy": round(st.get("mean_jct_us", 0.0), 1),
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
        if aux:
            out["aux"] = aux
        print(json.dumps(out))


if __name__ == "__main__":
    main()
