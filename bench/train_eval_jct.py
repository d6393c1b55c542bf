"""RL-quality experiment: train PPO on the 64-GPU sim, then evaluate mean
JCT against FIFO/SJF/random on IDENTICAL trace pools (same seeds).

Usage (GPU box): python bench/train_eval_jct.py [--iters 300]
Writes results JSON to stdout; used for docs/PERF.md quality table.
"""
import argparse
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch  # noqa: E402

from rlgpuschedule_amd.agents.evaluate import (evaluate_baseline,  # noqa: E402
                                               evaluate_policy)
from rlgpuschedule_amd.env.spec import cluster64  # noqa: E402
from rlgpuschedule_amd.trainer import CONFIGS, make_policy  # noqa: E402


def run(iters=300, envs=2048, eval_envs=16, eval_eps=16, jobs=96,
        interarrival=90.0, duration=600.0, seed=7, phase2_iters=0,
        phase2_ent=0.002, phase2_lr=1e-4, dtype="fp32", hidden=256,
        save_ckpt=None, ent_coef=0.01, lr=3e-4, select_every=0,
        select_envs=8, select_eps=8, kl_coef=0.0,
        obs_version=1, elastic=False, ent_floor=0.0, ent_floor_coef=0.0,
        bc_rollouts=0, bc_epochs=4, bc_lr=1e-3, bc_ent=0.0):
    """Programmatic version returning the full comparison dict.
    Optional phase 2: entropy/lr anneal via a fresh learner (Adam state
    resets; the captured graph bakes coefficients, so annealing means a
    new capture).
    select_every>0 enables eval-based checkpoint selection: every
    select_every iters run a quick greedy eval on a SELECTION trace pool
    (seed 555, disjoint from the final-eval seed 991 so the reported
    numbers carry no selection bias) and keep the best-JCT snapshot;
    the final evaluation then uses that snapshot. This is the standard
    answer to high-variance RL training (atomic-accumulation
    nondeterminism makes late training chaotic; see docs/PERF.md)."""
    from rlgpuschedule_amd.agents.ppo import PPOConfig
    from rlgpuschedule_amd.agents.ppo_hip import HipPPOLearner
    from rlgpuschedule_amd.agents.rollout_gpu import GpuRolloutEngine
    from rlgpuschedule_amd.env.hip_env import HipVectorEnv

    tkw = dict(mean_interarrival_s=interarrival, median_duration_s=duration)
    spec = cluster64(horizon=200_000, obs_version=obs_version,
                     elastic=elastic, elastic_resize=elastic)
    if elastic:
        tkw["elastic"] = True  # traces with min_gpus < gpus
    torch.manual_seed(seed)
    env = HipVectorEnv(spec, envs, trace_kind="philly", jobs_per_env=jobs,
                       seed=seed, **tkw)
    import dataclasses
    cfg0 = dataclasses.replace(CONFIGS["c2_ppo_mlp"], hidden=hidden,
                               spec=spec)
    policy = make_policy(cfg0, torch.device("cuda"))
    bc_stats = {}
    if bc_rollouts:
        # SJF warm start (agents/bc.py): clone the heuristic + value
        # warmup on a separate env pool, so PPO fine-tunes FROM the
        # baseline it must beat
        from rlgpuschedule_amd.agents.bc import pretrain_bc
        bc_env = HipVectorEnv(spec, min(envs, 1024), trace_kind="philly",
                              jobs_per_env=jobs, seed=seed + 50_000, **tkw)
        bc_stats = pretrain_bc(policy, bc_env, T=64,
                               n_rollouts=bc_rollouts, epochs=bc_epochs,
                               lr=bc_lr, ent_coef=bc_ent, device="cuda")
    learner = HipPPOLearner(policy, PPOConfig(epochs=1, minibatches=4,
                                              ent_coef=ent_coef, lr=lr,
                                              kl_coef=kl_coef,
                                              ent_floor=ent_floor,
                                              ent_floor_coef=ent_floor_coef),
                            dtype=dtype)
    engine = GpuRolloutEngine(env, policy, 64, seed=seed)
    t0 = time.perf_counter()
    stats = {}
    best = {"jct": float("inf"), "state": None, "at": -1}
    # max_steps bounds the cost of evaluating a degenerate (never-schedule)
    # policy: episodes then never finish and the eval would run 200k steps
    skw = dict(num_envs=select_envs, min_episodes=select_eps,
               jobs_per_env=jobs, seed=555, max_steps=30_000, **tkw)

    def _select(it):
        st = evaluate_policy(spec, policy, greedy=True, device="cuda", **skw)
        # guard: a degenerate never-schedule policy completes 0 jobs and
        # reports mean_jct_us == 0 — that must never win selection
        if st["completed"] > 0 and st["mean_jct_us"] < best["jct"]:
            best["jct"] = st["mean_jct_us"]
            best["state"] = {k: v.detach().clone()
                             for k, v in policy.state_dict().items()}
            best["at"] = it

    if select_every and bc_rollouts:
        _select(0)   # the BC(SJF) policy itself is a selection candidate
    for it in range(iters):
        buf = engine.collect()
        stats = learner.update(buf)
        if select_every and (it + 1) % select_every == 0:
            _select(it + 1)
    if phase2_iters:
        # low-lr settle phase: fresh learner (Adam state + captured graph
        # bake the coefficients); selection keeps running
        learner = HipPPOLearner(
            policy, PPOConfig(epochs=1, minibatches=4, lr=phase2_lr,
                              ent_coef=phase2_ent, kl_coef=kl_coef),
            dtype=dtype)
        for it in range(phase2_iters):
            buf = engine.collect()
            stats = learner.update(buf)
            if select_every and (it + 1) % select_every == 0:
                _select(iters + it + 1)
    torch.cuda.synchronize()
    res = {"train_s": round(time.perf_counter() - t0, 1),
           "final": {k: round(v, 4) for k, v in stats.items()}}
    if bc_stats:
        res["bc"] = bc_stats
    if select_every and best["state"] is not None:
        policy.load_state_dict(best["state"])
        res["selected_at"] = best["at"]
        res["selection_jct_us"] = round(best["jct"], 1)
    if save_ckpt:
        from rlgpuschedule_amd.ckpt.io import save_policy
        save_policy(save_ckpt, policy, spec=spec,
                    meta={"iters": iters, "hidden": hidden, "dtype": dtype})
    ekw = dict(num_envs=eval_envs, min_episodes=eval_eps,
               jobs_per_env=jobs, seed=991, **tkw)
    res["ppo_greedy"] = evaluate_policy(spec, policy, greedy=True,
                                        device="cuda", **ekw)
    res["ppo_sample"] = evaluate_policy(spec, policy, greedy=False,
                                        device="cuda", **ekw)
    for b in ("sjf", "fifo", "random"):
        res[b] = evaluate_baseline(spec, b, **ekw)
    for k in ("ppo_greedy", "ppo_sample", "sjf", "fifo", "random"):
        res[k] = {kk: (round(vv, 1) if isinstance(vv, float) else vv)
                  for kk, vv in res[k].items()}
    return res


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=300)
    ap.add_argument("--envs", type=int, default=2048)
    a = ap.parse_args()
    print(json.dumps(run(iters=a.iters, envs=a.envs), indent=1))
