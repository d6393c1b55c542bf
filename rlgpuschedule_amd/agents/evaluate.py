"""Policy evaluation: run a trained (or baseline) scheduler and report
episode stats (mean JCT, return, completed jobs).

Works with the torch policies (greedy or sampling) on the HIP env (GPU) or
the CPU reference env, and with the non-RL baselines.
"""
from __future__ import annotations

import numpy as np
import torch

from ..env.gym_api import CpuVectorEnv
from ..env.spec import EnvSpec
from ..models.policy import sample_categorical
from .baselines import FifoAgent, RandomAgent, SjfAgent


@torch.no_grad()
def evaluate_policy(spec: EnvSpec, policy, num_envs: int = 64,
                    min_episodes: int = 64, max_steps: int = 200_000,
                    trace_kind: str = "philly", jobs_per_env: int = 64,
                    seed: int = 1234, greedy: bool = True,
                    device: str | None = None, **trace_kw) -> dict:
    device = device or ("cuda" if torch.cuda.is_available() else "cpu")
    if device == "cuda":
        from ..env.hip_env import HipVectorEnv
        env = HipVectorEnv(spec, num_envs, trace_kind=trace_kind,
                           jobs_per_env=jobs_per_env, seed=seed, **trace_kw)
        obs = env.reset().to(torch.float32)
        mask = env.action_masks()
    else:
        env = CpuVectorEnv(spec, num_envs, trace_kind=trace_kind,
                           jobs_per_env=jobs_per_env, seed=seed, **trace_kw)
        obs = torch.from_numpy(env.reset())
        mask = torch.from_numpy(env.action_masks().astype(np.uint8))
    policy = policy.to(device)
    hx = policy.initial_state(num_envs, device)
    gen = torch.Generator(device=device)
    gen.manual_seed(seed)
    steps = 0
    while steps < max_steps:
        logits, _, hx_next = policy(obs.to(device), mask.to(device), hx)
        if greedy:
            actions = logits.argmax(-1)
        else:
            u = torch.rand(num_envs, device=device, generator=gen)
            actions, _, _ = sample_categorical(logits, u)
        if device == "cuda":
            obs, rew, done, info = env.step(actions)
            obs = obs.to(torch.float32)
            mask = info["action_mask"]
            if hx_next is not None:
                hx = hx_next * (1.0 - done.float()).unsqueeze(-1)
            steps += 1
            if steps % 512 == 0 and env.episode_stats()["episodes"] >= min_episodes:
                break
        else:
            obs_np, rew, done, info = env.step(actions.cpu().numpy())
            obs = torch.from_numpy(obs_np)
            mask = torch.from_numpy(info["action_mask"].astype(np.uint8))
            if hx_next is not None:
                hx = hx_next * torch.from_numpy(
                    1.0 - done.astype(np.float32)).unsqueeze(-1)
            steps += 1
            if env.stats.count >= min_episodes:
                break
    if device == "cuda":
        st = env.episode_stats()
    else:
        st = {"episodes": env.stats.count,
              "mean_return": env.stats.mean_return(),
              "mean_jct_us": env.stats.mean_jct_us(),
              "completed": env.stats.completed_sum}
    st["steps"] = steps
    return st


def evaluate_baseline(spec: EnvSpec, name: str, num_envs: int = 16,
                      min_episodes: int = 16, max_steps: int = 200_000,
                      trace_kind: str = "philly", jobs_per_env: int = 64,
                      seed: int = 1234, device: str | None = None,
                      **trace_kw) -> dict:
    """FIFO/SJF/random baselines. On a GPU the batched HIP env is used
    with vectorized torch agent rules (bit-parity env + same
    deterministic rules => same trajectories as the CPU reference for
    fifo/sjf), which makes big-pool evals (hundreds of trace windows)
    cheap; otherwise the CPU reference env."""
    device = device or ("cuda" if torch.cuda.is_available() else "cpu")
    if device == "cuda":
        return _evaluate_baseline_gpu(spec, name, num_envs, min_episodes,
                                      max_steps, trace_kind, jobs_per_env,
                                      seed, **trace_kw)
    env = CpuVectorEnv(spec, num_envs, trace_kind=trace_kind,
                       jobs_per_env=jobs_per_env, seed=seed, **trace_kw)
    agent = {"fifo": lambda: FifoAgent(spec.queue_visible),
             "sjf": lambda: SjfAgent(spec.queue_visible),
             "random": lambda: RandomAgent(spec.num_actions, seed)}[name]()
    obs = env.reset()
    mask = env.action_masks()
    steps = 0
    while env.stats.count < min_episodes and steps < max_steps:
        acts = agent.act(obs, mask)
        obs, rew, done, info = env.step(acts)
        mask = info["action_mask"]
        steps += 1
    return {"episodes": env.stats.count,
            "mean_return": env.stats.mean_return(),
            "mean_jct_us": env.stats.mean_jct_us(),
            "completed": env.stats.completed_sum, "steps": steps}


@torch.no_grad()
def _evaluate_baseline_gpu(spec, name, num_envs, min_episodes, max_steps,
                           trace_kind, jobs_per_env, seed, **trace_kw):
    from ..env.hip_env import HipVectorEnv
    from .bc import sjf_actions
    env = HipVectorEnv(spec, num_envs, trace_kind=trace_kind,
                       jobs_per_env=jobs_per_env, seed=seed, **trace_kw)
    env.reset()
    obs = env.b["obs"]
    mask = env.action_masks()
    K = spec.queue_visible
    gen = torch.Generator(device="cuda")
    gen.manual_seed(seed)
    noop = torch.full((num_envs,), K, dtype=torch.int64, device="cuda")
    steps = 0
    while steps < max_steps:
        valid = mask[:, :K].bool()
        if name == "sjf":
            acts = sjf_actions(obs, mask, K)
        elif name == "fifo":
            first = valid.float().argmax(dim=1)   # first True
            acts = torch.where(valid.any(dim=1), first, noop)
        else:   # random: uniform over ALL mask-valid actions (incl. noop+)
            m = mask.bool().float()
            acts = torch.multinomial(m, 1, generator=gen).squeeze(1)
        obs, rew, done, info = env.step(acts)
        mask = info["action_mask"]
        steps += 1
        if steps % 512 == 0 and env.episode_stats()["episodes"] >= min_episodes:
            break
    st = env.episode_stats()
    st["steps"] = steps
    return st
