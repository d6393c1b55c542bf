"""Gym-style env/agent API surface.

`ClusterSchedEnv` is the single-env Gym-style facade (reset/step/spaces),
`CpuVectorEnv` a batched wrapper over N reference envs with auto-reset.
The GPU batched env (`rlgpuschedule_amd.env.hip_env.HipVectorEnv`)
implements the same VectorEnv protocol on-device.
"""
from __future__ import annotations

import numpy as np

from ..traces.synthetic import JobTrace, make_pool
from ..utils.spaces import Box, Discrete
from .cpu_ref import ClusterEnvRef
from .spec import EnvSpec


class ClusterSchedEnv:
    """Single-env Gym-style API: obs = reset(); obs, r, done, info = step(a)."""

    metadata: dict = {}

    def __init__(self, spec: EnvSpec, trace: JobTrace, job_lo: int = 0,
                 job_hi: int | None = None):
        self._impl = ClusterEnvRef(spec, trace, job_lo, job_hi)
        self.spec = spec
        self.observation_space = Box(-np.inf, np.inf, (spec.obs_dim,), np.float32)
        self.action_space = Discrete(spec.num_actions)

    def reset(self) -> np.ndarray:
        return self._impl.reset()

    def step(self, action):
        return self._impl.step(action)

    def action_mask(self) -> np.ndarray:
        return self._impl.action_mask()

    @property
    def unwrapped(self) -> ClusterEnvRef:
        return self._impl


class EpisodeStats:
    """Accumulates finished-episode stats across a vector env."""

    def __init__(self):
        self.count = 0
        self.return_sum = 0.0
        self.jct_sum_us = 0.0
        self.completed_sum = 0
        self.decisions_sum = 0

    def add(self, ep: dict) -> None:
        self.count += 1
        self.return_sum += ep["return"]
        self.jct_sum_us += ep["mean_jct_us"] * max(ep["completed"], 1)
        self.completed_sum += ep["completed"]
        self.decisions_sum += ep["decisions"]

    def mean_jct_us(self) -> float:
        return self.jct_sum_us / max(self.completed_sum, 1)

    def mean_return(self) -> float:
        return self.return_sum / max(self.count, 1)


class CpuVectorEnv:
    """Batched CPU env over per-env trace windows, with auto-reset."""

    def __init__(self, spec: EnvSpec, num_envs: int, trace_kind: str = "philly",
                 jobs_per_env: int = 128, seed: int = 0,
                 pool: JobTrace | None = None, job_lo=None, job_hi=None,
                 **trace_kw):
        self.spec = spec
        self.num_envs = num_envs
        if pool is None:
            pool, lo, hi = make_pool(trace_kind, num_envs, jobs_per_env, seed,
                                     spec.max_job_gpus, **trace_kw)
        else:
            lo, hi = job_lo, job_hi
        self.pool = pool
        self.job_lo, self.job_hi = lo, hi
        self.envs = [ClusterEnvRef(spec, pool, int(lo[e]), int(hi[e]))
                     for e in range(num_envs)]
        self.observation_space = Box(-np.inf, np.inf, (spec.obs_dim,), np.float32)
        self.action_space = Discrete(spec.num_actions)
        self.stats = EpisodeStats()

    def reset(self) -> np.ndarray:
        return np.stack([e.reset() for e in self.envs])

    def step(self, actions) -> tuple[np.ndarray, np.ndarray, np.ndarray, dict]:
        actions = np.asarray(actions)
        obs = np.empty((self.num_envs, self.spec.obs_dim), np.float32)
        rew = np.empty(self.num_envs, np.float32)
        done = np.zeros(self.num_envs, bool)
        masks = np.empty((self.num_envs, self.spec.num_actions), bool)
        for i, e in enumerate(self.envs):
            o, r, d, info = e.step(int(actions[i]))
            rew[i] = r
            done[i] = d
            if d:
                self.stats.add(info["episode"])
                o = e.reset()
            obs[i] = o
            masks[i] = e.action_mask()
        return obs, rew, done, {"action_mask": masks}

    def action_masks(self) -> np.ndarray:
        return np.stack([e.action_mask() for e in self.envs])
