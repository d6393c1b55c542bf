"""Non-RL baseline agents: random, FIFO first-fit, SJF, Tiresias-like LAS.

These operate on the Gym-style obs/action-mask interface and serve both as
sanity baselines for mean-JCT and as test fixtures.
"""
from __future__ import annotations

import numpy as np


class RandomAgent:
    """Uniform over valid actions (mask-aware)."""

    def __init__(self, num_actions: int, seed: int = 0):
        self.rng = np.random.default_rng(seed)
        self.num_actions = num_actions

    def act(self, obs: np.ndarray, mask: np.ndarray | None = None) -> np.ndarray:
        n = obs.shape[0]
        if mask is None:
            return self.rng.integers(0, self.num_actions, size=n)
        acts = np.empty(n, np.int64)
        for i in range(n):
            valid = np.flatnonzero(mask[i])
            acts[i] = self.rng.choice(valid)
        return acts


class FifoAgent:
    """Schedule the oldest schedulable job; else NOOP. (first-fit FIFO)"""

    def __init__(self, queue_visible: int):
        self.K = queue_visible

    def act(self, obs: np.ndarray, mask: np.ndarray) -> np.ndarray:
        n = obs.shape[0]
        acts = np.full(n, self.K, np.int64)  # noop
        for i in range(n):
            valid = np.flatnonzero(mask[i, : self.K])
            if len(valid):
                acts[i] = valid[0]
        return acts


class SjfAgent:
    """Shortest (remaining-work) schedulable job first; else NOOP.

    Reads the per-slot work feature from the observation (ENV_SPEC obs
    layout: slot k work at index 4+5k+3)."""

    def __init__(self, queue_visible: int):
        self.K = queue_visible

    def act(self, obs: np.ndarray, mask: np.ndarray) -> np.ndarray:
        n = obs.shape[0]
        acts = np.full(n, self.K, np.int64)
        for i in range(n):
            valid = np.flatnonzero(mask[i, : self.K])
            if len(valid):
                works = obs[i, 4 + 5 * valid + 3]
                acts[i] = valid[int(np.argmin(works))]
        return acts
