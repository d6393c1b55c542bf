"""Minimal Gym-style spaces (gym/gymnasium is not in the offline image).

API-compatible subset: `Discrete(n)`, `Box(low, high, shape, dtype)` with
`.sample(rng)`, `.contains(x)`, `.shape`, `.n`.
"""
from __future__ import annotations

import numpy as np


class Space:
    pass


class Discrete(Space):
    def __init__(self, n: int):
        self.n = int(n)
        self.shape = ()
        self.dtype = np.int64

    def sample(self, rng: np.random.Generator | None = None) -> int:
        rng = rng or np.random.default_rng()
        return int(rng.integers(0, self.n))

    def contains(self, x) -> bool:
        return 0 <= int(x) < self.n

    def __repr__(self):
        return f"Discrete({self.n})"


class Box(Space):
    def __init__(self, low, high, shape=None, dtype=np.float32):
        self.low = np.asarray(low, dtype)
        self.high = np.asarray(high, dtype)
        self.shape = tuple(shape) if shape is not None else self.low.shape
        self.dtype = dtype

    def sample(self, rng: np.random.Generator | None = None) -> np.ndarray:
        rng = rng or np.random.default_rng()
        return rng.uniform(self.low, self.high, self.shape).astype(self.dtype)

    def contains(self, x) -> bool:
        x = np.asarray(x)
        return x.shape == self.shape

    def __repr__(self):
        return f"Box{self.shape}"
