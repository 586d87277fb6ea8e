"""Minimal gym-compatible spaces (gym is not a dependency of this rebuild;
interface mirrors ``gym.spaces`` as used by the reference envs)."""
from __future__ import annotations

import numpy as np


class Space:
    def contains(self, x) -> bool:
        raise NotImplementedError

    def sample(self):
        raise NotImplementedError


class Discrete(Space):
    def __init__(self, n: int):
        self.n = int(n)

    def contains(self, x) -> bool:
        return 0 <= int(x) < self.n

    def sample(self):
        return int(np.random.randint(self.n))

    def __repr__(self):
        return f"Discrete({self.n})"


class Box(Space):
    def __init__(self, low, high, shape=None, dtype=np.float32):
        self.low, self.high = low, high
        self.shape = tuple(shape) if shape is not None else None
        self.dtype = dtype

    def contains(self, x) -> bool:
        x = np.asarray(x)
        if self.shape is not None and x.shape != self.shape:
            return False
        return bool(np.all(x >= self.low) and np.all(x <= self.high))

    def sample(self):
        return np.random.uniform(self.low, self.high, size=self.shape).astype(self.dtype)

    def __repr__(self):
        return f"Box(low={self.low}, high={self.high}, shape={self.shape})"


class Dict(Space):
    def __init__(self, spaces: dict):
        self.spaces = dict(spaces)

    def __getitem__(self, k):
        return self.spaces[k]

    def items(self):
        return self.spaces.items()

    def contains(self, x) -> bool:
        return all(k in x and s.contains(x[k]) for k, s in self.spaces.items())

    def sample(self):
        return {k: s.sample() for k, s in self.spaces.items()}

    def __repr__(self):
        return f"Dict({self.spaces})"
