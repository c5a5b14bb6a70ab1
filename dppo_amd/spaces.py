"""Minimal gym-compatible space classes.

The reference depends on gym.spaces only for dispatch in make_pdtype
(reference Others/distributions.py:231-243) and for action_space.n /
.shape at call sites (Worker.py:149, PPO.py:14-15).  gym is not
installable in this environment, so these four classes provide the same
duck-typed surface.
"""

from __future__ import annotations

import numpy as np


class Space:
    pass


class Box(Space):
    """Continuous space; only shape/low/high/dtype are consumed."""

    def __init__(self, low=-np.inf, high=np.inf, shape=None, dtype=np.float32):
        if shape is None:
            raise ValueError("Box requires an explicit shape")
        self.shape = tuple(shape)
        self.dtype = np.dtype(dtype)
        self.low = np.full(self.shape, low, dtype=self.dtype)
        self.high = np.full(self.shape, high, dtype=self.dtype)

    def __repr__(self):
        return f"Box{self.shape}"

    def __eq__(self, other):
        return isinstance(other, Box) and self.shape == other.shape

    def __hash__(self):
        return hash(("Box", self.shape))


class Discrete(Space):
    def __init__(self, n: int):
        self.n = int(n)
        self.shape = ()
        self.dtype = np.dtype(np.int64)

    def __repr__(self):
        return f"Discrete({self.n})"

    def __eq__(self, other):
        return isinstance(other, Discrete) and self.n == other.n

    def __hash__(self):
        return hash(("Discrete", self.n))


class MultiDiscrete(Space):
    def __init__(self, nvec):
        self.nvec = np.asarray(nvec, dtype=np.int64)
        self.shape = self.nvec.shape
        self.dtype = np.dtype(np.int64)

    def __repr__(self):
        return f"MultiDiscrete({self.nvec.tolist()})"


class MultiBinary(Space):
    def __init__(self, n: int):
        self.n = int(n)
        self.shape = (self.n,)
        self.dtype = np.dtype(np.int8)

    def __repr__(self):
        return f"MultiBinary({self.n})"
