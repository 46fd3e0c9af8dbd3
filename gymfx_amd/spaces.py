"""Gymnasium spaces — thin import layer.

When gymnasium is installed, its real spaces/Env are used (full API
compatibility with reference consumers).  This image has no gymnasium, so a
minimal API-compatible shim (Discrete / Box / Dict spaces + Env base) is
provided: enough for agents that sample actions, check shapes/dtypes and
flatten Dict observations.
"""
from __future__ import annotations

try:  # pragma: no cover - exercised only when gymnasium is installed
    import gymnasium as _gym
    from gymnasium import spaces as _spaces

    Env = _gym.Env
    Discrete = _spaces.Discrete
    Box = _spaces.Box
    Dict = _spaces.Dict
    HAVE_GYMNASIUM = True
except ImportError:
    HAVE_GYMNASIUM = False

    import numpy as np

    class Space:
        def __init__(self, shape=None, dtype=None):
            self.shape = tuple(shape) if shape is not None else None
            self.dtype = np.dtype(dtype) if dtype is not None else None
            self._rng = np.random.default_rng()

        def seed(self, seed=None):
            self._rng = np.random.default_rng(seed)
            return [seed]

        def contains(self, x) -> bool:  # pragma: no cover - simple shim
            return True

    class Discrete(Space):
        def __init__(self, n: int, seed=None, start: int = 0):
            super().__init__(shape=(), dtype=np.int64)
            self.n = int(n)
            self.start = int(start)
            if seed is not None:
                self.seed(seed)

        def sample(self):
            return int(self._rng.integers(self.start, self.start + self.n))

        def contains(self, x) -> bool:
            try:
                xi = int(x)
            except (TypeError, ValueError):
                return False
            return self.start <= xi < self.start + self.n

        def __repr__(self):
            return f"Discrete({self.n})"

    class Box(Space):
        def __init__(self, low, high, shape=None, dtype=np.float32, seed=None):
            if shape is None:
                shape = np.broadcast(np.asarray(low), np.asarray(high)).shape
            super().__init__(shape=shape, dtype=dtype)
            self.low = np.broadcast_to(np.asarray(low, dtype=dtype), shape).copy()
            self.high = np.broadcast_to(np.asarray(high, dtype=dtype), shape).copy()
            if seed is not None:
                self.seed(seed)

        def sample(self):
            lo = np.where(np.isfinite(self.low), self.low, -1.0)
            hi = np.where(np.isfinite(self.high), self.high, 1.0)
            return self._rng.uniform(lo, hi).astype(self.dtype)

        def contains(self, x) -> bool:
            arr = np.asarray(x)
            return (
                arr.shape == self.shape
                and bool(np.all(arr >= self.low - 1e-6))
                and bool(np.all(arr <= self.high + 1e-6))
            )

        def __repr__(self):
            return f"Box(shape={self.shape}, dtype={self.dtype})"

    class Dict(Space):
        def __init__(self, spaces=None, seed=None, **kwargs):
            super().__init__()
            if spaces is None:
                spaces = kwargs
            self.spaces = dict(spaces)
            if seed is not None:
                self.seed(seed)

        def seed(self, seed=None):
            # gymnasium semantics: seeding a Dict seeds every subspace
            # with a derived seed, so sample() is deterministic
            rng = np.random.default_rng(seed)
            seeds = [seed]
            for sub in self.spaces.values():
                sub_seed = int(rng.integers(0, 2**31 - 1))
                seeds.extend(sub.seed(sub_seed))
            return seeds

        def sample(self):
            return {k: s.sample() for k, s in self.spaces.items()}

        def contains(self, x) -> bool:
            if not isinstance(x, dict):
                return False
            return all(k in x and s.contains(x[k]) for k, s in self.spaces.items())

        def __getitem__(self, key):
            return self.spaces[key]

        def keys(self):
            return self.spaces.keys()

        def items(self):
            return self.spaces.items()

        def __repr__(self):
            return f"Dict({list(self.spaces)})"

    class Env:
        """Minimal gymnasium.Env-compatible base."""

        metadata: dict = {"render_modes": []}
        action_space: Space
        observation_space: Space

        def reset(self, *, seed=None, options=None):
            if seed is not None:
                self._np_random = np.random.default_rng(seed)
            return None, {}

        def step(self, action):
            raise NotImplementedError

        def render(self):
            return None

        def close(self):
            pass
