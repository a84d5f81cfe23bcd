"""First-party observation/action space primitives.

The reference framework leans on Gymnasium's ``spaces`` module
(``gymnasium.spaces.Box/Discrete/...``; see reference
``agilerl/utils/evolvable_networks.py`` and ``agilerl/typing.py``). This
repo is MI355X-native and self-contained: spaces are small, torch-friendly
dataclass-like objects with ``sample()`` helpers used by the first-party
batched vector envs and by the network builders.

Supported (parity with reference §2.4 of SURVEY.md): Box, Discrete,
MultiDiscrete, MultiBinary, and nested Dict / Tuple observation spaces.
"""

from __future__ import annotations

from collections import OrderedDict
from typing import Any, Iterable, Optional, Sequence, Union

import numpy as np

__all__ = [
    "Space",
    "Box",
    "Discrete",
    "MultiDiscrete",
    "MultiBinary",
    "DictSpace",
    "TupleSpace",
    "is_image_space",
    "is_vector_space",
    "flatdim",
    "space_shape",
]


class Space:
    """Base class for all spaces."""

    shape: tuple
    dtype: np.dtype

    def __init__(self, shape: Optional[Sequence[int]] = None, dtype: Any = np.float32, seed: Optional[int] = None):
        self.shape = tuple(shape) if shape is not None else ()
        self.dtype = np.dtype(dtype)
        self._rng = np.random.default_rng(seed)

    def seed(self, seed: Optional[int] = None) -> None:
        self._rng = np.random.default_rng(seed)

    def sample(self):  # pragma: no cover - overridden
        raise NotImplementedError

    def contains(self, x) -> bool:  # pragma: no cover - overridden
        raise NotImplementedError

    def __contains__(self, x) -> bool:
        return self.contains(x)

    def __eq__(self, other) -> bool:
        return type(self) is type(other) and self.shape == other.shape and self.dtype == other.dtype

    def __repr__(self) -> str:  # pragma: no cover
        return f"{type(self).__name__}(shape={self.shape}, dtype={self.dtype})"


class Box(Space):
    """Continuous (possibly bounded) n-dimensional space."""

    def __init__(
        self,
        low: Union[float, np.ndarray],
        high: Union[float, np.ndarray],
        shape: Optional[Sequence[int]] = None,
        dtype: Any = np.float32,
        seed: Optional[int] = None,
    ):
        if shape is None:
            if np.isscalar(low) and np.isscalar(high):
                shape = (1,)
            else:
                shape = np.broadcast(np.asarray(low), np.asarray(high)).shape
        super().__init__(shape, dtype, seed)
        self.low = np.broadcast_to(np.asarray(low, dtype=self.dtype), self.shape).copy()
        self.high = np.broadcast_to(np.asarray(high, dtype=self.dtype), self.shape).copy()

    @property
    def bounded_below(self) -> np.ndarray:
        return np.isfinite(self.low)

    @property
    def bounded_above(self) -> np.ndarray:
        return np.isfinite(self.high)

    def sample(self) -> np.ndarray:
        lo = np.where(np.isfinite(self.low), self.low, -1.0)
        hi = np.where(np.isfinite(self.high), self.high, 1.0)
        return self._rng.uniform(lo, hi).astype(self.dtype)

    def contains(self, x) -> bool:
        x = np.asarray(x)
        return x.shape == self.shape and bool(np.all(x >= self.low - 1e-6) and np.all(x <= self.high + 1e-6))

    def __eq__(self, other) -> bool:
        return (
            isinstance(other, Box)
            and self.shape == other.shape
            and np.allclose(self.low, other.low)
            and np.allclose(self.high, other.high)
        )


class Discrete(Space):
    """{0, 1, ..., n-1}."""

    def __init__(self, n: int, seed: Optional[int] = None, start: int = 0):
        super().__init__((), np.int64, seed)
        self.n = int(n)
        self.start = int(start)

    def sample(self) -> int:
        return int(self._rng.integers(self.start, self.start + self.n))

    def contains(self, x) -> bool:
        try:
            xi = int(x)
        except (TypeError, ValueError):
            return False
        return self.start <= xi < self.start + self.n

    def __eq__(self, other) -> bool:
        return isinstance(other, Discrete) and self.n == other.n and self.start == other.start


class MultiDiscrete(Space):
    """Vector of discrete sub-spaces with per-dim cardinalities ``nvec``."""

    def __init__(self, nvec: Sequence[int], seed: Optional[int] = None):
        nvec = np.asarray(nvec, dtype=np.int64)
        super().__init__(nvec.shape, np.int64, seed)
        self.nvec = nvec

    def sample(self) -> np.ndarray:
        return (self._rng.random(self.nvec.shape) * self.nvec).astype(np.int64)

    def contains(self, x) -> bool:
        x = np.asarray(x)
        return x.shape == self.shape and bool(np.all(x >= 0) and np.all(x < self.nvec))

    def __eq__(self, other) -> bool:
        return isinstance(other, MultiDiscrete) and np.array_equal(self.nvec, other.nvec)


class MultiBinary(Space):
    """{0,1}^n."""

    def __init__(self, n: Union[int, Sequence[int]], seed: Optional[int] = None):
        shape = (int(n),) if np.isscalar(n) else tuple(n)
        super().__init__(shape, np.int8, seed)
        self.n = n if np.isscalar(n) else tuple(n)

    def sample(self) -> np.ndarray:
        return self._rng.integers(0, 2, size=self.shape, dtype=np.int8)

    def contains(self, x) -> bool:
        x = np.asarray(x)
        return x.shape == self.shape and bool(np.all((x == 0) | (x == 1)))

    def __eq__(self, other) -> bool:
        return isinstance(other, MultiBinary) and self.shape == other.shape


class DictSpace(Space):
    """Ordered mapping of named sub-spaces."""

    def __init__(self, spaces: Union[dict, OrderedDict, None] = None, seed: Optional[int] = None, **kwargs):
        super().__init__((), np.object_, seed)
        if spaces is None:
            spaces = kwargs
        self.spaces = OrderedDict(spaces)

    def sample(self) -> dict:
        return OrderedDict((k, s.sample()) for k, s in self.spaces.items())

    def contains(self, x) -> bool:
        return isinstance(x, dict) and all(k in x and s.contains(x[k]) for k, s in self.spaces.items())

    def items(self):
        return self.spaces.items()

    def keys(self):
        return self.spaces.keys()

    def values(self):
        return self.spaces.values()

    def __getitem__(self, key):
        return self.spaces[key]

    def __iter__(self):
        return iter(self.spaces)

    def __len__(self):
        return len(self.spaces)

    def __eq__(self, other) -> bool:
        return isinstance(other, DictSpace) and self.spaces == other.spaces

    def __repr__(self) -> str:  # pragma: no cover
        return f"DictSpace({dict(self.spaces)})"


class TupleSpace(Space):
    """Fixed-length tuple of sub-spaces."""

    def __init__(self, spaces: Iterable[Space], seed: Optional[int] = None):
        super().__init__((), np.object_, seed)
        self.spaces = tuple(spaces)

    def sample(self) -> tuple:
        return tuple(s.sample() for s in self.spaces)

    def contains(self, x) -> bool:
        return isinstance(x, (tuple, list)) and len(x) == len(self.spaces) and all(
            s.contains(xi) for s, xi in zip(self.spaces, x)
        )

    def __getitem__(self, i):
        return self.spaces[i]

    def __iter__(self):
        return iter(self.spaces)

    def __len__(self):
        return len(self.spaces)

    def __eq__(self, other) -> bool:
        return isinstance(other, TupleSpace) and self.spaces == other.spaces

    def __repr__(self) -> str:  # pragma: no cover
        return f"TupleSpace({list(self.spaces)})"


# ---------------------------------------------------------------------------
# Space predicates / helpers (reference parity: agilerl/utils/evolvable_networks.py)
# ---------------------------------------------------------------------------

def is_image_space(space: Space) -> bool:
    """3D Box observations are treated as images (C,H,W)."""
    return isinstance(space, Box) and len(space.shape) == 3


def is_vector_space(space: Space) -> bool:
    return isinstance(space, (Discrete, MultiDiscrete, MultiBinary)) or (
        isinstance(space, Box) and len(space.shape) in (0, 1)
    )


def space_shape(space: Space) -> tuple:
    """Shape of a single observation drawn from ``space``."""
    if isinstance(space, Discrete):
        return (space.n,)  # one-hot width for network input sizing
    if isinstance(space, MultiDiscrete):
        return (int(np.sum(space.nvec)),)
    if isinstance(space, MultiBinary):
        return space.shape
    if isinstance(space, Box):
        return space.shape
    raise TypeError(f"space_shape undefined for {type(space)}")


def flatdim(space: Space) -> int:
    """Flattened input width a network sees for one observation."""
    if isinstance(space, (DictSpace,)):
        return sum(flatdim(s) for s in space.spaces.values())
    if isinstance(space, TupleSpace):
        return sum(flatdim(s) for s in space.spaces)
    return int(np.prod(space_shape(space)))
