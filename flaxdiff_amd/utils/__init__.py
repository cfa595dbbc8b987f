"""Core utilities: deterministic RNG chain, image helpers.

MI355X-native re-design of the reference's RandomMarkovState
(/root/reference/flaxdiff/utils.py) on torch.Generator instead of JAX PRNG.
The semantics preserved: a *functional* random-state chain — every draw
returns a new state, so training is reproducible and per-rank streams can be
derived by fold_in (reference: jax.random.fold_in, diffusion_trainer.py:158).
"""
from __future__ import annotations

import dataclasses
from typing import Tuple

import torch


def _splitmix64(x: int) -> int:
    """SplitMix64 mixing — stateless, high-quality seed derivation."""
    x = (x + 0x9E3779B97F4A7C15) & 0xFFFFFFFFFFFFFFFF
    z = x
    z = ((z ^ (z >> 30)) * 0xBF58476D1CE4E5B9) & 0xFFFFFFFFFFFFFFFF
    z = ((z ^ (z >> 27)) * 0x94D049BB133111EB) & 0xFFFFFFFFFFFFFFFF
    return (z ^ (z >> 31)) & 0xFFFFFFFFFFFFFFFF


@dataclasses.dataclass(frozen=True)
class RandomKey:
    """A single-use random key (analog of a JAX PRNGKey)."""

    seed: int

    def generator(self, device=None) -> torch.Generator:
        g = torch.Generator(device=device if device is not None else "cpu")
        g.manual_seed(self.seed & 0x7FFFFFFFFFFFFFFF)
        return g

    def fold_in(self, data: int) -> "RandomKey":
        return RandomKey(_splitmix64(self.seed ^ _splitmix64(data + 0x1234567)))

    # -- convenience draws ---------------------------------------------------
    def normal(self, shape, dtype=torch.float32, device=None) -> torch.Tensor:
        dev = device if device is not None else "cpu"
        g = self.generator(dev)
        return torch.randn(*shape, generator=g, dtype=torch.float32, device=dev).to(dtype)

    def uniform(self, shape, low=0.0, high=1.0, dtype=torch.float32, device=None) -> torch.Tensor:
        dev = device if device is not None else "cpu"
        g = self.generator(dev)
        u = torch.rand(*shape, generator=g, dtype=torch.float32, device=dev)
        return (low + (high - low) * u).to(dtype)

    def randint(self, shape, low, high, device=None) -> torch.Tensor:
        dev = device if device is not None else "cpu"
        g = self.generator(dev)
        return torch.randint(low, high, tuple(shape), generator=g, device=dev)

    def bernoulli(self, shape, p, device=None) -> torch.Tensor:
        dev = device if device is not None else "cpu"
        g = self.generator(dev)
        return torch.rand(*shape, generator=g, device=dev) < p


@dataclasses.dataclass(frozen=True)
class MarkovState:
    pass


@dataclasses.dataclass(frozen=True)
class RandomMarkovState(MarkovState):
    """Functional random-state chain (reference: flaxdiff/utils.py RandomMarkovState).

    `get_random_key` returns (next_state, key); keys are single-use.
    """

    seed: int = 42

    def get_random_key(self) -> Tuple["RandomMarkovState", RandomKey]:
        nxt = _splitmix64(self.seed)
        key = RandomKey(_splitmix64(nxt ^ 0xA5A5A5A5A5A5A5A5))
        return RandomMarkovState(nxt), key

    def fold_in(self, data: int) -> "RandomMarkovState":
        return RandomMarkovState(_splitmix64(self.seed ^ _splitmix64(data + 0xDEADBEEF)))


def clip_images(images: torch.Tensor, clip_min: float = -1.0, clip_max: float = 1.0) -> torch.Tensor:
    """Reference: flaxdiff/utils.py clip_images."""
    return torch.clamp(images, clip_min, clip_max)


def get_coeff_shapes_tuple(array: torch.Tensor):
    """Reference: schedulers/common.py:6-8 — broadcast shape (-1, 1, 1, ...)."""
    return (-1,) + (1,) * (array.ndim - 1)
