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


def clip_images(images: torch.Tensor, clip_min: float = -1.0,
                clip_max: float = 1.0) -> torch.Tensor:
    """Reference: flaxdiff/utils.py clip_images."""
    return torch.clamp(images, clip_min, clip_max)


def get_coeff_shapes_tuple(array: torch.Tensor):
    """Reference: schedulers/common.py:6-8 — broadcast shape (-1, 1, 1, ...)."""
    return (-1,) + (1,) * (array.ndim - 1)


def denormalize_images(images: torch.Tensor, target_type=torch.uint8,
                       source_range=(-1.0, 1.0), target_range=(0, 255)) -> torch.Tensor:
    """Map images from a normalized range to a display range
    (reference: flaxdiff/utils.py:114-135)."""
    s_lo, s_hi = source_range
    t_lo, t_hi = target_range
    x = (images - s_lo) / (s_hi - s_lo)
    x = x * (t_hi - t_lo) + t_lo
    if target_type in (torch.uint8, torch.int32, torch.int64):
        x = x.round().clamp(t_lo, t_hi)
    return x.to(target_type)


def normalize_images(images: torch.Tensor) -> torch.Tensor:
    """uint8 [0,255] -> fp32 [-1,1] (the trainer's input normalization,
    reference diffusion_trainer.py:171)."""
    return images.float() / 127.5 - 1.0


def get_latest_checkpoint(checkpoint_path: str) -> str:
    """Latest step subdirectory of a checkpoint dir
    (reference: flaxdiff/utils.py:84-91)."""
    import os
    steps = sorted(int(d) for d in os.listdir(checkpoint_path)
                   if d.isdigit())
    if not steps:
        raise FileNotFoundError(f"no checkpoints under {checkpoint_path}")
    return os.path.join(checkpoint_path, str(steps[-1]))


def serialize_model(model) -> dict:
    """JSON-safe dict of a module's constructor config
    (reference: flaxdiff/utils.py:60-82). Uses the module's `config` attr if
    present, else its non-private, non-tensor attributes."""
    def _clean(v):
        if isinstance(v, dict):
            return {k: _clean(x) for k, x in v.items()}
        if isinstance(v, (list, tuple)):
            return [_clean(x) for x in v]
        if isinstance(v, (str, int, float, bool)) or v is None:
            return v
        if callable(v):
            return getattr(v, "__name__", str(v))
        return str(v)
    if hasattr(model, "config") and isinstance(model.config, dict):
        return _clean(model.config)
    src = {k: v for k, v in vars(model).items()
           if not k.startswith("_") and not torch.is_tensor(v)}
    return _clean(src)


class AutoTextTokenizer:
    """CLIP tokenizer to the trainer's batch contract
    (reference: flaxdiff/utils.py:239-258)."""

    def __init__(self, tensor_type: str = "pt",
                 modelname: str = "openai/clip-vit-large-patch14"):
        from transformers import AutoTokenizer
        self.tokenizer = AutoTokenizer.from_pretrained(modelname)
        self.tensor_type = tensor_type

    def __call__(self, inputs):
        tokens = self.tokenizer(
            inputs, padding="max_length",
            max_length=self.tokenizer.model_max_length, truncation=True,
            return_tensors=self.tensor_type)
        return {"input_ids": tokens["input_ids"],
                "attention_mask": tokens["attention_mask"],
                "caption": inputs}

    def __repr__(self):
        return self.__class__.__name__ + "()"


def defaultTextEncodeModel(modelname: str = "openai/clip-vit-large-patch14"):
    """Default CLIP text encoder (reference: flaxdiff/utils.py:261-263)."""
    from ..inputs.encoders import CLIPTextEncoder
    return CLIPTextEncoder.from_modelname(modelname=modelname)
