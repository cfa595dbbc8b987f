"""Conditioning input configs — declarative description of the sample tensor
and each conditioning modality.

Contract: reference /root/reference/flaxdiff/inputs/__init__.py:16-173
(ConditionalInputConfig / DiffusionInputConfig with serialize/deserialize for
checkpoint-config round-trips).
"""
from __future__ import annotations

import dataclasses
from typing import Any, Dict, List, Optional, Tuple

import torch

from .encoders import ConditioningEncoder, CLIPTextEncoder, DummyTextEncoder, ENCODER_REGISTRY


@dataclasses.dataclass
class ConditionalInputConfig:
    """Per-condition config (reference inputs/__init__.py:16-74)."""

    encoder: ConditioningEncoder
    conditioning_data_key: Optional[str] = None
    pretokenized: bool = False
    unconditional_input: Any = ""
    model_key_override: Optional[str] = None

    def __post_init__(self):
        # cache the unconditional embedding once (reference :26-31)
        emb = self.encoder([self.unconditional_input])
        self.unconditional = torch.as_tensor(emb[0])

    def serialize(self) -> dict:
        return {
            "encoder": self.encoder.serialize(),
            "conditioning_data_key": self.conditioning_data_key,
            "pretokenized": self.pretokenized,
            "unconditional_input": self.unconditional_input,
            "model_key_override": self.model_key_override,
        }

    @classmethod
    def deserialize(cls, data: dict) -> "ConditionalInputConfig":
        enc = ConditioningEncoder.deserialize(data["encoder"])
        return cls(encoder=enc,
                   conditioning_data_key=data.get("conditioning_data_key"),
                   pretokenized=data.get("pretokenized", False),
                   unconditional_input=data.get("unconditional_input", ""),
                   model_key_override=data.get("model_key_override"))


@dataclasses.dataclass
class DiffusionInputConfig:
    """Sample key/shape + conditions (reference inputs/__init__.py:77-173)."""

    sample_data_key: str
    sample_data_shape: Tuple[int, ...]
    conditions: List[ConditionalInputConfig] = dataclasses.field(default_factory=list)

    def get_input_shapes(self, autoencoder=None, sample_model_key: str = "x") -> Dict[str, Tuple]:
        shape = tuple(self.sample_data_shape)
        if autoencoder is not None:
            # VAE-aware: latent spatial dims (reference :96-100)
            h, w, c = shape
            f = autoencoder.downscale_factor
            shape = (h // f, w // f, autoencoder.latent_channels)
        shapes = {sample_model_key: shape}
        for cond in self.conditions:
            key = cond.model_key_override or cond.encoder.key
            shapes[key] = tuple(cond.unconditional.shape)
        return shapes

    def get_unconditionals(self) -> List[torch.Tensor]:
        return [c.unconditional for c in self.conditions]

    def process_conditioning(self, batch: Dict[str, Any], uncond_mask: torch.Tensor
                             ) -> List[torch.Tensor]:
        """Masked replacement of conditioning with the null embedding
        (reference :123-146)."""
        outs = []
        for cond in self.conditions:
            key = cond.conditioning_data_key or cond.encoder.key
            data = batch[key]
            if cond.pretokenized:
                emb = cond.encoder.encode_from_tokens(data)
            else:
                emb = cond.encoder(data)
            emb = torch.as_tensor(emb)
            null = cond.unconditional.to(emb.device, emb.dtype)
            mask = uncond_mask.to(emb.device)
            shape = (-1,) + (1,) * (emb.dim() - 1)
            emb = torch.where(mask.reshape(shape), null.unsqueeze(0), emb)
            outs.append(emb)
        return outs

    def encode_conditions(self, conditioning: List, device=None, dtype=None) -> Tuple:
        """Encode raw conditioning values (sampler path, reference
        samplers/common.py:315-349)."""
        separated: Dict[str, List] = {c.encoder.key: [] for c in self.conditions}
        for vals in conditioning:
            if isinstance(vals, (tuple, list)):
                for cond, val in zip(self.conditions, vals):
                    separated[cond.encoder.key].append(val)
            elif isinstance(vals, dict):
                for cond in self.conditions:
                    separated[cond.encoder.key].append(vals[cond.encoder.key])
            else:
                for cond in self.conditions:
                    separated[cond.encoder.key].append(vals)
        finals = []
        for cond in self.conditions:
            emb = torch.as_tensor(cond.encoder(separated[cond.encoder.key]))
            if device is not None:
                emb = emb.to(device)
            if dtype is not None:
                emb = emb.to(dtype)
            finals.append(emb)
        return tuple(finals)

    def serialize(self) -> dict:
        return {
            "sample_data_key": self.sample_data_key,
            "sample_data_shape": list(self.sample_data_shape),
            "conditions": [c.serialize() for c in self.conditions],
        }

    @classmethod
    def deserialize(cls, data: dict) -> "DiffusionInputConfig":
        return cls(
            sample_data_key=data["sample_data_key"],
            sample_data_shape=tuple(data["sample_data_shape"]),
            conditions=[ConditionalInputConfig.deserialize(c)
                        for c in data.get("conditions", [])],
        )


__all__ = ["ConditionalInputConfig", "DiffusionInputConfig", "ConditioningEncoder",
           "CLIPTextEncoder", "DummyTextEncoder", "ENCODER_REGISTRY"]
