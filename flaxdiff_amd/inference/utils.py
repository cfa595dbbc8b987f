"""Config manifest -> components (parse_config).

Behavior contract: reference /root/reference/flaxdiff/inference/utils.py:61-268
(model class registry :120-134 incl. +2d/+hilbert/+zigzag arch-suffix
canonicalization :173-180, dtype/activation string maps :92-117, autoencoder
reconstruction :182-198, input_config deserialize / back-compat default
:200-227, noise-schedule selection edm/karras -> KarrasVE +
KarrasPredictionTransform, cosine -> Cosine + VPrediction :244-254).

The config dict is the run manifest stored with each checkpoint
(utils/checkpoints.py config.json); wandb is optional, not required.
"""
from __future__ import annotations

import json
from typing import Any, Dict, Optional

import torch
import torch.nn.functional as F

from ..inputs import ConditionalInputConfig, DiffusionInputConfig
from ..inputs.encoders import get_text_encoder
from ..models import (HierarchicalMMDiT, HybridSSMAttentionDiT, SimpleDiT,
                      SimpleMMDiT, SimpleUDiT, Unet, UViT)
from ..predictors import KarrasPredictionTransform, VPredictionTransform
from ..schedulers import CosineNoiseScheduler, KarrasVENoiseScheduler

DTYPE_MAP = {
    "bfloat16": torch.bfloat16,
    "float32": torch.float32,
    "float16": torch.float16,
    "torch.bfloat16": torch.bfloat16,
    "torch.float32": torch.float32,
    # reference (JAX) config strings round-trip too
    "jax.numpy.float32": torch.float32,
    "jax.numpy.bfloat16": torch.bfloat16,
    "None": None,
    None: None,
}

ACTIVATION_MAP = {
    "swish": F.silu,
    "silu": F.silu,
    "jax._src.nn.functions.silu": F.silu,
    "mish": F.mish,
    "gelu": F.gelu,
}

MODEL_CLASSES = {
    "unet": Unet,
    "uvit": UViT,
    "simple_dit": SimpleDiT,
    "simple_mmdit": SimpleMMDiT,
    "simple_udit": SimpleUDiT,
    "hierarchical_mmdit": HierarchicalMMDiT,
    "hybrid_dit": HybridSSMAttentionDiT,
}

# constructor args that are JAX-only in reference configs — dropped on load
_IGNORED_MODEL_KEYS = {"dtype", "precision", "use_flash_attention", "use_remat",
                       "dropout_rate", "force_fp32_for_softmax"}


def map_nested_config(config: Dict[str, Any]) -> Dict[str, Any]:
    out = {}
    for key, value in config.items():
        if isinstance(value, dict):
            out[key] = map_nested_config(value)
        elif isinstance(value, list):
            out[key] = [map_nested_config(v) if isinstance(v, dict) else v
                        for v in value]
        elif isinstance(value, str):
            if value in DTYPE_MAP:
                out[key] = DTYPE_MAP[value]
            elif value in ACTIVATION_MAP:
                out[key] = ACTIVATION_MAP[value]
            elif value == "None":
                out[key] = None
            else:
                out[key] = value
        else:
            out[key] = value
    return out


def canonicalize_architecture(architecture: str) -> str:
    for suffix in ("+2d", "+hilbert", "+zigzag"):
        architecture = architecture.replace(suffix, "")
    return architecture


def parse_config(config: Dict[str, Any], overrides: Optional[Dict] = None
                 ) -> Dict[str, Any]:
    conf = dict(config)
    if overrides:
        if "arguments" in conf:
            conf["arguments"] = {**conf["arguments"], **overrides}
        for key in overrides:
            if key in conf:
                conf[key] = overrides[key]

    args = conf.get("arguments", {})
    model_config = dict(conf.get("model", {}))
    architecture = conf.get("architecture", args.get("architecture", "unet"))
    architecture = canonicalize_architecture(str(architecture))

    # autoencoder
    autoencoder = None
    autoencoder_name = conf.get("autoencoder", args.get("autoencoder"))
    if autoencoder_name:
        opts = conf.get("autoencoder_opts", args.get("autoencoder_opts", "{}"))
        if isinstance(opts, str):
            opts = json.loads(opts)
        from ..models.autoencoder import get_autoencoder
        autoencoder = get_autoencoder(autoencoder_name, **map_nested_config(opts))

    # input config (back-compat default mirrors reference :200-227)
    input_config = conf.get("input_config")
    if input_config is None:
        image_size = args.get("image_size", 128)
        encoder = get_text_encoder()
        input_config = DiffusionInputConfig(
            sample_data_key="image",
            sample_data_shape=(image_size, image_size, 3),
            conditions=[ConditionalInputConfig(
                encoder=encoder, conditioning_data_key="text",
                pretokenized=True, unconditional_input="",
                model_key_override="textcontext")])
    elif isinstance(input_config, dict):
        input_config = DiffusionInputConfig.deserialize(input_config)

    model_kwargs = {k: v for k, v in map_nested_config(model_config).items()
                    if k not in _IGNORED_MODEL_KEYS}
    model_class = MODEL_CLASSES.get(architecture)
    if model_class is None:
        raise ValueError(f"Unknown architecture: {architecture}. Supported: "
                         f"{', '.join(MODEL_CLASSES)}")
    model = model_class(**model_kwargs)

    # noise schedule selection (reference :244-254)
    noise_schedule_type = conf.get("noise_schedule",
                                   args.get("noise_schedule", "edm"))
    if noise_schedule_type in ("edm", "karras"):
        noise_schedule = KarrasVENoiseScheduler(1, sigma_max=80, rho=7,
                                                sigma_data=0.5)
        prediction_transform = KarrasPredictionTransform(
            sigma_data=noise_schedule.sigma_data)
    elif noise_schedule_type == "cosine":
        noise_schedule = CosineNoiseScheduler(1000, beta_end=1)
        prediction_transform = VPredictionTransform()
    else:
        raise ValueError(f"Unknown noise schedule: {noise_schedule_type}")

    return {
        "model": model,
        "model_config": model_kwargs,
        "architecture": architecture,
        "autoencoder": autoencoder,
        "noise_schedule": noise_schedule,
        "prediction_transform": prediction_transform,
        "input_config": input_config,
        "raw_config": conf,
    }


def load_from_checkpoint(checkpoint_dir: str, step: Optional[int] = None,
                         overrides: Optional[Dict] = None) -> Dict[str, Any]:
    """Local-checkpoint equivalent of the reference's wandb artifact load
    (inference/utils.py:270-349): reads config.json + state.pt from the
    step-numbered directory layout of utils.checkpoints.CheckpointManager."""
    from pathlib import Path
    base = Path(checkpoint_dir)
    mgr_dir = base
    steps = sorted(int(d.name) for d in mgr_dir.iterdir()
                   if d.is_dir() and d.name.isdigit())
    if not steps:
        raise FileNotFoundError(f"no step dirs under {checkpoint_dir}")
    step = step if step is not None else steps[-1]
    with open(mgr_dir / str(step) / "config.json") as f:
        config = json.load(f)
    payload = torch.load(mgr_dir / str(step) / "state.pt", map_location="cpu",
                         weights_only=False)
    parsed = parse_config(config, overrides)
    return {**parsed, "checkpoint": payload, "step": step}
