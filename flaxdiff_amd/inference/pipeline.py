"""DiffusionInferencePipeline — reconstruct model+scheduler+sampler from a
saved run manifest and generate samples.

Behavior contract: reference /root/reference/flaxdiff/inference/pipeline.py:42-272
(from_wandb_run restore -> parse_config -> sampler cache keyed by
(class, guidance) :176-215 -> generate_samples with ema/best param selection
:248-257). Here the primary source is a LOCAL checkpoint directory written by
utils.checkpoints.CheckpointManager; `from_wandb_run` is provided when wandb
is importable and the artifact is already on disk.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple, Type

import torch

from ..samplers import DiffusionSampler, EulerAncestralSampler
from ..utils import RandomMarkovState
from .utils import load_from_checkpoint


class InferencePipeline:
    """Generic base marker (reference pipeline.py:26-40 — a thin base the
    diffusion pipeline subclasses)."""


class DiffusionInferencePipeline(InferencePipeline):
    def __init__(self, model, noise_schedule, prediction_transform,
                 input_config=None, autoencoder=None, checkpoint: Optional[dict] = None,
                 config: Optional[dict] = None, step: Optional[int] = None):
        self.model = model
        self.noise_schedule = noise_schedule
        self.prediction_transform = prediction_transform
        self.input_config = input_config
        self.autoencoder = autoencoder
        self.checkpoint = checkpoint
        self.config = config or {}
        self.step = step
        self._sampler_cache: Dict[Tuple[type, float], DiffusionSampler] = {}
        self._param_source = "params"
        # GPU inference runs the model in bf16 (the HIP kernel set is bf16;
        # sampler-side state stays fp32)
        self.compute_dtype = (torch.bfloat16 if torch.cuda.is_available()
                              else torch.float32)
        self.device = "cuda" if torch.cuda.is_available() else "cpu"

    # ------------------------------------------------------------------
    @classmethod
    def from_checkpoint(cls, checkpoint_dir: str, step: Optional[int] = None,
                        use_ema: bool = True, use_best: bool = False,
                        overrides: Optional[dict] = None,
                        device: Optional[str] = None) -> "DiffusionInferencePipeline":
        parts = load_from_checkpoint(checkpoint_dir, step, overrides)
        pipe = cls(parts["model"], parts["noise_schedule"],
                   parts["prediction_transform"], parts["input_config"],
                   parts["autoencoder"], parts["checkpoint"],
                   parts["raw_config"], parts["step"])
        pipe.load_params(use_ema=use_ema, use_best=use_best)
        dev = device or ("cuda" if torch.cuda.is_available() else "cpu")
        pipe.device = dev
        pipe.model.to(dev).eval()
        return pipe

    @classmethod
    def from_wandb_run(cls, run_id: str, project: str, entity: str,
                       **kwargs) -> "DiffusionInferencePipeline":
        """Wandb-artifact restore (reference pipeline.py:59-114). Downloads
        the checkpoint artifact then defers to from_checkpoint."""
        import wandb
        api = wandb.Api()
        run = api.run(f"{entity}/{project}/{run_id}")
        artifact = next(a for a in run.logged_artifacts()
                        if a.type == "model" or "checkpoint" in a.name)
        path = artifact.download()
        return cls.from_checkpoint(path, **kwargs)

    @classmethod
    def from_wandb_registry(cls, modelname: str, project: str,
                            entity: Optional[str] = None,
                            version: str = "latest",
                            registry: str = "wandb-registry-model",
                            **kwargs) -> "DiffusionInferencePipeline":
        """Model-registry restore (reference pipeline.py:104-143): fetch the
        named registry artifact, then defer to from_checkpoint."""
        import wandb
        api = wandb.Api()
        prefix = f"{entity}/" if entity else ""
        artifact = api.artifact(
            f"{prefix}{registry}/{modelname}:{version}", type="model")
        path = artifact.download()
        return cls.from_checkpoint(path, **kwargs)

    # ------------------------------------------------------------------
    def load_params(self, use_ema: bool = True, use_best: bool = False):
        """Select params/ema_params (optionally from best_state) and load them
        into the model (reference :248-257)."""
        if self.checkpoint is None:
            return
        state = self.checkpoint.get("state", {})
        params = state.get("params", {})
        if use_best and self.checkpoint.get("best_state") is not None:
            # best_state stores the flat master + ema buffers; rebuild params
            best = self.checkpoint["best_state"]
            flat = best["ema"] if use_ema else best["flat"]
            self._load_flat(flat, params)
            self._param_source = "best_" + ("ema" if use_ema else "params")
            return
        if use_ema and state.get("ema_params") is not None:
            self._load_flat(state["ema_params"], params)
            self._param_source = "ema"
            return
        self.model.load_state_dict(params)
        self._param_source = "params"

    def _load_flat(self, flat: torch.Tensor, params: dict):
        """Unpack a flat fp32 master buffer using the known param order
        (reverse registration order; trainer/optim.py)."""
        self.model.load_state_dict(params)  # shapes/buffers
        named = [p for _, p in self.model.named_parameters() if p.requires_grad]
        named = list(reversed(named))
        off = 0
        flat = torch.as_tensor(flat)
        with torch.no_grad():
            for p in named:
                n = p.numel()
                p.copy_(flat[off:off + n].view(p.shape).to(p.dtype))
                off += n

    # ------------------------------------------------------------------
    def get_sampler(self, sampler_class: Type[DiffusionSampler] = EulerAncestralSampler,
                    guidance_scale: float = 0.0,
                    timestep_spacing: str = "linear") -> DiffusionSampler:
        key = (sampler_class, guidance_scale)
        if key not in self._sampler_cache:
            dt = self.compute_dtype
            model = self.model
            if dt != torch.float32:
                def model_fn(x, t, *conds, _m=self.model, _dt=dt):
                    return _m(x.to(_dt), t,
                              *(c.to(_dt) for c in conds)).float()
                model = model_fn
            sampler = sampler_class(
                model=model,
                noise_schedule=self.noise_schedule,
                model_output_transform=self.prediction_transform,
                input_config=self.input_config,
                guidance_scale=guidance_scale,
                autoencoder=self.autoencoder,
                timestep_spacing=timestep_spacing)
            # hipGraph-replay the per-step model eval (bit-exact vs eager,
            # verified in tests/test_gpu_ops.py::test_graph_captured_*)
            sampler.enable_graph_capture()
            self._sampler_cache[key] = sampler
        return self._sampler_cache[key]

    # ------------------------------------------------------------------
    def generate_samples(self,
                         num_samples: int,
                         resolution: int,
                         conditioning: Optional[List] = None,
                         sampler_class: Type[DiffusionSampler] = EulerAncestralSampler,
                         guidance_scale: float = 0.0,
                         diffusion_steps: int = 50,
                         timestep_spacing: str = "linear",
                         sequence_length: Optional[int] = None,
                         start_step: Optional[int] = None,
                         rngstate: Optional[RandomMarkovState] = None,
                         progress: bool = False) -> torch.Tensor:
        sampler = self.get_sampler(sampler_class, guidance_scale, timestep_spacing)
        return sampler.generate_samples(
            num_samples=num_samples, resolution=resolution,
            sequence_length=sequence_length, diffusion_steps=diffusion_steps,
            start_step=start_step, conditioning=conditioning,
            device=self.device, rngstate=rngstate, progress=progress)
