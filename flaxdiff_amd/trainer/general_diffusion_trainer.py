"""GeneralDiffusionTrainer — input-config-driven trainer for image AND video
diffusion with arbitrary conditioning (reference:
/root/reference/flaxdiff/trainer/general_diffusion_trainer.py:108-727).

Adds over DiffusionTrainer: DiffusionInputConfig-driven conditioning
(any number of modalities), eval metrics with best-direction tracking, and
experiment-name templating. The wandb model-registry push of the reference
becomes a local "export best checkpoint" operation (no hard wandb dep).
"""
from __future__ import annotations

from typing import Dict, List, Optional

import torch

from .diffusion_trainer import DiffusionTrainer


def generate_modelname(dataset_name: str, noise_schedule_name: str = "",
                       architecture_name: str = "", model=None,
                       input_config=None, autoencoder=None,
                       frames_per_sample: Optional[int] = None) -> str:
    """Canonical run name (reference general_diffusion_trainer.py:32-106 —
    its richer template is commented out upstream; the live format is
    `diffusion-{dataset}-res{H}`)."""
    res = ""
    if input_config is not None:
        res = f"-res{input_config.sample_data_shape[-2]}"
    return f"diffusion-{dataset_name}{res}"


class GeneralDiffusionTrainer(DiffusionTrainer):
    def __init__(self, model, noise_schedule, model_output_transform=None, *,
                 input_config=None, eval_metrics: Optional[List] = None, **kwargs):
        self.input_config = input_config
        self.eval_metrics = eval_metrics or []
        self.best_metric_values: Dict[str, float] = {}
        null_context = None
        text_shape = (77, 768)
        if input_config is not None and input_config.conditions:
            unconds = input_config.get_unconditionals()
            if unconds:
                null_context = torch.as_tensor(unconds[0])
                text_shape = tuple(null_context.shape)
        super().__init__(model, noise_schedule, model_output_transform,
                         null_context=null_context, text_context_shape=text_shape,
                         **kwargs)

    # ------------------------------------------------------------------
    def train_step(self, batch: Dict[str, torch.Tensor]) -> Dict[str, float]:
        if self.input_config is not None:
            sample_key = self.input_config.sample_data_key
            if sample_key in batch and "image" not in batch:
                batch = dict(batch)
                batch["image"] = batch[sample_key]
            # N-modality conditioning: CFG-dropout mask shared across
            # modalities, each replaced by its own null embedding
            # (reference inputs/__init__.py:123-146 process_conditioning)
            if self.input_config.conditions and "cond_embs" not in batch \
                    and "text_emb" not in batch:
                keys = [c.conditioning_data_key or c.encoder.key
                        for c in self.input_config.conditions]
                if all(k in batch for k in keys):
                    # batch size from the SAMPLE tensor (a tokenizer
                    # BatchEncoding's len() is its key count, not B)
                    B = len(batch.get("image", batch[sample_key]))
                    self.rngs, key = self.rngs.get_random_key()
                    mask = key.bernoulli((B,), self.unconditional_prob,
                                         device="cpu").bool()
                    conds = self.input_config.process_conditioning(batch, mask)
                    batch = dict(batch)
                    batch["cond_embs"] = tuple(conds)
        # video batches [B,T,H,W,C] fold time into batch for the 2D model path
        img = batch["image"]
        if torch.is_tensor(img) and img.dim() == 5 and not hasattr(self.model, "is_video_model"):
            batch = dict(batch)
            B, T = img.shape[0], img.shape[1]
            batch["image"] = img.reshape(B * T, *img.shape[2:])
        return super().train_step(batch)

    # ------------------------------------------------------------------
    def evaluate(self, samples: torch.Tensor, conditioning=None) -> Dict[str, float]:
        """Run eval metrics + best-direction tracking (reference :420-519)."""
        results = {}
        for metric in self.eval_metrics:
            val = float(metric.function(samples, conditioning))
            results[metric.name] = val
            prev = self.best_metric_values.get(metric.name)
            better = (prev is None or
                      (val > prev if metric.higher_is_better else val < prev))
            if better:
                self.best_metric_values[metric.name] = val
            if self.dist.is_main:
                self.wandb.log({f"val/{metric.name}": val})
        return results

    # ------------------------------------------------------------------
    def make_validation_fn(self, sampler_class=None, num_samples: int = 4,
                           resolution: int = 64, diffusion_steps: int = 50,
                           guidance_scale: float = 3.0,
                           conditioning_context=None):
        """Epoch-end validation hook for fit(): EMA sample generation +
        eval metrics + wandb logging (reference general_diffusion_trainer
        :378-519; guidance 3.0 default, :375)."""
        if sampler_class is None:
            from ..samplers import EulerAncestralSampler
            sampler_class = EulerAncestralSampler

        def val_fn(trainer):
            samples = trainer.validation_sample(
                sampler_class, num_samples=num_samples, resolution=resolution,
                diffusion_steps=diffusion_steps, guidance_scale=guidance_scale,
                conditioning_context=conditioning_context, use_ema=True)
            batch = {"image": ((samples + 1) * 127.5).clamp(0, 255).byte()}
            if conditioning_context is not None:
                batch["text"] = conditioning_context
            results = trainer.evaluate(samples, batch)
            if trainer.dist.is_main:
                trainer.wandb.log({"val/samples_mean": float(samples.mean()),
                                   **{f"val/{k}": v for k, v in results.items()}})
            return results

        return val_fn

    # ------------------------------------------------------------------
    def push_to_registry(self, registry_dir: str = "./registry",
                         compare_metric: Optional[str] = None,
                         aliases: Optional[List[str]] = None):
        """Export the latest checkpoint to a local model registry (stand-in
        for the reference's wandb registry push + best-run comparison,
        general_diffusion_trainer.py:560-703).

        With compare_metric set, the push only replaces the registry entry
        when this run's best value of that metric beats the stored one
        (direction from eval_metrics); the registry keeps a meta.json with
        the winning run's metrics, and "best"/alias symlink-style dirs.
        Returns the destination path, or None when the incumbent wins.
        """
        import json
        import shutil
        from pathlib import Path
        if not self.dist.is_main:
            return None
        self.save(block=True)
        latest = self._ckpt.latest_step()
        if latest is None:
            return None

        root = Path(registry_dir) / self.name
        meta_path = root / "meta.json"
        if compare_metric is not None and meta_path.exists():
            prev = json.loads(meta_path.read_text())
            prev_val = prev.get("metrics", {}).get(compare_metric)
            cur_val = self.best_metric_values.get(compare_metric)
            higher = True
            for m in self.eval_metrics:
                if m.name == compare_metric:
                    higher = m.higher_is_better
            if prev_val is not None and cur_val is not None:
                if (cur_val <= prev_val) if higher else (cur_val >= prev_val):
                    return None        # incumbent run stays the registry best

        src = self._ckpt.dir / str(latest)
        dst = root / str(latest)
        dst.parent.mkdir(parents=True, exist_ok=True)
        if dst.exists():
            shutil.rmtree(dst)
        shutil.copytree(src, dst)
        meta = {"step": int(latest), "run": self.name,
                "metrics": dict(self.best_metric_values),
                "aliases": ["latest"] + list(aliases or [])}
        meta_path.write_text(json.dumps(meta, indent=1))
        return str(dst)

    @classmethod
    def resume_from_registry(cls, registry_dir: str, name: str):
        """Locate the registry's best checkpoint for `name` (the reference's
        wandb artifact auto-download resume becomes a local path lookup).
        Returns (checkpoint_dir, meta dict) for
        DiffusionInferencePipeline.from_checkpoint or trainer.load()."""
        import json
        from pathlib import Path
        root = Path(registry_dir) / name
        meta_path = root / "meta.json"
        if not meta_path.exists():
            return None, None
        meta = json.loads(meta_path.read_text())
        return str(root / str(meta["step"])), meta
