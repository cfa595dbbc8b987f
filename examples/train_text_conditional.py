#!/usr/bin/env python3
"""Text-conditional diffusion with CFG dropout — the reference's main
text-to-image flow (GeneralDiffusionTrainer + DiffusionInputConfig).

Uses the offline DummyTextEncoder by default; pass --clip to use CLIP-L/14
(needs cached HF weights)."""
import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from flaxdiff_amd.inputs import (ConditionalInputConfig, DiffusionInputConfig,
                                 DummyTextEncoder)
from flaxdiff_amd.metrics import EvaluationMetric
from flaxdiff_amd.models import Unet
from flaxdiff_amd.predictors import KarrasPredictionTransform
from flaxdiff_amd.samplers import EulerAncestralSampler
from flaxdiff_amd.schedulers import EDMNoiseScheduler
from flaxdiff_amd.trainer import GeneralDiffusionTrainer


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--size", type=int, default=16)
    ap.add_argument("--batch", type=int, default=4)
    ap.add_argument("--clip", action="store_true")
    args = ap.parse_args()

    if args.clip:
        from flaxdiff_amd.inputs import CLIPTextEncoder
        encoder = CLIPTextEncoder.from_modelname()
    else:
        encoder = DummyTextEncoder()

    input_config = DiffusionInputConfig(
        sample_data_key="image",
        sample_data_shape=(args.size, args.size, 3),
        conditions=[ConditionalInputConfig(
            encoder=encoder, conditioning_data_key="text",
            pretokenized=True, unconditional_input="",
            model_key_override="textcontext")])

    model_cfg = dict(emb_features=32, feature_depths=[8, 16],
                     attention_configs=[None, {"heads": 2}], num_res_blocks=1,
                     norm_groups=4, context_dim=768)
    model = Unet(**model_cfg)
    trainer = GeneralDiffusionTrainer(
        model, EDMNoiseScheduler(1, sigma_max=80, sigma_data=0.5),
        KarrasPredictionTransform(sigma_data=0.5),
        input_config=input_config, name="textcond",
        checkpoint_base_path="./checkpoints",
        compute_dtype=torch.bfloat16 if torch.cuda.is_available()
        else torch.float32,
        distributed=int(os.environ.get("WORLD_SIZE", "1")) > 1)

    captions = ["a red square", "a blue circle", "green noise", "white"]
    toks = encoder.tokenize(captions[: args.batch])

    def batches():
        g = torch.Generator().manual_seed(0)
        while True:
            yield {"image": torch.randint(0, 255,
                                          (args.batch, args.size, args.size, 3),
                                          generator=g, dtype=torch.uint8),
                   "text": toks}

    # epoch-end validation: real multi-step EMA sampling + an eval metric
    # with best-direction tracking (reference general_diffusion_trainer
    # :378-519). pixel_std flags degenerate (collapsed/blank) samples.
    trainer.eval_metrics = [EvaluationMetric(
        function=lambda samples, batch: float(samples.float().std()),
        name="pixel_std", higher_is_better=True)]
    val_fn = trainer.make_validation_fn(
        sampler_class=EulerAncestralSampler, num_samples=args.batch,
        resolution=args.size, diffusion_steps=10, guidance_scale=3.0,
        conditioning_context=encoder(captions[: args.batch]))
    cfg = {"architecture": "unet", "model": model_cfg,
           "noise_schedule": "edm",
           "input_config": input_config.serialize(),
           "arguments": {"image_size": args.size}}
    trainer.fit(batches(), steps_per_epoch=args.steps, epochs=1,
                val_fn=val_fn, config=cfg)
    print("val metrics (best):", trainer.best_metric_values)
    assert trainer.best_metric_values.get("pixel_std", 0) > 0

    out = trainer.validation_sample(
        EulerAncestralSampler, num_samples=args.batch, resolution=args.size,
        diffusion_steps=10, guidance_scale=3.0,
        conditioning_context=encoder(captions[: args.batch]))
    print("CFG samples:", tuple(out.shape))


if __name__ == "__main__":
    main()
