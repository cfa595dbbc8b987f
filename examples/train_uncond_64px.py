#!/usr/bin/env python3
"""Minimal unconditional diffusion training — the BASELINE config-3 shape.

Mirrors the reference's "simple diffusion" tutorial notebook flow:
build scheduler + transform + UNet, train on a dataset, sample at the end.
Runs on 1 GPU as-is; on CPU pass --tiny for a smoke-scale config.
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from flaxdiff_amd.data import get_dataset
from flaxdiff_amd.models import Unet
from flaxdiff_amd.predictors import KarrasPredictionTransform
from flaxdiff_amd.samplers import EulerAncestralSampler
from flaxdiff_amd.schedulers import EDMNoiseScheduler
from flaxdiff_amd.trainer import DiffusionTrainer


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=1000)
    ap.add_argument("--batch", type=int, default=256)
    ap.add_argument("--tiny", action="store_true", help="CPU smoke scale")
    args = ap.parse_args()

    if args.tiny:
        size, depths, batch = 16, [8, 16], 4
        heads, res_blocks, groups = 2, 1, 4
    else:
        size, depths, batch = 64, [64, 128, 256, 512], args.batch
        heads, res_blocks, groups = 4, 2, 8

    model_cfg = dict(emb_features=depths[0] * 4, feature_depths=depths,
                     attention_configs=[{"heads": heads}] * len(depths),
                     num_res_blocks=res_blocks, norm_groups=groups,
                     context_dim=768)
    model = Unet(**model_cfg)
    trainer = DiffusionTrainer(
        model,
        EDMNoiseScheduler(1, sigma_max=80, sigma_data=0.5),
        KarrasPredictionTransform(sigma_data=0.5),
        name="uncond64", checkpoint_base_path="./checkpoints",
        compute_dtype=torch.bfloat16 if torch.cuda.is_available()
        else torch.float32,
        distributed=int(os.environ.get("WORLD_SIZE", "1")) > 1)

    loader = get_dataset("synthetic-64",
                         global_batch_size=batch, image_size=size,
                         worker_count=0)
    it = iter(loader)

    def batches():
        nonlocal it
        while True:
            try:
                yield next(it)
            except StopIteration:
                it = iter(loader)

    trainer.train_loop(batches(), steps=args.steps)
    # the manifest lets DiffusionInferencePipeline.from_checkpoint rebuild
    # the model (see examples/sample_from_checkpoint.py)
    trainer.save(config={"architecture": "unet", "model": model_cfg,
                         "noise_schedule": "edm",
                         "arguments": {"image_size": size}}, block=True)

    # sample with EMA weights (the trainer's validation path)
    out = trainer.validation_sample(EulerAncestralSampler, num_samples=4,
                                    resolution=size, diffusion_steps=10)
    print("samples:", tuple(out.shape), "range",
          float(out.min()), float(out.max()))


if __name__ == "__main__":
    main()
