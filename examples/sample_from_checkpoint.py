#!/usr/bin/env python3
"""Restore a trained run and generate samples — the inference-pipeline flow
(reference tutorial: load wandb artifact -> pipeline -> generate)."""
import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from flaxdiff_amd.inference import DiffusionInferencePipeline
from flaxdiff_amd.samplers import EulerAncestralSampler


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("checkpoint", help="checkpoint dir (e.g. ./checkpoints/uncond64)")
    ap.add_argument("--num", type=int, default=4)
    ap.add_argument("--resolution", type=int, default=64)
    ap.add_argument("--steps", type=int, default=50)
    ap.add_argument("--guidance", type=float, default=0.0)
    ap.add_argument("--no-ema", action="store_true")
    args = ap.parse_args()

    pipe = DiffusionInferencePipeline.from_checkpoint(
        args.checkpoint, use_ema=not args.no_ema)
    out = pipe.generate_samples(
        num_samples=args.num, resolution=args.resolution,
        diffusion_steps=args.steps, guidance_scale=args.guidance,
        sampler_class=EulerAncestralSampler)
    print("samples:", tuple(out.shape), "range",
          float(out.min()), float(out.max()))


if __name__ == "__main__":
    main()
