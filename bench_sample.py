#!/usr/bin/env python3
"""Sampling benchmark: EDM 50-step samples/sec on the 64px UNet (the second
half of BASELINE.json's headline metric). Compares eager vs hipGraph-captured
sampling. Not part of the driver's bench.py contract — run manually:
    python bench_sample.py [--batch 64] [--steps 50] [--reps 3]
"""
import argparse
import json
import time

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=64)
    ap.add_argument("--steps", type=int, default=50)
    ap.add_argument("--reps", type=int, default=3)
    ap.add_argument("--resolution", type=int, default=64)
    ap.add_argument("--guidance", type=float, default=0.0)
    args = ap.parse_args()

    from flaxdiff_amd.models import Unet
    from flaxdiff_amd.predictors import KarrasPredictionTransform
    from flaxdiff_amd.samplers import HeunSampler, EulerAncestralSampler
    from flaxdiff_amd.schedulers import KarrasVENoiseScheduler
    from flaxdiff_amd.utils import RandomMarkovState

    use_gpu = torch.cuda.is_available()
    dev = "cuda" if use_gpu else "cpu"
    dtype = torch.bfloat16 if use_gpu else torch.float32

    torch.manual_seed(0)
    model = Unet(emb_features=256, feature_depths=[64, 128, 256, 512],
                 attention_configs=[{"heads": 4}] * 4, num_res_blocks=2,
                 num_middle_res_blocks=1, norm_groups=8, context_dim=768).to(dev)
    # inference runs the whole net in bf16 (same compute dtype as training;
    # the sampler's transform math stays fp32 on the outputs)
    if use_gpu:
        model = model.to(torch.bfloat16)
    model.eval()

    schedule = KarrasVENoiseScheduler(1, sigma_max=80, rho=7, sigma_data=0.5)
    transform = KarrasPredictionTransform(sigma_data=0.5)

    null_ctx = torch.zeros(1, 77, 768, device=dev)

    def bench(sampler_cls, graph: bool, label: str):
        calls = {"n": 0}

        def wrapped_model(x, t, *c):
            calls["n"] += 1  # counts evals (graph path: captures, not replays)
            return model(x.to(dtype), t,
                         null_ctx.expand(x.shape[0], -1, -1).to(dtype)).float()

        sampler = sampler_cls(model=wrapped_model,
                              noise_schedule=schedule,
                              model_output_transform=transform,
                              guidance_scale=args.guidance,
                              timestep_spacing="linear")  # KarrasVE scheduler: linear t IS the rho ramp
        if graph:
            sampler.enable_graph_capture()
        kw = dict(num_samples=args.batch, resolution=args.resolution,
                  diffusion_steps=args.steps, device=dev, dtype=torch.float32)
        sampler.generate_samples(rngstate=RandomMarkovState(1), **kw)  # warmup
        if not graph:
            # NFE self-check: the warmup batch must have evaluated the model
            # >= diffusion_steps times (Heun: 2/step). Guards against the
            # round-1 regression where continuous schedulers collapsed to 1.
            nfe_min = args.steps
            assert calls["n"] >= nfe_min, \
                f"only {calls['n']} model evals for {args.steps}-step sampling"
        if use_gpu:
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for r in range(args.reps):
            out = sampler.generate_samples(rngstate=RandomMarkovState(2 + r), **kw)
        if use_gpu:
            torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / args.reps
        print(json.dumps({
            "metric": f"EDM {args.steps}-step samples/sec ({label})",
            "value": args.batch / dt, "unit": "samples/s",
            "sec_per_batch": dt, "batch": args.batch,
            "sampler": sampler_cls.__name__, "graph": graph,
            "nfe_per_step": 2 if sampler_cls is HeunSampler else 1,
            "checksum": float(out.float().mean()),
        }))
        return out

    out_eager = bench(EulerAncestralSampler, False, "euler_ancestral eager")
    out_graph = bench(EulerAncestralSampler, True, "euler_ancestral hipGraph")
    if use_gpu:
        diff = (out_eager - out_graph).abs().max().item()
        print(json.dumps({"graph_vs_eager_max_abs_diff": diff}))
    bench(HeunSampler, True, "heun hipGraph")


if __name__ == "__main__":
    main()
