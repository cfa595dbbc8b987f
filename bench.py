#!/usr/bin/env python3
"""Flagship training benchmark — the driver contract.

Measures the BASELINE.json metric: train images/sec/node for the 64x64
unconditional EDM UNet (feature_depths [64,128,256,512], heads 4, 2 res
blocks — README.md:363-367 of the reference) at global batch 256, bf16,
synthetic data, random-init weights. N>1 runs one rank per GPU over RCCL
(launched by torch.distributed.run); global batch stays 256 (strong scaling,
BASELINE config 3).

Usage: python bench.py [--gpus N] [--steps K] [--warmup W]
"""
import argparse
import json
import os
import time

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--global-batch", type=int, default=256)
    ap.add_argument("--resolution", type=int, default=64)
    ap.add_argument("--dtype", type=str, default="bf16", choices=["bf16", "fp32"])
    ap.add_argument("--profile", action="store_true",
                    help="per-step hipEvent timings to stderr (SURVEY §5.1)")
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    n_gpus = max(world, 1)
    if world <= 1:
        n_gpus = 1

    use_gpu = torch.cuda.is_available()
    compute_dtype = torch.bfloat16 if (args.dtype == "bf16" and use_gpu) else torch.float32

    from flaxdiff_amd import parallel
    from flaxdiff_amd.models import Unet
    from flaxdiff_amd.predictors import KarrasPredictionTransform
    from flaxdiff_amd.schedulers import EDMNoiseScheduler
    from flaxdiff_amd.trainer import DiffusionTrainer

    torch.manual_seed(1234)
    # FD_BENCH_TINY: plumbing dry-run only (CI launches the full torchrun
    # rendezvous + DP path on CPU with a toy model; never a perf number)
    tiny = os.environ.get("FD_BENCH_TINY") == "1"
    if tiny:
        args.global_batch = min(args.global_batch, 4 * n_gpus)
        args.resolution = 16
    model = Unet(
        output_channels=3,
        emb_features=32 if tiny else 256,
        feature_depths=[8, 16] if tiny else [64, 128, 256, 512],
        attention_configs=[{"heads": 2}] * (2 if tiny else 4),
        num_res_blocks=1 if tiny else 2,
        num_middle_res_blocks=1,
        norm_groups=4 if tiny else 8,
        context_dim=16 if tiny else 768,
    )
    trainer = DiffusionTrainer(
        model,
        EDMNoiseScheduler(1, sigma_max=80, sigma_data=0.5),
        KarrasPredictionTransform(sigma_data=0.5),
        name="bench",
        checkpoint_base_path="/tmp/fdiff_bench_ckpt",
        compute_dtype=compute_dtype,
        distributed=(world > 1),
        optimizer_kwargs={"lr": 2.7e-4},
        **({"text_context_shape": (4, 16)} if tiny else {}),
    )
    dev = trainer.device

    local_batch = max(args.global_batch // n_gpus, 1)
    g = torch.Generator().manual_seed(42 + rank)
    batch = {
        "image": torch.randint(0, 255, (local_batch, args.resolution,
                                        args.resolution, 3),
                               generator=g, dtype=torch.uint8).to(dev),
    }

    # warmup
    for _ in range(args.warmup):
        trainer.train_step(batch)

    parallel.barrier()
    if use_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    loss = 0.0
    ev = []
    if args.profile and use_gpu:
        for _ in range(args.steps):
            a = torch.cuda.Event(enable_timing=True)
            b = torch.cuda.Event(enable_timing=True)
            a.record()
            loss = trainer.train_step(batch)["loss"]
            b.record()
            ev.append((a, b))
    else:
        for _ in range(args.steps):
            loss = trainer.train_step(batch)["loss"]
    parallel.barrier()
    if use_gpu:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    if ev:
        import sys
        times = [a.elapsed_time(b) for a, b in ev]
        mean = sum(times) / len(times)
        var = sum((x - mean) ** 2 for x in times) / len(times)
        print(json.dumps({"profile": {"per_step_ms": [round(x, 3) for x in times],
                                      "mean_ms": round(mean, 3),
                                      "std_ms": round(var ** 0.5, 3),
                                      "rank": rank}}), file=sys.stderr)

    # max over ranks
    el = torch.tensor([elapsed], dtype=torch.float64)
    if world > 1:
        import torch.distributed as dist
        el_dev = el.to(dev) if use_gpu else el
        dist.all_reduce(el_dev, op=dist.ReduceOp.MAX)
        elapsed = float(el_dev.item())

    images_total = args.global_batch if world > 1 else local_batch
    value = images_total * args.steps / elapsed
    ms_per_step = elapsed / args.steps * 1000

    if rank == 0:
        print(json.dumps({
            "metric": "train images/sec/node (64x64 uncond EDM UNet, bs=256)",
            "value": value,
            "unit": "images/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": "bf16" if compute_dtype == torch.bfloat16 else "fp32",
            "data": "synthetic",
            "config": {
                "model": "unet_64px_[64,128,256,512]_res2_heads4",
                "global_batch": images_total,
                "resolution": args.resolution,
                "parallelism": f"dp{n_gpus}",
                "schedule": "EDM + KarrasPredictionTransform(sigma_data=0.5)",
                "final_loss": loss,
            },
        }))


if __name__ == "__main__":
    main()
