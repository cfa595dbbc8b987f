#!/usr/bin/env python3
"""Training CLI — the reference's single entry point, MI355X-native.

Flag-surface contract: reference /root/reference/training.py:83-218 (data
tuning, model arch + dims, dtype, distributed, checkpoint/resume, optimizer +
LR schedule, LDM, grad clip, wandb, val metrics, dropout/EMA/augment hygiene),
arch registry with per-arch kwargs :383-488, experiment-name templating
:555-582, warmup-cosine LR :597-601, trainer construction + fit :622-665.

Launch (one process per GPU over RCCL):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 training.py --dataset synthetic-64 ...
Single process: python training.py --distributed_training False ...
"""
from __future__ import annotations

import argparse
import json
import os
import time
from datetime import datetime

import torch


def boolean_string(s):
    if str(s).lower() not in {"false", "true"}:
        raise ValueError("Not a valid boolean string")
    return str(s).lower() == "true"


def build_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(description="Train a diffusion model (MI355X)")
    # data pipeline (grain flags map to DataLoader workers/prefetch)
    p.add_argument("--GRAIN_WORKER_COUNT", type=int, default=8,
                   help="DataLoader worker processes")
    p.add_argument("--GRAIN_READ_THREAD_COUNT", type=int, default=4)
    p.add_argument("--GRAIN_READ_BUFFER_SIZE", type=int, default=32)
    p.add_argument("--GRAIN_WORKER_BUFFER_SIZE", type=int, default=8)
    p.add_argument("--batch_size", type=int, default=32, help="GLOBAL batch size")
    p.add_argument("--image_size", type=int, default=128)
    p.add_argument("--epochs", type=int, default=100)
    p.add_argument("--steps_per_epoch", type=int, default=None)
    p.add_argument("--val_steps_per_epoch", type=int, default=4)
    p.add_argument("--dataset", type=str, default="synthetic-64")
    p.add_argument("--dataset_path", type=str, default=None)
    p.add_argument("--dataset_seed", type=int, default=0)
    p.add_argument("--dataset_test", type=boolean_string, default=False,
                   help="Iterate the loader for throughput/leak checking only")
    # schedules / arch
    p.add_argument("--noise_schedule", type=str, default="edm",
                   choices=["edm", "karras", "cosine"])
    p.add_argument("--architecture", type=str, default="unet",
                   help="unet|uvit|simple_dit|simple_mmdit|simple_udit|"
                        "hierarchical_mmdit|hybrid_dit (+hilbert/+zigzag/+2d)")
    p.add_argument("--emb_features", type=int, default=256)
    p.add_argument("--feature_depths", type=int, nargs="+",
                   default=[64, 128, 256, 512])
    p.add_argument("--attention_heads", type=int, default=8)
    p.add_argument("--flash_attention", type=boolean_string, default=False)
    p.add_argument("--use_projection", type=boolean_string, default=False)
    p.add_argument("--use_self_and_cross", type=boolean_string, default=True)
    p.add_argument("--only_pure_attention", type=boolean_string, default=True)
    p.add_argument("--norm_groups", type=int, default=8,
                   help="0 selects RMSNorm")
    p.add_argument("--num_res_blocks", type=int, default=2)
    p.add_argument("--num_middle_res_blocks", type=int, default=1)
    p.add_argument("--activation", type=str, default="swish")
    p.add_argument("--patch_size", type=int, default=16)
    p.add_argument("--num_layers", type=int, default=12)
    p.add_argument("--num_heads", type=int, default=12)
    p.add_argument("--mlp_ratio", type=int, default=4)
    p.add_argument("--use_hilbert", type=boolean_string, default=False)
    p.add_argument("--use_zigzag", type=boolean_string, default=False)
    p.add_argument("--use_2d_fusion", type=boolean_string, default=False)
    p.add_argument("--ssm_attention_ratio", type=str, default="3:1")
    p.add_argument("--ssm_state_dim", type=int, default=64)
    p.add_argument("--add_residualblock_output", type=boolean_string, default=False)
    # precision / distribution
    p.add_argument("--dtype", type=str, default=None,
                   help="bfloat16|float32 (compute dtype)")
    p.add_argument("--distributed_training", type=boolean_string, default=True)
    # experiment management
    p.add_argument("--experiment_name", type=str, default=None)
    p.add_argument("--load_from_checkpoint", type=str, default=None)
    p.add_argument("--checkpoint_dir", type=str, default="./checkpoints")
    p.add_argument("--max_checkpoints_to_keep", type=int, default=1)
    p.add_argument("--save_every", type=int, default=1000)
    # optimizer
    p.add_argument("--optimizer", type=str, default="adamw", choices=["adamw"])
    p.add_argument("--optimizer_opts", type=str, default="{}")
    p.add_argument("--learning_rate_schedule", type=str, default=None,
                   choices=[None, "cosine"])
    p.add_argument("--learning_rate", type=float, default=2.7e-4)
    p.add_argument("--learning_rate_peak", type=float, default=3e-4)
    p.add_argument("--learning_rate_end", type=float, default=2e-4)
    p.add_argument("--learning_rate_warmup_steps", type=int, default=10000)
    p.add_argument("--clip_grads", type=float, default=0)
    # LDM
    p.add_argument("--autoencoder", type=str, default=None,
                   choices=[None, "stable_diffusion", "simple"])
    p.add_argument("--autoencoder_opts", type=str, default="{}")
    # logging / eval
    p.add_argument("--wandb_project", type=str, default=None)
    p.add_argument("--wandb_entity", type=str, default=None)
    p.add_argument("--val_metrics", type=str, nargs="+", default=[])
    p.add_argument("--best_tracker_metric", type=str, default="val/loss")
    # hygiene
    p.add_argument("--dropout_rate", type=float, default=0.1)
    p.add_argument("--ema_decay", type=float, default=0.999)
    p.add_argument("--unconditional_prob", type=float, default=0.12)
    p.add_argument("--augmentation_mode", type=str, default="flip_jitter",
                   choices=["none", "flip_only", "flip_jitter"])
    return p


# ---------------------------------------------------------------------------
# architecture registry (reference training.py:383-488)
# ---------------------------------------------------------------------------

def build_model(args):
    from flaxdiff_amd.inference.utils import canonicalize_architecture
    from flaxdiff_amd.models import (HierarchicalMMDiT, HybridSSMAttentionDiT,
                                     SimpleDiT, SimpleMMDiT, SimpleUDiT, Unet,
                                     UViT)
    arch_raw = args.architecture
    use_hilbert = args.use_hilbert or "+hilbert" in arch_raw
    use_zigzag = args.use_zigzag or "+zigzag" in arch_raw
    use_2d = args.use_2d_fusion or "+2d" in arch_raw
    arch = canonicalize_architecture(arch_raw)

    if arch == "unet":
        cfg = dict(emb_features=args.emb_features,
                   feature_depths=list(args.feature_depths),
                   attention_configs=[{
                       "heads": args.attention_heads,
                       "use_projection": args.use_projection,
                       "use_self_and_cross": args.use_self_and_cross,
                       "only_pure_attention": args.only_pure_attention,
                   }] * len(args.feature_depths),
                   num_res_blocks=args.num_res_blocks,
                   num_middle_res_blocks=args.num_middle_res_blocks,
                   norm_groups=args.norm_groups, context_dim=768)
        return Unet(**cfg), cfg, arch
    tok = dict(patch_size=args.patch_size, emb_features=args.emb_features,
               num_layers=args.num_layers, num_heads=args.num_heads,
               context_dim=768)
    if arch == "uvit":
        cfg = dict(tok, use_projection=args.use_projection,
                   use_self_and_cross=args.use_self_and_cross,
                   norm_groups=args.norm_groups, use_hilbert=use_hilbert,
                   add_residualblock_output=args.add_residualblock_output)
        return UViT(**cfg), cfg, arch
    if arch == "simple_dit":
        cfg = dict(tok, mlp_ratio=args.mlp_ratio, use_hilbert=use_hilbert,
                   use_zigzag=use_zigzag)
        return SimpleDiT(**cfg), cfg, arch
    if arch == "simple_udit":
        cfg = dict(tok, mlp_ratio=args.mlp_ratio, use_hilbert=use_hilbert)
        return SimpleUDiT(**cfg), cfg, arch
    if arch == "simple_mmdit":
        cfg = dict(tok, mlp_ratio=args.mlp_ratio, use_hilbert=use_hilbert)
        return SimpleMMDiT(**cfg), cfg, arch
    if arch == "hierarchical_mmdit":
        cfg = dict(base_patch_size=args.patch_size // 2 or 8,
                   context_dim=768, mlp_ratio=args.mlp_ratio,
                   use_hilbert=use_hilbert)
        return HierarchicalMMDiT(**cfg), cfg, arch
    if arch == "hybrid_dit":
        cfg = dict(tok, mlp_ratio=args.mlp_ratio, use_hilbert=use_hilbert,
                   use_zigzag=use_zigzag, use_2d_fusion=use_2d,
                   ssm_state_dim=args.ssm_state_dim,
                   ssm_attention_ratio=args.ssm_attention_ratio)
        return HybridSSMAttentionDiT(**cfg), cfg, arch
    raise ValueError(f"unknown architecture {arch_raw}")


def make_experiment_name(args, arch: str) -> str:
    """Reference-style templating (training.py:555-582)."""
    if args.experiment_name:
        return args.experiment_name
    stamp = datetime.now().strftime("%Y-%m-%d_%H-%M")
    return (f"{arch}_{args.dataset}_{args.image_size}px_"
            f"bs{args.batch_size}_{args.noise_schedule}-{stamp}")


def main(argv=None):
    args = build_parser().parse_args(argv)

    # env bridge for augment hygiene (reference :221-223)
    mode_map = {"none": "none", "flip_only": "flip", "flip_jitter": "all"}
    os.environ["FLAXDIFF_AUGMENT_MODE"] = mode_map[args.augmentation_mode]

    from flaxdiff_amd import parallel
    from flaxdiff_amd.data import get_dataset
    from flaxdiff_amd.inputs import (ConditionalInputConfig,
                                     DiffusionInputConfig)
    from flaxdiff_amd.inputs.encoders import get_text_encoder
    from flaxdiff_amd.predictors import (KarrasPredictionTransform,
                                         VPredictionTransform)
    from flaxdiff_amd.samplers import EulerAncestralSampler
    from flaxdiff_amd.schedulers import (CosineNoiseScheduler,
                                         EDMNoiseScheduler,
                                         KarrasVENoiseScheduler)
    from flaxdiff_amd.trainer import GeneralDiffusionTrainer
    from flaxdiff_amd.trainer.optim import warmup_cosine_schedule

    dist = parallel.init_distributed() if args.distributed_training else None
    rank = dist.rank if dist else 0
    world = dist.world_size if dist else 1

    model, model_cfg, arch = build_model(args)
    experiment_name = make_experiment_name(args, arch)

    # data
    tokenizer = None
    encoder = get_text_encoder(prefer_clip=False)  # offline default
    if hasattr(encoder, "tokenizer") and encoder.tokenizer is not None:
        tokenizer = encoder.tokenizer
    loader = get_dataset(args.dataset, global_batch_size=args.batch_size,
                         rank=rank, world_size=world,
                         worker_count=args.GRAIN_WORKER_COUNT,
                         seed=args.dataset_seed, tokenizer=tokenizer)

    if args.dataset_test:
        t0 = time.time()
        it = iter(loader)
        for i in range(2000):
            next(it)
            if i % 100 == 0 and rank == 0:
                print(f"{i} batches, {i / max(time.time() - t0, 1e-9):.1f} b/s")
        return

    # schedule + transform
    if args.noise_schedule == "edm":
        schedule = EDMNoiseScheduler(1, sigma_max=80, rho=7, sigma_data=0.5)
        transform = KarrasPredictionTransform(sigma_data=0.5)
    elif args.noise_schedule == "karras":
        schedule = KarrasVENoiseScheduler(1, sigma_max=80, rho=7, sigma_data=0.5)
        transform = KarrasPredictionTransform(sigma_data=0.5)
    else:
        schedule = CosineNoiseScheduler(1000, beta_end=1)
        transform = VPredictionTransform()

    # LDM
    autoencoder = None
    if args.autoencoder:
        from flaxdiff_amd.models.autoencoder import get_autoencoder
        autoencoder = get_autoencoder(args.autoencoder,
                                      **json.loads(args.autoencoder_opts))

    # input config (text conditioning)
    input_config = DiffusionInputConfig(
        sample_data_key="image",
        sample_data_shape=(args.image_size, args.image_size, 3),
        conditions=[ConditionalInputConfig(
            encoder=encoder, conditioning_data_key="text", pretokenized=True,
            unconditional_input="", model_key_override="textcontext")])

    # optimizer opts + LR schedule
    opt_opts = json.loads(args.optimizer_opts)
    opt_kwargs = dict(lr=args.learning_rate, ema_decay=args.ema_decay,
                      **opt_opts)
    if args.clip_grads:
        opt_kwargs["grad_clip_norm"] = args.clip_grads
    if args.learning_rate_schedule == "cosine":
        total = (args.steps_per_epoch or 1000) * args.epochs
        opt_kwargs["lr_schedule"] = warmup_cosine_schedule(
            args.learning_rate_peak, args.learning_rate_warmup_steps, total,
            final_scale=args.learning_rate_end / args.learning_rate_peak)

    compute_dtype = {"bfloat16": torch.bfloat16, "float32": torch.float32,
                     None: (torch.bfloat16 if torch.cuda.is_available()
                            else torch.float32)}[args.dtype]

    eval_metrics = []
    for m in args.val_metrics:
        from flaxdiff_amd import metrics as M
        if m == "clip":
            eval_metrics.append(M.get_clip_score_metric())
        elif m == "psnr":
            eval_metrics.append(M.get_psnr_metric())
        elif m == "ssim":
            eval_metrics.append(M.get_ssim_metric())

    run_config = {
        "arguments": vars(args),
        "architecture": arch,
        "model": model_cfg,
        "noise_schedule": args.noise_schedule,
        "autoencoder": args.autoencoder,
        "autoencoder_opts": args.autoencoder_opts,
        "input_config": input_config.serialize(),
    }

    trainer = GeneralDiffusionTrainer(
        model, schedule, transform,
        input_config=input_config,
        eval_metrics=eval_metrics,
        autoencoder=autoencoder,
        unconditional_prob=args.unconditional_prob,
        name=experiment_name,
        checkpoint_base_path=args.checkpoint_dir,
        max_checkpoints_to_keep=args.max_checkpoints_to_keep,
        load_from_checkpoint=args.load_from_checkpoint is not None,
        compute_dtype=compute_dtype,
        distributed=args.distributed_training,
        optimizer_kwargs=opt_kwargs,
        wandb_project=args.wandb_project,
        wandb_config=run_config,
    )

    steps_per_epoch = args.steps_per_epoch or max(len(loader), 1)

    # H2D copies of the next batch overlap the current step on a side stream
    from flaxdiff_amd.data import DevicePrefetcher
    prefetched = DevicePrefetcher(loader, trainer.device)

    def data_iter():
        epoch = 0
        while True:
            if hasattr(loader, "sampler") and hasattr(loader.sampler, "set_epoch"):
                loader.sampler.set_epoch(epoch)
            for batch in prefetched:
                yield batch
            epoch += 1

    trainer.fit(data_iter(), steps_per_epoch, args.epochs,
                save_every=args.save_every, config=run_config)
    if rank == 0:
        print(f"done: {experiment_name}")


if __name__ == "__main__":
    main()
