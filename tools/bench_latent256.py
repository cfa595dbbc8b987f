#!/usr/bin/env python3
"""BASELINE config-5 shape at 1 GPU: 256x256 latent diffusion.

Frozen (random-init — offline env) SD-VAE on the native HIP kernel stack
encodes 256x256 uint8 batches to 32x32x4 latents; a text-conditional LDM
UNet trains in latent space (CFG dropout), and DDIM 50-step sampling
decodes back through the VAE. Reports train images/sec and DDIM
samples/sec as one JSON line each.
"""
import json
import time

import torch

from flaxdiff_amd.models import Unet
from flaxdiff_amd.models.autoencoder import StableDiffusionVAE
from flaxdiff_amd.predictors import KarrasPredictionTransform
from flaxdiff_amd.samplers import DDIMSampler
from flaxdiff_amd.schedulers import EDMNoiseScheduler
from flaxdiff_amd.trainer import DiffusionTrainer
from flaxdiff_amd.utils import RandomMarkovState

BATCH = 32
STEPS = 10
WARMUP = 3

torch.manual_seed(0)
dev = "cuda"
vae = StableDiffusionVAE(dtype=torch.bfloat16, device=dev)

model = Unet(in_channels=4, output_channels=4, emb_features=512,
             feature_depths=[192, 384, 768],
             attention_configs=[{"heads": 8}] * 3,
             num_res_blocks=2, num_middle_res_blocks=1,
             norm_groups=32, context_dim=768)
trainer = DiffusionTrainer(
    model, EDMNoiseScheduler(1, sigma_max=80, sigma_data=0.5),
    KarrasPredictionTransform(sigma_data=0.5), name="ldm256",
    checkpoint_base_path="/tmp/ldm256", compute_dtype=torch.bfloat16,
    autoencoder=vae, distributed=False)

g = torch.Generator().manual_seed(1)
batch = {"image": torch.randint(0, 255, (BATCH, 256, 256, 3), generator=g,
                                dtype=torch.uint8),
         "text_emb": torch.randn(BATCH, 77, 768, generator=g)}

for _ in range(WARMUP):
    trainer.train_step(batch)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(STEPS):
    out = trainer.train_step(batch)
torch.cuda.synchronize()
dt = time.perf_counter() - t0
print(json.dumps({
    "metric": "train images/sec (256px latent text-cond UNet [192,384,768], "
              "frozen SD-VAE encode, CFG dropout)",
    "value": BATCH * STEPS / dt, "ms_per_step": dt / STEPS * 1e3,
    "batch": BATCH, "dtype": "bf16", "final_loss": out["loss"],
    "vae": "random-init SD-VAE shape [128,256,512,512] (offline env)"}))

# ---- DDIM 50-step sampling incl. VAE decode ------------------------------
model.eval()
nfe = {"n": 0}
base_fwd = model.forward


def counting(*a, **k):
    nfe["n"] += 1
    return base_fwd(*a, **k)


model.forward = counting
sampler = DDIMSampler(
    model=lambda x, t, *c: model(x.to(torch.bfloat16), t,
                                 *(ci.to(torch.bfloat16) for ci in c)).float(),
    noise_schedule=trainer.noise_schedule,
    model_output_transform=trainer.model_output_transform,
    autoencoder=vae, guidance_scale=1.5, timestep_spacing="linear")
SB, SSTEPS = 16, 50
text = torch.randn(SB, 77, 768, device=dev, dtype=torch.float32)
kw = dict(num_samples=SB, resolution=256, diffusion_steps=SSTEPS,
          model_conditioning_inputs=(text,), device=dev)
sampler.unconditionals = [torch.zeros(77, 768)]
out = sampler.generate_samples(rngstate=RandomMarkovState(2), **kw)  # warmup
torch.cuda.synchronize()
nfe["n"] = 0
t0 = time.perf_counter()
out = sampler.generate_samples(rngstate=RandomMarkovState(3), **kw)
torch.cuda.synchronize()
dt = time.perf_counter() - t0
assert out.shape[1] == 256 and out.shape[3] == 3, out.shape
assert nfe["n"] >= SSTEPS, f"NFE {nfe['n']} < {SSTEPS}"
print(json.dumps({
    "metric": "DDIM 50-step 256px latent samples/sec (CFG 1.5, VAE decode)",
    "value": SB / dt, "sec_per_batch": dt, "batch": SB, "nfe": nfe["n"]}))
