"""300-step convergence evidence: EDM UNet-64 on a fixed synthetic set."""
import json
import torch
from flaxdiff_amd.models import Unet
from flaxdiff_amd.predictors import KarrasPredictionTransform
from flaxdiff_amd.schedulers import EDMNoiseScheduler
from flaxdiff_amd.trainer import DiffusionTrainer

torch.manual_seed(0)
model = Unet(emb_features=256, feature_depths=[64, 128, 256, 512],
             attention_configs=[{"heads": 4}] * 4, num_res_blocks=2,
             norm_groups=8, context_dim=768)
tr = DiffusionTrainer(model, EDMNoiseScheduler(1, sigma_max=80, sigma_data=0.5),
                      KarrasPredictionTransform(sigma_data=0.5),
                      name="conv300", checkpoint_base_path="/tmp/conv300",
                      compute_dtype=torch.bfloat16, distributed=False)
# fixed dataset of 2048 synthetic structured images (circles/gradients beat
# pure noise: the model has something to learn)
N = 2048
xs = torch.linspace(-1, 1, 64)
gy, gx = torch.meshgrid(xs, xs, indexing="ij")
imgs = []
g = torch.Generator().manual_seed(1)
for i in range(N):
    cx, cy, r = torch.rand(3, generator=g) * 1.6 - 0.8
    rad = 0.2 + 0.5 * torch.rand(1, generator=g)
    d = ((gx - cx) ** 2 + (gy - cy) ** 2).sqrt()
    img = torch.stack([(d < rad).float(),
                       0.5 + 0.5 * gx * torch.rand(1, generator=g),
                       0.5 + 0.5 * gy], 0)
    imgs.append((img.permute(1, 2, 0) * 255).to(torch.uint8))
data = torch.stack(imgs)

losses = []
for step in range(300):
    idx = torch.randint(0, N, (256,))
    out = tr.train_step({"image": data[idx]})
    losses.append(out["loss"])
rec = {"steps": 300, "batch": 256, "model": "unet_64px_[64,128,256,512]",
       "loss_first10": [round(l, 4) for l in losses[:10]],
       "loss_last10": [round(l, 4) for l in losses[-10:]],
       "loss_mean_0_50": round(sum(losses[:50]) / 50, 4),
       "loss_mean_250_300": round(sum(losses[250:]) / 50, 4),
       "graphed": tr._graph is not None,
       "optimizer_steps": tr.optimizer.step_count,
       "skipped_steps": tr.optimizer.skipped_steps}
print(json.dumps(rec))
with open("gpurun_out/r2_convergence.json", "w") as f:
    json.dump(rec, f, indent=1)
