import sys, os; sys.path.insert(0, os.getcwd())
"""BASELINE config-4 datapoint: 128px text-conditional UNet + CFG dropout."""
import json, time
import torch
from flaxdiff_amd.models import Unet
from flaxdiff_amd.predictors import KarrasPredictionTransform
from flaxdiff_amd.schedulers import EDMNoiseScheduler
from flaxdiff_amd.trainer import DiffusionTrainer

torch.manual_seed(0)
model = Unet(emb_features=512, feature_depths=[128, 256, 512, 1024],
             attention_configs=[{"heads": 8}] * 4, num_res_blocks=2,
             num_middle_res_blocks=1, norm_groups=8, context_dim=768)
tr = DiffusionTrainer(model, EDMNoiseScheduler(1, sigma_max=80),
                      KarrasPredictionTransform(sigma_data=0.5),
                      name="cond128", checkpoint_base_path="/tmp/fd_c128",
                      compute_dtype=torch.bfloat16, distributed=False)
B = 64
batch = {"image": torch.randint(0, 255, (B, 128, 128, 3), dtype=torch.uint8),
         "text_emb": torch.randn(B, 77, 768)}
for _ in range(3):
    tr.train_step(batch)
torch.cuda.synchronize()
t0 = time.perf_counter()
K = 8
for _ in range(K):
    out = tr.train_step(batch)
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / K
print(json.dumps({"metric": "train images/sec (128px text-cond UNet [128,256,512,1024], CFG dropout)",
                  "value": B / dt, "ms_per_step": dt * 1e3, "batch": B,
                  "dtype": "bf16", "final_loss": out["loss"]}))
