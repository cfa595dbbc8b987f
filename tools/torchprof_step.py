"""torch.profiler one bench step on GPU: top CUDA ops with shapes."""
import torch
from torch.profiler import profile, ProfilerActivity
from flaxdiff_amd.models import Unet
from flaxdiff_amd.schedulers import EDMNoiseScheduler
from flaxdiff_amd.predictors import KarrasPredictionTransform
from flaxdiff_amd.trainer import DiffusionTrainer

model = Unet(emb_features=256, feature_depths=[64, 128, 256, 512],
             attention_configs=[{"heads": 4}] * 4, num_res_blocks=2,
             num_middle_res_blocks=1, norm_groups=8, context_dim=768)
tr = DiffusionTrainer(model, EDMNoiseScheduler(1, sigma_max=80),
                      KarrasPredictionTransform(sigma_data=0.5),
                      name="prof", checkpoint_base_path="/tmp/profck",
                      compute_dtype=torch.bfloat16, distributed=False)
batch = {"image": torch.randint(0, 255, (256, 64, 64, 3), dtype=torch.uint8)}
for _ in range(3):
    tr.train_step(batch)
torch.cuda.synchronize()
with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
             record_shapes=True) as prof:
    tr.train_step(batch)
    torch.cuda.synchronize()
print(prof.key_averages(group_by_input_shape=True).table(
    sort_by="cuda_time_total", row_limit=40, max_src_column_width=40))
