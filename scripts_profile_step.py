import time, torch
from flaxdiff_amd.models import Unet
from flaxdiff_amd.predictors import KarrasPredictionTransform
from flaxdiff_amd.schedulers import EDMNoiseScheduler
from flaxdiff_amd.trainer import DiffusionTrainer
from flaxdiff_amd import ops

torch.manual_seed(0)
dev = torch.device('cuda:0')
model = Unet(emb_features=256, feature_depths=[64,128,256,512],
             attention_configs=[{"heads":4}]*4, num_res_blocks=2,
             num_middle_res_blocks=1, norm_groups=8, context_dim=768).to(dev)
print("freqs device after .to:", model.time_embed.freqs.device)
tr = DiffusionTrainer(model, EDMNoiseScheduler(1, sigma_max=80, sigma_data=0.5),
                      KarrasPredictionTransform(sigma_data=0.5), name="p",
                      checkpoint_base_path="/tmp/x", compute_dtype=torch.bfloat16,
                      distributed=False)
print("freqs device after trainer:", tr.model.time_embed.freqs.device)
B = 64
batch = {"image": torch.randint(0,255,(B,64,64,3),dtype=torch.uint8).to(dev)}

# granular timing via monkeypatched phases
import flaxdiff_amd.trainer.diffusion_trainer as dt
orig_backward = torch.Tensor.backward
def sync(): torch.cuda.synchronize()

out = tr.train_step(batch)
print("first step ok", out)

# time phases by patching
t0=time.perf_counter(); sync()
for _ in range(2):
    tr.optimizer.zero_grad()
sync(); print("zero_grad x2: %.1f ms" % ((time.perf_counter()-t0)*500))

import flaxdiff_amd.utils as U
images = (batch["image"].to(torch.bfloat16)-127.5)/127.5
text = tr.null_context.unsqueeze(0).expand(B, *tr.null_context.shape)
timesteps, _ = tr.noise_schedule.generate_timesteps(B, tr.rngs, device=dev)
rates = tuple(r.to(dev) for r in tr.noise_schedule.get_rates(timesteps, U.get_coeff_shapes_tuple(images)))
noise = torch.randn_like(images)
x_t, c_in, target = tr.model_output_transform.forward_diffusion(images, noise, rates)
x_in, t_in = tr.noise_schedule.transform_inputs(x_t * c_in.to(x_t.dtype), timesteps)

sync(); t0=time.perf_counter()
pred = tr.model(x_in, t_in.to(dev), text)
sync(); print("fwd: %.1f ms" % ((time.perf_counter()-t0)*1000))

pred2 = tr.model_output_transform.pred_transform(x_t, pred, rates)
w = tr.noise_schedule.get_weights(timesteps, U.get_coeff_shapes_tuple(images)).to(dev)
loss = (dt.l2_loss(pred2.float(), target.float()) * w).mean()
sync(); t0=time.perf_counter()
loss.backward()
sync(); print("bwd: %.1f ms" % ((time.perf_counter()-t0)*1000))

sync(); t0=time.perf_counter()
tr.optimizer.step(grad_scale=1.0)
sync(); print("opt: %.1f ms" % ((time.perf_counter()-t0)*1000))

sync(); t0=time.perf_counter()
out = tr.train_step(batch)
sync(); print("full train_step: %.1f ms" % ((time.perf_counter()-t0)*1000))
