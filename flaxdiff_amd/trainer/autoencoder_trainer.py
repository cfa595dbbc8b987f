"""AutoEncoderTrainer — reconstruction training for the VAE
(reference: /root/reference/flaxdiff/trainer/autoencoder_trainer.py:20-181;
the reference version is partial/legacy — this one trains a plain
reconstruction + KL objective on the SimpleTrainer loop)."""
from __future__ import annotations

from typing import Dict

import torch

from .simple_trainer import SimpleTrainer


class AutoEncoderTrainer(SimpleTrainer):
    def __init__(self, model, *, kl_weight: float = 1e-6, **kwargs):
        kwargs.setdefault("name", "AutoEncoder")
        super().__init__(model, **kwargs)
        self.kl_weight = kl_weight

    def train_step(self, batch: Dict[str, torch.Tensor]) -> Dict[str, float]:
        images = batch["image"].to(self.device)
        if images.dtype == torch.uint8:
            images = images.to(self.compute_dtype)
            images = (images - 127.5) / 127.5
        else:
            images = images.to(self.compute_dtype)

        self.optimizer.zero_grad()
        out = self.model(images)
        if isinstance(out, tuple):
            recon, mean, logvar = out
            kl = -0.5 * torch.mean(1 + logvar - mean ** 2 - logvar.exp())
        else:
            recon, kl = out, torch.zeros((), device=self.device)
        rec_loss = torch.nn.functional.mse_loss(recon.float(), images.float())
        loss = rec_loss + self.kl_weight * kl
        loss.backward()
        self.grad_sync.sync()
        self.optimizer.step(grad_scale=1.0 / self.dist.world_size)
        return {"loss": float(loss.detach()), "rec_loss": float(rec_loss.detach()),
                "kl": float(kl.detach())}
