"""Karras / EDM schedulers (reference: schedulers/karras.py:7-83)."""
import math

import torch

from ..utils import RandomMarkovState
from .common import GeneralizedNoiseScheduler


def _t(steps):
    if not torch.is_tensor(steps):
        steps = torch.as_tensor(steps, dtype=torch.float32)
    return steps.float()


class KarrasVENoiseScheduler(GeneralizedNoiseScheduler):
    """karras.py:7-56 — rho-spaced sigma ramp; c_noise = log(sigma)/4."""

    def __init__(self, timesteps=1.0, sigma_min=0.002, sigma_max=80, rho=7.0,
                 sigma_data=0.5, *args, **kwargs):
        super().__init__(timesteps=timesteps, sigma_min=sigma_min, sigma_max=sigma_max,
                         sigma_data=sigma_data, *args, **kwargs)
        self.min_inv_rho = sigma_min ** (1 / rho)
        self.max_inv_rho = sigma_max ** (1 / rho)
        self.rho = rho

    def get_sigmas(self, steps):
        ramp = torch.clamp(1 - _t(steps) / self.max_timesteps, 0.0, 1.0)
        return (self.max_inv_rho + ramp * (self.min_inv_rho - self.max_inv_rho)) ** self.rho

    def get_weights(self, steps, shape=(-1, 1, 1, 1)):
        """karras.py:20-26 — EDM lambda(sigma) = (sigma^2+sd^2)/(sigma*sd)^2."""
        sigma = self.get_sigmas(steps)
        eps = 1e-6
        w = (sigma ** 2 + self.sigma_data ** 2) / ((sigma * self.sigma_data) ** 2 + eps)
        return w.reshape(shape)

    def transform_inputs(self, x, steps, num_discrete_chunks=1000):
        """karras.py:28-33 — c_noise = log(sigma)/4."""
        sigmas = self.get_sigmas(_t(steps))
        return x, torch.log(sigmas + 1e-12) / 4

    def get_timesteps(self, sigmas):
        """karras.py:35-46 — inverse of get_sigmas."""
        sigmas = _t(sigmas).reshape(-1)
        inv_rho = (sigmas + 1e-12) ** (1 / self.rho)
        denom = self.min_inv_rho - self.max_inv_rho
        if abs(denom) < 1e-7:
            denom = math.copysign(1e-7, denom)
        ramp = torch.clamp((inv_rho - self.max_inv_rho) / denom, 0.0, 1.0)
        return torch.clamp(1 - ramp, 0.0, 1.0) * self.max_timesteps

    def generate_timesteps(self, batch_size, state: RandomMarkovState, device=None):
        timesteps, state = super().generate_timesteps(batch_size, state, device=device)
        return timesteps.float(), state

    def sample_timesteps_device(self, batch_size, device) -> torch.Tensor:
        return super().sample_timesteps_device(batch_size, device).float()


class SimpleExpNoiseScheduler(KarrasVENoiseScheduler):
    """karras.py:52-63 — log-spaced sigma table indexed by integer steps."""

    def __init__(self, timesteps, sigma_min=0.002, sigma_max=80, rho=7.0,
                 sigma_data=0.5, *args, **kwargs):
        super().__init__(timesteps=timesteps, sigma_min=sigma_min, sigma_max=sigma_max,
                         rho=rho, sigma_data=sigma_data, *args, **kwargs)
        n = timesteps if isinstance(timesteps, int) and timesteps > 1 else 1000
        self.sigmas_table = torch.exp(torch.linspace(math.log(sigma_min), math.log(sigma_max), n))

    def get_sigmas(self, steps):
        idx = _t(steps).long().clamp(0, self.sigmas_table.numel() - 1)
        return self.sigmas_table.to(idx.device)[idx]


class EDMNoiseScheduler(KarrasVENoiseScheduler):
    """karras.py:65-83 — sigma(t)=exp(1.2*t - 1.2) with t ~ N(0,1) at train time."""

    def get_sigmas(self, steps, std=1.2, mean=-1.2):
        space = _t(steps) / self.max_timesteps
        return torch.exp(space * std + mean)

    def generate_timesteps(self, batch_size, state: RandomMarkovState, device=None):
        state, key = state.get_random_key()
        timesteps = key.normal((batch_size,), device=device)
        return timesteps, state

    def sample_timesteps_device(self, batch_size, device) -> torch.Tensor:
        return torch.randn(batch_size, device=device)
