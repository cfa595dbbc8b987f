"""Cosine schedules (reference: schedulers/cosine.py:8-39)."""
import math

import numpy as np
import torch

from .common import GeneralizedNoiseScheduler, reshape_rates
from .continuous import ContinuousNoiseScheduler
from .discrete import DiscreteNoiseScheduler


def cosine_beta_schedule(timesteps, start_angle=0.008, end_angle=0.999):
    """cosine.py:8-13 — Nichol & Dhariwal cosine alpha-bar schedule."""
    ts = np.linspace(0, 1, timesteps + 1, dtype=np.float64)
    alphas_bar = np.cos((ts + start_angle) / (1 + start_angle) * np.pi / 2) ** 2
    alphas_bar = alphas_bar / alphas_bar[0]
    betas = 1 - (alphas_bar[1:] / alphas_bar[:-1])
    return np.clip(betas, 0, end_angle)


class CosineNoiseScheduler(DiscreteNoiseScheduler):
    def __init__(self, timesteps, beta_start=0.008, beta_end=0.999, *args, **kwargs):
        super().__init__(timesteps, beta_start, beta_end, schedule_fn=cosine_beta_schedule,
                         *args, **kwargs)


class CosineGeneralNoiseScheduler(GeneralizedNoiseScheduler):
    """cosine.py:19-29 — sigma(t) = tan(theta_min + t*(theta_max-theta_min))/kappa."""

    def __init__(self, sigma_min=0.02, sigma_max=80.0, kappa=1.0, *args, **kwargs):
        kwargs.pop("timesteps", None)
        super().__init__(timesteps=1, sigma_min=sigma_min, sigma_max=sigma_max, *args, **kwargs)
        self.kappa = kappa
        logsnr_max = 2 * (math.log(kappa) - math.log(sigma_max))
        self.theta_max = math.atan(math.exp(-0.5 * logsnr_max))
        logsnr_min = 2 * (math.log(kappa) - math.log(sigma_min))
        self.theta_min = math.atan(math.exp(-0.5 * logsnr_min))

    def get_sigmas(self, steps):
        if not torch.is_tensor(steps):
            steps = torch.as_tensor(steps, dtype=torch.float32)
        return torch.tan(self.theta_min
                         + steps.float() * (self.theta_max - self.theta_min)) / self.kappa


class CosineContinuousNoiseScheduler(ContinuousNoiseScheduler):
    """cosine.py:31-39 — alpha=cos(pi t / 2), sigma=sin(pi t / 2)."""

    def get_rates(self, steps, shape=(-1, 1, 1, 1)):
        if not torch.is_tensor(steps):
            steps = torch.as_tensor(steps, dtype=torch.float32)
        steps = steps.float()
        signal_rates = torch.cos((math.pi * steps) / (2 * self.max_timesteps))
        noise_rates = torch.sin((math.pi * steps) / (2 * self.max_timesteps))
        return reshape_rates((signal_rates, noise_rates), shape=shape)

    def get_weights(self, steps, shape=(-1, 1, 1, 1)):
        alpha, sigma = self.get_rates(steps, shape=shape)
        return 1 / (1 + (alpha ** 2 / sigma ** 2))
