"""Noise-scheduler base classes.

Math contract follows the reference (/root/reference/flaxdiff/schedulers/common.py:16-102)
exactly; implementation is torch-native with device-resident coefficient
tables. Rate lookups return fp32 tensors broadcastable against NHWC batches.
"""
from __future__ import annotations

from typing import Tuple

import torch

from ..utils import RandomMarkovState, get_coeff_shapes_tuple


def reshape_rates(rates: Tuple[torch.Tensor, torch.Tensor], shape=(-1, 1, 1, 1)):
    """Reference: schedulers/common.py:10-14."""
    signal_rates, noise_rates = rates
    return signal_rates.reshape(shape), noise_rates.reshape(shape)


def _as_tensor(x, device=None) -> torch.Tensor:
    if not torch.is_tensor(x):
        x = torch.as_tensor(x, dtype=torch.float32)
    if device is not None and x.device != device:
        x = x.to(device)
    return x.float()


class NoiseScheduler:
    """Base noise scheduler (reference: schedulers/common.py:16-64).

    Subclasses define `get_rates` (signal/noise rate per timestep) and may
    define posterior statistics for ancestral sampling.
    """

    def __init__(self, timesteps, dtype=torch.float32, clip_min=-1.0, clip_max=1.0,
                 *args, **kwargs):
        self.max_timesteps = timesteps
        self.dtype = dtype
        self.clip_min = clip_min
        self.clip_max = clip_max

    # -- timestep sampling ---------------------------------------------------
    def generate_timesteps(self, batch_size, state: RandomMarkovState,
                           device=None) -> Tuple[torch.Tensor, RandomMarkovState]:
        state, key = state.get_random_key()
        if isinstance(self.max_timesteps, int) and self.max_timesteps > 1:
            timesteps = key.randint((batch_size,), 0, self.max_timesteps, device=device)
        else:
            timesteps = key.uniform((batch_size,), 0.0, float(self.max_timesteps), device=device)
        return timesteps, state

    def sample_timesteps_device(self, batch_size, device) -> torch.Tensor:
        """Graph-safe timestep draw: same distribution as generate_timesteps
        but through torch's default (capture-aware) CUDA generator — used by
        the hipGraph-captured train step, where per-call seeded Generators
        would bake one fixed draw into the graph."""
        if isinstance(self.max_timesteps, int) and self.max_timesteps > 1:
            return torch.randint(0, self.max_timesteps, (batch_size,),
                                 device=device)
        return torch.rand(batch_size, device=device) * float(self.max_timesteps)

    # -- rates / weights -----------------------------------------------------
    def get_weights(self, steps, shape=(-1, 1, 1, 1)) -> torch.Tensor:
        raise NotImplementedError

    def get_rates(self, steps, shape=(-1, 1, 1, 1)) -> Tuple[torch.Tensor, torch.Tensor]:
        raise NotImplementedError

    def add_noise(self, images, noise, steps) -> torch.Tensor:
        """q(x_t | x_0) sample: signal*x0 + noise_rate*eps (common.py:43-45)."""
        signal_rates, noise_rates = self.get_rates(steps, shape=get_coeff_shapes_tuple(images))
        return signal_rates.to(images.dtype) * images + noise_rates.to(images.dtype) * noise

    def remove_all_noise(self, noisy_images, noise, steps, clip_denoised=True, rates=None):
        signal_rates, noise_rates = self.get_rates(
            steps, shape=get_coeff_shapes_tuple(noisy_images))
        return (noisy_images - noise * noise_rates) / signal_rates

    def transform_inputs(self, x, steps):
        return x, steps

    def get_posterior_mean(self, x_0, x_t, steps):
        raise NotImplementedError

    def get_posterior_variance(self, steps, shape=(-1, 1, 1, 1)):
        raise NotImplementedError

    def get_max_variance(self, shape=(-1, 1, 1, 1), device=None):
        """sqrt(alpha_T^2 + sigma_T^2) — noise-init scale (common.py:61-64)."""
        steps = _as_tensor([float(self.max_timesteps)], device)
        alpha_n, sigma_n = self.get_rates(steps, shape=shape)
        return torch.sqrt(alpha_n ** 2 + sigma_n ** 2)


class GeneralizedNoiseScheduler(NoiseScheduler):
    """EDM-style schedule: signal rate == 1, parameterized by sigma
    (reference: schedulers/common.py:66-102)."""

    def __init__(self, timesteps, sigma_min=0.002, sigma_max=80.0, sigma_data=1.0,
                 *args, **kwargs):
        super().__init__(timesteps, *args, **kwargs)
        self.sigma_min = sigma_min
        self.sigma_max = sigma_max
        self.sigma_data = sigma_data

    def get_weights(self, steps, shape=(-1, 1, 1, 1)):
        """common.py:80-82 (kept verbatim incl. the sigma_max^2 denominator)."""
        sigma = self.get_sigmas(steps)
        w = 1 + (1 / (1 + ((1 - sigma ** 2) / (sigma ** 2)))) / (self.sigma_max ** 2)
        return w.reshape(shape)

    def get_sigmas(self, steps) -> torch.Tensor:
        raise NotImplementedError

    def get_rates(self, steps, shape=(-1, 1, 1, 1)):
        sigmas = self.get_sigmas(steps)
        signal_rates = torch.ones_like(sigmas)
        return reshape_rates((signal_rates, sigmas), shape=shape)

    def transform_inputs(self, x, steps, num_discrete_chunks=1000):
        """common.py:94-97: map t to integer chunks for the model's t input."""
        steps = _as_tensor(steps, device=x.device if torch.is_tensor(x) else None)
        sigmas_discrete = (steps / self.max_timesteps) * num_discrete_chunks
        return x, sigmas_discrete.to(torch.int32)

    def get_timesteps(self, sigmas) -> torch.Tensor:
        raise NotImplementedError
