"""DDPM samplers (reference: samplers/ddpm.py:5-36)."""
import torch

from ..utils import RandomMarkovState
from .common import DiffusionSampler


class DDPMSampler(DiffusionSampler):
    """Ancestral posterior sampling with precomputed posterior mean/var."""

    def take_next_step(self, current_samples, reconstructed_samples,
                       model_conditioning_inputs, pred_noise, current_step,
                       state: RandomMarkovState, sample_model_fn, next_step=1):
        mean = self.noise_schedule.get_posterior_mean(reconstructed_samples,
                                                      current_samples, current_step)
        variance = self.noise_schedule.get_posterior_variance(steps=current_step)
        variance = variance.to(mean.device, mean.dtype)
        state, key = state.get_random_key()
        noise = key.normal(reconstructed_samples.shape, device=mean.device).to(mean.dtype)
        return mean + noise * variance, state


class SimpleDDPMSampler(DiffusionSampler):
    """Rate-space DDPM reformulation (ddpm.py:20-36)."""

    def take_next_step(self, current_samples, reconstructed_samples,
                       model_conditioning_inputs, pred_noise, current_step,
                       state: RandomMarkovState, sample_model_fn, next_step=1):
        state, key = state.get_random_key()
        noise = key.normal(reconstructed_samples.shape,
                           device=current_samples.device).to(current_samples.dtype)

        a_t, s_t = self.noise_schedule.get_rates(current_step)
        a_n, s_n = self.noise_schedule.get_rates(next_step)
        dev, dt = current_samples.device, current_samples.dtype
        a_t, s_t, a_n, s_n = (r.to(dev, dt) for r in (a_t, s_t, a_n, s_n))

        pred_noise_coeff = ((s_n ** 2) * a_t) / (s_t * a_n)
        noise_ratio_sq = (s_n ** 2) / (s_t ** 2)
        signal_ratio_sq = (a_t ** 2) / (a_n ** 2)
        gamma = torch.sqrt(noise_ratio_sq * (1 - signal_ratio_sq))

        next_samples = a_n * reconstructed_samples + pred_noise_coeff * pred_noise \
            + noise * gamma
        return next_samples, state
