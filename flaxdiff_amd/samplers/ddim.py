"""DDIM sampler (reference: samplers/ddim.py:7-49)."""
import torch

from ..utils import RandomMarkovState, get_coeff_shapes_tuple
from .common import DiffusionSampler


class DDIMSampler(DiffusionSampler):
    def __init__(self, *args, eta: float = 0.0, **kwargs):
        super().__init__(*args, **kwargs)
        self.eta = eta

    def take_next_step(self, current_samples, reconstructed_samples,
                       model_conditioning_inputs, pred_noise, current_step,
                       state: RandomMarkovState, sample_model_fn, next_step=1):
        shape = get_coeff_shapes_tuple(current_samples)
        a_t, s_t = self.noise_schedule.get_rates(current_step, shape)
        a_n, s_n = self.noise_schedule.get_rates(next_step, shape)
        dev, dt = current_samples.device, current_samples.dtype
        a_t, s_t, a_n, s_n = (r.to(dev, dt) for r in (a_t, s_t, a_n, s_n))

        if self.eta > 0:
            sigma_tilde = self.eta * s_n * torch.sqrt(1 - a_t ** 2 / a_n ** 2) \
                / torch.sqrt(1 - a_t ** 2)
            state, key = state.get_random_key()
            noise = key.normal(current_samples.shape, device=dev).to(dt)
            stochastic = sigma_tilde * noise
        else:
            stochastic = 0
        new_samples = a_n * reconstructed_samples + s_n * pred_noise + stochastic
        return new_samples, state
