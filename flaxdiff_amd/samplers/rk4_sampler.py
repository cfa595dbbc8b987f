"""RK4 sampler — 4 NFE/step (reference: samplers/rk4_sampler.py:7-33)."""
import torch

from ..schedulers import GeneralizedNoiseScheduler
from ..utils import MarkovState
from .common import DiffusionSampler


class RK4Sampler(DiffusionSampler):
    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)
        assert isinstance(self.noise_schedule, GeneralizedNoiseScheduler), \
            "Noise schedule must be a GeneralizedNoiseScheduler"

    def _derivative(self, sample_model_fn, x_t, sigma, model_conditioning_inputs):
        t = self.noise_schedule.get_timesteps(sigma)
        _, eps, _ = sample_model_fn(x_t, t.to(x_t.device), *model_conditioning_inputs)
        return eps

    def sample_step(self, sample_model_fn, current_samples, current_step,
                    model_conditioning_inputs, next_step=None, state: MarkovState = None):
        B = current_samples.shape[0]
        dev, dt = current_samples.device, current_samples.dtype
        ones = torch.ones(B, device=dev)
        cur = ones * float(current_step)
        nxt = ones * float(next_step)
        _, s_t = self.noise_schedule.get_rates(cur)
        _, s_n = self.noise_schedule.get_rates(nxt)
        s_t, s_n = s_t.to(dev, dt), s_n.to(dev, dt)
        dtau = s_n - s_t

        k1 = self._derivative(sample_model_fn, current_samples, s_t, model_conditioning_inputs)
        k2 = self._derivative(sample_model_fn, current_samples + 0.5 * k1 * dtau,
                              s_t + 0.5 * dtau, model_conditioning_inputs)
        k3 = self._derivative(sample_model_fn, current_samples + 0.5 * k2 * dtau,
                              s_t + 0.5 * dtau, model_conditioning_inputs)
        k4 = self._derivative(sample_model_fn, current_samples + k3 * dtau,
                              s_t + dtau, model_conditioning_inputs)

        next_samples = current_samples + ((k1 + 2 * k2 + 2 * k3 + k4) * dtau) / 6
        return next_samples, state
