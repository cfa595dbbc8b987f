"""Heun 2nd-order sampler — 2 NFE/step (reference: samplers/heun_sampler.py:6-27)."""

from ..utils import RandomMarkovState
from .common import DiffusionSampler


class HeunSampler(DiffusionSampler):
    def take_next_step(self, current_samples, reconstructed_samples,
                       model_conditioning_inputs, pred_noise, current_step,
                       state: RandomMarkovState, sample_model_fn, next_step=1):
        dev, dt = current_samples.device, current_samples.dtype
        a_t, s_t = self.noise_schedule.get_rates(current_step)
        a_n, s_n = self.noise_schedule.get_rates(next_step)
        a_t, s_t, a_n, s_n = (r.to(dev, dt) for r in (a_t, s_t, a_n, s_n))

        dtau = s_n - s_t
        x0_coeff = (a_t * s_n - a_n * s_t) / dtau

        dx_0 = (current_samples - x0_coeff * reconstructed_samples) / s_t
        next_samples_0 = current_samples + dx_0 * dtau

        # second model evaluation at the predicted point
        estimated_x0, _, _ = sample_model_fn(next_samples_0, next_step,
                                             *model_conditioning_inputs)
        dx_1 = (next_samples_0 - x0_coeff * estimated_x0) / s_n
        return current_samples + 0.5 * (dx_0 + dx_1) * dtau, state
