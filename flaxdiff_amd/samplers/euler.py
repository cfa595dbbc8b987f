"""Euler-family ODE/SDE samplers (reference: samplers/euler.py:6-55)."""

from ..utils import RandomMarkovState
from .common import DiffusionSampler


def _rates(ns, step, dev, dt):
    a, s = ns.get_rates(step)
    return a.to(dev, dt), s.to(dev, dt)


class EulerSampler(DiffusionSampler):
    """DDIM parameterized as an ODE (euler.py:6-18)."""

    def take_next_step(self, current_samples, reconstructed_samples,
                       model_conditioning_inputs, pred_noise, current_step,
                       state: RandomMarkovState, sample_model_fn, next_step=1):
        dev, dt = current_samples.device, current_samples.dtype
        a_t, s_t = _rates(self.noise_schedule, current_step, dev, dt)
        a_n, s_n = _rates(self.noise_schedule, next_step, dev, dt)
        dtau = s_n - s_t
        x0_coeff = (a_t * s_n - a_n * s_t) / dtau
        dx = (current_samples - x0_coeff * reconstructed_samples) / s_t
        return current_samples + dx * dtau, state


class SimplifiedEulerSampler(DiffusionSampler):
    """VE-simplified Euler (euler.py:20-33)."""

    def take_next_step(self, current_samples, reconstructed_samples,
                       model_conditioning_inputs, pred_noise, current_step,
                       state: RandomMarkovState, sample_model_fn, next_step=1):
        dev, dt = current_samples.device, current_samples.dtype
        _, s_t = _rates(self.noise_schedule, current_step, dev, dt)
        _, s_n = _rates(self.noise_schedule, next_step, dev, dt)
        dtau = s_n - s_t
        dx = (current_samples - reconstructed_samples) / s_t
        return current_samples + dx * dtau, state


class EulerAncestralSampler(DiffusionSampler):
    """Euler with ancestral noise injection (euler.py:35-55) — the reference's
    production sampler for text2img."""

    def take_next_step(self, current_samples, reconstructed_samples,
                       model_conditioning_inputs, pred_noise, current_step,
                       state: RandomMarkovState, sample_model_fn, next_step=1):
        dev, dt = current_samples.device, current_samples.dtype
        a_t, s_t = _rates(self.noise_schedule, current_step, dev, dt)
        a_n, s_n = _rates(self.noise_schedule, next_step, dev, dt)

        sigma_up = (s_n ** 2 * (s_t ** 2 - s_n ** 2) / s_t ** 2) ** 0.5
        sigma_down = (s_n ** 2 - sigma_up ** 2) ** 0.5
        dtau = sigma_down - s_t

        x0_coeff = (a_t * s_n - a_n * s_t) / (s_n - s_t)
        dx = (current_samples - x0_coeff * reconstructed_samples) / s_t

        state, key = state.get_random_key()
        dW = key.normal(current_samples.shape, device=dev).to(dt) * sigma_up
        return current_samples + dx * dtau + dW, state
