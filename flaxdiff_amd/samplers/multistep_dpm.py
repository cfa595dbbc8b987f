"""Multistep DPM sampler with host-side history (reference: samplers/multistep_dpm.py:6-58)."""

from ..utils import RandomMarkovState
from .common import DiffusionSampler


class MultiStepDPM(DiffusionSampler):
    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)
        self.history = []

    def take_next_step(self, current_samples, reconstructed_samples,
                       model_conditioning_inputs, pred_noise, current_step,
                       state: RandomMarkovState, sample_model_fn, next_step=1):
        dev, dt = current_samples.device, current_samples.dtype
        a_t, s_t = self.noise_schedule.get_rates(current_step)
        a_n, s_n = self.noise_schedule.get_rates(next_step)
        s_t, s_n = s_t.to(dev, dt), s_n.to(dev, dt)
        dtau = s_n - s_t

        def second_order(eps, sigma, last_eps, last_sigma):
            return (eps - last_eps) / (sigma - last_sigma)

        if len(self.history) == 0:
            next_samples = current_samples + pred_noise * dtau
        elif len(self.history) == 1:
            last = self.history[-1]
            dx_2 = second_order(pred_noise, s_t, last["eps"], last["sigma"])
            next_samples = current_samples + pred_noise * dtau + 0.5 * dx_2 * dtau ** 2
        else:
            last = self.history[-1]
            second_last = self.history[-2]
            dx_2 = second_order(pred_noise, s_t, last["eps"], last["sigma"])
            dx_2_last = second_order(last["eps"], last["sigma"],
                                     second_last["eps"], second_last["sigma"])
            denom = 0.5 * ((s_t + last["sigma"]) - (last["sigma"] + second_last["sigma"]))
            dx_3 = (dx_2 - dx_2_last) / denom
            next_samples = current_samples + pred_noise * dtau + 0.5 * dx_2 * dtau ** 2 \
                + (1.0 / 6.0) * dx_3 * dtau ** 3

        self.history.append({"eps": pred_noise, "sigma": s_t})
        return next_samples, state
