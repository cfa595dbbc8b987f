"""Prediction transforms: what the model predicts and how x0/eps are recovered.

Math contract: reference /root/reference/flaxdiff/predictors/__init__.py:9-96.
On GPU the c_in / c_skip / c_out scalings and the forward-diffusion axpy are
folded into fused HIP elementwise kernels (ops.fused.forward_diffusion); the
classes here define the math and serve as the CPU oracle.
"""
from __future__ import annotations

from typing import Tuple

import torch

from ..schedulers import NoiseScheduler, get_coeff_shapes_tuple

__all__ = [
    "DiffusionPredictionTransform",
    "EpsilonPredictionTransform",
    "DirectPredictionTransform",
    "VPredictionTransform",
    "KarrasPredictionTransform",
]


class DiffusionPredictionTransform:
    """Base transform (predictors/__init__.py:9-33)."""

    def pred_transform(self, x_t, preds, rates) -> torch.Tensor:
        return preds

    def __call__(self, x_t, preds, current_step, noise_schedule: NoiseScheduler):
        rates = noise_schedule.get_rates(current_step, shape=get_coeff_shapes_tuple(x_t))
        rates = tuple(r.to(device=x_t.device, dtype=x_t.dtype) for r in rates)
        preds = self.pred_transform(x_t, preds, rates)
        x_0, epsilon = self.backward_diffusion(x_t, preds, rates)
        return x_0, epsilon

    def forward_diffusion(self, x_0, epsilon, rates: Tuple[torch.Tensor, torch.Tensor]):
        """x_t = a*x0 + s*eps; returns (x_t, c_in, target) (:19-24)."""
        signal_rate, noise_rate = rates
        signal_rate = signal_rate.to(x_0.dtype)
        noise_rate = noise_rate.to(x_0.dtype)
        x_t = signal_rate * x_0 + noise_rate * epsilon
        expected_output = self.get_target(x_0, epsilon, (signal_rate, noise_rate))
        c_in = self.get_input_scale((signal_rate, noise_rate))
        return x_t, c_in, expected_output

    def backward_diffusion(self, x_t, preds, rates):
        raise NotImplementedError

    def get_target(self, x_0, epsilon, rates) -> torch.Tensor:
        return x_0

    def get_input_scale(self, rates):
        return 1


class EpsilonPredictionTransform(DiffusionPredictionTransform):
    """Model predicts the noise (:35-44)."""

    def backward_diffusion(self, x_t, preds, rates):
        signal_rates, noise_rates = rates
        epsilon = preds
        x_0 = (x_t - epsilon * noise_rates) / signal_rates
        return x_0, epsilon

    def get_target(self, x_0, epsilon, rates):
        return epsilon


class DirectPredictionTransform(DiffusionPredictionTransform):
    """Model predicts x_0 directly (:46-52)."""

    def backward_diffusion(self, x_t, preds, rates):
        signal_rate, noise_rate = rates
        x_0 = preds
        epsilon = (x_t - x_0 * signal_rate) / noise_rate
        return x_0, epsilon


class VPredictionTransform(DiffusionPredictionTransform):
    """v-prediction (:54-71)."""

    def backward_diffusion(self, x_t, preds, rates):
        signal_rate, noise_rate = rates
        variance = signal_rate ** 2 + noise_rate ** 2
        v = preds * torch.sqrt(variance)
        x_0 = signal_rate * x_t - noise_rate * v
        eps_0 = signal_rate * v + noise_rate * x_t
        return x_0 / variance, eps_0 / variance

    def get_target(self, x_0, epsilon, rates):
        signal_rate, noise_rate = rates
        v = signal_rate * epsilon - noise_rate * x_0
        variance = signal_rate ** 2 + noise_rate ** 2
        return v / torch.sqrt(variance)


class KarrasPredictionTransform(DiffusionPredictionTransform):
    """EDM preconditioning: c_in / c_skip / c_out (:73-96)."""

    def __init__(self, sigma_data=0.5):
        super().__init__()
        self.sigma_data = sigma_data

    def backward_diffusion(self, x_t, preds, rates):
        signal_rate, noise_rate = rates
        x_0 = preds
        epsilon = (x_t - x_0 * signal_rate) / noise_rate
        return x_0, epsilon

    def pred_transform(self, x_t, preds, rates, epsilon=1e-8):
        _, sigma = rates
        # scalar arithmetic only — graph-capture-safe (no H2D tensor creation)
        c_out = sigma * self.sigma_data / (torch.sqrt(sigma ** 2 + self.sigma_data ** 2) + epsilon)
        c_skip = self.sigma_data ** 2 / (self.sigma_data ** 2 + sigma ** 2 + epsilon)
        c_out = c_out.reshape(get_coeff_shapes_tuple(preds))
        c_skip = c_skip.reshape(get_coeff_shapes_tuple(x_t))
        return c_out * preds + c_skip * x_t

    def get_input_scale(self, rates, epsilon=1e-8):
        _, sigma = rates
        # scalar arithmetic only — graph-capture-safe (no H2D tensor creation)
        return 1 / (torch.sqrt(sigma ** 2 + self.sigma_data ** 2) + epsilon)
