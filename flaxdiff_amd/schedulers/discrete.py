"""Discrete variance-preserving scheduler.

Coefficient tables precomputed once on host in fp64, held as fp32 tensors and
moved to device on first use — the GPU train step then does a pure gather
(the MI355X plan from SURVEY.md §2.1). Math: reference
/root/reference/flaxdiff/schedulers/discrete.py:7-71.
"""
from __future__ import annotations


import numpy as np
import torch

from ..utils import RandomMarkovState, get_coeff_shapes_tuple
from .common import NoiseScheduler, reshape_rates

_TABLE_NAMES = (
    "betas", "alphas", "alpha_cumprod", "alpha_cumprod_prev",
    "sqrt_alpha_cumprod", "sqrt_one_minus_alpha_cumprod",
    "posterior_variance", "posterior_log_variance_clipped",
    "posterior_mean_coef1", "posterior_mean_coef2", "p2_loss_weights",
)


class DiscreteNoiseScheduler(NoiseScheduler):
    """Variance preserving: signal_rate^2 + noise_rate^2 = 1 (discrete.py:7-41)."""

    def __init__(self, timesteps, beta_start=0.0001, beta_end=0.02, schedule_fn=None,
                 p2_loss_weight_k: float = 1, p2_loss_weight_gamma: float = 1,
                 *args, **kwargs):
        super().__init__(timesteps, *args, **kwargs)
        betas = np.asarray(schedule_fn(timesteps, beta_start, beta_end), dtype=np.float64)
        alphas = 1.0 - betas
        alpha_cumprod = np.cumprod(alphas, axis=0)
        alpha_cumprod_prev = np.append(1.0, alpha_cumprod[:-1])

        posterior_variance = betas * (1 - alpha_cumprod_prev) / (1 - alpha_cumprod)
        tables = {
            "betas": betas,
            "alphas": alphas,
            "alpha_cumprod": alpha_cumprod,
            "alpha_cumprod_prev": alpha_cumprod_prev,
            "sqrt_alpha_cumprod": np.sqrt(alpha_cumprod),
            "sqrt_one_minus_alpha_cumprod": np.sqrt(1 - alpha_cumprod),
            "posterior_variance": posterior_variance,
            "posterior_log_variance_clipped": np.log(np.maximum(posterior_variance, 1e-20)),
            "posterior_mean_coef1": betas * np.sqrt(alpha_cumprod_prev) / (1 - alpha_cumprod),
            "posterior_mean_coef2":
                (1 - alpha_cumprod_prev) * np.sqrt(alphas) / (1 - alpha_cumprod),
            "p2_loss_weights":
                (p2_loss_weight_k + alpha_cumprod / (1 - alpha_cumprod))
                ** (-p2_loss_weight_gamma),
        }
        self._tables = {k: torch.from_numpy(np.ascontiguousarray(v)).float()
                        for k, v in tables.items()}
        self._tables_device = torch.device("cpu")

    # -- device management ---------------------------------------------------
    def _table(self, name: str, device) -> torch.Tensor:
        if device is not None and torch.device(device) != self._tables_device:
            self._tables = {k: v.to(device) for k, v in self._tables.items()}
            self._tables_device = torch.device(device)
        return self._tables[name]

    def __getattr__(self, name):
        tables = self.__dict__.get("_tables")
        if tables is not None and name in tables:
            return tables[name]
        raise AttributeError(name)

    def _steps_index(self, steps, device=None) -> torch.Tensor:
        if not torch.is_tensor(steps):
            steps = torch.as_tensor(steps)
        idx = steps.long().clamp_(0, self.max_timesteps - 1)
        return idx.to(device) if device is not None else idx

    # -- API -----------------------------------------------------------------
    def generate_timesteps(self, batch_size, state: RandomMarkovState, device=None):
        state, key = state.get_random_key()
        timesteps = key.randint((batch_size,), 0, self.max_timesteps, device=device)
        return timesteps, state

    def sample_timesteps_device(self, batch_size, device) -> torch.Tensor:
        return torch.randint(0, self.max_timesteps, (batch_size,), device=device)

    def get_p2_weights(self, k, gamma):
        ac = self._tables["alpha_cumprod"]
        return (k + ac / (1 - ac)) ** (-gamma)

    def get_weights(self, steps, shape=(-1, 1, 1, 1)):
        idx = self._steps_index(steps)
        return self._table("p2_loss_weights", idx.device)[idx].reshape(shape)

    def get_rates(self, steps, shape=(-1, 1, 1, 1)):
        idx = self._steps_index(steps)
        signal = self._table("sqrt_alpha_cumprod", idx.device)[idx]
        noise = self._table("sqrt_one_minus_alpha_cumprod", idx.device)[idx]
        return reshape_rates((signal, noise), shape=shape)

    def get_posterior_mean(self, x_0, x_t, steps):
        idx = self._steps_index(steps, device=x_0.device)
        c0 = self._table("posterior_mean_coef1", x_0.device)[idx]
        ct = self._table("posterior_mean_coef2", x_0.device)[idx]
        c0, ct = reshape_rates((c0, ct), shape=get_coeff_shapes_tuple(x_0))
        return c0.to(x_0.dtype) * x_0 + ct.to(x_t.dtype) * x_t

    def get_posterior_variance(self, steps, shape=(-1, 1, 1, 1)):
        idx = self._steps_index(steps)
        logvar = self._table("posterior_log_variance_clipped", idx.device)[idx]
        return torch.exp(0.5 * logvar).reshape(shape)
