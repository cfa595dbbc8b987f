"""Golden tests for the scheduler math core against closed-form values
(scheduler math is spec'd by reference §2.1)."""
import math

import numpy as np
import pytest
import torch

from flaxdiff_amd.schedulers import (
    ContinuousNoiseScheduler, CosineContinuousNoiseScheduler,
    CosineGeneralNoiseScheduler, CosineNoiseScheduler, DiscreteNoiseScheduler,
    EDMNoiseScheduler, ExpNoiseSchedule, KarrasVENoiseScheduler,
    LinearNoiseSchedule, SimpleExpNoiseScheduler, SqrtContinuousNoiseScheduler,
    cosine_beta_schedule, linear_beta_schedule)
from flaxdiff_amd.utils import RandomMarkovState


def test_linear_beta_schedule_values():
    betas = linear_beta_schedule(1000)
    assert betas[0] == pytest.approx(1e-4)
    assert betas[-1] == pytest.approx(0.02)
    assert np.all(np.diff(betas) > 0)


def test_cosine_beta_schedule_range():
    betas = cosine_beta_schedule(1000)
    assert betas.min() >= 0
    assert betas.max() <= 0.999
    assert betas[-1] == pytest.approx(0.999)  # clipped tail


def test_discrete_vp_identity():
    """signal^2 + noise^2 == 1 for the VP scheduler."""
    ns = LinearNoiseSchedule(1000)
    steps = torch.arange(0, 1000, 37)
    a, s = ns.get_rates(steps, shape=(-1,))
    assert torch.allclose(a ** 2 + s ** 2, torch.ones_like(a), atol=1e-5)


def test_discrete_alpha_cumprod_matches_numpy():
    ns = LinearNoiseSchedule(100)
    betas = linear_beta_schedule(100)
    ac = np.cumprod(1 - betas)
    a, s = ns.get_rates(torch.tensor([0, 50, 99]), shape=(-1,))
    np.testing.assert_allclose(a.numpy() ** 2, ac[[0, 50, 99]], rtol=1e-5)


def test_posterior_coeffs():
    """DDPM posterior mean coefficients: closed form q(x_{t-1}|x_t, x_0)."""
    ns = LinearNoiseSchedule(100)
    betas = linear_beta_schedule(100)
    alphas = 1 - betas
    ac = np.cumprod(alphas)
    ac_prev = np.append(1.0, ac[:-1])
    t = 42
    c1 = betas[t] * np.sqrt(ac_prev[t]) / (1 - ac[t])
    c2 = (1 - ac_prev[t]) * np.sqrt(alphas[t]) / (1 - ac[t])
    x0 = torch.ones(1, 2, 2, 1)
    xt = torch.full((1, 2, 2, 1), 2.0)
    mean = ns.get_posterior_mean(x0, xt, torch.tensor([t]))
    expected = c1 * 1.0 + c2 * 2.0
    assert mean.flatten()[0].item() == pytest.approx(expected, rel=1e-5)
    var = ns.get_posterior_variance(torch.tensor([t]), shape=(-1,))
    pv = betas[t] * (1 - ac_prev[t]) / (1 - ac[t])
    assert var.item() == pytest.approx(math.sqrt(pv), rel=1e-5)


def test_karras_sigma_ramp_endpoints():
    ns = KarrasVENoiseScheduler(timesteps=1000, sigma_min=0.002, sigma_max=80, rho=7)
    s_max = ns.get_sigmas(torch.tensor([1000.0]))
    s_min = ns.get_sigmas(torch.tensor([0.0]))
    assert s_max.item() == pytest.approx(80.0, rel=1e-4)
    assert s_min.item() == pytest.approx(0.002, rel=1e-3)


def test_karras_timestep_inverse():
    ns = KarrasVENoiseScheduler(timesteps=1000)
    steps = torch.tensor([100.0, 500.0, 900.0])
    sig = ns.get_sigmas(steps)
    rec = ns.get_timesteps(sig)
    assert torch.allclose(rec, steps, rtol=1e-3)


def test_karras_weights_edm_lambda():
    ns = KarrasVENoiseScheduler(timesteps=1000, sigma_data=0.5)
    steps = torch.tensor([500.0])
    sigma = ns.get_sigmas(steps).item()
    w = ns.get_weights(steps, shape=(-1,)).item()
    expected = (sigma ** 2 + 0.25) / ((sigma * 0.5) ** 2 + 1e-6)
    assert w == pytest.approx(expected, rel=1e-5)


def test_karras_cnoise_transform():
    ns = KarrasVENoiseScheduler(timesteps=1000)
    x = torch.zeros(2, 4, 4, 3)
    _, cn = ns.transform_inputs(x, torch.tensor([500.0, 800.0]))
    sig = ns.get_sigmas(torch.tensor([500.0, 800.0]))
    assert torch.allclose(cn, torch.log(sig + 1e-12) / 4)


def test_edm_sigma_formula():
    ns = EDMNoiseScheduler(1)
    t = torch.tensor([0.0, 1.0, -1.0])
    sig = ns.get_sigmas(t)
    expected = torch.exp(t * 1.2 - 1.2)
    assert torch.allclose(sig, expected)


def test_edm_timesteps_are_normal():
    ns = EDMNoiseScheduler(1)
    ts, _ = ns.generate_timesteps(10000, RandomMarkovState(0))
    assert abs(ts.mean().item()) < 0.05
    assert abs(ts.std().item() - 1) < 0.05


def test_cosine_continuous_rates():
    ns = CosineContinuousNoiseScheduler()
    a, s = ns.get_rates(torch.tensor([0.0, 0.5, 1.0]), shape=(-1,))
    assert a[0].item() == pytest.approx(1.0)
    assert s[0].item() == pytest.approx(0.0, abs=1e-6)
    assert a[2].item() == pytest.approx(0.0, abs=1e-6)
    assert s[2].item() == pytest.approx(1.0)
    assert a[1].item() == pytest.approx(math.cos(math.pi / 4), rel=1e-5)


def test_sqrt_scheduler():
    ns = SqrtContinuousNoiseScheduler()
    a, s = ns.get_rates(torch.tensor([0.25]), shape=(-1,))
    assert a.item() == pytest.approx(math.sqrt(0.75), rel=1e-5)
    assert s.item() == pytest.approx(0.5, rel=1e-5)


def test_add_noise_axpy():
    ns = LinearNoiseSchedule(1000)
    x0 = torch.randn(4, 8, 8, 3)
    eps = torch.randn(4, 8, 8, 3)
    t = torch.tensor([10, 100, 500, 999])
    xt = ns.add_noise(x0, eps, t)
    a, s = ns.get_rates(t)
    assert torch.allclose(xt, a * x0 + s * eps, atol=1e-6)


def test_max_variance():
    ns = CosineNoiseScheduler(1000)
    v = ns.get_max_variance(shape=(-1,))
    assert v.item() == pytest.approx(1.0, rel=1e-3)  # VP: always 1


def test_simple_exp_scheduler_table():
    ns = SimpleExpNoiseScheduler(1000)
    s0 = ns.get_sigmas(torch.tensor([0]))
    sN = ns.get_sigmas(torch.tensor([999]))
    assert s0.item() == pytest.approx(0.002, rel=1e-4)
    assert sN.item() == pytest.approx(80.0, rel=1e-4)


def test_cosine_general_sigmas_monotone():
    ns = CosineGeneralNoiseScheduler(sigma_min=0.02, sigma_max=80.0)
    s = ns.get_sigmas(torch.linspace(0, 1, 10))
    assert (s[1:] > s[:-1]).all()
    assert s[0].item() == pytest.approx(0.02, rel=1e-3)
    assert s[-1].item() == pytest.approx(80.0, rel=1e-2)


def test_timestep_generation_discrete_bounds():
    ns = LinearNoiseSchedule(1000)
    ts, state = ns.generate_timesteps(1000, RandomMarkovState(7))
    assert ts.min() >= 0 and ts.max() < 1000
    ts2, _ = state.get_random_key()[0], None  # chain continues
    # determinism
    ts_again, _ = ns.generate_timesteps(1000, RandomMarkovState(7))
    assert torch.equal(ts, ts_again)
