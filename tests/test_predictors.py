"""Golden tests for prediction transforms (reference math §2.1/predictors)."""
import pytest
import torch

from flaxdiff_amd.predictors import (DirectPredictionTransform,
                                     EpsilonPredictionTransform,
                                     KarrasPredictionTransform,
                                     VPredictionTransform)
from flaxdiff_amd.schedulers import (CosineNoiseScheduler,
                                     KarrasVENoiseScheduler)


def _rates(a, s):
    return (torch.full((2, 1, 1, 1), a), torch.full((2, 1, 1, 1), s))


def test_epsilon_roundtrip():
    tr = EpsilonPredictionTransform()
    x0 = torch.randn(2, 4, 4, 3)
    eps = torch.randn(2, 4, 4, 3)
    rates = _rates(0.8, 0.6)
    x_t, c_in, target = tr.forward_diffusion(x0, eps, rates)
    assert c_in == 1
    assert torch.allclose(target, eps)
    x0_rec, eps_rec = tr.backward_diffusion(x_t, eps, rates)
    assert torch.allclose(x0_rec, x0, atol=1e-5)
    assert torch.allclose(eps_rec, eps)


def test_direct_roundtrip():
    tr = DirectPredictionTransform()
    x0 = torch.randn(2, 4, 4, 3)
    eps = torch.randn(2, 4, 4, 3)
    rates = _rates(0.8, 0.6)
    x_t, _, target = tr.forward_diffusion(x0, eps, rates)
    assert torch.allclose(target, x0)
    x0_rec, eps_rec = tr.backward_diffusion(x_t, x0, rates)
    assert torch.allclose(x0_rec, x0)
    assert torch.allclose(eps_rec, eps, atol=1e-5)


def test_v_prediction_vp_roundtrip():
    """For VP rates (a^2+s^2=1): v = a*eps - s*x0; recovery is exact."""
    tr = VPredictionTransform()
    a, s = 0.8, 0.6
    x0 = torch.randn(2, 4, 4, 3)
    eps = torch.randn(2, 4, 4, 3)
    rates = _rates(a, s)
    x_t, _, v = tr.forward_diffusion(x0, eps, rates)
    assert torch.allclose(v, a * eps - s * x0, atol=1e-5)
    x0_rec, eps_rec = tr.backward_diffusion(x_t, v, rates)
    assert torch.allclose(x0_rec, x0, atol=1e-5)
    assert torch.allclose(eps_rec, eps, atol=1e-5)


def test_karras_preconditioning():
    sd = 0.5
    tr = KarrasPredictionTransform(sigma_data=sd)
    sigma = 2.0
    rates = _rates(1.0, sigma)
    c_in = tr.get_input_scale(rates)
    assert torch.allclose(c_in, 1 / torch.sqrt(torch.tensor(sd ** 2 + sigma ** 2)) , atol=1e-5)
    x_t = torch.randn(2, 4, 4, 3)
    f = torch.randn(2, 4, 4, 3)
    x0 = tr.pred_transform(x_t, f, rates)
    c_out = sigma * sd / (sd ** 2 + sigma ** 2) ** 0.5
    c_skip = sd ** 2 / (sd ** 2 + sigma ** 2)
    assert torch.allclose(x0, c_out * f + c_skip * x_t, atol=1e-4)


def test_transform_call_with_schedule():
    ns = KarrasVENoiseScheduler(timesteps=1000, sigma_data=0.5)
    tr = KarrasPredictionTransform(sigma_data=0.5)
    x_t = torch.randn(2, 4, 4, 3)
    preds = torch.randn(2, 4, 4, 3)
    steps = torch.tensor([300.0, 700.0])
    x0, eps = tr(x_t, preds, steps, ns)
    # backward identity: x_t == x0 + sigma*eps  (signal rate 1)
    _, sigma = ns.get_rates(steps, (-1, 1, 1, 1))
    assert torch.allclose(x0 + sigma * eps, x_t, atol=1e-4)
