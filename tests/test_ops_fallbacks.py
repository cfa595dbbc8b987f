"""CPU-path behavior of ops that have fused GPU kernels: the torch
fallbacks must match the op contract exactly (these run in every CI pass)."""
import torch

from flaxdiff_amd import ops


def test_cat_channels_cpu():
    a = torch.randn(2, 4, 4, 6)
    b = torch.randn(2, 4, 4, 10)
    assert torch.equal(ops.cat_channels(a, b), torch.cat([a, b], dim=-1))


def test_geglu_cpu():
    h = torch.randn(5, 32, dtype=torch.float64)
    x, g = h.chunk(2, dim=-1)
    ref = x * torch.nn.functional.gelu(g)
    assert torch.allclose(ops.geglu(h), ref)


def test_weighted_l2_loss_cpu():
    p = torch.randn(3, 4, 4, 2)
    t = torch.randn(3, 4, 4, 2)
    w = torch.rand(3, 1, 1, 1) + 0.1
    ref = (0.5 * (p - t) ** 2 * w).mean()
    assert torch.allclose(ops.weighted_l2_loss(p, t, w), ref, atol=1e-6)


def test_conv2d_add_badd_cpu():
    x = torch.randn(2, 6, 6, 4)
    w = torch.randn(3, 3, 4, 8) * 0.2
    b = torch.randn(8) * 0.1
    res = torch.randn(2, 6, 6, 8)
    t = torch.randn(2, 8)
    y0 = ops.conv2d(x, w, b, stride=1)
    y1 = ops.conv2d(x, w, b, stride=1, add=res, badd=t)
    assert torch.allclose(y1, y0 + t[:, None, None, :] + res, atol=1e-5)


def test_dense_add_cpu():
    x = torch.randn(5, 4)
    w = torch.randn(4, 6)
    b = torch.randn(6)
    res = torch.randn(5, 6)
    assert torch.allclose(ops.dense(x, w, b, add=res),
                          x @ w + b + res, atol=1e-5)


def test_scheduler_device_sampling_distribution():
    """sample_timesteps_device matches generate_timesteps' distribution
    family per scheduler class (graph-path RNG parity)."""
    from flaxdiff_amd.schedulers import (CosineNoiseScheduler,
                                         EDMNoiseScheduler,
                                         KarrasVENoiseScheduler)
    torch.manual_seed(0)
    d = CosineNoiseScheduler(1000).sample_timesteps_device(512, "cpu")
    assert d.dtype in (torch.int64, torch.int32) and d.min() >= 0 and d.max() < 1000
    k = KarrasVENoiseScheduler(1).sample_timesteps_device(512, "cpu")
    assert k.dtype == torch.float32 and k.min() >= 0 and k.max() <= 1.0
    e = EDMNoiseScheduler(1).sample_timesteps_device(4096, "cpu")
    assert abs(e.mean().item()) < 0.2 and abs(e.std().item() - 1) < 0.2


def test_conv2d_upsample2x_cpu_fallback():
    """conv2d_upsample2x == conv2d(nearest_upsample_2x(x)) off the GPU path
    (also covers non-64-divisible channels where the fused kernel declines)."""
    torch.manual_seed(2)
    x = torch.randn(2, 4, 4, 6)
    w = torch.randn(3, 3, 6, 10) * 0.2
    b = torch.randn(10) * 0.1
    ref = ops.conv2d(ops.nearest_upsample_2x(x), w, b, stride=1)
    assert torch.allclose(ops.conv2d_upsample2x(x, w, b), ref, atol=1e-6)
