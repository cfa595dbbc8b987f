"""CPU tests for metrics: PSNR/SSIM math, Fréchet distance, FID accumulator."""
import numpy as np
import pytest
import torch

from flaxdiff_amd.metrics import (EvaluationMetric, FrechetInceptionDistance,
                                  frechet_distance, get_psnr_metric,
                                  get_ssim_metric, psnr, ssim)


def test_psnr_identical_is_infinite_like():
    x = torch.rand(2, 8, 8, 3) * 2 - 1
    assert (psnr(x, x) > 100).all()


def test_psnr_known_value():
    a = torch.zeros(1, 4, 4, 1)
    b = torch.full((1, 4, 4, 1), 0.5)
    # mse=0.25, range=2 -> 10*log10(4/0.25) = 12.04
    assert abs(float(psnr(a, b)) - 12.0412) < 1e-3


def test_ssim_identical_is_one():
    x = torch.rand(2, 16, 16, 3) * 2 - 1
    s = ssim(x, x)
    assert torch.allclose(s, torch.ones_like(s), atol=1e-4)


def test_ssim_decreases_with_noise():
    torch.manual_seed(0)
    x = torch.rand(1, 32, 32, 3) * 2 - 1
    s_small = float(ssim(x, (x + 0.05 * torch.randn_like(x)).clamp(-1, 1)))
    s_big = float(ssim(x, (x + 0.5 * torch.randn_like(x)).clamp(-1, 1)))
    assert s_big < s_small < 1.0


def test_metric_wrappers():
    m = get_psnr_metric()
    assert isinstance(m, EvaluationMetric) and m.higher_is_better
    gen = torch.rand(2, 8, 8, 3) * 2 - 1
    batch = {"image": ((gen + 1) * 127.5).byte()}
    val = m.function(gen, batch)
    assert val > 40  # uint8 quantization only
    s = get_ssim_metric().function(gen, batch)
    assert s > 0.9


def test_frechet_distance_zero_for_identical():
    mu = np.zeros(4)
    sigma = np.eye(4)
    assert abs(frechet_distance(mu, sigma, mu, sigma)) < 1e-8


def test_frechet_distance_mean_shift():
    mu1, mu2 = np.zeros(3), np.ones(3) * 2
    sigma = np.eye(3)
    assert abs(frechet_distance(mu1, sigma, mu2, sigma) - 12.0) < 1e-6


def test_fid_accumulator_with_custom_features():
    torch.manual_seed(0)

    def feat(images):  # cheap deterministic extractor
        return images.reshape(images.shape[0], -1)[:, :8]

    fid = FrechetInceptionDistance(feature_fn=feat)
    real = torch.rand(64, 4, 4, 3)
    fid.update_real(real)
    fid.update_fake(real + 0.0)
    assert fid.compute() < 1e-6
    fid.reset()
    fid.update_real(torch.rand(64, 4, 4, 3))
    fid.update_fake(torch.rand(64, 4, 4, 3) + 3.0)
    assert fid.compute() > 10


@pytest.mark.slow
def test_inception_v3_shapes():
    from flaxdiff_amd.metrics.fid import InceptionV3Features
    net = InceptionV3Features()
    out = net(torch.rand(2, 64, 64, 3) * 2 - 1)
    assert out.shape == (2, 2048)
