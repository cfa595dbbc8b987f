"""CPU tests for the S5/SSM-DiT family.

The sequential-recurrence oracle: x_k = a_k x_{k-1} + b_k computed with a
plain python loop must match the log-depth parallel scan exactly (same fp32
complex math).
"""
import math

import pytest
import torch

from flaxdiff_amd.models.ssm_dit import (BidirectionalS5Layer,
                                         HybridSSMAttentionDiT, S5Layer,
                                         SSMDiTBlock, SpatialFusionConv,
                                         associative_scan_diag,
                                         _build_block_pattern)


@pytest.mark.parametrize("S", [1, 2, 3, 7, 16, 33])
def test_associative_scan_matches_sequential(S):
    torch.manual_seed(0)
    B, N = 2, 5
    a = torch.complex(torch.randn(B, S, N) * 0.3, torch.randn(B, S, N) * 0.3)
    b = torch.complex(torch.randn(B, S, N), torch.randn(B, S, N))
    out = associative_scan_diag(a, b)
    x = torch.zeros(B, N, dtype=torch.complex64)
    for k in range(S):
        x = a[:, k] * x + b[:, k]
        assert torch.allclose(out[:, k], x, atol=1e-4), f"step {k}"


def test_s5_layer_shapes_and_stability():
    torch.manual_seed(0)
    layer = S5Layer(features=16, state_dim=8)
    u = torch.randn(2, 32, 16)
    y = layer(u)
    assert y.shape == (2, 32, 16)
    assert torch.isfinite(y).all()
    # A_real must be negative (stable recurrence): A_bar magnitudes < 1
    dt = torch.exp(layer.log_dt)
    A = torch.complex(-torch.exp(layer.log_A_real), layer.A_imag)
    assert (torch.exp(A * dt).abs() < 1).all()


def test_s5_hippo_init_values():
    layer = S5Layer(features=4, state_dim=4)
    n = torch.arange(4, dtype=torch.float32)
    assert torch.allclose(torch.exp(layer.log_A_real), n + 0.5)
    assert torch.allclose(layer.A_imag, math.pi * n)


def test_s5_causality():
    """Changing u at position k must not affect outputs before k."""
    torch.manual_seed(0)
    layer = S5Layer(features=8, state_dim=4)
    u = torch.randn(1, 16, 8)
    y1 = layer(u)
    u2 = u.clone()
    u2[:, 10:] += 1.0
    y2 = layer(u2)
    assert torch.allclose(y1[:, :10], y2[:, :10], atol=1e-5)
    assert not torch.allclose(y1[:, 10:], y2[:, 10:], atol=1e-3)


def test_bidirectional_s5():
    torch.manual_seed(0)
    layer = BidirectionalS5Layer(features=8, state_dim=4)
    y = layer(torch.randn(2, 16, 8))
    assert y.shape == (2, 16, 8)
    assert torch.isfinite(y).all()


def test_spatial_fusion_zero_init_passthrough():
    fusion = SpatialFusionConv(features=8)
    x = torch.randn(2, 4, 4, 8)
    assert torch.equal(fusion(x), x)  # zero-init kernels -> identity


def test_block_pattern():
    assert _build_block_pattern(None, "3:1", 8) == \
        ["ssm", "ssm", "ssm", "attn"] * 2
    assert _build_block_pattern(None, "all-ssm", 3) == ["ssm"] * 3
    assert _build_block_pattern(["attn", "ssm"], "3:1", 2) == ["attn", "ssm"]


def test_ssm_dit_block_forward():
    torch.manual_seed(0)
    blk = SSMDiTBlock(features=32, num_heads=4, state_dim=8)
    x = torch.randn(2, 16, 32)
    y = blk(x, torch.randn(2, 32), None)
    assert y.shape == x.shape
    # zero-init AdaLN gates -> block is identity at init
    assert torch.allclose(y, x, atol=1e-5)


@pytest.mark.parametrize("kwargs", [
    dict(),
    dict(use_hilbert=True),
    dict(use_zigzag=True, use_2d_fusion=True),
    dict(ssm_attention_ratio="all-ssm", bidirectional_ssm=False),
])
def test_hybrid_ssm_dit_forward_backward(kwargs):
    torch.manual_seed(0)
    model = HybridSSMAttentionDiT(patch_size=4, emb_features=64, num_layers=4,
                                  num_heads=4, ssm_state_dim=8, context_dim=32,
                                  **kwargs)
    x = torch.randn(2, 16, 16, 3)
    y = model(x, torch.rand(2), torch.randn(2, 7, 32))
    assert y.shape == (2, 16, 16, 3)
    (y ** 2).mean().backward()
    grads = [p.grad for p in model.parameters() if p.grad is not None]
    assert grads and all(torch.isfinite(g).all() for g in grads)
