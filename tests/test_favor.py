"""CPU tests for FAVOR+ linear attention (Performer)."""
import torch

from flaxdiff_amd.models.favor_fastattn import (
    FastAttention, gaussian_orthogonal_random_matrix,
    make_fast_generalized_attention, make_fast_softmax_attention)


def test_orthogonal_random_matrix_blocks():
    W = gaussian_orthogonal_random_matrix(16, 8, scaling=1)
    assert W.shape == (16, 8)
    # scaling=1: every row has norm sqrt(d)
    assert torch.allclose(W.norm(dim=1), torch.full((16,), 8 ** 0.5), atol=1e-4)
    # first block rows orthogonal
    g = W[:8] @ W[:8].t()
    assert torch.allclose(g - torch.diag(torch.diagonal(g)),
                          torch.zeros(8, 8), atol=1e-4)


def test_favor_approximates_softmax_attention():
    torch.manual_seed(0)
    B, H, S, D = 2, 2, 32, 16
    q = torch.randn(B, H, S, D) * 0.3
    k = torch.randn(B, H, S, D) * 0.3
    v = torch.randn(B, H, S, D)
    exact = torch.softmax(q @ k.transpose(-1, -2) * D ** -0.5, -1) @ v
    fast = make_fast_softmax_attention(D, nb_features=256)(q, k, v)
    rel = (fast - exact).norm() / exact.norm()
    assert rel < 0.35, f"rel err {rel:.3f}"


def test_favor_causal_is_causal():
    torch.manual_seed(0)
    attn = make_fast_softmax_attention(8, nb_features=64, causal=True)
    q = torch.randn(1, 1, 16, 8)
    k = torch.randn(1, 1, 16, 8)
    v = torch.randn(1, 1, 16, 8)
    y1 = attn(q, k, v)
    k2, v2 = k.clone(), v.clone()
    k2[:, :, 10:] += 3.0
    v2[:, :, 10:] += 3.0
    y2 = attn(q, k2, v2)
    assert torch.allclose(y1[:, :, :10], y2[:, :, :10], atol=1e-5)
    assert not torch.allclose(y1[:, :, 10:], y2[:, :, 10:], atol=1e-2)


def test_generalized_relu_attention_runs():
    attn = make_fast_generalized_attention(8, nb_features=32)
    y = attn(torch.randn(1, 2, 8, 8), torch.randn(1, 2, 8, 8),
             torch.randn(1, 2, 8, 8))
    assert y.shape == (1, 2, 8, 8)
    assert torch.isfinite(y).all()


def test_redraw_changes_projection():
    attn = FastAttention(8, nb_features=16)
    before = attn.projection.clone()
    attn.redraw_projection(torch.Generator().manual_seed(123))
    assert not torch.allclose(before, attn.projection)
