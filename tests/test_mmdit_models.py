"""CPU tests for the MM-DiT family (SimpleMMDiT, HierarchicalMMDiT)."""
import pytest
import torch

from flaxdiff_amd.models import (HierarchicalMMDiT, MMAdaLNZero, PatchExpanding,
                                 PatchMerging, SimpleMMDiT)


def test_mm_adaln_zero_init_is_identity_modulation():
    torch.manual_seed(0)
    m = MMAdaLNZero(features=32)
    x = torch.randn(2, 5, 32)
    x_attn, gate_attn, x_mlp, gate_mlp = m(x, torch.randn(2, 32), torch.randn(2, 7, 32))
    # zero-init projections -> scale=shift=gate=0 -> modulated == layernorm(x)
    ln = torch.nn.functional.layer_norm(x, (32,))
    assert torch.allclose(x_attn, ln, atol=1e-5)
    assert torch.allclose(x_mlp, ln, atol=1e-5)
    assert gate_attn.abs().max() == 0 and gate_mlp.abs().max() == 0


def test_patch_merge_expand_roundtrip_shapes():
    merge = PatchMerging(16, 32)
    expand = PatchExpanding(32, 16)
    x = torch.randn(2, 8 * 8, 16)
    y, h, w = merge(x, 8, 8)
    assert y.shape == (2, 16, 32) and (h, w) == (4, 4)
    z, h2, w2 = expand(y, h, w)
    assert z.shape == (2, 64, 16) and (h2, w2) == (8, 8)


def test_patch_merging_groups_spatial_neighbors():
    # token value = row-major index; after merge the first token must combine
    # tokens {0, 1, w_p, w_p+1}
    x = torch.arange(16, dtype=torch.float32).reshape(1, 16, 1).repeat(1, 1, 4)
    m = PatchMerging(4, 8)
    with torch.no_grad():
        m.projection.weight.fill_(0)
        m.projection.weight[0::4, 0].fill_(1.0)  # sum channel-0 of the 4 merged
    merged = x.reshape(1, 2, 2, 2, 2, 4).permute(0, 1, 3, 2, 4, 5).reshape(1, 4, 16)
    assert merged[0, 0, 0] == 0 and merged[0, 0, 4] == 1
    assert merged[0, 0, 8] == 4 and merged[0, 0, 12] == 5


@pytest.mark.parametrize("kwargs", [dict(), dict(use_hilbert=True),
                                    dict(learn_sigma=True)])
def test_simple_mmdit_forward(kwargs):
    torch.manual_seed(0)
    model = SimpleMMDiT(patch_size=4, emb_features=64, num_layers=2, num_heads=4,
                        context_dim=32, **kwargs)
    y = model(torch.randn(2, 16, 16, 3), torch.rand(2), torch.randn(2, 7, 32))
    assert y.shape == (2, 16, 16, 3)
    assert y.abs().max() == 0  # zero-init final proj


def test_simple_mmdit_backward():
    model = SimpleMMDiT(patch_size=4, emb_features=64, num_layers=2, num_heads=4,
                        context_dim=32)
    y = model(torch.randn(2, 16, 16, 3), torch.rand(2), torch.randn(2, 7, 32))
    (y ** 2).mean().backward()
    assert all(torch.isfinite(p.grad).all() for p in model.parameters()
               if p.grad is not None)


@pytest.mark.parametrize("kwargs", [dict(), dict(use_hilbert=True)])
def test_hierarchical_mmdit_forward_backward(kwargs):
    torch.manual_seed(0)
    model = HierarchicalMMDiT(base_patch_size=2, emb_features=(32, 48, 64),
                              num_layers=(1, 1, 2), num_heads=(4, 4, 4),
                              context_dim=32, **kwargs)
    x = torch.randn(2, 16, 16, 3)
    y = model(x, torch.rand(2), torch.randn(2, 7, 32))
    assert y.shape == (2, 16, 16, 3)
    (y ** 2).mean().backward()
    grads = [p.grad for p in model.parameters() if p.grad is not None]
    assert grads and all(torch.isfinite(g).all() for g in grads)


def test_hierarchical_requires_divisible_dims():
    model = HierarchicalMMDiT(base_patch_size=2, emb_features=(32, 48, 64),
                              num_layers=(1, 1, 1), num_heads=(4, 4, 4),
                              context_dim=32)
    with pytest.raises(AssertionError):
        model(torch.randn(1, 10, 10, 3), torch.rand(1), torch.randn(1, 7, 32))
