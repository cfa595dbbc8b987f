"""CPU tests for the DiT family + hilbert/zigzag scan utils.

Math oracles: patchify/unpatchify round trips, permutation inverses, RoPE
identities, AdaLN zero-init => DiT output == zero-init-proj output at init.
"""
import numpy as np
import pytest
import torch

from flaxdiff_amd.models import (SimpleDiT, SimpleUDiT, UViT)
from flaxdiff_amd.models import hilbert as hb
from flaxdiff_amd.models.vit_common import (RotaryEmbedding,
                                            apply_rotary_embedding)


def test_patchify_unpatchify_roundtrip():
    x = torch.randn(2, 16, 24, 3)
    tokens = hb.patchify(x, 4)
    assert tokens.shape == (2, 4 * 6, 4 * 4 * 3)
    back = hb.unpatchify(tokens, 4, 16, 24, 3)
    assert torch.equal(back, x)


@pytest.mark.parametrize("hp,wp", [(4, 4), (8, 8), (4, 6), (5, 3), (16, 16)])
def test_hilbert_indices_are_permutation(hp, wp):
    idx = hb.hilbert_indices(hp, wp)
    assert sorted(idx.tolist()) == list(range(hp * wp))
    inv = hb.inverse_permutation(idx, hp * wp)
    assert torch.equal(idx[inv], torch.arange(hp * wp))


def test_hilbert_locality():
    # consecutive Hilbert indices are 2-D neighbors on power-of-2 grids
    idx = hb.hilbert_indices(8, 8).numpy()
    r, c = idx // 8, idx % 8
    d = np.abs(np.diff(r)) + np.abs(np.diff(c))
    assert (d == 1).all()


def test_zigzag_indices():
    idx = hb.zigzag_indices(3, 4).tolist()
    assert idx == [0, 1, 2, 3, 7, 6, 5, 4, 8, 9, 10, 11]


@pytest.mark.parametrize("fn", [hb.hilbert_patchify, hb.zigzag_patchify])
def test_scan_patchify_roundtrip(fn):
    x = torch.randn(2, 16, 16, 3)
    tokens, inv = fn(x, 4)
    back = hb.hilbert_unpatchify(tokens, inv, 4, 16, 16, 3)
    assert torch.allclose(back, x)


def test_sincos_pos_embed_shape_and_range():
    pe = hb.build_2d_sincos_pos_embed(64, 4, 6)
    assert pe.shape == (24, 64)
    assert np.abs(pe).max() <= 1.0 + 1e-6


def test_rope_identity_at_position_zero():
    rope = RotaryEmbedding(dim=16, max_seq_len=32)
    cos, sin = rope(8)
    x = torch.randn(1, 2, 8, 16)
    y = apply_rotary_embedding(x, cos, sin)
    # position 0 has angle 0 -> unchanged
    assert torch.allclose(y[:, :, 0], x[:, :, 0], atol=1e-6)
    # rotation preserves per-pair norms
    assert torch.allclose(y.norm(dim=-1), x.norm(dim=-1), atol=1e-5)


def test_rope_dynamic_extension():
    rope = RotaryEmbedding(dim=8, max_seq_len=4)
    cos, sin = rope(16)
    assert cos.shape == (16, 4) and sin.shape == (16, 4)


@pytest.mark.parametrize("kwargs", [
    dict(),                      # raster + RoPE
    dict(use_hilbert=True),
    dict(use_zigzag=True),
    dict(learn_sigma=True),
])
def test_simple_dit_forward(kwargs):
    torch.manual_seed(0)
    model = SimpleDiT(patch_size=4, emb_features=64, num_layers=2, num_heads=4,
                      context_dim=32, **kwargs)
    x = torch.randn(2, 16, 16, 3)
    t = torch.rand(2)
    ctx = torch.randn(2, 7, 32)
    y = model(x, t, ctx)
    assert y.shape == (2, 16, 16, 3)
    assert torch.isfinite(y).all()
    # zero-init final proj -> output is exactly zero at init
    assert y.abs().max() == 0


def test_simple_dit_backward():
    model = SimpleDiT(patch_size=4, emb_features=64, num_layers=2, num_heads=4,
                      context_dim=32)
    x = torch.randn(2, 16, 16, 3)
    y = model(x, torch.rand(2), torch.randn(2, 7, 32))
    (y ** 2).mean().backward()
    grads = [p.grad for p in model.parameters() if p.grad is not None]
    assert len(grads) > 0
    assert all(torch.isfinite(g).all() for g in grads)


@pytest.mark.parametrize("kwargs", [dict(), dict(use_hilbert=True),
                                    dict(add_residualblock_output=True)])
def test_uvit_forward(kwargs):
    torch.manual_seed(0)
    model = UViT(patch_size=4, emb_features=64, num_layers=4, num_heads=4,
                 context_dim=32, **kwargs)
    x = torch.randn(2, 16, 16, 3)
    y = model(x, torch.rand(2), torch.randn(2, 7, 32))
    assert y.shape == (2, 16, 16, 3)
    assert torch.isfinite(y).all()


def test_uvit_no_text():
    model = UViT(patch_size=4, emb_features=64, num_layers=2, num_heads=4)
    y = model(torch.randn(2, 16, 16, 3), torch.rand(2))
    assert y.shape == (2, 16, 16, 3)


@pytest.mark.parametrize("kwargs", [dict(), dict(use_hilbert=True),
                                    dict(learn_sigma=True)])
def test_simple_udit_forward(kwargs):
    torch.manual_seed(0)
    model = SimpleUDiT(patch_size=4, emb_features=64, num_layers=4, num_heads=4,
                       context_dim=32, **kwargs)
    x = torch.randn(2, 16, 16, 3)
    y = model(x, torch.rand(2), torch.randn(2, 7, 32))
    assert y.shape == (2, 16, 16, 3)
    assert y.dtype == torch.float32
    assert torch.isfinite(y).all()


def test_dit_trains_one_step():
    """End-to-end: SimpleDiT through the diffusion trainer on CPU."""
    from flaxdiff_amd.predictors import KarrasPredictionTransform
    from flaxdiff_amd.schedulers import EDMNoiseScheduler
    from flaxdiff_amd.trainer import DiffusionTrainer

    model = SimpleDiT(patch_size=4, emb_features=64, num_layers=2, num_heads=4,
                      context_dim=768)
    trainer = DiffusionTrainer(
        model, EDMNoiseScheduler(1, sigma_max=80),
        KarrasPredictionTransform(sigma_data=0.5),
        name="dit-cpu", checkpoint_base_path="/tmp/fdiff_test_dit",
        distributed=False)
    batch = {"image": torch.randint(0, 255, (2, 16, 16, 3), dtype=torch.uint8)}
    out1 = trainer.train_step(batch)
    out2 = trainer.train_step(batch)
    assert np.isfinite(out1["loss"]) and np.isfinite(out2["loss"])
