"""GPU sampling correctness: real multi-step hipGraph sampling.

Covers the round-1 weaknesses:
  * graph-vs-eager equivalence over a REAL 50-step EDM run (round 1 only ever
    compared a degenerate 1-step path);
  * hipGraph conditioning staleness — a second generate_samples call with a
    DIFFERENT prompt of the same shape must not replay the old embeddings
    (samplers/common.py update_conds).
"""
import pytest
import torch

pytestmark = [pytest.mark.gpu, pytest.mark.timeout(300, method="thread")]


def _make_sampler(model_fn, guidance=0.0):
    from flaxdiff_amd.predictors import KarrasPredictionTransform
    from flaxdiff_amd.samplers import EulerSampler
    from flaxdiff_amd.schedulers import KarrasVENoiseScheduler
    ns = KarrasVENoiseScheduler(1, sigma_max=80, rho=7, sigma_data=0.5)
    return EulerSampler(model=model_fn, noise_schedule=ns,
                        model_output_transform=KarrasPredictionTransform(sigma_data=0.5),
                        guidance_scale=guidance, timestep_spacing="linear")


def _small_unet(context_dim=None):
    from flaxdiff_amd.models import Unet
    torch.manual_seed(0)
    m = Unet(emb_features=64, feature_depths=[32, 64],
             attention_configs=[{"heads": 4}] * 2, num_res_blocks=1,
             num_middle_res_blocks=1, norm_groups=8,
             context_dim=context_dim).cuda().eval()
    return m


def test_graph_matches_eager_50_steps():
    from flaxdiff_amd.utils import RandomMarkovState
    model = _small_unet()
    calls = {"n": 0}

    def fn(x, t, *c):
        calls["n"] += 1
        return model(x.to(torch.bfloat16), t, None).float()

    kw = dict(num_samples=4, resolution=32, diffusion_steps=50,
              device="cuda", dtype=torch.float32)
    s_eager = _make_sampler(fn)
    out_eager = s_eager.generate_samples(rngstate=RandomMarkovState(3), **kw)
    assert calls["n"] == 50  # NFE semantics on GPU

    s_graph = _make_sampler(fn).enable_graph_capture()
    out_graph = s_graph.generate_samples(rngstate=RandomMarkovState(3), **kw)
    # hipBLASLt split-k GEMMs use atomics: run-to-run nondeterminism
    # accumulates over 50 bf16 steps, so this is a closeness check.
    diff = (out_eager - out_graph).abs().max().item()
    assert diff < 5e-2, f"graph vs eager max abs diff {diff} over 50 steps"


def test_graph_conditioning_refreshed_between_calls():
    """Second call with different conditioning must change the output."""
    from flaxdiff_amd.utils import RandomMarkovState
    model = _small_unet(context_dim=64)

    def fn(x, t, *c):
        return model(x.to(torch.bfloat16), t,
                     *(ci.to(torch.bfloat16) for ci in c)).float()

    s = _make_sampler(fn).enable_graph_capture()
    torch.manual_seed(1)
    cond_a = torch.randn(4, 8, 64, device="cuda")
    cond_b = torch.randn(4, 8, 64, device="cuda")
    kw = dict(num_samples=4, resolution=32, diffusion_steps=8,
              device="cuda", dtype=torch.float32)

    out_a = s.generate_samples(rngstate=RandomMarkovState(5),
                               model_conditioning_inputs=(cond_a,), **kw)
    graphed = s._graphed
    assert graphed is not None
    out_b = s.generate_samples(rngstate=RandomMarkovState(5),
                               model_conditioning_inputs=(cond_b,), **kw)
    assert s._graphed is graphed  # same shape: no recapture, buffers refreshed
    diff_ab = (out_a - out_b).abs().max().item()
    assert diff_ab > 1e-4, \
        "different conditioning replayed identical (stale) embeddings"

    # same conditioning + same rng re-runs the same trajectory up to
    # nondeterministic split-k GEMM atomics; the cond effect must dominate
    out_a2 = s.generate_samples(rngstate=RandomMarkovState(5),
                                model_conditioning_inputs=(cond_a,), **kw)
    diff_aa = (out_a - out_a2).abs().max().item()
    assert diff_ab > 5 * diff_aa, (
        f"conditioning refresh suspect: cond-change diff {diff_ab} vs "
        f"replay jitter {diff_aa}")


def test_graph_recaptures_on_cond_shape_change():
    from flaxdiff_amd.utils import RandomMarkovState
    model = _small_unet(context_dim=64)

    def fn(x, t, *c):
        return model(x.to(torch.bfloat16), t,
                     *(ci.to(torch.bfloat16) for ci in c)).float()

    s = _make_sampler(fn).enable_graph_capture()
    kw = dict(num_samples=2, resolution=32, diffusion_steps=4,
              device="cuda", dtype=torch.float32)
    s.generate_samples(rngstate=RandomMarkovState(1),
                       model_conditioning_inputs=(torch.randn(2, 8, 64, device="cuda"),),
                       **kw)
    g1 = s._graphed
    s.generate_samples(rngstate=RandomMarkovState(1),
                       model_conditioning_inputs=(torch.randn(2, 16, 64, device="cuda"),),
                       **kw)
    assert s._graphed is not g1  # seq-len change must trigger recapture
