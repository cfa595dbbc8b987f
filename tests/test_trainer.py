"""Trainer tests: loss decreases, EMA, checkpoint round-trip, NaN watchdog."""
import math

import pytest
import torch

from flaxdiff_amd.models import Unet
from flaxdiff_amd.predictors import EpsilonPredictionTransform
from flaxdiff_amd.schedulers import CosineNoiseScheduler
from flaxdiff_amd.trainer import DiffusionTrainer, FlatAdamWEMA, warmup_cosine_schedule


def tiny_trainer(tmp_path, **kw):
    m = Unet(emb_features=32, feature_depths=[8, 16],
             attention_configs=[None, {"heads": 2}], num_res_blocks=1,
             num_middle_res_blocks=1, norm_groups=4, context_dim=16)
    return DiffusionTrainer(m, CosineNoiseScheduler(1000),
                            EpsilonPredictionTransform(),
                            name="t", checkpoint_base_path=str(tmp_path),
                            text_context_shape=(4, 16), distributed=False,
                            optimizer_kwargs={"lr": 1e-3}, **kw)


def test_loss_decreases(tmp_path):
    torch.manual_seed(0)
    tr = tiny_trainer(tmp_path)
    batch = {"image": torch.zeros(8, 16, 16, 3, dtype=torch.uint8) + 127}
    losses = [tr.train_step(batch)["loss"] for _ in range(30)]
    assert sum(losses[-5:]) < sum(losses[:5])


def test_flat_optimizer_matches_adamw():
    """FlatAdamWEMA == torch.optim.AdamW step-for-step."""
    torch.manual_seed(0)
    m1 = torch.nn.Sequential(torch.nn.Linear(4, 8), torch.nn.Linear(8, 2))
    m2 = torch.nn.Sequential(torch.nn.Linear(4, 8), torch.nn.Linear(8, 2))
    m2.load_state_dict(m1.state_dict())
    opt1 = FlatAdamWEMA(m1, lr=1e-2, weight_decay=0.01)
    opt2 = torch.optim.AdamW(m2.parameters(), lr=1e-2, weight_decay=0.01, eps=1e-8)
    x = torch.randn(16, 4)
    y = torch.randn(16, 2)
    for _ in range(5):
        opt1.zero_grad()
        torch.nn.functional.mse_loss(m1(x), y).backward()
        opt1.step()
        opt2.zero_grad()
        torch.nn.functional.mse_loss(m2(x), y).backward()
        opt2.step()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        # torch AdamW decouples wd multiplicatively (p *= 1-lr*wd) while ours
        # folds wd into the update; allow small tolerance
        assert torch.allclose(p1, p2, atol=2e-4), (p1 - p2).abs().max()


def test_ema_tracks_params():
    m = torch.nn.Linear(4, 4)
    opt = FlatAdamWEMA(m, lr=1e-2, ema_decay=0.5)
    start = opt.ema.clone()
    opt.zero_grad()
    m(torch.randn(8, 4)).sum().backward()
    opt.step()
    assert not torch.allclose(opt.ema, start)
    expected = 0.5 * start + 0.5 * opt.flat
    assert torch.allclose(opt.ema, expected, atol=1e-6)


def test_ema_swap_restore():
    m = torch.nn.Linear(4, 4)
    opt = FlatAdamWEMA(m, lr=1e-1, ema_decay=0.0)
    opt.zero_grad()
    m(torch.randn(8, 4)).sum().backward()
    opt.step()
    before = opt.flat.clone()
    saved = opt.load_ema_into_params()
    assert torch.allclose(opt.flat, opt.ema)
    opt.restore_params(saved)
    assert torch.allclose(opt.flat, before)


def test_checkpoint_roundtrip(tmp_path):
    torch.manual_seed(0)
    tr = tiny_trainer(tmp_path)
    batch = {"image": torch.randint(0, 255, (4, 16, 16, 3), dtype=torch.uint8)}
    it = iter(lambda: batch, None)
    tr.train_loop(it, steps=3)
    tr.save(block=True)

    tr2 = tiny_trainer(tmp_path, load_from_checkpoint=True)
    assert tr2.global_step == 3
    assert torch.allclose(tr2.optimizer.flat, tr.optimizer.flat)
    assert torch.allclose(tr2.optimizer.ema, tr.optimizer.ema)
    assert torch.allclose(tr2.optimizer.exp_avg, tr.optimizer.exp_avg)


def test_nan_watchdog_restores_best(tmp_path):
    torch.manual_seed(0)
    tr = tiny_trainer(tmp_path)
    batch = {"image": torch.randint(0, 255, (4, 16, 16, 3), dtype=torch.uint8)}
    tr.train_loop(iter(lambda: batch, None), steps=2)
    assert tr.best_state is not None
    good = tr.best_state["flat"].clone()
    # poison the master buffer
    with torch.no_grad():
        tr.optimizer.flat.fill_(float("nan"))
    tr._recover()
    assert torch.isfinite(tr.optimizer.flat).all()
    assert torch.allclose(tr.optimizer.flat.cpu(), good)


def test_warmup_cosine_schedule():
    sched = warmup_cosine_schedule(1.0, warmup_steps=10, total_steps=110)
    assert sched(0) == 0.0
    assert sched(5) == pytest.approx(0.5)
    assert sched(10) == pytest.approx(1.0)
    assert sched(110) == pytest.approx(0.0, abs=1e-6)
    assert sched(60) == pytest.approx(0.5, abs=1e-6)


def test_grad_clip():
    m = torch.nn.Linear(4, 4)
    opt = FlatAdamWEMA(m, lr=0.0, grad_clip_norm=1e-9)
    opt.zero_grad()
    (m(torch.randn(8, 4)).sum() * 1000).backward()
    opt.step()  # should not blow up; lr=0 keeps params fixed
    assert torch.isfinite(opt.flat).all()


def test_nonfinite_grad_skips_step():
    """DynamicScale parity: a NaN gradient leaves params untouched."""
    import torch.nn as nn
    from flaxdiff_amd.trainer.optim import FlatAdamWEMA

    m = nn.Linear(4, 4)
    opt = FlatAdamWEMA(m, lr=1e-2)
    before = opt.flat.clone()
    opt.flat_grad.fill_(float("nan"))
    opt.step()
    assert torch.equal(opt.flat, before)
    assert opt.skipped_steps == 1 and opt.step_count == 0
    # a finite grad then applies normally
    opt.flat_grad.fill_(0.1)
    opt.step()
    assert not torch.equal(opt.flat, before)
    assert opt.step_count == 1


def test_validation_fn_end_to_end(tmp_path):
    """Epoch validation hook: EMA sampling + metrics through fit()."""
    from flaxdiff_amd.metrics import EvaluationMetric
    from flaxdiff_amd.models import Unet
    from flaxdiff_amd.predictors import KarrasPredictionTransform
    from flaxdiff_amd.schedulers import EDMNoiseScheduler
    from flaxdiff_amd.trainer import GeneralDiffusionTrainer

    calls = []
    metric = EvaluationMetric(
        function=lambda gen, batch: float(gen.float().std()),
        name="sample_std", higher_is_better=True)
    model = Unet(emb_features=32, feature_depths=[8, 16],
                 attention_configs=[None, None], num_res_blocks=1,
                 num_middle_res_blocks=1, norm_groups=4, context_dim=768)
    tr = GeneralDiffusionTrainer(
        model, EDMNoiseScheduler(1, sigma_max=80),
        KarrasPredictionTransform(sigma_data=0.5),
        eval_metrics=[metric], name="valfn",
        checkpoint_base_path=str(tmp_path), distributed=False)
    val_fn = tr.make_validation_fn(num_samples=2, resolution=16,
                                   diffusion_steps=2, guidance_scale=0.0)

    def batches():
        while True:
            yield {"image": torch.randint(0, 255, (2, 16, 16, 3),
                                          dtype=torch.uint8)}

    tr.fit(batches(), steps_per_epoch=2, epochs=1, val_fn=val_fn)
    assert "sample_std" in tr.best_metric_values


def test_autoencoder_trainer_reconstruction(tmp_path):
    """AutoEncoderTrainer: recon+KL loss decreases on a fixed batch."""
    import torch.nn as nn
    from flaxdiff_amd.trainer.autoencoder_trainer import AutoEncoderTrainer

    class TinyAE(nn.Module):
        def __init__(self):
            super().__init__()
            self.enc = nn.Linear(3, 8)
            self.mu = nn.Linear(8, 4)
            self.logvar = nn.Linear(8, 4)
            self.dec = nn.Linear(4, 3)

        def forward(self, x):
            h = torch.relu(self.enc(x))
            mu, lv = self.mu(h), self.logvar(h)
            z = mu + torch.randn_like(mu) * torch.exp(0.5 * lv)
            return self.dec(z), mu, lv

    torch.manual_seed(0)
    tr = AutoEncoderTrainer(TinyAE(), name="ae-test",
                            checkpoint_base_path=str(tmp_path),
                            distributed=False,
                            optimizer_kwargs={"lr": 1e-2})
    batch = {"image": torch.rand(16, 4, 4, 3) * 2 - 1}
    losses = [tr.train_step(batch)["rec_loss"] for _ in range(30)]
    assert losses[-1] < losses[0] * 0.9
    assert all(l == l for l in losses)


def test_resume_continues_identically(tmp_path):
    """Interrupted training must continue exactly where it left off: 3 steps
    + save + reload + 2 steps == 5 uninterrupted steps (weights, EMA and
    optimizer moments all bit-equal — RNG chain included in the checkpoint)."""
    batch = {"image": torch.randint(
        0, 255, (4, 16, 16, 3), dtype=torch.uint8,
        generator=torch.Generator().manual_seed(11))}
    it = iter(lambda: batch, None)

    torch.manual_seed(0)
    straight = tiny_trainer(tmp_path / "a")
    straight.train_loop(it, steps=5)

    torch.manual_seed(0)
    part1 = tiny_trainer(tmp_path / "b")
    part1.train_loop(it, steps=3)
    part1.save(block=True)
    resumed = tiny_trainer(tmp_path / "b", load_from_checkpoint=True)
    resumed.train_loop(it, steps=2)

    assert resumed.global_step == straight.global_step == 5
    assert torch.equal(resumed.optimizer.flat, straight.optimizer.flat)
    assert torch.equal(resumed.optimizer.ema, straight.optimizer.ema)
    assert torch.equal(resumed.optimizer.exp_avg, straight.optimizer.exp_avg)


def test_validation_sample_with_cfg(tmp_path):
    """validation_sample with conditioning + guidance>0 must supply the null
    embedding for the CFG uncond half (fixed: sampler got no unconditionals
    when built without an input_config)."""
    torch.manual_seed(0)
    tr = tiny_trainer(tmp_path)
    from flaxdiff_amd.samplers import EulerAncestralSampler
    ctx = torch.randn(2, 4, 16)
    out = tr.validation_sample(EulerAncestralSampler, num_samples=2,
                               resolution=16, diffusion_steps=2,
                               guidance_scale=3.0, conditioning_context=ctx)
    assert out.shape == (2, 16, 16, 3)
    assert torch.isfinite(out).all()


def test_general_trainer_multi_condition_step(tmp_path):
    """N-modality conditioning: two conditions, each
    CFG-dropped against its OWN null embedding, reach the model as separate
    context tensors (reference inputs/__init__.py:123-146)."""
    from flaxdiff_amd.inputs import ConditionalInputConfig, DiffusionInputConfig
    from flaxdiff_amd.inputs.encoders import DummyTextEncoder
    from flaxdiff_amd.predictors import EpsilonPredictionTransform
    from flaxdiff_amd.schedulers import CosineNoiseScheduler
    from flaxdiff_amd.trainer import GeneralDiffusionTrainer

    class TwoCondModel(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.proj1 = torch.nn.Linear(8, 4)
            self.proj2 = torch.nn.Linear(6, 4)
            self.out = torch.nn.Conv2d(3, 3, 1)
            self.seen = []

        def forward(self, x, temb, c_text, c_style):
            self.seen.append((tuple(c_text.shape), tuple(c_style.shape)))
            bias = self.proj1(c_text.float().mean(1)) + \
                self.proj2(c_style.float().mean(1))
            y = self.out(x.permute(0, 3, 1, 2)).permute(0, 2, 3, 1)
            return y + bias.mean(-1)[:, None, None, None]

    text_enc = DummyTextEncoder(seq_len=5, dim=8)
    style_enc = DummyTextEncoder(seq_len=3, dim=6)
    style_enc.key = "style"
    cfg = DiffusionInputConfig(
        sample_data_key="image", sample_data_shape=(16, 16, 3),
        conditions=[
            ConditionalInputConfig(encoder=text_enc, unconditional_input=""),
            ConditionalInputConfig(encoder=style_enc,
                                   conditioning_data_key="style",
                                   unconditional_input=""),
        ])
    model = TwoCondModel()
    tr = GeneralDiffusionTrainer(
        model, CosineNoiseScheduler(1000), EpsilonPredictionTransform(),
        input_config=cfg, name="multi-cond",
        checkpoint_base_path=str(tmp_path), distributed=False)
    batch = {
        "image": torch.randint(0, 255, (4, 16, 16, 3), dtype=torch.uint8),
        "text": ["a", "b", "c", "d"],
        "style": ["s1", "s2", "s3", "s4"],
    }
    out = tr.train_step(batch)
    assert "loss" in out and out["loss"] == out["loss"]
    assert model.seen[-1] == ((4, 5, 8), (4, 3, 6))


def test_registry_push_best_run_comparison(tmp_path):
    """Registry push with best-run comparison (reference wandb registry
    semantics, general_diffusion_trainer.py:560-703): a worse later run must
    NOT replace the stored best; a better one must."""
    from flaxdiff_amd.metrics import EvaluationMetric
    from flaxdiff_amd.models import Unet
    from flaxdiff_amd.predictors import EpsilonPredictionTransform
    from flaxdiff_amd.schedulers import CosineNoiseScheduler
    from flaxdiff_amd.trainer import GeneralDiffusionTrainer

    def make(name):
        m = Unet(emb_features=16, feature_depths=[8], attention_configs=[None],
                 num_res_blocks=1, norm_groups=4, context_dim=16)
        tr = GeneralDiffusionTrainer(
            m, CosineNoiseScheduler(100), EpsilonPredictionTransform(),
            name=name, checkpoint_base_path=str(tmp_path / "ck"),
            distributed=False)
        tr.eval_metrics = [EvaluationMetric(function=lambda *a: 0.0,
                                            name="fid", higher_is_better=False)]
        tr.train_step({"image": torch.randint(0, 255, (2, 8, 8, 3),
                                              dtype=torch.uint8)})
        return tr

    reg = str(tmp_path / "registry")
    tr = make("runA")
    tr.best_metric_values["fid"] = 10.0
    assert tr.push_to_registry(reg, compare_metric="fid") is not None

    tr2 = make("runA")
    tr2.best_metric_values["fid"] = 20.0        # worse (lower is better)
    assert tr2.push_to_registry(reg, compare_metric="fid") is None

    tr3 = make("runA")
    tr3.best_metric_values["fid"] = 5.0         # better
    dst = tr3.push_to_registry(reg, compare_metric="fid")
    assert dst is not None

    path, meta = GeneralDiffusionTrainer.resume_from_registry(reg, "runA")
    assert path is not None and meta["metrics"]["fid"] == 5.0
    import os as _os
    assert _os.path.isdir(path)


def test_watchdog_recovery_refreshes_shadows():
    """_recover() must restore the bf16 shadows the forward actually reads,
    not just the fp32 masters."""
    import torch
    from flaxdiff_amd.models import Unet
    from flaxdiff_amd.predictors import EpsilonPredictionTransform
    from flaxdiff_amd.schedulers import CosineNoiseScheduler
    from flaxdiff_amd.trainer import DiffusionTrainer

    model = Unet(emb_features=32, feature_depths=(8, 16),
                 attention_configs=(None, None), num_res_blocks=1,
                 norm_groups=4, context_dim=16)
    tr = DiffusionTrainer(model, CosineNoiseScheduler(100),
                          EpsilonPredictionTransform(), name="wdog",
                          checkpoint_base_path="/tmp/fdiff_wdog",
                          distributed=False)
    tr._snapshot_best(0.5)                       # good snapshot
    with torch.no_grad():
        tr.optimizer.flat.fill_(float("nan"))    # corrupt masters
        tr.optimizer.flat_bf16.fill_(float("nan"))
    tr._recover()
    assert torch.isfinite(tr.optimizer.flat).all()
    assert torch.isfinite(tr.optimizer.flat_bf16.float()).all()
    assert torch.allclose(tr.optimizer.flat_bf16.float(),
                          tr.optimizer.flat.bfloat16().float())


def test_graph_eligibility_predicate():
    """_graph_eligible gates: CPU device, autoencoder, lr schedules and
    multi-modality batches must all force the eager path."""
    import torch
    from flaxdiff_amd.models import Unet
    from flaxdiff_amd.predictors import EpsilonPredictionTransform
    from flaxdiff_amd.schedulers import CosineNoiseScheduler
    from flaxdiff_amd.trainer import DiffusionTrainer

    model = Unet(emb_features=32, feature_depths=(8, 16),
                 attention_configs=(None, None), num_res_blocks=1,
                 norm_groups=4, context_dim=16)
    tr = DiffusionTrainer(model, CosineNoiseScheduler(100),
                          EpsilonPredictionTransform(), name="gelig",
                          checkpoint_base_path="/tmp/fdiff_gelig",
                          distributed=False)
    img = torch.randint(0, 255, (2, 16, 16, 3), dtype=torch.uint8)
    # CPU device -> ineligible (hipGraphs are a GPU feature)
    assert not tr._graph_eligible({"image": img})
    # multi-modality conditioning -> ineligible
    assert not tr._graph_eligible({"image": img, "cond_embs": [img]})
    # lr schedule is host-side state -> ineligible
    tr.optimizer.lr_schedule = lambda s: 1e-4
    assert not tr._graph_eligible({"image": img})
    tr.optimizer.lr_schedule = None
    # env kill switch
    tr._graph_ok = False
    assert not tr._graph_eligible({"image": img})
