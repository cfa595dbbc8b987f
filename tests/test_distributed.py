"""Multi-process data-parallel tests on gloo (CPU, world_size=2).

Verifies the RCCL-path code (bucketed all-reduce, rank-fold-in RNG, grad
mean semantics) is correct by construction — the same code runs over
backend 'nccl' (RCCL) on the GPU nodes.
"""
import os

import pytest
import torch
import torch.multiprocessing as mp

from flaxdiff_amd.models import Unet


def _worker(rank, world, port, results):
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(world), "LOCAL_RANK": str(rank),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    import torch.distributed as dist
    from flaxdiff_amd.predictors import EpsilonPredictionTransform
    from flaxdiff_amd.schedulers import CosineNoiseScheduler
    from flaxdiff_amd.trainer import DiffusionTrainer

    torch.manual_seed(1234)  # same init on both ranks pre-broadcast
    m = Unet(emb_features=32, feature_depths=[8, 16],
             attention_configs=[None, {"heads": 2}], num_res_blocks=1,
             num_middle_res_blocks=1, norm_groups=4, context_dim=16)
    tr = DiffusionTrainer(m, CosineNoiseScheduler(1000),
                          EpsilonPredictionTransform(), name=f"ddp",
                          checkpoint_base_path="/tmp/fdiff_ddp_test",
                          text_context_shape=(4, 16), distributed=True)
    # per-rank distinct data
    g = torch.Generator().manual_seed(100 + rank)
    batch = {"image": torch.randint(0, 255, (4, 16, 16, 3), generator=g,
                                    dtype=torch.uint8)}
    losses = []
    for _ in range(3):
        losses.append(tr.train_step(batch)["loss"])
    # after sync steps params must be identical across ranks
    flat = tr.optimizer.flat.clone()
    gathered = [torch.zeros_like(flat) for _ in range(world)]
    dist.all_gather(gathered, flat)
    same = all(torch.allclose(gathered[0], gi, atol=1e-6) for gi in gathered)
    results[rank] = {"losses": losses, "params_equal": bool(same)}
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_data_parallel_two_ranks():
    world = 2
    ctx = mp.get_context("spawn")
    with ctx.Manager() as man:
        results = man.dict()
        port = 29731
        procs = [ctx.Process(target=_worker, args=(r, world, port, results))
                 for r in range(world)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=240)
        for p in procs:
            assert p.exitcode == 0, f"worker failed: {p.exitcode}"
        res = dict(results)
    assert res[0]["params_equal"] and res[1]["params_equal"]
    # losses are all-reduced means -> identical across ranks
    assert res[0]["losses"] == pytest.approx(res[1]["losses"], rel=1e-5)


def _rng_worker(rank, world, port, results):
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(world), "LOCAL_RANK": str(rank),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    import torch.distributed as dist
    from flaxdiff_amd.parallel import init_distributed
    from flaxdiff_amd.utils import RandomMarkovState
    init_distributed()
    st = RandomMarkovState(0).fold_in(rank)
    _, key = st.get_random_key()
    results[rank] = float(key.normal((4,)).sum())
    dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_per_rank_rng_decorrelated():
    """fold_in(rank) gives distinct random streams (reference
    diffusion_trainer.py:158 per-device fold-in)."""
    ctx = mp.get_context("spawn")
    with ctx.Manager() as man:
        results = man.dict()
        procs = [ctx.Process(target=_rng_worker, args=(r, 2, 29732, results))
                 for r in range(2)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=100)
        res = dict(results)
    assert res[0] != res[1]


def _cfg_worker(rank, world, port, results):
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(world), "LOCAL_RANK": str(rank),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    import torch.distributed as dist
    from flaxdiff_amd.predictors import KarrasPredictionTransform
    from flaxdiff_amd.schedulers import EDMNoiseScheduler
    from flaxdiff_amd.trainer import DiffusionTrainer

    torch.manual_seed(7)
    m = Unet(emb_features=32, feature_depths=[8, 16],
             attention_configs=[{"heads": 2}, {"heads": 2}], num_res_blocks=1,
             num_middle_res_blocks=1, norm_groups=4, context_dim=16)
    tr = DiffusionTrainer(m, EDMNoiseScheduler(1, sigma_max=80),
                          KarrasPredictionTransform(sigma_data=0.5),
                          name="ddp-cfg", checkpoint_base_path="/tmp/fd_ddp_cfg",
                          text_context_shape=(4, 16), unconditional_prob=0.5,
                          distributed=True)
    g = torch.Generator().manual_seed(300 + rank)
    batch = {"image": torch.randint(0, 255, (4, 16, 16, 3), generator=g,
                                    dtype=torch.uint8),
             "text_emb": torch.randn(4, 4, 16, generator=g)}
    losses = [tr.train_step(batch)["loss"] for _ in range(2)]
    flat = tr.optimizer.flat.clone()
    gathered = [torch.zeros_like(flat) for _ in range(world)]
    dist.all_gather(gathered, flat)
    same = all(torch.allclose(gathered[0], gi, atol=1e-6) for gi in gathered)
    results[rank] = {"losses": losses, "params_equal": bool(same)}
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_data_parallel_text_conditional_cfg():
    """Text-conditional + CFG-dropout training stays in sync across ranks
    (BASELINE config 4's DP path, EDM schedule)."""
    world = 2
    ctx = mp.get_context("spawn")
    with ctx.Manager() as man:
        results = man.dict()
        procs = [ctx.Process(target=_cfg_worker, args=(r, world, 29733, results))
                 for r in range(world)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=240)
        for p in procs:
            assert p.exitcode == 0, f"worker failed: {p.exitcode}"
        res = dict(results)
    assert res[0]["params_equal"] and res[1]["params_equal"]
    assert res[0]["losses"] == pytest.approx(res[1]["losses"], rel=1e-5)


def _resume_worker(rank, world, port, ckpt_dir, results):
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(world), "LOCAL_RANK": str(rank),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    import torch.distributed as dist
    from flaxdiff_amd.predictors import EpsilonPredictionTransform
    from flaxdiff_amd.schedulers import CosineNoiseScheduler
    from flaxdiff_amd.trainer import DiffusionTrainer

    def make():
        torch.manual_seed(1234)
        m = Unet(emb_features=32, feature_depths=[8, 16],
                 attention_configs=[None, {"heads": 2}], num_res_blocks=1,
                 num_middle_res_blocks=1, norm_groups=4, context_dim=16)
        return m

    g = torch.Generator().manual_seed(100 + rank)
    batch = {"image": torch.randint(0, 255, (4, 16, 16, 3), generator=g,
                                    dtype=torch.uint8)}
    it = iter(lambda: batch, None)

    tr = DiffusionTrainer(make(), CosineNoiseScheduler(1000),
                          EpsilonPredictionTransform(), name="ddp-resume",
                          checkpoint_base_path=ckpt_dir,
                          text_context_shape=(4, 16), distributed=True)
    tr.train_loop(it, steps=2)
    tr.save(block=True)
    dist.barrier()

    tr2 = DiffusionTrainer(make(), CosineNoiseScheduler(1000),
                           EpsilonPredictionTransform(), name="ddp-resume",
                           checkpoint_base_path=ckpt_dir,
                           text_context_shape=(4, 16), distributed=True,
                           load_from_checkpoint=True)
    tr2.train_loop(it, steps=1)
    # ranks agree after resume + one more synced step
    flat = tr2.optimizer.flat.clone()
    gathered = [torch.zeros_like(flat) for _ in range(world)]
    dist.all_gather(gathered, flat)
    same = all(torch.equal(gathered[0], gi) for gi in gathered)
    results[rank] = {"step": tr2.global_step, "params_equal": bool(same)}
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_data_parallel_resume(tmp_path):
    """Rank-0 checkpoints + resume keep all ranks in lockstep (the driver's
    multi-GPU runs restart from rank-0 checkpoints)."""
    world = 2
    ctx = mp.get_context("spawn")
    with ctx.Manager() as man:
        results = man.dict()
        procs = [ctx.Process(target=_resume_worker,
                             args=(r, world, 29741, str(tmp_path), results))
                 for r in range(world)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(240)
            assert p.exitcode == 0
        for r in range(world):
            assert results[r]["step"] == 3
            assert results[r]["params_equal"]


@pytest.mark.timeout(420)
def test_data_parallel_four_ranks():
    """4-rank gloo lockstep (beyond world_size=2)."""
    world = 4
    ctx = mp.get_context("spawn")
    with ctx.Manager() as man:
        results = man.dict()
        procs = [ctx.Process(target=_worker, args=(r, world, 29741, results))
                 for r in range(world)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=360)
        for p in procs:
            assert p.exitcode == 0
        res = dict(results)
    assert all(res[r]["params_equal"] for r in range(world))
    for r in range(1, world):
        assert res[0]["losses"] == pytest.approx(res[r]["losses"], rel=1e-5)


@pytest.mark.timeout(600)
@pytest.mark.slow
def test_data_parallel_eight_ranks():
    """8-rank gloo lockstep — the driver's scaling-run world size."""
    world = 8
    ctx = mp.get_context("spawn")
    with ctx.Manager() as man:
        results = man.dict()
        procs = [ctx.Process(target=_worker, args=(r, world, 29751, results))
                 for r in range(world)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=540)
        for p in procs:
            assert p.exitcode == 0
        res = dict(results)
    assert all(res[r]["params_equal"] for r in range(world))


@pytest.mark.timeout(420)
def test_bench_dry_run_torchrun_cpu():
    """bench.py under the driver's exact torchrun launch, 4 CPU ranks.

    FD_BENCH_TINY=1 shrinks the model so this is a PLUMBING test of the
    rendezvous + DP + JSON contract, not a perf number."""
    import json
    import subprocess
    import sys

    env = dict(os.environ)
    env.update({"FD_BENCH_TINY": "1", "MASTER_ADDR": "127.0.0.1"})
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "4", "--master-addr", "127.0.0.1",
           "--master-port", "29761", os.path.join(repo, "bench.py"),
           "--gpus", "4", "--steps", "2", "--warmup", "1"]
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=360,
                         env=env, cwd=repo)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    rec = json.loads(line)
    assert rec["n_gpus"] == 4
    assert rec["steps"] == 2
    assert rec["value"] > 0


def _syncflat_worker(rank, world, port, results):
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(world), "LOCAL_RANK": str(rank),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    import torch.distributed as dist
    from flaxdiff_amd.predictors import EpsilonPredictionTransform
    from flaxdiff_amd.schedulers import CosineNoiseScheduler
    from flaxdiff_amd.trainer import DiffusionTrainer

    torch.manual_seed(7)
    m = Unet(emb_features=32, feature_depths=[8, 16],
             attention_configs=[None, None], num_res_blocks=1,
             num_middle_res_blocks=1, norm_groups=4, context_dim=16)
    tr = DiffusionTrainer(m, CosineNoiseScheduler(1000),
                          EpsilonPredictionTransform(), name="syncflat",
                          checkpoint_base_path="/tmp/fdiff_syncflat",
                          text_context_shape=(4, 16), distributed=True)
    g = torch.Generator().manual_seed(300 + rank)
    batch = {"image": torch.randint(0, 255, (4, 16, 16, 3), generator=g,
                                    dtype=torch.uint8)}
    # the hipGraph-replay protocol: hooks suspended so backward fires no
    # collectives; sync_flat() reduces the whole flat buffer afterwards
    tr.grad_sync.suspended = True
    tr.optimizer.zero_grad()
    images = batch["image"].float()
    images = (images - 127.5) / 127.5
    B = images.shape[0]
    text = tr.null_context.unsqueeze(0).expand(B, *tr.null_context.shape)
    from flaxdiff_amd.utils import get_coeff_shapes_tuple
    timesteps = tr.noise_schedule.sample_timesteps_device(B, "cpu")
    noise = torch.randn(images.shape)
    rates = tr.noise_schedule.get_rates(timesteps,
                                        get_coeff_shapes_tuple(images))
    x_t, c_in, target = tr.model_output_transform.forward_diffusion(
        images, noise, rates)
    x_in, t_in = tr.noise_schedule.transform_inputs(x_t * c_in, timesteps)
    pred = tr.model(x_in, t_in, text)
    pred = tr.model_output_transform.pred_transform(x_t, pred, rates)
    w = tr.noise_schedule.get_weights(timesteps,
                                      get_coeff_shapes_tuple(images))
    ((0.5 * (pred.float() - target.float()) ** 2) * w).mean().backward()
    # no collective fired yet: local grads differ across ranks
    local = tr.optimizer.flat_grad.clone()
    tr.grad_sync.sync_flat()
    tr.grad_sync.suspended = False
    reduced = tr.optimizer.flat_grad.clone()
    gathered = [torch.zeros_like(reduced) for _ in range(world)]
    dist.all_gather(gathered, reduced)
    same = all(torch.allclose(gathered[0], gi, atol=1e-6) for gi in gathered)
    results[rank] = {"reduced_equal": bool(same),
                     "was_local": not torch.allclose(local, reduced)}
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_sync_flat_matches_reduction():
    """sync_flat() (the graph-replay reduction path) produces identical
    summed grads on every rank, and suspended hooks fire no collectives."""
    world = 2
    ctx = mp.get_context("spawn")
    with ctx.Manager() as man:
        results = man.dict()
        port = 29741
        procs = [ctx.Process(target=_syncflat_worker,
                             args=(r, world, port, results))
                 for r in range(world)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=240)
        for p in procs:
            assert p.exitcode == 0, f"worker failed: {p.exitcode}"
        res = dict(results)
    assert res[0]["reduced_equal"] and res[1]["reduced_equal"]
    assert res[0]["was_local"] and res[1]["was_local"]
