"""SimpleTrainer — generic supervised training orchestration.

Behavior contract: reference /root/reference/flaxdiff/trainer/simple_trainer.py:148-677:
distributed autodetect, rank-0-only checkpointing/logging, best-state
tracking, the NaN/abnormal-loss watchdog with best-state restore
(:542-575), periodic save, epoch-level fit loop.

MI355X-native execution: one process per GPU, RCCL over xGMI
(parallel.init_distributed), flat-master FlatAdamWEMA optimizer, bucketed
grad all-reduce overlapped with backward (parallel.GradBucketSynchronizer).
"""
from __future__ import annotations

import math
import time
from typing import Callable, Dict, Optional

import torch

from .. import parallel
from ..parallel import DistContext, GradBucketSynchronizer, init_distributed
from ..utils import RandomMarkovState
from ..utils.checkpoints import CheckpointManager
from .optim import FlatAdamWEMA


class _NoopLogger:
    def log(self, *a, **k):
        pass

    def finish(self):
        pass


def _make_wandb(project, experiment_name, config, rank):
    if rank != 0 or project is None:
        return _NoopLogger()
    try:
        import wandb
        return wandb.init(project=project, name=experiment_name, config=config,
                          resume="allow")
    except Exception:
        return _NoopLogger()


class SimpleTrainer:
    def __init__(self,
                 model: torch.nn.Module,
                 optimizer: Optional[FlatAdamWEMA] = None,
                 *,
                 name: str = "run",
                 checkpoint_base_path: str = "./checkpoints",
                 checkpoint_step: Optional[int] = None,
                 load_from_checkpoint: bool = False,
                 max_checkpoints_to_keep: int = 2,
                 rngs: Optional[RandomMarkovState] = None,
                 wandb_project: Optional[str] = None,
                 wandb_config: Optional[dict] = None,
                 loss_fn: Callable = torch.nn.functional.mse_loss,
                 compute_dtype: torch.dtype = torch.float32,
                 optimizer_kwargs: Optional[dict] = None,
                 distributed: bool = True,
                 train_start_step_override: Optional[int] = None):
        self.name = name
        self.dist: DistContext = init_distributed() if distributed else DistContext()
        self.device = self.dist.device
        self.compute_dtype = compute_dtype
        self.loss_fn = loss_fn

        self.model = model.to(self.device)
        parallel.broadcast_module(self.model)

        self.optimizer = optimizer if optimizer is not None else \
            FlatAdamWEMA(self.model, **(optimizer_kwargs or {}))
        self.grad_sync = GradBucketSynchronizer(self.optimizer.params,
                                                self.optimizer.flat_grad,
                                                self.optimizer.offsets)

        self.rngs = (rngs if rngs is not None else RandomMarkovState(0)).fold_in(self.dist.rank)
        self.global_step = 0
        self.best_loss = math.inf
        self.best_state: Optional[dict] = None
        self._ckpt = CheckpointManager(checkpoint_base_path, name, max_checkpoints_to_keep)
        self.wandb = _make_wandb(wandb_project, name, wandb_config, self.dist.rank)

        if load_from_checkpoint:
            self.load(step=checkpoint_step)
        if train_start_step_override is not None:
            self.global_step = train_start_step_override

    # ------------------------------------------------------------------
    # checkpointing (schema: SURVEY.md §5.4)
    # ------------------------------------------------------------------
    def _state_payload(self) -> dict:
        return {
            "rngs": self.rngs.seed,
            "state": {
                "params": {k: v for k, v in self.model.state_dict().items()},
                "opt_state": self.optimizer.state_dict(),
                "ema_params": self.optimizer.ema,
                "step": self.global_step,
                "rngs": self.rngs.seed,
            },
            "best_state": self.best_state,
            "best_loss": self.best_loss,
            "epoch": getattr(self, "epoch", 0),
        }

    def save(self, config: Optional[dict] = None, block: bool = False):
        if not self.dist.is_main:
            return
        self._ckpt.save(self.global_step, self._state_payload(), config=config, block=block)

    def load(self, step: Optional[int] = None):
        payload = self._ckpt.load(step)
        if payload is None:
            return False
        state = payload["state"]
        self.model.load_state_dict(state["params"])
        # re-flatten into the master buffer
        with torch.no_grad():
            for p, off in zip(self.optimizer.params, self.optimizer.offsets):
                self.optimizer.flat[off:off + p.numel()].copy_(p.data.reshape(-1).float())
                p.data = self.optimizer.flat[off:off + p.numel()].view(p.shape)
        self.optimizer.load_state_dict(state["opt_state"])
        self.global_step = int(state["step"])
        self.best_loss = float(payload.get("best_loss", math.inf))
        if payload.get("best_loss", 0) == 0:  # corrupt best (reference :363-365)
            self.best_loss = math.inf
        self.best_state = payload.get("best_state")
        rngs = payload.get("rngs")
        if rngs is not None:
            self.rngs = RandomMarkovState(int(rngs))
        return True

    # ------------------------------------------------------------------
    # train step — subclasses override
    # ------------------------------------------------------------------
    def train_step(self, batch: Dict[str, torch.Tensor]) -> Dict[str, float]:
        """Generic supervised step: batch = {'x': ..., 'y': ...}."""
        x = batch["x"].to(self.device, self.compute_dtype)
        y = batch["y"].to(self.device, self.compute_dtype)
        self.optimizer.zero_grad()
        pred = self.model(x)
        loss = self.loss_fn(pred.float(), y.float())
        loss.backward()
        self.grad_sync.sync()
        self.optimizer.step(grad_scale=1.0 / self.dist.world_size)
        loss = parallel.all_reduce_mean_scalar(loss.detach())
        return {"loss": float(loss)}

    def val_step(self, batch) -> Dict[str, float]:
        with torch.no_grad():
            x = batch["x"].to(self.device, self.compute_dtype)
            y = batch["y"].to(self.device, self.compute_dtype)
            loss = self.loss_fn(self.model(x).float(), y.float())
        return {"loss": float(loss)}

    # ------------------------------------------------------------------
    # watchdog (reference simple_trainer.py:542-575)
    # ------------------------------------------------------------------
    def _snapshot_best(self, loss: float):
        if loss < self.best_loss:
            self.best_loss = loss
            self.best_state = {
                "flat": self.optimizer.flat.detach().cpu().clone(),
                "ema": self.optimizer.ema.detach().cpu().clone(),
            }

    def _abnormal_loss(self, loss: float) -> bool:
        return not math.isfinite(loss) or loss <= 1e-8

    def _recover(self):
        nan_params = int(torch.isnan(self.optimizer.flat).sum())
        if self.dist.is_main:
            print(f"[watchdog] abnormal loss; nan params in master buffer: {nan_params}; "
                  f"restoring best state (best_loss={self.best_loss})")
        if self.best_state is not None:
            with torch.no_grad():
                self.optimizer.flat.copy_(self.best_state["flat"].to(self.device))
                self.optimizer.ema.copy_(self.best_state["ema"].to(self.device))
                # the forward reads the bf16 shadows, not the fp32 masters —
                # without this refresh the next steps still run on the
                # corrupted weights until an optimizer step rewrites them
                self.optimizer.flat_bf16.copy_(self.optimizer.flat)
                self.optimizer._refresh_t()
        self.optimizer.exp_avg.zero_()
        self.optimizer.exp_avg_sq.zero_()
        if torch.cuda.is_available():
            torch.cuda.empty_cache()

    # ------------------------------------------------------------------
    # loops (reference simple_trainer.py:500-677)
    # ------------------------------------------------------------------
    def train_loop(self, data_iter, steps: int, save_every: Optional[int] = None,
                   log_every: int = 100, config: Optional[dict] = None) -> float:
        self.model.train()
        running = 0.0
        count = 0
        t0 = time.time()
        for _ in range(steps):
            batch = next(data_iter)
            metrics = self.train_step(batch)
            self.global_step += 1
            loss = metrics["loss"]

            if self._abnormal_loss(loss):
                self._recover()
                continue
            self._snapshot_best(loss)
            running += loss
            count += 1

            if self.global_step % log_every == 0 and self.dist.is_main:
                dt = (time.time() - t0) / max(count, 1)
                self.wandb.log({"train/loss": loss,
                                "train/avg_time_per_step": dt,
                                "train/step": self.global_step})
            if save_every and self.global_step % save_every == 0:
                self.save(config=config)
        return running / max(count, 1)

    def fit(self, data_iter, steps_per_epoch: int, epochs: int,
            val_fn: Optional[Callable] = None, save_every: Optional[int] = None,
            config: Optional[dict] = None):
        for epoch in range(epochs):
            self.epoch = epoch
            t0 = time.time()
            avg_loss = self.train_loop(data_iter, steps_per_epoch,
                                       save_every=save_every, config=config)
            epoch_time = time.time() - t0
            if self.dist.is_main:
                self.wandb.log({"train/epoch_loss": avg_loss,
                                "train/epoch_time": epoch_time,
                                "train/epoch": epoch})
                print(f"epoch {epoch}: loss {avg_loss:.5f} ({epoch_time:.1f}s)")
            if val_fn is not None:
                val_fn(self)
            self.save(config=config)
        self._ckpt.wait()
        return self
