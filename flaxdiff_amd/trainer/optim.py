"""Flat-buffer AdamW + EMA optimizer.

All parameters are re-parented into ONE contiguous fp32 master buffer, laid
out in REVERSE registration order so backward fills the matching flat
gradient buffer front-to-back (enables bucketed all-reduce overlap —
parallel.GradBucketSynchronizer). The Adam update + EMA lerp + grad-mean
scaling run as ONE fused HIP kernel pass over the flat buffers on GPU
(ops.fused_adamw_ema); on CPU the same math runs as flat torch ops.

Replaces the reference's optax.adamw + apply_ema tree_map
(diffusion_trainer.py:31-37, training.py:594-608).
"""
from __future__ import annotations

import math
from typing import Callable, Iterable, List, Optional

import torch

from .. import ops


class FlatAdamWEMA:
    def __init__(self, module: torch.nn.Module, lr: float = 2.7e-4,
                 betas=(0.9, 0.999), eps: float = 1e-8, weight_decay: float = 0.0,
                 ema_decay: float = 0.999, lr_schedule: Optional[Callable[[int], float]] = None,
                 grad_clip_norm: Optional[float] = None,
                 skip_nonfinite: bool = True):
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.ema_decay = ema_decay
        self.lr_schedule = lr_schedule
        self.grad_clip_norm = grad_clip_norm
        self.skip_nonfinite = skip_nonfinite
        self._skipped_host = 0
        self._skip_dev: Optional[torch.Tensor] = None  # GPU-path skip counter
        self._step_dev: Optional[torch.Tensor] = None  # hipGraph-path counter
        self.step_count = 0

        params = [p for p in module.parameters() if p.requires_grad]
        params = list(reversed(params))  # backward-production order
        total = sum(p.numel() for p in params)
        device = params[0].device if params else torch.device("cpu")

        self.flat = torch.zeros(total, dtype=torch.float32, device=device)
        self.flat_grad = torch.zeros(total, dtype=torch.float32, device=device)
        self.exp_avg = torch.zeros(total, dtype=torch.float32, device=device)
        self.exp_avg_sq = torch.zeros(total, dtype=torch.float32, device=device)
        self.ema = torch.zeros(total, dtype=torch.float32, device=device)

        # bf16 shadow of the fp32 masters, refreshed in the SAME pass as the
        # fused Adam kernel. Forward reads the shadow directly (via the
        # autograd shim in models.common), which removes the per-call
        # fp32->bf16 weight-cast kernels (~500/step on the 64px UNet).
        self.flat_bf16 = torch.zeros(total, dtype=torch.bfloat16, device=device)

        self.params: List[torch.nn.Parameter] = params
        self.offsets: List[int] = []
        off = 0
        with torch.no_grad():
            for p in params:
                n = p.numel()
                self.flat[off:off + n].copy_(p.data.reshape(-1).float())
                p.data = self.flat[off:off + n].view(p.shape)
                p.grad = self.flat_grad[off:off + n].view(p.shape)
                p._shadow_bf16 = self.flat_bf16[off:off + n].view(p.shape)
                self.offsets.append(off)
                off += n
        self.ema.copy_(self.flat)
        self.flat_bf16.copy_(self.flat)
        self.total = total

        # Transposed bf16 shadows for 64-divisible 2-D dense weights: the
        # MFMA GEMM forward consumes W^T [N,K] (fast NT operand layout); one
        # batched 64x64 transpose kernel refreshes them per step instead of
        # a per-call transpose (ops/hip/gemm_bf16.hip transpose_shadows).
        self.flat_bf16_t = None
        self._t_tiles = None
        def _t_shape(p):
            if p.dim() == 2:
                kn = (p.shape[0], p.shape[1])
            elif p.dim() == 4 and p.shape[0] == 1 and p.shape[1] == 1:
                kn = (p.shape[2], p.shape[3])     # 1x1 conv weight
            else:
                return None
            return kn if kn[0] % 64 == 0 and kn[1] % 64 == 0 else None

        t_params = [(i, p, _t_shape(p)) for i, p in enumerate(params)
                    if _t_shape(p) is not None]
        if t_params and device.type == "cuda":
            tot_t = sum(p.numel() for _, p, _kn in t_params)
            self.flat_bf16_t = torch.zeros(tot_t, dtype=torch.bfloat16,
                                           device=device)
            tiles = []
            off_t = 0
            for i, p, (K, N) in t_params:
                src = self.offsets[i]
                for n0 in range(0, N, 64):
                    for k0 in range(0, K, 64):
                        tiles.append((src + k0 * N + n0, off_t + n0 * K + k0,
                                      N, K))
                p._shadow_bf16_t = self.flat_bf16_t[off_t:off_t + p.numel()] \
                    .view(N, K)
                off_t += p.numel()
            self._t_tiles = torch.tensor(tiles, dtype=torch.int32,
                                         device=device)
            self._refresh_t()

    def _refresh_t(self):
        if self.flat_bf16_t is not None and ops.hip_available():
            ops.transpose_shadows(self.flat_bf16, self.flat_bf16_t,
                                  self._t_tiles)

    # ------------------------------------------------------------------
    def zero_grad(self):
        self.flat_grad.zero_()

    def current_lr(self) -> float:
        if self.lr_schedule is not None:
            return self.lr_schedule(self.step_count)
        return self.lr

    @property
    def skipped_steps(self) -> int:
        n = self._skipped_host
        if self._skip_dev is not None:
            n += int(self._skip_dev.item())
        return n

    @torch.no_grad()
    def step(self, grad_scale: float = 1.0, use_step_dev: bool = False):
        """grad_scale multiplies gradients before use (e.g. 1/world_size so the
        all-reduce SUM becomes the reference's pmean).

        use_step_dev (hipGraph capture): keep the Adam step counter in a
        device int32 so bias correction is recomputed per graph REPLAY
        instead of baked at capture. The caller owns advancing the host
        `step_count` mirror per replay (checkpointing still works)."""
        self.step_count += 1
        lr = self.current_lr()

        if use_step_dev:
            assert self.flat.is_cuda and ops.hip_available(), \
                "use_step_dev needs the GPU fused-optimizer path"
            assert self.lr_schedule is None, \
                "hipGraph step requires a constant lr (schedule is host-side)"
            if self._step_dev is None:
                self._step_dev = torch.tensor([self.step_count - 1],
                                              dtype=torch.int32,
                                              device=self.flat.device)
            self._step_dev.add_(1)

        if self.flat.is_cuda and ops.hip_available():
            # Clip factor and nonfinite gate are built ON DEVICE so the hot
            # step never does a GPU->host sync. A gated (skipped)
            # step leaves step_count advanced on the host — bias correction
            # deviates by one exponent tick per (rare, abnormal) skip.
            scale_dev = None
            if self.grad_clip_norm is not None or self.skip_nonfinite:
                gnorm = self.flat_grad.norm() * grad_scale
                scale_dev = gnorm.new_full((1,), grad_scale)
                if self.grad_clip_norm is not None:
                    scale_dev = scale_dev * torch.clamp(
                        self.grad_clip_norm / (gnorm + 1e-6), max=1.0)
                if self.skip_nonfinite:
                    if self._skip_dev is None:
                        self._skip_dev = torch.zeros(1, dtype=torch.int32,
                                                     device=self.flat.device)
                    scale_dev = torch.where(torch.isfinite(gnorm), scale_dev,
                                            torch.zeros_like(scale_dev))
                scale_dev = scale_dev.reshape(1).float().contiguous()
            ops.fused_adamw_ema(self.flat, self.flat_grad, self.exp_avg,
                                self.exp_avg_sq, self.ema, self.flat_bf16,
                                lr=lr, beta1=self.beta1, beta2=self.beta2,
                                eps=self.eps, weight_decay=self.weight_decay,
                                step=self.step_count, ema_decay=self.ema_decay,
                                grad_scale=grad_scale, scale_dev=scale_dev,
                                skip_ctr=self._skip_dev if self.skip_nonfinite else None,
                                step_dev=self._step_dev if use_step_dev else None)
            self._refresh_t()
            return

        if self.grad_clip_norm is not None or self.skip_nonfinite:
            gnorm = float(self.flat_grad.norm()) * grad_scale
            if self.skip_nonfinite and not math.isfinite(gnorm):
                # DynamicScale skip-on-nonfinite parity (reference
                # diffusion_trainer.py:229-240): drop the step, keep state.
                self.step_count -= 1
                self._skipped_host += 1
                return
            if self.grad_clip_norm is not None:
                clip = self.grad_clip_norm / (gnorm + 1e-6)
                if clip < 1.0:
                    grad_scale = grad_scale * clip

        g = self.flat_grad
        if grad_scale != 1.0:
            g = g * grad_scale
        self.exp_avg.mul_(self.beta1).add_(g, alpha=1 - self.beta1)
        self.exp_avg_sq.mul_(self.beta2).addcmul_(g, g, value=1 - self.beta2)
        bc1 = 1 - self.beta1 ** self.step_count
        bc2 = 1 - self.beta2 ** self.step_count
        denom = (self.exp_avg_sq / bc2).sqrt_().add_(self.eps)
        update = (self.exp_avg / bc1) / denom
        if self.weight_decay:
            update = update + self.weight_decay * self.flat
        self.flat.add_(update, alpha=-lr)
        self.ema.mul_(self.ema_decay).add_(self.flat, alpha=1 - self.ema_decay)
        self.flat_bf16.copy_(self.flat)

    # ------------------------------------------------------------------
    def ema_view(self, param: torch.nn.Parameter) -> torch.Tensor:
        i = self.params.index(param)
        off = self.offsets[i]
        return self.ema[off:off + param.numel()].view(param.shape)

    def load_ema_into_params(self):
        """Swap EMA weights into the live parameters (for eval); returns a
        tensor holding the previous params so they can be restored."""
        saved = self.flat.clone()
        self.flat.copy_(self.ema)
        self.flat_bf16.copy_(self.flat)
        self._refresh_t()
        return saved

    def restore_params(self, saved: torch.Tensor):
        self.flat.copy_(saved)
        self.flat_bf16.copy_(self.flat)
        self._refresh_t()

    # ------------------------------------------------------------------
    def state_dict(self):
        return {
            "flat": self.flat, "exp_avg": self.exp_avg,
            "exp_avg_sq": self.exp_avg_sq, "ema": self.ema,
            "step_count": self.step_count,
            "hyper": {"lr": self.lr, "betas": (self.beta1, self.beta2),
                      "eps": self.eps, "weight_decay": self.weight_decay,
                      "ema_decay": self.ema_decay},
        }

    def load_state_dict(self, sd):
        with torch.no_grad():
            self.flat.copy_(sd["flat"])
            self.exp_avg.copy_(sd["exp_avg"])
            self.exp_avg_sq.copy_(sd["exp_avg_sq"])
            self.ema.copy_(sd["ema"])
            self.flat_bf16.copy_(self.flat)
        self._refresh_t()
        self.step_count = int(sd["step_count"])
        if self._step_dev is not None:
            self._step_dev.fill_(self.step_count)


def warmup_cosine_schedule(base_lr: float, warmup_steps: int, total_steps: int,
                           final_scale: float = 0.0) -> Callable[[int], float]:
    """Warmup->cosine LR (reference training.py:597-601)."""
    def sched(step: int) -> float:
        if warmup_steps > 0 and step < warmup_steps:
            return base_lr * step / warmup_steps
        if total_steps <= warmup_steps:
            return base_lr
        t = (step - warmup_steps) / max(1, total_steps - warmup_steps)
        t = min(max(t, 0.0), 1.0)
        return base_lr * (final_scale + (1 - final_scale) * 0.5 * (1 + math.cos(math.pi * t)))
    return sched
