"""Data parallelism over RCCL/xGMI — one process per GPU.

Replaces the reference's JAX SPMD mesh + lax.pmean with an
explicit design sized for MI355X topology: gradients live in ONE flat
buffer laid out in backward order, so bucketed `all_reduce` slices launch as
backward produces them and overlap with the remaining backward on RCCL's
comm stream. xGMI is point-to-point (7 links x ~153 GB/s per GPU), so buckets
default large (64 MB) — ring all-reduce is per-link bound and per-launch
latency matters more than fine-grained overlap.

Collective mapping:
  pmean(grads)  -> bucketed all_reduce(SUM) + 1/N folded into the optimizer
  pmean(loss)   -> single scalar all_reduce piggybacked after the last bucket
  fold_in(rank) -> RandomMarkovState.fold_in(rank)
"""
from __future__ import annotations

import os
from dataclasses import dataclass
from typing import List, Optional

import torch
import torch.distributed as dist


def _default_device() -> torch.device:
    return torch.device("cuda:0") if torch.cuda.is_available() else torch.device("cpu")


@dataclass
class DistContext:
    rank: int = 0
    world_size: int = 1
    local_rank: int = 0
    device: torch.device = None  # type: ignore[assignment]

    def __post_init__(self):
        if self.device is None:
            self.device = _default_device()
            if self.device.type == "cuda":
                torch.cuda.set_device(self.device)

    @property
    def is_distributed(self) -> bool:
        return self.world_size > 1

    @property
    def is_main(self) -> bool:
        return self.rank == 0


def init_distributed(backend: Optional[str] = None) -> DistContext:
    """Process-group bootstrap (reference: jax.distributed.initialize(),
    training.py:235). env:// rendezvous, one rank per GPU."""
    if "WORLD_SIZE" not in os.environ or int(os.environ.get("WORLD_SIZE", "1")) <= 1:
        dev = torch.device("cuda:0") if torch.cuda.is_available() else torch.device("cpu")
        if dev.type == "cuda":
            torch.cuda.set_device(dev)
        return DistContext(device=dev)

    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if not dist.is_initialized():
        dist.init_process_group(backend=backend, rank=rank, world_size=world)
    if torch.cuda.is_available():
        device = torch.device(f"cuda:{local_rank}")
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")
    return DistContext(rank=rank, world_size=world, local_rank=local_rank, device=device)


_RANK_COLORS = [32, 33, 34, 35, 36, 31, 92, 93]  # ANSI per-rank


def rank_print(*args, rank: Optional[int] = None, **kwargs):
    """Rank-colored console logging (reference simple_trainer.py:32-41)."""
    if rank is None:
        rank = dist.get_rank() if dist.is_initialized() else 0
    color = _RANK_COLORS[rank % len(_RANK_COLORS)]
    print(f"\033[{color}m[rank {rank}]\033[0m", *args, **kwargs)


def barrier():
    if dist.is_initialized():
        dist.barrier()


def all_reduce_mean_scalar(x: torch.Tensor) -> torch.Tensor:
    if dist.is_initialized():
        dist.all_reduce(x, op=dist.ReduceOp.SUM)
        x /= dist.get_world_size()
    return x


def broadcast_module(module: torch.nn.Module, src: int = 0):
    """Make rank-0's initialization authoritative."""
    if not dist.is_initialized():
        return
    for p in module.state_dict().values():
        if torch.is_tensor(p) and p.is_floating_point():
            dist.broadcast(p.data, src=src)


class GradBucketSynchronizer:
    """Bucketed all-reduce over a flat gradient buffer.

    Offsets are assigned in reverse parameter order (last layer first =
    backward production order), so bucket k's gradients are complete before
    bucket k+1's. Per-parameter post-accumulate hooks count arrivals; when a
    bucket fills, its slice all_reduces with async_op=True — NCCL runs it on
    its own stream, overlapping the remaining backward.
    """

    def __init__(self, params: List[torch.nn.Parameter], flat_grad: torch.Tensor,
                 offsets: List[int], bucket_bytes: Optional[int] = None):
        if bucket_bytes is None:
            # xGMI tuning knob (docs/multigpu_tuning.md): ring all-reduce on
            # 8 GPUs is per-link bound (7 x ~153 GB/s point-to-point), so the
            # bucket must be big enough to amortize per-step latency but
            # small enough to start reducing early in backward.
            bucket_bytes = int(os.environ.get("FD_BUCKET_MB", "64")) << 20
        self.flat_grad = flat_grad
        self.enabled = dist.is_initialized()
        elem = flat_grad.element_size()
        bucket_elems = max(1, bucket_bytes // elem)

        # params with their flat ranges, ordered by offset
        ranges = sorted(zip(offsets, [p.numel() for p in params], params))
        self.buckets = []          # list of (start, end, n_params)
        self._param_bucket = {}    # param -> bucket idx
        start = 0
        cur_params = 0
        end = 0
        for off, n, p in ranges:
            self._param_bucket[p] = len(self.buckets)
            cur_params += 1
            end = off + n
            if end - start >= bucket_elems:
                self.buckets.append([start, end, cur_params])
                start = end
                cur_params = 0
        if cur_params:
            self.buckets.append([start, end, cur_params])

        self._arrived = [0] * len(self.buckets)
        self._launched = [False] * len(self.buckets)
        self._works = []
        # suspended: hooks become no-ops. The hipGraph-captured train step
        # replays backward without firing collectives (RCCL must stay out of
        # the captured graph); the caller then reduces with sync_flat().
        self.suspended = False

        if self.enabled:
            for off, n, p in ranges:
                p.register_post_accumulate_grad_hook(self._hook)

    def _hook(self, p):
        if self.suspended:
            return
        bi = self._param_bucket[p]
        self._arrived[bi] += 1
        if self._arrived[bi] >= self.buckets[bi][2] and not self._launched[bi]:
            self._launch(bi)

    def _launch(self, bi):
        start, end, _ = self.buckets[bi]
        w = dist.all_reduce(self.flat_grad[start:end], op=dist.ReduceOp.SUM,
                            async_op=True)
        self._works.append(w)
        self._launched[bi] = True

    def sync(self):
        """Call after backward: launches any straggler buckets and waits."""
        if not self.enabled or self.suspended:
            return
        for bi in range(len(self.buckets)):
            if not self._launched[bi]:
                self._launch(bi)
        for w in self._works:
            w.wait()
        self._works.clear()
        self._arrived = [0] * len(self.buckets)
        self._launched = [False] * len(self.buckets)

    def sync_flat(self):
        """All-reduce the whole flat gradient buffer bucket-by-bucket in one
        burst (async launches, then wait). Used after a hipGraph replay of
        backward, where the per-param hooks were suspended — same bucket
        sizing as the overlapped path so xGMI ring bandwidth is unchanged,
        only the backward overlap is given up."""
        if not self.enabled:
            return
        works = [dist.all_reduce(self.flat_grad[s0:e0], op=dist.ReduceOp.SUM,
                                 async_op=True)
                 for s0, e0, _ in self.buckets]
        for w in works:
            w.wait()
