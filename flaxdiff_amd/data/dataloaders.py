"""Dataloaders — torch DataLoader stack with the reference's batch contract.

Behavior contract: reference /root/reference/flaxdiff/data/dataloaders.py
(get_dataset_grain :261-358 with per-process sharding :297-305, collate with
shape-mismatch resize and dummy-batch fallback :85-252, background-queue
DataLoaderWithMesh :28-82) and online_loader.py :836-992 (.shard per process,
prefetch threads).

Design: one process per GPU (RCCL DP), so each rank gets a DataLoader over a
rank-sharded sampler; local batch = global / world_size. The collate returns
    {"image": uint8 NHWC tensor, "text": {"input_ids", "attention_mask"}}
and on per-sample failure substitutes a dummy sample instead of killing the
epoch (reference :203-247).
"""
from __future__ import annotations

import queue
import threading
from typing import Any, Dict, Iterator, List

import numpy as np
import torch
from torch.utils.data import DataLoader, Dataset

from .sources import (DataAugmenter, DataSource, ImageAugmenter,
                      SyntheticImageSource, datasetMap)


class AugmentedDataset(Dataset):
    """Source + augmenter with per-sample failure fallback."""

    def __init__(self, source: DataSource, augmenter: DataAugmenter,
                 fallback_image_size: int = 64):
        self.source = source
        self.augmenter = augmenter
        self.fallback_image_size = fallback_image_size
        self._fallback = None

    def __len__(self):
        return len(self.source)

    def _dummy(self) -> Dict[str, Any]:
        if self._fallback is None:
            img = np.zeros((self.fallback_image_size, self.fallback_image_size, 3),
                           dtype=np.uint8)
            self._fallback = self.augmenter({"image": img, "caption": ""})
        return self._fallback

    def __getitem__(self, idx):
        try:
            return self.augmenter(self.source[idx])
        except Exception:  # noqa: BLE001 — reference returns dummy batches (:203-247)
            return self._dummy()


def collate_image_batch(samples: List[Dict[str, Any]]) -> Dict[str, Any]:
    """Stack to the trainer contract; resize stragglers to the majority shape
    (reference dataloaders.py:85-180)."""
    shapes = [s["image"].shape for s in samples]
    target = max(set(shapes), key=shapes.count)
    imgs = []
    for s in samples:
        img = s["image"]
        if img.shape != target:
            t = torch.from_numpy(np.ascontiguousarray(img)).permute(2, 0, 1).float()
            t = torch.nn.functional.interpolate(
                t.unsqueeze(0), size=target[:2], mode="bilinear",
                align_corners=False)
            img = t.squeeze(0).permute(1, 2, 0).clamp(0, 255).byte().numpy()
        imgs.append(torch.from_numpy(np.ascontiguousarray(img)))
    batch = {"image": torch.stack(imgs)}
    if "text" in samples[0]:
        batch["text"] = {
            "input_ids": torch.stack(
                [torch.as_tensor(s["text"]["input_ids"]) for s in samples]),
            "attention_mask": torch.stack(
                [torch.as_tensor(s["text"]["attention_mask"]) for s in samples]),
        }
    return batch


class CaptionDeletionTransform:
    """Drop the caption key from samples — unconditional training at the data
    level (reference dataloaders.py CaptionDeletionTransform)."""

    def __call__(self, element: Dict[str, Any]) -> Dict[str, Any]:
        element.pop("caption", None)
        return element

    map = __call__  # grain-style MapTransform interface


def generate_collate_fn(media_type: str = "image", tokenizer=None):
    """Collate factory keyed by media type (reference dataloaders.py
    generate_collate_fn): tokenizes raw captions when a tokenizer is given."""
    from .videos import collate_video_batch

    base = collate_image_batch if media_type == "image" else collate_video_batch

    def collate(samples: List[Dict[str, Any]]) -> Dict[str, Any]:
        batch = base(samples)
        if tokenizer is not None and "text" not in batch:
            captions = [s.get("caption", "") for s in samples]
            toks = tokenizer(captions)
            batch["text"] = {"input_ids": torch.as_tensor(toks["input_ids"]),
                             "attention_mask": torch.as_tensor(toks["attention_mask"])}
        return batch

    return collate


class ShardedSampler(torch.utils.data.Sampler):
    """Shuffled, rank-sharded, drop-remainder sampler (the grain
    IndexSampler + ShardByJaxProcess equivalent, reference :299-305)."""

    def __init__(self, n: int, rank: int = 0, world_size: int = 1,
                 shuffle: bool = True, seed: int = 0):
        self.n = n
        self.rank = rank
        self.world_size = world_size
        self.shuffle = shuffle
        self.seed = seed
        self.epoch = 0
        self.per_rank = n // world_size

    def set_epoch(self, epoch: int):
        self.epoch = epoch

    def __len__(self):
        return self.per_rank

    def __iter__(self):
        if self.shuffle:
            g = torch.Generator().manual_seed(self.seed * 100003 + self.epoch)
            order = torch.randperm(self.n, generator=g)
        else:
            order = torch.arange(self.n)
        shard = order[self.rank:self.per_rank * self.world_size:self.world_size]
        return iter(shard.tolist())


def get_dataset(dataset_name: str = "synthetic-64", *, global_batch_size: int = 16,
                rank: int = 0, world_size: int = 1, worker_count: int = 4,
                shuffle: bool = True, seed: int = 0, tokenizer=None,
                **source_kwargs) -> DataLoader:
    """Named-dataset loader (get_dataset_grain equivalent, reference :261-358)."""
    if dataset_name not in datasetMap:
        raise KeyError(f"unknown dataset {dataset_name!r}; "
                       f"known: {sorted(datasetMap)}")
    source, augmenter = datasetMap[dataset_name](tokenizer=tokenizer,
                                                 **source_kwargs)
    return make_dataloader(source, augmenter,
                           global_batch_size=global_batch_size, rank=rank,
                           world_size=world_size, worker_count=worker_count,
                           shuffle=shuffle, seed=seed)


def make_dataloader(source: DataSource, augmenter: DataAugmenter, *,
                    global_batch_size: int = 16, rank: int = 0,
                    world_size: int = 1, worker_count: int = 4,
                    shuffle: bool = True, seed: int = 0) -> DataLoader:
    local_bs = max(global_batch_size // world_size, 1)
    ds = AugmentedDataset(source, augmenter)
    sampler = ShardedSampler(len(ds), rank, world_size, shuffle, seed)
    return DataLoader(ds, batch_size=local_bs, sampler=sampler,
                      num_workers=worker_count, collate_fn=collate_image_batch,
                      drop_last=True, persistent_workers=worker_count > 0)


def get_dataset_online(dataset, *, image_key: str = "image",
                       caption_key: str = "caption", image_size: int = 64,
                       global_batch_size: int = 16, rank: int = 0,
                       world_size: int = 1, worker_count: int = 2,
                       tokenizer=None) -> "PrefetchLoader":
    """Streaming loader over a HF dataset (OnlineStreamingDataLoader
    equivalent, reference online_loader.py:836-992): shards by rank, augments
    in a thread pool, prefetches into a bounded queue."""
    if hasattr(dataset, "shard") and world_size > 1:
        dataset = dataset.shard(num_shards=world_size, index=rank)
    aug = ImageAugmenter(image_size=image_size, tokenizer=tokenizer)
    local_bs = max(global_batch_size // world_size, 1)
    return PrefetchLoader(dataset, aug, image_key, caption_key, local_bs,
                          worker_count)


class PrefetchLoader:
    """Thread-prefetched batch iterator over an iterable/indexable dataset.

    Mirrors the reference's queue-based loader (dataloaders.py:28-82 +
    online_loader.py:589-704): daemon workers fill a bounded queue; dead
    workers surface as StopIteration rather than a hang.
    """

    def __init__(self, dataset, augmenter: DataAugmenter, image_key: str,
                 caption_key: str, batch_size: int, workers: int = 2,
                 queue_size: int = 8):
        self.dataset = dataset
        self.augmenter = augmenter
        self.image_key = image_key
        self.caption_key = caption_key
        self.batch_size = batch_size
        self.queue: "queue.Queue" = queue.Queue(maxsize=queue_size)
        self._stop = threading.Event()
        self._thread = threading.Thread(target=self._fill, daemon=True)
        self._thread.start()

    def _samples(self) -> Iterator[Dict[str, Any]]:
        while True:
            for row in self.dataset:
                img = row[self.image_key]
                cap = row.get(self.caption_key, "") if isinstance(row, dict) else ""
                try:
                    yield self.augmenter({"image": np.asarray(img), "caption": cap})
                except Exception:  # noqa: BLE001
                    continue

    def _fill(self):
        batch: List[Dict[str, Any]] = []
        for s in self._samples():
            if self._stop.is_set():
                return
            batch.append(s)
            if len(batch) == self.batch_size:
                self.queue.put(collate_image_batch(batch))
                batch = []

    def __iter__(self):
        return self

    def __next__(self):
        while True:
            try:
                return self.queue.get(timeout=5.0)
            except queue.Empty:
                if not self._thread.is_alive():
                    raise StopIteration from None

    def stop(self):
        self._stop.set()


class DevicePrefetcher:
    """Wraps a host loader and keeps the NEXT batch's H2D copies in flight on
    a side HIP stream while the current step computes (the reference's
    DataLoaderWithMesh pushed device arrays from a background thread,
    dataloaders.py:28-82)."""

    def __init__(self, loader, device, compute_dtype=None):
        self.loader = loader
        self.device = torch.device(device)
        self.compute_dtype = compute_dtype
        self._use_stream = self.device.type == "cuda"
        self.stream = torch.cuda.Stream() if self._use_stream else None

    def _to_device(self, batch):
        out = {}
        for k, v in batch.items():
            if torch.is_tensor(v):
                out[k] = v.to(self.device, non_blocking=True)
            elif isinstance(v, dict):
                out[k] = {kk: vv.to(self.device, non_blocking=True)
                          if torch.is_tensor(vv) else vv
                          for kk, vv in v.items()}
            else:
                out[k] = v
        return out

    def __iter__(self):
        it = iter(self.loader)
        if not self._use_stream:
            for batch in it:
                yield self._to_device(batch)
            return
        nxt = None
        for batch in it:
            with torch.cuda.stream(self.stream):
                staged = self._to_device(batch)
            if nxt is not None:
                yield nxt
            torch.cuda.current_stream().wait_stream(self.stream)
            nxt = staged
        if nxt is not None:
            yield nxt

    def __len__(self):
        return len(self.loader)
