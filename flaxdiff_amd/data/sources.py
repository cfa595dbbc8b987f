"""Data sources + augmenters.

Behavior contract: reference /root/reference/flaxdiff/data/sources/base.py:8-141
(DataSource / DataAugmenter ABCs + factories), sources/images.py (TFDS/GCS
sources :100-270, augmenters :166-313), dataset_map.py:19-174 (registry).

MI355X-native stack: torch Dataset / DataLoader instead of grain; sources
yield {"image": uint8 HWC numpy/tensor, "caption"/"text": str or tokens} and
the augmenter produces the trainer batch contract
    {"image": uint8 NHWC, "text": {input_ids, attention_mask}}
(reference sources/images.py:190-196). Synthetic sources serve the BASELINE
benchmark configs (no network in the target environment).
"""
from __future__ import annotations

import abc
import os
from pathlib import Path
from typing import Any, Callable, Dict, List, Optional, Tuple

import numpy as np
import torch

# Augment hygiene env bridge (reference training.py:221-223 ->
# sources/images.py:166,304): "all" | "flip" | "none"
AUGMENT_MODE_ENV = "FLAXDIFF_AUGMENT_MODE"


def _augment_mode() -> str:
    return os.environ.get(AUGMENT_MODE_ENV, "all")


class DataSource(abc.ABC):
    """Indexable source of raw samples (reference sources/base.py:8-43)."""

    @abc.abstractmethod
    def __len__(self) -> int: ...

    @abc.abstractmethod
    def __getitem__(self, idx: int) -> Dict[str, Any]: ...


class DataAugmenter(abc.ABC):
    """Per-sample transform raw -> model-ready (reference sources/base.py:45-80)."""

    @abc.abstractmethod
    def __call__(self, sample: Dict[str, Any]) -> Dict[str, Any]: ...


# ---------------------------------------------------------------------------
# sources
# ---------------------------------------------------------------------------

class SyntheticImageSource(DataSource):
    """Deterministic synthetic uint8 images + captions — the benchmark source
    (BASELINE runs on synthetic data; there is no network for datasets)."""

    def __init__(self, image_size: int = 64, num_samples: int = 10000,
                 channels: int = 3, seed: int = 0):
        self.image_size = image_size
        self.num_samples = num_samples
        self.channels = channels
        self.seed = seed

    def __len__(self):
        return self.num_samples

    def __getitem__(self, idx):
        rng = np.random.default_rng(self.seed * 1_000_003 + idx)
        img = rng.integers(0, 256, (self.image_size, self.image_size,
                                    self.channels), dtype=np.uint8)
        return {"image": img, "caption": f"synthetic sample {idx}"}


class ImageFolderSource(DataSource):
    """Local directory of images; caption from a sidecar .txt or the filename
    (stands in for the reference's GCS ArrayRecord source)."""

    EXTS = (".png", ".jpg", ".jpeg", ".bmp", ".webp")

    def __init__(self, root: str):
        self.root = Path(root)
        self.files: List[Path] = sorted(
            p for p in self.root.rglob("*") if p.suffix.lower() in self.EXTS)
        if not self.files:
            raise FileNotFoundError(f"no images under {root}")

    def __len__(self):
        return len(self.files)

    def __getitem__(self, idx):
        from PIL import Image
        p = self.files[idx]
        img = np.asarray(Image.open(p).convert("RGB"))
        cap_file = p.with_suffix(".txt")
        caption = cap_file.read_text().strip() if cap_file.exists() else p.stem
        return {"image": img, "caption": caption}


class HFDatasetSource(DataSource):
    """HuggingFace `datasets`-backed source (reference online_loader.py's
    load_dataset path, :836-921). Works offline with on-disk datasets."""

    def __init__(self, dataset, image_key: str = "image",
                 caption_key: str = "caption"):
        self.ds = dataset
        self.image_key = image_key
        self.caption_key = caption_key

    def __len__(self):
        return len(self.ds)

    def __getitem__(self, idx):
        row = self.ds[int(idx)]
        img = row[self.image_key]
        if not isinstance(img, np.ndarray):
            img = np.asarray(img)
        cap = row.get(self.caption_key, "") if isinstance(row, dict) else ""
        return {"image": img, "caption": cap}


class TensorSource(DataSource):
    """Wraps pre-loaded tensors/arrays (tests, golden data)."""

    def __init__(self, images: np.ndarray, captions: Optional[List[str]] = None):
        self.images = images
        self.captions = captions

    def __len__(self):
        return len(self.images)

    def __getitem__(self, idx):
        cap = self.captions[idx] if self.captions else ""
        return {"image": np.asarray(self.images[idx]), "caption": cap}


# ---------------------------------------------------------------------------
# augmenters
# ---------------------------------------------------------------------------

def _resize_uint8(img: np.ndarray, size: int) -> np.ndarray:
    """Bilinear uint8 HWC resize via torch (cv2-free)."""
    if img.shape[0] == size and img.shape[1] == size:
        return img
    t = torch.from_numpy(np.ascontiguousarray(img)).permute(2, 0, 1).float()
    t = torch.nn.functional.interpolate(t.unsqueeze(0), size=(size, size),
                                        mode="bilinear", align_corners=False)
    return t.squeeze(0).permute(1, 2, 0).clamp(0, 255).byte().numpy()


class ImageAugmenter(DataAugmenter):
    """Resize + optional random flip / color jitter, then tokenize the caption
    (reference sources/images.py:166-196,304-335)."""

    def __init__(self, image_size: int, tokenizer: Optional[Callable] = None,
                 max_length: int = 77, rng: Optional[np.random.Generator] = None):
        self.image_size = image_size
        self.tokenizer = tokenizer
        self.max_length = max_length
        self.rng = rng or np.random.default_rng()

    def _tokenize(self, caption: str) -> Dict[str, np.ndarray]:
        if self.tokenizer is not None:
            tok = self.tokenizer(caption, padding="max_length",
                                 max_length=self.max_length, truncation=True,
                                 return_tensors="np")
            return {"input_ids": tok["input_ids"][0],
                    "attention_mask": tok["attention_mask"][0]}
        # deterministic hash-token fallback (offline tests)
        ids = np.zeros(self.max_length, dtype=np.int64)
        mask = np.zeros(self.max_length, dtype=np.int64)
        words = caption.split()[: self.max_length]
        for i, w in enumerate(words):
            ids[i] = (hash(w) % 30000) + 1
            mask[i] = 1
        return {"input_ids": ids, "attention_mask": mask}

    def __call__(self, sample: Dict[str, Any]) -> Dict[str, Any]:
        img = _resize_uint8(np.asarray(sample["image"]), self.image_size)
        mode = _augment_mode()
        if mode in ("all", "flip") and self.rng.random() < 0.5:
            img = img[:, ::-1].copy()
        if mode == "all" and self.rng.random() < 0.1:
            jitter = self.rng.uniform(0.9, 1.1)
            img = np.clip(img.astype(np.float32) * jitter, 0, 255).astype(np.uint8)
        return {"image": img, "text": self._tokenize(str(sample.get("caption", "")))}


# ---------------------------------------------------------------------------
# registry (reference dataset_map.py:19-174)
# ---------------------------------------------------------------------------

def _synthetic_entry(image_size: int):
    def build(**kwargs):
        src = SyntheticImageSource(image_size=image_size,
                                   num_samples=kwargs.get("num_samples", 10000))
        aug = ImageAugmenter(image_size=image_size,
                             tokenizer=kwargs.get("tokenizer"))
        return src, aug
    return build


datasetMap: Dict[str, Callable[..., Tuple[DataSource, DataAugmenter]]] = {
    "synthetic-64": _synthetic_entry(64),
    "synthetic-128": _synthetic_entry(128),
    "synthetic-256": _synthetic_entry(256),
}


def register_dataset(name: str, builder: Callable[..., Tuple[DataSource, DataAugmenter]]):
    datasetMap[name] = builder


def register_image_folder(name: str, root: str, image_size: int):
    def build(**kwargs):
        return (ImageFolderSource(root),
                ImageAugmenter(image_size=image_size,
                               tokenizer=kwargs.get("tokenizer")))
    register_dataset(name, build)
