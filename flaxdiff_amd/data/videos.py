"""Video / audio-video data sources.

Behavior contract: reference /root/reference/flaxdiff/data/sources/videos.py
(local dir walk + cache :79-154, AudioVideoAugmenter random clip + audio
tokens :156-223), av_utils.py (decord/pyav/opencv readers, random-clip
samplers), audio_utils.py (mel spectrograms), voxceleb2.py (AV dataset for
audio-guided video diffusion).

This environment has no decord/pyav/opencv; readers are structured with
capability probing: decord/av are used when importable, otherwise videos are
read from frame directories (PIL) or .npy clip files. Audio mel spectrograms
are computed with torch.stft (no librosa dependency).
"""
from __future__ import annotations

import math
from pathlib import Path
from typing import Any, Dict, List, Optional

import numpy as np
import torch

from .sources import DataAugmenter, DataSource, _resize_uint8


# ---------------------------------------------------------------------------
# clip readers (av_utils equivalents, capability-probed)
# ---------------------------------------------------------------------------

def _have(mod: str) -> bool:
    try:
        __import__(mod)
        return True
    except Exception:  # noqa: BLE001
        return False


def read_video_clip(path: str, num_frames: Optional[int] = None,
                    start: int = 0) -> np.ndarray:
    """Returns [T, H, W, 3] uint8. Prefers decord, then PyAV, then a frame
    directory of images, then a .npy clip (reference av_utils.py readers)."""
    p = Path(path)
    if p.is_dir():
        from PIL import Image
        files = sorted(f for f in p.iterdir()
                       if f.suffix.lower() in (".png", ".jpg", ".jpeg"))
        files = files[start:(start + num_frames) if num_frames else None]
        return np.stack([np.asarray(Image.open(f).convert("RGB"))
                         for f in files])
    if p.suffix == ".npy":
        arr = np.load(p)
        end = start + num_frames if num_frames else None
        return np.asarray(arr[start:end], dtype=np.uint8)
    if _have("decord"):
        import decord
        vr = decord.VideoReader(str(p))
        n = len(vr)
        idx = list(range(start, min(start + (num_frames or n), n)))
        return vr.get_batch(idx).asnumpy()
    if _have("av"):
        import av
        frames = []
        with av.open(str(p)) as container:
            for i, frame in enumerate(container.decode(video=0)):
                if i < start:
                    continue
                frames.append(frame.to_ndarray(format="rgb24"))
                if num_frames and len(frames) >= num_frames:
                    break
        return np.stack(frames)
    raise RuntimeError(f"no video reader available for {path} "
                       "(install decord/av, or use frame dirs / .npy clips)")


def count_video_frames(path: str) -> int:
    p = Path(path)
    if p.is_dir():
        return len([f for f in p.iterdir()
                    if f.suffix.lower() in (".png", ".jpg", ".jpeg")])
    if p.suffix == ".npy":
        return int(np.load(p, mmap_mode="r").shape[0])
    if _have("decord"):
        import decord
        return len(decord.VideoReader(str(p)))
    clip = read_video_clip(path)
    return clip.shape[0]


# ---------------------------------------------------------------------------
# audio utils (audio_utils.py equivalent; torch.stft, no librosa)
# ---------------------------------------------------------------------------

def mel_filterbank(n_mels: int, n_fft: int, sample_rate: int,
                   fmin: float = 0.0, fmax: Optional[float] = None
                   ) -> torch.Tensor:
    """Slaney-style triangular mel filterbank [n_mels, n_fft//2+1]."""
    fmax = fmax or sample_rate / 2

    def hz_to_mel(f):
        return 2595.0 * math.log10(1.0 + f / 700.0)

    def mel_to_hz(m):
        return 700.0 * (10.0 ** (m / 2595.0) - 1.0)

    mels = torch.linspace(hz_to_mel(fmin), hz_to_mel(fmax), n_mels + 2)
    freqs = torch.tensor([mel_to_hz(float(m)) for m in mels])
    fft_freqs = torch.linspace(0, sample_rate / 2, n_fft // 2 + 1)
    fb = torch.zeros(n_mels, n_fft // 2 + 1)
    for i in range(n_mels):
        lo, ctr, hi = freqs[i], freqs[i + 1], freqs[i + 2]
        up = (fft_freqs - lo) / (ctr - lo + 1e-9)
        down = (hi - fft_freqs) / (hi - ctr + 1e-9)
        fb[i] = torch.clamp(torch.minimum(up, down), min=0)
    return fb


def mel_spectrogram(audio: torch.Tensor, sample_rate: int = 16000,
                    n_fft: int = 400, hop_length: int = 160,
                    n_mels: int = 80) -> torch.Tensor:
    """[T_samples] -> log-mel [n_mels, T_frames]."""
    audio = torch.as_tensor(audio, dtype=torch.float32)
    spec = torch.stft(audio, n_fft=n_fft, hop_length=hop_length,
                      window=torch.hann_window(n_fft), return_complex=True)
    power = spec.abs() ** 2
    mel = mel_filterbank(n_mels, n_fft, sample_rate) @ power
    return torch.log(mel + 1e-6)


# ---------------------------------------------------------------------------
# sources / augmenters
# ---------------------------------------------------------------------------

class VideoFolderSource(DataSource):
    """Directory of videos (files, frame-dirs or .npy clips) with optional
    caption sidecars (reference sources/videos.py:79-154)."""

    VIDEO_EXTS = (".mp4", ".avi", ".mkv", ".webm", ".npy")

    def __init__(self, root: str):
        self.root = Path(root)
        self.items: List[Path] = sorted(
            [p for p in self.root.rglob("*")
             if p.suffix.lower() in self.VIDEO_EXTS] +
            [p for p in self.root.iterdir()
             if p.is_dir() and any(f.suffix.lower() in (".png", ".jpg")
                                   for f in p.iterdir())])
        if not self.items:
            raise FileNotFoundError(f"no videos under {root}")

    def __len__(self):
        return len(self.items)

    def __getitem__(self, idx):
        p = self.items[idx]
        cap = p.with_suffix(".txt")
        caption = cap.read_text().strip() if cap.exists() else p.stem
        return {"video_path": str(p), "caption": caption}


class VideoAugmenter(DataAugmenter):
    """Random clip of `num_frames`, resized (reference videos.py:156-223)."""

    def __init__(self, image_size: int, num_frames: int = 16,
                 rng: Optional[np.random.Generator] = None):
        self.image_size = image_size
        self.num_frames = num_frames
        self.rng = rng or np.random.default_rng()

    def __call__(self, sample: Dict[str, Any]) -> Dict[str, Any]:
        path = sample["video_path"]
        total = count_video_frames(path)
        start = 0
        if total > self.num_frames:
            start = int(self.rng.integers(0, total - self.num_frames + 1))
        clip = read_video_clip(path, self.num_frames, start)
        if clip.shape[0] < self.num_frames:  # loop-pad short clips
            reps = -(-self.num_frames // clip.shape[0])
            clip = np.concatenate([clip] * reps)[: self.num_frames]
        frames = np.stack([_resize_uint8(f, self.image_size) for f in clip])
        return {"image": frames, "caption": sample.get("caption", "")}


class AudioVideoAugmenter(VideoAugmenter):
    """Adds a log-mel audio conditioning track aligned with the clip
    (reference videos.py:156-223 + voxceleb2.py)."""

    def __init__(self, image_size: int, num_frames: int = 16, fps: int = 25,
                 sample_rate: int = 16000, n_mels: int = 80, **kw):
        super().__init__(image_size, num_frames, **kw)
        self.fps = fps
        self.sample_rate = sample_rate
        self.n_mels = n_mels

    def __call__(self, sample: Dict[str, Any]) -> Dict[str, Any]:
        out = super().__call__(sample)
        audio = sample.get("audio")
        if audio is None:
            n = int(self.num_frames / self.fps * self.sample_rate)
            audio = np.zeros(n, dtype=np.float32)
        out["audio_mel"] = mel_spectrogram(torch.as_tensor(audio),
                                           self.sample_rate,
                                           n_mels=self.n_mels)
        return out


def collate_video_batch(samples: List[Dict[str, Any]]) -> Dict[str, Any]:
    batch = {"image": torch.stack(
        [torch.from_numpy(np.ascontiguousarray(s["image"])) for s in samples])}
    if "audio_mel" in samples[0]:
        batch["audio_mel"] = torch.stack([s["audio_mel"] for s in samples])
    return batch


# ---------------------------------------------------------------------------
# VoxCeleb2-style talking-face AV dataset (reference sources/voxceleb2.py:24-412)
# ---------------------------------------------------------------------------

def lower_half_mask(num_frames: int, height: int, width: int,
                    pad: int = 0) -> np.ndarray:
    """[F, H, W, 1] float mask: 1 on the (to-be-inpainted) lower face half,
    shrunk by `pad` rows (reference voxceleb2.py:177-203 get_simple_mask)."""
    m = np.zeros((num_frames, height, width, 1), dtype=np.float32)
    m[:, max(height // 2 - pad, 0):, :, :] = 1.0
    return m


class VoxCeleb2LikeSource(DataSource):
    """Talking-face AV clips in the voxceleb2 layout (id/.../clip.*), each
    clip a video (any read_video_clip format) with optional audio:
      * `<clip>.wav.npy` — raw waveform, or
      * `.npz` clips with `frames` [T,H,W,3] uint8 and `audio` [S] float.

    Per sample (reference voxceleb2.py:281-383):
      instance_images [F,H,W,3]  — the training window
      reference_images [F,H,W,3] — a DISJOINT window of the same identity
      mask / instance_masks [F,H,W,1] — lower-half inpainting mask
      instance_masked_images     — instance * (mask < 0.5)
      mels [n_mels, T_a]         — log-mel of the window's audio (if present)
      identity                   — speaker id (top-level dir name)
    """

    def __init__(self, root: str, num_frames: int = 8, resolution: int = 64,
                 sample_rate: int = 16000, fps: float = 25.0,
                 n_mels: int = 80, audio: bool = True):
        self.root = Path(root)
        self.num_frames = num_frames
        self.resolution = resolution
        self.sample_rate = sample_rate
        self.fps = fps
        self.n_mels = n_mels
        self.audio = audio
        exts = VideoFolderSource.VIDEO_EXTS + (".npz",)
        self.items = sorted(p for p in self.root.rglob("*")
                            if p.suffix.lower() in exts)
        if not self.items:
            raise FileNotFoundError(f"no clips under {root}")

    def __len__(self):
        return len(self.items)

    def _load_clip(self, p: Path):
        if p.suffix == ".npz":
            z = np.load(p)
            return np.asarray(z["frames"], dtype=np.uint8), \
                (np.asarray(z["audio"], dtype=np.float32)
                 if "audio" in z.files else None)
        frames = read_video_clip(str(p))
        wav = None
        side = p.with_suffix(p.suffix + ".wav.npy")
        alt = p.with_name(p.stem + ".wav.npy")
        for cand in (side, alt):
            if cand.exists():
                wav = np.load(cand).astype(np.float32)
                break
        return frames, wav

    def __getitem__(self, idx: int) -> Dict[str, Any]:
        rng = np.random.default_rng(idx)
        for attempt in range(10):
            p = self.items[(idx + attempt) % len(self.items)]
            frames, wav = self._load_clip(p)
            F_ = self.num_frames
            if frames.shape[0] >= 3 * F_:
                break
        else:
            raise RuntimeError("no clip with >= 3x num_frames frames")

        total = frames.shape[0]
        start = int(rng.integers(F_ // 2, total - F_ - F_ // 2 + 1))
        inst = frames[start:start + F_]
        # disjoint reference window of the SAME identity (voxceleb2.py:205-242)
        ref_lo = 0 if start >= F_ else start + F_
        ref_hi = start - F_ if start >= F_ else total - F_
        ref_start = int(rng.integers(ref_lo, max(ref_hi, ref_lo) + 1))
        ref = frames[ref_start:ref_start + F_]

        def _resize(clip):
            t = torch.from_numpy(np.ascontiguousarray(clip)).permute(0, 3, 1, 2).float()
            t = torch.nn.functional.interpolate(t, size=(self.resolution,
                                                         self.resolution),
                                                mode="bilinear",
                                                align_corners=False)
            return t.permute(0, 2, 3, 1).round().clamp(0, 255).byte().numpy()

        inst = _resize(inst)
        ref = _resize(ref)
        mask = lower_half_mask(F_, self.resolution, self.resolution)
        out: Dict[str, Any] = {
            "instance_images": inst,
            "reference_images": ref,
            "mask": mask,
            "instance_masks": mask,
            "instance_masked_images": (inst.astype(np.float32)
                                       * (mask < 0.5)).astype(np.uint8),
            "identity": p.relative_to(self.root).parts[0]
            if len(p.relative_to(self.root).parts) > 1 else p.stem,
            "video": inst,      # generic video key for the existing augmenters
            "caption": "",
        }
        if self.audio and wav is not None:
            spf = self.sample_rate / self.fps      # samples per frame
            a0 = int(start * spf)
            a1 = int((start + F_) * spf)
            window = wav[a0:min(a1, len(wav))]
            if len(window) > 0:
                out["raw_audio"] = window
                out["mels"] = mel_spectrogram(
                    torch.from_numpy(window), sample_rate=self.sample_rate,
                    n_mels=self.n_mels).numpy()
        return out
