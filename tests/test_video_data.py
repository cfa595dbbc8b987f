"""CPU tests for video data sources (npy clips, frame dirs, mel audio)."""
import numpy as np
import torch

from flaxdiff_amd.data.videos import (AudioVideoAugmenter, VideoAugmenter,
                                      VideoFolderSource, collate_video_batch,
                                      count_video_frames, mel_spectrogram,
                                      read_video_clip)


def _make_npy_video(tmp_path, name, T=8, size=12):
    clip = np.random.randint(0, 255, (T, size, size, 3), dtype=np.uint8)
    np.save(tmp_path / f"{name}.npy", clip)
    (tmp_path / f"{name}.txt").write_text(f"caption for {name}")
    return clip


def test_read_npy_clip(tmp_path):
    clip = _make_npy_video(tmp_path, "a")
    got = read_video_clip(str(tmp_path / "a.npy"), num_frames=4, start=2)
    assert np.array_equal(got, clip[2:6])
    assert count_video_frames(str(tmp_path / "a.npy")) == 8


def test_frame_dir_clip(tmp_path):
    from PIL import Image
    d = tmp_path / "vid"
    d.mkdir()
    for i in range(5):
        Image.fromarray(np.full((8, 8, 3), i * 10, np.uint8)).save(
            d / f"{i:03d}.png")
    clip = read_video_clip(str(d))
    assert clip.shape == (5, 8, 8, 3)
    assert clip[3, 0, 0, 0] == 30


def test_video_source_and_augmenter(tmp_path):
    _make_npy_video(tmp_path, "a", T=10)
    _make_npy_video(tmp_path, "b", T=3)  # shorter than clip -> loop pad
    src = VideoFolderSource(str(tmp_path))
    assert len(src) == 2
    aug = VideoAugmenter(image_size=8, num_frames=6,
                         rng=np.random.default_rng(0))
    out0 = aug(src[0])
    assert out0["image"].shape == (6, 8, 8, 3)
    assert out0["caption"] == "caption for a"
    out1 = aug(src[1])  # loop-padded
    assert out1["image"].shape == (6, 8, 8, 3)


def test_audio_video_augmenter(tmp_path):
    _make_npy_video(tmp_path, "a", T=8)
    src = VideoFolderSource(str(tmp_path))
    aug = AudioVideoAugmenter(image_size=8, num_frames=4)
    out = aug(src[0])
    assert out["audio_mel"].shape[0] == 80
    batch = collate_video_batch([out, out])
    assert batch["image"].shape == (2, 4, 8, 8, 3)
    assert batch["audio_mel"].shape[0] == 2


def test_mel_spectrogram_tone():
    sr = 16000
    t = torch.arange(sr // 4, dtype=torch.float32) / sr
    tone = torch.sin(2 * np.pi * 440.0 * t)
    mel = mel_spectrogram(tone, sr)
    assert mel.shape[0] == 80
    assert torch.isfinite(mel).all()
    # energy concentrated in low-mid mel bins for 440 Hz
    assert mel.mean(dim=1).argmax() < 40
