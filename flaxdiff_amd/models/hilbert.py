"""Scan-order utilities: Hilbert / zigzag patch serialization + 2-D sin-cos PE.

Behavior contract: reference /root/reference/flaxdiff/models/hilbert.py
(hilbert_indices :87, inverse_permutation :132, patchify :162, unpatchify :184,
hilbert_patchify :213, zigzag_* :248-299, hilbert_unpatchify :302,
build_2d_sincos_pos_embed :12).

MI355X design: index tables are computed ONCE on the host with numpy and
cached; on device the reorder is a single `index_select` gather (HBM-bound,
fused by torch) — there is no per-step host work. The rectangular Hilbert
order is the standard power-of-2 square curve filtered to in-bounds cells.
"""
from __future__ import annotations

from functools import lru_cache
from typing import Tuple

import numpy as np
import torch


def build_2d_sincos_pos_embed(emb_dim: int, h_p: int, w_p: int) -> np.ndarray:
    """MAE-style fixed 2-D sin-cos PE, row-major [h_p*w_p, emb_dim].

    First half of the channels encodes the row (sin then cos), second half the
    column, each with 10000^-k frequencies.
    """
    assert emb_dim % 4 == 0, f"emb_dim must be divisible by 4, got {emb_dim}"
    quarter = emb_dim // 4
    omega = 1.0 / (10000.0 ** (np.arange(quarter, dtype=np.float32) / quarter))
    row = np.outer(np.arange(h_p, dtype=np.float32), omega)  # [H_P, q]
    col = np.outer(np.arange(w_p, dtype=np.float32), omega)  # [W_P, q]
    pe = np.zeros((h_p, w_p, emb_dim), dtype=np.float32)
    pe[..., 0 * quarter:1 * quarter] = np.sin(row)[:, None, :]
    pe[..., 1 * quarter:2 * quarter] = np.cos(row)[:, None, :]
    pe[..., 2 * quarter:3 * quarter] = np.sin(col)[None, :, :]
    pe[..., 3 * quarter:4 * quarter] = np.cos(col)[None, :, :]
    return pe.reshape(h_p * w_p, emb_dim)


def _hilbert_d2xy(n: int, d: int) -> Tuple[int, int]:
    """Hilbert index d -> (x=col, y=row) on an n x n grid (n power of 2)."""
    x = y = 0
    t = d
    s = 1
    while s < n:
        rx = 1 & (t // 2)
        ry = 1 & (t ^ rx)
        if ry == 0:  # rotate quadrant
            if rx == 1:
                x, y = s - 1 - x, s - 1 - y
            x, y = y, x
        x += s * rx
        y += s * ry
        t //= 4
        s *= 2
    return x, y


@lru_cache(maxsize=64)
def _hilbert_indices_np(h_p: int, w_p: int) -> np.ndarray:
    total = h_p * w_p
    if total == 0:
        return np.zeros(0, dtype=np.int64)
    n = 1
    while n < max(h_p, w_p):
        n <<= 1
    out = np.empty(total, dtype=np.int64)
    k = 0
    for d in range(n * n):
        x, y = _hilbert_d2xy(n, d)
        if x < w_p and y < h_p:
            out[k] = y * w_p + x
            k += 1
            if k == total:
                break
    return out


def hilbert_indices(h_p: int, w_p: int) -> torch.Tensor:
    """result[i] = row-major index of the i-th patch in Hilbert order."""
    return torch.from_numpy(_hilbert_indices_np(h_p, w_p).copy())


@lru_cache(maxsize=64)
def _zigzag_indices_np(h_p: int, w_p: int) -> np.ndarray:
    grid = np.arange(h_p * w_p, dtype=np.int64).reshape(h_p, w_p)
    grid[1::2] = grid[1::2, ::-1]  # odd rows right-to-left (ZigMa serpentine)
    return grid.reshape(-1)


def zigzag_indices(h_p: int, w_p: int) -> torch.Tensor:
    return torch.from_numpy(_zigzag_indices_np(h_p, w_p).copy())


def inverse_permutation(idx: torch.Tensor, total_size: int) -> torch.Tensor:
    """inv[k] = i where idx[i] == k (-1 for absent targets)."""
    inv = torch.full((total_size,), -1, dtype=torch.long)
    inv[idx] = torch.arange(idx.shape[0], dtype=torch.long)
    return inv


def patchify(x: torch.Tensor, patch_size: int) -> torch.Tensor:
    """NHWC image -> row-major patch tokens [B, H_P*W_P, p*p*C]."""
    B, H, W, C = x.shape
    p = patch_size
    x = x.reshape(B, H // p, p, W // p, p, C)
    x = x.permute(0, 1, 3, 2, 4, 5)  # B, H_P, W_P, p, p, C
    return x.reshape(B, (H // p) * (W // p), p * p * C)


def unpatchify(x: torch.Tensor, patch_size: int, H: int, W: int, C: int) -> torch.Tensor:
    """Row-major patch tokens [B, N, p*p*C] -> NHWC image."""
    B = x.shape[0]
    p = patch_size
    x = x.reshape(B, H // p, W // p, p, p, C)
    x = x.permute(0, 1, 3, 2, 4, 5)
    return x.reshape(B, H, W, C)


def _scan_patchify(x: torch.Tensor, patch_size: int, idx_np: np.ndarray):
    B, H, W, C = x.shape
    h_p, w_p = H // patch_size, W // patch_size
    tokens = patchify(x, patch_size)
    idx = torch.from_numpy(idx_np.copy()).to(x.device)
    inv = inverse_permutation(idx.cpu(), h_p * w_p).to(x.device)
    return tokens.index_select(1, idx), inv


def hilbert_patchify(x: torch.Tensor, patch_size: int):
    """Returns (patches in Hilbert order, inverse permutation row-major->seq)."""
    B, H, W, C = x.shape
    return _scan_patchify(x, patch_size,
                          _hilbert_indices_np(H // patch_size, W // patch_size))


def zigzag_patchify(x: torch.Tensor, patch_size: int):
    B, H, W, C = x.shape
    return _scan_patchify(x, patch_size,
                          _zigzag_indices_np(H // patch_size, W // patch_size))


def hilbert_unpatchify(x: torch.Tensor, inv_idx: torch.Tensor, patch_size: int,
                       H: int, W: int, C: int) -> torch.Tensor:
    """Scatter scan-ordered tokens back to row-major and unpatchify.

    inv_idx[k] = sequence position holding row-major patch k, so a single
    gather along dim 1 restores raster order (scan-agnostic; works for both
    Hilbert and zigzag inv tables).
    """
    row_major = x.index_select(1, inv_idx.to(x.device).clamp(min=0))
    return unpatchify(row_major, patch_size, H, W, C)


zigzag_unpatchify = hilbert_unpatchify


# ---- visualization helpers (reference hilbert.py:373+) ---------------------

def create_patch_grid(patches_np: np.ndarray, patch_size: int, channels: int,
                      grid_cols: int = 10, border: int = 1) -> np.ndarray:
    """Tile a [N, P*P*C] patch sequence into one grid image (numpy), with a
    border between cells — patches appear in sequence order, which makes the
    scan order visible."""
    n = patches_np.shape[0]
    rows = (n + grid_cols - 1) // grid_cols
    cell = patch_size + border
    grid = np.ones((rows * cell + border, grid_cols * cell + border, channels),
                   dtype=np.float32) * 0.5
    for k in range(n):
        r, c = divmod(k, grid_cols)
        patch = patches_np[k].reshape(patch_size, patch_size, channels)
        y = border + r * cell
        x = border + c * cell
        grid[y:y + patch_size, x:x + patch_size] = patch
    return grid


def visualize_hilbert_curve(h: int, w: int, patch_size: int = 1):
    """Plot the Hilbert traversal over an h x w grid (requires matplotlib)."""
    import matplotlib.pyplot as plt  # optional dep; not in the base image
    idx = _hilbert_indices_np(h // patch_size, w // patch_size)
    w_p = w // patch_size
    ys, xs = np.divmod(idx, w_p)
    fig, ax = plt.subplots(figsize=(6, 6))
    ax.plot(xs + 0.5, ys + 0.5, "-o", markersize=3)
    ax.set_xlim(0, w_p)
    ax.set_ylim(w_p, 0)
    ax.set_title(f"Hilbert curve {h//patch_size}x{w//patch_size}")
    ax.set_aspect("equal")
    return fig


def demo_hilbert_patching(image: np.ndarray, patch_size: int = 8):
    """Show an image next to its Hilbert-ordered patch grid (matplotlib)."""
    import matplotlib.pyplot as plt  # optional dep
    x = torch.from_numpy(np.asarray(image, dtype=np.float32))[None]
    patches, inv = hilbert_patchify(x, patch_size)
    grid = create_patch_grid(patches[0].numpy(), patch_size, x.shape[-1])
    fig, axes = plt.subplots(1, 2, figsize=(12, 6))
    axes[0].imshow(np.asarray(image))
    axes[0].set_title("input")
    axes[1].imshow(grid.squeeze())
    axes[1].set_title("patches in Hilbert order")
    for a in axes:
        a.axis("off")
    return fig
