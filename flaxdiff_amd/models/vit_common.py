"""ViT/DiT shared pieces: patch embedding, RoPE, AdaLN-Zero.

Behavior contract: reference /root/reference/flaxdiff/models/vit_common.py
(PatchEmbedding :20, unpatchify :10, PositionalEncoding :40, RotaryEmbedding
:86, apply_rotary_embedding :62, RoPEAttention :123, AdaLNZero :189,
AdaLNParams :240).

MI355X notes: patch embedding is a strided conv == one MFMA GEMM over
[p*p*C -> D] after a host-free reshape; RoPE cos/sin tables are registered
buffers (precomputed once, live in HBM) and the rotation is fused by torch
into two FMAs per element; attention runs the hand-written flash kernel via
ops.attention.
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.nn as nn

from .. import ops
from .common import Dense


def norm_fp32(norm: nn.LayerNorm, x: torch.Tensor) -> torch.Tensor:
    """Run a LayerNorm in fp32 whatever the module/activation dtype (the
    AdaLN pre-norms are fp32 for stability; after model.bfloat16() the params
    are bf16, and torch rejects mixed-dtype layer_norm on CPU)."""
    w = norm.weight.float() if norm.weight is not None else None
    b = norm.bias.float() if norm.bias is not None else None
    return nn.functional.layer_norm(
        x.float(), norm.normalized_shape, w, b, norm.eps).to(x.dtype)


def unpatchify(x: torch.Tensor, channels: int = 3) -> torch.Tensor:
    """[B, N, p*p*C] (square raster grid) -> NHWC image."""
    patch_size = int((x.shape[2] // channels) ** 0.5)
    h = w = int(x.shape[1] ** 0.5)
    assert h * w == x.shape[1] and patch_size ** 2 * channels == x.shape[2], \
        f"Invalid shape: {tuple(x.shape)}"
    B = x.shape[0]
    x = x.reshape(B, h, w, patch_size, patch_size, channels)
    x = x.permute(0, 1, 3, 2, 4, 5)
    return x.reshape(B, h * patch_size, w * patch_size, channels)


class PatchEmbedding(nn.Module):
    """Non-overlapping patchify + linear embed (one GEMM).

    The reference uses a strided conv (vit_common.py:31-36); with stride ==
    kernel == patch_size that is exactly a reshape + [p*p*C, D] GEMM, which is
    how it executes here (MFMA library GEMM, no im2col).
    """

    def __init__(self, patch_size: int, embedding_dim: int, in_channels: int = 3):
        super().__init__()
        self.patch_size = patch_size
        self.proj = Dense(patch_size * patch_size * in_channels, embedding_dim)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, H, W, C = x.shape
        p = self.patch_size
        assert H % p == 0 and W % p == 0, "image dims must be divisible by patch size"
        x = x.reshape(B, H // p, p, W // p, p, C).permute(0, 1, 3, 2, 4, 5)
        x = x.reshape(B, (H // p) * (W // p), p * p * C)
        return self.proj(x)


class PositionalEncoding(nn.Module):
    """Learned additive PE, zero-init (vit_common.py:40-49)."""

    def __init__(self, max_len: int, embedding_dim: int):
        super().__init__()
        self.pos_encoding = nn.Parameter(torch.zeros(1, max_len, embedding_dim))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return x + self.pos_encoding[:, :x.shape[1], :].to(x.dtype)


def _rotate_half(x: torch.Tensor) -> torch.Tensor:
    x1, x2 = x.chunk(2, dim=-1)
    return torch.cat((-x2, x1), dim=-1)


def apply_rotary_embedding(x: torch.Tensor, freqs_cos: torch.Tensor,
                           freqs_sin: torch.Tensor) -> torch.Tensor:
    """x: [..., S, D]; freqs: [S, D/2]. x*cos + rotate_half(x)*sin."""
    shape = (1,) * (x.dim() - 2) + freqs_cos.shape
    cos = torch.cat([freqs_cos, freqs_cos], dim=-1).reshape(*shape[:-1], -1)
    sin = torch.cat([freqs_sin, freqs_sin], dim=-1).reshape(*shape[:-1], -1)
    return (x * cos.to(x.dtype) + _rotate_half(x) * sin.to(x.dtype))


class RotaryEmbedding(nn.Module):
    """Precomputed cos/sin tables [max_seq_len, dim/2] (vit_common.py:86-117)."""

    def __init__(self, dim: int, max_seq_len: int = 4096, base: int = 10000):
        super().__init__()
        self.dim = dim
        self.max_seq_len = max_seq_len
        self.base = base
        cos, sin = self._tables(max_seq_len)
        self.register_buffer("freqs_cos", cos, persistent=False)
        self.register_buffer("freqs_sin", sin, persistent=False)

    def _tables(self, seq_len: int):
        inv_freq = 1.0 / (self.base ** (torch.arange(0, self.dim, 2,
                                                     dtype=torch.float32) / self.dim))
        t = torch.arange(seq_len, dtype=torch.float32)
        freqs = torch.outer(t, inv_freq)
        return torch.cos(freqs), torch.sin(freqs)

    def forward(self, seq_len: int) -> Tuple[torch.Tensor, torch.Tensor]:
        if seq_len > self.max_seq_len:  # dynamic extension, uncached
            cos, sin = self._tables(seq_len)
            return cos.to(self.freqs_cos.device), sin.to(self.freqs_sin.device)
        return self.freqs_cos[:seq_len], self.freqs_sin[:seq_len]


class RoPEAttention(nn.Module):
    """QKV + RoPE on q/k + flash attention + out proj (vit_common.py:123-186)."""

    def __init__(self, query_dim: int, heads: int, dim_head: int,
                 use_bias: bool = True, context_dim: Optional[int] = None,
                 rope_emb: Optional[RotaryEmbedding] = None):
        super().__init__()
        self.heads = heads
        self.dim_head = dim_head
        inner = heads * dim_head
        cdim = context_dim if context_dim is not None else query_dim
        self.to_q = Dense(query_dim, inner, use_bias=use_bias)
        self.to_k = Dense(cdim, inner, use_bias=use_bias)
        self.to_v = Dense(cdim, inner, use_bias=use_bias)
        self.to_out = Dense(inner, query_dim, use_bias=use_bias)
        self.rope_emb = rope_emb  # shared module, not owned

    def forward(self, x: torch.Tensor, context: Optional[torch.Tensor] = None,
                freqs_cis: Optional[Tuple[torch.Tensor, torch.Tensor]] = None):
        orig_shape = x.shape
        if x.dim() == 4:
            B, H, W, C = x.shape
            x = x.reshape(B, H * W, C)
        B, S, _ = x.shape
        ctx = x if context is None else context
        if ctx.dim() == 4:
            ctx = ctx.reshape(ctx.shape[0], -1, ctx.shape[-1])
        ctx = ctx.to(x.dtype)
        Sk = ctx.shape[1]

        q = self.to_q(x).reshape(B, S, self.heads, self.dim_head).permute(0, 2, 1, 3)
        k = self.to_k(ctx).reshape(B, Sk, self.heads, self.dim_head).permute(0, 2, 1, 3)
        v = self.to_v(ctx).reshape(B, Sk, self.heads, self.dim_head).permute(0, 2, 1, 3)

        if freqs_cis is None:
            if self.rope_emb is None:
                raise ValueError("RoPE frequencies not provided")
            freqs_cis = self.rope_emb(S)
        cos, sin = freqs_cis
        dev_cos, dev_sin = cos.to(q.device), sin.to(q.device)
        q = apply_rotary_embedding(q, dev_cos, dev_sin)
        k = apply_rotary_embedding(k, dev_cos[:Sk], dev_sin[:Sk])

        o = ops.attention(q, k, v)
        o = o.permute(0, 2, 1, 3).reshape(B, S, self.heads * self.dim_head)
        return self.to_out(o).reshape(orig_shape)


class AdaLNParams(nn.Module):
    """Zero-init projection conditioning -> 6*features modulation params
    (vit_common.py:240-261)."""

    def __init__(self, cond_features: int, features: int):
        super().__init__()
        self.ada_proj = Dense(cond_features, 6 * features, zero_init=True)

    def forward(self, conditioning: torch.Tensor) -> torch.Tensor:
        if conditioning.dim() == 2:
            conditioning = conditioning.unsqueeze(1)  # [B, 1, D_cond]
        return self.ada_proj(conditioning)  # [B, 1, 6F]


class AdaLNZero(nn.Module):
    """LayerNorm (no affine) + 6-way modulate; clips the MLP scale/shift to
    [-10, 10] like the reference (vit_common.py:189-237)."""

    def __init__(self, cond_features: int, features: int, norm_epsilon: float = 1e-5):
        super().__init__()
        self.features = features
        self.eps = norm_epsilon
        self.ada_proj = Dense(cond_features, 6 * features, zero_init=True)

    def forward(self, x: torch.Tensor, conditioning: torch.Tensor):
        if x.dim() == 3 and conditioning.dim() == 2:
            conditioning = conditioning.unsqueeze(1)
        params = self.ada_proj(conditioning)
        scale_mlp, shift_mlp, gate_mlp, scale_attn, shift_attn, gate_attn = \
            params.chunk(6, dim=-1)
        scale_mlp = scale_mlp.clamp(-10.0, 10.0)
        shift_mlp = shift_mlp.clamp(-10.0, 10.0)
        norm_x = torch.nn.functional.layer_norm(x.float(), (self.features,),
                                                eps=self.eps).to(x.dtype)
        x_attn = norm_x * (1 + scale_attn) + shift_attn
        x_mlp = norm_x * (1 + scale_mlp) + shift_mlp
        return x_attn, gate_attn, x_mlp, gate_mlp


def layer_norm_noaffine(x: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    """fp32 LayerNorm without scale/bias (the DiT pre-norm)."""
    return torch.nn.functional.layer_norm(x.float(), (x.shape[-1],), eps=eps).to(x.dtype)
