"""MM-DiT family: SimpleMMDiT and HierarchicalMMDiT (PixArt-style pyramid).

Behavior contract: reference /root/reference/flaxdiff/models/simple_mmdit.py
(MMAdaLNZero :17-90, MMDiTBlock :94-160, SimpleMMDiT :162-333,
PatchMerging :336, PatchExpanding :384, HierarchicalMMDiT :433-729).

Key semantics preserved:
  * MMAdaLNZero projects time and text embeddings with SEPARATE zero-init
    Denses and sums the 6F modulation params; MLP scale/shift clipped to
    [-10,10]; text mean-pooled when its sequence differs from x's.
  * MMDiTBlock = MMAdaLNZero -> RoPE self-attention -> gate; MLP -> gate.
  * HierarchicalMMDiT: 3-stage U pyramid with per-stage embeddings/heads/RoPE,
    2x2 PatchMerging (LN -> Dense), PatchExpanding (Dense -> LN), encoder
    skips fused by LN+Dense on the channel concat.

MI355X notes: pure token-GEMM models — MFMA library GEMMs + the flash
attention kernel; merging/expanding are reshapes + one GEMM each.
"""
from __future__ import annotations

from typing import Optional, Sequence

import torch
import torch.nn as nn
import torch.nn.functional as F

from .common import Dense, FourierEmbedding, TimeProjection
from .hilbert import (hilbert_indices, hilbert_patchify, hilbert_unpatchify,
                      inverse_permutation)
from .simple_dit import DiTBlock  # noqa: F401  (re-export convenience)
from .vit_common import (PatchEmbedding, RoPEAttention, RotaryEmbedding, norm_fp32,
                         layer_norm_noaffine, unpatchify)


class MMAdaLNZero(nn.Module):
    """Separate time/text zero-init AdaLN projections, summed (ref :17-90)."""

    def __init__(self, features: int, t_features: Optional[int] = None,
                 text_features: Optional[int] = None, norm_epsilon: float = 1e-5,
                 use_mean_pooling: bool = True):
        super().__init__()
        self.features = features
        self.eps = norm_epsilon
        self.use_mean_pooling = use_mean_pooling
        self.ada_t_proj = Dense(t_features or features, 6 * features, zero_init=True)
        self.ada_text_proj = Dense(text_features or features, 6 * features,
                                   zero_init=True)

    def forward(self, x: torch.Tensor, t_emb: torch.Tensor, text_emb: torch.Tensor):
        norm_x = layer_norm_noaffine(x, self.eps)

        if t_emb.dim() == 2:
            t_emb = t_emb.unsqueeze(1)
        if text_emb.dim() == 2:
            text_emb = text_emb.unsqueeze(1)
        elif (text_emb.dim() == 3 and self.use_mean_pooling
              and text_emb.shape[1] != x.shape[1]):
            text_emb = text_emb.mean(dim=1, keepdim=True)

        t_params = self.ada_t_proj(t_emb)
        text_params = self.ada_text_proj(text_emb)
        if t_params.shape[1] != text_params.shape[1]:
            text_params = text_params.mean(dim=1, keepdim=True)
        ada = t_params + text_params

        scale_mlp, shift_mlp, gate_mlp, scale_attn, shift_attn, gate_attn = \
            ada.chunk(6, dim=-1)
        scale_mlp = scale_mlp.clamp(-10.0, 10.0)
        shift_mlp = shift_mlp.clamp(-10.0, 10.0)
        x_attn = norm_x * (1 + scale_attn) + shift_attn
        x_mlp = norm_x * (1 + scale_mlp) + shift_mlp
        return x_attn, gate_attn, x_mlp, gate_mlp


class MMDiTBlock(nn.Module):
    def __init__(self, features: int, num_heads: int, rope_emb: RotaryEmbedding,
                 mlp_ratio: int = 4, norm_epsilon: float = 1e-5):
        super().__init__()
        self.ada_ln_zero = MMAdaLNZero(features, norm_epsilon=norm_epsilon)
        self.attention = RoPEAttention(features, num_heads, features // num_heads,
                                       use_bias=True, rope_emb=rope_emb)
        hidden = features * mlp_ratio
        self.mlp_in = Dense(features, hidden)
        self.mlp_out = Dense(hidden, features)

    def forward(self, x, t_emb, text_emb, freqs_cis):
        x_attn, gate_attn, x_mlp, gate_mlp = self.ada_ln_zero(x, t_emb, text_emb)
        x = x + gate_attn * self.attention(x_attn, context=None, freqs_cis=freqs_cis)
        x = x + gate_mlp * self.mlp_out(F.gelu(self.mlp_in(x_mlp)))
        return x


class SimpleMMDiT(nn.Module):
    def __init__(self,
                 output_channels: int = 3,
                 in_channels: int = 3,
                 patch_size: int = 16,
                 emb_features: int = 768,
                 num_layers: int = 12,
                 num_heads: int = 12,
                 mlp_ratio: int = 4,
                 context_dim: int = 768,
                 norm_epsilon: float = 1e-5,
                 learn_sigma: bool = False,
                 use_hilbert: bool = False):
        super().__init__()
        self.output_channels = output_channels
        self.patch_size = patch_size
        self.learn_sigma = learn_sigma
        self.use_hilbert = use_hilbert

        if use_hilbert:
            self.hilbert_proj = Dense(patch_size * patch_size * in_channels,
                                      emb_features)
        else:
            self.patch_embed = PatchEmbedding(patch_size, emb_features, in_channels)

        self.time_fourier = FourierEmbedding(features=emb_features)
        self.time_proj = TimeProjection(emb_features, emb_features * mlp_ratio)
        self.time_out = Dense(emb_features * mlp_ratio, emb_features)
        self.text_proj = Dense(context_dim, emb_features)

        self.rope = RotaryEmbedding(dim=emb_features // num_heads, max_seq_len=4096)
        self.blocks = nn.ModuleList([
            MMDiTBlock(emb_features, num_heads, self.rope, mlp_ratio, norm_epsilon)
            for _ in range(num_layers)])

        self.final_norm = nn.LayerNorm(emb_features, eps=norm_epsilon)
        out_dim = patch_size * patch_size * output_channels
        if learn_sigma:
            out_dim *= 2
        self.final_proj = Dense(emb_features, out_dim, zero_init=True)

    def forward(self, x: torch.Tensor, temb: torch.Tensor,
                textcontext: torch.Tensor) -> torch.Tensor:
        assert textcontext is not None, "SimpleMMDiT requires textcontext"
        B, H, W, C = x.shape
        p = self.patch_size

        inv_idx = None
        if self.use_hilbert:
            patches_raw, inv_idx = hilbert_patchify(x, p)
            seq = self.hilbert_proj(patches_raw)
        else:
            seq = self.patch_embed(x)

        t_emb = self.time_out(self.time_proj(self.time_fourier(temb).to(seq.dtype)))
        text_emb = self.text_proj(textcontext.to(seq.dtype))

        freqs_cis = self.rope(seq.shape[1])
        freqs_cis = (freqs_cis[0].to(seq.device), freqs_cis[1].to(seq.device))
        for block in self.blocks:
            seq = block(seq, t_emb, text_emb, freqs_cis)

        out = norm_fp32(self.final_norm, seq)
        out = self.final_proj(out)
        if self.learn_sigma:
            out, _ = out.chunk(2, dim=-1)
        if self.use_hilbert:
            return hilbert_unpatchify(out, inv_idx, p, H, W, self.output_channels)
        return unpatchify(out, channels=self.output_channels)


class PatchMerging(nn.Module):
    """2x2 merge + LN + Dense (ref :336-383)."""

    def __init__(self, in_features: int, out_features: int, merge_size: int = 2,
                 norm_epsilon: float = 1e-5):
        super().__init__()
        self.merge_size = merge_size
        merged_dim = merge_size * merge_size * in_features
        self.norm = nn.LayerNorm(merged_dim, eps=norm_epsilon)
        self.projection = Dense(merged_dim, out_features)

    def forward(self, x: torch.Tensor, h_p: int, w_p: int):
        B, L, C = x.shape
        m = self.merge_size
        assert L == h_p * w_p and h_p % m == 0 and w_p % m == 0
        x = x.reshape(B, h_p // m, m, w_p // m, m, C)
        x = x.permute(0, 1, 3, 2, 4, 5).reshape(B, (h_p // m) * (w_p // m),
                                                m * m * C)
        x = norm_fp32(self.norm, x)
        x = self.projection(x)
        return x, h_p // m, w_p // m


class PatchExpanding(nn.Module):
    """Dense -> LN -> 2x2 expand (ref :384-430)."""

    def __init__(self, in_features: int, out_features: int, expand_size: int = 2,
                 norm_epsilon: float = 1e-5):
        super().__init__()
        self.expand_size = expand_size
        self.out_features = out_features
        expanded = expand_size * expand_size * out_features
        self.projection = Dense(in_features, expanded)
        self.norm = nn.LayerNorm(expanded, eps=norm_epsilon)

    def forward(self, x: torch.Tensor, h_p: int, w_p: int):
        B, L, C = x.shape
        e = self.expand_size
        x = self.projection(x)
        x = norm_fp32(self.norm, x)
        x = x.reshape(B, h_p, w_p, e, e, self.out_features)
        x = x.permute(0, 1, 3, 2, 4, 5).reshape(B, (h_p * e) * (w_p * e),
                                                self.out_features)
        return x, h_p * e, w_p * e


class HierarchicalMMDiT(nn.Module):
    """PixArt-alpha-style 3-stage pyramid MM-DiT (ref :433-729)."""

    def __init__(self,
                 output_channels: int = 3,
                 in_channels: int = 3,
                 base_patch_size: int = 8,
                 emb_features: Sequence[int] = (512, 768, 1024),
                 num_layers: Sequence[int] = (4, 4, 14),
                 num_heads: Sequence[int] = (8, 12, 16),
                 mlp_ratio: int = 4,
                 context_dim: int = 768,
                 norm_epsilon: float = 1e-5,
                 learn_sigma: bool = False,
                 use_hilbert: bool = False):
        super().__init__()
        assert len(emb_features) == len(num_layers) == len(num_heads)
        n = len(emb_features)
        self.n_stages = n
        self.output_channels = output_channels
        self.base_patch_size = base_patch_size
        self.learn_sigma = learn_sigma
        self.use_hilbert = use_hilbert

        self.patch_embed = PatchEmbedding(base_patch_size, emb_features[0],
                                          in_channels)
        if use_hilbert:
            self.hilbert_proj = Dense(
                base_patch_size * base_patch_size * in_channels, emb_features[0])

        base_dim = emb_features[-1]
        self.time_fourier = FourierEmbedding(features=base_dim)
        self.time_proj = TimeProjection(base_dim, base_dim * mlp_ratio)
        self.time_out = Dense(base_dim * mlp_ratio, base_dim)
        self.text_proj_base = Dense(context_dim, base_dim)
        self.t_emb_projs = nn.ModuleList(
            [Dense(base_dim, emb_features[i]) for i in range(n)])
        self.text_emb_projs = nn.ModuleList(
            [Dense(base_dim, emb_features[i]) for i in range(n)])

        self.ropes = nn.ModuleList([
            RotaryEmbedding(dim=emb_features[i] // num_heads[i], max_seq_len=4096)
            for i in range(n)])

        self.encoder_blocks = nn.ModuleList([
            nn.ModuleList([MMDiTBlock(emb_features[s], num_heads[s],
                                      self.ropes[s], mlp_ratio, norm_epsilon)
                           for _ in range(num_layers[s])])
            for s in range(n)])
        self.patch_mergers = nn.ModuleList([
            PatchMerging(emb_features[s], emb_features[s + 1],
                         norm_epsilon=norm_epsilon)
            for s in range(n - 1)])

        # decoder lists ordered for stages n-2 .. 0
        self.patch_expanders = nn.ModuleList([
            PatchExpanding(emb_features[s + 1], emb_features[s],
                           norm_epsilon=norm_epsilon)
            for s in range(n - 2, -1, -1)])
        self.fusion_norms = nn.ModuleList([
            nn.LayerNorm(2 * emb_features[s], eps=norm_epsilon)
            for s in range(n - 2, -1, -1)])
        self.fusion_denses = nn.ModuleList([
            Dense(2 * emb_features[s], emb_features[s])
            for s in range(n - 2, -1, -1)])
        self.decoder_blocks = nn.ModuleList([
            nn.ModuleList([MMDiTBlock(emb_features[s], num_heads[s],
                                      self.ropes[s], mlp_ratio, norm_epsilon)
                           for _ in range(num_layers[s])])
            for s in range(n - 2, -1, -1)])

        self.final_norm = nn.LayerNorm(emb_features[0], eps=norm_epsilon)
        out_dim = base_patch_size * base_patch_size * output_channels
        if learn_sigma:
            out_dim *= 2
        self.final_proj = Dense(emb_features[0], out_dim, zero_init=True)

    def forward(self, x: torch.Tensor, temb: torch.Tensor,
                textcontext: torch.Tensor) -> torch.Tensor:
        assert textcontext is not None, "HierarchicalMMDiT requires textcontext"
        B, H, W, C = x.shape
        n = self.n_stages
        p = self.base_patch_size
        assert H % (p * 2 ** (n - 1)) == 0 and W % (p * 2 ** (n - 1)) == 0

        h_p, w_p = H // p, W // p
        inv_idx = None
        if self.use_hilbert:
            fine_idx = hilbert_indices(h_p, w_p)
            inv_idx = inverse_permutation(fine_idx, h_p * w_p)
            patches_raw, _ = hilbert_patchify(x, p)
            seq = self.hilbert_proj(patches_raw)
        else:
            seq = self.patch_embed(x)

        t_base = self.time_out(self.time_proj(self.time_fourier(temb).to(seq.dtype)))
        text_base = self.text_proj_base(textcontext.to(seq.dtype))
        t_embs = [proj(t_base) for proj in self.t_emb_projs]
        text_embs = [proj(text_base) for proj in self.text_emb_projs]

        def rope_for(stage, seq_len, device):
            cos, sin = self.ropes[stage](seq_len)
            return cos.to(device), sin.to(device)

        skips = {}
        ch, cw = h_p, w_p
        for stage in range(n):
            freqs = rope_for(stage, seq.shape[1], seq.device)
            for block in self.encoder_blocks[stage]:
                seq = block(seq, t_embs[stage], text_embs[stage], freqs)
            skips[stage] = seq
            if stage < n - 1:
                seq, ch, cw = self.patch_mergers[stage](seq, ch, cw)

        for i, stage in enumerate(range(n - 2, -1, -1)):
            seq, ch, cw = self.patch_expanders[i](seq, ch, cw)
            seq = torch.cat([seq, skips[stage]], dim=-1)
            seq = norm_fp32(self.fusion_norms[i], seq)
            seq = self.fusion_denses[i](seq)
            freqs = rope_for(stage, seq.shape[1], seq.device)
            for block in self.decoder_blocks[i]:
                seq = block(seq, t_embs[stage], text_embs[stage], freqs)

        out = norm_fp32(self.final_norm, seq)
        out = self.final_proj(out)
        if self.learn_sigma:
            out, _ = out.chunk(2, dim=-1)
        if self.use_hilbert:
            return hilbert_unpatchify(out, inv_idx, p, H, W, self.output_channels)
        return unpatchify(out, channels=self.output_channels)
