"""UViT and SimpleUDiT — U-Net-shaped transformers.

Behavior contract: reference /root/reference/flaxdiff/models/simple_vit.py
(UViT :18-250, SimpleUDiT :255-447).

UViT: token sequence = patch tokens + one time token + text tokens; N/2 down
TransformerBlocks (skips pushed), one mid block, N/2 up blocks with
skip-concat + Dense fusion; learned (normal 0.02) additive PE over patches;
zero-init final projection; optional output ResBlock-ish conv head.

SimpleUDiT: same U shape but with RoPE + AdaLN-Zero DiTBlocks and pooled
text+time conditioning; final proj fp32 zero-init.

MI355X notes: these are pure token-GEMM models — everything lowers to MFMA
library GEMMs + the flash-attention kernel; skip fusion is one [2D -> D] GEMM.
"""
from __future__ import annotations

from typing import Callable, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .attention import TransformerBlock
from .common import Conv, Dense, FourierEmbedding, TimeProjection
from .hilbert import hilbert_patchify, hilbert_unpatchify
from .simple_dit import DiTBlock
from .vit_common import PatchEmbedding, RotaryEmbedding, norm_fp32, unpatchify


class UViT(nn.Module):
    def __init__(self,
                 output_channels: int = 3,
                 in_channels: int = 3,
                 patch_size: int = 16,
                 emb_features: int = 768,
                 num_layers: int = 12,
                 num_heads: int = 12,
                 use_projection: bool = False,
                 use_self_and_cross: bool = False,
                 activation: Callable = F.silu,
                 norm_groups: int = 8,
                 add_residualblock_output: bool = False,
                 norm_inputs: bool = False,
                 explicitly_add_residual: bool = True,
                 norm_epsilon: float = 1e-5,
                 use_hilbert: bool = False,
                 context_dim: int = 768):
        super().__init__()
        assert num_layers % 2 == 0, "num_layers must be even for U-Net structure"
        half = num_layers // 2
        self.patch_size = patch_size
        self.output_channels = output_channels
        self.add_residualblock_output = add_residualblock_output
        self.use_hilbert = use_hilbert
        self.activation = activation

        self.patch_embed = PatchEmbedding(patch_size, emb_features, in_channels)
        if use_hilbert:
            self.hilbert_proj = Dense(patch_size * patch_size * in_channels,
                                      emb_features)
        max_patches = (512 // patch_size) ** 2
        self.pos_encoding = nn.Parameter(
            torch.randn(1, max_patches, emb_features) * 0.02)

        self.time_embed = FourierEmbedding(features=emb_features)
        self.time_proj = TimeProjection(emb_features, emb_features)
        self.text_proj = Dense(context_dim, emb_features)

        def block():
            return TransformerBlock(
                in_channels=emb_features, heads=num_heads,
                dim_head=emb_features // num_heads,
                use_projection=use_projection,
                use_self_and_cross=use_self_and_cross,
                only_pure_attention=False, norm_inputs=norm_inputs,
                explicitly_add_residual=explicitly_add_residual,
                norm_epsilon=norm_epsilon)

        self.down_blocks = nn.ModuleList([block() for _ in range(half)])
        self.mid_block = block()
        self.up_dense = nn.ModuleList(
            [Dense(emb_features * 2, emb_features) for _ in range(half)])
        self.up_blocks = nn.ModuleList([block() for _ in range(half)])

        self.final_norm = nn.LayerNorm(emb_features, eps=norm_epsilon)
        patch_dim = patch_size ** 2 * output_channels
        self.final_proj = Dense(emb_features, patch_dim, zero_init=True)

        if add_residualblock_output:
            self.final_conv1 = Conv(output_channels + in_channels, 64, (3, 3), (1, 1))
            self.final_norm_conv = nn.LayerNorm(64, eps=norm_epsilon)
            self.final_conv2 = Conv(64, output_channels, (3, 3), (1, 1))
        else:
            self.final_conv_direct = Conv(output_channels, output_channels,
                                          (1, 1), (1, 1))

    def forward(self, x: torch.Tensor, temb: torch.Tensor,
                textcontext: Optional[torch.Tensor] = None) -> torch.Tensor:
        original_img = x
        B, H, W, C = x.shape
        p = self.patch_size
        assert H % p == 0 and W % p == 0
        num_patches = (H // p) * (W // p)

        inv_idx = None
        if self.use_hilbert:
            patches_raw, inv_idx = hilbert_patchify(x, p)
            tokens = self.hilbert_proj(patches_raw)
        else:
            tokens = self.patch_embed(x)
        assert num_patches <= self.pos_encoding.shape[1]
        tokens = tokens + self.pos_encoding[:, :num_patches, :].to(tokens.dtype)

        time_token = self.time_proj(self.time_embed(temb).to(tokens.dtype))
        time_token = time_token.unsqueeze(1)
        if textcontext is not None:
            text_tokens = self.text_proj(textcontext.to(tokens.dtype))
            seq = torch.cat([tokens, time_token, text_tokens], dim=1)
        else:
            seq = torch.cat([tokens, time_token], dim=1)

        skips = []
        for blk in self.down_blocks:
            seq = blk(seq)
            skips.append(seq)
        seq = self.mid_block(seq)
        for dense, blk in zip(self.up_dense, self.up_blocks):
            seq = torch.cat([seq, skips.pop()], dim=-1)
            seq = dense(seq)
            seq = blk(seq)

        seq = norm_fp32(self.final_norm, seq)
        patches_out = self.final_proj(seq[:, :num_patches, :])

        if self.use_hilbert:
            img = hilbert_unpatchify(patches_out, inv_idx, p, H, W,
                                     self.output_channels)
        else:
            img = unpatchify(patches_out, channels=self.output_channels)

        if self.add_residualblock_output:
            img = torch.cat([original_img.to(img.dtype), img], dim=-1)
            img = self.final_conv1(img)
            img = self.final_norm_conv(img.float()).to(img.dtype)
            img = self.activation(img)
            img = self.final_conv2(img)
        return img


class SimpleUDiT(nn.Module):
    def __init__(self,
                 output_channels: int = 3,
                 in_channels: int = 3,
                 patch_size: int = 16,
                 emb_features: int = 768,
                 num_layers: int = 12,
                 num_heads: int = 12,
                 mlp_ratio: int = 4,
                 norm_epsilon: float = 1e-5,
                 learn_sigma: bool = False,
                 use_hilbert: bool = False,
                 context_dim: int = 768):
        super().__init__()
        assert num_layers % 2 == 0, "num_layers must be even for U-Net structure"
        half = num_layers // 2
        self.patch_size = patch_size
        self.output_channels = output_channels
        self.learn_sigma = learn_sigma
        self.use_hilbert = use_hilbert

        self.patch_embed = PatchEmbedding(patch_size, emb_features, in_channels)
        if use_hilbert:
            self.hilbert_proj = Dense(patch_size * patch_size * in_channels,
                                      emb_features)

        self.time_fourier = FourierEmbedding(features=emb_features)
        self.time_proj = TimeProjection(emb_features, emb_features * mlp_ratio)
        self.time_out = Dense(emb_features * mlp_ratio, emb_features)
        self.text_proj = Dense(context_dim, emb_features)

        max_patches = (512 // patch_size) ** 2
        self.rope = RotaryEmbedding(dim=emb_features // num_heads,
                                    max_seq_len=max_patches)

        def block():
            return DiTBlock(emb_features, num_heads, self.rope, mlp_ratio,
                            norm_epsilon)

        self.down_blocks = nn.ModuleList([block() for _ in range(half)])
        self.mid_block = block()
        self.up_dense = nn.ModuleList(
            [Dense(emb_features * 2, emb_features) for _ in range(half)])
        self.up_blocks = nn.ModuleList([block() for _ in range(half)])

        self.final_norm = nn.LayerNorm(emb_features, eps=norm_epsilon)
        out_dim = patch_size * patch_size * output_channels
        if learn_sigma:
            out_dim *= 2
        self.final_proj = Dense(emb_features, out_dim, zero_init=True)

    def forward(self, x: torch.Tensor, temb: torch.Tensor,
                textcontext: Optional[torch.Tensor] = None) -> torch.Tensor:
        B, H, W, C = x.shape
        p = self.patch_size

        inv_idx = None
        if self.use_hilbert:
            patches_raw, inv_idx = hilbert_patchify(x, p)
            seq = self.hilbert_proj(patches_raw)
        else:
            seq = self.patch_embed(x)

        cond = self.time_out(self.time_proj(self.time_fourier(temb).to(seq.dtype)))
        if textcontext is not None:
            text_emb = self.text_proj(textcontext.to(seq.dtype))
            if text_emb.dim() == 3:
                text_emb = text_emb.mean(dim=1)
            cond = cond + text_emb

        skips = []
        for blk in self.down_blocks:
            seq = blk(seq, cond, None)
            skips.append(seq)
        seq = self.mid_block(seq, cond, None)
        for dense, blk in zip(self.up_dense, self.up_blocks):
            seq = torch.cat([seq, skips.pop()], dim=-1)
            seq = dense(seq)
            seq = blk(seq, cond, None)

        out = norm_fp32(self.final_norm, seq)
        out = self.final_proj(out)
        if self.learn_sigma:
            out, _ = out.chunk(2, dim=-1)
        if self.use_hilbert:
            img = hilbert_unpatchify(out, inv_idx, p, H, W, self.output_channels)
        else:
            img = unpatchify(out, channels=self.output_channels)
        return img.float()
