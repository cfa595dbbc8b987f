"""SimpleDiT — classic DiT with RoPE + AdaLN-Zero + optional Hilbert/zigzag scan.

Behavior contract: reference /root/reference/flaxdiff/models/simple_dit.py
(DiTBlock :23-95, SimpleDiT :103-307). Key semantics preserved:
  * patchify via conv-embed (raster) or raw-patch + Dense (hilbert/zigzag);
  * additive MAE-style 2-D sin-cos PE always applied, reordered to the scan
    order so each token carries its TRUE 2-D position (:246-258);
  * RoPE per head-dim on q/k; overridden to identity for non-raster scans
    (:282-284) since sequence index is not a 2-D position there;
  * conditioning = Fourier time embed -> TimeProjection -> Dense, plus
    mean-pooled projected text (:263-271);
  * AdaLN-Zero modulate -> attn -> gate; modulate -> MLP -> gate (:74-95);
  * zero-init final projection, optional learn_sigma doubling (:199-208).

MI355X notes: all GEMM-shaped work (patch embed, QKV/out, MLP, AdaLN proj)
runs as MFMA library GEMMs; attention is the hand-written flash kernel; the
scan reorder is a device gather with host-cached index tables (hilbert.py).
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .common import Dense, FourierEmbedding, TimeProjection
from .hilbert import (build_2d_sincos_pos_embed, hilbert_indices,
                      hilbert_patchify, hilbert_unpatchify, zigzag_indices,
                      zigzag_patchify)
from .vit_common import (AdaLNParams, PatchEmbedding, RoPEAttention, norm_fp32,
                         RotaryEmbedding, layer_norm_noaffine, unpatchify)


class DiTBlock(nn.Module):
    """AdaLN-Zero DiT block: modulate->RoPE-attn->gate; modulate->MLP->gate."""

    def __init__(self, features: int, num_heads: int, rope_emb: RotaryEmbedding,
                 mlp_ratio: int = 4, norm_epsilon: float = 1e-5,
                 use_gating: bool = True):
        super().__init__()
        self.eps = norm_epsilon
        self.use_gating = use_gating
        self.ada_params = AdaLNParams(features, features)
        self.attention = RoPEAttention(features, num_heads, features // num_heads,
                                       use_bias=True, rope_emb=rope_emb)
        hidden = features * mlp_ratio
        self.mlp_in = Dense(features, hidden)
        self.mlp_out = Dense(hidden, features)

    def forward(self, x, conditioning, freqs_cis):
        scale_mlp, shift_mlp, gate_mlp, scale_attn, shift_attn, gate_attn = \
            self.ada_params(conditioning).chunk(6, dim=-1)

        norm_x = layer_norm_noaffine(x, self.eps)
        attn_out = self.attention(norm_x * (1 + scale_attn) + shift_attn,
                                  context=None, freqs_cis=freqs_cis)
        x = x + (gate_attn * attn_out if self.use_gating else attn_out)

        norm_x = layer_norm_noaffine(x, self.eps)
        h = self.mlp_in(norm_x * (1 + scale_mlp) + shift_mlp)
        mlp_out = self.mlp_out(F.gelu(h))
        x = x + (gate_mlp * mlp_out if self.use_gating else mlp_out)
        return x


class SimpleDiT(nn.Module):
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
                 use_hilbert: bool = False,
                 use_zigzag: bool = False):
        super().__init__()
        assert not (use_hilbert and use_zigzag), \
            "use_hilbert and use_zigzag are mutually exclusive"
        self.output_channels = output_channels
        self.patch_size = patch_size
        self.emb_features = emb_features
        self.learn_sigma = learn_sigma
        self.use_hilbert = use_hilbert
        self.use_zigzag = use_zigzag

        if use_hilbert or use_zigzag:
            self.hilbert_proj = Dense(patch_size * patch_size * in_channels,
                                      emb_features)
        else:
            self.patch_embed = PatchEmbedding(patch_size, emb_features, in_channels)

        # time conditioning: Fourier -> 2-layer GELU proj at mlp_ratio width
        # -> Dense back to emb (reference simple_dit.py:149-156)
        self.time_fourier = FourierEmbedding(features=emb_features)
        self.time_proj = TimeProjection(emb_features, emb_features * mlp_ratio)
        self.time_out = Dense(emb_features * mlp_ratio, emb_features)

        self.text_proj = Dense(context_dim, emb_features)

        self.rope = RotaryEmbedding(dim=emb_features // num_heads, max_seq_len=4096)
        self.blocks = nn.ModuleList([
            DiTBlock(emb_features, num_heads, self.rope, mlp_ratio, norm_epsilon)
            for _ in range(num_layers)])

        # final norm is a full affine LayerNorm in the reference (:199-201)
        self.final_norm = nn.LayerNorm(emb_features, eps=norm_epsilon)
        out_dim = patch_size * patch_size * output_channels
        if learn_sigma:
            out_dim *= 2
        self.final_proj = Dense(emb_features, out_dim, zero_init=True)
        self._pe_cache = {}

    def _pos_embed(self, h_p: int, w_p: int, device, dtype) -> torch.Tensor:
        key = (h_p, w_p, self.use_hilbert, self.use_zigzag)
        pe = self._pe_cache.get(key)
        if pe is None:
            pe = torch.from_numpy(
                build_2d_sincos_pos_embed(self.emb_features, h_p, w_p).copy())
            if self.use_hilbert:
                pe = pe[hilbert_indices(h_p, w_p)]
            elif self.use_zigzag:
                pe = pe[zigzag_indices(h_p, w_p)]
            self._pe_cache[key] = pe
        return pe.to(device=device, dtype=dtype)

    def forward(self, x: torch.Tensor, temb: torch.Tensor,
                textcontext: Optional[torch.Tensor] = None) -> torch.Tensor:
        B, H, W, C = x.shape
        p = self.patch_size
        assert H % p == 0 and W % p == 0
        h_p, w_p = H // p, W // p

        inv_idx = None
        if self.use_hilbert:
            tokens_raw, inv_idx = hilbert_patchify(x, p)
            tokens = self.hilbert_proj(tokens_raw)
        elif self.use_zigzag:
            tokens_raw, inv_idx = zigzag_patchify(x, p)
            tokens = self.hilbert_proj(tokens_raw)
        else:
            tokens = self.patch_embed(x)
        n_tokens = tokens.shape[1]

        tokens = tokens + self._pos_embed(h_p, w_p, tokens.device, tokens.dtype)[None]

        cond = self.time_out(self.time_proj(self.time_fourier(temb).to(tokens.dtype)))
        if textcontext is not None:
            cond = cond + self.text_proj(textcontext.to(tokens.dtype)).mean(dim=1)

        cos, sin = self.rope(n_tokens)
        if self.use_hilbert or self.use_zigzag:
            cos, sin = torch.ones_like(cos), torch.zeros_like(sin)
        freqs_cis = (cos.to(tokens.device), sin.to(tokens.device))

        for block in self.blocks:
            tokens = block(tokens, cond, freqs_cis)

        out = norm_fp32(self.final_norm, tokens)
        out = self.final_proj(out)

        if self.learn_sigma:
            out, _logvar = out.chunk(2, dim=-1)
        if inv_idx is not None:
            return hilbert_unpatchify(out, inv_idx, p, H, W, self.output_channels)
        return unpatchify(out, channels=self.output_channels)
