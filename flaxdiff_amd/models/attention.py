"""Attention blocks (NHWC / token-sequence), running on the CDNA4 attention
kernel through ops.attention.

Structure contract: reference /root/reference/flaxdiff/models/attention.py.
The UNet's default block (only_pure_attention=True, simple_unet.py:81) is
RMSNorm -> cross-attention(x, textcontext) -> +residual; this maps to the
small-KV (S_kv=77) fused attention kernel. EfficientAttention and
NormalAttention collapse into one implementation here — the reference's
Pallas-vs-naive split is a TPU artifact; on MI355X both run the same
hand-written flash kernel.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops
from .common import Dense, RMSNorm


class NormalAttention(nn.Module):
    """QKV projections -> fp32-softmax attention -> out projection
    (reference attention.py:117-177)."""

    def __init__(self, query_dim: int, heads: int = 4, dim_head: int = 64,
                 use_bias: bool = True, force_fp32_for_softmax: bool = True,
                 context_dim: Optional[int] = None):
        super().__init__()
        self.heads = heads
        self.dim_head = dim_head
        inner = heads * dim_head
        cdim = context_dim if context_dim is not None else query_dim
        self.to_q = Dense(query_dim, inner, use_bias=use_bias)
        self.to_k = Dense(cdim, inner, use_bias=use_bias)
        self.to_v = Dense(cdim, inner, use_bias=use_bias)
        self.to_out = Dense(inner, query_dim, use_bias=use_bias)

    def forward(self, x: torch.Tensor, context: Optional[torch.Tensor] = None,
                add: Optional[torch.Tensor] = None):
        # `add` (same shape as x) rides the out-projection's GEMM epilogue —
        # the pre-norm residual add costs no extra kernel.
        orig_shape = x.shape
        if x.dim() == 4:
            B, H, W, C = x.shape
            x = x.reshape(B, H * W, C)
        B, S, _ = x.shape
        ctx = x if context is None else context
        if ctx.dim() == 4:
            ctx = ctx.reshape(ctx.shape[0], -1, ctx.shape[-1])
        ctx = ctx.to(x.dtype)

        q = self.to_q(x).reshape(B, S, self.heads, self.dim_head).permute(0, 2, 1, 3)
        k = self.to_k(ctx).reshape(B, ctx.shape[1], self.heads, self.dim_head).permute(0, 2, 1, 3)
        v = self.to_v(ctx).reshape(B, ctx.shape[1], self.heads, self.dim_head).permute(0, 2, 1, 3)

        o = ops.attention(q, k, v)
        o = o.permute(0, 2, 1, 3).reshape(B, S, self.heads * self.dim_head)
        o = self.to_out(o, add=add.reshape(B, S, -1) if add is not None else None)
        return o.reshape(orig_shape)


# The reference's Pallas flash-attention wrapper — same math, same kernel here.
EfficientAttention = NormalAttention


class GEGLU(nn.Module):
    """proj -> split -> x * gelu(g) (reference attention.py:179-205)."""

    def __init__(self, dim: int, mult: int = 4):
        super().__init__()
        self.proj = Dense(dim, dim * mult * 2)
        self.inner = dim * mult

    def forward(self, x):
        return ops.geglu(self.proj(x))


class FeedForward(nn.Module):
    """GEGLU MLP, dim*4 inner (reference attention.py:207-238)."""

    def __init__(self, dim: int):
        super().__init__()
        self.net_0 = GEGLU(dim)
        self.net_2 = Dense(dim * 4, dim)

    def forward(self, x, add=None):
        return self.net_2(self.net_0(x), add=add)


class BasicTransformerBlock(nn.Module):
    """Self-attn + cross-attn + GEGLU FF with RMSNorm pre-norms
    (reference attention.py:240-303). only_pure_attention => just cross-attn."""

    def __init__(self, query_dim: int, heads: int = 4, dim_head: int = 64,
                 use_bias: bool = True, use_cross_only: bool = False,
                 only_pure_attention: bool = False,
                 force_fp32_for_softmax: bool = True, norm_epsilon: float = 1e-4,
                 context_dim: Optional[int] = None):
        super().__init__()
        self.only_pure_attention = only_pure_attention
        self.use_cross_only = use_cross_only
        # Match the reference's lazy init: submodules that the chosen mode
        # never calls are not materialized (flax only creates params for
        # traced modules, so only_pure_attention checkpoints hold attention2 only).
        self.attention2 = NormalAttention(query_dim, heads, dim_head, use_bias,
                                          force_fp32_for_softmax, context_dim=context_dim)
        if not only_pure_attention:
            if not use_cross_only:
                self.attention1 = NormalAttention(query_dim, heads, dim_head, use_bias,
                                                  force_fp32_for_softmax)
                self.norm1 = RMSNorm(query_dim, eps=norm_epsilon)
            self.ff = FeedForward(query_dim)
            self.norm2 = RMSNorm(query_dim, eps=norm_epsilon)
            self.norm3 = RMSNorm(query_dim, eps=norm_epsilon)

    def forward(self, hidden_states, context=None, add=None):
        # residual adds fused into each sub-block's final GEMM epilogue;
        # `add` rides the LAST sub-block's epilogue (the TransformerBlock
        # outer residual in the UNet's only_pure_attention default)
        if self.only_pure_attention:
            return self.attention2(hidden_states, context, add=add)
        if not self.use_cross_only:
            hidden_states = self.attention1(self.norm1(hidden_states),
                                            add=hidden_states)
        hidden_states = self.attention2(self.norm2(hidden_states), context,
                                        add=hidden_states)
        hidden_states = self.ff(self.norm3(hidden_states), add=hidden_states)
        return hidden_states if add is None else hidden_states + add


class TransformerBlock(nn.Module):
    """Optional input RMSNorm + optional in/out projection around a
    BasicTransformerBlock, with residual add (reference attention.py:305-380)."""

    def __init__(self, in_channels: int, heads: int = 4, dim_head: int = 32,
                 use_linear_attention: bool = True, use_projection: bool = False,
                 use_self_and_cross: bool = True, only_pure_attention: bool = False,
                 force_fp32_for_softmax: bool = True, norm_inputs: bool = True,
                 explicitly_add_residual: bool = True, norm_epsilon: float = 1e-4,
                 context_dim: Optional[int] = None):
        super().__init__()
        self.norm_inputs = norm_inputs
        self.use_projection = use_projection
        self.only_pure_attention = only_pure_attention
        self.explicitly_add_residual = explicitly_add_residual
        inner_dim = heads * dim_head if use_projection else in_channels
        if norm_inputs:
            self.input_norm = RMSNorm(in_channels, eps=norm_epsilon)
        if use_projection:
            # linear-vs-1x1-conv both lower to the same GEMM in NHWC
            self.project_in = Dense(in_channels, inner_dim, use_bias=False)
            self.project_out = Dense(inner_dim, in_channels, use_bias=False)
        self.block = BasicTransformerBlock(
            query_dim=inner_dim, heads=heads, dim_head=dim_head, use_bias=False,
            use_cross_only=not use_self_and_cross,
            only_pure_attention=only_pure_attention,
            force_fp32_for_softmax=force_fp32_for_softmax,
            norm_epsilon=norm_epsilon, context_dim=context_dim)

    def forward(self, x, context=None):
        if self.norm_inputs:
            x = self.input_norm(x)
        projected = self.project_in(x) if self.use_projection else x
        if context is None:
            context = projected
        if self.use_projection:
            projected = self.block(projected, context)
            if self.only_pure_attention or self.explicitly_add_residual:
                return self.project_out(projected, add=x)
            return self.project_out(projected)
        if self.only_pure_attention or self.explicitly_add_residual:
            return self.block(projected, context, add=x)
        return self.block(projected, context)
