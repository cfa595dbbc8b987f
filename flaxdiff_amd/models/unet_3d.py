"""UNet3D — conditional video UNet.

Behavior contract: reference /root/reference/flaxdiff/models/unet_3d.py:24-445
+ unet_3d_blocks.py (FlaxTransformerTemporalModel :26-101 — attention across
frames at every pixel; TemporalConvLayer :103-167 — stack of (3,1,1) convs
with ZERO-INIT final so the block starts as identity; CrossAttn down/up/mid
3-D blocks :170-505 that interleave spatial resnets, spatial cross-attention,
temporal conv and temporal attention).

MI355X design: video [B,T,H,W,C] folds frames into the batch for every
SPATIAL op — so the 2-D HIP kernel set (GN+SiLU, implicit-GEMM conv, flash
attention) runs unchanged on [B*T,H,W,C] — and unfolds only for the two
temporal mixers:
  * TemporalConvLayer: (3,1,1) depth conv == a 1-D conv over T at each
    (h,w,c); executed as 3 shifted GEMM-free adds (torch) over the folded
    layout — HBM-bound, cheap;
  * TemporalAttention: [B*H*W, T, C] sequences through the flash-attn kernel.
`is_video_model = True` tells GeneralDiffusionTrainer NOT to fold time.
"""
from __future__ import annotations

from typing import Callable, Optional, Sequence

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops
from .attention import TransformerBlock
from .common import (Conv, Dense, Downsample, FourierEmbedding, GroupNorm,
                     ResidualBlock, TimeProjection, Upsample)


class _TemporalConv(nn.Module):
    """One FULL (3,1,1) conv over frames: three [Cin,Cout] taps applied to
    the previous/current/next frame (zero-padded edges) — exactly a 1-D
    temporal conv with complete channel mixing, expressed as three dense
    MFMA GEMMs over the folded pixels (reference uses nn.Conv k=(3,1,1))."""

    def __init__(self, cin: int, cout: int, zero_init: bool = False):
        super().__init__()
        def mk():
            if zero_init:
                return nn.Parameter(torch.zeros(cin, cout))
            w = torch.randn(cin, cout) * (1.0 / (3 * cin)) ** 0.5
            return nn.Parameter(w)
        self.w_prev = mk()
        self.w_cur = mk()
        self.w_next = mk()
        self.bias = nn.Parameter(torch.zeros(cout))

    def forward(self, v: torch.Tensor) -> torch.Tensor:
        # v: [B, T, H, W, Cin]
        prev = torch.cat([torch.zeros_like(v[:, :1]), v[:, :-1]], dim=1)
        nxt = torch.cat([v[:, 1:], torch.zeros_like(v[:, :1])], dim=1)
        out = ops.dense(prev, self.w_prev) + ops.dense(v, self.w_cur) \
            + ops.dense(nxt, self.w_next)
        return out + self.bias.to(out.dtype)


class TemporalConvLayer(nn.Module):
    """4x [GN -> SiLU -> full (3,1,1) conv] with ZERO-INIT final conv so the
    block starts as identity, residual (reference unet_3d_blocks.py:103-167:
    conv1 in->out, conv2 out->in, conv3 in->in, conv4 in->in zero-init)."""

    def __init__(self, channels: int, norm_groups: int = 32,
                 out_channels: Optional[int] = None):
        super().__init__()
        cout = out_channels or channels
        chans = [(channels, cout), (cout, channels),
                 (channels, channels), (channels, channels)]
        def gn(c):
            g = min(norm_groups, c)
            while c % g:
                g -= 1
            return GroupNorm(g, c, eps=1e-5)
        self.norms = nn.ModuleList([gn(ci) for ci, _ in chans])
        self.convs = nn.ModuleList(
            [_TemporalConv(ci, co, zero_init=(i == 3))
             for i, (ci, co) in enumerate(chans)])

    def forward(self, x: torch.Tensor, num_frames: int) -> torch.Tensor:
        # x: [B*T, H, W, C]
        BT, H, W, C = x.shape
        B = BT // num_frames
        v = x.reshape(B, num_frames, H, W, C)
        identity = v
        for norm, conv in zip(self.norms, self.convs):
            h = norm(v.reshape(-1, H, W, v.shape[-1]), silu=True)
            v = conv(h.reshape(B, num_frames, H, W, -1))
        return (identity + v).reshape(BT, H, W, C)


class TemporalAttention(nn.Module):
    """GN -> proj_in -> self-attention over frames at each pixel -> proj_out,
    residual (reference unet_3d_blocks.py:26-101)."""

    def __init__(self, channels: int, heads: int = 4, norm_groups: int = 32):
        super().__init__()
        g = min(norm_groups, channels)
        while channels % g:
            g -= 1
        self.norm = GroupNorm(g, channels, eps=1e-5)
        self.proj_in = Dense(channels, channels)
        self.block = TransformerBlock(channels, heads=heads,
                                      dim_head=channels // heads,
                                      use_self_and_cross=False,
                                      only_pure_attention=False,
                                      norm_inputs=False,
                                      explicitly_add_residual=False)
        self.proj_out = Dense(channels, channels, zero_init=True)

    def forward(self, x: torch.Tensor, num_frames: int) -> torch.Tensor:
        BT, H, W, C = x.shape
        B = BT // num_frames
        residual = x
        h = self.norm(x)
        h = h.reshape(B, num_frames, H * W, C).permute(0, 2, 1, 3)
        h = h.reshape(B * H * W, num_frames, C)
        h = self.proj_out(self.block(self.proj_in(h)))
        h = h.reshape(B, H * W, num_frames, C).permute(0, 2, 1, 3)
        return residual + h.reshape(BT, H, W, C)


class UNet3D(nn.Module):
    """Video UNet: spatial levels from the 2-D stack + temporal mixers."""

    is_video_model = True

    def __init__(self,
                 output_channels: int = 3,
                 in_channels: int = 3,
                 emb_features: int = 256,
                 feature_depths: Sequence[int] = (64, 128, 256),
                 attention_configs: Sequence[Optional[dict]] = ({"heads": 4},) * 3,
                 num_res_blocks: int = 1,
                 norm_groups: int = 8,
                 context_dim: int = 768,
                 temporal_attention: bool = True,
                 activation: Callable = F.silu):
        super().__init__()
        self.output_channels = output_channels
        self.time_embed = FourierEmbedding(features=emb_features)
        self.time_proj = TimeProjection(emb_features, emb_features)

        f0 = feature_depths[0]
        self.conv_in = Conv(in_channels, f0, (3, 3), (1, 1))

        def res(cin, cout):
            return ResidualBlock("conv", cin, cout, emb_features,
                                 norm_groups=norm_groups, activation=activation)

        def attn(dim, cfg):
            return TransformerBlock(dim, heads=cfg["heads"],
                                    dim_head=dim // cfg["heads"],
                                    only_pure_attention=True,
                                    context_dim=context_dim)

        def tattn(dim, cfg):
            return (TemporalAttention(dim, heads=cfg["heads"],
                                      norm_groups=norm_groups)
                    if temporal_attention else nn.Identity())

        ch = f0
        skips = [f0]
        self.down = nn.ModuleList()
        for i, (dim, cfg) in enumerate(zip(feature_depths, attention_configs)):
            level = nn.ModuleDict()
            level["res"] = nn.ModuleList()
            level["tconv"] = nn.ModuleList()
            level["attn"] = nn.ModuleList()
            level["tattn"] = nn.ModuleList()
            for _ in range(num_res_blocks):
                level["res"].append(res(ch, dim))
                ch = dim
                level["tconv"].append(TemporalConvLayer(ch, norm_groups))
                level["attn"].append(attn(ch, cfg) if cfg else nn.Identity())
                level["tattn"].append(tattn(ch, cfg) if cfg else nn.Identity())
                skips.append(ch)
            if i != len(feature_depths) - 1:
                level["down"] = Downsample(ch, dim)
            self.down.append(level)

        self.mid_res1 = res(ch, ch)
        self.mid_tconv = TemporalConvLayer(ch, norm_groups)
        self.mid_attn = attn(ch, attention_configs[-1] or {"heads": 4})
        self.mid_tattn = tattn(ch, attention_configs[-1] or {"heads": 4})
        self.mid_res2 = res(ch, ch)

        self.up = nn.ModuleList()
        for i, (dim, cfg) in enumerate(zip(reversed(feature_depths),
                                           reversed(attention_configs))):
            level = nn.ModuleDict()
            level["res"] = nn.ModuleList()
            level["tconv"] = nn.ModuleList()
            level["attn"] = nn.ModuleList()
            level["tattn"] = nn.ModuleList()
            for _ in range(num_res_blocks):
                level["res"].append(res(ch + skips.pop(), dim))
                ch = dim
                level["tconv"].append(TemporalConvLayer(ch, norm_groups))
                level["attn"].append(attn(ch, cfg) if cfg else nn.Identity())
                level["tattn"].append(tattn(ch, cfg) if cfg else nn.Identity())
            if i != len(feature_depths) - 1:
                level["up"] = Upsample(ch, dim)
            self.up.append(level)

        x0 = feature_depths[0]
        self.final_res = res(ch + skips.pop(), x0)
        self.final_norm = GroupNorm(min(norm_groups, x0), x0, eps=1e-5)
        self.conv_out = Conv(x0, output_channels, (3, 3), (1, 1))
        assert not skips

    def forward(self, x: torch.Tensor, temb: torch.Tensor,
                textcontext: Optional[torch.Tensor] = None) -> torch.Tensor:
        assert x.dim() == 5, "UNet3D expects [B, T, H, W, C]"
        B, T, H, W, C = x.shape
        x = x.reshape(B * T, H, W, C)
        temb_rep = temb.repeat_interleave(T, dim=0)
        t_vec = self.time_proj(self.time_embed(temb_rep).to(x.dtype))
        ctx = None
        if textcontext is not None:
            ctx = textcontext.repeat_interleave(T, dim=0).to(x.dtype)

        h = self.conv_in(x)
        stack = [h]
        for level in self.down:
            for rb, tc, ab, tb in zip(level["res"], level["tconv"],
                                      level["attn"], level["tattn"]):
                h = rb(h, t_vec)
                h = tc(h, T)
                if not isinstance(ab, nn.Identity):
                    h = ab(h, ctx)
                if not isinstance(tb, nn.Identity):
                    h = tb(h, T)
                stack.append(h)
            if "down" in level:
                h = level["down"](h)

        h = self.mid_res1(h, t_vec)
        h = self.mid_tconv(h, T)
        h = self.mid_attn(h, ctx)
        if not isinstance(self.mid_tattn, nn.Identity):
            h = self.mid_tattn(h, T)
        h = self.mid_res2(h, t_vec)

        for level in self.up:
            for rb, tc, ab, tb in zip(level["res"], level["tconv"],
                                      level["attn"], level["tattn"]):
                h = ops.cat_channels(h, stack.pop())
                h = rb(h, t_vec)
                h = tc(h, T)
                if not isinstance(ab, nn.Identity):
                    h = ab(h, ctx)
                if not isinstance(tb, nn.Identity):
                    h = tb(h, T)
            if "up" in level:
                h = level["up"](h)

        h = ops.cat_channels(h, stack.pop())
        h = self.final_res(h, t_vec)
        h = self.final_norm(h, silu=True)
        h = self.conv_out(h)
        return h.reshape(B, T, H, W, self.output_channels)
