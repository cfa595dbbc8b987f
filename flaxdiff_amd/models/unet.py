"""UNet — the primary architecture (BASELINE configs 1-4).

Structure contract: reference /root/reference/flaxdiff/models/simple_unet.py:11-222,
reproduced exactly including its channel flow quirks:
  * per-level ResidualBlocks run at the INCOMING channel count (dim_in), the
    Downsample at the end of level i projects to feature_depths[i]
    (simple_unet.py:60-100);
  * the i-th Upsample projects to feature_depths[-i] (note -0 == 0,
    simple_unet.py:176-184);
  * attention (default: pure cross-attention to the text context,
    only_pure_attention=True) on the last res block of every level.

Call signature: model(x[B,H,W,C] NHWC, temb[B], textcontext[B,S,Ctx]) -> [B,H,W,out].
"""
from __future__ import annotations

from typing import Callable, List, Optional, Sequence

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops
from .attention import TransformerBlock
from .common import (ConvLayer, Downsample, FourierEmbedding, GroupNorm,
                     RMSNorm, ResidualBlock, TimeProjection, Upsample)


def _make_attention(dim: int, attention_config: dict, context_dim: int,
                    default_fp32_softmax: bool = False) -> TransformerBlock:
    heads = attention_config["heads"]
    return TransformerBlock(
        in_channels=dim,
        heads=heads,
        dim_head=dim // heads,
        use_projection=attention_config.get("use_projection", False),
        use_self_and_cross=attention_config.get("use_self_and_cross", True),
        only_pure_attention=attention_config.get("only_pure_attention", True),
        force_fp32_for_softmax=attention_config.get("force_fp32_for_softmax",
                                                    default_fp32_softmax),
        norm_inputs=attention_config.get("norm_inputs", True),
        explicitly_add_residual=attention_config.get("explicitly_add_residual", True),
        context_dim=context_dim,
    )


class Unet(nn.Module):
    def __init__(self,
                 output_channels: int = 3,
                 in_channels: int = 3,
                 emb_features: int = 64 * 4,
                 feature_depths: Sequence[int] = (64, 128, 256, 512),
                 attention_configs: Sequence[Optional[dict]] = ({"heads": 8},) * 4,
                 num_res_blocks: int = 2,
                 num_middle_res_blocks: int = 1,
                 activation: Callable = F.silu,
                 norm_groups: int = 8,
                 context_dim: int = 768,
                 conv_type: str = "conv"):
        super().__init__()
        self.output_channels = output_channels
        self.emb_features = emb_features
        self.feature_depths = list(feature_depths)
        self.attention_configs = list(attention_configs)
        self.num_res_blocks = num_res_blocks
        self.num_middle_res_blocks = num_middle_res_blocks
        self.activation = activation
        self.norm_groups = norm_groups

        self.time_embed = FourierEmbedding(features=emb_features)
        self.time_proj = TimeProjection(emb_features, emb_features)

        f0 = self.feature_depths[0]
        self.conv_in = ConvLayer(conv_type, in_channels, f0, (3, 3), (1, 1))

        def res_block(cin, cout):
            return ResidualBlock(conv_type, cin, cout, emb_features,
                                 kernel_size=(3, 3), strides=(1, 1),
                                 activation=activation, norm_groups=norm_groups)

        # ---- encoder --------------------------------------------------------
        # skip-connection channel bookkeeping mirrors the reference `downs` stack
        ch = f0
        skip_channels: List[int] = [f0]
        self.down_blocks = nn.ModuleList()
        for i, (dim_out, att_cfg) in enumerate(zip(self.feature_depths, self.attention_configs)):
            dim_in = ch
            level = nn.ModuleDict()
            res_list = nn.ModuleList()
            attn_list = nn.ModuleList()
            for j in range(num_res_blocks):
                res_list.append(res_block(dim_in, dim_in))
                if att_cfg is not None and j == num_res_blocks - 1:
                    attn_list.append(_make_attention(dim_in, att_cfg, context_dim))
                else:
                    attn_list.append(nn.Identity())
                skip_channels.append(dim_in)
            level["res"] = res_list
            level["attn"] = attn_list
            if i != len(self.feature_depths) - 1:
                level["down"] = Downsample(dim_in, dim_out, scale=2, activation=activation)
                ch = dim_out
            self.down_blocks.append(level)

        # ---- middle ---------------------------------------------------------
        middle_dim = self.feature_depths[-1]
        middle_att = self.attention_configs[-1]
        self.middle_res1 = nn.ModuleList()
        self.middle_attn = nn.ModuleList()
        self.middle_res2 = nn.ModuleList()
        for j in range(num_middle_res_blocks):
            self.middle_res1.append(res_block(ch, middle_dim))
            ch = middle_dim
            if middle_att is not None and j == num_middle_res_blocks - 1:
                cfg = dict(middle_att)
                cfg.setdefault("use_self_and_cross", False)
                self.middle_attn.append(_make_attention(middle_dim, cfg, context_dim))
            else:
                self.middle_attn.append(nn.Identity())
            self.middle_res2.append(res_block(middle_dim, middle_dim))

        # ---- decoder --------------------------------------------------------
        self.up_blocks = nn.ModuleList()
        rev_depths = list(reversed(self.feature_depths))
        rev_attn = list(reversed(self.attention_configs))
        skips = list(skip_channels)  # consumed from the end (stack pop)
        for i, (dim_out, att_cfg) in enumerate(zip(rev_depths, rev_attn)):
            level = nn.ModuleDict()
            res_list = nn.ModuleList()
            attn_list = nn.ModuleList()
            for j in range(num_res_blocks):
                skip_ch = skips.pop()
                res_list.append(res_block(ch + skip_ch, dim_out))
                ch = dim_out
                if att_cfg is not None and j == num_res_blocks - 1:
                    attn_list.append(_make_attention(dim_out, att_cfg, context_dim))
                else:
                    attn_list.append(nn.Identity())
            level["res"] = res_list
            level["attn"] = attn_list
            if i != len(self.feature_depths) - 1:
                # reference quirk: Upsample features = feature_depths[-i]
                up_features = self.feature_depths[-i] if i > 0 else self.feature_depths[0]
                level["up"] = Upsample(ch, up_features, scale=2, activation=activation)
                ch = up_features
            self.up_blocks.append(level)

        # ---- output head ----------------------------------------------------
        self.conv_mid = ConvLayer(conv_type, ch, f0, (3, 3), (1, 1))
        final_skip = skips.pop()  # the conv_in activation
        assert not skips
        self.final_residual = res_block(f0 + final_skip, f0)
        if norm_groups > 0:
            self.conv_out_norm = GroupNorm(norm_groups, f0, eps=1e-5)
        else:
            self.conv_out_norm = RMSNorm(f0, eps=1e-5)
        self.conv_out = ConvLayer(conv_type, f0, output_channels, (3, 3), (1, 1))

    def forward(self, x: torch.Tensor, temb: torch.Tensor,
                textcontext: torch.Tensor) -> torch.Tensor:
        temb = self.time_embed(temb).to(x.dtype)
        temb = self.time_proj(temb)

        x = self.conv_in(x)
        downs = [x]

        for level in self.down_blocks:
            for res, attn in zip(level["res"], level["attn"]):
                x = res(x, temb)
                if not isinstance(attn, nn.Identity):
                    x = attn(x, textcontext)
                downs.append(x)
            if "down" in level:
                x = level["down"](x)

        for res1, attn, res2 in zip(self.middle_res1, self.middle_attn, self.middle_res2):
            x = res1(x, temb)
            if not isinstance(attn, nn.Identity):
                x = attn(x, textcontext)
            x = res2(x, temb)

        for level in self.up_blocks:
            for res, attn in zip(level["res"], level["attn"]):
                x = ops.cat_channels(x, downs.pop())
                x = res(x, temb)
                if not isinstance(attn, nn.Identity):
                    x = attn(x, textcontext)
            if "up" in level:
                x = level["up"](x)

        x = self.conv_mid(x)
        x = ops.cat_channels(x, downs.pop())
        x = self.final_residual(x, temb)

        if isinstance(self.conv_out_norm, GroupNorm):
            x = self.conv_out_norm(x, silu=True)
        else:
            x = self.activation(self.conv_out_norm(x))
        return self.conv_out(x)
