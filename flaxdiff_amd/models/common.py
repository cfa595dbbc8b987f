"""Model building blocks (NHWC, torch modules over flaxdiff_amd.ops).

Structure contract: reference /root/reference/flaxdiff/models/common.py.
Every block keeps the reference's math (GN->SiLU->3x3 conv ResBlock with
temb bias, nearest-2x Upsample + conv, stride-2 Downsample, sinusoidal /
Fourier time embeddings) but executes through the CDNA4 kernel set: GroupNorm
and SiLU run as one fused HIP kernel, 3x3 convs as LDS-tiled implicit GEMM,
1x1 convs / Dense as MFMA library GEMMs.

Parameters are stored fp32 (master weights); forward casts to the activation
dtype, so bf16 training accumulates gradients into fp32 masters.
"""
from __future__ import annotations

import math
from typing import Callable, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops

# ---------------------------------------------------------------------------
# init helpers (flax-compatible variance scaling, truncated normal)
# ---------------------------------------------------------------------------

_TRUNC_STD_CORRECTION = 0.87962566103423978  # std of N(0,1) truncated to [-2, 2]


def variance_scaling_(tensor: torch.Tensor, fan_in: int, fan_out: int,
                      scale: float = 1.0, mode: str = "fan_in") -> torch.Tensor:
    """Truncated-normal variance scaling (flax kernel_init, common.py:13-15)."""
    if mode == "fan_in":
        denom = max(1, fan_in)
    elif mode == "fan_out":
        denom = max(1, fan_out)
    else:  # fan_avg
        denom = max(1.0, (fan_in + fan_out) / 2)
    std = math.sqrt(max(scale, 1e-10) / denom) / _TRUNC_STD_CORRECTION
    with torch.no_grad():
        tensor.normal_(0.0, std)
        tensor.clamp_(-2 * std, 2 * std)
    return tensor


class _ShadowCast(torch.autograd.Function):
    """fp32 master param -> its pre-computed bf16 shadow (zero-copy forward).

    The shadow view is refreshed by the fused Adam kernel each optimizer step
    (trainer/optim.py), so the forward is a no-op instead of a cast kernel;
    the backward casts the incoming bf16 grad to fp32 so autograd accumulates
    into the flat fp32 grad buffer.
    """

    @staticmethod
    def forward(ctx, p):
        return p._shadow_bf16

    @staticmethod
    def backward(ctx, dy):
        return dy.float()


def _cast(p: Optional[torch.Tensor], dtype) -> Optional[torch.Tensor]:
    if p is None:
        return None
    if p.dtype == dtype:
        return p
    if dtype == torch.bfloat16 and getattr(p, "_shadow_bf16", None) is not None:
        return _ShadowCast.apply(p)
    return p.to(dtype)


# ---------------------------------------------------------------------------
# Dense (flax convention: weight [in, out], y = x @ W + b)
# ---------------------------------------------------------------------------

class Dense(nn.Module):
    def __init__(self, in_features: int, out_features: int, use_bias: bool = True,
                 kernel_scale: float = 1.0, zero_init: bool = False):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        w = torch.empty(in_features, out_features)
        if zero_init:
            nn.init.zeros_(w)
        else:
            variance_scaling_(w, in_features, out_features, scale=kernel_scale)
        self.weight = nn.Parameter(w)
        self.bias = nn.Parameter(torch.zeros(out_features)) if use_bias else None

    def forward(self, x: torch.Tensor, add: torch.Tensor = None) -> torch.Tensor:
        if x.is_cuda and x.dtype == torch.bfloat16:
            # fp32 masters go straight in: ops.dense uses the bf16 shadow for
            # compute and returns fp32 grads (no per-call cast kernels)
            return ops.dense(x, self.weight, self.bias, add=add)
        return ops.dense(x, _cast(self.weight, x.dtype),
                         _cast(self.bias, x.dtype), add=add)


# ---------------------------------------------------------------------------
# Norms
# ---------------------------------------------------------------------------

class GroupNorm(nn.Module):
    """NHWC GroupNorm; `forward(x, silu=True)` runs the fused GN+SiLU kernel."""

    def __init__(self, num_groups: int, num_channels: int, eps: float = 1e-5):
        super().__init__()
        assert num_channels % num_groups == 0, (num_groups, num_channels)
        self.num_groups = num_groups
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(num_channels))
        self.bias = nn.Parameter(torch.zeros(num_channels))

    def forward(self, x: torch.Tensor, silu: bool = False) -> torch.Tensor:
        if x.is_cuda:
            # fp32 masters go straight to the HIP kernel (it consumes fp32
            # gamma/beta and emits fp32 dgamma/dbeta) — no bf16 shadow round
            # trip, no per-call cast kernels
            return ops.group_norm(x, self.num_groups, self.weight, self.bias,
                                  self.eps, silu)
        return ops.group_norm(x, self.num_groups, _cast(self.weight, x.dtype),
                              _cast(self.bias, x.dtype), self.eps, silu)


class RMSNorm(nn.Module):
    def __init__(self, dim: int, eps: float = 1e-5):
        super().__init__()
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(dim))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if x.is_cuda:
            return ops.rms_norm(x, self.weight, self.eps)
        return ops.rms_norm(x, _cast(self.weight, x.dtype), self.eps)


# ---------------------------------------------------------------------------
# Time embeddings (reference: common.py:81-124)
# ---------------------------------------------------------------------------

class TimeEmbedding(nn.Module):
    """Sinusoidal timestep embedding -> [B, features]."""

    def __init__(self, features: int, max_positions: int = 10000):
        super().__init__()
        self.features = features
        self.max_positions = max_positions

    def forward(self, t: torch.Tensor) -> torch.Tensor:
        return ops.sinusoidal_time_embedding(t, self.features, self.max_positions)


class FourierEmbedding(nn.Module):
    """Random-Fourier timestep embedding with a FIXED frequency table
    (reference fixes PRNGKey(42), scale 16 — common.py:97-108)."""

    def __init__(self, features: int, scale: float = 16.0):
        super().__init__()
        g = torch.Generator().manual_seed(42)
        freqs = torch.randn(features // 2, generator=g) * scale
        self.register_buffer("freqs", freqs, persistent=True)

    def forward(self, t: torch.Tensor) -> torch.Tensor:
        return ops.fourier_time_embedding(t, self.freqs)


class TimeProjection(nn.Module):
    """Dense -> GELU -> Dense -> GELU (common.py:110-124)."""

    def __init__(self, in_features: int, features: int):
        super().__init__()
        self.dense1 = Dense(in_features, features)
        self.dense2 = Dense(features, features)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = F.gelu(self.dense1(x))
        x = F.gelu(self.dense2(x))
        return x


# ---------------------------------------------------------------------------
# Convolutions
# ---------------------------------------------------------------------------

class Conv(nn.Module):
    """NHWC conv, HWIO weights, flax-SAME padding."""

    def __init__(self, in_features: int, features: int, kernel_size=(3, 3),
                 strides=(1, 1), use_bias: bool = True):
        super().__init__()
        kh, kw = kernel_size if isinstance(kernel_size, (tuple, list)) \
            else (kernel_size, kernel_size)
        self.stride = strides[0] if isinstance(strides, (tuple, list)) else strides
        w = torch.empty(kh, kw, in_features, features)
        variance_scaling_(w, kh * kw * in_features, kh * kw * features)
        self.weight = nn.Parameter(w)
        self.bias = nn.Parameter(torch.zeros(features)) if use_bias else None

    def forward(self, x: torch.Tensor, add: torch.Tensor = None,
                badd: torch.Tensor = None) -> torch.Tensor:
        if x.is_cuda and x.dtype == torch.bfloat16:
            return ops.conv2d(x, self.weight, self.bias, stride=self.stride,
                              add=add, badd=badd)
        return ops.conv2d(x, _cast(self.weight, x.dtype), _cast(self.bias, x.dtype),
                          stride=self.stride, add=add, badd=badd)


class ConvTranspose(nn.Module):
    def __init__(self, in_features: int, features: int, kernel_size=(3, 3), strides=(2, 2)):
        super().__init__()
        kh, kw = kernel_size
        self.stride = strides[0]
        w = torch.empty(kh, kw, in_features, features)
        variance_scaling_(w, kh * kw * in_features, kh * kw * features)
        self.weight = nn.Parameter(w)
        self.bias = nn.Parameter(torch.zeros(features))

    def forward(self, x):
        return ops.conv2d_transpose(x, _cast(self.weight, x.dtype),
                                    _cast(self.bias, x.dtype), stride=self.stride)


class SeparableConv(nn.Module):
    """Depthwise + pointwise conv (common.py:126-153)."""

    def __init__(self, in_features: int, features: int, kernel_size=(3, 3),
                 strides=(1, 1), use_bias: bool = False):
        super().__init__()
        kh, kw = kernel_size
        self.stride = strides[0] if isinstance(strides, (tuple, list)) else strides
        dw = torch.empty(kh, kw, in_features, 1)
        variance_scaling_(dw, kh * kw, kh * kw)
        self.depthwise = nn.Parameter(dw)
        pw = torch.empty(1, 1, in_features, features)
        variance_scaling_(pw, in_features, features)
        self.pointwise = nn.Parameter(pw)
        self.bias = nn.Parameter(torch.zeros(features)) if use_bias else None

    def forward(self, x):
        x = ops.depthwise_conv2d(x, _cast(self.depthwise, x.dtype), None, self.stride)
        return ops.conv2d(x, _cast(self.pointwise, x.dtype), _cast(self.bias, x.dtype), 1)


class WeightStandardizedConv(nn.Module):
    """Conv with weight standardization (common.py:18-66)."""

    def __init__(self, in_features: int, features: int, kernel_size=(3, 3), strides=(1, 1)):
        super().__init__()
        kh, kw = kernel_size
        self.stride = strides[0] if isinstance(strides, (tuple, list)) else strides
        w = torch.empty(kh, kw, in_features, features)
        variance_scaling_(w, kh * kw * in_features, kh * kw * features)
        self.weight = nn.Parameter(w)
        self.bias = nn.Parameter(torch.zeros(features))

    def forward(self, x):
        w = self.weight
        eps = 1e-5 if x.dtype == torch.float32 else 1e-3
        mean = w.mean(dim=(0, 1, 2), keepdim=True)
        var = w.var(dim=(0, 1, 2), unbiased=False, keepdim=True)
        w = (w - mean) / torch.sqrt(var + eps)
        return ops.conv2d(x, _cast(w, x.dtype), _cast(self.bias, x.dtype), self.stride)


def ConvLayer(conv_type: str, in_features: int, features: int, kernel_size=(3, 3),
              strides=(1, 1)) -> nn.Module:
    """Dispatch conv/w_conv/separable/conv_transpose (common.py:155-201)."""
    if conv_type == "conv":
        return Conv(in_features, features, kernel_size, strides)
    if conv_type == "w_conv":
        return WeightStandardizedConv(in_features, features, kernel_size, strides)
    if conv_type == "separable":
        return SeparableConv(in_features, features, kernel_size, strides)
    if conv_type == "conv_transpose":
        return ConvTranspose(in_features, features, kernel_size, strides)
    raise ValueError(f"unknown conv_type {conv_type}")


class PixelShuffle(nn.Module):
    """b h w (h2 w2 c) -> b (h h2) (w w2) c (common.py:68-79)."""

    def __init__(self, scale: int):
        super().__init__()
        self.scale = scale

    def forward(self, x):
        B, H, W, C = x.shape
        s = self.scale
        c = C // (s * s)
        x = x.reshape(B, H, W, s, s, c)
        x = x.permute(0, 1, 3, 2, 4, 5)
        return x.reshape(B, H * s, W * s, c)


class Upsample(nn.Module):
    """Nearest-2x + 3x3 conv, optional residual concat (common.py:203-226)."""

    def __init__(self, in_features: int, features: int, scale: int = 2,
                 activation: Callable = F.silu):
        super().__init__()
        assert scale == 2
        self.conv = Conv(in_features, features, (3, 3), (1, 1))

    def forward(self, x, residual=None):
        if x.is_cuda and x.dtype == torch.bfloat16:
            # upsample fused into the conv's halo staging (ops.conv2d_upsample2x)
            out = ops.conv2d_upsample2x(x, self.conv.weight, self.conv.bias)
        else:
            out = self.conv(ops.nearest_upsample_2x(x))
        if residual is not None:
            out = ops.cat_channels(out, residual)
        return out


class Downsample(nn.Module):
    """3x3 stride-2 conv, optional residual concat (common.py:228-249)."""

    def __init__(self, in_features: int, features: int, scale: int = 2,
                 activation: Callable = F.silu):
        super().__init__()
        assert scale == 2
        self.conv = Conv(in_features, features, (3, 3), (2, 2))

    def forward(self, x, residual=None):
        out = self.conv(x)
        if residual is not None:
            if residual.shape[1] > out.shape[1]:
                residual = ops.avg_pool_2x(residual)
            out = ops.cat_channels(out, residual)
        return out


# ---------------------------------------------------------------------------
# ResidualBlock — the UNet workhorse (common.py:258-337)
# ---------------------------------------------------------------------------

class ResidualBlock(nn.Module):
    """GN(+SiLU fused) -> 3x3 conv -> +temb -> GN(+SiLU) -> 3x3 conv -> +skip."""

    def __init__(self, conv_type: str, in_features: int, features: int,
                 temb_features: int, kernel_size=(3, 3), strides=(1, 1),
                 activation: Callable = F.silu, norm_groups: int = 8,
                 norm_epsilon: float = 1e-4):
        super().__init__()
        self.features = features
        self.activation = activation
        self._fused_silu = activation in (F.silu, torch.nn.functional.silu)
        if norm_groups > 0:
            self.norm1 = GroupNorm(norm_groups, in_features, eps=norm_epsilon)
            self.norm2 = GroupNorm(norm_groups, features, eps=norm_epsilon)
        else:
            self.norm1 = RMSNorm(in_features, eps=norm_epsilon)
            self.norm2 = RMSNorm(features, eps=norm_epsilon)
        self.conv1 = ConvLayer(conv_type, in_features, features, kernel_size, strides)
        self.temb_projection = Dense(temb_features, features)
        self.conv2 = ConvLayer(conv_type, features, features, kernel_size, strides)
        self.residual_conv = (ConvLayer(conv_type, in_features, features, (1, 1), (1, 1))
                              if in_features != features else None)

    def _norm_act(self, norm, x):
        if isinstance(norm, GroupNorm) and self._fused_silu:
            return norm(x, silu=True)
        return self.activation(norm(x))

    def forward(self, x: torch.Tensor, temb: torch.Tensor,
                textemb: torch.Tensor = None, extra_features: torch.Tensor = None):
        residual = x
        out = self._norm_act(self.norm1, x)
        t = self.temb_projection(temb.to(out.dtype))
        if isinstance(self.conv1, Conv):
            # temb broadcast fused into the conv epilogue (one fewer full
            # HBM read+write elementwise pass per block)
            out = self.conv1(out, badd=t)
        else:
            out = self.conv1(out) + t[:, None, None, :]

        out = self._norm_act(self.norm2, out)
        if self.residual_conv is not None:
            residual = self.residual_conv(residual)
        if isinstance(self.conv2, Conv) and residual.shape[-1] == self.features:
            # residual add fused into the conv epilogue (saves one full
            # HBM read+write elementwise pass per block on GPU)
            out = self.conv2(out, add=residual)
        else:
            out = self.conv2(out) + residual

        if extra_features is not None:
            out = ops.cat_channels(out, extra_features)
        return out


def l2norm(t: torch.Tensor, dim=1, eps=1e-6):
    return t / torch.clamp(t.norm(p=2, dim=dim, keepdim=True), min=eps)
