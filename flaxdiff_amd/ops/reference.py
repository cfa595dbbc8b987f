"""Pure-PyTorch reference implementations of every custom op.

These are the *oracle*: numerics tests compare each HIP kernel against these
run in fp32 (SURVEY.md §4 test strategy). They are also the CPU execution
path, so the whole framework runs end-to-end without a GPU.

All image tensors are NHWC [B, H, W, C]; conv weights are HWIO
[kh, kw, Cin, Cout] (matches the implicit-GEMM B-panel layout the CDNA4
kernels consume).
"""
from __future__ import annotations

import torch
import torch.nn.functional as F


# -- convolution -------------------------------------------------------------

def conv2d_nhwc(x: torch.Tensor, w: torch.Tensor, b: torch.Tensor | None = None,
                stride: int = 1, padding: int | str = "same") -> torch.Tensor:
    """NHWC conv with HWIO weights. padding='same' matches flax SAME semantics.

    Reference semantics: flax nn.Conv (models/common.py:155-201).
    """
    kh, kw, ci, co = w.shape
    xc = x.permute(0, 3, 1, 2).contiguous()          # NCHW
    wc = w.permute(3, 2, 0, 1).contiguous()          # OIHW
    if padding == "same":
        # flax SAME: total pad = k - 1 for stride 1; for stride s, pads so
        # out = ceil(in / s). torch supports 'same' only for stride 1.
        if stride == 1:
            pt = (kh - 1) // 2
            pl = (kw - 1) // 2
            pb = kh - 1 - pt
            pr = kw - 1 - pl
            xc = F.pad(xc, (pl, pr, pt, pb))
            out = F.conv2d(xc, wc, b, stride=1)
        else:
            ih, iw = xc.shape[2], xc.shape[3]
            oh = -(-ih // stride)
            ow = -(-iw // stride)
            pad_h = max((oh - 1) * stride + kh - ih, 0)
            pad_w = max((ow - 1) * stride + kw - iw, 0)
            pt, pb = pad_h // 2, pad_h - pad_h // 2
            pl, pr = pad_w // 2, pad_w - pad_w // 2
            xc = F.pad(xc, (pl, pr, pt, pb))
            out = F.conv2d(xc, wc, b, stride=stride)
    else:
        out = F.conv2d(xc, wc, b, stride=stride, padding=padding)
    return out.permute(0, 2, 3, 1).contiguous()


def conv2d_transpose_nhwc(x: torch.Tensor, w: torch.Tensor, b: torch.Tensor | None = None,
                          stride: int = 2) -> torch.Tensor:
    """NHWC transposed conv (flax nn.ConvTranspose SAME)."""
    kh, kw, ci, co = w.shape
    xc = x.permute(0, 3, 1, 2).contiguous()
    # torch convT weight layout: (in, out, kh, kw)
    wc = w.permute(2, 3, 0, 1).contiguous()
    out = F.conv_transpose2d(xc, wc, b, stride=stride)
    # flax SAME convT output = in * stride; crop symmetric
    target_h = x.shape[1] * stride
    target_w = x.shape[2] * stride
    eh = out.shape[2] - target_h
    ew = out.shape[3] - target_w
    if eh > 0 or ew > 0:
        t = eh // 2
        l = ew // 2
        out = out[:, :, t:t + target_h, l:l + target_w]
    return out.permute(0, 2, 3, 1).contiguous()


def depthwise_conv2d_nhwc(x: torch.Tensor, w: torch.Tensor, b: torch.Tensor | None = None,
                          stride: int = 1, dilation: int = 1) -> torch.Tensor:
    """Depthwise NHWC conv; w is [kh, kw, C, 1] (SeparableConv, common.py:126-153).
    Symmetric torch-style padding ((k-1)*dilation)//2 per side."""
    kh, kw, c = w.shape[0], w.shape[1], w.shape[2]
    xc = x.permute(0, 3, 1, 2).contiguous()
    wc = w.reshape(kh, kw, c).permute(2, 0, 1).unsqueeze(1).contiguous()  # (C,1,kh,kw)
    pt = ((kh - 1) * dilation) // 2
    pl = ((kw - 1) * dilation) // 2
    xc = F.pad(xc, (pl, (kw - 1) * dilation - pl, pt, (kh - 1) * dilation - pt))
    out = F.conv2d(xc, wc, b, stride=stride, groups=c, dilation=dilation)
    return out.permute(0, 2, 3, 1).contiguous()


# -- normalization -----------------------------------------------------------

def group_norm_nhwc(x: torch.Tensor, num_groups: int, gamma: torch.Tensor,
                    beta: torch.Tensor, eps: float = 1e-5,
                    silu: bool = False) -> torch.Tensor:
    """GroupNorm over an NHWC tensor, optional fused SiLU epilogue.

    Matches flax nn.GroupNorm (channels grouped contiguously, normalized over
    spatial dims and the channels of each group). Reference use:
    models/common.py:275,288 (GN -> SiLU pairs in ResidualBlock).
    """
    B = x.shape[0]
    C = x.shape[-1]
    spatial = x.shape[1:-1]
    xg = x.reshape(B, -1, num_groups, C // num_groups)
    mean = xg.mean(dim=(1, 3), keepdim=True)
    var = xg.var(dim=(1, 3), unbiased=False, keepdim=True)
    xn = (xg - mean) / torch.sqrt(var + eps)
    xn = xn.reshape(B, *spatial, C)
    out = xn * gamma + beta
    if silu:
        out = F.silu(out)
    return out


def rms_norm(x: torch.Tensor, gamma: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    """RMSNorm over the last dim (flax nn.RMSNorm semantics)."""
    dtype = x.dtype
    x32 = x.float()
    rrms = torch.rsqrt(x32.pow(2).mean(dim=-1, keepdim=True) + eps)
    return (x32 * rrms).to(dtype) * gamma


def layer_norm(x: torch.Tensor, gamma: torch.Tensor, beta: torch.Tensor,
               eps: float = 1e-5) -> torch.Tensor:
    return F.layer_norm(x, (x.shape[-1],), gamma, beta, eps)


# -- attention ---------------------------------------------------------------

def attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
              scale: float | None = None) -> torch.Tensor:
    """Softmax attention, fp32 softmax. q,k,v: [B, H, S, D] -> [B, H, Sq, D].

    Reference: nn.dot_product_attention with force_fp32_for_softmax
    (models/attention.py:156-177).
    """
    d = q.shape[-1]
    if scale is None:
        scale = d ** -0.5
    s = torch.einsum("bhqd,bhkd->bhqk", q.float(), k.float()) * scale
    p = torch.softmax(s, dim=-1)
    return torch.einsum("bhqk,bhkd->bhqd", p, v.float()).to(q.dtype)


# -- embeddings --------------------------------------------------------------

def sinusoidal_time_embedding(t: torch.Tensor, features: int,
                              max_positions: int = 10000) -> torch.Tensor:
    """[B] -> [B, features]: concat(sin, cos) (models/common.py:81-95)."""
    import math
    half_dim = features // 2
    f = math.log(max_positions) / (half_dim - 1)
    freqs = torch.exp(-f * torch.arange(half_dim, dtype=torch.float32, device=t.device))
    emb = t.float()[:, None] * freqs[None, :]
    return torch.cat([torch.sin(emb), torch.cos(emb)], dim=-1)


def fourier_time_embedding(t: torch.Tensor, freqs: torch.Tensor) -> torch.Tensor:
    """[B], freqs [F/2] -> [B, F]: concat(sin, cos) of 2*pi*f*t (common.py:97-108)."""
    import math
    emb = t.float()[:, None] * (2 * math.pi * freqs)[None, :]
    return torch.cat([torch.sin(emb), torch.cos(emb)], dim=-1)


# -- elementwise fusions -----------------------------------------------------

def forward_diffusion(x0: torch.Tensor, eps: torch.Tensor, signal_rate: torch.Tensor,
                      noise_rate: torch.Tensor):
    """x_t = a*x0 + s*eps (predictors/__init__.py:19-24); rates broadcast [B,1,1,1]."""
    return signal_rate.to(x0.dtype) * x0 + noise_rate.to(x0.dtype) * eps


def nearest_upsample_2x_nhwc(x: torch.Tensor) -> torch.Tensor:
    """Nearest 2x upsample of NHWC (models/common.py:211-215)."""
    B, H, W, C = x.shape
    return (x.reshape(B, H, 1, W, 1, C)
             .expand(B, H, 2, W, 2, C)
             .reshape(B, 2 * H, 2 * W, C).contiguous())


def avg_pool_2x_nhwc(x: torch.Tensor) -> torch.Tensor:
    """2x2 stride-2 avg pool NHWC (models/common.py:246-248)."""
    xc = x.permute(0, 3, 1, 2)
    out = F.avg_pool2d(xc, 2, 2)
    return out.permute(0, 2, 3, 1).contiguous()
