"""Autoencoders for latent diffusion.

Behavior contract: reference /root/reference/flaxdiff/models/autoencoder/
(AutoEncoder ABC with 5-D video reshape handling, autoencoder.py:11-120;
StableDiffusionVAE wrapper around the diffusers Flax VAE with reparameterized
sampling, downscale-factor probing and scaling factor, diffusers.py:13-110;
SimpleAutoEncoder stub, simple_autoenc.py:25-57).

MI355X design: instead of wrapping diffusers (not a dependency here), the SD
VAE ARCHITECTURE (encoder 128-256-512-512 with mid attention, KL head,
decoder mirror) is implemented natively on this framework's NHWC kernel
stack — GroupNorm+SiLU fused kernel, LDS-tiled implicit-GEMM convs, the
flash-attention kernel — so frozen VAE encode/decode runs on the same HIP
path as the UNet (BASELINE config 5). Pretrained SD weights load from a
local diffusers-format safetensors/pt file via `load_diffusers_weights`
(no network in the target environment).
"""
from __future__ import annotations

import abc
from typing import Optional, Sequence

import torch
import torch.nn as nn

from .attention import NormalAttention
from .common import Conv, GroupNorm
from .. import ops


class AutoEncoder(abc.ABC):
    """encode/decode with [B,T,H,W,C] video folding (reference :48-118)."""

    @abc.abstractmethod
    def __encode__(self, x: torch.Tensor, key=None, **kw) -> torch.Tensor: ...

    @abc.abstractmethod
    def __decode__(self, z: torch.Tensor, key=None, **kw) -> torch.Tensor: ...

    @property
    def downscale_factor(self) -> int:
        return self.__downscale_factor__

    @property
    def latent_channels(self) -> int:
        return self.__latent_channels__

    def encode(self, x: torch.Tensor, key=None, **kw) -> torch.Tensor:
        if x.dim() == 5:
            B, T = x.shape[:2]
            z = self.__encode__(x.reshape(-1, *x.shape[2:]), key=key, **kw)
            return z.reshape(B, T, *z.shape[1:])
        return self.__encode__(x, key=key, **kw)

    def decode(self, z: torch.Tensor, key=None, **kw) -> torch.Tensor:
        if z.dim() == 5:
            B, T = z.shape[:2]
            x = self.__decode__(z.reshape(-1, *z.shape[2:]), key=key, **kw)
            return x.reshape(B, T, *x.shape[1:])
        return self.__decode__(z, key=key, **kw)

    def __call__(self, x: torch.Tensor, key=None, **kw) -> torch.Tensor:
        return self.decode(self.encode(x, key=key, **kw), **kw)


# ---------------------------------------------------------------------------
# native SD-VAE building blocks (NHWC, our kernel stack)
# ---------------------------------------------------------------------------

class VAEResnetBlock(nn.Module):
    """GN(32)+SiLU -> 3x3 -> GN+SiLU -> 3x3 (+1x1 shortcut)."""

    def __init__(self, cin: int, cout: int, groups: int = 32):
        super().__init__()
        self.norm1 = GroupNorm(groups, cin, eps=1e-6)
        self.conv1 = Conv(cin, cout, (3, 3), (1, 1))
        self.norm2 = GroupNorm(groups, cout, eps=1e-6)
        self.conv2 = Conv(cout, cout, (3, 3), (1, 1))
        self.shortcut = Conv(cin, cout, (1, 1), (1, 1)) if cin != cout else None

    def forward(self, x):
        h = self.conv1(self.norm1(x, silu=True))
        h = self.conv2(self.norm2(h, silu=True))
        skip = self.shortcut(x) if self.shortcut is not None else x
        return h + skip


class VAEAttnBlock(nn.Module):
    """Single-head spatial self-attention on the mid block."""

    def __init__(self, channels: int, groups: int = 32):
        super().__init__()
        self.norm = GroupNorm(groups, channels, eps=1e-6)
        self.attn = NormalAttention(channels, heads=1, dim_head=channels,
                                    use_bias=True)

    def forward(self, x):
        return x + self.attn(self.norm(x))


class VAEDownsample(nn.Module):
    def __init__(self, channels: int):
        super().__init__()
        self.conv = Conv(channels, channels, (3, 3), (2, 2))

    def forward(self, x):
        return self.conv(x)


class VAEUpsample(nn.Module):
    def __init__(self, channels: int):
        super().__init__()
        self.conv = Conv(channels, channels, (3, 3), (1, 1))

    def forward(self, x):
        return self.conv(ops.nearest_upsample_2x(x.contiguous()))


class VAEEncoder(nn.Module):
    def __init__(self, in_channels=3, latent_channels=4,
                 block_out_channels: Sequence[int] = (128, 256, 512, 512),
                 layers_per_block=2, groups=32, double_z=True):
        super().__init__()
        self.conv_in = Conv(in_channels, block_out_channels[0], (3, 3), (1, 1))
        blocks = []
        ch = block_out_channels[0]
        for i, cout in enumerate(block_out_channels):
            stage = nn.ModuleDict()
            res = nn.ModuleList()
            for _ in range(layers_per_block):
                res.append(VAEResnetBlock(ch, cout, groups))
                ch = cout
            stage["res"] = res
            if i < len(block_out_channels) - 1:
                stage["down"] = VAEDownsample(ch)
            blocks.append(stage)
        self.down_blocks = nn.ModuleList(blocks)
        self.mid_res1 = VAEResnetBlock(ch, ch, groups)
        self.mid_attn = VAEAttnBlock(ch, groups)
        self.mid_res2 = VAEResnetBlock(ch, ch, groups)
        self.norm_out = GroupNorm(groups, ch, eps=1e-6)
        out_ch = latent_channels * (2 if double_z else 1)
        self.conv_out = Conv(ch, out_ch, (3, 3), (1, 1))

    def forward(self, x):
        x = self.conv_in(x)
        for stage in self.down_blocks:
            for res in stage["res"]:
                x = res(x)
            if "down" in stage:
                x = stage["down"](x)
        x = self.mid_res2(self.mid_attn(self.mid_res1(x)))
        return self.conv_out(self.norm_out(x, silu=True))


class VAEDecoder(nn.Module):
    def __init__(self, latent_channels=4, out_channels=3,
                 block_out_channels: Sequence[int] = (128, 256, 512, 512),
                 layers_per_block=2, groups=32):
        super().__init__()
        rev = list(reversed(block_out_channels))   # (512, 512, 256, 128)
        ch = rev[0]
        self.conv_in = Conv(latent_channels, ch, (3, 3), (1, 1))
        self.mid_res1 = VAEResnetBlock(ch, ch, groups)
        self.mid_attn = VAEAttnBlock(ch, groups)
        self.mid_res2 = VAEResnetBlock(ch, ch, groups)
        blocks = []
        for i, cout in enumerate(rev):
            stage = nn.ModuleDict()
            res = nn.ModuleList()
            for _ in range(layers_per_block + 1):  # decoder has 3 resnets/stage
                res.append(VAEResnetBlock(ch, cout, groups))
                ch = cout
            stage["res"] = res
            if i < len(rev) - 1:
                stage["up"] = VAEUpsample(ch)
            blocks.append(stage)
        self.up_blocks = nn.ModuleList(blocks)
        self.norm_out = GroupNorm(groups, ch, eps=1e-6)
        self.conv_out = Conv(ch, out_channels, (3, 3), (1, 1))

    def forward(self, z):
        x = self.conv_in(z)
        x = self.mid_res2(self.mid_attn(self.mid_res1(x)))
        for stage in self.up_blocks:
            for res in stage["res"]:
                x = res(x)
            if "up" in stage:
                x = stage["up"](x)
        return self.conv_out(self.norm_out(x, silu=True))


class StableDiffusionVAE(AutoEncoder):
    """SD-VAE running on the native NHWC HIP kernel stack.

    weights_path: optional local diffusers-format state dict
    (safetensors or .pt). scaling_factor matches SD v1 (0.18215).
    """

    def __init__(self, weights_path: Optional[str] = None,
                 scaling_factor: float = 0.18215, latent_channels: int = 4,
                 block_out_channels: Sequence[int] = (128, 256, 512, 512),
                 device: Optional[str] = None, dtype: torch.dtype = torch.float32,
                 modelname: str = "", revision: str = ""):
        self.modelname = modelname
        self.revision = revision
        self.scaling_factor = scaling_factor
        self.dtype = dtype
        self.encoder = VAEEncoder(3, latent_channels, block_out_channels)
        self.decoder = VAEDecoder(latent_channels, 3, block_out_channels)
        self.quant_conv = Conv(2 * latent_channels, 2 * latent_channels, (1, 1))
        self.post_quant_conv = Conv(latent_channels, latent_channels, (1, 1))
        if weights_path:
            load_diffusers_weights(self, weights_path)
        dev = device or ("cuda" if torch.cuda.is_available() else "cpu")
        for m in (self.encoder, self.decoder, self.quant_conv, self.post_quant_conv):
            m.to(dev).eval()
            for p in m.parameters():
                p.requires_grad_(False)
        self.__downscale_factor__ = 2 ** (len(block_out_channels) - 1)
        self.__latent_channels__ = latent_channels

    @torch.no_grad()
    def __encode__(self, images: torch.Tensor, key=None, **kw) -> torch.Tensor:
        x = images.to(next(self.encoder.parameters()).device, self.dtype)
        h = self.quant_conv(self.encoder(x))
        mean, log_std = h.chunk(2, dim=-1)
        if key is not None:
            log_std = log_std.clamp(-30, 20)
            std = torch.exp(0.5 * log_std)
            if hasattr(key, "normal"):
                noise = key.normal(mean.shape, device=mean.device).to(mean.dtype)
            else:
                noise = torch.randn_like(mean)
            latents = mean + std * noise
        else:
            latents = mean
        return latents * self.scaling_factor

    @torch.no_grad()
    def __decode__(self, latents: torch.Tensor, key=None, **kw) -> torch.Tensor:
        z = latents.to(next(self.decoder.parameters()).device, self.dtype)
        z = z / self.scaling_factor
        return self.decoder(self.post_quant_conv(z))

    def serialize(self) -> dict:
        return {"type": "stable_diffusion", "scaling_factor": self.scaling_factor,
                "modelname": self.modelname}


class SimpleAutoEncoder(AutoEncoder):
    """Trainable conv autoencoder. The reference's SimpleAutoEncoder is an
    UNIMPLEMENTED stub returning zeros (simple_autoenc.py:25-57); this one is
    functional but keeps the same name/role."""

    def __init__(self, latent_channels: int = 4, feature_depths=(64, 128),
                 device: Optional[str] = None):
        self.encoder = VAEEncoder(3, latent_channels, feature_depths,
                                  layers_per_block=1, groups=8)
        self.decoder = VAEDecoder(latent_channels, 3, feature_depths,
                                  layers_per_block=1, groups=8)
        self.quant_conv = Conv(2 * latent_channels, 2 * latent_channels, (1, 1))
        self.post_quant_conv = Conv(latent_channels, latent_channels, (1, 1))
        dev = device or "cpu"
        for m in (self.encoder, self.decoder, self.quant_conv, self.post_quant_conv):
            m.to(dev)
        self.__downscale_factor__ = 2 ** (len(feature_depths) - 1)
        self.__latent_channels__ = latent_channels

    def to(self, *args, **kwargs) -> "SimpleAutoEncoder":
        """Move/cast the wrapped modules (not an nn.Module itself — the
        AutoEncoder interface is functional, reference autoencoder.py:11)."""
        for m in (self.encoder, self.decoder, self.quant_conv,
                  self.post_quant_conv):
            m.to(*args, **kwargs)
        return self

    def parameters(self):
        for m in (self.encoder, self.decoder, self.quant_conv,
                  self.post_quant_conv):
            yield from m.parameters()

    def __encode__(self, x, key=None, **kw):
        h = self.quant_conv(self.encoder(x))
        mean, _ = h.chunk(2, dim=-1)
        return mean

    def __decode__(self, z, key=None, **kw):
        return self.decoder(self.post_quant_conv(z))


def load_diffusers_weights(vae: StableDiffusionVAE, path: str):
    """Load a diffusers AutoencoderKL state dict (NCHW OIHW) into the native
    NHWC/HWIO modules. Accepts .safetensors or torch .pt/.bin."""
    if path.endswith(".safetensors"):
        from safetensors.torch import load_file
        sd = load_file(path)
    else:
        sd = torch.load(path, map_location="cpu")

    def put_conv(mod: Conv, prefix: str):
        w = sd[prefix + ".weight"]            # [O, I, KH, KW]
        mod.weight.data.copy_(w.permute(2, 3, 1, 0))  # -> [KH, KW, I, O]
        if mod.bias is not None and prefix + ".bias" in sd:
            mod.bias.data.copy_(sd[prefix + ".bias"])

    def put_gn(mod: GroupNorm, prefix: str):
        mod.weight.data.copy_(sd[prefix + ".weight"])
        mod.bias.data.copy_(sd[prefix + ".bias"])

    def put_resnet(block: VAEResnetBlock, prefix: str):
        put_gn(block.norm1, prefix + ".norm1")
        put_conv(block.conv1, prefix + ".conv1")
        put_gn(block.norm2, prefix + ".norm2")
        put_conv(block.conv2, prefix + ".conv2")
        if block.shortcut is not None:
            put_conv(block.shortcut, prefix + ".conv_shortcut")

    def put_attn(block: VAEAttnBlock, prefix: str):
        put_gn(block.norm, prefix + ".group_norm")
        for ours, theirs in (("to_q", "to_q"), ("to_k", "to_k"),
                             ("to_v", "to_v")):
            lin = getattr(block.attn, ours)
            lin.weight.data.copy_(sd[f"{prefix}.{theirs}.weight"].t())
            lin.bias.data.copy_(sd[f"{prefix}.{theirs}.bias"])
        block.attn.to_out.weight.data.copy_(sd[prefix + ".to_out.0.weight"].t())
        block.attn.to_out.bias.data.copy_(sd[prefix + ".to_out.0.bias"])

    # encoder
    e = vae.encoder
    put_conv(e.conv_in, "encoder.conv_in")
    for i, stage in enumerate(e.down_blocks):
        for j, res in enumerate(stage["res"]):
            put_resnet(res, f"encoder.down_blocks.{i}.resnets.{j}")
        if "down" in stage:
            put_conv(stage["down"].conv,
                     f"encoder.down_blocks.{i}.downsamplers.0.conv")
    put_resnet(e.mid_res1, "encoder.mid_block.resnets.0")
    put_attn(e.mid_attn, "encoder.mid_block.attentions.0")
    put_resnet(e.mid_res2, "encoder.mid_block.resnets.1")
    put_gn(e.norm_out, "encoder.conv_norm_out")
    put_conv(e.conv_out, "encoder.conv_out")
    # decoder
    d = vae.decoder
    put_conv(d.conv_in, "decoder.conv_in")
    put_resnet(d.mid_res1, "decoder.mid_block.resnets.0")
    put_attn(d.mid_attn, "decoder.mid_block.attentions.0")
    put_resnet(d.mid_res2, "decoder.mid_block.resnets.1")
    for i, stage in enumerate(d.up_blocks):
        for j, res in enumerate(stage["res"]):
            put_resnet(res, f"decoder.up_blocks.{i}.resnets.{j}")
        if "up" in stage:
            put_conv(stage["up"].conv,
                     f"decoder.up_blocks.{i}.upsamplers.0.conv")
    put_gn(d.norm_out, "decoder.conv_norm_out")
    put_conv(d.conv_out, "decoder.conv_out")
    put_conv(vae.quant_conv, "quant_conv")
    put_conv(vae.post_quant_conv, "post_quant_conv")


def get_autoencoder(name: str, **opts) -> AutoEncoder:
    if name in ("stable_diffusion", "sd", "sd_vae"):
        return StableDiffusionVAE(**opts)
    if name == "simple":
        return SimpleAutoEncoder(**opts)
    raise ValueError(f"unknown autoencoder {name!r}")
