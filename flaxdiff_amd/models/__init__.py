from .common import (Conv, ConvLayer, Dense, Downsample, FourierEmbedding,
                     GroupNorm, PixelShuffle, RMSNorm, ResidualBlock,
                     SeparableConv, TimeEmbedding, TimeProjection, Upsample,
                     WeightStandardizedConv)
from .attention import (BasicTransformerBlock, EfficientAttention, FeedForward,
                        GEGLU, NormalAttention, TransformerBlock)
from .unet import Unet
from .unet_3d import UNet3D
from .simple_dit import DiTBlock, SimpleDiT
from .simple_mmdit import (HierarchicalMMDiT, MMAdaLNZero, MMDiTBlock,
                           PatchExpanding, PatchMerging, SimpleMMDiT)
from .simple_vit import SimpleUDiT, UViT
from .ssm_dit import (BidirectionalS5Layer, HybridSSMAttentionDiT, S5Layer,
                      SSMDiTBlock, SpatialFusionConv)
from .vit_common import (AdaLNParams, AdaLNZero, PatchEmbedding,
                         PositionalEncoding, RoPEAttention, RotaryEmbedding,
                         apply_rotary_embedding)

__all__ = [
    "Unet", "UNet3D", "Conv", "ConvLayer", "Dense", "Downsample", "FourierEmbedding",
    "GroupNorm", "PixelShuffle", "RMSNorm", "ResidualBlock", "SeparableConv",
    "TimeEmbedding", "TimeProjection", "Upsample", "WeightStandardizedConv",
    "BasicTransformerBlock", "EfficientAttention", "FeedForward", "GEGLU",
    "NormalAttention", "TransformerBlock",
    "DiTBlock", "SimpleDiT", "SimpleUDiT", "UViT",
    "HierarchicalMMDiT", "MMAdaLNZero", "MMDiTBlock", "PatchExpanding",
    "PatchMerging", "SimpleMMDiT",
    "BidirectionalS5Layer", "HybridSSMAttentionDiT", "S5Layer", "SSMDiTBlock",
    "SpatialFusionConv",
    "AdaLNParams", "AdaLNZero", "PatchEmbedding", "PositionalEncoding",
    "RoPEAttention", "RotaryEmbedding", "apply_rotary_embedding",
]
