from .common import (Conv, ConvLayer, Dense, Downsample, FourierEmbedding,
                     GroupNorm, PixelShuffle, RMSNorm, ResidualBlock,
                     SeparableConv, TimeEmbedding, TimeProjection, Upsample,
                     WeightStandardizedConv)
from .attention import (BasicTransformerBlock, EfficientAttention, FeedForward,
                        GEGLU, NormalAttention, TransformerBlock)
from .unet import Unet

__all__ = [
    "Unet", "Conv", "ConvLayer", "Dense", "Downsample", "FourierEmbedding",
    "GroupNorm", "PixelShuffle", "RMSNorm", "ResidualBlock", "SeparableConv",
    "TimeEmbedding", "TimeProjection", "Upsample", "WeightStandardizedConv",
    "BasicTransformerBlock", "EfficientAttention", "FeedForward", "GEGLU",
    "NormalAttention", "TransformerBlock",
]
