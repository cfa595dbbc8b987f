from .common import EvaluationMetric
from .images import (get_clip_metric, get_clip_score_metric, get_psnr_metric,
                     get_ssim_metric, psnr, ssim)
from .fid import FrechetInceptionDistance, frechet_distance

__all__ = [
    "EvaluationMetric", "get_clip_metric", "get_clip_score_metric",
    "get_psnr_metric", "get_ssim_metric", "psnr", "ssim",
    "FrechetInceptionDistance", "frechet_distance",
]
