"""FID — Fréchet Inception Distance.

Behavior contract: reference /root/reference/flaxdiff/metrics/inception.py
(full InceptionV3 port for FID :21-657, pretrained-pickle loading, pool3
features). Here: the canonical pool3 InceptionV3 (FID variant) as native
torch modules (NCHW — eval-only path, not a perf path), plus the Fréchet
distance math and a streaming accumulator. Pretrained weights load from a
local state-dict path (`weights_path`) since the target environment has no
network; any callable feature extractor is also accepted (e.g. CLIP image
tower), matching common FID-variant practice.
"""
from __future__ import annotations

from typing import Callable, Optional

import numpy as np
import torch
import torch.nn as nn
import torch.nn.functional as F


def frechet_distance(mu1: np.ndarray, sigma1: np.ndarray,
                     mu2: np.ndarray, sigma2: np.ndarray,
                     eps: float = 1e-6) -> float:
    """||mu1-mu2||^2 + Tr(S1 + S2 - 2 sqrt(S1 S2)) (Heusel et al. 2017)."""
    import scipy.linalg
    diff = mu1 - mu2
    covmean, _ = scipy.linalg.sqrtm(sigma1 @ sigma2, disp=False)
    if not np.isfinite(covmean).all():
        offset = np.eye(sigma1.shape[0]) * eps
        covmean = scipy.linalg.sqrtm((sigma1 + offset) @ (sigma2 + offset))
    if np.iscomplexobj(covmean):
        covmean = covmean.real
    return float(diff @ diff + np.trace(sigma1) + np.trace(sigma2)
                 - 2 * np.trace(covmean))


# ---------------------------------------------------------------------------
# InceptionV3 (FID pool3 variant)
# ---------------------------------------------------------------------------

class _ConvBN(nn.Module):
    def __init__(self, cin, cout, **kw):
        super().__init__()
        self.conv = nn.Conv2d(cin, cout, bias=False, **kw)
        self.bn = nn.BatchNorm2d(cout, eps=1e-3)

    def forward(self, x):
        return F.relu(self.bn(self.conv(x)), inplace=True)


class _InceptionA(nn.Module):
    def __init__(self, cin, pool_features):
        super().__init__()
        self.branch1x1 = _ConvBN(cin, 64, kernel_size=1)
        self.branch5x5_1 = _ConvBN(cin, 48, kernel_size=1)
        self.branch5x5_2 = _ConvBN(48, 64, kernel_size=5, padding=2)
        self.branch3x3dbl_1 = _ConvBN(cin, 64, kernel_size=1)
        self.branch3x3dbl_2 = _ConvBN(64, 96, kernel_size=3, padding=1)
        self.branch3x3dbl_3 = _ConvBN(96, 96, kernel_size=3, padding=1)
        self.branch_pool = _ConvBN(cin, pool_features, kernel_size=1)

    def forward(self, x):
        b1 = self.branch1x1(x)
        b5 = self.branch5x5_2(self.branch5x5_1(x))
        b3 = self.branch3x3dbl_3(self.branch3x3dbl_2(self.branch3x3dbl_1(x)))
        bp = self.branch_pool(F.avg_pool2d(x, 3, 1, 1))
        return torch.cat([b1, b5, b3, bp], 1)


class _InceptionB(nn.Module):
    def __init__(self, cin):
        super().__init__()
        self.branch3x3 = _ConvBN(cin, 384, kernel_size=3, stride=2)
        self.branch3x3dbl_1 = _ConvBN(cin, 64, kernel_size=1)
        self.branch3x3dbl_2 = _ConvBN(64, 96, kernel_size=3, padding=1)
        self.branch3x3dbl_3 = _ConvBN(96, 96, kernel_size=3, stride=2)

    def forward(self, x):
        b3 = self.branch3x3(x)
        bd = self.branch3x3dbl_3(self.branch3x3dbl_2(self.branch3x3dbl_1(x)))
        bp = F.max_pool2d(x, 3, 2)
        return torch.cat([b3, bd, bp], 1)


class _InceptionC(nn.Module):
    def __init__(self, cin, c7):
        super().__init__()
        self.branch1x1 = _ConvBN(cin, 192, kernel_size=1)
        self.branch7x7_1 = _ConvBN(cin, c7, kernel_size=1)
        self.branch7x7_2 = _ConvBN(c7, c7, kernel_size=(1, 7), padding=(0, 3))
        self.branch7x7_3 = _ConvBN(c7, 192, kernel_size=(7, 1), padding=(3, 0))
        self.branch7x7dbl_1 = _ConvBN(cin, c7, kernel_size=1)
        self.branch7x7dbl_2 = _ConvBN(c7, c7, kernel_size=(7, 1), padding=(3, 0))
        self.branch7x7dbl_3 = _ConvBN(c7, c7, kernel_size=(1, 7), padding=(0, 3))
        self.branch7x7dbl_4 = _ConvBN(c7, c7, kernel_size=(7, 1), padding=(3, 0))
        self.branch7x7dbl_5 = _ConvBN(c7, 192, kernel_size=(1, 7), padding=(0, 3))
        self.branch_pool = _ConvBN(cin, 192, kernel_size=1)

    def forward(self, x):
        b1 = self.branch1x1(x)
        b7 = self.branch7x7_3(self.branch7x7_2(self.branch7x7_1(x)))
        bd = self.branch7x7dbl_5(self.branch7x7dbl_4(self.branch7x7dbl_3(
            self.branch7x7dbl_2(self.branch7x7dbl_1(x)))))
        bp = self.branch_pool(F.avg_pool2d(x, 3, 1, 1))
        return torch.cat([b1, b7, bd, bp], 1)


class _InceptionD(nn.Module):
    def __init__(self, cin):
        super().__init__()
        self.branch3x3_1 = _ConvBN(cin, 192, kernel_size=1)
        self.branch3x3_2 = _ConvBN(192, 320, kernel_size=3, stride=2)
        self.branch7x7x3_1 = _ConvBN(cin, 192, kernel_size=1)
        self.branch7x7x3_2 = _ConvBN(192, 192, kernel_size=(1, 7), padding=(0, 3))
        self.branch7x7x3_3 = _ConvBN(192, 192, kernel_size=(7, 1), padding=(3, 0))
        self.branch7x7x3_4 = _ConvBN(192, 192, kernel_size=3, stride=2)

    def forward(self, x):
        b3 = self.branch3x3_2(self.branch3x3_1(x))
        b7 = self.branch7x7x3_4(self.branch7x7x3_3(self.branch7x7x3_2(
            self.branch7x7x3_1(x))))
        bp = F.max_pool2d(x, 3, 2)
        return torch.cat([b3, b7, bp], 1)


class _InceptionE(nn.Module):
    def __init__(self, cin, pool: str = "avg"):
        super().__init__()
        self.pool = pool
        self.branch1x1 = _ConvBN(cin, 320, kernel_size=1)
        self.branch3x3_1 = _ConvBN(cin, 384, kernel_size=1)
        self.branch3x3_2a = _ConvBN(384, 384, kernel_size=(1, 3), padding=(0, 1))
        self.branch3x3_2b = _ConvBN(384, 384, kernel_size=(3, 1), padding=(1, 0))
        self.branch3x3dbl_1 = _ConvBN(cin, 448, kernel_size=1)
        self.branch3x3dbl_2 = _ConvBN(448, 384, kernel_size=3, padding=1)
        self.branch3x3dbl_3a = _ConvBN(384, 384, kernel_size=(1, 3), padding=(0, 1))
        self.branch3x3dbl_3b = _ConvBN(384, 384, kernel_size=(3, 1), padding=(1, 0))
        self.branch_pool = _ConvBN(cin, 192, kernel_size=1)

    def forward(self, x):
        b1 = self.branch1x1(x)
        b3 = self.branch3x3_1(x)
        b3 = torch.cat([self.branch3x3_2a(b3), self.branch3x3_2b(b3)], 1)
        bd = self.branch3x3dbl_2(self.branch3x3dbl_1(x))
        bd = torch.cat([self.branch3x3dbl_3a(bd), self.branch3x3dbl_3b(bd)], 1)
        if self.pool == "max":  # FID variant uses max-pool in the last E block
            bp = F.max_pool2d(x, 3, 1, 1)
        else:
            bp = F.avg_pool2d(x, 3, 1, 1)
        bp = self.branch_pool(bp)
        return torch.cat([b1, b3, bd, bp], 1)


class InceptionV3Features(nn.Module):
    """InceptionV3 up to pool3 (2048-dim), the canonical FID feature net."""

    def __init__(self, weights_path: Optional[str] = None):
        super().__init__()
        self.Conv2d_1a_3x3 = _ConvBN(3, 32, kernel_size=3, stride=2)
        self.Conv2d_2a_3x3 = _ConvBN(32, 32, kernel_size=3)
        self.Conv2d_2b_3x3 = _ConvBN(32, 64, kernel_size=3, padding=1)
        self.Conv2d_3b_1x1 = _ConvBN(64, 80, kernel_size=1)
        self.Conv2d_4a_3x3 = _ConvBN(80, 192, kernel_size=3)
        self.Mixed_5b = _InceptionA(192, 32)
        self.Mixed_5c = _InceptionA(256, 64)
        self.Mixed_5d = _InceptionA(288, 64)
        self.Mixed_6a = _InceptionB(288)
        self.Mixed_6b = _InceptionC(768, 128)
        self.Mixed_6c = _InceptionC(768, 160)
        self.Mixed_6d = _InceptionC(768, 160)
        self.Mixed_6e = _InceptionC(768, 192)
        self.Mixed_7a = _InceptionD(768)
        self.Mixed_7b = _InceptionE(1280, pool="avg")
        self.Mixed_7c = _InceptionE(2048, pool="max")
        if weights_path:
            sd = torch.load(weights_path, map_location="cpu")
            self.load_state_dict(sd, strict=False)
        self.eval()

    @torch.no_grad()
    def forward(self, x: torch.Tensor) -> torch.Tensor:
        """x: NHWC float in [-1,1] -> [B, 2048] pool3 features."""
        x = x.permute(0, 3, 1, 2).float()
        if x.shape[-1] != 299 or x.shape[-2] != 299:
            x = F.interpolate(x, size=(299, 299), mode="bilinear",
                              align_corners=False)
        x = self.Conv2d_1a_3x3(x)
        x = self.Conv2d_2a_3x3(x)
        x = self.Conv2d_2b_3x3(x)
        x = F.max_pool2d(x, 3, 2)
        x = self.Conv2d_3b_1x1(x)
        x = self.Conv2d_4a_3x3(x)
        x = F.max_pool2d(x, 3, 2)
        for blk in (self.Mixed_5b, self.Mixed_5c, self.Mixed_5d, self.Mixed_6a,
                    self.Mixed_6b, self.Mixed_6c, self.Mixed_6d, self.Mixed_6e,
                    self.Mixed_7a, self.Mixed_7b, self.Mixed_7c):
            x = blk(x)
        x = F.adaptive_avg_pool2d(x, 1)
        return x.flatten(1)


class FrechetInceptionDistance:
    """Streaming FID accumulator over two image sets.

    feature_fn: images NHWC in [-1,1] -> [B, D] features. Defaults to the
    pool3 InceptionV3 above. In DP runs each rank accumulates its shard and
    `reduce_across_ranks()` all-reduces the sufficient statistics (the
    reference instead pmean-synced InceptionV3 BN stats, inception.py:552).
    """

    def __init__(self, feature_fn: Optional[Callable] = None,
                 weights_path: Optional[str] = None):
        self.feature_fn = feature_fn or InceptionV3Features(weights_path)
        self.reset()

    def reset(self):
        self._sums = [np.zeros(0), np.zeros(0)]
        self._outers = [np.zeros((0, 0)), np.zeros((0, 0))]
        self._counts = [0, 0]

    def _update(self, which: int, images: torch.Tensor):
        feats = self.feature_fn(images)
        f = feats.detach().cpu().double().numpy()
        if self._counts[which] == 0:
            d = f.shape[1]
            self._sums[which] = np.zeros(d)
            self._outers[which] = np.zeros((d, d))
        self._sums[which] += f.sum(0)
        self._outers[which] += f.T @ f
        self._counts[which] += f.shape[0]

    def update_real(self, images: torch.Tensor):
        self._update(0, images)

    def update_fake(self, images: torch.Tensor):
        self._update(1, images)

    def reduce_across_ranks(self):
        import torch.distributed as dist
        if not (dist.is_available() and dist.is_initialized()):
            return
        for which in (0, 1):
            t = torch.from_numpy(self._sums[which])
            o = torch.from_numpy(self._outers[which])
            c = torch.tensor([self._counts[which]], dtype=torch.float64)
            for x in (t, o, c):
                dist.all_reduce(x)
            self._sums[which] = t.numpy()
            self._outers[which] = o.numpy()
            self._counts[which] = int(c.item())

    def compute(self) -> float:
        assert min(self._counts) >= 2, "need >=2 samples per distribution"
        stats = []
        for which in (0, 1):
            n = self._counts[which]
            mu = self._sums[which] / n
            sigma = (self._outers[which] - n * np.outer(mu, mu)) / (n - 1)
            stats.append((mu, sigma))
        return frechet_distance(stats[0][0], stats[0][1], stats[1][0], stats[1][1])
