"""Image eval metrics: CLIP similarity / CLIPScore, PSNR, SSIM.

Behavior contract: reference /root/reference/flaxdiff/metrics/images.py
(_get_clip cache :16-29, cosine helper :32-46, legacy clip_similarity
1-cos lower-better :49-77, canonical CLIPScore 100*max(cos,0) higher-better
:80-112). The reference declared psnr.py/ssim.py but left them EMPTY
 — implemented properly here.

CLIP runs as a frozen HF transformers PyTorch model on ROCm (not a perf
path). Weight download needs network; in offline environments pass a local
checkpoint path as `modelname`.
"""
from __future__ import annotations


import torch
import torch.nn.functional as F

from .common import EvaluationMetric

_clip_cache: dict = {}


def _get_clip(modelname: str):
    """Cached (model, processor); loaded once per process (reference :16-29)."""
    if modelname not in _clip_cache:
        from transformers import AutoProcessor, CLIPModel
        model = CLIPModel.from_pretrained(modelname, torch_dtype=torch.float16)
        model.eval()
        if torch.cuda.is_available():
            model = model.cuda()
        processor = AutoProcessor.from_pretrained(modelname, use_fast=False)
        _clip_cache[modelname] = (model, processor)
    return _clip_cache[modelname]


@torch.no_grad()
def _clip_cosine(model, pixel_values, input_ids, attention_mask) -> torch.Tensor:
    dev = next(model.parameters()).device
    out = model(pixel_values=pixel_values.to(dev),
                input_ids=input_ids.to(dev),
                attention_mask=attention_mask.to(dev))
    img = F.normalize(out.image_embeds.float(), dim=-1, eps=1e-6)
    txt = F.normalize(out.text_embeds.float(), dim=-1, eps=1e-6)
    return (img * txt).sum(-1)


def _prep_images(generated: torch.Tensor, processor):
    imgs = (((generated.float() + 1.0) / 2.0) * 255).clamp(0, 255).byte()
    return processor(images=[im.cpu().numpy() for im in imgs],
                     return_tensors="pt", padding=True)["pixel_values"]


def get_clip_metric(modelname: str = "openai/clip-vit-large-patch14") -> EvaluationMetric:
    """Legacy CLIP distance: mean(1 - cos). LOWER is better (reference :49-77)."""
    model, processor = _get_clip(modelname)

    def clip_metric(generated, batch):
        text = batch["text"]
        pv = _prep_images(generated, processor)
        cos = _clip_cosine(model, pv, text["input_ids"], text["attention_mask"])
        return float((1.0 - cos).mean())

    return EvaluationMetric(function=clip_metric, name="clip_similarity")


def get_clip_score_metric(modelname: str = "openai/clip-vit-large-patch14") -> EvaluationMetric:
    """Canonical CLIPScore: mean(100*max(cos,0)). HIGHER is better (:80-112)."""
    model, processor = _get_clip(modelname)

    def clip_score_metric(generated, batch):
        text = batch["text"]
        pv = _prep_images(generated, processor)
        cos = _clip_cosine(model, pv, text["input_ids"], text["attention_mask"])
        return float((100.0 * cos.clamp(min=0)).mean())

    return EvaluationMetric(function=clip_score_metric, name="clip_score",
                            higher_is_better=True)


# ---------------------------------------------------------------------------
# PSNR / SSIM — declared-but-empty in the reference (psnr.py/ssim.py, 0 LoC)
# ---------------------------------------------------------------------------

def psnr(a: torch.Tensor, b: torch.Tensor, data_range: float = 2.0) -> torch.Tensor:
    """Peak signal-to-noise ratio; default range 2.0 for [-1,1] images."""
    mse = ((a.float() - b.float()) ** 2).mean(dim=tuple(range(1, a.dim())))
    return 10.0 * torch.log10(data_range ** 2 / mse.clamp_min(1e-12))


def _gaussian_kernel(size: int = 11, sigma: float = 1.5) -> torch.Tensor:
    x = torch.arange(size, dtype=torch.float32) - (size - 1) / 2
    g = torch.exp(-x ** 2 / (2 * sigma ** 2))
    g = g / g.sum()
    return torch.outer(g, g)


def ssim(a: torch.Tensor, b: torch.Tensor, data_range: float = 2.0,
         window_size: int = 11, sigma: float = 1.5) -> torch.Tensor:
    """Mean SSIM per sample over NHWC images (Wang et al. 2004)."""
    C1 = (0.01 * data_range) ** 2
    C2 = (0.03 * data_range) ** 2
    x = a.float().permute(0, 3, 1, 2)
    y = b.float().permute(0, 3, 1, 2)
    C = x.shape[1]
    w = _gaussian_kernel(window_size, sigma).to(x.device)
    w = w.expand(C, 1, window_size, window_size).contiguous()
    pad = window_size // 2

    def filt(t):
        return F.conv2d(t, w, padding=pad, groups=C)

    mu_x, mu_y = filt(x), filt(y)
    mu_x2, mu_y2, mu_xy = mu_x * mu_x, mu_y * mu_y, mu_x * mu_y
    sig_x = filt(x * x) - mu_x2
    sig_y = filt(y * y) - mu_y2
    sig_xy = filt(x * y) - mu_xy
    s = ((2 * mu_xy + C1) * (2 * sig_xy + C2)) / \
        ((mu_x2 + mu_y2 + C1) * (sig_x + sig_y + C2))
    return s.mean(dim=(1, 2, 3))


def get_psnr_metric() -> EvaluationMetric:
    def fn(generated, batch):
        ref = batch["image"]
        if ref.dtype == torch.uint8:
            ref = ref.float() / 127.5 - 1.0
        return float(psnr(generated, ref.to(generated.device)).mean())
    return EvaluationMetric(function=fn, name="psnr", higher_is_better=True)


def get_ssim_metric() -> EvaluationMetric:
    def fn(generated, batch):
        ref = batch["image"]
        if ref.dtype == torch.uint8:
            ref = ref.float() / 127.5 - 1.0
        return float(ssim(generated, ref.to(generated.device)).mean())
    return EvaluationMetric(function=fn, name="ssim", higher_is_better=True)
