#!/usr/bin/env python3
"""Parity + A/B timing for the conv wgrad v3 kernel (run on MI355X).

Compares dw from the v3 path (default) against the round-1 kernel
(FD_WGRAD_NO_V3=1 path is selected per-call here via env toggling is not
possible in-process, so we call the extension's two paths by shape) and
against a plain fp32 torch reference.
"""
import json
import os
import time

import torch

import flaxdiff_amd.ops as ops
from flaxdiff_amd.ops import _require_ext


def torch_ref(dy, x, stride=1):
    # NHWC bf16 -> fp32 reference dw[r,s,ci,co]
    xf = x.float().permute(0, 3, 1, 2)      # NCHW
    dyf = dy.float().permute(0, 3, 1, 2)
    dw = torch.nn.grad.conv2d_weight(
        xf, (dyf.shape[1], xf.shape[1], 3, 3), dyf, stride=1, padding=1)
    return dw.permute(2, 3, 1, 0).contiguous()   # [KH,KW,Ci,Co]


def run(B, H, W, Ci, Co, reps=20, check=True):
    torch.manual_seed(0)
    x = (torch.randn(B, H, W, Ci) * 0.5).bfloat16().cuda()
    dy = (torch.randn(B, H, W, Co) * 0.5).bfloat16().cuda()
    ext = _require_ext()

    dw3, _ = ext.conv2d_wgrad(dy, x, 3, 3, 1)
    torch.cuda.synchronize()

    out = {"shape": f"B{B} {H}x{W} Ci{Ci} Co{Co}"}
    if check:
        ref = torch_ref(dy, x).cuda()
        rel = (dw3 - ref).abs().max().item() / (ref.abs().max().item() + 1e-9)
        out["rel_err_vs_fp32ref"] = rel
        assert rel < 4e-2, f"v3 wgrad mismatch: rel={rel}"

    def timeit(fn):
        for _ in range(3):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(reps):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / reps * 1e3

    out["v3_ms"] = timeit(lambda: ext.conv2d_wgrad(dy, x, 3, 3, 1))
    # old kernel path: force ineligibility is env-based at first call; instead
    # time it via the dedicated env in a subprocess-free way: the old kernel
    # is still reachable for Ci%64!=0 shapes, so time the same shape through
    # FD_WGRAD_NO_V3 only when set before import (bench harness does that).
    if os.environ.get("FD_WGRAD_NO_V3"):
        out["path"] = "v2(old)"
    else:
        out["path"] = "v3"
    print(json.dumps(out))
    return out


def main():
    shapes = [
        (8, 64, 64, 64, 64),
        (8, 32, 32, 128, 128),
        (8, 16, 16, 256, 256),
        (8, 16, 16, 512, 512),
        (8, 8, 8, 512, 512),
        (256, 64, 64, 64, 64),     # bench level-0 shape
        (256, 32, 32, 128, 128),   # bench level-1
        (256, 16, 16, 256, 256),   # level-2
        (256, 8, 8, 512, 512),     # level-3 (W=8: falls back to old kernel)
    ]
    for i, (B, H, W, Ci, Co) in enumerate(shapes):
        run(B, H, W, Ci, Co, check=(i < 5))


if __name__ == "__main__":
    main()
