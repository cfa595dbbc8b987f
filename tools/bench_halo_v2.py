#!/usr/bin/env python3
"""Parity + timing for the halo conv v2 kernel (fwd + dgrad) on MI355X."""
import json
import os
import time

import torch

from flaxdiff_amd.ops import _require_ext


def run(B, HW, Ci, Co, reps=20, check=True):
    torch.manual_seed(0)
    ext = _require_ext()
    x = (torch.randn(B, HW, HW, Ci) * 0.5).bfloat16().cuda()
    w = (torch.randn(3, 3, Ci, Co) * 0.1).bfloat16()
    bias = torch.randn(Co).float()
    wT = w.permute(0, 1, 3, 2).contiguous().cuda()
    wc = w.contiguous().cuda()
    bias_c = bias.cuda()
    dy = (torch.randn(B, HW, HW, Co) * 0.5).bfloat16().cuda()

    out = {"shape": f"B{B} {HW}x{HW} Ci{Ci} Co{Co}",
           "path": "v1" if os.environ.get("FD_HALO_V1") else "v2"}
    if check:
        xf = x.float().permute(0, 3, 1, 2).cpu()
        wf = w.float().permute(3, 2, 0, 1)
        ref = torch.nn.functional.conv2d(xf, wf, bias, 1, 1)
        ref = ref.permute(0, 2, 3, 1)
        y = ext.conv2d_fwd(x, wc, bias_c, 1, torch.Tensor(), torch.Tensor())
        rel = (y.float().cpu() - ref).abs().max().item() / (ref.abs().max().item() + 1e-9)
        out["fwd_rel"] = rel
        assert rel < 4e-2, rel
        dyf = dy.float().permute(0, 3, 1, 2).cpu()
        refdx = torch.nn.grad.conv2d_input(
            (B, Ci, HW, HW), wf, dyf, stride=1, padding=1).permute(0, 2, 3, 1)
        dx = ext.conv2d_dgrad(dy, wc, 1, HW, HW)
        reld = (dx.float().cpu() - refdx).abs().max().item() / (refdx.abs().max().item() + 1e-9)
        out["dgrad_rel"] = reld
        assert reld < 4e-2, reld

    def timeit(fn):
        for _ in range(3):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(reps):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / reps * 1e3

    out["fwd_ms"] = timeit(lambda: ext.conv2d_fwd(x, wc, bias_c, 1, torch.Tensor(), torch.Tensor()))
    out["dgrad_ms"] = timeit(lambda: ext.conv2d_dgrad(dy, wc, 1, HW, HW))
    print(json.dumps(out))


def main():
    for i, (B, HW, Ci, Co) in enumerate([
        (4, 64, 64, 64),
        (4, 32, 128, 128),
        (4, 16, 256, 256),
        (4, 16, 512, 512),
        (256, 64, 64, 64),
        (256, 32, 128, 128),
        (256, 16, 256, 256),
        (256, 8, 512, 512),
    ]):
        run(B, HW, Ci, Co, check=(i < 4))


if __name__ == "__main__":
    main()
