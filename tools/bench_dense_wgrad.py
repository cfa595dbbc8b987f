#!/usr/bin/env python3
"""Parity + timing for dense_wgrad_v3 (run on MI355X)."""
import json
import os
import time

import torch

from flaxdiff_amd.ops import _require_ext


def run(M, Ci, Co, reps=30, check=True):
    torch.manual_seed(0)
    ext = _require_ext()
    x = (torch.randn(M, 1, 1, Ci) * 0.5).bfloat16().cuda()
    dy = (torch.randn(M, 1, 1, Co) * 0.5).bfloat16().cuda()
    dw, _ = ext.conv2d_wgrad(dy, x, 1, 1, 1)
    out = {"shape": f"M{M} Ci{Ci} Co{Co}",
           "path": "old" if os.environ.get("FD_WGRAD_NO_V3") else "v3"}
    if check:
        ref = x.reshape(M, Ci).float().t() @ dy.reshape(M, Co).float()
        rel = (dw.reshape(Ci, Co) - ref).abs().max().item() / (ref.abs().max().item() + 1e-9)
        out["rel"] = rel
        assert rel < 1e-2, rel

    def timeit(fn):
        for _ in range(3):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(reps):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / reps * 1e3

    out["ms"] = timeit(lambda: ext.conv2d_wgrad(dy, x, 1, 1, 1))
    print(json.dumps(out))


for i, (M, Ci, Co) in enumerate([
        (1048576, 64, 64),      # level-0 sized 1x1
        (262144, 128, 128),
        (65536, 256, 256),
        (65536, 256, 768),      # cross-attn K/V proj-ish
        (16384, 512, 512),
        (16384, 512, 2048),     # GEGLU FF
]):
    run(M, Ci, Co, check=(i < 4))
