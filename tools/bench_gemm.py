#!/usr/bin/env python3
"""Parity + A/B: hand-written MFMA GEMM vs hipBLASLt (run on MI355X)."""
import json
import time

import torch

from flaxdiff_amd.ops import _require_ext


def run(M, K, N, check=True, reps=30):
    torch.manual_seed(0)
    ext = _require_ext()
    x = (torch.randn(M, K) * 0.5).bfloat16().cuda()
    w = (torch.randn(K, N) * 0.1).bfloat16().cuda()
    bias = torch.randn(N).float().cuda()
    dy = (torch.randn(M, N) * 0.5).bfloat16().cuda()

    out = {"MKN": [M, K, N]}
    if check:
        y = ext.gemm_fwd(x, w, bias)
        ref = (x.float() @ w.float() + bias).cuda()
        rel = (y.float() - ref).abs().max().item() / (ref.abs().max().item() + 1e-9)
        out["fwd_rel"] = round(rel, 5)
        assert rel < 3e-2, rel
        dx = ext.gemm_dx(dy, w)
        refdx = dy.float() @ w.float().t()
        reld = (dx.float() - refdx).abs().max().item() / (refdx.abs().max().item() + 1e-9)
        out["dx_rel"] = round(reld, 5)
        assert reld < 3e-2, reld

    def t(fn):
        for _ in range(3):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(reps):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / reps * 1e3

    out["fwd_ms"] = round(t(lambda: ext.gemm_fwd(x, w, bias)), 4)
    out["lib_fwd_ms"] = round(t(lambda: torch.matmul(x, w)), 4)
    out["dx_ms"] = round(t(lambda: ext.gemm_dx(dy, w)), 4)
    out["lib_dx_ms"] = round(t(lambda: torch.matmul(dy, w.t())), 4)
    print(json.dumps(out))


for i, (M, K, N) in enumerate([
        (4096, 64, 64),
        (1048576, 64, 64),       # level-0 1x1
        (262144, 128, 128),
        (65536, 256, 256),
        (65536, 768, 256),       # cross-attn K/V proj
        (16384, 512, 2048),      # GEGLU up
        (16384, 2048, 512),      # hmm K=2048
        (19712, 768, 128),       # ctx proj level-1
]):
    run(M, K, N, check=(i < 5))
