#!/usr/bin/env python3
"""Parity + A/B timing: attn_bwd_smallkv v1 vs v2 (run on MI355X)."""
import json
import time

import torch

from flaxdiff_amd.ops import _require_ext


def ref_bwd(q, k, v, do, scale):
    qf, kf, vf, dof = q.float(), k.float(), v.float(), do.float()
    qf.requires_grad_(True); kf.requires_grad_(True); vf.requires_grad_(True)
    s = torch.einsum("bhqd,bhkd->bhqk", qf, kf) * scale
    p = torch.softmax(s, dim=-1)
    o = torch.einsum("bhqk,bhkd->bhqd", p, vf)
    o.backward(dof)
    return qf.grad, kf.grad, vf.grad


def run(B, H, Sq, Skv, D, reps=20, check=True):
    torch.manual_seed(0)
    ext = _require_ext()
    scale = D ** -0.5
    q = (torch.randn(B, H, Sq, D) * 0.5).bfloat16().cuda()
    k = (torch.randn(B, H, Skv, D) * 0.5).bfloat16().cuda()
    v = (torch.randn(B, H, Skv, D) * 0.5).bfloat16().cuda()
    do = (torch.randn(B, H, Sq, D) * 0.5).bfloat16().cuda()
    _, lse = ext.attn_fwd(q, k, v, scale)

    out = {"shape": f"B{B} H{H} Sq{Sq} Skv{Skv} D{D}"}
    if check:
        dq2, dk2, dv2 = ext.attn_bwd_smallkv_v2(q, k, v, do, lse, scale)
        rq, rk, rv = ref_bwd(q, k, v, do, scale)
        for name, got, want in [("dq", dq2, rq), ("dk", dk2, rk), ("dv", dv2, rv)]:
            rel = (got.float().cpu() - want.cpu()).abs().max().item() / \
                  (want.abs().max().item() + 1e-9)
            out[f"{name}_rel"] = round(rel, 5)
            assert rel < 5e-2, (name, rel)

    def timeit(fn):
        for _ in range(3):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(reps):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / reps * 1e3

    if D <= 32:
        out["v1_ms"] = timeit(lambda: ext.attn_bwd_smallkv(q, k, v, do, lse, scale))
    out["v2_ms"] = timeit(lambda: ext.attn_bwd_smallkv_v2(q, k, v, do, lse, scale))
    print(json.dumps(out))


for i, (B, H, Sq, Skv, D) in enumerate([
        (2, 4, 64, 77, 16),
        (2, 4, 100, 13, 16),
        (2, 4, 128, 128, 32),
        (2, 4, 96, 64, 24),
        (2, 4, 96, 77, 64),        # D=64 parity
        (256, 4, 4096, 77, 16),    # level-0 cross (the hot shape)
        (256, 4, 1024, 77, 32),    # level-1 cross
        (256, 4, 256, 77, 64),     # level-2 cross (was the composed path)
]):
    run(B, H, Sq, Skv, D, check=(i < 5))
