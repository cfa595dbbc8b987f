"""A/B the wgrad bias fold + kernel timings on the bench shape zoo."""
import sys, os; sys.path.insert(0, os.getcwd())
import os, time, json
import torch
from flaxdiff_amd.ops import _require_ext

ext = _require_ext()
shapes = [  # (B,H,W,Ci,Co,st) — 64px UNet zoo
    (256, 64, 64, 64, 64, 1),
    (256, 32, 32, 128, 128, 1),
    (256, 16, 16, 256, 256, 1),
    (256, 8, 8, 512, 512, 1),
]
def bench(fn, reps=10):
    for _ in range(3): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps * 1e3

out = {}
for (B,H,W,Ci,Co,st) in shapes:
    x = torch.randn(B,H,W,Ci, device="cuda").bfloat16()
    dy = torch.randn(B,H,W,Co, device="cuda").bfloat16()
    ms = bench(lambda: ext.conv2d_wgrad(dy, x, 3, 3, st))
    out[f"wgrad_{Ci}x{H}"] = round(ms, 3)
print(json.dumps({"mode": "bias" if "FD_WGRAD_NO_BIAS" not in os.environ else "nobias", **out}))
