import sys, os; sys.path.insert(0, os.getcwd())
"""Per-shape conv fwd/dgrad/wgrad timings on the 64px UNet zoo."""
import time, json
import torch
from flaxdiff_amd.ops import _require_ext

ext = _require_ext()
shapes = [  # B,H,W,Ci,Co,st
    (256, 64, 64, 64, 64, 1),
    (256, 32, 32, 128, 128, 1),
    (256, 16, 16, 256, 256, 1),
    (256, 8, 8, 512, 512, 1),
    (256, 16, 16, 768, 512, 1),   # decoder concat
    (256, 32, 32, 64, 128, 2),    # downsample
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
    w = (torch.randn(3,3,Ci,Co, device="cuda") * 0.05).bfloat16()
    wT = w.permute(0,1,3,2).contiguous()
    OH, OW = H//st, W//st
    dy = torch.randn(B,OH,OW,Co, device="cuda").bfloat16()
    key = f"{Ci}->{Co}@{H}s{st}"
    fwd = bench(lambda: ext.conv2d_fwd(x, w, torch.Tensor(), st))
    dgr = bench(lambda: ext.conv2d_dgrad(dy, w, st, H, W))
    wgr = bench(lambda: ext.conv2d_wgrad(dy, x, 3, 3, st))
    # speed-of-light @ 2.5 PF
    fl = 2*B*OH*OW*Ci*Co*9
    out[key] = {"fwd_ms": round(fwd,3), "dgrad_ms": round(dgr,3),
                "wgrad_ms": round(wgr,3),
                "tflops_fwd": round(fl/fwd/1e9,1),
                "pct_peak": round(fl/fwd/1e9/2500*100,1)}
print(json.dumps(out, indent=1))
