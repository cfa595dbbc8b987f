import sys, os; sys.path.insert(0, os.getcwd())
"""A/B: fused flash attn bwd vs composed GEMM path, bench shapes."""
import time, json
import torch
from flaxdiff_amd.ops import _require_ext

ext = _require_ext()
shapes = [  # B,H,Sq,Skv,D — 64px UNet levels 0,1 + middle
    (256, 4, 4096, 77, 16),
    (256, 4, 1024, 77, 32),
    (256, 4, 64, 64, 32),
]
def bench(fn, reps=5):
    for _ in range(2): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps * 1e3

out = {}
for (B,H,Sq,Skv,D) in shapes:
    q = (torch.randn(B,H,Sq,D, device="cuda")*0.5).bfloat16()
    k = (torch.randn(B,H,Skv,D, device="cuda")*0.5).bfloat16()
    v = (torch.randn(B,H,Skv,D, device="cuda")*0.5).bfloat16()
    do = (torch.randn(B,H,Sq,D, device="cuda")*0.5).bfloat16()
    scale = D ** -0.5
    o, lse = ext.attn_fwd(q, k, v, scale)

    fused_ms = bench(lambda: ext.attn_bwd_smallkv(q, k, v, do, lse, scale))

    def composed():
        qf, kf, vf, dof = q.float(), k.float(), v.float(), do.float()
        s = torch.einsum("bhqd,bhkd->bhqk", qf, kf) * scale
        p = torch.exp(s - lse.unsqueeze(-1))
        dv = torch.einsum("bhqk,bhqd->bhkd", p, dof)
        dp = torch.einsum("bhqd,bhkd->bhqk", dof, vf)
        dsum = (dp * p).sum(-1, keepdim=True)
        ds = (dp - dsum) * p * scale
        dq = torch.einsum("bhqk,bhkd->bhqd", ds, kf)
        dk = torch.einsum("bhqk,bhqd->bhkd", ds, qf)
        return dq, dk, dv
    comp_ms = bench(composed)
    # parity check vs composed
    dq, dk, dv = ext.attn_bwd_smallkv(q, k, v, do, lse, scale)
    cdq, cdk, cdv = composed()
    errs = [float((a.float()-b).abs().max()/(b.abs().max()+1e-6))
            for a, b in ((dq,cdq),(dk,cdk),(dv,cdv))]
    out[f"S{Sq}_D{D}"] = {"fused_ms": round(fused_ms,3),
                          "composed_ms": round(comp_ms,3),
                          "rel_err": [round(e,4) for e in errs]}
print(json.dumps(out))
