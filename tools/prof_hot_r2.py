#!/usr/bin/env python3
"""Exercise the round-2 hot kernels at bench-representative shapes for a
rocprofv3 --pmc stall decomposition (SQ_WAVE_CYCLES / SQ_WAIT_ANY / ...)."""
import torch
from flaxdiff_amd.ops import _require_ext

ext = _require_ext()
torch.manual_seed(0)
dev = "cuda"
E = torch.Tensor()

def reps(fn, n=30):
    fn()
    torch.cuda.synchronize()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()

# halo v2 fwd/dgrad — level-1 (32x32, C128) and level-2 (16x16, C256)
for HW, C in ((32, 128), (16, 256)):
    x = (torch.randn(256, HW, HW, C) * 0.5).bfloat16().to(dev)
    w = (torch.randn(3, 3, C, C) * 0.05).bfloat16().to(dev)
    wT = w.permute(0, 1, 3, 2).contiguous()
    bias = torch.randn(C).float().to(dev)
    dy = (torch.randn(256, HW, HW, C) * 0.5).bfloat16().to(dev)
    reps(lambda: ext.conv2d_fwd(x, w, bias, 1, E, E))
    reps(lambda: ext.conv2d_dgrad(dy, w, 1, HW, HW))

# gemm_nt — attention proj shape at level-2 (M=B*S, K=N=256) and FF (K=256,N=2048)
for M, K, N in ((256 * 256, 256, 256), (256 * 256, 256, 2048)):
    xg = (torch.randn(M, K) * 0.5).bfloat16().to(dev)
    wt = (torch.randn(N, K) * 0.05).bfloat16().to(dev)
    bg = torch.randn(N).float().to(dev)
    reps(lambda: ext.gemm_nt(xg, wt, bg, E), 20)

# gn3 bwd — level-0 (the biggest GN tensor)
xn = (torch.randn(256, 64, 64, 64) * 0.5).bfloat16().to(dev)
dyn = (torch.randn(256, 64, 64, 64) * 0.5).bfloat16().to(dev)
gamma = torch.randn(64).float().to(dev)
beta = torch.randn(64).float().to(dev)
y, mean, rstd = ext.gn_silu_fwd(xn, gamma, beta, 8, 1e-4, True)
torch.cuda.synchronize()
reps(lambda: ext.gn_silu_bwd(dyn, xn, gamma, beta, mean, rstd, 8, True))

# attention fwd — level-2 self-attention (B=256, h=4, S=256, D=64)
q = (torch.randn(256, 4, 256, 64) * 0.3).bfloat16().to(dev)
reps(lambda: ext.attn_fwd(q, q, q, 0.125), 20)
print("done")
