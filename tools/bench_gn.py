import time, torch
from flaxdiff_amd.ops import _require_ext
ext = _require_ext()
torch.manual_seed(0)
for (B,HW,C,G) in ((256,64,64,8),(256,32,128,8),(256,16,256,8)):
    x = (torch.randn(B, HW, HW, C) * 0.5).bfloat16().cuda()
    dy = torch.randn_like(x)
    gamma = torch.randn(C).float().cuda(); beta = torch.randn(C).float().cuda()
    y, mean, rstd = ext.gn_silu_fwd(x, gamma, beta, G, 1e-4, True)
    for _ in range(5): ext.gn_silu_bwd(dy, x, gamma, beta, mean, rstd, G, True)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(30): ext.gn_silu_bwd(dy, x, gamma, beta, mean, rstd, G, True)
    torch.cuda.synchronize()
    bw = time.perf_counter() - t0
    t0 = time.perf_counter()
    for _ in range(30): ext.gn_silu_fwd(x, gamma, beta, G, 1e-4, True)
    torch.cuda.synchronize()
    fw = time.perf_counter() - t0
    print(f"B{B} {HW}x{HW} C{C}: bwd {bw/30*1e6:.1f}us fwd {fw/30*1e6:.1f}us")
