import torch
from flaxdiff_amd.ops import _require_ext
ext = _require_ext()
torch.manual_seed(0)
# level-0 GN shape: B=256, 64x64, C=64, groups 8
x = (torch.randn(256, 64, 64, 64) * 0.5).bfloat16().cuda()
dy = (torch.randn(256, 64, 64, 64) * 0.5).bfloat16().cuda()
gamma = torch.randn(64).float().cuda()
beta = torch.randn(64).float().cuda()
y, mean, rstd = ext.gn_silu_fwd(x, gamma, beta, 8, 1e-4, True)
torch.cuda.synchronize()
for _ in range(10):
    ext.gn_silu_bwd(dy, x, gamma, beta, mean, rstd, 8, True)
torch.cuda.synchronize()
print("done")
