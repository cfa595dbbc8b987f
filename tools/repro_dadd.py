import torch
from flaxdiff_amd import ops

dev = "cuda"
torch.manual_seed(11)
B,H,W,Ci,Co = 2,64,64,64,64
x = (torch.randn(B, H, W, Ci) * 0.5).bfloat16().to(dev)
w = (torch.randn(3, 3, Ci, Co) / (9 * Ci) ** 0.5).bfloat16().to(dev)
bias = (torch.randn(Co) * 0.1).bfloat16().to(dev)
res = torch.randn(B, H, W, Co).bfloat16().to(dev)

xa = x.clone().requires_grad_(True)
wa = w.clone().requires_grad_(True)
ra = res.clone().requires_grad_(True)
ya = ops.conv2d(xa, wa, bias, stride=1, add=ra)
dy = torch.randn_like(ya)
ya.backward(dy)
torch.cuda.synchronize()
print("ra.grad is dy storage:", ra.grad.data_ptr() == dy.data_ptr())
print("ra.grad[0,0,0,:6]:", ra.grad[0,0,0,:6].float().tolist())
print("dy     [0,0,0,:6]:", dy[0,0,0,:6].float().tolist())
print("wa.grad[0,0,0,:6]:", wa.grad[0,0,0,:6].float().tolist())
print("equal ra dy:", torch.equal(ra.grad, dy))
print("max diff:", (ra.grad.float()-dy.float()).abs().max().item())
nz = (ra.grad.float()-dy.float()).abs().flatten()
idx = torch.nonzero(nz > 0.1).flatten()
print("num bad:", idx.numel(), "of", nz.numel(), "first idx:", idx[:5].tolist())
