import torch
from flaxdiff_amd.ops import _require_ext
ext = _require_ext()
torch.manual_seed(0)
B,H,Sq,Skv,D = 256,4,4096,77,16
scale = D ** -0.5
q = (torch.randn(B,H,Sq,D)*0.5).bfloat16().cuda()
k = (torch.randn(B,H,Skv,D)*0.5).bfloat16().cuda()
v = (torch.randn(B,H,Skv,D)*0.5).bfloat16().cuda()
do = (torch.randn(B,H,Sq,D)*0.5).bfloat16().cuda()
_, lse = ext.attn_fwd(q, k, v, scale)
for _ in range(3):
    ext.attn_bwd_smallkv_v2(q, k, v, do, lse, scale)
torch.cuda.synchronize()
for _ in range(5):
    ext.attn_bwd_smallkv_v2(q, k, v, do, lse, scale)
torch.cuda.synchronize()
print("done")
