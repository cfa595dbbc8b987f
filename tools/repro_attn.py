import torch
from flaxdiff_amd.ops import _require_ext
ext = _require_ext()
torch.manual_seed(0)
B,H,Sq,Skv,D = 2,4,64,77,16
scale = D ** -0.5
q = (torch.randn(B,H,Sq,D)*0.5).bfloat16().cuda()
k = (torch.randn(B,H,Skv,D)*0.5).bfloat16().cuda()
v = (torch.randn(B,H,Skv,D)*0.5).bfloat16().cuda()
do = (torch.randn(B,H,Sq,D)*0.5).bfloat16().cuda()
_, lse = ext.attn_fwd(q, k, v, scale)
print("fwd ok")
dq, dk, dv = ext.attn_bwd_smallkv_v2(q, k, v, do, lse, scale)
torch.cuda.synchronize()
print("bwd direct ok", dq.shape)
# now the script's exact order: ref_bwd torch autograd between fwd and bwd
qf = q.float().requires_grad_(True); kf = k.float().requires_grad_(True); vf = v.float().requires_grad_(True)
s = torch.einsum("bhqd,bhkd->bhqk", qf, kf) * scale
p = torch.softmax(s, dim=-1)
o = torch.einsum("bhqk,bhkd->bhqd", p, vf)
o.backward(do.float())
print("ref bwd ok")
dq, dk, dv = ext.attn_bwd_smallkv_v2(q, k, v, do, lse, scale)
torch.cuda.synchronize()
print("bwd after-ref ok")
