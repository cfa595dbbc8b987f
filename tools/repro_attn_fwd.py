import torch
from flaxdiff_amd.ops import _require_ext
ext = _require_ext()

def probe(B,H,Sq,Skv,D):
    torch.manual_seed(0)
    q = (torch.randn(B,H,Sq,D)*0.3).bfloat16().cuda()
    k = (torch.randn(B,H,Skv,D)*0.3).bfloat16().cuda()
    v = (torch.randn(B,H,Skv,D)*0.3).bfloat16().cuda()
    scale = D ** -0.5
    ref = torch.nn.functional.scaled_dot_product_attention(
        q.float(), k.float(), v.float(), scale=scale)
    o, lse = ext.attn_fwd(q, k, v, scale)
    lse_ref = (torch.einsum('bhqd,bhkd->bhqk', q.float(), k.float())*scale).logsumexp(-1)
    err = (o.float()-ref).abs().amax(dim=(0,1,3))
    lerr = (lse-lse_ref).abs().amax(dim=(0,1))
    groups = [(g, float(err[g:g+64].max()), float(lerr[g:g+64].max()))
              for g in range(0, Sq, 64)]
    bad = [f"q{g//64}:{e:.3f}/{l:.3f}" for g,e,l in groups if l > 1e-3 or e > 2e-3]
    print(f"Sq={Sq:4d} Skv={Skv:4d} D={D:3d}: {'OK' if not bad else ' '.join(bad)}")

probe(2,4,256,77,16)
probe(2,4,128,77,16)
probe(2,4,64,77,16)
probe(2,4,256,64,16)
probe(2,4,256,128,16)
probe(2,4,256,13,16)
probe(2,4,256,77,32)
probe(1,1,256,77,16)
