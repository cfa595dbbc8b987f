import sys, os; sys.path.insert(0, os.getcwd())
import torch
from flaxdiff_amd.ops import _require_ext
ext = _require_ext()
x = torch.randn(256,64,64,64, device="cuda").bfloat16()
dy = torch.randn(256,64,64,64, device="cuda").bfloat16()
for _ in range(10):
    ext.conv2d_wgrad(dy, x, 3, 3, 1)
torch.cuda.synchronize()
print("done")
