#!/usr/bin/env python3
"""Run the level-0 halo fwd shape repeatedly for rocprofv3 capture."""
import torch
from flaxdiff_amd.ops import _require_ext

ext = _require_ext()
torch.manual_seed(0)
B, HW, Ci, Co = 256, 64, 64, 64
x = (torch.randn(B, HW, HW, Ci) * 0.5).bfloat16().cuda()
w = (torch.randn(3, 3, Ci, Co) * 0.1).bfloat16().cuda()
bias = torch.randn(Co).float().cuda()
for _ in range(3):
    ext.conv2d_fwd(x, w, bias, 1, torch.Tensor(), torch.Tensor())
torch.cuda.synchronize()
for _ in range(10):
    ext.conv2d_fwd(x, w, bias, 1, torch.Tensor(), torch.Tensor())
torch.cuda.synchronize()
print("done")
