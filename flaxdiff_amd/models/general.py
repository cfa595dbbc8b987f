"""BCHWModelWrapper — NHWC<->NCHW adapter (reference models/general.py:5-20).

The native stack is NHWC end to end; this adapter exists only to host
external NCHW models (e.g. diffusers UNets) behind the standard
model(x, temb, textcontext) signature.
"""
from __future__ import annotations

import torch
import torch.nn as nn


class BCHWModelWrapper(nn.Module):
    def __init__(self, model: nn.Module):
        super().__init__()
        self.model = model

    def forward(self, x: torch.Tensor, temb: torch.Tensor,
                textcontext: torch.Tensor = None) -> torch.Tensor:
        out = self.model(x.permute(0, 3, 1, 2), temb, textcontext)
        if hasattr(out, "sample"):  # diffusers output objects
            out = out.sample
        return out.permute(0, 2, 3, 1)
