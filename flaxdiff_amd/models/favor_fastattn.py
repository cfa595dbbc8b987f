"""FAVOR+ linear attention (Performer).

Behavior contract: reference /root/reference/flaxdiff/models/favor_fastattn.py
(vendored google-research code: make_fast_softmax_attention :206,
make_fast_generalized_attention :268, GaussianOrthogonalRandomMatrix :341,
prefix-sum numerator/denominator, FastAttentionviaLowRankDecomposition :527).
Unused in the reference's main path — provided for API
parity as torch modules.

Math (Choromanski et al., ICLR 2021): softmax kernel approximated by
positive random features
    phi(x) = h(x)/sqrt(m) * exp(W x - ||x||^2 / 2),
attention(Q,K,V) ~= phi(Q) [phi(K)^T V] / (phi(Q) [phi(K)^T 1]) — O(n) in
sequence length. Non-causal path is two GEMMs (MFMA library GEMMs on ROCm);
causal path is a prefix-sum over the sequence.
"""
from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn


def gaussian_orthogonal_random_matrix(nb_rows: int, nb_cols: int,
                                      scaling: int = 0,
                                      generator: Optional[torch.Generator] = None
                                      ) -> torch.Tensor:
    """Orthogonal Gaussian feature matrix (reference :341-400)."""
    nb_full = nb_rows // nb_cols
    blocks = []
    for _ in range(nb_full):
        g = torch.randn(nb_cols, nb_cols, generator=generator)
        q, _ = torch.linalg.qr(g)
        blocks.append(q.t())
    rem = nb_rows - nb_full * nb_cols
    if rem > 0:
        g = torch.randn(nb_cols, nb_cols, generator=generator)
        q, _ = torch.linalg.qr(g)
        blocks.append(q.t()[:rem])
    W = torch.cat(blocks, dim=0)
    if scaling == 0:  # row norms ~ chi(d)
        norms = torch.randn(nb_rows, nb_cols, generator=generator).norm(dim=1)
    elif scaling == 1:
        norms = torch.full((nb_rows,), math.sqrt(nb_cols))
    else:
        raise ValueError(f"invalid scaling {scaling}")
    return W * norms.unsqueeze(1)


def softmax_kernel_features(x: torch.Tensor, projection: torch.Tensor,
                            is_query: bool, eps: float = 1e-4) -> torch.Tensor:
    """Positive random features of the softmax kernel (FAVOR+).

    x: [..., S, D]; projection: [M, D] -> [..., S, M]."""
    d = x.shape[-1]
    m = projection.shape[0]
    x = x * (d ** -0.25)
    wx = torch.einsum("...sd,md->...sm", x, projection.to(x.dtype))
    norm = (x ** 2).sum(-1, keepdim=True) / 2
    if is_query:  # stabilize per-row
        z = wx - norm - wx.amax(dim=-1, keepdim=True)
    else:         # stabilize globally over the sequence
        z = wx - norm - wx.amax(dim=(-2, -1), keepdim=True)
    return (torch.exp(z) + eps) / math.sqrt(m)


def relu_kernel_features(x: torch.Tensor, projection: torch.Tensor,
                         is_query: bool, eps: float = 1e-4) -> torch.Tensor:
    """Generalized attention with ReLU features (reference :268-340)."""
    m = projection.shape[0]
    wx = torch.einsum("...sd,md->...sm", x, projection.to(x.dtype))
    return torch.relu(wx) / math.sqrt(m) + eps


def _noncausal_attention(qp, kp, v):
    kv = torch.einsum("...sm,...sd->...md", kp, v)
    z = 1.0 / (torch.einsum("...sm,...m->...s", qp,
                            kp.sum(dim=-2)) + 1e-6)
    return torch.einsum("...sm,...md,...s->...sd", qp, kv, z)


def _causal_attention(qp, kp, v):
    """Prefix-sum numerator/denominator (the reference's lax.scan :450-520)."""
    kv = torch.einsum("...sm,...sd->...smd", kp, v).cumsum(dim=-3)
    ks = kp.cumsum(dim=-2)
    num = torch.einsum("...sm,...smd->...sd", qp, kv)
    den = torch.einsum("...sm,...sm->...s", qp, ks) + 1e-6
    return num / den.unsqueeze(-1)


class FastAttention(nn.Module):
    """Performer attention over [B, H, S, D] tensors
    (FastAttentionviaLowRankDecomposition, reference :527-716)."""

    def __init__(self, dim_head: int, nb_features: Optional[int] = None,
                 causal: bool = False, kernel: str = "softmax",
                 ortho_scaling: int = 0, redraw: bool = False, seed: int = 0):
        super().__init__()
        self.dim_head = dim_head
        self.nb_features = nb_features or int(dim_head * math.log(max(dim_head, 2)))
        self.causal = causal
        self.kernel = kernel
        self.ortho_scaling = ortho_scaling
        self.redraw = redraw
        g = torch.Generator().manual_seed(seed)
        self.register_buffer(
            "projection",
            gaussian_orthogonal_random_matrix(self.nb_features, dim_head,
                                              ortho_scaling, g),
            persistent=True)

    def redraw_projection(self, generator: Optional[torch.Generator] = None):
        self.projection.copy_(gaussian_orthogonal_random_matrix(
            self.nb_features, self.dim_head, self.ortho_scaling, generator))

    def forward(self, q: torch.Tensor, k: torch.Tensor, v: torch.Tensor
                ) -> torch.Tensor:
        feat = softmax_kernel_features if self.kernel == "softmax" \
            else relu_kernel_features
        qp = feat(q.float(), self.projection, is_query=True)
        # causal: keys must be stabilized per-row (a global max over the
        # sequence would leak future keys into past outputs)
        kp = feat(k.float(), self.projection, is_query=self.causal)
        out = (_causal_attention if self.causal else _noncausal_attention)(
            qp, kp, v.float())
        return out.to(v.dtype)


def make_fast_softmax_attention(dim_head: int, nb_features: Optional[int] = None,
                                causal: bool = False, **kw) -> FastAttention:
    """reference :206-265."""
    return FastAttention(dim_head, nb_features, causal, kernel="softmax", **kw)


def make_fast_generalized_attention(dim_head: int,
                                    nb_features: Optional[int] = None,
                                    causal: bool = False, **kw) -> FastAttention:
    """reference :268-340 (ReLU generalized attention)."""
    return FastAttention(dim_head, nb_features, causal, kernel="relu", **kw)
