"""SSM-DiT: S5 diagonal state-space blocks + hybrid SSM/attention DiT.

Behavior contract: reference /root/reference/flaxdiff/models/ssm_dit.py
(HiPPO init :37-55, S5Layer :58-221 with ZOH discretization :161-169 and
associative scan :196-200, BidirectionalS5Layer :225-286, SpatialFusionConv
:293-350, SSMDiTBlock :357-540, HybridSSMAttentionDiT :545-779).

Math preserved exactly:
  * A_real = -exp(log_A_real) with HiPPO-diag init log(n+0.5); A_imag = pi*n;
  * ZOH: A_bar = exp(A*dt), B_bar = ((A_bar-1)/(A+1e-8)) * B, per-state dt
    ~ logU(dt_min, dt_max);
  * scan x_k = A_bar x_{k-1} + B u_k via the associative operator
    (a1,b1)*(a2,b2) = (a1 a2, a2 b1 + b2);
  * y = Re(C x) + D u, fp32 scan regardless of compute dtype.

MI355X notes: the scan here is a log-depth Hillis-Steele doubling over
[B,S,N] complex64 tensors — log2(S) fused complex-FMA sweeps, HBM-bound and
fine at the <=4k token lengths this model family uses. A single-kernel LDS
Blelchley scan is a possible follow-up, flagged in SURVEY.md §2.3.
"""
from __future__ import annotations

import math
from typing import Optional, Sequence, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops
from .common import Dense, FourierEmbedding, TimeProjection
from .hilbert import (build_2d_sincos_pos_embed, hilbert_indices,
                      hilbert_patchify, hilbert_unpatchify, inverse_permutation,
                      zigzag_indices, zigzag_patchify)
from .simple_dit import DiTBlock
from .vit_common import (AdaLNParams, PatchEmbedding, RotaryEmbedding, norm_fp32,
                         layer_norm_noaffine, unpatchify)


def associative_scan_diag(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """Inclusive scan of x_k = a_k * x_{k-1} + b_k along dim 1.

    a, b: [B, S, N] (complex). Hillis-Steele doubling: log2(S) vectorized
    sweeps with the S5 binary operator (a1,b1)*(a2,b2) = (a1a2, a2 b1 + b2).
    """
    a = a.clone()
    b = b.clone()
    S = a.shape[1]
    d = 1
    while d < S:
        # combine element i with element i-d (new b before updating a)
        b = torch.cat([b[:, :d], b[:, d:] + a[:, d:] * b[:, :-d]], dim=1)
        a = torch.cat([a[:, :d], a[:, d:] * a[:, :-d]], dim=1)
        d *= 2
    return b


class S5Layer(nn.Module):
    """Diagonal S5 layer (fp32 complex scan)."""

    def __init__(self, features: int, state_dim: int = 64,
                 dt_min: float = 0.001, dt_max: float = 0.1):
        super().__init__()
        self.features = features
        self.state_dim = state_dim
        n = torch.arange(state_dim, dtype=torch.float32)
        self.log_A_real = nn.Parameter(torch.log(n + 0.5))
        self.A_imag = nn.Parameter(math.pi * n)

        def lecun(rows, cols):
            return torch.randn(rows, cols) * (1.0 / math.sqrt(cols))

        self.B_re = nn.Parameter(lecun(state_dim, features))
        self.B_im = nn.Parameter(lecun(state_dim, features))
        self.C_re = nn.Parameter(lecun(features, state_dim))
        self.C_im = nn.Parameter(lecun(features, state_dim))
        self.D = nn.Parameter(torch.randn(features))
        log_dt = torch.rand(state_dim) * (math.log(dt_max) - math.log(dt_min)) \
            + math.log(dt_min)
        self.log_dt = nn.Parameter(log_dt)

    def forward(self, u: torch.Tensor) -> torch.Tensor:
        B, S, Fdim = u.shape
        in_dtype = u.dtype
        u = u.float()

        dt = torch.exp(self.log_dt.float())                      # [N]
        A = torch.complex(-torch.exp(self.log_A_real.float()),
                          self.A_imag.float())                   # [N]
        A_bar = torch.exp(A * dt)                                # [N]
        B_c = torch.complex(self.B_re.float(), self.B_im.float())  # [N, F]
        B_bar = ((A_bar - 1.0) / (A + 1e-8)).unsqueeze(-1) * B_c   # [N, F]
        C_c = torch.complex(self.C_re.float(), self.C_im.float())  # [F, N]

        Bu = torch.einsum("bsf,nf->bsn", u.to(B_bar.dtype), B_bar)  # [B,S,N] cplx
        if Bu.is_cuda:
            from .. import ops
            x_states = ops.s5_scan(A_bar, Bu)
        else:
            a = A_bar.reshape(1, 1, -1).expand(B, S, self.state_dim)
            x_states = associative_scan_diag(a, Bu)

        y = torch.einsum("fn,bsn->bsf", C_c, x_states).real
        y = y + self.D.float().reshape(1, 1, -1) * u
        return y.to(in_dtype)


class BidirectionalS5Layer(nn.Module):
    """Forward + reversed S5 scans, concat then project (ref :225-286)."""

    def __init__(self, features: int, state_dim: int = 64,
                 dt_min: float = 0.001, dt_max: float = 0.1):
        super().__init__()
        self.s5_forward = S5Layer(features, state_dim, dt_min, dt_max)
        self.s5_backward = S5Layer(features, state_dim, dt_min, dt_max)
        self.out_proj = Dense(2 * features, features)

    def forward(self, u: torch.Tensor) -> torch.Tensor:
        y_fwd = self.s5_forward(u)
        y_bwd = self.s5_backward(u.flip(1)).flip(1)
        return self.out_proj(torch.cat([y_fwd, y_bwd], dim=-1))


class SpatialFusionConv(nn.Module):
    """Spatial-Mamba multi-dilation zero-init depthwise conv fusion
    (ref :293-350; arXiv:2410.15091)."""

    def __init__(self, features: int, dilations: Tuple[int, ...] = (1, 2, 3),
                 kernel_size: int = 3):
        super().__init__()
        self.dilations = dilations
        self.kernel_size = kernel_size
        self.weights = nn.ParameterList([
            nn.Parameter(torch.zeros(features, 1, kernel_size, kernel_size))
            for _ in dilations])

    def forward(self, y_2d: torch.Tensor) -> torch.Tensor:
        # NHWC throughout: the dilated depthwise runs on the HIP kernels of
        # depthwise.hip on GPU (weights stay in the torch (C,1,k,k) layout
        # for state-dict stability; permuted to [k,k,C,1] at call).
        out = y_2d
        for w, dil in zip(self.weights, self.dilations):
            wk = w.permute(2, 3, 0, 1).to(y_2d.dtype)
            out = out + ops.depthwise_conv2d(y_2d, wk, None, 1, dil)
        return out


class SSMDiTBlock(nn.Module):
    """DiTBlock with the attention replaced by (bidirectional) S5
    (ref :357-540). Same call signature as DiTBlock."""

    def __init__(self, features: int, num_heads: int,
                 rope_emb: Optional[RotaryEmbedding] = None,
                 state_dim: int = 64, mlp_ratio: int = 4,
                 norm_epsilon: float = 1e-5, use_gating: bool = True,
                 bidirectional: bool = True, use_2d_fusion: bool = False,
                 scan_order: str = "raster"):
        super().__init__()
        assert scan_order in ("raster", "hilbert", "zigzag")
        self.eps = norm_epsilon
        self.use_gating = use_gating
        self.use_2d_fusion = use_2d_fusion
        self.scan_order = scan_order
        self.ada_params = AdaLNParams(features, features)
        ssm_cls = BidirectionalS5Layer if bidirectional else S5Layer
        self.ssm = ssm_cls(features, state_dim)
        if use_2d_fusion:
            self.spatial_fusion = SpatialFusionConv(features)
        hidden = features * mlp_ratio
        self.mlp_in = Dense(features, hidden)
        self.mlp_out = Dense(hidden, features)

    def _apply_2d_fusion(self, y: torch.Tensor) -> torch.Tensor:
        B, S, Fdim = y.shape
        h_p = math.isqrt(S)
        assert h_p * h_p == S, "2D fusion requires a square patch grid"
        if self.scan_order == "hilbert":
            fwd = hilbert_indices(h_p, h_p).to(y.device)
            inv = inverse_permutation(fwd.cpu(), S).to(y.device)
        elif self.scan_order == "zigzag":
            fwd = zigzag_indices(h_p, h_p).to(y.device)
            inv = inverse_permutation(fwd.cpu(), S).to(y.device)
        else:
            fwd = inv = None
        y_rm = y if inv is None else y.index_select(1, inv)
        fused = self.spatial_fusion(y_rm.reshape(B, h_p, h_p, Fdim))
        fused = fused.reshape(B, S, Fdim)
        return fused if fwd is None else fused.index_select(1, fwd)

    def forward(self, x, conditioning, freqs_cis=None):
        scale_mlp, shift_mlp, gate_mlp, scale_attn, shift_attn, gate_attn = \
            self.ada_params(conditioning).chunk(6, dim=-1)

        residual = x
        norm_x = layer_norm_noaffine(x, self.eps)
        ssm_out = self.ssm(norm_x * (1 + scale_attn) + shift_attn)
        if self.use_2d_fusion:
            ssm_out = self._apply_2d_fusion(ssm_out)
        x = residual + (gate_attn * ssm_out if self.use_gating else ssm_out)

        residual = x
        norm_x = layer_norm_noaffine(x, self.eps)
        mlp_out = self.mlp_out(F.gelu(self.mlp_in(norm_x * (1 + scale_mlp) + shift_mlp)))
        x = residual + (gate_mlp * mlp_out if self.use_gating else mlp_out)
        return x


def _build_block_pattern(pattern: Optional[Sequence[str]], ratio: str,
                         num_layers: int):
    if pattern is not None:
        return list(pattern)
    if ratio == "all-ssm":
        return ["ssm"] * num_layers
    if ratio == "all-attn":
        return ["attn"] * num_layers
    n_ssm, n_attn = (int(p) for p in ratio.split(":"))
    unit = ["ssm"] * n_ssm + ["attn"] * n_attn
    return (unit * (num_layers // len(unit) + 1))[:num_layers]


class HybridSSMAttentionDiT(nn.Module):
    """Interleaved SSM/attention DiT (ref :545-779), e.g. ratio "3:1"."""

    def __init__(self,
                 output_channels: int = 3,
                 in_channels: int = 3,
                 patch_size: int = 16,
                 emb_features: int = 768,
                 num_layers: int = 12,
                 num_heads: int = 12,
                 mlp_ratio: int = 4,
                 ssm_state_dim: int = 64,
                 context_dim: int = 768,
                 norm_epsilon: float = 1e-5,
                 learn_sigma: bool = False,
                 use_hilbert: bool = False,
                 use_zigzag: bool = False,
                 block_pattern: Optional[Sequence[str]] = None,
                 ssm_attention_ratio: str = "3:1",
                 bidirectional_ssm: bool = True,
                 use_2d_fusion: bool = False):
        super().__init__()
        assert not (use_hilbert and use_zigzag)
        self.output_channels = output_channels
        self.patch_size = patch_size
        self.emb_features = emb_features
        self.learn_sigma = learn_sigma
        self.use_hilbert = use_hilbert
        self.use_zigzag = use_zigzag

        if use_hilbert or use_zigzag:
            self.hilbert_proj = Dense(patch_size * patch_size * in_channels,
                                      emb_features)
        else:
            self.patch_embed = PatchEmbedding(patch_size, emb_features, in_channels)

        self.time_fourier = FourierEmbedding(features=emb_features)
        self.time_proj = TimeProjection(emb_features, emb_features * mlp_ratio)
        self.time_out = Dense(emb_features * mlp_ratio, emb_features)
        self.text_proj = Dense(context_dim, emb_features)
        self.rope = RotaryEmbedding(dim=emb_features // num_heads, max_seq_len=4096)

        scan_order = "hilbert" if use_hilbert else ("zigzag" if use_zigzag
                                                    else "raster")
        self.block_pattern = _build_block_pattern(block_pattern,
                                                  ssm_attention_ratio, num_layers)
        blocks = []
        for kind in self.block_pattern:
            if kind == "ssm":
                blocks.append(SSMDiTBlock(
                    emb_features, num_heads, self.rope, ssm_state_dim,
                    mlp_ratio, norm_epsilon, bidirectional=bidirectional_ssm,
                    use_2d_fusion=use_2d_fusion, scan_order=scan_order))
            else:
                blocks.append(DiTBlock(emb_features, num_heads, self.rope,
                                       mlp_ratio, norm_epsilon))
        self.blocks = nn.ModuleList(blocks)

        self.final_norm = nn.LayerNorm(emb_features, eps=norm_epsilon)
        out_dim = patch_size * patch_size * output_channels
        if learn_sigma:
            out_dim *= 2
        self.final_proj = Dense(emb_features, out_dim, zero_init=True)
        self._pe_cache = {}

    def _pos_embed(self, h_p: int, w_p: int, device, dtype) -> torch.Tensor:
        key = (h_p, w_p, self.use_hilbert, self.use_zigzag)
        pe = self._pe_cache.get(key)
        if pe is None:
            pe = torch.from_numpy(
                build_2d_sincos_pos_embed(self.emb_features, h_p, w_p).copy())
            if self.use_hilbert:
                pe = pe[hilbert_indices(h_p, w_p)]
            elif self.use_zigzag:
                pe = pe[zigzag_indices(h_p, w_p)]
            self._pe_cache[key] = pe
        return pe.to(device=device, dtype=dtype)

    def forward(self, x: torch.Tensor, temb: torch.Tensor,
                textcontext: Optional[torch.Tensor] = None) -> torch.Tensor:
        B, H, W, C = x.shape
        p = self.patch_size
        h_p, w_p = H // p, W // p

        inv_idx = None
        if self.use_hilbert:
            raw, inv_idx = hilbert_patchify(x, p)
            seq = self.hilbert_proj(raw)
        elif self.use_zigzag:
            raw, inv_idx = zigzag_patchify(x, p)
            seq = self.hilbert_proj(raw)
        else:
            seq = self.patch_embed(x)
        n_tokens = seq.shape[1]

        seq = seq + self._pos_embed(h_p, w_p, seq.device, seq.dtype)[None]

        cond = self.time_out(self.time_proj(self.time_fourier(temb).to(seq.dtype)))
        if textcontext is not None:
            cond = cond + self.text_proj(textcontext.to(seq.dtype)).mean(dim=1)

        cos, sin = self.rope(n_tokens)
        if self.use_hilbert or self.use_zigzag:
            cos, sin = torch.ones_like(cos), torch.zeros_like(sin)
        freqs = (cos.to(seq.device), sin.to(seq.device))

        for block in self.blocks:
            seq = block(seq, cond, freqs)

        out = norm_fp32(self.final_norm, seq)
        out = self.final_proj(out)
        if self.learn_sigma:
            out, _ = out.chunk(2, dim=-1)
        if inv_idx is not None:
            return hilbert_unpatchify(out, inv_idx, p, H, W, self.output_channels)
        return unpatchify(out, channels=self.output_channels)
