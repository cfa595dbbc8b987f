"""Op dispatch layer: HIP/CDNA4 kernels on GPU, pure-torch reference on CPU.

Contract:
  * All ops take NHWC activations and HWIO conv weights.
  * On CUDA (ROCm) tensors the hand-written gfx950 HIP kernels in
    `flaxdiff_amd/ops/hip/` are REQUIRED — if the extension is not built the
    op raises instead of silently falling back (the GPU path must be native).
  * On CPU tensors the fp32 torch reference (`ops.reference`) runs; it is
    also the oracle every kernel is tested against.

Hot-op inventory (SURVEY.md §2.10 native-code inventory):
  group_norm (fused SiLU), conv2d 3x3/stride-2/1x1 implicit GEMM, attention
  (CFG-doubled self+cross), rms_norm, time embeddings, forward-diffusion axpy,
  fused Adam+EMA.
"""
from __future__ import annotations

import os
from typing import Optional

import torch

from . import reference

_EXT = None
_EXT_ERR: Optional[str] = None


def _load_ext():
    """Import the in-tree HIP extension (built by __graft_entry__.build())."""
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        from .hip import build as hip_build
        _EXT = hip_build.load_extension()
    except Exception as e:  # noqa: BLE001
        _EXT_ERR = f"{type(e).__name__}: {e}"
        _EXT = None
    return _EXT


def hip_available() -> bool:
    return _load_ext() is not None


def _require_ext():
    ext = _load_ext()
    if ext is None:
        raise RuntimeError(
            "flaxdiff_amd HIP extension is not available on a GPU tensor path "
            f"(build error: {_EXT_ERR}). Run __graft_entry__.build() / "
            "python -m flaxdiff_amd.ops.hip.build first — the GPU path never "
            "silently falls back to stock torch ops.")
    return ext


_FORCE_TORCH = os.environ.get("FLAXDIFF_FORCE_TORCH_OPS", "0") == "1"
_GEMM_LIB = os.environ.get("FD_GEMM_LIB", "0") == "1"  # revert dense GEMMs to hipBLASLt


def _use_hip(x: torch.Tensor) -> bool:
    return x.is_cuda and not _FORCE_TORCH


# ---------------------------------------------------------------------------
# GroupNorm (+ optional fused SiLU)
# ---------------------------------------------------------------------------

class _GroupNormSiLUFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, gamma, beta, groups, eps, silu):
        ext = _require_ext()
        y, mean, rstd = ext.gn_silu_fwd(x, gamma, beta, groups, eps, silu)
        ctx.save_for_backward(x, gamma, mean, rstd)
        ctx.groups = groups
        ctx.silu = silu
        ctx.has_beta = beta is not None
        ctx.beta = beta
        return y

    @staticmethod
    def backward(ctx, dy):
        x, gamma, mean, rstd = ctx.saved_tensors
        ext = _require_ext()
        beta = ctx.beta if ctx.has_beta else torch.zeros_like(gamma)
        dx, dgamma, dbeta = ext.gn_silu_bwd(dy.contiguous(), x, gamma, beta,
                                            mean, rstd, ctx.groups, ctx.silu)
        return dx, dgamma, (dbeta if ctx.has_beta else None), None, None, None


def group_norm(x: torch.Tensor, groups: int, gamma: torch.Tensor,
               beta: torch.Tensor, eps: float = 1e-5, silu: bool = False) -> torch.Tensor:
    if _use_hip(x):
        return _GroupNormSiLUFn.apply(x.contiguous(), gamma, beta, groups, eps, silu)
    return reference.group_norm_nhwc(x, groups, gamma, beta, eps, silu)


# ---------------------------------------------------------------------------
# RMSNorm (last-dim)
# ---------------------------------------------------------------------------

class _RMSNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, gamma, eps):
        ext = _require_ext()
        y, rrms = ext.rms_norm_fwd(x, gamma, eps)
        ctx.save_for_backward(x, gamma, rrms)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, gamma, rrms = ctx.saved_tensors
        ext = _require_ext()
        dx, dgamma = ext.rms_norm_bwd(dy.contiguous(), x, gamma, rrms)
        return dx, dgamma, None


def rms_norm(x: torch.Tensor, gamma: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    if _use_hip(x):
        return _RMSNormFn.apply(x.contiguous(), gamma, eps)
    return reference.rms_norm(x, gamma, eps)


# ---------------------------------------------------------------------------
# Conv2d NHWC (3x3 SAME / 1x1 / stride-2) — LDS-tiled implicit GEMM on MFMA
# ---------------------------------------------------------------------------

def _conv2d_wgrad_gemms(dy: torch.Tensor, x: torch.Tensor, kh: int, kw: int,
                        stride: int) -> torch.Tensor:
    """Weight gradient as KH*KW shifted-view GEMMs on MFMA (hipBLASLt).

    dw[r,s,ci,co] = sum_m x[b, oh*st+r-pt, ow*st+s-pl, ci] * dy[b,oh,ow,co]
    over the in-bounds output positions. Each (r,s) tap is one [Ci x M']@[M' x Co]
    library GEMM over a strided view — no im2col materialization. A fully
    hand-written LDS-tiled wgrad kernel is the planned replacement.
    """
    B, H, W, Ci = x.shape
    _, OH, OW, Co = dy.shape
    pt = max((OH - 1) * stride + kh - H, 0) // 2
    pl = max((OW - 1) * stride + kw - W, 0) // 2
    dw = torch.zeros(kh, kw, Ci, Co, dtype=torch.float32, device=x.device)
    for r in range(kh):
        oh_lo = max(0, -((pt - r) // -stride))  # ceil((pt-r)/stride), clamped
        oh_hi = min(OH - 1, (H - 1 - r + pt) // stride)
        if oh_hi < oh_lo:
            continue
        ih_lo = oh_lo * stride + r - pt
        noh = oh_hi - oh_lo + 1
        for s in range(kw):
            ow_lo = max(0, -((pl - s) // -stride))
            ow_hi = min(OW - 1, (W - 1 - s + pl) // stride)
            if ow_hi < ow_lo:
                continue
            iw_lo = ow_lo * stride + s - pl
            now = ow_hi - ow_lo + 1
            x_sub = x[:, ih_lo:ih_lo + (noh - 1) * stride + 1:stride,
                      iw_lo:iw_lo + (now - 1) * stride + 1:stride, :]
            dy_sub = dy[:, oh_lo:oh_hi + 1, ow_lo:ow_hi + 1, :]
            a = x_sub.reshape(-1, Ci)
            b = dy_sub.reshape(-1, Co)
            dw[r, s] += (a.transpose(0, 1) @ b).float()
    return dw


def _kernel_view(p: torch.Tensor) -> torch.Tensor:
    """bf16 tensor the compute kernels consume. An fp32 master param with a
    bf16 shadow (trainer/optim.py) uses the shadow — no cast kernel; the
    grads then flow back in fp32 straight into the flat grad buffer (no
    fp32->bf16->fp32 round trip through a _ShadowCast node)."""
    if p.dtype == torch.bfloat16:
        return p
    sh = getattr(p, "_shadow_bf16", None)
    if sh is not None:
        return sh
    return p.to(torch.bfloat16)


def _kernel_view_t(p: torch.Tensor) -> torch.Tensor:
    """[N,K] transposed bf16 view of a [K,N] dense weight for the NT GEMM.
    Optimizer-managed params carry a per-step-refreshed transposed shadow;
    otherwise (inference) transpose on the fly."""
    t = getattr(p, "_shadow_bf16_t", None)
    if t is not None:
        return t
    return _kernel_view(p).t().contiguous()


def transpose_shadows(src_flat, dst_flat, tiles):
    _require_ext().transpose_shadows(src_flat, dst_flat, tiles)


def _routable_param(p: Optional[torch.Tensor]) -> bool:
    return p is None or p.dtype == torch.bfloat16 or \
        getattr(p, "_shadow_bf16", None) is not None


class _Conv2dFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b, stride, add, badd):
        ext = _require_ext()
        wk = _kernel_view(w)
        # fp32 bias feeds the kernel's fp32 epilogue directly (no cast);
        # `add` (residual, same shape as y) and `badd` (per-sample [B,C]
        # broadcast, the temb projection) are fused into the same epilogue
        y = ext.conv2d_fwd(x, wk, b if b is not None else torch.Tensor(),
                           stride,
                           add if add is not None else torch.Tensor(),
                           badd if badd is not None else torch.Tensor())
        ctx.save_for_backward(x, wk)
        ctx.stride = stride
        ctx.has_bias = b is not None
        ctx.has_add = add is not None
        ctx.has_badd = badd is not None
        ctx.w_dtype = w.dtype
        ctx.b_dtype = b.dtype if b is not None else None
        return y

    @staticmethod
    def backward(ctx, dy):
        x, wk = ctx.saved_tensors
        ext = _require_ext()
        dy = dy.contiguous()
        dx = ext.conv2d_dgrad(dy, wk, ctx.stride, x.shape[1], x.shape[2]) \
            if ctx.needs_input_grad[0] else None
        dw = db = None
        if ctx.needs_input_grad[1]:
            dw, _db_unused = ext.conv2d_wgrad(dy, x, wk.shape[0], wk.shape[1],
                                              ctx.stride)
            dw = dw if ctx.w_dtype == torch.float32 else dw.to(ctx.w_dtype)
        if ctx.has_bias:
            db = _bias_grad(dy)          # fp32
            if ctx.b_dtype != torch.float32:
                db = db.to(ctx.b_dtype)
        dadd = dy if ctx.has_add else None   # y = conv + add -> d(add) = dy
        dbadd = None
        if ctx.has_badd:                     # d(badd)[b,c] = sum_hw dy
            B = dy.shape[0]
            dbadd = dy.reshape(B, -1, dy.shape[-1]).sum(1)
        return dx, dw, db, None, dadd, dbadd


class _DenseFn(torch.autograd.Function):
    """y = x @ w (+ b) with the WEIGHT grad routed through the split-M
    wgrad kernel (1x1-conv case). hipBLASLt picks a no-split tile for the
    [Cin x M] @ [M x Cout] reduction (M ~ 1e6, tiny output) and serializes
    the K loop in one workgroup per tile — measured 590us/call vs ~60us
    through the split-M kernel (profiles/ MT64x64x256 rows)."""

    @staticmethod
    def forward(ctx, x2d, w2d, b, add):
        wk = _kernel_view(w2d)
        if not _GEMM_LIB and x2d.shape[1] % 64 == 0:
            # hand-written MFMA GEMM (fused fp32 bias + optional residual) —
            # no hipBLASLt on the declared hot path (BASELINE.json north
            # star). The weight comes in transposed ([N,K] NT layout).
            ext = _require_ext()
            bf = b if b is not None else torch.Tensor()
            if bf.numel() and bf.dtype != torch.float32:
                bf = bf.float()
            y = ext.gemm_nt(x2d, _kernel_view_t(w2d), bf,
                            add if add is not None else torch.Tensor())
        else:
            y = torch.matmul(x2d, wk)
            if b is not None:
                y = y + _kernel_view(b)
            if add is not None:
                y = y + add
        ctx.save_for_backward(x2d, wk)
        ctx.has_bias = b is not None
        ctx.has_add = add is not None
        ctx.w_dtype = w2d.dtype
        ctx.b_dtype = b.dtype if b is not None else None
        return y

    @staticmethod
    def backward(ctx, dy):
        x2d, wk = ctx.saved_tensors
        dy = dy.contiguous()
        ext = _require_ext()
        if ctx.needs_input_grad[0]:
            if not _GEMM_LIB and dy.shape[1] % 64 == 0:
                dx = ext.gemm_dx(dy, wk)
            else:
                dx = torch.matmul(dy, wk.t())
        else:
            dx = None
        dw = None
        if ctx.needs_input_grad[1]:
            M = x2d.shape[0]
            dw4, _ = ext.conv2d_wgrad(dy.view(M, 1, 1, dy.shape[1]),
                                      x2d.view(M, 1, 1, x2d.shape[1]), 1, 1, 1)
            dw = dw4.view(x2d.shape[1], dy.shape[1])
            if ctx.w_dtype != torch.float32:
                dw = dw.to(ctx.w_dtype)
        db = None
        if ctx.has_bias:
            db = _bias_grad(dy)
            if ctx.b_dtype != torch.float32:
                db = db.to(ctx.b_dtype)
        dadd = dy if ctx.has_add else None   # y = xW + add -> d(add) = dy
        return dx, dw, db, dadd


def dense(x: torch.Tensor, w: torch.Tensor, b: Optional[torch.Tensor] = None,
          add: Optional[torch.Tensor] = None) -> torch.Tensor:
    """x [..., Cin] @ w [Cin, Cout] (+ b): MFMA GEMM forward, split-M
    wgrad backward on GPU. w/b may be fp32 masters with bf16 shadows.
    `add` fuses a same-shape residual into the GEMM epilogue (transformer
    residual adds, reference attention.py:240-303) in fp32 before the single
    bf16 rounding."""
    if _use_hip(x) and x.dtype == torch.bfloat16 and _routable_param(w) \
            and _routable_param(b):
        lead = x.shape[:-1]
        add2 = None
        if add is not None and add.dtype == torch.bfloat16:
            add2 = add.reshape(-1, w.shape[1]).contiguous()
        y = _DenseFn.apply(x.reshape(-1, x.shape[-1]).contiguous(), w, b, add2)
        y = y.reshape(*lead, w.shape[1])
        if add is not None and add2 is None:
            y = y + add
        return y
    if w.dtype != x.dtype:
        w = w.to(x.dtype)
    y = torch.matmul(x, w)
    if b is not None:
        y = y + b.to(y.dtype)
    if add is not None:
        y = y + add.to(y.dtype)
    return y


class _WL2LossFn(torch.autograd.Function):
    """mean(w[b] * 0.5*(pred-target)^2) in one pass each way (the trainer
    loss, reference diffusion_trainer.py:203-211)."""

    @staticmethod
    def forward(ctx, pred, target, wb):
        ctx.save_for_backward(pred, target, wb)
        return _require_ext().wl2_loss_fwd(pred, target, wb)

    @staticmethod
    def backward(ctx, go):
        pred, target, wb = ctx.saved_tensors
        dpred = _require_ext().wl2_loss_bwd(pred, target, wb,
                                            go.reshape(1).float())
        return dpred, None, None


def weighted_l2_loss(pred: torch.Tensor, target: torch.Tensor,
                     weights: torch.Tensor) -> torch.Tensor:
    """Fused loss = mean(weights * 0.5*(pred-target)^2); weights broadcast
    per sample [B,1,1,...]. Falls back to torch ops off the bf16 GPU path."""
    if _use_hip(pred) and pred.dtype == torch.bfloat16 \
            and target.dtype == torch.bfloat16:
        wb = weights.reshape(weights.shape[0]).float().contiguous()
        return _WL2LossFn.apply(pred.contiguous(), target.contiguous(), wb)
    return (0.5 * (pred.float() - target.float()) ** 2
            * weights.to(torch.float32)).mean()


class _ConvUp2xFn(torch.autograd.Function):
    """nearest-2x upsample + 3x3 SAME conv as ONE kernel: the upsample is an
    address remap inside the conv's halo staging (no materialized big
    tensor on the forward; reference common.py:203-226 Upsample).
    Backward rematerializes the upsampled activation once for wgrad and
    folds the upsample adjoint (2x2 window sum) over the conv dgrad."""

    @staticmethod
    def forward(ctx, x, w, b):
        ext = _require_ext()
        wk = _kernel_view(w)
        y = ext.conv2d_fwd_up2x(x, wk, b if b is not None else torch.Tensor())
        if y is None or y.numel() == 0:   # undefined tensor -> None via pybind
            raise _ConvUpIneligibleError
        ctx.save_for_backward(x, wk)
        ctx.has_bias = b is not None
        ctx.w_dtype = w.dtype
        ctx.b_dtype = b.dtype if b is not None else None
        return y

    @staticmethod
    def backward(ctx, dy):
        x, wk = ctx.saved_tensors
        ext = _require_ext()
        dy = dy.contiguous()
        H2, W2 = x.shape[1] * 2, x.shape[2] * 2
        dx = db = dw = None
        if ctx.needs_input_grad[0]:
            dx_up = ext.conv2d_dgrad(dy, wk, 1, H2, W2)
            dx = ext.upsample2x_bwd(dx_up)
        if ctx.needs_input_grad[1]:
            x_up = ext.upsample2x_fwd(x)
            dw, _ = ext.conv2d_wgrad(dy, x_up, 3, 3, 1)
            dw = dw if ctx.w_dtype == torch.float32 else dw.to(ctx.w_dtype)
        if ctx.has_bias:
            db = _bias_grad(dy)
            if ctx.b_dtype != torch.float32:
                db = db.to(ctx.b_dtype)
        return dx, dw, db


class _ConvUpIneligibleError(Exception):
    pass


def conv2d_upsample2x(x: torch.Tensor, w: torch.Tensor,
                      b: Optional[torch.Tensor] = None) -> torch.Tensor:
    """conv2d(nearest_upsample_2x(x), w, b) with the upsample fused into the
    conv's staging on the GPU halo path; falls back to the two-op form."""
    if _use_hip(x) and x.dtype == torch.bfloat16 and _routable_param(w) \
            and w.shape[0] == 3 and w.shape[1] == 3:
        try:
            return _ConvUp2xFn.apply(
                x.contiguous(), w.contiguous(),
                b.contiguous() if b is not None else None)
        except _ConvUpIneligibleError:
            pass
    return conv2d(nearest_upsample_2x(x), w, b, stride=1)


class _GEGLUFn(torch.autograd.Function):
    """y = h[..., :N] * gelu(h[..., N:]) — one fused pass each way
    (reference attention.py:207-238)."""

    @staticmethod
    def forward(ctx, h):
        ctx.save_for_backward(h)
        return _require_ext().geglu_fwd(h)

    @staticmethod
    def backward(ctx, dy):
        (h,) = ctx.saved_tensors
        return _require_ext().geglu_bwd(dy.contiguous(), h)


def geglu(h: torch.Tensor) -> torch.Tensor:
    """GEGLU gate: split the last dim in half, y = a * gelu(b)."""
    if _use_hip(h) and h.dtype == torch.bfloat16 and h.shape[-1] % 16 == 0:
        return _GEGLUFn.apply(h.contiguous())
    a, b = h.chunk(2, dim=-1)
    return a * torch.nn.functional.gelu(b)


class _Cat2Fn(torch.autograd.Function):
    """Last-dim 2-tensor concat with a one-pass split backward (the UNet
    skip concat, reference simple_unet.py:132-160)."""

    @staticmethod
    def forward(ctx, a, b):
        ctx.ca = a.shape[-1]
        ctx.cb = b.shape[-1]
        return _require_ext().cat2_lastdim(a, b)

    @staticmethod
    def backward(ctx, dy):
        da, db = _require_ext().split2_lastdim(dy.contiguous(), ctx.ca, ctx.cb)
        return da, db


def cat_channels(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """torch.cat([a, b], -1) with fused fwd/bwd kernels on the GPU path."""
    if _use_hip(a) and a.dtype == torch.bfloat16 and b.dtype == torch.bfloat16 \
            and (a.shape[-1] + b.shape[-1]) % 8 == 0:
        return _Cat2Fn.apply(a.contiguous(), b.contiguous())
    return torch.cat([a, b], dim=-1)


def _bias_grad(dy: torch.Tensor) -> torch.Tensor:
    """Column sum over all but the channel dim (coalesced HIP kernel)."""
    d2 = dy.reshape(-1, dy.shape[-1])
    C = d2.shape[1]
    if d2.is_cuda and d2.dtype == torch.bfloat16 and C % 8 == 0 and C <= 2048:
        return _require_ext().colsum(d2.contiguous())
    return d2.float().sum(0)


def conv2d(x: torch.Tensor, w: torch.Tensor, b: Optional[torch.Tensor] = None,
           stride: int = 1, add: Optional[torch.Tensor] = None,
           badd: Optional[torch.Tensor] = None) -> torch.Tensor:
    """SAME-padded NHWC conv. 1x1 convs route to a plain GEMM.
    w/b may be fp32 master params with bf16 shadows (no cast kernels).
    `add` fuses a same-shape residual into the conv epilogue (ResidualBlock's
    `conv2(out) + residual`, reference common.py:268); `badd` [B, Cout] fuses
    the per-sample temb-projection broadcast (`out + t[:,None,None,:]`,
    reference common.py:257) — each saves one full HBM read+write pass per
    block on GPU; both fall back to plain adds elsewhere."""
    kh, kw = w.shape[0], w.shape[1]
    if kh == 1 and kw == 1 and stride == 1:
        w2 = w.reshape(w.shape[2], w.shape[3])
        # reshape drops tensor attributes: re-attach the bf16 shadow so the
        # fp32 master still routes through _DenseFn (grads flow through the
        # reshape view back to the param)
        sh = getattr(w, "_shadow_bf16", None)
        if sh is not None and w.dtype != torch.bfloat16:
            w2._shadow_bf16 = sh.reshape(w.shape[2], w.shape[3])
            sht = getattr(w, "_shadow_bf16_t", None)
            if sht is not None:
                w2._shadow_bf16_t = sht
        y = dense(x.reshape(-1, w.shape[2]), w2, b)
        y = y.reshape(*x.shape[:-1], w.shape[3])
        if badd is not None:
            y = y + badd[:, None, None, :].to(y.dtype)
        return y if add is None else y + add
    if _use_hip(x) and x.dtype == torch.bfloat16 and _routable_param(w):
        if add is not None and add.dtype != torch.bfloat16:
            add_in, add_out = None, add
        else:
            add_in, add_out = add, None
        if badd is not None and badd.dtype != torch.bfloat16:
            badd_in, badd_out = None, badd
        else:
            badd_in, badd_out = badd, None
        out = _Conv2dFn.apply(
            x.contiguous(), w.contiguous(),
            b.contiguous() if b is not None else None, stride,
            add_in.contiguous() if add_in is not None else None,
            badd_in.contiguous() if badd_in is not None else None)
        if badd_out is not None:
            out = out + badd_out[:, None, None, :].to(out.dtype)
        if add_out is not None:
            out = out + add_out
        return out
    if w.dtype != x.dtype:
        w = w.to(x.dtype)
    if b is not None and b.dtype != x.dtype:
        b = b.to(x.dtype)
    y = reference.conv2d_nhwc(x, w, b, stride=stride, padding="same")
    if badd is not None:
        y = y + badd[:, None, None, :].to(y.dtype)
    return y if add is None else y + add


class _ConvTransposeFn(torch.autograd.Function):
    """stride-2 transposed conv as ONE dgrad-kernel call.

    convT(x, w) is exactly the input-gradient of the stride-2 SAME conv with
    weight wf = w.permute(0,1,3,2) (verified bit-exact against the
    zero-stuff + flipped-conv form on CPU, incl. odd sizes) — so the forward
    runs the implicit-GEMM dgrad kernel directly with no zero-stuffed pixels
    (the former zero-stuff path spent 3/4 of its MACs on zeros). Adjoints:
    d/dx = the forward conv, d/dwf = the split-M wgrad with roles swapped.
    """

    @staticmethod
    def forward(ctx, x, wf, stride):
        ext = _require_ext()
        H2 = x.shape[1] * stride
        W2 = x.shape[2] * stride
        y = ext.conv2d_dgrad(x, wf, stride, H2, W2)
        ctx.save_for_backward(x, wf)
        ctx.stride = stride
        return y

    @staticmethod
    def backward(ctx, dout):
        x, wf = ctx.saved_tensors
        ext = _require_ext()
        dout = dout.contiguous()
        dx = ext.conv2d_fwd(dout, wf, torch.Tensor(), ctx.stride,
                            torch.Tensor(), torch.Tensor()) \
            if ctx.needs_input_grad[0] else None
        dwf = None
        if ctx.needs_input_grad[1]:
            dwf, _ = ext.conv2d_wgrad(x, dout, wf.shape[0], wf.shape[1],
                                      ctx.stride)
            dwf = dwf.to(wf.dtype)
        return dx, dwf, None


def conv2d_transpose(x: torch.Tensor, w: torch.Tensor, b: Optional[torch.Tensor] = None,
                     stride: int = 2) -> torch.Tensor:
    """flax-SAME transposed conv; w is [kh, kw, Cin, Cout]. On GPU the
    stride-2 3x3 case runs the implicit-GEMM dgrad kernel directly
    (_ConvTransposeFn)."""
    kh, kw = w.shape[0], w.shape[1]
    if _use_hip(x) and stride == 2 and kh == 3 and kw == 3:
        wf = w.permute(0, 1, 3, 2).contiguous()
        out = _ConvTransposeFn.apply(x.contiguous(), wf, stride)
        if b is not None:
            out = out + b
        return out
    return reference.conv2d_transpose_nhwc(x, w, b, stride)


class _DepthwiseConvFn(torch.autograd.Function):
    """3x3 depthwise NHWC conv on the HIP kernels (depthwise.hip)."""

    @staticmethod
    def forward(ctx, x, w, stride, dilation):
        ext = _require_ext()
        y = ext.dwconv_fwd(x, w, None, stride, dilation)
        ctx.save_for_backward(x, w)
        ctx.stride = stride
        ctx.dilation = dilation
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        ext = _require_ext()
        dy = dy.contiguous()
        dx = ext.dwconv_dgrad(dy, w, x.shape[1], x.shape[2],
                              ctx.stride, ctx.dilation)
        dw = ext.dwconv_wgrad(dy, x, ctx.stride, ctx.dilation)
        return dx, dw.to(w.dtype), None, None


def depthwise_conv2d(x: torch.Tensor, w: torch.Tensor, b: Optional[torch.Tensor] = None,
                     stride: int = 1, dilation: int = 1) -> torch.Tensor:
    """Depthwise NHWC conv; w is [kh, kw, C, 1] (SeparableConv,
    reference common.py:126-153). GPU bf16 3x3 runs the hand-written
    kernels; other kernel sizes compose torch ops (cold path)."""
    if (_use_hip(x) and x.dtype == torch.bfloat16 and b is None
            and w.shape[0] == 3 and w.shape[1] == 3 and x.shape[-1] % 8 == 0):
        w3 = w.reshape(3, 3, -1).contiguous()
        return _DepthwiseConvFn.apply(x, w3, stride, dilation)
    return reference.depthwise_conv2d_nhwc(x, w, b, stride, dilation)


# ---------------------------------------------------------------------------
# Attention (softmax fp32) — hand-written flash kernel forward; backward
# recomputes P from saved LSE and runs the 5 grads as MFMA GEMMs.
# ---------------------------------------------------------------------------

class _AttentionFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, scale):
        ext = _require_ext()
        o, lse = ext.attn_fwd(q, k, v, scale)
        ctx.save_for_backward(q, k, v, lse)
        ctx.scale = scale
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, lse = ctx.saved_tensors
        scale = ctx.scale
        # Small-KV fused MFMA flash backward: 3x the composed GEMM path at
        # the bench shapes (2.34 vs 7.12 ms at Sq=4096 D=16 — tools/
        # attn_bwd_ab.py). Larger head dims fall through to the GEMMs.
        use_v1 = os.environ.get("FD_ATTN_BWD_V1")
        dmax = 32 if use_v1 else 64
        if k.shape[2] <= 128 and q.shape[3] <= dmax:
            ext = _require_ext()
            fn = ext.attn_bwd_smallkv if use_v1 else ext.attn_bwd_smallkv_v2
            dq, dk, dv = fn(q, k, v, do.contiguous(), lse, scale)
            return dq, dk, dv, None
        # General shapes: recompute P row-exactly from the saved log-sum-exp,
        # then the grads are plain batched GEMMs (library GEMM on MFMA).
        qf, kf, vf, dof = q.float(), k.float(), v.float(), do.float()
        s = torch.einsum("bhqd,bhkd->bhqk", qf, kf) * scale
        p = torch.exp(s - lse.unsqueeze(-1))
        dv = torch.einsum("bhqk,bhqd->bhkd", p, dof)
        dp = torch.einsum("bhqd,bhkd->bhqk", dof, vf)
        dsum = (dp * p).sum(-1, keepdim=True)
        ds = (dp - dsum) * p * scale
        dq = torch.einsum("bhqk,bhkd->bhqd", ds, kf)
        dk = torch.einsum("bhqk,bhqd->bhkd", ds, qf)
        return dq.to(q.dtype), dk.to(k.dtype), dv.to(v.dtype), None


def attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
              scale: Optional[float] = None) -> torch.Tensor:
    """q,k,v: [B, H, S, D] -> [B, H, Sq, D]; fp32 softmax.

    The flash kernel covers D in 8..128 multiples of 8 (every shipped config).
    Other head dims (e.g. emb 48 / 4 heads = 12) compose library GEMMs +
    fp32 softmax on-device — a shape-gated route, not an eager fallback."""
    if scale is None:
        scale = q.shape[-1] ** -0.5
    if _use_hip(q):
        D = q.shape[-1]
        if D % 8 == 0 and D <= 128:
            return _AttentionFn.apply(q.contiguous(), k.contiguous(),
                                      v.contiguous(), scale)
    return reference.attention(q, k, v, scale)


# ---------------------------------------------------------------------------
# Elementwise fusions
# ---------------------------------------------------------------------------

def forward_diffusion(x0, eps, signal_rate, noise_rate):
    if _use_hip(x0):
        ext = _require_ext()
        return ext.fwd_diffusion(x0, eps, signal_rate.reshape(-1).float().contiguous(),
                                 noise_rate.reshape(-1).float().contiguous())
    return reference.forward_diffusion(x0, eps, signal_rate, noise_rate)


def nearest_upsample_2x(x: torch.Tensor) -> torch.Tensor:
    if _use_hip(x):
        ext = _require_ext()
        return _Upsample2xFn.apply(x.contiguous())
    return reference.nearest_upsample_2x_nhwc(x)


class _Upsample2xFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ext = _require_ext()
        return ext.upsample2x_fwd(x)

    @staticmethod
    def backward(ctx, dy):
        ext = _require_ext()
        return ext.upsample2x_bwd(dy.contiguous())


def avg_pool_2x(x: torch.Tensor) -> torch.Tensor:
    return reference.avg_pool_2x_nhwc(x)


def sinusoidal_time_embedding(t: torch.Tensor, features: int,
                              max_positions: int = 10000) -> torch.Tensor:
    if _use_hip(t):
        ext = _require_ext()
        return ext.time_embed(t, torch.Tensor(), features, float(max_positions))
    return reference.sinusoidal_time_embedding(t, features, max_positions)


def fourier_time_embedding(t: torch.Tensor, freqs: torch.Tensor) -> torch.Tensor:
    if _use_hip(t):
        ext = _require_ext()
        return ext.time_embed(t, freqs, freqs.numel() * 2, 0.0)
    return reference.fourier_time_embedding(t, freqs)


# ---------------------------------------------------------------------------
# Fused Adam + EMA optimizer step (HIP kernel, one pass over params)
# ---------------------------------------------------------------------------

class _S5ScanFn(torch.autograd.Function):
    """x_s = a * x_{s-1} + bu_s (complex, a constant per channel) — serial
    per-(b,n) HIP scan (one read + one write of the tensor) replacing the
    log-depth doubling. Backward is the exact reverse scan with conj(a)."""

    @staticmethod
    def forward(ctx, a_ri, bu_ri):
        ext = _require_ext()
        x = ext.s5_scan_fwd(a_ri.contiguous(), bu_ri.contiguous())
        ctx.save_for_backward(a_ri, x)
        return x

    @staticmethod
    def backward(ctx, dx):
        a_ri, x = ctx.saved_tensors
        ext = _require_ext()
        dbu, dap = ext.s5_scan_bwd(a_ri.contiguous(), x, dx.contiguous())
        return dap.sum(dim=0), dbu


def s5_scan(a: torch.Tensor, bu: torch.Tensor) -> torch.Tensor:
    """a: [N] complex64; bu: [B,S,N] complex64 -> inclusive scan [B,S,N]."""
    if bu.is_cuda and not _FORCE_TORCH:
        a_ri = torch.view_as_real(a.contiguous()).float()
        bu_ri = torch.view_as_real(bu.contiguous()).float()
        x = _S5ScanFn.apply(a_ri, bu_ri)
        return torch.view_as_complex(x)
    from ..models.ssm_dit import associative_scan_diag
    return associative_scan_diag(a.reshape(1, 1, -1).expand_as(bu), bu)


def fused_adamw_ema(params_f32, grads, exp_avg, exp_avg_sq, ema_f32, params_bf16,
                    *, lr, beta1, beta2, eps, weight_decay, step, ema_decay,
                    grad_scale: float = 1.0, scale_dev=None, skip_ctr=None,
                    step_dev=None):
    """In-place AdamW + EMA lerp over flat fp32 master buffers.

    grads may be bf16 (gets scaled by grad_scale, e.g. 1/world_size folded in).
    params_bf16 is refreshed from the fp32 master in the same pass (may be None).
    scale_dev (optional fp32 device scalar) overrides grad_scale — built
    on-device as grad_scale * clip-factor * finite-gate so the optimizer step
    needs no GPU→host sync; a nonfinite/zero value skips the whole update and
    bumps skip_ctr (optional int32 device scalar).
    """
    ext = _require_ext()
    ext.fused_adamw_ema(params_f32, grads, exp_avg, exp_avg_sq, ema_f32,
                        params_bf16 if params_bf16 is not None else torch.Tensor(),
                        lr, beta1, beta2, eps, weight_decay, step, ema_decay,
                        grad_scale, scale_dev, skip_ctr, step_dev)
