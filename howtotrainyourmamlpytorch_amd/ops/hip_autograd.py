"""autograd.Function wrappers around the HIP/CDNA4 kernels.

Double-backward policy (second-order MAML needs ``create_graph=True``
through the inner-loop support forward):

* ops whose backward is *bilinear in saved tensors* (LSLR update, maxpool
  scatter) use custom kernels at every order;
* BN+act and softmax-CE check ``torch.is_grad_enabled()`` inside
  ``backward`` (True exactly under create_graph) and fall back to a
  differentiable torch composition there, using the fused backward kernels
  on every ordinary backward (eval, first-order, and the big outer
  backward).

Conv uses the fully-custom bilinear trio in ``tconv.hip`` once built; until
then the grouped-conv reference composition runs (MIOpen under ATen).
"""

from __future__ import annotations

from typing import Optional

import torch

from . import hip_ext
from . import reference as ref


def _ext():
    e = hip_ext()
    assert e is not None, "hip_autograd used without loaded extension"
    return e


# ---------------------------------------------------------------------------
# fused BN + leaky-ReLU
# ---------------------------------------------------------------------------
class _BNActFn(torch.autograd.Function):
    """x3 [T, M, C] -> (y, mean, var).  Fused kernels forward and on plain
    backward; differentiable torch composition under create_graph."""

    @staticmethod
    def forward(ctx, x3, gamma, beta, eps, slope, act):
        y, mean, var, rstd = _ext().bn_act_fwd(x3, gamma.float(), beta.float(),
                                               eps, slope, act)
        ctx.save_for_backward(x3, mean, rstd, gamma, beta)
        ctx.eps, ctx.slope, ctx.act = eps, slope, act
        return y, mean, var

    @staticmethod
    def backward(ctx, dy, dmean, dvar):
        x3, mean, rstd, gamma, beta = ctx.saved_tensors
        per_task = gamma.dim() == 2
        if torch.is_grad_enabled():
            T, M, C = x3.shape
            xf = x3.float()
            mu = xf.mean(dim=1, keepdim=True)
            var_t = xf.var(dim=1, unbiased=False, keepdim=True)
            inv = torch.rsqrt(var_t + ctx.eps)
            xhat = (xf - mu) * inv
            g = gamma.float().view(T if per_task else 1, 1, C)
            b = beta.float().view(T if per_task else 1, 1, C)
            dy_eff = dy.float()
            if ctx.act:
                pre = xhat * g + b
                dy_eff = dy_eff * torch.where(
                    pre > 0, torch.ones_like(pre), torch.full_like(pre, ctx.slope))
            s1 = dy_eff.mean(dim=1, keepdim=True)
            s2 = (dy_eff * xhat).mean(dim=1, keepdim=True)
            dx = (g * inv * (dy_eff - s1 - xhat * s2)).to(x3.dtype)
            dgamma_t = (dy_eff * xhat).sum(dim=1)
            dbeta_t = dy_eff.sum(dim=1)
        else:
            dx, dgamma_t, dbeta_t = _ext().bn_act_bwd(
                dy, x3, mean, rstd, gamma.float(), beta.float(),
                ctx.slope, ctx.act)
        if per_task:
            dgamma, dbeta = dgamma_t, dbeta_t
        else:
            dgamma, dbeta = dgamma_t.sum(0), dbeta_t.sum(0)
        return dx, dgamma.to(gamma.dtype), dbeta.to(beta.dtype), None, None, None


def task_bn_act(x, gamma, beta, eps=1e-5, negative_slope=0.01, apply_act=True):
    T, NS, H, W, C = x.shape
    x3 = x.reshape(T, NS * H * W, C).contiguous()
    y, mean, var = _BNActFn.apply(x3, gamma, beta, eps, negative_slope, apply_act)
    return y.view(T, NS, H, W, C), mean, var


# ---------------------------------------------------------------------------
# maxpool 2x2 — mask is fixed after forward, so backward (scatter) and
# double-backward (gather) are both linear custom kernels.
# ---------------------------------------------------------------------------
class _PoolBwdFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, dy, mask, H, W):
        ctx.save_for_backward(mask)
        ctx.HW = (H, W)
        return _ext().maxpool2x2_bwd(dy, mask, H, W)

    @staticmethod
    def backward(ctx, ddx):
        (mask,) = ctx.saved_tensors
        # gather: d(dy) = ddx at the argmax positions == pool-forward of ddx
        # restricted to saved argmax — implemented by re-running fwd gather
        g = _gather_by_mask(ddx.contiguous(), mask)
        return g, None, None, None


def _gather_by_mask(x4, mask):
    # x4: [N, H, W, C]; mask: [N, Ho, Wo, C] -> [N, Ho, Wo, C]
    N, H, W, C = x4.shape
    Ho, Wo = mask.shape[1], mask.shape[2]
    win = x4[:, :2 * Ho, :2 * Wo, :].reshape(N, Ho, 2, Wo, 2, C)
    win = win.permute(0, 1, 3, 5, 2, 4).reshape(N, Ho, Wo, C, 4)
    return torch.gather(win, 4, mask.long().unsqueeze(-1)).squeeze(-1)


class _PoolFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x4):
        y, mask = _ext().maxpool2x2_fwd(x4)
        ctx.save_for_backward(mask)
        ctx.HW = (x4.shape[1], x4.shape[2])
        return y

    @staticmethod
    def backward(ctx, dy):
        (mask,) = ctx.saved_tensors
        H, W = ctx.HW
        return _PoolBwdFn.apply(dy.contiguous(), mask, H, W)


def task_maxpool2x2(x):
    T, NS, H, W, C = x.shape
    y = _PoolFn.apply(x.reshape(T * NS, H, W, C).contiguous())
    return y.view(T, NS, H // 2, W // 2, C)


# ---------------------------------------------------------------------------
# fused softmax-CE
# ---------------------------------------------------------------------------
class _CEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, labels):
        loss, probs = _ext().ce_fwd(logits, labels)
        ctx.save_for_backward(logits, probs, labels)
        return loss

    @staticmethod
    def backward(ctx, gtask):
        logits, probs, labels = ctx.saved_tensors
        if torch.is_grad_enabled():
            T, M, ways = logits.shape
            lf = logits.float()
            p = torch.softmax(lf, dim=-1)
            onehot = torch.nn.functional.one_hot(labels, ways).float()
            dlogits = (p - onehot) * (gtask.view(T, 1, 1) / M)
            return dlogits.to(logits.dtype), None
        return _ext().ce_bwd(probs, labels, gtask.contiguous()).to(logits.dtype), None


def softmax_cross_entropy(logits, labels):
    return _CEFn.apply(logits.contiguous(), labels.contiguous())


# ---------------------------------------------------------------------------
# fused LSLR update — backward components are bilinear; custom at all orders
# ---------------------------------------------------------------------------
class _LSLRFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, arena, grad, lr_vec):
        ctx.save_for_backward(grad, lr_vec)
        return _ext().lslr_fwd(arena.contiguous(), grad, lr_vec)

    @staticmethod
    def backward(ctx, gout):
        grad, lr_vec = ctx.saved_tensors
        if torch.is_grad_enabled():
            dgrad = -lr_vec.unsqueeze(0) * gout
            dlr = -(gout * grad).sum(dim=0)
            return gout, dgrad, dlr
        dgrad, dlr = _ext().lslr_bwd(gout, grad, lr_vec)
        return gout, dgrad, dlr


def lslr_update(arena, grad, lr_vec):
    return _LSLRFn.apply(arena, grad.contiguous(), lr_vec.contiguous())


# ---------------------------------------------------------------------------
# conv trio — MFMA implicit GEMM.  The three ops (fwd, dgrad, wgrad) are
# mutually bilinear, so each backward is a composition of the other two:
# custom kernels at EVERY derivative order (second-order MAML's
# create_graph included), no torch fallback on the hot path.
# ---------------------------------------------------------------------------
class _ConvFwdFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b, pad):
        ctx.save_for_backward(x, w)
        ctx.pad = pad
        ctx.has_bias = b is not None
        wp = _ext().tconv_repack(w, False)
        H, W = x.shape[2], x.shape[3]
        Ho, Wo = H + 2 * pad - 2, W + 2 * pad - 2
        return _ext().tconv_mm(x, wp, b, pad, Ho, Wo)

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        dy = dy.contiguous()
        dx = dw = db = None
        if ctx.needs_input_grad[0]:
            dx = _ConvDgradFn.apply(dy, w, ctx.pad)
        if ctx.needs_input_grad[1]:
            dw = _ConvWgradFn.apply(dy, x, ctx.pad)
        if ctx.has_bias and ctx.needs_input_grad[2]:
            db = dy.float().sum(dim=(1, 2, 3))
        return dx, dw, db, None


class _ConvDgradFn(torch.autograd.Function):
    """dx = dgrad(dy, w): the fwd kernel run with flipped/transposed
    repacked weights and pad' = 2 - pad (full-correlation identity)."""

    @staticmethod
    def forward(ctx, dy, w, pad):
        ctx.save_for_backward(dy, w)
        ctx.pad = pad
        wp = _ext().tconv_repack(w, True)
        Ho, Wo = dy.shape[2], dy.shape[3]
        H, W = Ho - 2 * pad + 2, Wo - 2 * pad + 2
        return _ext().tconv_mm(dy, wp, None, 2 - pad, H, W)

    @staticmethod
    def backward(ctx, g):
        dy, w = ctx.saved_tensors
        g = g.contiguous()
        d_dy = d_w = None
        if ctx.needs_input_grad[0]:
            d_dy = _ConvFwdFn.apply(g, w, None, ctx.pad)
        if ctx.needs_input_grad[1]:
            d_w = _ConvWgradFn.apply(dy, g, ctx.pad)
        return d_dy, d_w, None


class _ConvWgradFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, dy, x, pad):
        ctx.save_for_backward(dy, x)
        ctx.pad = pad
        return _ext().tconv_wgrad(dy, x, pad)

    @staticmethod
    def backward(ctx, gw):
        dy, x = ctx.saved_tensors
        gw = gw.contiguous()
        d_dy = d_x = None
        if ctx.needs_input_grad[0]:
            d_dy = _ConvFwdFn.apply(x, gw, None, ctx.pad)
        if ctx.needs_input_grad[1]:
            d_x = _ConvDgradFn.apply(dy, gw, ctx.pad)
        return d_dy, d_x, None


def task_conv3x3(x, w, b=None, stride=1, padding=1):
    if stride != 1 or x.dtype != torch.bfloat16 or w.shape[1] > 64:
        # stride-2 (max_pooling=False configs) and fp32 compute use the
        # grouped-ATen composition; cast for dtype consistency
        wc = w.to(x.dtype)
        bc = b.to(x.dtype) if b is not None else None
        return ref.task_conv3x3(x, wc, bc, stride, padding)
    return _ConvFwdFn.apply(x.contiguous(), w.contiguous(),
                            b.contiguous() if b is not None else None, padding)


def task_linear(x, w, b=None):
    return ref.task_linear(x, w.to(x.dtype), b.to(x.dtype) if b is not None else None)


def fused_adam_step(params, grads, exp_avgs, exp_avg_sqs, step, lr,
                    beta1=0.9, beta2=0.999, eps=1e-8, weight_decay=0.0, clamp=None):
    return ref.fused_adam_step(params, grads, exp_avgs, exp_avg_sqs, step, lr,
                               beta1, beta2, eps, weight_decay, clamp)
