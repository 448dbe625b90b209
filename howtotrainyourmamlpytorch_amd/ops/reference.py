"""Pure-PyTorch task-batched reference implementations of every hot op.

These are (a) the CPU execution path, (b) the numerics oracle the HIP/CDNA4
kernels are tested against, and (c) the fallback that is *never* silently
used on a GPU box (``ops.__init__`` fails loudly there unless HIP kernels
are explicitly disabled).

Layout contract (shared with the HIP kernels): activations are
**task-batched NHWC** — ``x[T, NS, H, W, C]`` where ``T`` is the number of
tasks resident on this GPU and ``NS`` the flattened ways*shots image batch
of one task.  Weights keep the reference-compatible logical shapes
(``[T, F, C, 3, 3]`` conv, ``[T, ways, K]`` linear) so checkpoints
round-trip; kernels repack internally.

Task batching is the central MI355X design decision: the reference runs
each task's inner loop serially in Python
(``few_shot_learning_system.py:193``), which on tiny few-shot shapes is
kernel-launch bound.  Here one op call processes every resident task, so
each launch does ``T×`` the work.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.nn.functional as F


def task_conv3x3(x: torch.Tensor, w: torch.Tensor, b: Optional[torch.Tensor],
                 stride: int = 1, padding: int = 1) -> torch.Tensor:
    """Task-batched 3x3 conv: per-task weights (fast weights differ per task).

    x: [T, NS, H, W, C] ; w: [T, F, C, 3, 3] ; b: [T, F] or None
    returns [T, NS, Ho, Wo, F]

    Implemented as one grouped conv (groups=T) so even the oracle is a
    single ATen call, not a Python loop.
    """
    T, NS, H, W, C = x.shape
    Tw, Fo, Cw, kh, kw = w.shape
    assert Tw == T and Cw == C and kh == 3 and kw == 3, (x.shape, w.shape)
    # [T, NS, H, W, C] -> [NS, T*C, H, W]
    xg = x.permute(1, 0, 4, 2, 3).reshape(NS, T * C, H, W)
    wg = w.reshape(T * Fo, C, 3, 3)
    bg = b.reshape(T * Fo) if b is not None else None
    yg = F.conv2d(xg, wg, bg, stride=stride, padding=padding, groups=T)
    Ho, Wo = yg.shape[-2], yg.shape[-1]
    # [NS, T*F, Ho, Wo] -> [T, NS, Ho, Wo, F]
    return yg.reshape(NS, T, Fo, Ho, Wo).permute(1, 0, 3, 4, 2).contiguous()


def task_bn_act(x: torch.Tensor, gamma: torch.Tensor, beta: torch.Tensor,
                eps: float = 1e-5, negative_slope: float = 0.01,
                apply_act: bool = True) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Task-batched BatchNorm (batch statistics, training semantics — the
    reference *always* normalizes with batch stats,
    ``meta_neural_network_architectures.py:244-247``) fused with leaky-ReLU.

    x: [T, NS, H, W, C]
    gamma/beta: [C] (shared per-step meta-params) or [T, C] (inner-loop
    adapted fast weights)
    returns (y [T, NS, H, W, C], batch_mean [T, C], batch_var [T, C])
    — batch_var is the biased (1/M) variance used for normalization; the
    running-stat update uses it too, matching F.batch_norm's fused update.
    """
    T, NS, H, W, C = x.shape
    xf = x.float()
    mean = xf.mean(dim=(1, 2, 3))                     # [T, C]
    var = xf.var(dim=(1, 2, 3), unbiased=False)       # [T, C]
    inv = torch.rsqrt(var + eps)
    if gamma.dim() == 1:
        g = gamma.view(1, 1, 1, 1, C)
        bta = beta.view(1, 1, 1, 1, C)
    else:
        g = gamma.view(T, 1, 1, 1, C)
        bta = beta.view(T, 1, 1, 1, C)
    y = (xf - mean.view(T, 1, 1, 1, C)) * inv.view(T, 1, 1, 1, C) * g + bta
    if apply_act:
        y = F.leaky_relu(y, negative_slope=negative_slope)
    return y.to(x.dtype), mean, var


def task_layer_norm_act(x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor,
                        eps: float = 1e-5, negative_slope: float = 0.01,
                        apply_act: bool = True) -> torch.Tensor:
    """Task-batched LayerNorm over (H, W, C) per image + leaky-ReLU.

    Mirrors the reference's alternative norm (weight frozen at 1, bias
    adaptable — ``meta_neural_network_architectures.py:261-322``).
    x: [T, NS, H, W, C]; weight/bias: [C]-broadcastable or [T, C].
    """
    T, NS, H, W, C = x.shape
    xf = x.float()
    mean = xf.mean(dim=(2, 3, 4), keepdim=True)
    var = xf.var(dim=(2, 3, 4), unbiased=False, keepdim=True)
    y = (xf - mean) * torch.rsqrt(var + eps)
    if weight.dim() == 1:
        y = y * weight.view(1, 1, 1, 1, C) + bias.view(1, 1, 1, 1, C)
    else:
        y = y * weight.view(T, 1, 1, 1, C) + bias.view(T, 1, 1, 1, C)
    if apply_act:
        y = F.leaky_relu(y, negative_slope=negative_slope)
    return y.to(x.dtype)


def task_maxpool2x2(x: torch.Tensor) -> torch.Tensor:
    """2x2 stride-2 max pool on NHWC task-batched input (floor mode, like
    the reference's ``F.max_pool2d(out, 2)``)."""
    T, NS, H, W, C = x.shape
    xn = x.reshape(T * NS, H, W, C).permute(0, 3, 1, 2)
    yn = F.max_pool2d(xn, kernel_size=2, stride=2)
    Ho, Wo = yn.shape[-2], yn.shape[-1]
    return yn.permute(0, 2, 3, 1).reshape(T, NS, Ho, Wo, C)


def task_global_avgpool(x: torch.Tensor) -> torch.Tensor:
    """Global average pool over H, W (the reference's no-max-pool path,
    ``meta_neural_network_architectures.py:609,654-655``)."""
    return x.mean(dim=(2, 3))


def task_linear(x: torch.Tensor, w: torch.Tensor, b: Optional[torch.Tensor]) -> torch.Tensor:
    """Task-batched linear head: x [T, M, K] @ w [T, ways, K]^T + b [T, ways]."""
    y = torch.bmm(x, w.transpose(1, 2))
    if b is not None:
        y = y + b.unsqueeze(1)
    return y


def softmax_cross_entropy(logits: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
    """Per-task mean cross-entropy.  logits [T, M, ways], labels [T, M]
    (int64) -> loss [T]."""
    T, M, ways = logits.shape
    loss = F.cross_entropy(logits.reshape(T * M, ways).float(),
                           labels.reshape(T * M), reduction="none")
    return loss.view(T, M).mean(dim=1)


def lslr_update(arena: torch.Tensor, grad: torch.Tensor,
                lr_vec: torch.Tensor) -> torch.Tensor:
    """Fused LSLR fast-weight update over the flat arena:
    ``arena' = arena - lr_vec * grad`` with lr_vec [P] broadcast over tasks.
    arena/grad: [T, P]."""
    return arena - lr_vec.unsqueeze(0) * grad


def fused_adam_step(params, grads, exp_avgs, exp_avg_sqs, step: int,
                    lr: float, beta1: float = 0.9, beta2: float = 0.999,
                    eps: float = 1e-8, weight_decay: float = 0.0,
                    clamp: Optional[float] = None) -> None:
    """Reference Adam (in-place, fp32 state) with optional pre-clamp of
    gradients (the reference clamps to ±10 for imagenet datasets,
    ``few_shot_learning_system.py:332-335``).  Lists of flat tensors."""
    bc1 = 1.0 - beta1 ** step
    bc2 = 1.0 - beta2 ** step
    for p, g, m, v in zip(params, grads, exp_avgs, exp_avg_sqs):
        if g is None:
            continue
        g = g.float()
        if clamp is not None:
            g = g.clamp(-clamp, clamp)
        if weight_decay != 0.0:
            g = g + weight_decay * p.float()
        m.mul_(beta1).add_(g, alpha=1.0 - beta1)
        v.mul_(beta2).addcmul_(g, g, value=1.0 - beta2)
        denom = (v / bc2).sqrt_().add_(eps)
        p.data.addcdiv_(m, denom, value=-lr / bc1)
