"""Op dispatch layer.

Every hot op has two implementations:

* ``reference.py`` — pure PyTorch, task-batched.  Used on CPU and as the
  test oracle for the HIP kernels.
* the in-tree HIP/CDNA4 extension (``ops/hip``, built for gfx950 by
  ``ops/build.py``) — the mandatory GPU path.

Policy: on a CUDA/HIP device the extension **must** be importable unless
kernels were explicitly disabled (``disable_hip_kernels()`` or env
``MAML355_NO_HIP=1``); a silent eager fallback on the GPU box would
invalidate every benchmark, so we raise instead.
"""

from __future__ import annotations

import os

import torch

from . import reference as ref

_HIP_EXT = None
_HIP_TRIED = False
_HIP_DISABLED = os.environ.get("MAML355_NO_HIP", "0") == "1"


def disable_hip_kernels() -> None:
    global _HIP_DISABLED
    _HIP_DISABLED = True


def enable_hip_kernels() -> None:
    global _HIP_DISABLED, _HIP_TRIED
    _HIP_DISABLED = False
    _HIP_TRIED = False


def hip_ext():
    """Return the loaded HIP extension module, importing it on first use.
    Returns None when disabled or not on a GPU build."""
    global _HIP_EXT, _HIP_TRIED
    if _HIP_DISABLED:
        return None
    if not _HIP_TRIED:
        _HIP_TRIED = True
        try:
            from . import hip_loader
            _HIP_EXT = hip_loader.load()
        except Exception as e:  # noqa: BLE001 - surfaced via require_hip
            _HIP_EXT = None
            _HIP_IMPORT_ERROR[0] = e
    return _HIP_EXT


_HIP_IMPORT_ERROR: list = [None]


def _want_hip(x: torch.Tensor) -> bool:
    if not x.is_cuda:
        return False
    ext = hip_ext()
    if ext is None:
        if _HIP_DISABLED:
            return False
        raise RuntimeError(
            "maml355: tensor is on %s but the HIP/CDNA4 extension is not "
            "loaded (import error: %r). Build it with "
            "`python -m howtotrainyourmamlpytorch_amd.ops.build` or disable "
            "kernels explicitly with MAML355_NO_HIP=1." % (x.device, _HIP_IMPORT_ERROR[0])
        )
    return True


# ---------------------------------------------------------------------------
# public ops — HIP autograd wrappers are registered here as they land
# ---------------------------------------------------------------------------

def task_conv3x3(x, w, b=None, stride=1, padding=1, return_stats=False):
    if _want_hip(x):
        from . import hip_autograd
        return hip_autograd.task_conv3x3(x, w, b, stride, padding, return_stats)
    y = ref.task_conv3x3(x, w, b, stride, padding)
    return (y, None) if return_stats else y


def task_bn_act(x, gamma, beta, eps=1e-5, negative_slope=0.01, apply_act=True):
    if _want_hip(x):
        from . import hip_autograd
        return hip_autograd.task_bn_act(x, gamma, beta, eps, negative_slope, apply_act)
    return ref.task_bn_act(x, gamma, beta, eps, negative_slope, apply_act)


def task_layer_norm_act(x, weight, bias, eps=1e-5, negative_slope=0.01, apply_act=True):
    _want_hip(x)  # layer-norm path: reference composition is acceptable on GPU too
    return ref.task_layer_norm_act(x, weight, bias, eps, negative_slope, apply_act)


def task_bn_act_pool(x, gamma, beta, eps=1e-5, negative_slope=0.01, sums=None):
    """BN(batch stats) + leaky-ReLU + 2x2 maxpool.  Fused single-pass
    forward on GPU (C % 8 == 0), optionally consuming conv-epilogue
    precomputed stats; composition elsewhere."""
    if x.is_cuda and x.shape[-1] % 8 == 0 and _want_hip(x):
        from . import hip_autograd
        return hip_autograd.task_bn_act_pool(x, gamma, beta, eps,
                                             negative_slope, sums)
    y, mean, var = task_bn_act(x, gamma, beta, eps, negative_slope)
    return task_maxpool2x2(y), mean, var


def task_maxpool2x2(x):
    if _want_hip(x):
        from . import hip_autograd
        return hip_autograd.task_maxpool2x2(x)
    return ref.task_maxpool2x2(x)


def task_global_avgpool(x):
    return ref.task_global_avgpool(x)


def task_linear(x, w, b=None):
    if _want_hip(x):
        from . import hip_autograd
        return hip_autograd.task_linear(x, w, b)
    return ref.task_linear(x, w, b)


def softmax_cross_entropy(logits, labels):
    if _want_hip(logits):
        from . import hip_autograd
        return hip_autograd.softmax_cross_entropy(logits, labels)
    return ref.softmax_cross_entropy(logits, labels)


def lslr_update(arena, grad, lr_vec):
    if _want_hip(arena):
        from . import hip_autograd
        return hip_autograd.lslr_update(arena, grad, lr_vec)
    return ref.lslr_update(arena, grad, lr_vec)


def fused_adam_step(params, grads, exp_avgs, exp_avg_sqs, step, lr,
                    beta1=0.9, beta2=0.999, eps=1e-8, weight_decay=0.0, clamp=None):
    if params and _want_hip(params[0]):
        from . import hip_autograd
        return hip_autograd.fused_adam_step(params, grads, exp_avgs, exp_avg_sqs,
                                            step, lr, beta1, beta2, eps, weight_decay, clamp)
    return ref.fused_adam_step(params, grads, exp_avgs, exp_avg_sqs, step, lr,
                               beta1, beta2, eps, weight_decay, clamp)
