"""Task-batched functional VGG backbone (the reference's
``VGGReLUNormNetwork``, ``meta_neural_network_architectures.py:545-689``,
re-designed MI355X-first).

Differences from the reference that matter:

* **Task batching.**  ``forward`` takes a whole task batch
  ``x[T, NS, C, H, W]`` and a fast-weight arena ``[T, P]`` — every op
  processes all resident tasks in one kernel launch.  The reference runs
  one task at a time.
* **NHWC activations** internally (HIP kernels are channel-innermost for
  MFMA K-contiguity); the public input stays NCHW images.
* Block order matches the reference: conv -> norm -> leaky-ReLU -> maxpool
  (``MetaConvNormLayerReLU.forward:416-428``), with BN *always* using batch
  statistics (training semantics) and per-step running stats / per-step
  gamma,beta when ``per_step_bn_statistics``
  (``meta_neural_network_architectures.py:177-185,226-247``).
"""

from __future__ import annotations

import os
from typing import Dict, List, Optional, Tuple

import torch
import torch.nn as nn

from .. import ops
from .arena import ParamArena


def _xavier_uniform(shape, gen: Optional[torch.Generator] = None) -> torch.Tensor:
    t = torch.empty(*shape)
    nn.init.xavier_uniform_(t.view(shape[0], -1), generator=gen)
    return t


class _SlotGather(torch.autograd.Function):
    """Per-element LR vector from the per-slot LSLR table.

    Forward is a plain gather; backward is an ORDERED per-slot segment sum
    over the contiguous arena ranges (torch's index_select backward is an
    atomic index_add on CUDA — run-to-run nondeterministic, measured
    7e-7 drift on the LSLR grads).  The backward uses differentiable torch
    ops, so create_graph (second-order MAML) works through it."""

    @staticmethod
    def forward(ctx, lrs_step, slot_index, bounds):
        ctx.bounds = bounds
        return lrs_step.index_select(0, slot_index)

    @staticmethod
    def backward(ctx, gout):
        dl = torch.stack([gout[off:off + n].sum() for off, n in ctx.bounds])
        return dl, None, None


class TaskBatchedVGG(nn.Module):
    """4-stage conv backbone + linear head, fully functional over an arena.

    Inner-loop-adapted params (the arena): conv weights/biases + linear
    weight/bias (+ norm-layer affine params when
    ``enable_inner_loop_optimizable_bn_params``).  Meta-only params: the
    arena init ``theta`` plus per-step BN gamma/beta.
    """

    def __init__(self, im_shape: Tuple[int, int, int], num_output_classes: int,
                 num_stages: int = 4, num_filters: int = 64,
                 max_pooling: bool = True, conv_padding: bool = True,
                 norm_layer: str = "batch_norm", per_step_bn_statistics: bool = True,
                 num_steps: int = 5, learnable_bn_gamma: bool = True,
                 learnable_bn_beta: bool = True,
                 inner_loop_bn_params: bool = False,
                 negative_slope: float = 0.01, bn_momentum: float = 0.1,
                 bn_eps: float = 1e-5, generator: Optional[torch.Generator] = None):
        super().__init__()
        self.im_c, self.im_h, self.im_w = im_shape
        self.num_stages = num_stages
        self.num_filters = num_filters
        self.max_pooling = max_pooling
        self.conv_padding = conv_padding
        self.norm_layer_type = norm_layer
        self.per_step_bn_statistics = per_step_bn_statistics
        self.num_steps = num_steps
        self.inner_loop_bn_params = inner_loop_bn_params
        self.negative_slope = negative_slope
        self.bn_momentum = bn_momentum
        self.bn_eps = bn_eps
        self.num_output_classes = num_output_classes

        # ----- shape inference (reference does this with a dummy tensor,
        # :586-618; here it is closed-form) -----
        pad = 1 if conv_padding else 0
        stride = 1 if max_pooling else 2
        h, w = self.im_h, self.im_w
        self.stage_shapes: List[Tuple[int, int, int]] = []  # (C_in, H_in, W_in) per stage
        c = self.im_c
        for i in range(num_stages):
            self.stage_shapes.append((c, h, w))
            h = (h + 2 * pad - 3) // stride + 1
            w = (w + 2 * pad - 3) // stride + 1
            if max_pooling:
                h, w = h // 2, w // 2
            if h < 1 or w < 1:
                raise ValueError(
                    f"spatial dims underflow at stage {i}: image "
                    f"{self.im_h}x{self.im_w} cannot support {num_stages} stages")
            c = num_filters
        self.final_spatial = (h, w)
        self.feature_dim = num_filters if not max_pooling else num_filters * h * w

        # ----- arena spec (reference-style dotted names) -----
        named_shapes: List[Tuple[str, Tuple[int, ...]]] = []
        for i in range(num_stages):
            cin, _, _ = self.stage_shapes[i]
            named_shapes.append((f"layer_dict.conv{i}.conv.weight", (num_filters, cin, 3, 3)))
            named_shapes.append((f"layer_dict.conv{i}.conv.bias", (num_filters,)))
            if inner_loop_bn_params and norm_layer == "batch_norm":
                # reference overrides per-step gamma/beta to a single
                # [num_features] parameter when BN params are inner-loop
                # fast weights (meta_neural_network_architectures.py:194-198)
                bn_shape = (num_filters,)
                named_shapes.append((f"layer_dict.conv{i}.norm_layer.weight", bn_shape))
                named_shapes.append((f"layer_dict.conv{i}.norm_layer.bias", bn_shape))
        # linear head: weight stored [ways, K] with K = NHWC-flattened feature
        named_shapes.append(("layer_dict.linear.weights", (num_output_classes, self.feature_dim)))
        named_shapes.append(("layer_dict.linear.bias", (num_output_classes,)))
        self.arena = ParamArena(named_shapes)

        # ----- theta: the meta-learned arena initialization -----
        init = {}
        for i in range(num_stages):
            cin, _, _ = self.stage_shapes[i]
            init[f"layer_dict.conv{i}.conv.weight"] = _xavier_uniform((num_filters, cin, 3, 3), generator)
            init[f"layer_dict.conv{i}.conv.bias"] = torch.zeros(num_filters)
            if inner_loop_bn_params and norm_layer == "batch_norm":
                bn_shape = (num_filters,)
                init[f"layer_dict.conv{i}.norm_layer.weight"] = torch.ones(*bn_shape)
                init[f"layer_dict.conv{i}.norm_layer.bias"] = torch.zeros(*bn_shape)
        init["layer_dict.linear.weights"] = _xavier_uniform(
            (num_output_classes, self.feature_dim), generator)
        init["layer_dict.linear.bias"] = torch.zeros(num_output_classes)
        self.theta = nn.Parameter(self.arena.pack(init))
        slot_index = self.arena.slot_index()
        self.register_buffer("slot_index", slot_index, persistent=False)

        # ----- meta-only BN params / running-stat buffers -----
        if norm_layer == "batch_norm" and not inner_loop_bn_params:
            steps_dim = (num_steps,) if per_step_bn_statistics else ()
            for i in range(num_stages):
                g = torch.ones(*steps_dim, num_filters)
                b = torch.zeros(*steps_dim, num_filters)
                self.register_parameter(f"bn_weight_{i}",
                                        nn.Parameter(g, requires_grad=learnable_bn_gamma))
                self.register_parameter(f"bn_bias_{i}",
                                        nn.Parameter(b, requires_grad=learnable_bn_beta))
        if norm_layer == "batch_norm":
            steps_dim = (num_steps,) if per_step_bn_statistics else ()
            for i in range(num_stages):
                self.register_buffer(f"bn_running_mean_{i}", torch.zeros(*steps_dim, num_filters))
                self.register_buffer(f"bn_running_var_{i}", torch.ones(*steps_dim, num_filters))
                self.register_buffer(f"bn_backup_mean_{i}", torch.zeros(*steps_dim, num_filters))
                self.register_buffer(f"bn_backup_var_{i}", torch.ones(*steps_dim, num_filters))

    # ------------------------------------------------------------------
    def init_arena(self, num_tasks: int) -> torch.Tensor:
        """Fast-weight arena for a task batch: theta broadcast to [T, P].
        Gradients flow back to theta through the expand."""
        return self.theta.unsqueeze(0).expand(num_tasks, self.arena.numel)

    def lr_vector(self, lrs: torch.Tensor, num_step: int) -> torch.Tensor:
        """Per-element learning-rate vector [P] from the LSLR table
        [num_slots, num_steps+1] — differentiable gather with a
        deterministic segment-sum backward."""
        if not hasattr(self, "_slot_bounds"):
            self._slot_bounds = [(s.offset, s.numel) for s in self.arena.specs]
        return _SlotGather.apply(lrs[:, num_step], self.slot_index,
                                 self._slot_bounds)

    # ------------------------------------------------------------------
    def forward(self, x: torch.Tensor, num_step: int, arena: torch.Tensor,
                training: bool = True, backup_running_statistics: bool = False) -> torch.Tensor:
        """x: [T, NS, C, H, W] images; arena: [T, P] fast weights.
        Returns logits [T, NS, ways].

        Running BN statistics are updated on every call (the reference
        normalizes with batch stats and updates running stats even at eval,
        which is why backup/restore exists —
        ``meta_neural_network_architectures.py:240-255``)."""
        T, NS, C, H, W = x.shape
        v = self.arena.views(arena)
        if backup_running_statistics:
            self._backup_stats()

        out = x.permute(0, 1, 3, 4, 2)  # NHWC
        stride = 1 if self.max_pooling else 2
        pad = 1 if self.conv_padding else 0
        for i in range(self.num_stages):
            w_i = v[f"layer_dict.conv{i}.conv.weight"]
            b_i = v[f"layer_dict.conv{i}.conv.bias"]
            # conv-epilogue BN-stats fusion measured NET-NEGATIVE (-9%:
            # LDS atomics in the hot conv kernel cost more than the cheap
            # separate stats pass) — keep the capability but default off
            # (epilogue fusion uses LDS/global atomics, so it is disabled
            # under the deterministic-reduction mode)
            want_stats = (self.norm_layer_type == "batch_norm"
                          and self.max_pooling
                          and os.environ.get("MAML355_EPIFUSE", "0") == "1"
                          and os.environ.get("MAML355_DETERMINISTIC", "0") != "1")
            bn_sums = None
            if want_stats:
                out, bn_sums = ops.task_conv3x3(out, w_i, b_i, stride=stride,
                                                padding=pad, return_stats=True)
            else:
                out = ops.task_conv3x3(out, w_i, b_i, stride=stride, padding=pad)
            if self.norm_layer_type == "batch_norm":
                gamma, beta = self._bn_affine(i, num_step, v)
                stat_count = out[0].numel() // out.shape[-1]
                if self.max_pooling:
                    # fused BN+act+pool consuming the conv-epilogue stats
                    out, mean, var = ops.task_bn_act_pool(
                        out, gamma, beta, eps=self.bn_eps,
                        negative_slope=self.negative_slope, sums=bn_sums)
                else:
                    out, mean, var = ops.task_bn_act(
                        out, gamma, beta, eps=self.bn_eps,
                        negative_slope=self.negative_slope)
                self._update_running_stats(i, num_step, mean, var,
                                           count=stat_count)
            elif self.norm_layer_type == "layer_norm":
                wname = f"layer_dict.conv{i}.norm_layer.bias"
                bias = v[wname] if wname in v else torch.zeros(
                    out.shape[-1], device=out.device, dtype=torch.float32)
                weight = torch.ones_like(bias)
                out = ops.task_layer_norm_act(out, weight, bias,
                                              eps=self.bn_eps, negative_slope=self.negative_slope)
            else:
                out = torch.nn.functional.leaky_relu(out, negative_slope=self.negative_slope)
            if self.max_pooling and self.norm_layer_type != "batch_norm":
                out = ops.task_maxpool2x2(out)

        if self.max_pooling:
            feats = out.reshape(T, NS, -1)
        else:
            feats = ops.task_global_avgpool(out)
        logits = ops.task_linear(feats, v["layer_dict.linear.weights"],
                                 v["layer_dict.linear.bias"])
        return logits

    # ------------------------------------------------------------------
    def _bn_affine(self, i: int, num_step: int, v: Dict[str, torch.Tensor]):
        """Resolve gamma/beta for stage i at inner step ``num_step``.
        Reference semantics: per-step gamma/beta are used when BN params are
        NOT inner-loop fast weights (:229-234); otherwise the passed fast
        weights are used."""
        name_w = f"layer_dict.conv{i}.norm_layer.weight"
        if self.inner_loop_bn_params and name_w in v:
            # single [T, F] fast weight regardless of per-step stats
            # (reference override, meta_neural_network_architectures.py:194-198)
            return v[name_w], v[f"layer_dict.conv{i}.norm_layer.bias"]
        gamma = getattr(self, f"bn_weight_{i}")
        beta = getattr(self, f"bn_bias_{i}")
        if self.per_step_bn_statistics:
            gamma = gamma[num_step]          # [F]
            beta = beta[num_step]
        return gamma, beta

    def _update_running_stats(self, i: int, num_step: int, mean: torch.Tensor,
                              var: torch.Tensor, count: int) -> None:
        with torch.no_grad():
            m = mean.mean(dim=0)             # average the per-task batch stats
            # unbiased variance for the running buffer, like F.batch_norm
            bessel = count / max(1, count - 1)
            vv = var.mean(dim=0) * bessel
            rm = getattr(self, f"bn_running_mean_{i}")
            rv = getattr(self, f"bn_running_var_{i}")
            if self.per_step_bn_statistics:
                rm[num_step].mul_(1 - self.bn_momentum).add_(self.bn_momentum * m)
                rv[num_step].mul_(1 - self.bn_momentum).add_(self.bn_momentum * vv)
            else:
                rm.mul_(1 - self.bn_momentum).add_(self.bn_momentum * m)
                rv.mul_(1 - self.bn_momentum).add_(self.bn_momentum * vv)

    def _backup_stats(self) -> None:
        if self.norm_layer_type != "batch_norm":
            return
        with torch.no_grad():
            for i in range(self.num_stages):
                getattr(self, f"bn_backup_mean_{i}").copy_(getattr(self, f"bn_running_mean_{i}"))
                getattr(self, f"bn_backup_var_{i}").copy_(getattr(self, f"bn_running_var_{i}"))

    def restore_backup_stats(self) -> None:
        """Reset running stats to the pre-eval backup (reference:
        ``meta_neural_network_architectures.py:683-688``)."""
        if self.norm_layer_type != "batch_norm":
            return
        with torch.no_grad():
            for i in range(self.num_stages):
                getattr(self, f"bn_running_mean_{i}").copy_(getattr(self, f"bn_backup_mean_{i}"))
                getattr(self, f"bn_running_var_{i}").copy_(getattr(self, f"bn_backup_var_{i}"))
