"""The MAML / MAML++ meta-learning engine (reference:
``few_shot_learning_system.py``), re-designed task-batched.

All six MAML++ innovations are implemented (SURVEY.md top):
MSL (multi-step loss with annealed importance vector), LSLR (learnable
per-layer per-step inner LRs), BNRS/BNWB (per-step BN statistics and
weights — in the model), DA (first-order until
``first_order_to_second_order_epoch``), CA (cosine-annealed outer LR).

Key structural departure from the reference: the per-task Python loop
(``few_shot_learning_system.py:193``) is replaced by **task batching** — the
whole local meta-batch adapts simultaneously over one fast-weight arena
``[T, P]``, so each inner step is a handful of large fused kernel launches
instead of ``T x layers`` small ones.  The math is unchanged: tasks do not
interact (per-task BN stats, per-task CE means), and
``autograd.grad(sum_t L_t, arena)[t] == d L_t / d arena[t]``.
"""

from __future__ import annotations

import math
import os
from typing import Dict, List

import numpy as np
import torch
import torch.nn as nn

from .. import ops
from ..models.vgg import TaskBatchedVGG
from ..utils.seeding import set_torch_seed


class MAMLFewShotClassifier(nn.Module):
    """Reference-compatible public API: ``run_train_iter``,
    ``run_validation_iter``, ``save_model``, ``load_model``
    (``few_shot_learning_system.py:338,371,399,410``)."""

    def __init__(self, im_shape, device, args):
        super().__init__()
        self.args = args
        self.device = device
        self.current_epoch = 0
        self.current_iter = 0
        # im_shape comes in reference form (2, c, h, w) — batch dim ignored
        c, h, w = im_shape[-3], im_shape[-2], im_shape[-1]
        self.im_shape = (c, h, w)
        self.rng = set_torch_seed(seed=args.seed)

        num_steps = args.number_of_training_steps_per_iter
        self.classifier = TaskBatchedVGG(
            im_shape=(c, h, w),
            num_output_classes=args.num_classes_per_set,
            num_stages=args.num_stages,
            num_filters=args.cnn_num_filters,
            max_pooling=bool(args.max_pooling),
            conv_padding=bool(args.conv_padding),
            norm_layer=args.norm_layer,
            per_step_bn_statistics=bool(args.per_step_bn_statistics),
            num_steps=num_steps,
            learnable_bn_gamma=bool(args.learnable_bn_gamma),
            learnable_bn_beta=bool(args.learnable_bn_beta),
            inner_loop_bn_params=bool(args.enable_inner_loop_optimizable_bn_params),
            generator=self.rng,
        )

        # LSLR table: one learnable LR per inner-loop param tensor per step
        # (reference: one nn.Parameter of shape [num_steps+1] per weight,
    # ``inner_loop_optimizers.py:86-91``; here a single [slots, steps+1]
        # table so the fused update can gather it in one go).
        init_lr = float(getattr(args, "init_inner_loop_learning_rate", None)
                        or args.task_learning_rate)
        learnable = bool(args.learnable_per_layer_per_step_inner_loop_learning_rate)
        self.inner_loop_lrs = nn.Parameter(
            torch.full((self.classifier.arena.num_slots, num_steps + 1), init_lr),
            requires_grad=learnable)

        self.to(device)
        # fused multi-tensor Adam: one launch per meta-update, with the
        # reference's imagenet ±10 grad clamp fused in
        from .fused_adam import FusedAdam
        clamp = 10.0 if "imagenet" in args.dataset_name else None
        self.optimizer = FusedAdam(self.trainable_parameters(),
                                   lr=args.meta_learning_rate, grad_clamp=clamp)
        self.dist = None  # set by attach_distributed()
        self.timers = None
        if getattr(args, "enable_phase_timers", False):
            from ..utils.timers import PhaseTimers
            self.timers = PhaseTimers(use_cuda_events=(device.type == "cuda"))

    # ------------------------------------------------------------------
    def attach_distributed(self, dist_ctx) -> None:
        """Attach a parallel.DistContext: meta-gradients are then all-reduced
        (flat bucket over RCCL/xGMI) before each Adam step."""
        self.dist = dist_ctx

    def trainable_parameters(self):
        return [p for p in self.parameters() if p.requires_grad]

    # ------------------------------------------------------------------
    def get_per_step_loss_importance_vector(self) -> torch.Tensor:
        """Annealed MSL importance vector — exact reference formula
        (``few_shot_learning_system.py:83-103``)."""
        n = self.args.number_of_training_steps_per_iter
        msl_epochs = self.args.multi_step_loss_num_epochs
        loss_weights = np.ones(n) * (1.0 / n)
        decay_rate = 1.0 / n / msl_epochs
        min_value = 0.03 / n
        for i in range(n - 1):
            loss_weights[i] = np.maximum(loss_weights[i] - self.current_epoch * decay_rate,
                                         min_value)
        loss_weights[-1] = np.minimum(
            loss_weights[-1] + self.current_epoch * (n - 1) * decay_rate,
            1.0 - (n - 1) * min_value)
        return torch.tensor(loss_weights, dtype=torch.float32, device=self.device)

    def scheduled_meta_lr(self, epoch: int) -> float:
        """Cosine annealing of the outer LR, stepped per epoch with
        T_max=total_epochs (reference: ``few_shot_learning_system.py:69-71,346``)."""
        base = self.args.meta_learning_rate
        eta_min = self.args.min_learning_rate
        t = min(max(epoch, 0), self.args.total_epochs)
        return eta_min + 0.5 * (base - eta_min) * (1.0 + math.cos(math.pi * t / self.args.total_epochs))

    # ------------------------------------------------------------------
    def forward(self, data_batch, epoch: int, use_second_order: bool,
                use_multi_step_loss_optimization: bool, num_steps: int,
                training_phase: bool):
        """Run the full task-batched inner loop + outer loss.

        data_batch: (x_support [B,N,S,c,h,w], x_target [B,N,T,c,h,w],
                     y_support [B,N,S], y_target [B,N,T])
        Returns (losses dict, per_task_target_preds [B, N*T, ways]).
        """
        # real episode batches carry the per-episode seed as a 5th element
        # (reference data.py:478-524); synthetic batches are 4-tuples
        x_support, x_target, y_support, y_target = data_batch[:4]
        x_support = x_support.to(self.device, non_blocking=True)
        x_target = x_target.to(self.device, non_blocking=True)
        y_support = y_support.to(self.device, non_blocking=True)
        y_target = y_target.to(self.device, non_blocking=True)

        T = x_support.shape[0]
        # bf16 activations on GPU (fp32 accumulate in the MFMA kernels and
        # fp32 master weights in the arena); fp32 on CPU
        act_dtype = torch.bfloat16 if (
            self.device.type == "cuda"
            and getattr(self.args, "compute_dtype", "bf16") == "bf16"
        ) else torch.float32
        # optional mixed-precision inner loop: support passes (and their
        # create_graph backward chains, where second-order error
        # accumulates) in fp32, target passes bf16
        sup_dtype = torch.float32 if getattr(
            self.args, "fp32_support_pass", False) else act_dtype
        xs = x_support.reshape(T, -1, *x_support.shape[-3:]).to(sup_dtype)
        xt = x_target.reshape(T, -1, *x_target.shape[-3:]).to(act_dtype)
        ys = y_support.reshape(T, -1).long()
        yt = y_target.reshape(T, -1).long()

        msl_active = (use_multi_step_loss_optimization and training_phase
                      and epoch < self.args.multi_step_loss_num_epochs)
        importance = self.get_per_step_loss_importance_vector()

        arena = self.classifier.init_arena(T)
        per_step_target_loss: List[torch.Tensor] = []
        final_logits = None

        for step in range(num_steps):
            support_logits = self.classifier(xs, num_step=step, arena=arena,
                                             training=True,
                                             backup_running_statistics=(step == 0))
            support_loss = ops.softmax_cross_entropy(support_logits, ys)  # [T]
            # first-order / eval: the support graph is not traversed again,
            # so its activation buffers are freed here (retain only under
            # create_graph, where it is implied)
            grad = torch.autograd.grad(support_loss.sum(), arena,
                                       create_graph=use_second_order)[0]
            lr_vec = self.classifier.lr_vector(self.inner_loop_lrs, step)
            arena = ops.lslr_update(arena, grad, lr_vec)

            if msl_active:
                tl = self.classifier(xt, num_step=step, arena=arena, training=True)
                per_step_target_loss.append(ops.softmax_cross_entropy(tl, yt))
                if step == num_steps - 1:
                    final_logits = tl
            elif step == num_steps - 1:
                # eval / final-step-only: target loss is report-only unless
                # training — skip graph construction at eval
                with torch.set_grad_enabled(training_phase):
                    final_logits = self.classifier(xt, num_step=step, arena=arena,
                                                   training=True)
                    per_step_target_loss.append(
                        ops.softmax_cross_entropy(final_logits, yt))

        if msl_active:
            step_losses = torch.stack(per_step_target_loss, dim=0)        # [steps, T]
            task_losses = (importance.unsqueeze(1) * step_losses).sum(0)  # [T]
        else:
            task_losses = per_step_target_loss[-1]                        # [T]

        loss = task_losses.mean()
        with torch.no_grad():
            preds = final_logits.argmax(dim=-1)
            accuracy_per_task = (preds == yt).float().mean(dim=1)
            accuracy = accuracy_per_task.mean()

        if not training_phase:
            self.classifier.restore_backup_stats()

        losses = {"loss": loss, "accuracy": float(accuracy.item())}
        if msl_active:
            for i, wgt in enumerate(importance.tolist()):
                losses[f"loss_importance_vector_{i}"] = wgt
        return losses, final_logits.detach()

    # ------------------------------------------------------------------
    def train_forward_prop(self, data_batch, epoch: int):
        use_second_order = (self.args.second_order and
                            epoch > self.args.first_order_to_second_order_epoch)
        return self.forward(data_batch, epoch,
                            use_second_order=use_second_order,
                            use_multi_step_loss_optimization=self.args.use_multi_step_loss_optimization,
                            num_steps=self.args.number_of_training_steps_per_iter,
                            training_phase=True)

    def evaluation_forward_prop(self, data_batch, epoch: int):
        return self.forward(data_batch, epoch, use_second_order=False,
                            use_multi_step_loss_optimization=True,
                            num_steps=self.args.number_of_evaluation_steps_per_iter,
                            training_phase=False)

    # ------------------------------------------------------------------
    def meta_update(self, loss: torch.Tensor) -> None:
        """One outer (meta) update: backward through the whole unrolled
        task-batch graph, flat all-reduce across ranks, optional grad clamp
        (±10 for imagenet datasets, ``few_shot_learning_system.py:332-335``),
        Adam step."""
        self.optimizer.zero_grad(set_to_none=True)
        loss.backward()
        if self.dist is not None and self.dist.world_size > 1:
            self.dist.all_reduce_gradients(self.trainable_parameters())
        # imagenet grad clamp is fused into the Adam kernel (FusedAdam)
        self.optimizer.step()

    def _chunked_train_step(self, data_batch, epoch: int, chunk: int):
        """Per-chunk forward+backward with gradient accumulation.

        The outer loss is a mean over tasks, so running ``chunk`` tasks at a
        time and scaling each chunk's backward by ``chunk_size/total`` is
        mathematically identical to the single big graph (SURVEY.md §7
        "memory of the unrolled tape") — it caps the live second-order
        activation tape to one chunk, letting the meta-batch exceed what a
        single unrolled graph fits in HBM."""
        total = data_batch[0].shape[0]
        self.optimizer.zero_grad(set_to_none=True)
        agg: Dict[str, float] = {}
        weights_sum = 0.0
        preds = []
        overlap = (self.dist is not None and self.dist.world_size > 1)
        if overlap:
            self.dist.start_overlapped_reduction(self.trainable_parameters())
        for lo in range(0, total, chunk):
            hi = min(lo + chunk, total)
            sub = tuple(x[lo:hi] for x in data_batch)
            losses, p = self.train_forward_prop(sub, epoch)
            wgt = (hi - lo) / total
            (losses["loss"] * wgt).backward()
            if overlap:
                # north-star overlap: all-reduce this chunk's gradient
                # contribution on the comm stream while the next chunk's
                # inner loop computes (all-reduce is linear, so per-chunk
                # reduction == reduction of the accumulated sum)
                self.dist.reduce_chunk_gradients(self.trainable_parameters())
            # weighted mean over chunks for every scalar entry (the
            # importance-vector entries are identical across chunks, so the
            # weighted mean reproduces them exactly)
            for k, v in losses.items():
                val = float(v.detach()) if torch.is_tensor(v) else float(v)
                agg[k] = agg.get(k, 0.0) + val * wgt
            weights_sum += wgt
            preds.append(p)
        if overlap:
            self.dist.finish_overlapped_reduction(self.trainable_parameters())
        self.optimizer.step()
        return agg, torch.cat(preds, dim=0)

    def run_train_iter(self, data_batch, epoch):
        epoch = int(epoch)
        if self.current_epoch != epoch:
            self.current_epoch = epoch
        if not self.training:
            self.train()
        lr = self.scheduled_meta_lr(epoch)
        for group in self.optimizer.param_groups:
            group["lr"] = lr
        chunk = int(getattr(self.args, "task_chunk_size", 0) or 0)
        if chunk > 0 and chunk < data_batch[0].shape[0]:
            losses, per_task_preds = self._chunked_train_step(data_batch, epoch, chunk)
            losses["learning_rate"] = lr
            self.current_iter += 1
            return losses, per_task_preds
        if self.timers is not None:
            with self.timers.phase("inner_loop_fwd"):
                losses, per_task_preds = self.train_forward_prop(data_batch, epoch)
            with self.timers.phase("outer_bwd_and_opt"):
                self.meta_update(losses["loss"])
        else:
            losses, per_task_preds = self.train_forward_prop(data_batch, epoch)
            self.meta_update(losses["loss"])
        losses["loss"] = float(losses["loss"].detach().item())
        losses["learning_rate"] = lr
        self.current_iter += 1
        return losses, per_task_preds

    def run_validation_iter(self, data_batch):
        if self.training:
            self.eval()
        losses, per_task_preds = self.evaluation_forward_prop(data_batch, self.current_epoch)
        losses["loss"] = float(losses["loss"].detach().item())
        return losses, per_task_preds

    # ------------------------------------------------------------------
    # checkpointing — same file layout AND key naming as the reference: one
    # torch pickle per epoch holding the experiment ``state`` dict with
    # state['network'] using reference-style names
    # (``few_shot_learning_system.py:399-424``):
    #   classifier.layer_dict.conv{i}.conv.{weight,bias}
    #   classifier.layer_dict.conv{i}.norm_layer.{weight,bias,running_*}
    #   classifier.layer_dict.linear.{weights,bias}
    #   inner_loop_optimizer.names_learning_rates_dict.layer_dict-...-weight
    # The flat arena is expanded on save and re-packed on load; the linear
    # weight is permuted between our NHWC flatten order and the reference's
    # NCHW flatten order so checkpoints are interchangeable.
    def _linear_to_reference(self, w: torch.Tensor) -> torch.Tensor:
        cls = self.classifier
        if not cls.max_pooling:
            return w
        h, w_sp = cls.final_spatial
        ways = w.shape[0]
        return (w.view(ways, h, w_sp, cls.num_filters).permute(0, 3, 1, 2)
                .reshape(ways, -1))

    def _linear_from_reference(self, w: torch.Tensor) -> torch.Tensor:
        cls = self.classifier
        if not cls.max_pooling:
            return w
        h, w_sp = cls.final_spatial
        ways = w.shape[0]
        return (w.view(ways, cls.num_filters, h, w_sp).permute(0, 2, 3, 1)
                .reshape(ways, -1))

    def reference_state_dict(self) -> Dict[str, torch.Tensor]:
        cls = self.classifier
        out: Dict[str, torch.Tensor] = {}
        views = cls.arena.views(self.classifier.theta.detach())
        for name, v in views.items():
            t = v.clone()
            if name == "layer_dict.linear.weights":
                t = self._linear_to_reference(t)
            out["classifier." + name] = t
        if cls.norm_layer_type == "batch_norm":
            for i in range(cls.num_stages):
                prefix = f"classifier.layer_dict.conv{i}.norm_layer."
                if not cls.inner_loop_bn_params:
                    out[prefix + "weight"] = getattr(cls, f"bn_weight_{i}").detach().clone()
                    out[prefix + "bias"] = getattr(cls, f"bn_bias_{i}").detach().clone()
                out[prefix + "running_mean"] = getattr(cls, f"bn_running_mean_{i}").clone()
                out[prefix + "running_var"] = getattr(cls, f"bn_running_var_{i}").clone()
        for spec in cls.arena.specs:
            key = ("inner_loop_optimizer.names_learning_rates_dict."
                   + spec.name.replace(".", "-"))
            out[key] = self.inner_loop_lrs.detach()[spec.slot].clone()
        return out

    def load_reference_state_dict(self, sd: Dict[str, torch.Tensor]) -> None:
        cls = self.classifier
        named = {}
        for spec in cls.arena.specs:
            t = sd["classifier." + spec.name]
            if spec.name == "layer_dict.linear.weights":
                t = self._linear_from_reference(t)
            named[spec.name] = t
        with torch.no_grad():
            self.classifier.theta.copy_(cls.arena.pack(named).to(self.device))
            if cls.norm_layer_type == "batch_norm":
                for i in range(cls.num_stages):
                    prefix = f"classifier.layer_dict.conv{i}.norm_layer."
                    if not cls.inner_loop_bn_params and prefix + "weight" in sd:
                        getattr(cls, f"bn_weight_{i}").copy_(sd[prefix + "weight"])
                        getattr(cls, f"bn_bias_{i}").copy_(sd[prefix + "bias"])
                    if prefix + "running_mean" in sd:
                        getattr(cls, f"bn_running_mean_{i}").copy_(sd[prefix + "running_mean"])
                        getattr(cls, f"bn_running_var_{i}").copy_(sd[prefix + "running_var"])
            for spec in cls.arena.specs:
                key = ("inner_loop_optimizer.names_learning_rates_dict."
                       + spec.name.replace(".", "-"))
                if key in sd:
                    self.inner_loop_lrs[spec.slot].copy_(sd[key])

    def save_model(self, model_save_dir: str, state: Dict) -> None:
        state = dict(state)
        state["network"] = self.reference_state_dict()
        state["optimizer"] = self.optimizer.state_dict()
        torch.save(state, f=model_save_dir)

    def load_model(self, model_save_dir: str, model_name: str, model_idx) -> Dict:
        filepath = os.path.join(model_save_dir, f"{model_name}_{model_idx}")
        state = torch.load(filepath, map_location=self.device, weights_only=False)
        self.load_reference_state_dict(state["network"])
        if "optimizer" in state:
            self.optimizer.load_state_dict(state["optimizer"])
        return state
