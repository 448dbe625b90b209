"""Fused multi-tensor Adam for the meta-update.

State-dict compatible subclass of ``torch.optim.Adam`` whose ``step()`` is
ONE kernel launch over every trainable tensor (adam.hip), with the
reference's pre-step gradient clamp fused in
(``few_shot_learning_system.py:330-336``: ``grad.clamp_(-10,10)`` for
imagenet datasets followed by Adam).  On CPU the op-layer dispatch runs the
same math eagerly (``ops/reference.py:fused_adam_step``), so checkpoints
and numerics are identical across devices.
"""

from __future__ import annotations

from typing import Optional

import torch

from .. import ops


class FusedAdam(torch.optim.Adam):
    def __init__(self, params, lr: float = 1e-3, betas=(0.9, 0.999),
                 eps: float = 1e-8, weight_decay: float = 0.0,
                 grad_clamp: Optional[float] = None):
        super().__init__(params, lr=lr, betas=betas, eps=eps,
                         weight_decay=weight_decay, amsgrad=False)
        self.grad_clamp = grad_clamp

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            beta1, beta2 = group["betas"]
            # bucket by step count (normally one bucket: params advance
            # together; differing counts only after partial state loads)
            buckets = {}
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = torch.zeros((), dtype=torch.float32)
                    state["exp_avg"] = torch.zeros_like(p)
                    state["exp_avg_sq"] = torch.zeros_like(p)
                state["step"] += 1
                key = int(state["step"].item())
                buckets.setdefault(key, []).append(p)
            for step_count, plist in buckets.items():
                ops.fused_adam_step(
                    params=[p.data for p in plist],
                    grads=[p.grad for p in plist],
                    exp_avgs=[self.state[p]["exp_avg"] for p in plist],
                    exp_avg_sqs=[self.state[p]["exp_avg_sq"] for p in plist],
                    step=step_count, lr=group["lr"], beta1=beta1, beta2=beta2,
                    eps=group["eps"], weight_decay=group["weight_decay"],
                    clamp=self.grad_clamp)
        return loss
