"""Adam with shared-memory state for the CPU HogWild-parity mode.

Capability parity with /root/reference/shared_adam.py:3-17: moment buffers
are created eagerly and moved to POSIX shared memory so forked worker
processes update one optimizer state (A3C-style).  Kept quirks (SURVEY.md §2a
component 6): default ``betas=(0.9, 0.9)`` (the reference's non-standard
second beta) and a per-process plain-int step count.

This optimizer exists for the reference-parity shared-memory mode
(parallel/hogwild.py) and for CPU tests.  The MI355X learner path instead
uses the fused multi-tensor Adam HIP kernel (ops/hip/engine.hip, K8 in
SURVEY.md §2c) over a flat parameter slab.
"""

from __future__ import annotations

import torch


class SharedAdam(torch.optim.Adam):
    def __init__(self, params, lr: float = 1e-3, betas=(0.9, 0.9),
                 eps: float = 1e-8, weight_decay: float = 0):
        super().__init__(params, lr=lr, betas=betas, eps=eps,
                         weight_decay=weight_decay)
        for group in self.param_groups:
            for p in group["params"]:
                state = self.state[p]
                state["step"] = 0
                state["exp_avg"] = torch.zeros_like(p.data)
                state["exp_avg_sq"] = torch.zeros_like(p.data)
                state["exp_avg"].share_memory_()
                state["exp_avg_sq"].share_memory_()

    @torch.no_grad()
    def step(self, closure=None):
        """Plain Adam update against the (shared) moment buffers.

        Written out explicitly rather than deferring to torch's fused/foreach
        paths: those assume ``state['step']`` is a tensor, while parity
        requires a plain per-process int.
        """
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            beta1, beta2 = group["betas"]
            lr, eps, wd = group["lr"], group["eps"], group["weight_decay"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                grad = p.grad
                if wd != 0:
                    grad = grad.add(p, alpha=wd)
                state = self.state[p]
                state["step"] += 1
                t = state["step"]
                m, v = state["exp_avg"], state["exp_avg_sq"]
                m.mul_(beta1).add_(grad, alpha=1 - beta1)
                v.mul_(beta2).addcmul_(grad, grad, value=1 - beta2)
                bc1 = 1 - beta1 ** t
                bc2 = 1 - beta2 ** t
                denom = (v / bc2).sqrt_().add_(eps)
                p.addcdiv_(m, denom, value=-lr / bc1)
        return loss
