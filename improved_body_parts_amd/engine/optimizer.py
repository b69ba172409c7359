"""Fused SGD with momentum + weight decay.

Replaces the Apex FusedSGD alternative the reference comments about
(train_distributed.py:121-125). On MI355X the step runs as ONE multi-tensor HIP
kernel over chunked parameter/grad/momentum pointers (ops/csrc/sgd.hip), with
fp32 master weights maintained inside the optimizer when the model parameters
are bf16 (native-bf16 replacement for Apex O1). On CPU it falls back to the
mathematically identical eager loop.

Update rule (matches torch.optim.SGD, which the reference uses):
    g = grad + wd * w ; m = mu * m + g ; w -= lr * m
"""
from __future__ import annotations

import torch
from torch.optim import Optimizer

from ..ops._backend import hip_available


class FusedSGD(Optimizer):
    def __init__(self, params, lr=2.5e-5, momentum=0.9, weight_decay=0.0):
        defaults = dict(lr=lr, momentum=momentum, weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            lr = group["lr"]
            mu = group["momentum"]
            wd = group["weight_decay"]
            hip_batch = []  # (param_bf16, grad, momentum_fp32, master_fp32)
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if "momentum_buffer" not in state:
                    state["momentum_buffer"] = torch.zeros_like(
                        p, dtype=torch.float32, memory_format=torch.preserve_format)
                    if p.dtype != torch.float32:
                        state["master"] = p.detach().float().clone()
                buf = state["momentum_buffer"]
                master = state.get("master")
                if p.is_cuda and hip_available():
                    hip_batch.append((p, p.grad, buf, master))
                    continue
                # eager path (CPU oracle / debugging)
                w = master if master is not None else p
                g = p.grad.float()
                if wd != 0:
                    g = g.add(w, alpha=wd)
                buf.mul_(mu).add_(g)
                w.add_(buf, alpha=-lr)
                if master is not None:
                    p.copy_(master.to(p.dtype))
            if hip_batch:
                from ..ops import optim as _optim
                _optim.fused_sgd_step(hip_batch, lr=lr, momentum=mu,
                                      weight_decay=wd)
        return loss
