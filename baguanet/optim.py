"""FusedSGD — torch.optim.SGD semantics with the whole update in ONE
gfx950 kernel launch (csrc/hip/multi_tensor.hip multi_sgd_kernel).

torch's eager SGD walks param groups with foreach chains (several kernel
launches per step); on MI355X the fused multi-tensor kernel applies
weight-decay + momentum + update for every parameter in a single
grid-strided launch.  fp32 parameters on CUDA take the fused path; other
dtypes/devices fall back to torch.optim.SGD-equivalent eager math, so
the optimizer is always usable (CPU tests, bf16 experiments).
"""

from __future__ import annotations

from typing import Iterable

import torch


class FusedSGD(torch.optim.Optimizer):
    def __init__(self, params: Iterable[torch.nn.Parameter], lr: float,
                 momentum: float = 0.0, weight_decay: float = 0.0,
                 nesterov: bool = False):
        if nesterov and momentum <= 0:
            raise ValueError("nesterov requires momentum > 0")
        defaults = dict(lr=lr, momentum=momentum,
                        weight_decay=weight_decay, nesterov=nesterov)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            fused_p, fused_g, fused_m = [], [], []
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if "momentum_buffer" not in state:
                    state["momentum_buffer"] = torch.zeros_like(p)
                use_fused = (
                    p.is_cuda
                    and p.dtype == torch.float32
                    and p.grad.dtype == torch.float32
                    and p.is_contiguous()
                    and p.grad.is_contiguous()
                )
                if use_fused:
                    fused_p.append(p)
                    fused_g.append(p.grad)
                    fused_m.append(state["momentum_buffer"])
                else:
                    self._eager_step(p, state["momentum_buffer"], group)
            if fused_p:
                from . import ops

                ops.fused_sgd(
                    fused_p, fused_g, fused_m, group["lr"],
                    group["momentum"], group["weight_decay"],
                    group["nesterov"],
                )
        return loss

    @staticmethod
    def _eager_step(p, buf, group):
        g = p.grad
        if group["weight_decay"]:
            g = g.add(p, alpha=group["weight_decay"])
        buf.mul_(group["momentum"]).add_(g)
        if group["nesterov"]:
            g = g.add(buf, alpha=group["momentum"])
        else:
            g = buf
        p.add_(g, alpha=-group["lr"])
