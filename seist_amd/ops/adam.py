"""Fused Adam/AdamW optimizer (K17 of SURVEY §2.4).

Single multi-tensor HIP kernel per step on GPU (the SeisT models have
<= 1.1 M parameters spread over hundreds of small tensors — eager Adam is
pure launch overhead). Keeps fp32 master weights and fp32 moments when the
model parameters are bf16, so bf16 training matches fp32 Adam trajectories
to bf16 rounding.

CPU path is the same math in plain PyTorch (used for tests).
"""

import math
from typing import Optional

import torch

from . import ext, has_ext


class FusedAdam(torch.optim.Optimizer):
    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.0, adamw=False):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay, adamw=adamw)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure: Optional[callable] = None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        for group in self.param_groups:
            lr = group["lr"]
            beta1, beta2 = group["betas"]
            eps = group["eps"]
            wd = group["weight_decay"]
            adamw = group["adamw"]

            params, grads, ms, vs, masters, steps = [], [], [], [], [], []
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(p, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(p, dtype=torch.float32)
                    if p.dtype != torch.float32:
                        state["master"] = p.detach().clone().float()
                state["step"] += 1
                params.append(p)
                grads.append(p.grad)
                ms.append(state["exp_avg"])
                vs.append(state["exp_avg_sq"])
                masters.append(state.get("master", None))
                steps.append(state["step"])

            if not params:
                continue

            if params[0].is_cuda and has_ext() and hasattr(ext(), "adam_step"):
                # steps are uniform within a group after the first call
                step = steps[0]
                bc1 = 1.0 - beta1 ** step
                bc2 = 1.0 - beta2 ** step
                master_list = [m if m is not None else p
                               for p, m in zip(params, masters)]
                has_master = masters[0] is not None
                ext().adam_step(params, grads, ms, vs, master_list,
                                has_master, lr, beta1, beta2, eps, wd,
                                bc1, bc2, adamw)
            else:
                for p, g, m, v, master, step in zip(params, grads, ms, vs,
                                                    masters, steps):
                    w = master if master is not None else p
                    g32 = g.float()
                    if wd != 0.0:
                        if adamw:
                            w.mul_(1.0 - lr * wd)
                        else:
                            g32 = g32.add(w, alpha=wd)
                    m.mul_(beta1).add_(g32, alpha=1 - beta1)
                    v.mul_(beta2).addcmul_(g32, g32, value=1 - beta2)
                    bc1 = 1.0 - beta1 ** step
                    bc2 = 1.0 - beta2 ** step
                    denom = (v / bc2).sqrt_().add_(eps)
                    w.addcdiv_(m, denom, value=-lr / bc1)
                    if master is not None:
                        p.copy_(w.to(p.dtype))
        return loss
