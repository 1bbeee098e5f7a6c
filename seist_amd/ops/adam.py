"""Fused Adam/AdamW optimizer (K17 of SURVEY §2.4).

On GPU the whole update is one kernel launch per parameter dtype over
device-resident chunk metadata packed once (pointers are stable across
steps; repacked automatically if any grad pointer moves). Keeps fp32
master weights and fp32 moments when parameters are bf16, so bf16
training follows fp32 Adam trajectories to bf16 rounding.

CPU path is the same math in plain PyTorch (used by tests as reference).
"""

from typing import Optional

import torch

from . import ext, has_ext


class FusedAdam(torch.optim.Optimizer):
    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.0, adamw=False):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay, adamw=adamw)
        super().__init__(params, defaults)
        self._packed = None  # list of (meta, sample, has_master, ptr_sig)

    def load_state_dict(self, state_dict):
        # The packed chunk metadata bakes m/v/master device pointers in; a
        # restored state has fresh tensors, so force a repack.
        super().load_state_dict(state_dict)
        self._packed = None

    def _collect(self, group):
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
            elif torch.is_tensor(state["step"]):
                # torch.optim.Adam checkpoints store step as a 0-dim tensor
                state["step"] = int(state["step"])
            state["step"] += 1
            params.append(p)
            grads.append(p.grad)
            ms.append(state["exp_avg"])
            vs.append(state["exp_avg_sq"])
            masters.append(state.get("master", None))
            steps.append(state["step"])
        return params, grads, ms, vs, masters, steps

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
            params, grads, ms, vs, masters, steps = self._collect(group)
            if not params:
                continue

            native = (params[0].is_cuda and has_ext()
                      and hasattr(ext(), "adam_step_packed")
                      and len(set(steps)) == 1)
            if native:
                self._native_step(params, grads, ms, vs, masters, steps[0],
                                  lr, beta1, beta2, eps, wd, adamw)
            else:
                self._torch_step(params, grads, ms, vs, masters, steps,
                                 lr, beta1, beta2, eps, wd, adamw)
        return loss

    def _native_step(self, params, grads, ms, vs, masters, step, lr, beta1,
                     beta2, eps, wd, adamw):
        bc1 = 1.0 - beta1 ** step
        bc2 = 1.0 - beta2 ** step
        # partition by dtype (bf16 convs + fp32 norms can share one group)
        parts = {}
        for p, g, m, v, mst in zip(params, grads, ms, vs, masters):
            parts.setdefault(p.dtype, []).append((p, g, m, v, mst))
        sig = tuple(
            (p.data_ptr(), g.data_ptr(), m.data_ptr(), v.data_ptr(),
             mst.data_ptr() if mst is not None else 0)
            for p, g, m, v, mst in
            [t for lst in parts.values() for t in lst])
        if self._packed is None or self._packed[0] != sig:
            packed = []
            for dtype, lst in parts.items():
                ps = [t[0] for t in lst]
                gs = [t[1] for t in lst]
                mms = [t[2] for t in lst]
                vvs = [t[3] for t in lst]
                has_master = dtype != torch.float32
                msts = [t[4] if t[4] is not None else t[0] for t in lst]
                meta = ext().adam_pack(ps, gs, mms, vvs, msts, has_master)
                packed.append((meta, ps[0], has_master))
            self._packed = (sig, packed)
        for meta, sample, has_master in self._packed[1]:
            ext().adam_step_packed(meta, sample, has_master, lr, beta1,
                                   beta2, eps, wd, bc1, bc2, adamw)

    @staticmethod
    def _torch_step(params, grads, ms, vs, masters, steps, lr, beta1, beta2,
                    eps, wd, adamw):
        for p, g, m, v, master, step in zip(params, grads, ms, vs, masters,
                                            steps):
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
