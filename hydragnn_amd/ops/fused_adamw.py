"""Flat-buffer AdamW: the whole model's update in one HIP kernel.

``torch.optim.AdamW(foreach=True)`` still issues ~10 launches over a
list of tensors; for the launch-bound captured train step
(train/captured.py) this optimizer flattens all fp32 parameters into
ONE contiguous buffer at construction (parameters become views), keeps
a matching flat gradient buffer that autograd accumulates into, and
steps with the single fused kernel (csrc/fused_adamw.hip).  On CPU it
applies identical math with flat torch ops, so the trajectory is
testable against torch.optim.AdamW without a GPU.

Construct BEFORE any DDP wrapping so reducer bucket views are built
over the flattened storages.

Role parity: the reference trains with stock torch.optim selected by
``utils/optimizer`` (reference hydragnn/utils/optimizer/optimizer.py);
this fused optimizer is the MI355X-native drop-in used by the
launch-bound captured path (select_optimizer still provides every
reference optimizer type).
"""

from __future__ import annotations

import torch

from ._extension import get_extension, use_eager


class FusedAdamW(torch.optim.Optimizer):
    def __init__(self, params, lr: float = 1e-3,
                 betas=(0.9, 0.999), eps: float = 1e-8,
                 weight_decay: float = 1e-2):
        params = [p for p in params if p.requires_grad]
        if not params:
            raise ValueError("FusedAdamW: no parameters")
        dtypes = {p.dtype for p in params}
        if dtypes == {torch.float32}:
            self.param_dtype = torch.float32
        elif dtypes == {torch.bfloat16}:
            # pure-bf16 training: bf16 working params + fp32 master
            self.param_dtype = torch.bfloat16
        else:
            raise TypeError(
                "FusedAdamW supports uniform fp32 or bf16 parameters")
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay)
        super().__init__(params, defaults)

        device = params[0].device
        total = sum(p.numel() for p in params)
        self.flat_param = torch.empty(total, dtype=self.param_dtype,
                                      device=device)
        self.flat_grad = torch.zeros(total, dtype=self.param_dtype,
                                     device=device)
        self.exp_avg = torch.zeros(total, dtype=torch.float32,
                                   device=device)
        self.exp_avg_sq = torch.zeros(total, dtype=torch.float32,
                                      device=device)
        self.step_t = torch.zeros(1, dtype=torch.float32,
                                  device=device)
        off = 0
        self._params = params
        with torch.no_grad():
            for p in params:
                n = p.numel()
                self.flat_param[off:off + n].copy_(p.reshape(-1))
                p.data = self.flat_param[off:off + n].view_as(p)
                p.grad = self.flat_grad[off:off + n].view_as(p)
                off += n
        self.master = (self.flat_param.float()
                       if self.param_dtype == torch.bfloat16 else None)

    def zero_grad(self, set_to_none: bool = True):
        # grads are views into the flat buffer: never drop them
        self.flat_grad.zero_()

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        g = self.param_groups[0]
        lr, (b1, b2) = g["lr"], g["betas"]
        eps, wd = g["eps"], g["weight_decay"]
        if self.flat_param.is_cuda and not use_eager():
            ext = get_extension(required=True)
            if self.master is None:
                ext.fused_adamw(self.flat_param, self.flat_grad,
                                self.exp_avg, self.exp_avg_sq,
                                self.step_t, lr, b1, b2, eps, wd)
            else:
                ext.fused_adamw_bf16(self.flat_param, self.flat_grad,
                                     self.master, self.exp_avg,
                                     self.exp_avg_sq, self.step_t,
                                     lr, b1, b2, eps, wd)
            return loss
        # CPU / eager fallback: identical math on the flat buffers
        self.step_t += 1
        t = float(self.step_t.item())
        work = self.master if self.master is not None \
            else self.flat_param
        g32 = self.flat_grad.float() if self.master is not None \
            else self.flat_grad
        work.mul_(1.0 - lr * wd)
        self.exp_avg.mul_(b1).add_(g32, alpha=1.0 - b1)
        self.exp_avg_sq.mul_(b2).addcmul_(g32, g32, value=1.0 - b2)
        bc1 = 1.0 - b1 ** t
        bc2 = 1.0 - b2 ** t
        denom = (self.exp_avg_sq / bc2).sqrt_().add_(eps)
        work.addcdiv_(self.exp_avg / bc1, denom, value=-lr)
        if self.master is not None:
            self.flat_param.copy_(work)
        return loss

    def state_dict(self):
        return {
            "param_groups": self.param_groups,
            "flat": {"exp_avg": self.exp_avg,
                     "exp_avg_sq": self.exp_avg_sq,
                     "step": self.step_t,
                     "master": self.master},
        }

    def load_state_dict(self, sd):
        if sd["flat"].get("master") is not None \
                and self.master is not None:
            self.master.copy_(sd["flat"]["master"])
        self.exp_avg.copy_(sd["flat"]["exp_avg"])
        self.exp_avg_sq.copy_(sd["flat"]["exp_avg_sq"])
        self.step_t.copy_(sd["flat"]["step"])
        for g, gs in zip(self.param_groups, sd["param_groups"]):
            g.update({k: v for k, v in gs.items() if k != "params"})
