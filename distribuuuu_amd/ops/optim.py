"""Fused SGD-momentum optimizer.

One multi-tensor HIP kernel per step consumes a device-resident pointer table
covering every parameter (the reference launches one ATen kernel per tensor —
161 launches for ResNet-50). bf16 parameters carry fp32 master weights and fp32
momentum so the update math never loses precision (the bf16-end-to-end accuracy
requirement of SURVEY.md §7 hard-part 4).

CPU / no-extension path falls back to torch.optim.SGD semantics exactly
(reference utils.py:187-196: momentum + nesterov + weight decay + dampening).
"""

import torch
from torch.optim import SGD

from .dispatch import hip_op_available, ext


class HIPSGD(SGD):
    """torch.optim.SGD drop-in whose GPU step is one fused multi-tensor kernel."""

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        if not hip_op_available("sgd_step"):
            return self._fallback_step(loss)

        for group in self.param_groups:
            params, grads, moms, masters = [], [], [], []
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if "momentum_buffer" not in state:
                    state["momentum_buffer"] = torch.zeros_like(
                        p, dtype=torch.float32)
                    if p.dtype in (torch.bfloat16, torch.float16):
                        state["master"] = p.detach().float().clone()
                params.append(p)
                grads.append(p.grad)
                moms.append(state["momentum_buffer"])
                masters.append(state.get("master", p))  # fp32 view or self
            if params:
                ext().sgd_step(
                    params, grads, moms, masters,
                    group["lr"], group["momentum"], group["dampening"],
                    group["weight_decay"], group["nesterov"],
                )
        return loss

    def _fallback_step(self, loss):
        """Plain SGD math, with fp32 master handling for bf16 params."""
        for group in self.param_groups:
            lr = group["lr"]
            mu = group["momentum"]
            damp = group["dampening"]
            wd = group["weight_decay"]
            nesterov = group["nesterov"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                low_prec = p.dtype in (torch.bfloat16, torch.float16)
                if "momentum_buffer" not in state:
                    state["momentum_buffer"] = torch.zeros_like(
                        p, dtype=torch.float32)
                    if low_prec:
                        state["master"] = p.detach().float().clone()
                g = p.grad.float()
                w = state["master"] if low_prec else p.data
                if wd != 0:
                    g = g.add(w, alpha=wd)
                buf = state["momentum_buffer"]
                buf.mul_(mu).add_(g, alpha=1 - damp)
                upd = g.add(buf, alpha=mu) if nesterov else buf
                w.add_(upd, alpha=-lr)
                if low_prec:
                    p.data.copy_(w.to(p.dtype))
        return loss
