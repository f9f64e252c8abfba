"""Fused SGD-momentum optimizer.

One multi-tensor HIP kernel per step consumes a device-resident chunk table
covering every parameter (the reference launches one ATen kernel per tensor —
161 launches for ResNet-50). The table is built once and cached across steps
(no per-step host work; the step is hipGraph-capturable). bf16 parameters
carry fp32 master weights and fp32 momentum so the update math never loses
precision (SURVEY.md §7 hard-part 4).

CPU / no-extension path falls back to torch.optim.SGD semantics exactly
(reference utils.py:187-196: momentum + nesterov + weight decay + dampening).
"""

import torch
from torch.optim import SGD

from .dispatch import hip_op_available, ext

_CHUNK = 16384


class HIPSGD(SGD):
    """torch.optim.SGD drop-in whose GPU step is one fused multi-tensor kernel."""

    def _build_table(self, group):
        recs = []
        key = []
        for p in group["params"]:
            if p.grad is None:
                continue
            state = self.state[p]
            if "momentum_buffer" not in state:
                state["momentum_buffer"] = torch.zeros_like(
                    p, dtype=torch.float32)
                if p.dtype in (torch.bfloat16, torch.float16):
                    state["master"] = p.detach().float().clone()
            master = state.get("master", p)
            flags = (1 if p.dtype == torch.bfloat16 else 0) | (
                2 if p.grad.dtype == torch.bfloat16 else 0)
            key.append((p.data_ptr(), p.grad.data_ptr()))
            n = p.numel()
            off = 0
            while off < n:
                cnt = min(_CHUNK, n - off)
                recs.append([p.grad.data_ptr(), p.data_ptr(),
                             state["momentum_buffer"].data_ptr(),
                             master.data_ptr(), off,
                             cnt | (flags << 32)])
                off += _CHUNK
        device = group["params"][0].device
        # pinned staging + async copy. The pinned buffer and device table are
        # allocated on the FIRST build (outside any hipGraph capture, during
        # warmup); capture-time rebuilds (grad pointers change under
        # zero_grad(set_to_none=True)) only do host writes + one capturable
        # async H2D copy into the existing device tensor.
        cpu = torch.tensor(recs, dtype=torch.int64)
        pin = group.get("_hip_table_pin")
        table = group.get("_hip_table_dev")
        if pin is None or pin.numel() != cpu.numel():
            pin = cpu.pin_memory()
            table = pin.to(device, non_blocking=True)
            group["_hip_table_pin"] = pin
            group["_hip_table_dev"] = table
        else:
            pin.copy_(cpu)
            table.copy_(pin, non_blocking=True)
        return table, tuple(key)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        if not hip_op_available("sgd_step"):
            return self._fallback_step(loss)

        for group in self.param_groups:
            cache = group.get("_hip_table")
            key = tuple((p.data_ptr(),
                         p.grad.data_ptr() if p.grad is not None else 0)
                        for p in group["params"])
            if cache is None or cache[1] != key:
                table, _ = self._build_table(group)
                group["_hip_table"] = (table, key)
            else:
                table = cache[0]
            if table.numel():
                ext().sgd_step(table, group["lr"], group["momentum"],
                               group["dampening"], group["weight_decay"],
                               group["nesterov"])
        return loss

    def _drop_table_cache(self):
        for group in self.param_groups:
            group.pop("_hip_table", None)
            group.pop("_hip_table_pin", None)
            group.pop("_hip_table_dev", None)

    def state_dict(self):
        # drop the pointer cache from serialized state
        self._drop_table_cache()
        return super().state_dict()

    def load_state_dict(self, state_dict):
        # loading replaces momentum/master tensors; the cached device table
        # embeds their raw pointers, so it must be rebuilt on the next step
        super().load_state_dict(state_dict)
        self._drop_table_cache()
        # loaded state tensors may arrive in the wrong dtype/device; the
        # table kernel requires fp32 momentum/master on the param's device
        for group in self.param_groups:
            for p in group["params"]:
                state = self.state.get(p)
                if not state:
                    continue
                for k in ("momentum_buffer", "master"):
                    if k in state and state[k] is not None:
                        state[k] = state[k].to(device=p.device,
                                               dtype=torch.float32)

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
