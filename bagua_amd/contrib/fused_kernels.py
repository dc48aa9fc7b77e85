"""FusedSGD / FusedAdam — optimizers stepping whole flat param groups
with one hand-written CDNA4 HIP kernel per group.

MI355X-native addition beyond the reference's contiguity fusion
(contrib/fused_optimizer.py): parameters and state are flattened once at
construction, then every step is a single memory-bound kernel
(ops/csrc/kernels.hip fused_sgd_kernel / fused_adam_kernel) instead of
the foreach optimizer's 3-5 launches per group. Math matches
torch.optim.SGD / Adam / AdamW exactly (tests/test_fused_kernels.py).

CPU fallback runs the same math in torch, so the optimizers are usable
(and testable) everywhere.
"""

from typing import Iterable

import torch

from ..ops import native


class _FlatGroupOptimizer(torch.optim.Optimizer):
    """Flattens each param group into (flat_param, flat_grad) views."""

    def _flatten_group(self, group):
        params = [p for p in group["params"] if p.requires_grad]
        if not params:
            return None
        total = sum(p.numel() for p in params)
        p0 = params[0]
        flat_w = torch.zeros(total, dtype=p0.dtype, device=p0.device)
        flat_g = torch.zeros_like(flat_w)
        offset = 0
        for p in params:
            w_view = flat_w.narrow(0, offset, p.numel()).view_as(p)
            w_view.copy_(p.detach())
            p.data = w_view
            g_view = flat_g.narrow(0, offset, p.numel()).view_as(p)
            if p.grad is not None:
                g_view.copy_(p.grad.detach())
            p.grad = g_view
            p._bagua_grad_view = g_view
            offset += p.numel()
        return {"params": params, "flat_w": flat_w, "flat_g": flat_g}

    def _repair_grads(self, rec):
        for p in rec["params"]:
            view = p._bagua_grad_view
            if p.grad is None:
                view.zero_()
                p.grad = view
            elif p.grad.data_ptr() != view.data_ptr():
                view.copy_(p.grad.detach())
                p.grad = view

    # flat-group state lives outside optimizer.state; (de)serialize it so
    # checkpointing (bagua_amd.checkpoint) round-trips momentum/master
    _FLAT_STATE_KEYS = ("momentum_buffer", "master", "exp_avg",
                        "exp_avg_sq")

    def state_dict(self):
        sd = super().state_dict()
        sd["bagua_flat_state"] = [
            {k: (v.clone() if isinstance(v, torch.Tensor) else v)
             for k, v in rec.items()
             if k in self._FLAT_STATE_KEYS
             or k in ("momentum_initialized", "step")}
            if rec is not None else None
            for rec in self._flat
        ]
        return sd

    def load_state_dict(self, state_dict):
        flat_state = state_dict.pop("bagua_flat_state", None)
        super().load_state_dict(state_dict)
        if flat_state is None:
            return
        for rec, saved in zip(self._flat, flat_state):
            if rec is None or saved is None:
                continue
            for k, v in saved.items():
                if isinstance(v, torch.Tensor) and k in rec \
                        and isinstance(rec[k], torch.Tensor):
                    rec[k].copy_(v.to(rec[k].device))
                else:
                    rec[k] = v


class FusedSGD(_FlatGroupOptimizer):
    """Single-kernel SGD over flat param groups. bf16 parameters get an
    fp32 master copy automatically (pure-bf16 training: fwd/bwd and the
    gradient allreduce run at half the bytes, the update stays exact)."""

    def __init__(self, params: Iterable, lr: float, momentum: float = 0.0,
                 dampening: float = 0.0, weight_decay: float = 0.0,
                 nesterov: bool = False):
        defaults = dict(lr=lr, momentum=momentum, dampening=dampening,
                        weight_decay=weight_decay, nesterov=nesterov)
        super().__init__(params, defaults)
        self._flat = [self._flatten_group(g) for g in self.param_groups]
        for g, rec in zip(self.param_groups, self._flat):
            if rec is None:
                continue
            if rec["flat_w"].dtype == torch.bfloat16:
                rec["master"] = rec["flat_w"].float()
            if g["momentum"] != 0:
                rec["momentum_buffer"] = torch.zeros_like(
                    rec.get("master", rec["flat_w"]))
                rec["momentum_initialized"] = False

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group, rec in zip(self.param_groups, self._flat):
            if rec is None:
                continue
            self._repair_grads(rec)
            w, g = rec["flat_w"], rec["flat_g"]
            mu = group["momentum"]
            m = rec.get("momentum_buffer")
            master = rec.get("master")
            if w.is_cuda and native.available() \
                    and w.dtype == torch.float32:
                native.lib().fused_sgd_step(
                    w, g, m if m is not None else g, group["lr"], mu,
                    group["dampening"], group["weight_decay"],
                    group["nesterov"],
                    rec.get("momentum_initialized", False))
            elif w.is_cuda and native.available() \
                    and w.dtype == torch.bfloat16:
                native.lib().fused_sgd_mixed_step(
                    w, g, master, m if m is not None else master,
                    group["lr"], mu, group["dampening"],
                    group["weight_decay"], group["nesterov"],
                    rec.get("momentum_initialized", False))
            else:
                ref = master if master is not None else w
                grad = g.float() if master is not None else g
                if group["weight_decay"] != 0:
                    grad = grad.add(ref, alpha=group["weight_decay"])
                if mu != 0:
                    if rec["momentum_initialized"]:
                        m.mul_(mu).add_(grad, alpha=1 - group["dampening"])
                    else:
                        m.copy_(grad)
                    grad = grad.add(m, alpha=mu) if group["nesterov"] \
                        else m
                ref.add_(grad, alpha=-group["lr"])
                if master is not None:
                    w.copy_(master.to(w.dtype))
            if mu != 0:
                rec["momentum_initialized"] = True
        return loss


class FusedAdam(_FlatGroupOptimizer):
    def __init__(self, params: Iterable, lr: float = 1e-3,
                 betas=(0.9, 0.999), eps: float = 1e-8,
                 weight_decay: float = 0.0, adamw: bool = False):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay, adamw=adamw)
        super().__init__(params, defaults)
        self._flat = [self._flatten_group(g) for g in self.param_groups]
        for rec in self._flat:
            if rec is not None:
                if rec["flat_w"].dtype == torch.bfloat16:
                    rec["master"] = rec["flat_w"].float()
                ref = rec.get("master", rec["flat_w"])
                rec["exp_avg"] = torch.zeros_like(ref)
                rec["exp_avg_sq"] = torch.zeros_like(ref)
                rec["step"] = 0

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group, rec in zip(self.param_groups, self._flat):
            if rec is None:
                continue
            self._repair_grads(rec)
            rec["step"] += 1
            w, g = rec["flat_w"], rec["flat_g"]
            m, v = rec["exp_avg"], rec["exp_avg_sq"]
            master = rec.get("master")
            beta1, beta2 = group["betas"]
            if w.is_cuda and native.available() \
                    and w.dtype == torch.float32:
                native.lib().fused_adam_step(
                    w, g, m, v, rec["step"], group["lr"], beta1, beta2,
                    group["eps"], group["weight_decay"], group["adamw"])
            else:
                # bf16 params: update the fp32 master, write back bf16
                ref = master if master is not None else w
                grad = g.float() if master is not None else g
                if group["adamw"]:
                    ref.mul_(1 - group["lr"] * group["weight_decay"])
                elif group["weight_decay"] != 0:
                    grad = grad.add(ref, alpha=group["weight_decay"])
                m.mul_(beta1).add_(grad, alpha=1 - beta1)
                v.mul_(beta2).addcmul_(grad, grad, value=1 - beta2)
                bc1 = 1 - beta1 ** rec["step"]
                bc2 = 1 - beta2 ** rec["step"]
                denom = (v.sqrt() / (bc2 ** 0.5)).add_(group["eps"])
                ref.addcdiv_(m, denom, value=-group["lr"] / bc1)
                if master is not None:
                    w.copy_(master.to(w.dtype))
        return loss


class FusedAdamW(FusedAdam):
    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=1e-2):
        super().__init__(params, lr, betas, eps, weight_decay, adamw=True)
