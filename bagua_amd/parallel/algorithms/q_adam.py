"""QAdam — quantized-momentum Adam
(reference: bagua/torch_api/algorithms/q_adam.py:13-267).

Warmup phase: plain gradient allreduce + full Adam moment updates.
After warmup: the 2nd moment freezes, the *1st moment* (exp_avg) becomes
the communication tensor — updated by a python op on the scheduler path,
then synchronized with ByteGrad-style MinMaxUInt8 compression.
"""

import math
from typing import List, Tuple

import torch
from torch.optim.optimizer import Optimizer

from ...bucket import BaguaBucket
from ...communication import BaguaProcessGroup
from ...tensor import BaguaTensor
from .base import Algorithm, AlgorithmImpl


class QAdamOptimizer(Optimizer):
    def __init__(self, params, lr: float = 1e-3, warmup_steps: int = 100,
                 betas: Tuple[float, float] = (0.9, 0.999), eps: float = 1e-8,
                 weight_decay: float = 0.0):
        if not 0.0 <= lr:
            raise ValueError("Invalid learning rate: {}".format(lr))
        if not 0.0 <= eps:
            raise ValueError("Invalid epsilon value: {}".format(eps))
        if not 0.0 <= betas[0] < 1.0:
            raise ValueError("Invalid beta parameter 0: {}".format(betas[0]))
        if not 0.0 <= betas[1] < 1.0:
            raise ValueError("Invalid beta parameter 1: {}".format(betas[1]))
        if warmup_steps <= 0:
            raise ValueError("warmup_steps must be > 0")
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay)
        super().__init__(params, defaults)
        self.warmup_steps = warmup_steps

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        for group in self.param_groups:
            lr = group["lr"]
            weight_decay = group["weight_decay"]
            beta1, beta2 = group["betas"]
            eps = group["eps"]

            for param in group["params"]:
                state = self.state[param]
                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(param)
                    state["exp_avg_sq"] = torch.zeros_like(param)

                state["step"] += 1
                step_id = state["step"]
                grad = param.grad
                if weight_decay != 0:
                    grad = grad.add(param, alpha=weight_decay)

                if step_id < self.warmup_steps:
                    # full Adam moment updates only during warmup; after
                    # warmup exp_avg is maintained by the comm python-op
                    # and exp_avg_sq is frozen
                    state["exp_avg"].mul_(beta1).add_(grad, alpha=1 - beta1)
                    state["exp_avg_sq"].mul_(beta2).addcmul_(
                        grad, grad, value=1 - beta2)

                bias_correction1 = 1 - beta1 ** step_id
                bias_correction2 = 1 - beta2 ** step_id
                denom = (state["exp_avg_sq"].sqrt()
                         / math.sqrt(bias_correction2)).add_(eps)
                step_size = lr / bias_correction1
                param.data.addcdiv_(state["exp_avg"], denom,
                                    value=-step_size)
        return loss


class QAdamAlgorithmImpl(AlgorithmImpl):
    def __init__(self, process_group: BaguaProcessGroup,
                 q_adam_optimizer: QAdamOptimizer,
                 hierarchical: bool = True):
        super().__init__(process_group)
        self.hierarchical = hierarchical
        self.optimizer = q_adam_optimizer
        self.warmup_steps = q_adam_optimizer.warmup_steps

    @property
    def optimizer_step_id(self):
        param = self.optimizer.param_groups[0]["params"][0]
        return self.optimizer.state[param].get("step", 0)

    def need_reset(self):
        return self.optimizer_step_id == self.warmup_steps

    def bucket_alignment(self) -> int:
        n = self.process_group.get_global_communicator().nranks()
        return n * 32

    def init_tensors(self, ddp) -> List[BaguaTensor]:
        parameters = ddp.bagua_build_params()
        name_of, idx_of = {}, {}
        for idx, (name, param) in enumerate(reversed(parameters)):
            name_of[id(param)] = name
            idx_of[id(param)] = idx

        tensors = []
        warmup = self.optimizer_step_id < self.warmup_steps
        for group in self.optimizer.param_groups:
            for param in group["params"]:
                if param.grad is None:
                    param.grad = torch.zeros_like(param)
                if warmup:
                    t = ddp.ensure_bagua_tensor(
                        param, name_of[id(param)],
                        getter_closure=lambda p: p.grad,
                        setter_closure=lambda p, t: setattr(p, "grad", t))
                else:
                    if "exp_avg" not in self.optimizer.state[param]:
                        self.optimizer.state[param]["exp_avg"] = (
                            torch.zeros_like(param))

                    def set_momentum_fn(p, t):
                        self.optimizer.state[p]["exp_avg"] = t

                    t = ddp.ensure_bagua_tensor(
                        param, name_of[id(param)],
                        getter_closure=(
                            lambda p: self.optimizer.state[p]["exp_avg"]),
                        setter_closure=set_momentum_fn)
                t._q_adam_idx = idx_of[id(param)]
                tensors.append(t)
        tensors.sort(key=lambda t: t._q_adam_idx)
        return tensors

    def init_operations(self, ddp, bucket: BaguaBucket):
        bucket.clear_ops()
        if self.optimizer_step_id < self.warmup_steps:
            bucket.append_centralized_synchronous_op(
                hierarchical=False, average=True, group=self.process_group)
        else:

            def calculate_momentum(*args):
                beta1, _ = self.optimizer.param_groups[0]["betas"]
                for t in bucket.tensors:
                    t.tensor().mul_(beta1).add_(
                        t.proxy.grad, alpha=1 - beta1)

            bucket.append_python_op(calculate_momentum,
                                    group=self.process_group)
            bucket.append_centralized_synchronous_op(
                hierarchical=self.hierarchical, average=True,
                scattergather=True, compression="MinMaxUInt8",
                group=self.process_group)

    def init_backward_hook(self, ddp):
        warmup = self.optimizer_step_id < self.warmup_steps

        def hook(parameter_name, parameter):
            bt = ddp._bagua_tensor_map[parameter_name]
            if warmup:
                assert bt.data_ptr() == parameter.grad.data_ptr(), (
                    "registered tensor must be the grad during QAdam warmup")
            else:
                assert bt.data_ptr() == (
                    self.optimizer.state[parameter]["exp_avg"].data_ptr()), (
                    "registered tensor must be exp_avg after QAdam warmup")
            bt.mark_communication_ready(ddp.bagua_backend)

        return hook


class QAdamAlgorithm(Algorithm):
    def __init__(self, q_adam_optimizer: QAdamOptimizer,
                 hierarchical: bool = True):
        self.optimizer = q_adam_optimizer
        self.hierarchical = hierarchical

    def reify(self, process_group: BaguaProcessGroup):
        return QAdamAlgorithmImpl(process_group,
                                  q_adam_optimizer=self.optimizer,
                                  hierarchical=self.hierarchical)
