"""GShard-style top-k gated MoE layer
(reference: bagua/torch_api/model_parallel/moe/sharded_moe.py:77-375,
itself DeepSpeed-derived; re-implemented here from the GShard math).

Token flow per layer: gate -> dispatch einsum -> alltoall over the
expert-parallel group -> local experts -> alltoall back -> combine
einsum. On one 8xMI355X node the two alltoalls are single-hop xGMI
exchanges (fully-connected point-to-point topology).
"""

import math
from typing import Optional, Tuple

import torch
import torch.distributed as dist
import torch.nn.functional as F
from torch import Tensor
from torch.nn import Module

uniform_map = {}
gumbel_map = {}
exp_selection_uniform_map = {}


def multiplicative_jitter(x, device: torch.device, epsilon=1e-2):
    """Multiply by uniform noise in [1-eps, 1+eps] (training only)."""
    if epsilon == 0:
        return x
    uniform = uniform_map.get(device)
    if uniform is None:
        uniform = torch.distributions.uniform.Uniform(
            low=torch.tensor(1.0 - epsilon, device=device),
            high=torch.tensor(1.0 + epsilon, device=device),
        ).rsample
        uniform_map[device] = uniform
    return x * uniform(x.shape)


def gumbel_rsample(shape, device: torch.device) -> Tensor:
    gumbel = gumbel_map.get(device)
    if gumbel is None:
        one = torch.tensor(1.0, device=device)
        zero = torch.tensor(0.0, device=device)
        gumbel = torch.distributions.gumbel.Gumbel(zero, one).rsample
        gumbel_map[device] = gumbel
    return gumbel(shape)


class _AllToAll(torch.autograd.Function):
    """Autograd alltoall over the EP group
    (reference: sharded_moe.py:77-90)."""

    @staticmethod
    def forward(ctx, group, input: Tensor) -> Tensor:
        ctx.group = group
        input = input.contiguous()
        output = torch.empty_like(input)
        if group is None or not dist.is_initialized() \
                or dist.get_world_size(group) == 1:
            output.copy_(input)
        else:
            dist.all_to_all_single(output, input, group=group)
        return output

    @staticmethod
    def backward(ctx, *grad_output):
        return (None, _AllToAll.apply(ctx.group, *grad_output))


def _capacity(num_tokens: int, num_experts: int, capacity_factor: float,
              min_capacity: int) -> int:
    capacity = math.ceil(num_tokens / num_experts * capacity_factor)
    return max(capacity, min_capacity)


def top1gating(
    logits: Tensor,
    capacity_factor: float,
    min_capacity: int,
    used_token: Optional[Tensor] = None,
    noisy_gate_policy: Optional[str] = None,
    drop_tokens: bool = True,
) -> Tuple[Tensor, Tensor, Tensor, Tensor]:
    """Top-1 gating. Returns (l_aux, combine_weights, dispatch_mask,
    metadata)."""
    if noisy_gate_policy == "RSample":
        logits_w_noise = logits + gumbel_rsample(logits.shape,
                                                 device=logits.device)
    gates = F.softmax(logits, dim=1)
    num_tokens, num_experts = gates.shape
    capacity = _capacity(num_tokens, num_experts, capacity_factor,
                         min_capacity)

    indices1_s = torch.argmax(
        logits_w_noise if noisy_gate_policy == "RSample" else gates, dim=1)
    mask1 = F.one_hot(indices1_s, num_classes=num_experts)
    if used_token is not None:
        mask1 = used_token.reshape(-1, 1) * mask1

    # auxiliary load-balancing loss (GShard eq. 4)
    me = torch.mean(gates, dim=0)
    ce = torch.mean(mask1.float(), dim=0)
    l_aux = torch.sum(me * ce) * num_experts

    # position of each token inside its expert queue
    locations1 = torch.cumsum(mask1, dim=0) - 1
    if drop_tokens:
        mask1 = mask1 * torch.lt(locations1, capacity)
    locations1_s = torch.sum(locations1 * mask1, dim=1)

    mask1_float = mask1.float()
    gates1_s = (gates * mask1_float).sum(dim=1)

    locations1_sc = F.one_hot(locations1_s,
                              num_classes=capacity).float() * \
        mask1_float.sum(dim=1, keepdim=True)
    combine_weights = torch.einsum(
        "s,se,sc->sec", gates1_s, mask1_float, locations1_sc)
    dispatch_mask = combine_weights.bool()
    exp_counts = torch.sum(mask1, dim=0)
    return l_aux, combine_weights, dispatch_mask, exp_counts


def top2gating(
    logits: Tensor, capacity_factor: float, min_capacity: int,
) -> Tuple[Tensor, Tensor, Tensor, Tensor]:
    """Top-2 gating with normalized weights."""
    gates = F.softmax(logits, dim=1)
    num_tokens, num_experts = gates.shape
    capacity = _capacity(num_tokens, num_experts, 2 * capacity_factor,
                         min_capacity)

    indices1_s = torch.argmax(gates, dim=1)
    mask1 = F.one_hot(indices1_s, num_classes=num_experts)

    logits_w_noise = logits + gumbel_rsample(logits.shape,
                                             device=logits.device)
    logits_except1 = logits_w_noise.masked_fill(mask1.bool(),
                                                float("-inf"))
    indices2_s = torch.argmax(logits_except1, dim=1)
    mask2 = F.one_hot(indices2_s, num_classes=num_experts)

    locations1 = torch.cumsum(mask1, dim=0) - 1
    locations2 = torch.cumsum(mask2, dim=0) - 1
    locations2 += torch.sum(mask1, dim=0, keepdim=True)

    me = torch.mean(gates, dim=0)
    ce = torch.mean(mask1.float(), dim=0)
    l_aux = torch.mean(me * ce) * num_experts * num_experts

    mask1 = mask1 * torch.lt(locations1, capacity)
    mask2 = mask2 * torch.lt(locations2, capacity)
    locations1_s = torch.sum(locations1 * mask1, dim=1)
    locations2_s = torch.sum(locations2 * mask2, dim=1)

    mask1_float = mask1.float()
    mask2_float = mask2.float()
    gates1_s = torch.einsum("se,se->s", gates, mask1_float)
    gates2_s = torch.einsum("se,se->s", gates, mask2_float)
    denom_s = torch.clamp(gates1_s + gates2_s,
                          min=torch.finfo(gates.dtype).eps)
    gates1_s = gates1_s / denom_s
    gates2_s = gates2_s / denom_s

    gates1 = torch.einsum("s,se->se", gates1_s, mask1_float)
    gates2 = torch.einsum("s,se->se", gates2_s, mask2_float)
    locations1_sc = F.one_hot(locations1_s, num_classes=capacity).float()
    locations2_sc = F.one_hot(locations2_s, num_classes=capacity).float()
    combine1_sec = torch.einsum("se,sc->sec", gates1, locations1_sc)
    combine2_sec = torch.einsum("se,sc->sec", gates2, locations2_sc)
    combine_weights = combine1_sec + combine2_sec
    dispatch_mask = combine_weights.bool()
    exp_counts = torch.sum(mask1 + mask2, dim=0)
    return l_aux, combine_weights, dispatch_mask, exp_counts


class TopKGate(Module):
    """Learned router (reference: sharded_moe.py:93-303)."""

    def __init__(self, model_dim: int, num_experts: int, k: int = 1,
                 capacity_factor: float = 1.0,
                 eval_capacity_factor: float = 1.0, min_capacity: int = 4,
                 noisy_gate_policy: Optional[str] = None,
                 drop_tokens: bool = True):
        super().__init__()
        if k not in (1, 2):
            raise ValueError("Only top-1 and top-2 gatings are supported")
        self.wg = torch.nn.Linear(model_dim, num_experts, bias=False)
        self.k = k
        self.capacity_factor = capacity_factor
        self.eval_capacity_factor = eval_capacity_factor
        self.min_capacity = min_capacity
        self.noisy_gate_policy = noisy_gate_policy
        self.drop_tokens = drop_tokens

    def forward(self, input: Tensor, used_token: Optional[Tensor] = None):
        input_fp32 = input.float()
        if self.noisy_gate_policy == "Jitter" and self.training:
            input_fp32 = multiplicative_jitter(input_fp32,
                                               device=input.device)
        logits = F.linear(input_fp32, self.wg.weight.float())
        cap = (self.capacity_factor if self.training
               else self.eval_capacity_factor)
        if self.k == 1:
            return top1gating(
                logits, cap, self.min_capacity, used_token,
                self.noisy_gate_policy if self.training else None,
                self.drop_tokens)
        return top2gating(logits, cap, self.min_capacity)


class MOELayer(Module):
    """Mixture-of-experts layer (reference: sharded_moe.py:306-375)."""

    def __init__(self, gate: TopKGate, experts: Module, ep_group,
                 ep_size: int, num_local_experts: int):
        super().__init__()
        self.gate = gate
        self.experts = experts
        self.ep_group = ep_group
        self.ep_size = ep_size
        self.num_local_experts = num_local_experts
        self.num_experts = ep_size * num_local_experts
        self.l_aux = torch.tensor(0.0)
        self.exp_counts = None

    def forward(self, *input: Tensor, **kwargs) -> Tensor:
        d_model = input[0].shape[-1]
        reshaped_input = input[0].reshape(-1, d_model)

        self.l_aux, combine_weights, dispatch_mask, self.exp_counts = \
            self.gate(reshaped_input, kwargs.get("used_token"))

        dispatched = torch.einsum(
            "sec,sm->ecm", dispatch_mask.to(reshaped_input.dtype),
            reshaped_input)

        dispatched = _AllToAll.apply(self.ep_group, dispatched)
        dispatched = dispatched.reshape(
            self.ep_size, self.num_local_experts, -1, d_model)

        expert_output = self.experts(dispatched)
        expert_output = _AllToAll.apply(self.ep_group, expert_output)
        expert_output = expert_output.reshape(
            self.num_experts, -1, d_model)

        combined = torch.einsum(
            "sec,ecm->sm", combine_weights.to(expert_output.dtype),
            expert_output)
        return combined.reshape(input[0].shape)
