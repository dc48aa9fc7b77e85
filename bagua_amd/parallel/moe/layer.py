"""User-facing MoE wrapper (reference: model_parallel/moe/layer.py:22-110).

``MoE(hidden_size, expert, num_local_experts=k)`` builds
``num_local_experts * world_size`` experts sharded one group per rank
(expert parallelism over the default process group)."""

from typing import Optional

import torch
import torch.distributed as dist

from .experts import Experts
from .sharded_moe import MOELayer, TopKGate


class MoE(torch.nn.Module):
    def __init__(
        self,
        hidden_size: int,
        expert: torch.nn.Module,
        num_local_experts: int = 1,
        k: int = 1,
        output_dropout_prob: float = 0.0,
        capacity_factor: float = 1.0,
        eval_capacity_factor: float = 1.0,
        min_capacity: int = 4,
        noisy_gate_policy: Optional[str] = None,
        drop_tokens: bool = True,
        expert_parallel_group=None,
    ):
        super().__init__()
        assert noisy_gate_policy is None or noisy_gate_policy in (
            "None", "Jitter", "RSample"), (
            "Unsupported noisy_gate_policy: %s" % noisy_gate_policy)

        if dist.is_available() and dist.is_initialized():
            self.ep_size = (dist.get_world_size(expert_parallel_group)
                            if expert_parallel_group is not None
                            else dist.get_world_size())
        else:
            self.ep_size = 1
        self.num_local_experts = num_local_experts
        self.num_experts = self.ep_size * num_local_experts

        experts = Experts(expert, num_local_experts)
        gate = TopKGate(hidden_size, self.num_experts, k, capacity_factor,
                        eval_capacity_factor, min_capacity,
                        noisy_gate_policy, drop_tokens)
        self.bagua_moe = MOELayer(gate, experts, expert_parallel_group,
                                  self.ep_size, num_local_experts)
        self.dropout = torch.nn.Dropout(output_dropout_prob)

    def forward(self, hidden_states, used_token=None):
        """Returns (output, l_aux, exp_counts)."""
        output = self.bagua_moe(hidden_states, used_token=used_token)
        output = self.dropout(output)
        return output, self.bagua_moe.l_aux, self.bagua_moe.exp_counts
