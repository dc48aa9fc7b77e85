"""Local expert container (reference: model_parallel/moe/experts.py:10-40).

Each parameter is tagged ``expert=True`` so the DP engine excludes it
from gradient synchronization (bagua_amd/parallel/engine.py
bagua_build_params)."""

import copy

import torch


class Experts(torch.nn.Module):
    def __init__(self, expert: torch.nn.Module, num_local_experts: int = 1):
        super().__init__()
        self.bagua_experts = torch.nn.ModuleList(
            [copy.deepcopy(expert) for _ in range(num_local_experts)])
        self.num_local_experts = num_local_experts
        for expert_module in self.bagua_experts:
            for _, param in expert_module.named_parameters():
                param.expert = True
                param.allreduce = False

    def forward(self, inputs: torch.Tensor) -> torch.Tensor:
        # inputs: (ep_size, num_local_experts, capacity, d_model)
        chunks = inputs.chunk(self.num_local_experts, dim=1)
        outputs = []
        for chunk, expert in zip(chunks, self.bagua_experts):
            out = expert(chunk)
            if isinstance(out, tuple):
                out = out[0]
            outputs.append(out)
        return torch.cat(outputs, dim=1)
