"""MoE helpers (reference: bagua/torch_api/model_parallel/moe/utils.py:4-7)."""


def is_moe_param(param) -> bool:
    return hasattr(param, "expert") and param.expert
