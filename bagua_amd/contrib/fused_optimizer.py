"""Fused optimizer (placeholder — full implementation lands with contrib).

reference: bagua/torch_api/contrib/fuse/optimizer.py
"""


def is_fused_optimizer(optimizer) -> bool:
    return hasattr(optimizer, "_bagua_fused_count")


def fuse_optimizer(optimizer, do_flatten: bool = True, check_flatten: bool = True):
    raise NotImplementedError("fused optimizer lands in a later commit")


def fuse_step(optimizer, closure=None):
    raise NotImplementedError("fused optimizer lands in a later commit")
