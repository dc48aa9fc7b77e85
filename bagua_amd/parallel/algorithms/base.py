"""Algorithm abstraction + registry.

Mirrors the reference's pluggable-algorithm design
(bagua/torch_api/algorithms/base.py:13-263): an :class:`Algorithm` is a
factory of process-group-bound :class:`AlgorithmImpl` objects whose hook
methods the DDP engine calls at fixed points of the training step.
"""

from typing import Callable, Dict, List

import torch

from ...bucket import BaguaBucket
from ...communication import BaguaProcessGroup
from ...tensor import BaguaTensor


class Algorithm:
    """Factory. Subclasses set ctor args and override :meth:`reify`."""

    def reify(self, process_group: BaguaProcessGroup) -> "AlgorithmImpl":
        return AlgorithmImpl(process_group)


class AlgorithmImpl:
    def __init__(self, process_group: BaguaProcessGroup):
        self.process_group = process_group

    # ------------------------------------------------------------------
    def need_reset(self) -> bool:
        """True when the engine must re-run init (e.g. QAdam warmup ends)
        (reference: base.py:60-71)."""
        return False

    def init_tensors(self, ddp) -> List[BaguaTensor]:
        """Register communication tensors. Default: gradients, in reverse
        parameter order (backward completes roughly in that order)
        (reference: base.py:73-102)."""
        parameters = ddp.bagua_build_params()
        tensors = []
        for name, param in reversed(parameters):
            if param.grad is None:
                param.grad = torch.zeros_like(param)
            t = ddp.ensure_bagua_tensor(
                param, name,
                getter_closure=lambda p: p.grad,
                setter_closure=_set_grad)
            tensors.append(t)
        return tensors

    def tensors_to_buckets(
        self, tensors: List[List[BaguaTensor]], do_flatten: bool
    ) -> List[BaguaBucket]:
        """Turn the engine's grouping suggestion into buckets
        (reference: base.py:104-126)."""
        buckets = []
        for idx, group in enumerate(tensors):
            buckets.append(BaguaBucket(
                group, str(idx), flatten=do_flatten,
                alignment=self.bucket_alignment()))
        return buckets

    def bucket_alignment(self) -> int:
        """Pad bucket numel to a multiple of this. Chunked (scattergather/
        compressed) paths override with a multiple of world size."""
        return 1

    def init_forward_pre_hook(self, ddp) -> Callable:
        def hook(input):
            pass

        return hook

    def init_backward_hook(self, ddp) -> Callable:
        """Per-parameter hook after its grad is accumulated. Default: mark
        comm-ready (reference: base.py:144-163)."""

        def hook(parameter_name, parameter):
            bt = ddp._bagua_tensor_map.get(parameter_name)
            assert bt is not None, (
                "unexpected parameter %s" % parameter_name)
            bt.mark_communication_ready(ddp.bagua_backend)

        return hook

    def init_post_backward_hook(self, ddp) -> Callable:
        """After autograd drains: wait for scheduled comm
        (reference: base.py:165-179)."""

        def hook():
            ddp.bagua_backend.wait_pending_comm_ops()

        return hook

    def init_post_optimizer_step_hook(self, ddp) -> Callable:
        def hook(optimizer: torch.optim.Optimizer):
            pass

        return hook

    def init_operations(self, ddp, bucket: BaguaBucket):
        """Append comm ops onto a freshly-registered bucket."""
        pass


def _set_grad(param, new_grad):
    param.grad = new_grad


# ---------------------------------------------------------------------------
# Registry (reference: base.py:211-263, algorithms/__init__.py:8-33)
# ---------------------------------------------------------------------------


class GlobalAlgorithmRegistry:
    _registry: Dict[str, Callable] = {}
    _descriptions: Dict[str, str] = {}

    @classmethod
    def register(cls, name: str, factory: Callable, description: str = ""):
        cls._registry[name] = factory
        cls._descriptions[name] = description

    @classmethod
    def get(cls, name: str) -> Callable:
        if name not in cls._registry:
            raise KeyError(
                "unknown algorithm %r; known: %s"
                % (name, sorted(cls._registry)))
        return cls._registry[name]

    @classmethod
    def names(cls):
        return sorted(cls._registry)
