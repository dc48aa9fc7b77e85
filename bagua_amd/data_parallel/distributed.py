"""torch-DDP-compatible wrapper
(reference: bagua/torch_api/data_parallel/distributed.py:63-360).

``DistributedDataParallel(module, optimizers=[...], algorithm=...)`` is a
drop-in for ``torch.nn.parallel.DistributedDataParallel`` (v1.9 surface).
Arguments bagua cannot honor fall back to real torch DDP with a warning.
"""

import logging
from contextlib import contextmanager
from typing import List, Optional

import torch
from torch.nn.modules import Module

from ..communication import BaguaProcessGroup, _get_default_group, from_torch_group
from ..parallel.algorithms.base import Algorithm
from ..parallel.algorithms.gradient_allreduce import GradientAllReduceAlgorithm
from ..parallel.engine import BaguaDistributedDataParallel

logger = logging.getLogger(__name__)


def to_bagua_process_group(process_group=None) -> BaguaProcessGroup:
    """Convert a torch/bagua/None process group to a BaguaProcessGroup
    (reference: data_parallel/distributed.py:63-90)."""
    if process_group is None:
        return _get_default_group()
    if isinstance(process_group, BaguaProcessGroup):
        return process_group
    if isinstance(process_group, torch.distributed.ProcessGroup):
        return from_torch_group(process_group)
    raise TypeError("unsupported process group %r" % type(process_group))


class DistributedDataParallel_V1_9_0(Module):
    def __init__(
        self,
        module,
        device_ids=None,
        output_device=None,
        dim=0,
        broadcast_buffers: bool = True,
        process_group=None,
        bucket_cap_mb: int = 25,
        find_unused_parameters: bool = False,
        check_reduction: bool = False,
        gradient_as_bucket_view: bool = True,
        # bagua extensions
        optimizers: List[torch.optim.Optimizer] = [],
        algorithm: Optional[Algorithm] = None,
    ):
        super().__init__()
        assert any(p.requires_grad for p in module.parameters()), (
            "DistributedDataParallel is not needed when a module "
            "doesn't have any parameter that requires a gradient.")
        if device_ids is not None and len(device_ids) > 1:
            raise ValueError("device_ids can only be None or contain a "
                             "single element.")
        self.module = module
        self.device = next(module.parameters()).device
        self.broadcast_buffers = broadcast_buffers
        if algorithm is None:
            algorithm = GradientAllReduceAlgorithm()
        self.inner = BaguaDistributedDataParallel(
            self.module,
            optimizers=optimizers,
            algorithm=algorithm,
            process_group=to_bagua_process_group(process_group),
            gradient_as_bucket_view=gradient_as_bucket_view,
            find_unused_parameters=find_unused_parameters,
        )

    # -- parity surface -------------------------------------------------
    @property
    def require_backward_grad_sync(self):
        return self.inner.require_backward_grad_sync

    @property
    def parameters_to_ignore(self):
        return self.inner.parameters_to_ignore

    @property
    def bagua_algorithm(self):
        return self.inner.bagua_algorithm

    @property
    def bagua_module_name(self):
        return self.inner.bagua_module_name

    @property
    def bagua_optimizers(self):
        return self.inner.bagua_optimizers

    @property
    def bagua_buckets(self):
        return self.inner.bagua_buckets

    def _sync_buffers(self):
        # torch-DDP semantics: with broadcast_buffers=True, module buffers
        # (BN running stats etc.) are re-broadcast from rank 0 before each
        # authoritative forward so ranks can never drift
        # (test_c10d_common checks this behavior class).
        from ..communication import broadcast_coalesced

        bufs = [b.data for b in self.module.buffers() if b.numel() > 0]
        if bufs:
            comm = self.inner.process_group.get_global_communicator()
            broadcast_coalesced(bufs, src=0, comm=comm)

    def forward(self, *inputs, **kwargs):
        if (self.broadcast_buffers and self.module.training
                and self.inner.require_backward_grad_sync
                and torch.distributed.is_initialized()
                and torch.distributed.get_world_size() > 1):
            self._sync_buffers()
        return self.module(*inputs, **kwargs)

    def register_comm_hook(self, state, hook):
        """torch DDP comm hooks rewire c10d bucket allreduce; bagua's
        algorithm abstraction IS that layer — use a custom Algorithm
        instead (parity with the reference, which also rejected hooks)."""
        raise NotImplementedError(
            "bagua DistributedDataParallel does not support "
            "register_comm_hook; implement a bagua Algorithm instead")

    def _register_builtin_comm_hook(self, comm_hook_type):
        raise NotImplementedError(
            "bagua DistributedDataParallel does not support "
            "register_comm_hook; implement a bagua Algorithm instead")

    @contextmanager
    def no_sync(self):
        """Skip gradient sync inside the context
        (reference: data_parallel/distributed.py:174-195)."""
        old = self.inner.require_backward_grad_sync
        self.inner.require_backward_grad_sync = False
        try:
            yield
        finally:
            self.inner.require_backward_grad_sync = old


def DistributedDataParallel(
    module,
    device_ids=None,
    output_device=None,
    dim=0,
    broadcast_buffers: bool = True,
    process_group=None,
    bucket_cap_mb: int = 25,
    find_unused_parameters: bool = False,
    check_reduction: bool = False,
    gradient_as_bucket_view: bool = True,
    optimizers: List[torch.optim.Optimizer] = [],
    algorithm: Optional[Algorithm] = None,
):
    """Factory; falls back to torch DDP when asked for semantics bagua
    doesn't implement (reference: data_parallel/distributed.py:319-345)."""
    if check_reduction or (device_ids is not None and len(device_ids) > 1):
        logger.warning(
            "unsupported DDP arguments for bagua; falling back to "
            "torch.nn.parallel.DistributedDataParallel")
        return torch.nn.parallel.DistributedDataParallel(
            module=module, device_ids=device_ids,
            output_device=output_device, dim=dim,
            broadcast_buffers=broadcast_buffers,
            process_group=None if process_group is None else process_group,
            bucket_cap_mb=bucket_cap_mb,
            find_unused_parameters=find_unused_parameters,
            gradient_as_bucket_view=gradient_as_bucket_view)
    return DistributedDataParallel_V1_9_0(
        module=module, device_ids=device_ids, output_device=output_device,
        dim=dim, broadcast_buffers=broadcast_buffers,
        process_group=process_group, bucket_cap_mb=bucket_cap_mb,
        find_unused_parameters=find_unused_parameters,
        check_reduction=check_reduction,
        gradient_as_bucket_view=gradient_as_bucket_view,
        optimizers=optimizers, algorithm=algorithm)
