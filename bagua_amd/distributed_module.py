"""Legacy ``with_bagua`` API (reference: bagua/torch_api/distributed.py:21-147).

The reference gorilla-patched ``torch.nn.Module``. Here the primary entry
is the explicit :func:`with_bagua` function; :class:`BaguaModule` is a
mixin for users who want the method form. ``bagua_amd.patch_torch_module``
installs the method on ``torch.nn.Module`` for drop-in compatibility.
"""

from typing import List, Optional

import torch

from .communication import BaguaProcessGroup, _get_default_group
from .parallel.algorithms.base import Algorithm
from .parallel.engine import BaguaDistributedDataParallel


def with_bagua(
    module: torch.nn.Module,
    optimizers: List[torch.optim.Optimizer],
    algorithm: Algorithm,
    process_group: Optional[BaguaProcessGroup] = None,
    do_flatten: bool = True,
) -> torch.nn.Module:
    """Wrap ``module`` for bagua distributed training and return it.

    After this call the module exposes ``bagua_ddp``, ``bagua_algorithm``,
    ``bagua_optimizers`` and ``bagua_buckets`` attributes.
    """
    ddp = BaguaDistributedDataParallel(
        module,
        optimizers=optimizers,
        algorithm=algorithm,
        process_group=process_group or _get_default_group(),
        gradient_as_bucket_view=do_flatten,
    )
    module.bagua_ddp = ddp
    return module


class BaguaModule:
    """Mixin: ``class MyNet(nn.Module, BaguaModule)`` then
    ``model.with_bagua([opt], algorithm)``."""

    def with_bagua(self, optimizers, algorithm, process_group=None,
                   do_flatten=True):
        return with_bagua(self, optimizers, algorithm, process_group,
                          do_flatten)

    @property
    def bagua_algorithm(self):
        return self.bagua_ddp.bagua_algorithm

    @property
    def bagua_optimizers(self):
        return self.bagua_ddp.bagua_optimizers

    @property
    def bagua_buckets(self):
        return self.bagua_ddp.bagua_buckets


def patch_torch_module():
    """Install ``with_bagua`` on torch.nn.Module (opt-in drop-in compat)."""
    if not hasattr(torch.nn.Module, "with_bagua"):
        torch.nn.Module.with_bagua = BaguaModule.with_bagua
        torch.nn.Module.bagua_algorithm = BaguaModule.bagua_algorithm
        torch.nn.Module.bagua_optimizers = BaguaModule.bagua_optimizers
        torch.nn.Module.bagua_buckets = BaguaModule.bagua_buckets
