"""bagua_amd — MI355X-native distributed data-parallel training framework.

A from-scratch rebuild of BaguaSys/bagua's capabilities for AMD Instinct
MI355X (gfx950, CDNA4): PyTorch-ROCm Python API, hand-written HIP kernels
for the compression/fusion hot path, RCCL over xGMI for collectives.

Public API mirrors ``bagua.torch_api`` (reference:
bagua/torch_api/__init__.py:25-63) so reference users can switch with an
import change.
"""

__version__ = "0.1.0"

from .communication import (  # noqa: F401
    ReduceOp,
    allgather,
    allgather_inplace,
    allreduce,
    allreduce_inplace,
    alltoall,
    alltoall_inplace,
    alltoall_v,
    alltoall_v_inplace,
    barrier,
    broadcast,
    broadcast_object,
    deinit_process_group,
    from_torch_group,
    gather,
    gather_inplace,
    get_backend,
    init_process_group,
    is_initialized,
    new_group,
    recv,
    reduce,
    reduce_inplace,
    reduce_scatter,
    reduce_scatter_inplace,
    scatter,
    scatter_inplace,
    send,
)
from .env import (  # noqa: F401
    get_local_rank,
    get_local_size,
    get_node_rank,
    get_rank,
    get_world_size,
)
from .tensor import BaguaTensor, ensure_bagua_tensor, to_bagua_tensor  # noqa: F401
from .bucket import BaguaBucket  # noqa: F401

from . import communication  # noqa: F401
from . import env  # noqa: F401
from . import ops  # noqa: F401
from .parallel import algorithms  # noqa: F401
from . import data_parallel  # noqa: F401
from .data_parallel import (  # noqa: F401
    BaguaDistributedDataParallel,
    DistributedDataParallel,
)
from .distributed_module import (  # noqa: F401
    BaguaModule,
    patch_torch_module,
    with_bagua,
)

from . import torch_api  # noqa: E402,F401  (reference-style import alias)


def enable_logging(level=None):
    """Configure logging from LOG_LEVEL env (reference: the Rust core's
    tracing subscriber, bagua-core-py/src/lib.rs:542-547)."""
    import logging
    import os

    lvl = level or os.environ.get("LOG_LEVEL", "WARNING")
    logging.basicConfig(
        level=getattr(logging, str(lvl).upper(), logging.WARNING),
        format="%(asctime)s %(name)s %(levelname)s %(message)s")
