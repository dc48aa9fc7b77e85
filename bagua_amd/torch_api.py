"""Compatibility veneer: ``import bagua_amd.torch_api as bagua`` mirrors
the reference's ``import bagua.torch_api as bagua`` import style
(reference: bagua/torch_api/__init__.py:25-63)."""

from . import (  # noqa: F401
    communication,
    data_parallel,
    env,
)
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
    gather,
    gather_inplace,
    get_backend,
    init_process_group,
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
from .distributed_module import BaguaModule, with_bagua  # noqa: F401
from .tensor import BaguaTensor  # noqa: F401
from .bucket import BaguaBucket  # noqa: F401
from .env import (  # noqa: F401
    get_local_rank,
    get_local_size,
    get_rank,
    get_world_size,
)
from . import contrib  # noqa: F401
from . import checkpoint  # noqa: F401
from .parallel import algorithms  # noqa: F401
from .parallel import moe as model_parallel_moe  # noqa: F401
from .parallel import moe  # noqa: F401  (reference exports `moe` directly)
from . import data_parallel  # noqa: F401
