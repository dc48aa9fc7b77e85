from .distributed import DistributedDataParallel  # noqa: F401
from .distributed import DistributedDataParallel_V1_9_0  # noqa: F401
from .distributed import to_bagua_process_group  # noqa: F401
from ..parallel.engine import BaguaDistributedDataParallel  # noqa: F401
