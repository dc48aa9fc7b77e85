from . import algorithms  # noqa: F401
from .engine import BaguaDistributedDataParallel  # noqa: F401
