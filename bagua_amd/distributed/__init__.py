from . import run  # noqa: F401
from . import launch  # noqa: F401
from . import baguarun  # noqa: F401
from . import sys_perf  # noqa: F401
