"""Contrib utilities (reference: bagua/torch_api/contrib/)."""

from .fused_optimizer import fuse_optimizer, fuse_step, is_fused_optimizer  # noqa: F401
from .load_balancing_data_loader import (  # noqa: F401
    LoadBalancingDistributedBatchSampler,
    LoadBalancingDistributedSampler,
)
from .cache_loader import CacheLoader  # noqa: F401
from .cached_dataset import CachedDataset  # noqa: F401
from .sync_batchnorm import SyncBatchNorm  # noqa: F401
from . import utils  # noqa: F401
from .fused_kernels import FusedAdam, FusedAdamW, FusedSGD  # noqa: F401
