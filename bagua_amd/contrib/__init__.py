from .fused_optimizer import fuse_optimizer, fuse_step, is_fused_optimizer  # noqa: F401
