from .utils import is_moe_param  # noqa: F401

try:  # full MoE layer (needs torch.distributed initialized at use time)
    from .layer import MoE  # noqa: F401
    from .sharded_moe import MOELayer, TopKGate  # noqa: F401
    from .experts import Experts  # noqa: F401
except ImportError:  # pragma: no cover - partial builds
    pass
