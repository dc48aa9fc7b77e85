"""PyTorch-Lightning strategy for bagua_amd.

The reference's Lightning integration lived inside pytorch_lightning
itself (``pytorch_lightning.strategies.BaguaStrategy``, smoke-tested at
/root/reference/tests/pytorch_lightning/test_bagua_strategy.py:30-40).
pytorch_lightning is not installed in this image, so this module ships
the strategy in two layers:

* :class:`BaguaStrategyCore` — the full strategy logic (environment
  setup, algorithm reification, module wrapping, reduce/broadcast/
  barrier, teardown) with the public surface Lightning's Strategy
  protocol expects, usable standalone and fully testable on CPU/gloo.
* :class:`BaguaStrategy` — when ``pytorch_lightning`` IS importable,
  a ``DDPStrategy`` subclass delegating model configuration to the core
  (drop-in: ``Trainer(strategy=BaguaStrategy(algorithm="bytegrad"))``);
  otherwise an alias of the core.
"""

import logging
from typing import Any, List, Optional

import torch

logger = logging.getLogger(__name__)


class BaguaStrategyCore:
    """Strategy logic shared by the standalone and Lightning-bound forms.

    Parameters mirror the upstream Lightning BaguaStrategy:
    ``algorithm`` is a registry name (gradient_allreduce, bytegrad,
    decentralized, low_precision_decentralized, qadam, async),
    ``flatten`` controls bucket flattening, extra kwargs go to the
    algorithm constructor (e.g. ``sync_interval_ms`` for async).
    """

    strategy_name = "bagua"

    def __init__(self, algorithm: str = "gradient_allreduce",
                 flatten: bool = True, **bagua_kwargs: Any):
        self._algorithm_name = algorithm
        self._flatten = flatten
        self._bagua_kwargs = dict(bagua_kwargs)
        self.model = None
        self._wrapped = None

    # -- lifecycle ------------------------------------------------------
    def setup_environment(self):
        """Initialize the bagua process group (idempotent)."""
        import bagua_amd

        if torch.cuda.is_available():
            from . import env

            torch.cuda.set_device(env.get_local_rank())
        bagua_amd.init_process_group()

    def _make_algorithm(self, optimizers: List[torch.optim.Optimizer]):
        from .parallel.algorithms import GlobalAlgorithmRegistry

        factory = GlobalAlgorithmRegistry.get(self._algorithm_name)
        if self._algorithm_name == "qadam":
            from .parallel.algorithms.q_adam import QAdamOptimizer

            qadam = [opt for opt in optimizers
                     if isinstance(opt, QAdamOptimizer)]
            if len(qadam) != 1:
                raise ValueError(
                    "the qadam algorithm requires exactly one "
                    "QAdamOptimizer in configure_optimizers")
            return factory(qadam[0], **self._bagua_kwargs)
        return factory(**self._bagua_kwargs)

    def setup_module(self, module: torch.nn.Module,
                     optimizers: Optional[List] = None):
        """Wrap the (Lightning)Module in bagua DDP and return it."""
        import bagua_amd

        optimizers = list(optimizers or [])
        algorithm = self._make_algorithm(optimizers)
        self.model = module
        self._wrapped = bagua_amd.DistributedDataParallel(
            module, optimizers=optimizers, algorithm=algorithm,
            gradient_as_bucket_view=self._flatten)
        return self._wrapped

    # Lightning calls this name on Strategy subclasses
    configure_ddp = setup_module

    def teardown(self):
        if (self._wrapped is not None
                and self._algorithm_name == "async"):
            algo = self._wrapped.inner.bagua_algorithm
            algo.abort(self._wrapped)
        self._wrapped = None

    # -- collectives (Lightning Strategy protocol) ----------------------
    def barrier(self, name: Optional[str] = None):
        import bagua_amd

        bagua_amd.barrier()

    def reduce(self, tensor, group=None, reduce_op="mean"):
        if not isinstance(tensor, torch.Tensor):
            return tensor
        import bagua_amd

        op = (bagua_amd.ReduceOp.AVG if str(reduce_op) in ("mean", "avg")
              else bagua_amd.ReduceOp.SUM)
        bagua_amd.allreduce_inplace(tensor, op=op)
        return tensor

    def broadcast(self, obj, src: int = 0):
        from .communication import broadcast_object

        return broadcast_object(obj, src=src)

    @property
    def root_device(self):
        if torch.cuda.is_available():
            from . import env

            return torch.device("cuda", env.get_local_rank())
        return torch.device("cpu")

    def model_to_device(self):
        if self.model is not None:
            self.model.to(self.root_device)


def _make_lightning_subclass():
    try:
        from pytorch_lightning.strategies.ddp import DDPStrategy
    except Exception:  # pragma: no cover - lightning not in this image
        return None

    class BaguaStrategy(DDPStrategy):  # pragma: no cover - needs lightning
        """Lightning-bound strategy delegating to BaguaStrategyCore."""

        strategy_name = "bagua"

        def __init__(self, algorithm: str = "gradient_allreduce",
                     flatten: bool = True, **kwargs: Any):
            bagua_kwargs = {
                k: kwargs.pop(k) for k in list(kwargs)
                if k not in ("accelerator", "parallel_devices",
                             "cluster_environment", "checkpoint_io",
                             "precision_plugin")}
            super().__init__(**kwargs)
            self._core = BaguaStrategyCore(algorithm, flatten,
                                           **bagua_kwargs)

        def setup_distributed(self):
            self._core.setup_environment()

        def _setup_model(self, model):
            optimizers = getattr(self, "optimizers", [])
            return self._core.setup_module(model, optimizers)

        def teardown(self):
            self._core.teardown()
            super().teardown()

    return BaguaStrategy


BaguaStrategy = _make_lightning_subclass() or BaguaStrategyCore
