"""Algorithm registry (reference: bagua/torch_api/algorithms/__init__.py:8-33)."""

from .base import Algorithm, AlgorithmImpl, GlobalAlgorithmRegistry  # noqa: F401
from . import gradient_allreduce  # noqa: F401
from . import bytegrad  # noqa: F401
from . import decentralized  # noqa: F401
from . import q_adam  # noqa: F401
from . import async_model_average  # noqa: F401

from .gradient_allreduce import GradientAllReduceAlgorithm
from .bytegrad import ByteGradAlgorithm
from .decentralized import (
    DecentralizedAlgorithm,
    LowPrecisionDecentralizedAlgorithm,
)
from .q_adam import QAdamAlgorithm
from .async_model_average import AsyncModelAverageAlgorithm

GlobalAlgorithmRegistry.register(
    "gradient_allreduce", GradientAllReduceAlgorithm,
    description="full-precision synchronous gradient allreduce")
GlobalAlgorithmRegistry.register(
    "bytegrad", ByteGradAlgorithm,
    description="MinMaxUInt8-compressed centralized gradient sync")
GlobalAlgorithmRegistry.register(
    "decentralized", DecentralizedAlgorithm,
    description="decentralized model averaging SGD")
GlobalAlgorithmRegistry.register(
    "low_precision_decentralized", LowPrecisionDecentralizedAlgorithm,
    description="difference-compressed decentralized SGD")
GlobalAlgorithmRegistry.register(
    "qadam", QAdamAlgorithm,
    description="quantized-momentum Adam")
GlobalAlgorithmRegistry.register(
    "async", AsyncModelAverageAlgorithm,
    description="asynchronous model averaging")
