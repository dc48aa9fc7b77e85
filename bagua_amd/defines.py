"""Shared type and hyperparameter definitions.

Re-design of the reference's bagua/bagua_define.py:12-58 for the MI355X
build: dtypes grow bf16 (the native MI355X training dtype), and the
autotuned hyperparameter space is sized for xGMI bucket fusion.
"""

import enum
from typing import Dict, List, Optional

from pydantic import BaseModel


class TensorDtype(str, enum.Enum):
    F32 = "f32"
    F16 = "f16"
    BF16 = "bf16"
    U8 = "u8"
    I64 = "i64"


class TensorDeclaration(BaseModel):
    name: str
    num_elements: int
    dtype: TensorDtype


def dtype_bytes(dtype: TensorDtype) -> int:
    return {
        TensorDtype.F32: 4,
        TensorDtype.F16: 2,
        TensorDtype.BF16: 2,
        TensorDtype.U8: 1,
        TensorDtype.I64: 8,
    }[dtype]


class BaguaHyperparameter(BaseModel):
    """Runtime-tunable hyperparameters exchanged with the autotune service
    (reference: bagua_define.py:34-50)."""

    buckets: List[List[TensorDeclaration]] = []
    bucket_size: int = 32 * 1024 * 1024
    # None = the tuner is not searching this dimension (single node);
    # the engine then leaves the algorithm's own flag alone
    is_hierarchical_reduce: Optional[bool] = None

    def update(self, param_dict: Dict) -> "BaguaHyperparameter":
        tmp = self.dict()
        for key, value in param_dict.items():
            if key in tmp:
                self.__dict__[key] = type(tmp[key])(value)
        return self


class BaguaCoreTelemetrySpan(BaseModel):
    trace_id: int
    action: str
    tensor_name: str
    start_time: int
    end_time: int
