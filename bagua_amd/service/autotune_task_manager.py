"""Autotune bookkeeping per model
(reference: bagua/service/autotune_task_manager.py:40-185).

Holds the (iteration, hyperparameters, score) record list, re-sorts
tensors by the telemetry-derived execution order, and greedily packs them
into buckets of the trial bucket size.
"""

import csv
import logging
import os
import time
from typing import Dict, List, Optional, Tuple

from ..defines import (
    BaguaHyperparameter,
    TensorDeclaration,
    TensorDtype,
    dtype_bytes,
)
from .bayesian_optimizer import BayesianOptimizer, BoolParam, IntParam

logger = logging.getLogger(__name__)


def split_bucket_by_bucket_size(
    tensor_list: List[TensorDeclaration], bucket_size: int,
) -> List[List[TensorDeclaration]]:
    """Group by dtype, then greedy-pack preserving order
    (reference: autotune_task_manager.py:85-119)."""
    buckets: List[List[TensorDeclaration]] = []
    cur: List[TensorDeclaration] = []
    cur_bytes = 0
    cur_dtype: Optional[TensorDtype] = None
    for td in tensor_list:
        nb = td.num_elements * dtype_bytes(td.dtype)
        if cur and (cur_dtype != td.dtype or cur_bytes + nb > bucket_size):
            buckets.append(cur)
            cur, cur_bytes = [], 0
        cur.append(td)
        cur_bytes += nb
        cur_dtype = td.dtype
    if cur:
        buckets.append(cur)
    return buckets


class AutotuneTaskManager:
    def __init__(self, model_name: str, is_output_log: bool = False,
                 search_hierarchical: bool = False):
        self.model_name = model_name
        self.records: List[Tuple[int, BaguaHyperparameter, float]] = []
        self.tensor_list: List[TensorDeclaration] = []
        # tensor_name -> order index derived from telemetry spans
        self.tensor_order: Dict[str, int] = {}
        # hierarchical reduce only changes execution on MULTI-node runs
        # (one node: intra == global and the ops take the flat path), so
        # searching it on a single node injects pure noise into the
        # hill-climb — the reference searched-but-never-consumed it
        # (autotune_task_manager.py:48-59); here it is searched only when
        # it can matter and IS consumed (engine._ask_hyperparameters).
        space = {"bucket_size_2p": IntParam(25, (20, 31))}
        if search_hierarchical:
            space["is_hierarchical_reduce"] = BoolParam(False)
        self.search_hierarchical = search_hierarchical
        self.optimizer = BayesianOptimizer(space)
        self.sampling_start: Optional[float] = None
        self._log_writer = None
        if is_output_log:
            path = "autotune_logs_%s.csv" % model_name
            f = open(path, "a", newline="")
            self._log_writer = csv.writer(f)
            self._log_writer.writerow(
                ["time", "train_iter", "bucket_size_2p",
                 "is_hierarchical_reduce", "score"])

    # ------------------------------------------------------------------
    def set_tensor_list(self, tensor_list: List[TensorDeclaration]):
        self.tensor_list = tensor_list

    def report_span(self, tensor_name: str, start_time: int):
        """Telemetry span -> execution order. Smaller start time = earlier
        backward completion (reference derives a partial order,
        autotune_service.py:274-294)."""
        if tensor_name not in self.tensor_order:
            self.tensor_order[tensor_name] = start_time
        else:
            self.tensor_order[tensor_name] = min(
                self.tensor_order[tensor_name], start_time)

    def ordered_tensor_list(self) -> List[TensorDeclaration]:
        if not self.tensor_order:
            return self.tensor_list
        return sorted(
            self.tensor_list,
            key=lambda td: self.tensor_order.get(td.name, 1 << 62))

    # ------------------------------------------------------------------
    def record(self, train_iter: int, hp: BaguaHyperparameter, score: float):
        self.records.append((train_iter, hp, score))
        if self._log_writer is not None:
            self._log_writer.writerow([
                time.time(), train_iter, hp.bucket_size.bit_length() - 1,
                hp.is_hierarchical_reduce, score])

    def best_hyperparameters(self) -> Optional[BaguaHyperparameter]:
        if not self.records:
            return None
        return max(self.records, key=lambda r: r[2])[1]

    def tell_and_ask(self, prev_hp: BaguaHyperparameter,
                     prev_score: float) -> BaguaHyperparameter:
        point = {
            "bucket_size_2p": max(prev_hp.bucket_size, 1).bit_length() - 1,
        }
        if self.search_hierarchical:
            point["is_hierarchical_reduce"] = bool(
                prev_hp.is_hierarchical_reduce)
        self.optimizer.tell(point, prev_score)
        proposal = self.optimizer.ask()
        bucket_size = 1 << int(proposal["bucket_size_2p"])
        hier = (bool(proposal["is_hierarchical_reduce"])
                if self.search_hierarchical else None)
        hp = BaguaHyperparameter(
            bucket_size=bucket_size,
            is_hierarchical_reduce=hier,
            buckets=split_bucket_by_bucket_size(
                self.ordered_tensor_list(), bucket_size),
        )
        return hp
