"""Synchronous full-precision gradient allreduce — the torch-DDP-equivalent
algorithm (reference: bagua/torch_api/algorithms/gradient_allreduce.py:1-64).
"""

from ...communication import BaguaProcessGroup
from .base import Algorithm, AlgorithmImpl


class GradientAllReduceAlgorithmImpl(AlgorithmImpl):
    def __init__(self, process_group: BaguaProcessGroup,
                 hierarchical: bool = False, average: bool = True):
        super().__init__(process_group)
        self.hierarchical = hierarchical
        self.average = average

    def init_operations(self, ddp, bucket):
        bucket.clear_ops()
        bucket.append_centralized_synchronous_op(
            hierarchical=self.hierarchical,
            average=self.average,
            group=self.process_group,
        )


class GradientAllReduceAlgorithm(Algorithm):
    def __init__(self, hierarchical: bool = False, average: bool = True):
        self.hierarchical = hierarchical
        self.average = average

    def reify(self, process_group: BaguaProcessGroup):
        return GradientAllReduceAlgorithmImpl(
            process_group, hierarchical=self.hierarchical,
            average=self.average)
