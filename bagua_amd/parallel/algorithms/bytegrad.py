"""ByteGrad — MinMaxUInt8-compressed centralized gradient sync
(reference: bagua/torch_api/algorithms/bytegrad.py:1-82).

The wire path (compress -> alltoall -> reduce own chunk -> compress own ->
allgather -> decompress) maps unusually well onto one 8xMI355X node: xGMI
is fully connected point-to-point, so the alltoall is a single-hop
exchange on 7 links in parallel.
"""

from ...communication import BaguaProcessGroup
from .base import Algorithm, AlgorithmImpl


class ByteGradAlgorithmImpl(AlgorithmImpl):
    def __init__(self, process_group: BaguaProcessGroup,
                 hierarchical: bool = True, average: bool = True):
        super().__init__(process_group)
        self.hierarchical = hierarchical
        self.average = average

    def bucket_alignment(self) -> int:
        # chunked wire format needs numel % nranks == 0, and 32-byte-aligned
        # chunks for the compressed headers (reference: bytegrad.py:33-45)
        n = self.process_group.get_global_communicator().nranks()
        return n * 32

    def init_operations(self, ddp, bucket):
        bucket.clear_ops()
        bucket.append_centralized_synchronous_op(
            hierarchical=self.hierarchical,
            average=self.average,
            scattergather=True,
            compression="MinMaxUInt8",
            group=self.process_group,
        )


class ByteGradAlgorithm(Algorithm):
    def __init__(self, hierarchical: bool = True, average: bool = True):
        self.hierarchical = hierarchical
        self.average = average

    def reify(self, process_group: BaguaProcessGroup):
        return ByteGradAlgorithmImpl(
            process_group, hierarchical=self.hierarchical,
            average=self.average)
