"""Decentralized SGD — full-precision and low-precision (difference
compressed) variants (reference: bagua/torch_api/algorithms/decentralized.py).

Both communicate *weights*, not gradients, in ONE bucket:

* full precision: weights exchanged at forward-pre (overlapping the whole
  fwd+bwd), averaged peer weights copied back post-backward;
* low precision: gossip fires post-optimizer-step on a ring with
  MinMaxUInt8-compressed weight differences.
"""

from typing import List

import torch

from ...bucket import BaguaBucket
from ...communication import BaguaProcessGroup
from ...executor import copy_back_peer_weight
from ...tensor import BaguaTensor, ensure_bagua_tensor
from .base import Algorithm, AlgorithmImpl


class DecentralizedAlgorithmImpl(AlgorithmImpl):
    def __init__(self, process_group: BaguaProcessGroup,
                 hierarchical: bool = True, peer_selection_mode: str = "all",
                 communication_interval: int = 1):
        super().__init__(process_group)
        self.hierarchical = hierarchical
        self.peer_selection_mode = peer_selection_mode
        self.communication_interval = communication_interval
        self.tensors: List[BaguaTensor] = []

    def _should_communicate(self, ddp) -> bool:
        cur_step = ddp.bagua_train_step_counter - 1
        return cur_step % self.communication_interval == 0

    def init_tensors(self, ddp) -> List[BaguaTensor]:
        # register the WEIGHTS themselves (reference: decentralized.py:44-51)
        parameters = ddp.bagua_build_params()
        self.tensors = [
            ddp.ensure_bagua_tensor(param, name)
            for name, param in reversed(parameters)
        ]
        return self.tensors

    def tensors_to_buckets(self, tensors, do_flatten):
        # all params in ONE bucket (reference: decentralized.py:52-61)
        all_tensors = []
        for group in tensors:
            all_tensors.extend(group)
        return [BaguaBucket(all_tensors, str(0), flatten=do_flatten)]

    def init_forward_pre_hook(self, ddp):
        def hook(input):
            if self._should_communicate(ddp):
                for tensor in self.tensors:
                    tensor.mark_communication_ready(ddp.bagua_backend)

        return hook

    def init_backward_hook(self, ddp):
        def hook(parameter_name, parameter):
            return

        return hook

    def init_post_backward_hook(self, ddp):
        def hook():
            if self._should_communicate(ddp):
                ddp.bagua_backend.wait_pending_comm_ops_host()
                for bucket in ddp.bagua_buckets:
                    copy_back_peer_weight(
                        bucket._decentralized_op, bucket, self.process_group)

        return hook

    def _init_states(self, bucket: BaguaBucket):
        weight_tensor = bucket.flattened_tensor()
        bucket._peer_weight = ensure_bagua_tensor(weight_tensor, "peer_weight")

    def init_operations(self, ddp, bucket: BaguaBucket):
        self._init_states(bucket)
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        bucket.clear_ops()
        op = bucket.append_decentralized_synchronous_op(
            peer_weight=bucket._peer_weight,
            hierarchical=self.hierarchical,
            peer_selection_mode=self.peer_selection_mode,
            group=self.process_group,
        )
        bucket._decentralized_op = op


class DecentralizedAlgorithm(Algorithm):
    def __init__(self, hierarchical: bool = True,
                 peer_selection_mode: str = "all",
                 communication_interval: int = 1):
        self.hierarchical = hierarchical
        self.peer_selection_mode = peer_selection_mode
        self.communication_interval = communication_interval

    def reify(self, process_group: BaguaProcessGroup):
        return DecentralizedAlgorithmImpl(
            process_group,
            hierarchical=self.hierarchical,
            peer_selection_mode=self.peer_selection_mode,
            communication_interval=self.communication_interval,
        )


class LowPrecisionDecentralizedAlgorithmImpl(AlgorithmImpl):
    def __init__(self, process_group: BaguaProcessGroup,
                 hierarchical: bool = True, communication_interval: int = 1):
        super().__init__(process_group)
        self.hierarchical = hierarchical
        self.communication_interval = communication_interval
        self.tensors: List[BaguaTensor] = []

    def _should_communicate(self, ddp) -> bool:
        cur_step = ddp.bagua_train_step_counter - 1
        return cur_step % self.communication_interval == 0

    def init_tensors(self, ddp) -> List[BaguaTensor]:
        parameters = ddp.bagua_build_params()
        self.tensors = [
            ddp.ensure_bagua_tensor(param, name)
            for name, param in reversed(parameters)
        ]
        optimizer_param_ids = [
            id(param)
            for optimizer in ddp.bagua_optimizers
            for group in optimizer.param_groups
            for param in group["params"]
        ]
        for name, param in parameters:
            if id(param) not in optimizer_param_ids:
                raise RuntimeError(
                    "Module parameter %s is not used by your optimizer(s); "
                    "exclude it via _bagua_params_and_buffers_to_ignore"
                    % name)
        return self.tensors

    def tensors_to_buckets(self, tensors, do_flatten):
        all_tensors = []
        for group in tensors:
            all_tensors.extend(group)
        return [BaguaBucket(all_tensors, str(0), flatten=do_flatten,
                            alignment=32)]

    def init_backward_hook(self, ddp):
        def hook(parameter_name, parameter):
            pass

        return hook

    def init_post_backward_hook(self, ddp):
        def hook():
            pass

        return hook

    def init_post_optimizer_step_hook(self, ddp):
        from ...contrib.fused_optimizer import is_fused_optimizer

        def hook(optimizer: torch.optim.Optimizer):
            assert not is_fused_optimizer(optimizer), (
                "low-precision decentralized cannot run on a fused optimizer")
            if self._should_communicate(ddp):
                for tensor in self.tensors:
                    tensor.mark_communication_ready(ddp.bagua_backend)
                ddp.bagua_backend.wait_pending_comm_ops()

        return hook

    def _init_states(self, bucket: BaguaBucket):
        bucket._weight = ensure_bagua_tensor(
            bucket.flattened_tensor(), "weight")
        bucket._left_peer_weight = ensure_bagua_tensor(
            bucket.flattened_tensor(), "left_peer_weight")
        bucket._right_peer_weight = ensure_bagua_tensor(
            bucket.flattened_tensor(), "right_peer_weight")

    def init_operations(self, ddp, bucket: BaguaBucket):
        self._init_states(bucket)
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        bucket.clear_ops()
        bucket.append_low_precision_decentralized_synchronous_op(
            weight=bucket._weight,
            left_peer_weight=bucket._left_peer_weight,
            right_peer_weight=bucket._right_peer_weight,
            hierarchical=self.hierarchical,
            compression="MinMaxUInt8",
            group=self.process_group,
        )


class LowPrecisionDecentralizedAlgorithm(Algorithm):
    def __init__(self, hierarchical: bool = True,
                 communication_interval: int = 1):
        self.hierarchical = hierarchical
        self.communication_interval = communication_interval

    def reify(self, process_group: BaguaProcessGroup):
        return LowPrecisionDecentralizedAlgorithmImpl(
            process_group,
            hierarchical=self.hierarchical,
            communication_interval=self.communication_interval,
        )
