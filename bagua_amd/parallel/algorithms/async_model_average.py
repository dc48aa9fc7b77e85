"""Asynchronous model averaging
(reference: bagua/torch_api/algorithms/async_model_average.py:33-347).

A background thread continuously averages the model weights with all peers
(allreduce-SUM into scratch, then ``x += reduced/n - x_copy`` under a
weight lock) while the training loop runs uninterrupted. Abort/resume use
a distributed MIN-negotiation so all ranks stop consistently.
"""

import enum
import logging
import threading
import time
from concurrent.futures import ThreadPoolExecutor, wait
from typing import List

import torch

from ...bucket import BaguaBucket
from ...communication import BaguaProcessGroup, barrier, new_group
from ...tensor import BaguaTensor
from .base import Algorithm, AlgorithmImpl

logger = logging.getLogger(__name__)


class _AsyncInternalState(enum.IntEnum):
    NEW = 0
    SCHEDULED = 1
    STARTED = 2
    STOPPED = 3


class AsyncModelAverageAlgorithmImpl(AlgorithmImpl):
    def __init__(self, process_group: BaguaProcessGroup,
                 peer_selection_mode: str = "all",
                 sync_interval_ms: int = 500, warmup_steps: int = 0):
        super().__init__(process_group)
        assert peer_selection_mode == "all", (
            "async model average supports peer_selection_mode='all' only")
        self.peer_selection_mode = peer_selection_mode
        self.sync_interval_ms = sync_interval_ms
        self.step_id = 0
        self.warmup_steps = warmup_steps

        self.executor = ThreadPoolExecutor(max_workers=1)
        self.cv = threading.Condition()
        self.notified = False
        self.status = _AsyncInternalState.NEW
        self.future = None

        # dedicated group + stream for the background loop; ``dedicated``
        # gives it its own torch/RCCL channel so the loop's collectives
        # can never interleave with main-thread collectives on the default
        # communicator (reference: async_model_average.py:72-82 also used
        # a separate comm + stream)
        self.thread_group = new_group(
            process_group.ranks,
            group_name=process_group.group_name + "_async",
            dedicated=True)

    # ------------------------------------------------------------------
    def tensors_to_buckets(self, tensors, do_flatten) -> List[BaguaBucket]:
        assert do_flatten, "async algorithm supports do_flatten=True only"
        if self.step_id < self.warmup_steps:
            return super().tensors_to_buckets(tensors, do_flatten)
        all_tensors = []
        for group in tensors:
            all_tensors.extend(group)
        return [BaguaBucket(all_tensors, str(0), flatten=do_flatten)]

    def init_tensors(self, ddp) -> List[BaguaTensor]:
        parameters = ddp.bagua_build_params()
        tensors = []
        for name, param in reversed(parameters):
            if self.step_id < self.warmup_steps:
                if param.grad is None:
                    param.grad = torch.zeros_like(param)
                t = ddp.ensure_bagua_tensor(
                    param, name,
                    getter_closure=lambda p: p.grad,
                    setter_closure=lambda p, t: setattr(p, "grad", t))
            else:
                t = ddp.ensure_bagua_tensor(param, name)
            tensors.append(t)
        return tensors

    def init_forward_pre_hook(self, ddp):
        def hook(input):
            if self.step_id > self.warmup_steps and self.sync_interval_ms > 0:
                if self.status == _AsyncInternalState.NEW:
                    self.future = self.executor.submit(
                        self._run_async_loop, ddp)
                    self.status = _AsyncInternalState.SCHEDULED
                elif (self.status == _AsyncInternalState.SCHEDULED
                      and self.future.running()):
                    with self.cv:
                        self.notified = True
                        self.cv.notify()
                    self.status = _AsyncInternalState.STARTED
                self._lock_model(ddp)

        return hook

    def init_backward_hook(self, ddp):
        def hook(parameter_name, parameter):
            if self.step_id <= self.warmup_steps:
                ddp._bagua_tensor_map[parameter_name].mark_communication_ready(
                    ddp.bagua_backend)

        return hook

    def init_post_backward_hook(self, ddp):
        def hook():
            if self.step_id <= self.warmup_steps:
                ddp.bagua_backend.wait_pending_comm_ops()
            else:
                self._unlock_model(ddp)

        return hook

    def need_reset(self):
        self.step_id += 1
        if self.warmup_steps > 0 and self.step_id == self.warmup_steps + 1:
            logger.info("async model average starts at step %d", self.step_id)
            return True
        return False

    def init_operations(self, ddp, bucket: BaguaBucket):
        ddp.bagua_backend.wait_pending_comm_ops_host()
        bucket.clear_ops()
        if self.step_id < self.warmup_steps:
            bucket.append_centralized_synchronous_op(
                hierarchical=False, average=True, group=self.thread_group)
        else:
            op = bucket.append_asynchronous_model_average_op(
                peer_selection_mode=self.peer_selection_mode,
                sync_interval_ms=self.sync_interval_ms,
                group=self.thread_group)
            bucket._async_op = op
        # build the dedicated RCCL communicator NOW, on the main thread at
        # a point every rank reaches in the same order — constructing it
        # lazily from the background loop's first allreduce would race the
        # main thread's collectives on the default communicator
        # (ncclCommInitRank is itself collective).
        self.thread_group.ensure_native_communicators()

    # ------------------------------------------------------------------
    def _sync_device(self):
        if torch.cuda.is_available():
            torch.cuda.current_stream().synchronize()

    def _lock_model(self, ddp):
        self._sync_device()
        for bucket in ddp.bagua_buckets:
            bucket._async_op._weight_lock.acquire()

    def _unlock_model(self, ddp):
        self._sync_device()
        for bucket in ddp.bagua_buckets:
            bucket._async_op._weight_lock.release()

    def _op_active(self, ddp) -> bool:
        return (hasattr(ddp.bagua_buckets[0], "_async_op")
                and ddp.bagua_buckets[0]._async_op._status)

    def _run_async_loop(self, ddp):
        with self.cv:
            while not self.notified:
                self.cv.wait()
        comm_step = 0
        while self._op_active(ddp):
            start = time.time()
            for bucket in ddp.bagua_buckets:
                ddp.bagua_backend.execute_bucket_now(bucket)
            ddp.bagua_backend.wait_pending_comm_ops_host()
            logger.debug("async comm step %d took %.1f ms", comm_step,
                         (time.time() - start) * 1e3)
            comm_step += 1
            time.sleep(self.sync_interval_ms / 1000)

    # ------------------------------------------------------------------
    @staticmethod
    def _unwrap(ddp):
        from ..engine import BaguaDistributedDataParallel

        if isinstance(ddp, BaguaDistributedDataParallel):
            return ddp
        if hasattr(ddp, "inner"):
            return ddp.inner
        if hasattr(ddp, "bagua_ddp"):
            return ddp.bagua_ddp
        raise TypeError("unexpected ddp object %r" % type(ddp))

    def abort(self, ddp):
        """Stop the background loop consistently on all ranks."""
        ddp = self._unwrap(ddp)
        if self.status in (_AsyncInternalState.SCHEDULED,
                           _AsyncInternalState.STARTED):
            barrier(comm=self.process_group.get_global_communicator())
            if hasattr(ddp.bagua_buckets[0], "_async_op"):
                ddp.bagua_buckets[0]._async_op._status = False
            with self.cv:
                self.notified = True
                self.cv.notify()
            wait([self.future])
            self.status = _AsyncInternalState.STOPPED

    def resume(self, ddp):
        ddp = self._unwrap(ddp)
        if self.status in (_AsyncInternalState.NEW,
                           _AsyncInternalState.STOPPED):
            barrier(comm=self.process_group.get_global_communicator())
            if hasattr(ddp.bagua_buckets[0], "_async_op"):
                ddp.bagua_buckets[0]._async_op._status = True
            self.notified = False
            self.future = self.executor.submit(self._run_async_loop, ddp)
            self.status = _AsyncInternalState.SCHEDULED


class AsyncModelAverageAlgorithm(Algorithm):
    def __init__(self, peer_selection_mode: str = "all",
                 sync_interval_ms: int = 500, warmup_steps: int = 0):
        self.peer_selection_mode = peer_selection_mode
        self.sync_interval_ms = sync_interval_ms
        self.warmup_steps = warmup_steps

    def reify(self, process_group: BaguaProcessGroup):
        return AsyncModelAverageAlgorithmImpl(
            process_group,
            peer_selection_mode=self.peer_selection_mode,
            sync_interval_ms=self.sync_interval_ms,
            warmup_steps=self.warmup_steps)
