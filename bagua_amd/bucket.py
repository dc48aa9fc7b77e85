"""BaguaBucket — gradient fusion bucket plus declarative comm ops.

Re-design of the reference bucket (bagua/torch_api/bucket.py:18-366 and the
Rust-side op constructors, bagua-core-internal/src/datatypes/mod.rs:1126-1310).
Instead of reifying native op objects at append time, algorithms append
**op descriptors** (plain dataclasses). Two executors interpret them:

* ``bagua_amd.executor.TorchExecutor`` — torch.distributed-based, runs on
  CPU/gloo (tests) and as a GPU fallback;
* the native C++/RCCL scheduler (``bagua_amd._C``) — the MI355X hot path.

Flattening allocates one contiguous buffer per bucket, padded so its numel
is divisible by the world size times the dtype's 32-byte alignment quantum
(needed by the scatter-gather and compressed paths; reference pads with a
``bagua_padding_tensor`` — here padding is just buffer tail, auto-ready by
construction).
"""

import contextlib
from dataclasses import dataclass, field
from typing import Callable, List, Optional

import torch

from .tensor import BaguaTensor


# ---------------------------------------------------------------------------
# Op descriptors (interpreted by executors)
# ---------------------------------------------------------------------------


@dataclass
class CentralizedSyncOp:
    """Synchronous centralized communication (allreduce family).

    reference: bucket.py:167-213 /
    comm_ops/centralized_{full,low}_precision_synchronous.rs
    """

    hierarchical: bool = False
    average: bool = True
    scattergather: bool = False
    compression: Optional[str] = None  # None | "MinMaxUInt8"
    group: object = None  # BaguaProcessGroup


@dataclass
class DecentralizedSyncOp:
    """Decentralized model averaging (reference: bucket.py:215-263)."""

    peer_selection_mode: str = "all"  # all | shift_one
    peer_weight: Optional[BaguaTensor] = None
    hierarchical: bool = True
    group: object = None
    step: int = 0  # shift_one pairing counter, advanced per execution


@dataclass
class LowPrecisionDecentralizedSyncOp:
    """Difference-compressed ring gossip (reference: bucket.py:265-320)."""

    weight: Optional[BaguaTensor] = None
    left_peer_weight: Optional[BaguaTensor] = None
    right_peer_weight: Optional[BaguaTensor] = None
    hierarchical: bool = True
    compression: str = "MinMaxUInt8"
    group: object = None


@dataclass
class AsyncModelAverageOp:
    """Asynchronous model averaging (reference: bucket.py:322-352)."""

    peer_selection_mode: str = "all"
    sync_interval_ms: int = 500
    group: object = None
    # runtime state
    _status: bool = field(default=True, repr=False)
    _weight_lock: object = field(default=None, repr=False)


@dataclass
class PythonOp:
    """Run a Python callable(bucket_name) on the scheduler thread
    (reference: comm_ops/python_ffi_op.rs:12-25)."""

    fn: Callable = None


class BaguaBucket:
    def __init__(
        self,
        tensors: List[BaguaTensor],
        name: str,
        flatten: bool = True,
        alignment: int = 1,
    ):
        """Group ``tensors`` into one scheduling and fusion unit.

        Args:
            tensors: registration records, all same dtype/device.
            name: bucket name, unique within a backend.
            flatten: fuse effective tensors into one contiguous buffer.
            alignment: pad flattened numel to a multiple of this (algorithms
                pass world_size so chunked ops divide evenly).
        """
        self.tensors = list(tensors)
        self.name = name
        self.padding = 0
        self.backend_bucket = None  # native registration handle
        self.flatten = flatten
        self.alignment = alignment
        self.ops: List[object] = []
        self._flat: Optional[torch.Tensor] = None
        self._gather_buf: Optional[torch.Tensor] = None

        for t in self.tensors:
            t.bucket = self

        if flatten:
            self._flatten_(alignment)

    # ------------------------------------------------------------------
    @property
    def bytes(self) -> int:
        return sum(t.numel() * t.tensor().element_size() for t in self.tensors)

    def _flatten_(self, alignment: int):
        effs = [t.materialized() for t in self.tensors]
        total = sum(e.numel() for e in effs)
        if alignment > 1:
            total = ((total + alignment - 1) // alignment) * alignment
        self.padding = total - sum(e.numel() for e in effs)
        e0 = effs[0]
        flat = torch.zeros(total, dtype=e0.dtype, device=e0.device)
        offset = 0
        for t in self.tensors:
            t.set_storage(flat, offset)
            offset += t.numel()
        self._flat = flat

    def flattened_tensor(self) -> torch.Tensor:
        """A contiguous tensor containing all bucket tensors' data. If the
        bucket is flattened this is (a clone of) the fused buffer;
        otherwise a fresh gather (reference: bucket.py:112-133)."""
        if self._flat is not None:
            return self._flat.clone()
        effs = [t.tensor() for t in self.tensors]
        return torch.cat([e.reshape(-1) for e in effs])

    def comm_tensor(self) -> torch.Tensor:
        """The in-place communication view: the fused buffer when
        flattened; for non-flattened buckets a persistent gather buffer
        (use :meth:`comm_view` so data is staged/copied back — reference:
        datatypes/mod.rs:1011-1088 force_copy path)."""
        if self._flat is not None:
            return self._flat
        if self._gather_buf is None:
            total = sum(t.numel() for t in self.tensors)
            if self.alignment > 1:
                total = ((total + self.alignment - 1)
                         // self.alignment) * self.alignment
            e0 = self.tensors[0].materialized()
            self._gather_buf = torch.zeros(total, dtype=e0.dtype,
                                           device=e0.device)
        return self._gather_buf

    def comm_view(self):
        """Context manager yielding the communication buffer; stages
        tensors in and copies results back when the bucket is not
        flattened (zero-copy when it is)."""

        @contextlib.contextmanager
        def ctx():
            buf = self.comm_tensor()
            if self._flat is None:
                offset = 0
                for t in self.tensors:
                    eff = t.materialized()
                    buf.narrow(0, offset, eff.numel()).copy_(
                        eff.detach().reshape(-1))
                    offset += eff.numel()
            yield buf
            if self._flat is None:
                offset = 0
                for t in self.tensors:
                    eff = t.tensor()
                    eff.detach().reshape(-1).copy_(
                        buf.narrow(0, offset, eff.numel()))
                    offset += eff.numel()

        return ctx()

    def check_flatten(self) -> bool:
        if self._flat is None:
            return False
        ptr = self._flat.data_ptr()
        offset = 0
        for t in self.tensors:
            e = t.tensor()
            if e.data_ptr() != ptr + offset * e.element_size():
                return False
            offset += e.numel()
        return True

    # ------------------------------------------------------------------
    # op append API (reference: bucket.py:134-352)
    # ------------------------------------------------------------------
    def append_python_op(self, python_function, group=None):
        self.ops.append(PythonOp(fn=python_function))
        return self

    def append_centralized_synchronous_op(
        self,
        hierarchical: bool = False,
        average: bool = True,
        scattergather: bool = False,
        compression: Optional[str] = None,
        group=None,
    ):
        self.ops.append(CentralizedSyncOp(
            hierarchical=hierarchical, average=average,
            scattergather=scattergather, compression=compression, group=group))
        return self

    def append_decentralized_synchronous_op(
        self,
        peer_weight: BaguaTensor,
        hierarchical: bool = True,
        peer_selection_mode: str = "all",
        group=None,
    ):
        op = DecentralizedSyncOp(
            peer_selection_mode=peer_selection_mode, peer_weight=peer_weight,
            hierarchical=hierarchical, group=group)
        self.ops.append(op)
        return op

    def append_low_precision_decentralized_synchronous_op(
        self,
        weight: BaguaTensor,
        left_peer_weight: BaguaTensor,
        right_peer_weight: BaguaTensor,
        hierarchical: bool = True,
        compression: str = "MinMaxUInt8",
        group=None,
    ):
        op = LowPrecisionDecentralizedSyncOp(
            weight=weight, left_peer_weight=left_peer_weight,
            right_peer_weight=right_peer_weight, hierarchical=hierarchical,
            compression=compression, group=group)
        self.ops.append(op)
        return op

    def append_asynchronous_model_average_op(self, peer_selection_mode="all",
                                             sync_interval_ms=500, group=None):
        import threading

        op = AsyncModelAverageOp(
            peer_selection_mode=peer_selection_mode,
            sync_interval_ms=sync_interval_ms, group=group)
        op._weight_lock = threading.Lock()
        self.ops.append(op)
        return op

    def clear_ops(self) -> "BaguaBucket":
        self.ops = []
        return self

    # ------------------------------------------------------------------
    def ready_for_comm(self) -> bool:
        return all(t.ready for t in self.tensors)

    def reset_ready(self):
        for t in self.tensors:
            t.ready = False
