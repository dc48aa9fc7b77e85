"""Bucket-op executor.

Interprets the op descriptors appended to a :class:`BaguaBucket`
(bagua_amd/bucket.py) against the group's communicators. On GPU, ops run on
the group's dedicated high-priority HIP stream and fence with events; on
CPU everything is synchronous over gloo.

Semantics mirror the reference Rust comm ops:
 * centralized sync (full / low precision):
   comm_ops/centralized_full_precision_synchronous.rs:16-56,
   comm_ops/centralized_low_precision_synchronous.rs:16-74
 * decentralized (all / shift_one):
   comm_ops/decentralized_full_precision_synchronous.rs:37-124
 * low-precision decentralized ring gossip:
   comm_ops/decentralized_low_precision_synchronous.rs:37-153
 * async model average:
   comm_ops/decentralized_full_precision_asynchronous.rs:47-212
 * hierarchical pre/post: communicators/mod.rs:261-348
"""

import logging

import torch

from . import ops
from .bucket import (
    AsyncModelAverageOp,
    BaguaBucket,
    CentralizedSyncOp,
    DecentralizedSyncOp,
    LowPrecisionDecentralizedSyncOp,
    PythonOp,
)
from .communication import BaguaProcessGroup, ReduceOp

logger = logging.getLogger(__name__)


class _HierCtx:
    """Hierarchical pre/post around an inner collective.

    pre: intra-node reduce (averaged) to the local leader;
    inner op runs on leaders over the inter-node communicator;
    post: intra-node broadcast from the leader.
    On a single node this degenerates to the flat path.
    """

    def __init__(self, group: BaguaProcessGroup, enabled: bool):
        self.group = group
        glob = group.get_global_communicator()
        self.flat_comm = glob
        self.multi_node = False
        if enabled:
            intra = group.get_intra_node_communicator()
            self.multi_node = intra.nranks() < glob.nranks()
        if self.multi_node:
            self.intra = group.get_intra_node_communicator()
            self.inter = group.get_inter_node_communicator()
            self.is_leader = self.intra.rank_in_comm == 0

    def comm(self):
        return self.inter if self.multi_node else self.flat_comm

    def pre(self, flat, average=True):
        if self.multi_node:
            self.intra.reduce_inplace(
                flat, 0, ReduceOp.AVG if average else ReduceOp.SUM)

    def run_inner(self) -> bool:
        """True if this rank participates in the inner op."""
        return (not self.multi_node) or self.is_leader

    def post(self, flat):
        if self.multi_node:
            self.intra.broadcast(flat, 0)


def execute_ops(bucket: BaguaBucket, group: BaguaProcessGroup, backend=None):
    """Run every op appended to the bucket, in order."""
    for op in bucket.ops:
        if isinstance(op, CentralizedSyncOp):
            _exec_centralized(op, bucket, op.group or group)
        elif isinstance(op, DecentralizedSyncOp):
            _exec_decentralized(op, bucket, op.group or group)
        elif isinstance(op, LowPrecisionDecentralizedSyncOp):
            _exec_low_prec_decentralized(op, bucket, op.group or group)
        elif isinstance(op, AsyncModelAverageOp):
            _exec_async_model_average(op, bucket, op.group or group)
        elif isinstance(op, PythonOp):
            op.fn(bucket.name)
        else:
            raise TypeError("unknown bucket op %r" % (op,))


# ---------------------------------------------------------------------------


def _exec_centralized(op: CentralizedSyncOp, bucket: BaguaBucket,
                      group: BaguaProcessGroup):
    with bucket.comm_view() as flat:
        ctx = _HierCtx(group, op.hierarchical)
        ctx.pre(flat, average=op.average)
        if ctx.run_inner():
            comm = ctx.comm()
            if op.compression is None:
                if not op.scattergather:
                    comm.allreduce_inplace(
                        flat, ReduceOp.AVG if op.average else ReduceOp.SUM)
                else:
                    _scattergather_sync(comm, flat, op.average)
            elif op.compression == "MinMaxUInt8":
                _compressed_sync(comm, flat, op.average)
            else:
                raise ValueError("unknown compression %r" % op.compression)
        ctx.post(flat)


def _scattergather_sync(comm, flat, average):
    """alltoall -> local chunk reduce -> allgather
    (reference: centralized_full_precision_synchronous.rs:33-43)."""
    n = comm.nranks()
    rank = comm.rank_in_comm
    assert flat.numel() % n == 0, "bucket not padded to nranks"
    comm.alltoall_inplace(flat)
    _on_comm_stream(comm, lambda: ops.reduce_chunk_inplace(
        flat, n, rank, average))
    comm.allgather_inplace(flat)


def _compressed_sync(comm, flat, average):
    """ByteGrad wire protocol
    (reference: centralized_low_precision_synchronous.rs:16-74):
    compress all chunks -> alltoall(u8) -> decompress -> reduce own chunk ->
    compress own chunk -> allgather(u8) -> decompress."""
    n = comm.nranks()
    rank = comm.rank_in_comm
    assert flat.numel() % n == 0, "bucket not padded to nranks"

    def phase1():
        return ops.compress_chunked(flat, n)

    comp = _on_comm_stream(comm, phase1)
    comm.alltoall_inplace(comp)

    def phase2():
        # fused dequantize+reduce: non-target chunks are only reduction
        # inputs, so they are never materialized into flat (saves ~2
        # bucket passes of HBM traffic per step)
        ops.dequant_reduce(comp, flat, n, rank, average)
        ops.compress_chunked(flat, n, target_chunk=rank, out=comp)

    _on_comm_stream(comm, phase2)
    comm.allgather_inplace(comp)

    def phase3():
        ops.decompress_chunked_into(comp, flat, n)

    _on_comm_stream(comm, phase3)


def _on_comm_stream(comm, fn):
    """Run local device math on the comm stream so it is ordered with the
    collectives without host syncs."""
    if comm.stream is not None and torch.cuda.is_available():
        from .communication import _event_get, _event_put

        curr = torch.cuda.current_stream()
        ev = _event_get()
        ev.record(curr)
        comm.stream.wait_event(ev)
        _event_put(ev)
        with torch.cuda.stream(comm.stream):
            out = fn()
        done = _event_get()
        done.record(comm.stream)
        curr.wait_event(done)
        _event_put(done)
        return out
    return fn()


# ---------------------------------------------------------------------------


def _exec_decentralized(op: DecentralizedSyncOp, bucket: BaguaBucket,
                        group: BaguaProcessGroup):
    with bucket.comm_view() as flat:
        _exec_decentralized_inner(op, flat, group)
    op.step += 1


def _exec_decentralized_inner(op, flat, group):
    peer = op.peer_weight.tensor()
    ctx = _HierCtx(group, op.hierarchical)
    ctx.pre(flat, average=True)
    if ctx.run_inner():
        comm = ctx.comm()
        if op.peer_selection_mode == "all":
            peer.copy_(flat)
            comm.allreduce_inplace(peer, ReduceOp.AVG)
        elif op.peer_selection_mode == "shift_one":
            n = comm.nranks()
            rank = comm.rank_in_comm
            assert n % 2 == 0, (
                "shift_one needs an even number of peers (got %d)" % n)
            step = op.step
            # half-ring pairing (reference:
            # decentralized_full_precision_synchronous.rs:79-85)
            if rank < n // 2:
                peer_rank = ((step + rank) % ((n + 1) // 2)) + n // 2
            else:
                peer_rank = (rank - n // 2 - step) % (n // 2)
            comm.group_start()
            comm.send(flat, peer_rank)
            comm.recv(peer, peer_rank)
            comm.group_end()
            _on_comm_stream(comm, lambda: ops.average_inplace(peer, flat))
        else:
            raise ValueError(op.peer_selection_mode)


def copy_back_peer_weight(op: DecentralizedSyncOp, bucket: BaguaBucket,
                          group: BaguaProcessGroup):
    """Post-backward: install averaged weights
    (reference: decentralized_full_precision_synchronous.rs:105-124)."""
    with bucket.comm_view() as flat:
        ctx = _HierCtx(group, op.hierarchical)
        if ctx.run_inner():
            _on_comm_stream(ctx.comm(),
                            lambda: flat.copy_(op.peer_weight.tensor()))
        ctx.post(flat)


# ---------------------------------------------------------------------------


def _exec_low_prec_decentralized(op: LowPrecisionDecentralizedSyncOp,
                                 bucket: BaguaBucket,
                                 group: BaguaProcessGroup):
    """Difference-compressed ring gossip
    (reference: decentralized_low_precision_synchronous.rs:37-153):

        t  = x + L/3 + R/3 - 5W/3
        c  = compress(t);  send c to both ring neighbors
        L += decompress(c_left);  R += decompress(c_right)
        W  = W + decompress(c);   x = W
    """
    with bucket.comm_view() as flat:
        _exec_low_prec_inner(op, flat, group)


def _exec_low_prec_inner(op, flat, group):
    W = op.weight.tensor()
    L = op.left_peer_weight.tensor()
    R = op.right_peer_weight.tensor()
    ctx = _HierCtx(group, op.hierarchical)
    ctx.pre(flat, average=True)
    if ctx.run_inner():
        comm = ctx.comm()
        n = comm.nranks()
        rank = comm.rank_in_comm

        def diff_and_compress():
            ops.addmul_inplace(flat, L, 1.0 / 3.0)
            ops.addmul_inplace(flat, R, 1.0 / 3.0)
            ops.addmul_inplace(flat, W, -5.0 / 3.0)
            return ops.compress_chunked(flat, 1)

        comp = _on_comm_stream(comm, diff_and_compress)
        lrecv = torch.empty_like(comp)
        rrecv = torch.empty_like(comp)
        left = (rank + n - 1) % n
        right = (rank + 1) % n
        comm.group_start()
        comm.send(comp, left)
        comm.send(comp, right)
        comm.recv(lrecv, left)
        comm.recv(rrecv, right)
        comm.group_end()

        def apply():
            ops.decompress_chunked_into(lrecv, flat, 1)
            ops.add_inplace(L, flat)
            ops.decompress_chunked_into(rrecv, flat, 1)
            ops.add_inplace(R, flat)
            ops.decompress_chunked_into(comp, flat, 1)
            ops.add_inplace(flat, W)
            W.copy_(flat)

        _on_comm_stream(comm, apply)
    ctx.post(flat)


# ---------------------------------------------------------------------------


def _exec_async_model_average(op: AsyncModelAverageOp, bucket: BaguaBucket,
                              group: BaguaProcessGroup):
    """One async-averaging round
    (reference: decentralized_full_precision_asynchronous.rs:47-178).

    Abort negotiation (allreduce-MIN of the status byte) keeps ranks
    consistent; weights are snapshotted under the op's weight lock, the
    allreduce runs on the comm stream, and the correction
    ``x += reduced/n - x_copy`` is applied under the lock again so the
    training thread never sees a half-applied average.
    """
    with bucket.comm_view() as flat:
        _exec_async_inner(op, flat, group)


def _exec_async_inner(op, flat, group):
    comm = (op.group or group).get_global_communicator()
    n = comm.nranks()

    flag = torch.ones(1, device=flat.device) * (1.0 if op._status else 0.0)
    comm.allreduce_inplace(flag, ReduceOp.MIN)
    if flat.is_cuda:
        torch.cuda.current_stream().synchronize()
    if flag.item() < 0.5:
        op._status = False
        return

    with op._weight_lock:
        x_copy = flat.detach().clone()
    reduced = torch.empty_like(x_copy)
    comm.allreduce(x_copy, reduced, ReduceOp.SUM)
    with op._weight_lock:
        _on_comm_stream(comm, lambda: ops.async_model_average(
            flat, reduced, x_copy, n))
        if flat.is_cuda:
            torch.cuda.current_stream().synchronize()
