"""Direct one-hop p2p alltoall over xGMI (prototype, opt-in).

The 8-GPU MI355X node is fully connected point-to-point (7 xGMI links
per GPU), so an alltoall needs no ring: every rank pulls its chunk from
each peer's exported buffer in one hop. This module wraps the native
``_C.P2PAlltoAll`` (IPC-exported buffers + device sequence-number
barrier, see ops/csrc/core.cpp) with the store-based handle exchange.

Enable with ``BAGUA_P2P_ALLTOALL=1``; the default path stays
``ncclAllToAll`` (RCCL's own p2p transport). The 8-GPU A/B comparison
is the acceptance test for flipping the default
(SURVEY §5 "Distributed communication backend", hard part #4).
"""

import base64
import logging
import os

import torch.distributed.distributed_c10d as c10d

from . import native

logger = logging.getLogger(__name__)

_MIN_CAPACITY = 64 * 1024 * 1024  # one bucket of default size, padded


def enabled() -> bool:
    return os.environ.get("BAGUA_P2P_ALLTOALL", "0") == "1"


class P2pAllToAll:
    """One instance per (communicator, capacity-epoch). Construction is
    COLLECTIVE: every rank of the communicator must construct with the
    same arguments in the same order (the handle exchange goes through
    the c10d store)."""

    def __init__(self, name: str, rank: int, nranks: int, stream_ptr: int,
                 capacity: int, epoch: int = 0):
        self.rank = rank
        self.nranks = nranks
        self.capacity = capacity
        from .. import communication

        self.impl = native.lib().P2PAlltoAll(
            rank=rank, nranks=nranks, stream=stream_ptr, capacity=capacity)
        store = c10d._get_default_store()
        # keyed by BOTH the grow-epoch and the process-group deinit epoch
        # so stale handle blobs from a previous init can never be read
        key = "bagua_p2p_{}_{}_{}_{}".format(
            name, communication._uid_epoch, epoch, rank)
        store.set(key, base64.b64encode(bytes(self.impl.handles())).decode())
        handles = []
        for p in range(nranks):
            pkey = "bagua_p2p_{}_{}_{}_{}".format(
                name, communication._uid_epoch, epoch, p)
            handles.append(base64.b64decode(store.get(pkey)))
        self.impl.connect(handles)
        logger.info("p2p alltoall connected: %d ranks, %d MiB capacity",
                    nranks, capacity >> 20)

    def alltoall(self, t_in, t_out):
        self.impl.alltoall(t_in, t_out)

    def allgather_inplace(self, t):
        self.impl.allgather_inplace(t)


def get_for_communicator(comm, bytes_needed: int):
    """Cached, lazily grown instance on a BaguaCommunicator. Collective:
    all ranks call with identical sizes in identical order (bucket
    schedules are rank-identical by construction)."""
    capacity = max(_MIN_CAPACITY, 1 << (bytes_needed - 1).bit_length())
    cur = getattr(comm, "_p2p_a2a", None)
    if cur is not None and cur.capacity >= bytes_needed:
        return cur
    epoch = getattr(comm, "_p2p_epoch", 0) + 1
    comm._p2p_epoch = epoch
    comm._p2p_a2a = P2pAllToAll(
        comm.name, comm.rank_in_comm, comm.nranks(),
        comm.stream.cuda_stream, capacity, epoch)
    return comm._p2p_a2a
