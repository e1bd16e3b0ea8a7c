"""Multi-GPU hash repartition over RCCL/xGMI.

Replaces the reference's shuffle data plane (core.py:276-376 `push` ->
Arrow Flight do_put / flight.py cache -> do_get) for the join/group-by
repartition: rows are hash-partitioned by `key % world`
(quokka_runtime.py:222 semantics, bit-exact), columns are scattered into
partition-contiguous device buffers, and exchanged with ONE grouped
ncclSend/ncclRecv per column over the 7 p2p xGMI links (direct per-peer
sends, not a ring — SURVEY.md §5). Control (unique-id broadcast, count
exchange) rides on torch.distributed gloo, mirroring how the reference
keeps control state off the data path (Redis).

One rank per GPU. Works at world_size == 1 (self-exchange) so the code
path is testable on a single-GPU box.
"""
import ctypes

import numpy as np

from . import ops, shim
from .shim import DevColumn, c_u64, c_vp


class Comm:
    """RCCL communicator; unique id broadcast via torch.distributed (gloo),
    mirroring LocalCluster's host-side control plane (utils.py:96-161)."""

    def __init__(self, rank, world, dist=None):
        self.rank = rank
        self.world = world
        uid = (ctypes.c_uint8 * 128)()
        if rank == 0:
            shim.call("qk_comm_unique_id", uid)
        if world > 1:
            assert dist is not None, "need torch.distributed for rendezvous"
            import torch
            t = torch.tensor(np.frombuffer(bytes(uid), dtype=np.uint8))
            dist.broadcast(t, src=0)
            arr = t.numpy().tobytes()
            uid = (ctypes.c_uint8 * 128).from_buffer_copy(arr)
        self._dist = dist
        p = c_vp(0)
        shim.call("qk_comm_init", self.rank, self.world, uid,
                  ctypes.byref(p))
        self.handle = p

    def destroy(self):
        if self.handle is not None:
            shim.call("qk_comm_destroy", self.handle)
            self.handle = None

    def exchange_counts(self, send_counts):
        """all-to-all of per-peer row counts (host, gloo)."""
        if self.world == 1:
            return np.asarray(send_counts, dtype=np.uint64).copy()
        import torch
        inp = torch.tensor(np.asarray(send_counts, dtype=np.int64))
        out = torch.zeros_like(inp)
        self._dist.all_to_all_single(out, inp)
        return out.numpy().astype(np.uint64)

    def alltoallv_column(self, col, send_offsets, send_counts, recv_counts,
                         stream=None):
        """Exchange one partition-ordered DevColumn; returns the received
        DevColumn (partition-ordered by source rank)."""
        total = int(np.sum(recv_counts))
        recv = DevColumn(col.dtype, max(1, total))
        recv.n = total
        recv_offsets = np.zeros(self.world, dtype=np.uint64)
        np.cumsum(recv_counts[:-1], out=recv_offsets[1:])
        so = np.ascontiguousarray(send_offsets[: self.world], dtype=np.uint64)
        sc = np.ascontiguousarray(send_counts, dtype=np.uint64)
        ro = np.ascontiguousarray(recv_offsets, dtype=np.uint64)
        rc = np.ascontiguousarray(recv_counts, dtype=np.uint64)
        sh = stream.handle if stream else None
        shim.call("qk_alltoallv", sh, self.handle, self.world,
                  ctypes.c_uint32(col.dtype.itemsize), col.ptr,
                  so.ctypes.data_as(c_vp), sc.ctypes.data_as(c_vp),
                  recv.ptr, ro.ctypes.data_as(c_vp),
                  rc.ctypes.data_as(c_vp))
        return recv

    def allreduce_f64(self, buf_ptr, n, stream=None):
        sh = stream.handle if stream else None
        shim.call("qk_allreduce_f64", sh, self.handle, buf_ptr, c_u64(n))


def repartition(comm, key_col, payload_cols, stream=None):
    """Hash-repartition rows by key % world across ranks.

    key_col: DevColumn i64; payload_cols: dict name -> DevColumn (same n).
    Returns (recv_key_col, dict name -> recv DevColumn). This is the
    reference's partition_fn + push + pull (core.py:152-376) collapsed to:
    partition kernel -> gather per column -> grouped RCCL send/recv.
    """
    n = key_col.n
    world = comm.world
    offsets, idx = ops.partition_i64(key_col, world, stream, n)
    send_counts = np.diff(offsets).astype(np.uint64)
    recv_counts = comm.exchange_counts(send_counts)

    # per-column gather -> alltoallv are STREAM-ORDERED (RCCL grouped ops
    # and the D2D self-copy launch on `stream`), so the whole exchange
    # pipelines with ONE sync at the end; temps are freed only after it
    # (the pool would otherwise hand a buffer still being read by an
    # in-flight send to the next allocation)
    temps = []

    def exch(col):
        ordered = col.gather(idx, n, stream)
        temps.append(ordered)
        return comm.alltoallv_column(ordered, offsets, send_counts,
                                     recv_counts, stream)

    recv_key = exch(key_col)
    recv_payload = {name: exch(col) for name, col in payload_cols.items()}
    shim.call("qk_stream_sync", stream.handle if stream else None)
    for t in temps:
        t.free()
    idx.free()
    return recv_key, recv_payload
