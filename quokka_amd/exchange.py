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

    def alltoallv_into(self, col, recv, so, sc, ro, rc, stream=None):
        """alltoallv one column slice-set into a PREALLOCATED recv column
        (the overlapped exchange reuses one recv allocation across chunks;
        so/sc/ro/rc are per-peer element offsets/counts, uint64[world])."""
        sh = stream.handle if stream else None
        so = np.ascontiguousarray(so, dtype=np.uint64)
        sc = np.ascontiguousarray(sc, dtype=np.uint64)
        ro = np.ascontiguousarray(ro, dtype=np.uint64)
        rc = np.ascontiguousarray(rc, dtype=np.uint64)
        shim.call("qk_alltoallv", sh, self.handle, self.world,
                  ctypes.c_uint32(col.dtype.itemsize), col.ptr,
                  so.ctypes.data_as(c_vp), sc.ctypes.data_as(c_vp),
                  recv.ptr, ro.ctypes.data_as(c_vp),
                  rc.ctypes.data_as(c_vp))


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


def plan_chunks(send_offsets, send_counts, recv_counts, nchunks):
    """Deterministic chunk plan both sides of the exchange derive from the
    SAME exchanged counts (keeping RCCL grouped send/recv matching aligned
    across ranks, like the 256 MiB split inside qk_alltoallv).

    Peer p's count c splits into nchunks pieces of size
    c*(j+1)//nchunks - c*j//nchunks. The receive layout is CHUNK-MAJOR so
    each received chunk is one contiguous row range (probe-able while the
    next chunk is still in flight):
        chunk j rows = [chunk_start[j], chunk_start[j] + chunk_rows[j])
    Returns (per_chunk, chunk_start, chunk_rows) with per_chunk[j] =
    (so_j, sc_j, ro_j, rc_j) uint64[world] element offsets/counts."""
    world = len(send_counts)
    sc = np.asarray(send_counts, dtype=np.uint64)
    rc = np.asarray(recv_counts, dtype=np.uint64)
    so_base = np.asarray(send_offsets[:world], dtype=np.uint64)

    def split(c, j):
        return c * (j + 1) // nchunks - c * j // nchunks

    per_chunk = []
    chunk_start = np.zeros(nchunks, dtype=np.uint64)
    chunk_rows = np.zeros(nchunks, dtype=np.uint64)
    sent_so_far = np.zeros(world, dtype=np.uint64)
    recv_cursor = np.uint64(0)
    for j in range(nchunks):
        sc_j = np.array([split(int(sc[p]), j) for p in range(world)],
                        dtype=np.uint64)
        rc_j = np.array([split(int(rc[p]), j) for p in range(world)],
                        dtype=np.uint64)
        so_j = so_base + sent_so_far
        ro_j = recv_cursor + np.concatenate(
            ([np.uint64(0)], np.cumsum(rc_j[:-1])))
        per_chunk.append((so_j, sc_j, ro_j.astype(np.uint64), rc_j))
        chunk_start[j] = recv_cursor
        chunk_rows[j] = rc_j.sum()
        recv_cursor += chunk_rows[j]
        sent_so_far += sc_j
    return per_chunk, chunk_start, chunk_rows


def repartition_overlapped(comm, key_col, payload_cols, comp_stream,
                           comm_stream, consume=None, nchunks=4):
    """Hash-repartition by key % world with the RCCL exchange OVERLAPPED
    against `consume` (the probe) on the compute stream — the north_star
    requirement the reference meets with its 8-thread Flight do_put pool
    (core.py:324-371).

    Pipeline (all enqueued, the GPU overlaps):
      comp_stream:  partition -> per-column gather -> [record ready]
                    [wait chunk j done] -> consume(chunk j) ...
      comm_stream:  [wait ready] -> alltoallv chunk 0 -> [record done 0]
                    -> alltoallv chunk 1 -> [record done 1] -> ...

    consume(cols, start_row, n_rows, chunk_idx): cols maps the key column
    name '__key__' and each payload name to a DevColumnView of that
    received chunk. Returns (recv_key, recv_payload, chunk_start,
    chunk_rows); both streams are synced and temporaries freed before
    returning (one host sync per step, like `repartition`)."""
    n = key_col.n
    world = comm.world
    offsets, idx = ops.partition_i64(key_col, world, comp_stream, n)
    send_counts = np.diff(offsets).astype(np.uint64)
    recv_counts = comm.exchange_counts(send_counts)
    total_recv = int(recv_counts.sum())

    named = [("__key__", key_col)] + list(payload_cols.items())
    ordered = {}
    for name, col in named:
        ordered[name] = col.gather(idx, n, comp_stream)
    ready = shim.Event()
    ready.record(comp_stream)
    ready.wait(comm_stream)

    recvs = {name: DevColumn(col.dtype, max(1, total_recv))
             for name, col in named}
    for name, col in named:
        recvs[name].n = total_recv
    per_chunk, chunk_start, chunk_rows = plan_chunks(
        offsets, send_counts, recv_counts, nchunks)

    done_evs = [shim.Event() for _ in range(nchunks)]
    for j, (so_j, sc_j, ro_j, rc_j) in enumerate(per_chunk):
        for name, _ in named:
            comm.alltoallv_into(ordered[name], recvs[name],
                                so_j, sc_j, ro_j, rc_j, comm_stream)
        done_evs[j].record(comm_stream)
    if consume is not None:
        for j in range(nchunks):
            if not int(chunk_rows[j]):
                continue
            done_evs[j].wait(comp_stream)
            views = {name: recvs[name].view(int(chunk_start[j]),
                                            int(chunk_rows[j]))
                     for name, _ in named}
            consume(views, int(chunk_start[j]), int(chunk_rows[j]), j)
    comp_stream.sync()
    comm_stream.sync()
    for ev in done_evs:
        ev.destroy()
    ready.destroy()
    for t in ordered.values():
        t.free()
    idx.free()
    recv_key = recvs.pop("__key__")
    return recv_key, recvs, chunk_start, chunk_rows
