"""Multi-process (gloo, world_size=2) CPU tests of the distributed-path
bookkeeping: sharding rows across ranks + summing partial aggregates must
equal the single-process oracle (SURVEY.md §8e — Q1's only exchange is the
tiny partial-agg combine; Q3's repartition bucket assignment is key % N)."""
import os

import numpy as np
import pytest
import torch.multiprocessing as mp

from oracle import tpch_gen as G, queries as Q, executors as E


def _rank_main(rank, world, port, result_q):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    li = G.gen_lineitem(0.01, 42)
    n = len(li["l_orderkey"])
    lo, hi = rank * n // world, (rank + 1) * n // world
    shard = {k: v[lo:hi] for k, v in li.items()}
    import torch
    part = torch.from_numpy(Q.q1_partials(shard))
    dist.all_reduce(part, op=dist.ReduceOp.SUM)
    if rank == 0:
        result_q.put(part.numpy())
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_q1_partials_allreduce_equals_full():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29511
    procs = [ctx.Process(target=_rank_main, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    combined = q.get(timeout=100)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    full = Q.q1_partials(G.gen_lineitem(0.01, 42))
    np.testing.assert_allclose(combined, full, rtol=1e-12)


def test_partition_shards_are_disjoint_and_complete():
    """key % N sharding (quokka_runtime.py:222): each row lands on exactly
    one rank; per-rank group-by then concat equals global group-by."""
    rng = np.random.default_rng(3)
    keys = rng.integers(0, 10_000, 50_000).astype(np.int64)
    vals = rng.random(50_000)
    parts = E.partition_int(keys, 8)
    seen = 0
    all_keys, all_sums = [], []
    for p in range(8):
        m = parts == p
        seen += m.sum()
        uk, sums = E.groupby_sum_i64(keys[m], vals[m])
        assert np.all(uk % 8 == p)
        all_keys.append(uk)
        all_sums.append(sums)
    assert seen == len(keys)
    gk = np.concatenate(all_keys)
    gs = np.concatenate(all_sums)
    order = np.argsort(gk)
    uk_ref, sums_ref = E.groupby_sum_i64(keys, vals)
    assert np.array_equal(gk[order], uk_ref)
    np.testing.assert_allclose(gs[order], sums_ref, rtol=1e-12)


def test_repartition_count_bookkeeping():
    """The host-side plan of exchange.repartition (counts/offsets both
    sides) is consistent: what rank r sends to p equals what p receives
    from r, and every row lands exactly once (CPU, no GPU/RCCL)."""
    rng = np.random.default_rng(7)
    world = 4
    per_rank_keys = [rng.integers(0, 1000, rng.integers(50, 200)).astype(np.int64)
                     for _ in range(world)]
    send = np.zeros((world, world), dtype=np.int64)
    for r in range(world):
        parts = E.partition_int(per_rank_keys[r], world)
        for p in range(world):
            send[r, p] = (parts == p).sum()
    recv = send.T
    for p in range(world):
        got = int(recv[p].sum())
        want = sum(int((E.partition_int(k, world) == p).sum())
                   for k in per_rank_keys)
        assert got == want
    assert send.sum() == sum(len(k) for k in per_rank_keys)


def test_top10_merge_from_disjoint_rank_shards():
    """bench.py main_q3 merges per-rank Q3 top-10s by concatenating the
    candidates and re-running _topk; with disjoint per-rank orderkey
    groups (`orderkey % world` sharding) that must equal the global
    top-10 (revenue desc, orderdate asc, orderkey asc). Pure CPU check
    of the merge rule."""
    from quokka_amd.queries import _topk
    rng = np.random.default_rng(11)
    world = 4
    n = 5_000
    keys = rng.permutation(n).astype(np.int64)
    full = {
        "l_orderkey": keys,
        "revenue": np.round(rng.uniform(0, 1e6, n), 2),
        "o_orderdate": rng.integers(9000, 9300, n).astype(np.int32),
        "o_shippriority": np.zeros(n, dtype=np.int32),
    }
    # force some revenue ties across ranks to exercise the tie-break
    full["revenue"][keys % 17 == 0] = 999999.0
    sel = _topk(full, 10)
    want = {c: v[sel] for c, v in full.items()}
    per_rank = []
    for r in range(world):
        m = keys % world == r
        shard = {c: v[m] for c, v in full.items()}
        s = _topk(shard, 10)
        per_rank.append({c: v[s] for c, v in shard.items()})
    cand = {c: np.concatenate([g[c] for g in per_rank]) for c in full}
    msel = _topk(cand, 10)
    got = {c: v[msel] for c, v in cand.items()}
    for c in full:
        assert np.array_equal(got[c], want[c]), c


def test_plan_chunks_conservation_and_layout():
    """exchange.plan_chunks: both sides derive the SAME split from the
    exchanged counts. Check (a) per-peer chunk counts sum back to the
    full counts, (b) send offsets tile each peer's partition range
    exactly, (c) the chunk-major receive layout is contiguous and
    non-overlapping."""
    from quokka_amd.exchange import plan_chunks
    rng = np.random.default_rng(5)
    world, nchunks = 8, 4
    send_counts = rng.integers(0, 10_000, world).astype(np.uint64)
    send_offsets = np.zeros(world + 1, dtype=np.uint64)
    np.cumsum(send_counts, out=send_offsets[1:])
    recv_counts = rng.integers(0, 10_000, world).astype(np.uint64)
    per_chunk, chunk_start, chunk_rows = plan_chunks(
        send_offsets, send_counts, recv_counts, nchunks)

    sc_sum = np.zeros(world, dtype=np.uint64)
    rc_sum = np.zeros(world, dtype=np.uint64)
    covered = []
    for j, (so_j, sc_j, ro_j, rc_j) in enumerate(per_chunk):
        sc_sum += sc_j
        rc_sum += rc_j
        # send ranges tile the peer's partition contiguously
        for p in range(world):
            prior = sum(int(per_chunk[k][1][p]) for k in range(j))
            assert int(so_j[p]) == int(send_offsets[p]) + prior
        # recv ranges: chunk j occupies [chunk_start[j], +chunk_rows[j])
        lo = int(chunk_start[j])
        cur = lo
        for p in range(world):
            assert int(ro_j[p]) == cur
            cur += int(rc_j[p])
        assert cur - lo == int(chunk_rows[j])
        covered.append((lo, cur))
    assert np.array_equal(sc_sum, send_counts)
    assert np.array_equal(rc_sum, recv_counts)
    # chunk recv ranges are contiguous, in order, and cover the total
    assert covered[0][0] == 0
    for (a, b), (c, d) in zip(covered, covered[1:]):
        assert b == c
    assert covered[-1][1] == int(recv_counts.sum())


def test_plan_chunks_numpy_emulation_multiset():
    """Emulate the chunked overlapped exchange with numpy copies across a
    4-rank world: applying each rank's send plan into each peer's
    chunk-major recv layout must deliver every row exactly once, with
    per-rank received multisets equal to the unchunked exchange."""
    from quokka_amd.exchange import plan_chunks
    rng = np.random.default_rng(13)
    world, nchunks = 4, 3
    data = [rng.integers(0, 997, rng.integers(2_000, 4_000)).astype(np.int64)
            for _ in range(world)]
    ordered, offsets, send_counts = [], [], []
    for r in range(world):
        parts = E.partition_int(data[r], world)
        order = np.argsort(parts, kind="stable")
        ordered.append(data[r][order])
        sc = np.bincount(parts, minlength=world).astype(np.uint64)
        off = np.zeros(world + 1, dtype=np.uint64)
        np.cumsum(sc, out=off[1:])
        offsets.append(off)
        send_counts.append(sc)
    recv_counts = [np.array([send_counts[p][r] for p in range(world)],
                            dtype=np.uint64) for r in range(world)]
    recv = [np.full(int(rc.sum()), -1, dtype=np.int64)
            for rc in recv_counts]
    plans = [plan_chunks(offsets[r], send_counts[r], recv_counts[r],
                         nchunks) for r in range(world)]
    for j in range(nchunks):
        for r in range(world):              # r sends chunk j to every peer
            so_j, sc_j, _, _ = plans[r][0][j]
            for p in range(world):
                # receiver p's plan for chunk j names where r's piece lands
                _, _, ro_p, rc_p = plans[p][0][j]
                assert int(rc_p[r]) == int(sc_j[p])
                lo_s = int(so_j[p]); n = int(sc_j[p])
                lo_r = int(ro_p[r])
                recv[p][lo_r:lo_r + n] = ordered[r][lo_s:lo_s + n]
    for r in range(world):
        assert not np.any(recv[r] == -1)
        # multiset equality with the unchunked exchange result
        want = np.concatenate([data[p][E.partition_int(data[p], world) == r]
                               for p in range(world)])
        assert np.array_equal(np.sort(recv[r]), np.sort(want))
        assert np.all(recv[r] % world == r)


def test_plan_chunks_more_chunks_than_rows():
    """nchunks larger than some peers' counts: zero-size chunk pieces
    are planned consistently (conservation + layout hold)."""
    from quokka_amd.exchange import plan_chunks
    world, nchunks = 4, 7
    send_counts = np.array([3, 0, 1, 5], dtype=np.uint64)
    send_offsets = np.zeros(world + 1, dtype=np.uint64)
    np.cumsum(send_counts, out=send_offsets[1:])
    recv_counts = np.array([0, 2, 9, 1], dtype=np.uint64)
    per_chunk, chunk_start, chunk_rows = plan_chunks(
        send_offsets, send_counts, recv_counts, nchunks)
    sc_sum = sum(c[1] for c in per_chunk)
    rc_sum = sum(c[3] for c in per_chunk)
    assert np.array_equal(sc_sum, send_counts)
    assert np.array_equal(rc_sum, recv_counts)
    assert int(chunk_rows.sum()) == int(recv_counts.sum())
    assert int(chunk_start[0]) == 0
