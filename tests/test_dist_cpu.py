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
