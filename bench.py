"""Benchmark: TPC-H Q1 hot path on MI355X (BASELINE.json configs[1]).

A "step" = one full Q1 pass over the HBM-resident SF100 lineitem columns:
fused filter+group-by partial-agg kernel (qk_q1_agg) + partial combine +
host finalize — i.e. the whole reference pipeline of SURVEY.md §3.4 with
scan IO excluded (inputs resident, `data: synthetic`).

Usage:  python bench.py [--gpus N] [--steps K] [--warmup W] [--sf S]
Multi-rank: launched by torch.distributed.run, one rank per GPU (weak
scaling: each rank owns a full SF-sized shard; the only exchange on Q1's
path is the 384-byte partial-aggregate combine — SURVEY.md §8e / BASELINE
configs[1] is "no shuffle" — done host-side via gloo).

Prints ONE JSON line from rank 0 (driver contract), including:
  roofline      — dominant-kernel HBM roofline measured with HIP events on
                  the launching stream (38 algorithmic B/row)
  cpu_baseline  — the CPU oracle (numpy restatement, oracle/queries.py)
                  timed on this box's host cores over a bounded sample
"""
import argparse
import json
import os
import sys
import time

import numpy as np

ROOT = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, ROOT)

SF100_LINEITEM_ROWS = 600_037_902   # dbgen SF100 cardinality
Q1_BYTES_PER_ROW = 38               # 4*f64 + i32 + 2*u8 read once
HBM_PEAK_GBPS = 8000.0              # MI355X spec (measured achievable ~6290)


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=5)
    p.add_argument("--warmup", type=int, default=2)
    p.add_argument("--sf", type=float, default=100.0,
                   help="scale factor per GPU (rows = SF/100 * 600037902)")
    p.add_argument("--query", choices=["q1", "q3", "q4", "q5", "q6",
                                       "q18"],
                   default="q1")
    p.add_argument("--verify", action="store_true",
                   help="full-size property cross-checks (SURVEY.md §8c): "
                        "independent-kernel row counts, group-count bounds, "
                        "finite aggregates; asserts on failure")
    p.add_argument("--exchange", choices=["auto", "none", "rccl"],
                   default="auto",
                   help="q3 multi-rank repartition (auto: rccl when world>1)")
    p.add_argument("--chunks", type=int, default=4,
                   help="exchange pipeline depth: probe chunk j overlaps "
                        "the RCCL transfer of chunk j+1 (north_star)")
    p.add_argument("--no-overlap", action="store_true",
                   help="A/B: sequential exchange -> probe (round-1 path)")
    p.add_argument("--nt", type=int, default=1,
                   help="non-temporal probe loads (Q3/Q5)")
    p.add_argument("--cpu-sample-rows", type=int, default=12_000_000)
    p.add_argument("--skip-cpu-baseline", action="store_true")
    p.add_argument("--skip-subbench", action="store_true",
                   help="q1 default run: skip the folded q3/q6/e2e legs")
    p.add_argument("--e2e-sf", type=float, default=10.0,
                   help="scale factor of the end-to-end Parquet leg")
    return p.parse_args()


def gen_device_lineitem(shim, n, rank):
    """Device-side synthetic lineitem (Q1 columns) for this rank's shard."""
    from quokka_amd.shim import DevColumn, c_u64, c_i64
    cols = {name: DevColumn(dt, n) for name, dt in [
        ("l_quantity", np.float64), ("l_extendedprice", np.float64),
        ("l_discount", np.float64), ("l_tax", np.float64),
        ("l_returnflag", np.uint8), ("l_linestatus", np.uint8),
        ("l_shipdate", np.int32)]}
    shim.call("qk_gen_lineitem", None, c_u64(n), c_u64(rank * n), c_u64(42),
              c_i64(20_000_000), c_i64(1_000_000), c_i64(150_000_000),
              None, None,
              cols["l_quantity"].ptr, cols["l_extendedprice"].ptr,
              cols["l_discount"].ptr, cols["l_tax"].ptr,
              cols["l_returnflag"].ptr, cols["l_linestatus"].ptr,
              cols["l_shipdate"].ptr, None, None)
    return cols


def cpu_baseline(sample_rows, target_seconds=12.0):
    """Time the CPU oracle restatement (kind='port') on a bounded sample."""
    from oracle import tpch_gen as G, queries as OQ
    sf = sample_rows / 6_000_000
    li = G.gen_lineitem(sf, seed=42)
    n = len(li["l_shipdate"])
    # one calibration pass, then enough passes for ~target_seconds
    t0 = time.time()
    OQ.q1_partials(li)
    per = time.time() - t0
    passes = max(1, min(16, int(target_seconds / max(per, 1e-3))))
    t0 = time.time()
    for _ in range(passes):
        OQ.q1_partials(li)
    dt = time.time() - t0
    return {
        "value": n * passes / dt,
        "unit": "rows/s",
        "cores": 1,
        "kind": "port",
        "sample": "%.1fM-row seeded lineitem x %d passes of the numpy "
                  "oracle (oracle/queries.py:q1_partials), single-threaded"
                  % (n / 1e6, passes),
    }


def gen_device_q3_tables(shim, n, rank, world=1, n_total=None):
    """Lineitem (4 cols) + orders + customer for the fused Q3.

    world == 1: full tables on this rank. world > 1 (exchange mode,
    configs[3]): lineitem and orders are SHARDED by row range across
    ranks over the TOTAL key space (n_total rows overall); customer is
    replicated (broadcast side)."""
    from quokka_amd.shim import DevColumn, c_u64, c_i64
    n_total = n_total or n * world
    n_ord_total = max(1, n_total // 4)
    n_cust = max(1, n_ord_total // 10)
    li = {k: DevColumn(dt, n) for k, dt in [
        ("l_orderkey", np.int64), ("l_shipdate", np.int32),
        ("l_extendedprice", np.float64), ("l_discount", np.float64)]}
    shim.call("qk_gen_lineitem", None, c_u64(n), c_u64(rank * n), c_u64(42),
              c_i64(20_000_000), c_i64(1_000_000), c_i64(n_ord_total),
              li["l_orderkey"].ptr, None, None, li["l_extendedprice"].ptr,
              li["l_discount"].ptr, None, None, None, li["l_shipdate"].ptr,
              None, None)
    n_ord_local = n_ord_total // world if world > 1 else n_ord_total
    ord_off = rank * n_ord_local if world > 1 else 0
    if world > 1 and rank == world - 1:      # last rank takes the remainder
        n_ord_local = n_ord_total - ord_off
    od = {k: DevColumn(dt, n_ord_local) for k, dt in [
        ("o_orderkey", np.int64), ("o_custkey", np.int64),
        ("o_orderdate", np.int32), ("o_shippriority", np.int32)]}
    shim.call("qk_gen_orders", None, c_u64(n_ord_local), c_u64(ord_off),
              c_u64(42), c_i64(n_cust), od["o_orderkey"].ptr,
              od["o_custkey"].ptr, od["o_orderdate"].ptr,
              od["o_shippriority"].ptr, None, None, c_i64(1))
    cu = {"c_custkey": DevColumn(np.int64, n_cust),
          "c_mktsegment": DevColumn(np.uint8, n_cust)}
    shim.call("qk_gen_customer", None, c_u64(n_cust), c_u64(0), c_u64(42),
              cu["c_custkey"].ptr, cu["c_mktsegment"].ptr, None)
    return li, od, cu


def gen_device_q5_tables(shim, n, rank):
    from quokka_amd.shim import DevColumn, c_u64, c_i64
    n_ord, n_supp = max(1, n // 4), max(1, n // 600)
    n_cust = max(1, n_ord // 10)
    li = {k: DevColumn(dt, n) for k, dt in [
        ("l_orderkey", np.int64), ("l_suppkey", np.int64),
        ("l_extendedprice", np.float64), ("l_discount", np.float64)]}
    shim.call("qk_gen_lineitem", None, c_u64(n), c_u64(rank * n), c_u64(42),
              c_i64(20_000_000), c_i64(n_supp), c_i64(n_ord),
              li["l_orderkey"].ptr, li["l_suppkey"].ptr, None,
              li["l_extendedprice"].ptr, li["l_discount"].ptr, None,
              None, None, None, None, None)
    od = {k: DevColumn(dt, n_ord) for k, dt in [
        ("o_orderkey", np.int64), ("o_custkey", np.int64),
        ("o_orderdate", np.int32)]}
    shim.call("qk_gen_orders", None, c_u64(n_ord), c_u64(0), c_u64(42),
              c_i64(n_cust), od["o_orderkey"].ptr, od["o_custkey"].ptr,
              od["o_orderdate"].ptr, None, None, None, c_i64(1))
    cu = {"c_custkey": DevColumn(np.int64, n_cust),
          "c_nationkey": DevColumn(np.int32, n_cust)}
    shim.call("qk_gen_customer", None, c_u64(n_cust), c_u64(0), c_u64(42),
              cu["c_custkey"].ptr, None, cu["c_nationkey"].ptr)
    su = {"s_suppkey": DevColumn(np.int64, n_supp),
          "s_nationkey": DevColumn(np.int32, n_supp)}
    shim.call("qk_gen_supplier", None, c_u64(n_supp), c_u64(0), c_u64(42),
              su["s_suppkey"].ptr, su["s_nationkey"].ptr)
    return li, od, cu, su


def cpu_baseline_q5(sample_rows, target_seconds=12.0):
    from oracle import tpch_gen as G, queries as OQ
    sf = sample_rows / 6_000_000
    d = {"orders": G.gen_orders(sf, 42)}
    d["lineitem"] = G.gen_lineitem(sf, 42, d["orders"])
    d["customer"] = G.gen_customer(sf, 42)
    d["supplier"] = G.gen_supplier(sf, 42)
    nat, reg = G.gen_nation(), G.gen_region()
    n = len(d["lineitem"]["l_orderkey"])
    t0 = time.time()
    OQ.q5(d["lineitem"], d["orders"], d["customer"], d["supplier"], nat, reg)
    per = time.time() - t0
    passes = max(1, min(16, int(target_seconds / max(per, 1e-3))))
    t0 = time.time()
    for _ in range(passes):
        OQ.q5(d["lineitem"], d["orders"], d["customer"], d["supplier"],
              nat, reg)
    dt = time.time() - t0
    return {"value": n * passes / dt, "unit": "rows/s", "cores": 1,
            "kind": "port",
            "sample": "%.1fM-row seeded lineitem (+orders/customer/supplier)"
                      " x %d passes of the numpy oracle "
                      "(oracle/queries.py:q5), single-threaded"
                      % (n / 1e6, passes)}


def cpu_baseline_q3(sample_rows, target_seconds=12.0):
    from oracle import tpch_gen as G, queries as OQ
    sf = sample_rows / 6_000_000
    d = {"orders": G.gen_orders(sf, 42)}
    d["lineitem"] = G.gen_lineitem(sf, 42, d["orders"])
    d["customer"] = G.gen_customer(sf, 42)
    n = len(d["lineitem"]["l_orderkey"])
    t0 = time.time()
    OQ.q3(d["lineitem"], d["orders"], d["customer"])
    per = time.time() - t0
    passes = max(1, min(16, int(target_seconds / max(per, 1e-3))))
    t0 = time.time()
    for _ in range(passes):
        OQ.q3(d["lineitem"], d["orders"], d["customer"])
    dt = time.time() - t0
    return {
        "value": n * passes / dt, "unit": "rows/s", "cores": 1,
        "kind": "port",
        "sample": "%.1fM-row seeded lineitem (+orders/customer) x %d passes "
                  "of the numpy oracle (oracle/queries.py:q3), "
                  "single-threaded" % (n / 1e6, passes),
    }


def read_traffic(sf=None, query=None):
    """Per-launch HBM bytes from the committed rocprofv3 PMC measurement
    (profiles/traffic_<q>.json), or None before one exists. The PMC run
    was taken at the measured SF recorded in the file; at any other
    --sf the figure would be wrong, so return None instead."""
    path = os.path.join(ROOT, "profiles", "traffic_%s.json" % (query or QUERY))
    if os.path.exists(path):
        with open(path) as f:
            d = json.load(f)
        measured_sf = d.get("sf", 100)
        if sf is not None and abs(float(sf) - float(measured_sf)) > 1e-9:
            return None
        return d.get("traffic_bytes_per_launch")
    return None


QUERY = "q1"


def run_e2e(args, shim, DQ):
    """End-to-end leg: Parquet file ON DISK -> host metadata parse +
    file-bytes upload -> GPU decode (qk_pq_* kernels) -> fused Q1 ->
    result. The reference's hot loop includes the scan
    (unordered_readers.py:42-99); this is the number that exposes the
    PCIe-inclusive rate next to the HBM-resident headline.

    The file is WRITTEN untimed (synthetic lineitem, flags as
    dictionary-encoded strings — the realistic shape); each timed pass
    re-reads it from the page cache, so the measured path is
    host RAM -> HBM -> kernels, like the reference's warm scan."""
    import os
    import tempfile
    import pyarrow as pa
    import pyarrow.parquet as pq
    from quokka_amd import parquet_gpu
    from quokka_amd.shim import DevColumn, c_u64

    n = int(round(args.e2e_sf / 100.0 * SF100_LINEITEM_ROWS)) & ~3
    cols = gen_device_lineitem(shim, n, 0)
    host = {k: c.to_numpy() for k, c in cols.items()}
    for c in cols.values():
        c.free()
    RET = np.array(["A", "N", "R"])
    LIN = np.array(["F", "O"])
    t = pa.table({
        "l_quantity": host["l_quantity"],
        "l_extendedprice": host["l_extendedprice"],
        "l_discount": host["l_discount"],
        "l_tax": host["l_tax"],
        "l_shipdate": pa.array(host["l_shipdate"], type=pa.int32()),
        "l_returnflag": pa.array(RET[host["l_returnflag"]]).dictionary_encode(),
        "l_linestatus": pa.array(LIN[host["l_linestatus"]]).dictionary_encode(),
    })
    del host
    # TPC-H columns are non-null: write them as such so the decoder skips
    # the per-page definition-level walk (host-side cost on warm scans)
    t = t.cast(pa.schema([pa.field(f.name, f.type, nullable=False)
                          for f in t.schema]))
    tmpdir = tempfile.mkdtemp(prefix="qk_e2e_")
    path = os.path.join(tmpdir, "lineitem.parquet")
    pq.write_table(t, path, compression="NONE", data_page_version="1.0",
                   use_dictionary=["l_returnflag", "l_linestatus"])
    file_bytes = os.path.getsize(path)
    del t

    def one_pass():
        dec = parquet_gpu.read_table(path)
        # dict flag columns -> canonical u8 codes via a 3-entry device
        # gather (values order in the file is writer-dependent)
        def canon(name, order):
            codes_u32, values = dec[name]
            m = np.zeros(max(1, len(values)), dtype=np.uint8)
            for i, v in enumerate(values):
                m[i] = order.index(v)
            mcol = DevColumn.from_numpy(m)
            out = DevColumn(np.uint8, codes_u32.n)
            shim.call("qk_gather_u8", None, c_u64(codes_u32.n),
                      codes_u32.ptr, mcol.ptr, out.ptr)
            out.n = codes_u32.n
            mcol.free()
            codes_u32.free()
            return out
        dec["l_returnflag"] = canon("l_returnflag", ["A", "N", "R"])
        dec["l_linestatus"] = canon("l_linestatus", ["F", "O"])
        res = DQ.q1(dec)
        for c in dec.values():
            c.free()
        return res

    res = one_pass()                       # warm (page cache + pool)
    passes = 3
    t0 = time.time()
    for _ in range(passes):
        res = one_pass()
    dt = (time.time() - t0) / passes
    os.unlink(path)
    os.rmdir(tmpdir)
    return {
        "value": n / dt,
        "unit": "rows/s",
        "ms_per_pass": dt * 1e3,
        "file_gb_per_s": file_bytes / dt / 1e9,
        "file_bytes": file_bytes,
        "rows": n,
        "sf": args.e2e_sf,
        "q1_result_rows": len(res["count_order"]),
        "path": "parquet on disk (page-cache warm) -> upload -> "
                "qk_pq_* GPU decode -> fused Q1",
    }


def main_q3(args, n, world, rank, dist, shim, DQ, standalone=True):
    """TPC-H Q3 on the fused device path.

    world == 1 (BASELINE.json configs[2]): a step = rebuild customer+orders
    tables + fused probe/agg + extract top-10, all rows local.

    world > 1 or --exchange rccl (configs[3]): STRONG scaling over the
    total --sf: each rank owns 1/world of lineitem+orders rows, and every
    step hash-repartitions both by orderkey % world with RCCL grouped
    send/recv over xGMI (quokka_amd/exchange.py — the reference's Flight
    shuffle, core.py:276-376), rebuilds from the received partition,
    probes, extracts; per-rank groups are disjoint so the global top-10 is
    merged from per-rank top-10s."""
    import ctypes as ct
    from quokka_amd import ops, exchange
    from quokka_amd.shim import DevBuffer, c_u64

    use_exchange = args.exchange == "rccl" or (args.exchange == "auto"
                                               and world > 1)
    overlap = use_exchange and not args.no_overlap
    if use_exchange:
        n_local = (n // world) & ~3
    else:
        n_local = n
    li, od, cu = gen_device_q3_tables(shim, n_local, rank, world,
                                      n_total=n if use_exchange else None)
    stream = shim.Stream()
    comm_stream = shim.Stream() if use_exchange else None
    comm = exchange.Comm(rank, world, dist) if use_exchange else None

    def do_exchange():
        """Repartition lineitem + orders by orderkey % world; returns new
        (li, od) column dicts (received partitions)."""
        rk, rp = exchange.repartition(
            comm, li["l_orderkey"],
            {k: v for k, v in li.items() if k != "l_orderkey"}, stream)
        li2 = {"l_orderkey": rk, **rp}
        ok, op = exchange.repartition(
            comm, od["o_orderkey"],
            {k: v for k, v in od.items() if k != "o_orderkey"}, stream)
        od2 = {"o_orderkey": ok, **op}
        return li2, od2

    def do_step_overlapped(fused_obj):
        """One exchange step with the lineitem RCCL repartition OVERLAPPED
        against the fused probe, chunk by chunk, on comm_stream vs stream
        (exchange.repartition_overlapped — the north_star side-stream
        overlap; replaces the sequential round-1 exchange->probe)."""
        def consume(views, start, nrows, j):
            fused_obj.probe({
                "l_orderkey": views["__key__"],
                "l_shipdate": views["l_shipdate"],
                "l_extendedprice": views["l_extendedprice"],
                "l_discount": views["l_discount"]}, nt=(4 if args.nt == 4 else bool(args.nt)))
        rk, rp, _, _ = exchange.repartition_overlapped(
            comm, li["l_orderkey"],
            {k: v for k, v in li.items() if k != "l_orderkey"},
            stream, comm_stream, consume, nchunks=max(1, args.chunks))
        return {"l_orderkey": rk, **rp}

    if use_exchange:
        li_x, od_x = do_exchange()
        fused = DQ.Q3Fused(od_x, cu, stream)
    else:
        li_x, od_x = li, od
        fused = DQ.Q3Fused(od, cu, stream)
    stream.sync()
    timer = shim.Timer()

    # one-off counts for the algorithmic-byte accounting
    mc = ops._count_buf()
    fused.probe(li_x, mc)
    stream.sync()
    n_match = ops._read_u64(mc)
    mc.free()
    fused.reset_sums()
    idxbuf = shim.DevColumn(np.uint32, max(1, li_x["l_shipdate"].n))
    cntbuf = ops._count_buf()
    shim.call("qk_filter_i32", stream.handle, c_u64(li_x["l_shipdate"].n),
              li_x["l_shipdate"].ptr, 2, ct.c_int32(DQ.Q3_DATE),
              idxbuf.ptr, cntbuf.ptr)
    stream.sync()
    n_pass = ops._read_u64(cntbuf)
    idxbuf.free(); cntbuf.free()

    def step(timed):
        nonlocal li_x, od_x, fused
        if use_exchange:
            for c in list(li_x.values()) + list(od_x.values()):
                c.free()
            # orders (build side) must be fully received before any probe:
            # exchange + rebuild first, then the lineitem repartition
            # overlapped with the probe chunk pipeline
            ok, op = exchange.repartition(
                comm, od["o_orderkey"],
                {k: v for k, v in od.items() if k != "o_orderkey"}, stream)
            od_x = {"o_orderkey": ok, **op}
            fused.free()
            fused = DQ.Q3Fused(od_x, cu, stream)
            if timed:
                timer.start(stream)
            if overlap:
                li_x = do_step_overlapped(fused)
            else:
                rk, rp = exchange.repartition(
                    comm, li["l_orderkey"],
                    {k: v for k, v in li.items() if k != "l_orderkey"},
                    stream)
                li_x = {"l_orderkey": rk, **rp}
                fused.probe(li_x, nt=(4 if args.nt == 4 else bool(args.nt)))
            if timed:
                timer.stop(stream)
        else:
            fused.rebuild()
            if timed:
                timer.start(stream)
            fused.probe(li_x, nt=(4 if args.nt == 4 else bool(args.nt)))
            if timed:
                timer.stop(stream)
        n_groups, top10 = fused.extract_top10(10)
        if dist is not None:
            # per-rank orderkey partitions are disjoint: merge top-10s
            import torch.distributed as _d
            gathered = [None] * world
            _d.all_gather_object(gathered, top10)
            if rank == 0:
                cand = {k: np.concatenate([g[k] for g in gathered])
                        for k in top10}
                sel = DQ._topk(cand, 10)
                top10 = {k: v[sel] for k, v in cand.items()}
        return n_groups, top10, (timer.elapsed_ms() if timed else None)

    for _ in range(args.warmup):
        step(False)
    if dist is not None:
        dist.barrier()
    stream.sync()
    t0 = time.time()
    kernel_ms = []
    n_groups = 0
    for _ in range(args.steps):
        n_groups, top10, kms = step(True)
        kernel_ms.append(kms)
    stream.sync()
    elapsed = time.time() - t0
    if dist is not None:
        import torch
        t = torch.tensor([elapsed], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t[0])
        dist.barrier()

    if args.verify:
        # full-size property checks: group count bounded by build rows and
        # by joined rows; joined rows bounded by ship-passing rows; the
        # ship-pass count comes from the standalone filter kernel
        assert 0 < n_groups <= fused.n_build, (n_groups, fused.n_build)
        assert n_groups <= n_match <= n_pass, (n_groups, n_match, n_pass)
        assert np.isfinite(top10["revenue"]).all() and             (top10["revenue"] > 0).all()
        # revenues sorted desc
        assert np.all(np.diff(top10["revenue"]) <= 0)
        if rank == 0:
            print("# verify ok: q3 groups=%d <= matches=%d <= ship_pass=%d"
                  % (n_groups, n_match, n_pass), flush=True)

    if rank == 0:
        # probe-kernel algorithmic bytes: 12 B/row (orderkey+shipdate) every
        # row + 12 B bucket read per ship-passing row + 16 B (price+disc)
        # per matched row
        alg_bytes = 12 * li_x["l_shipdate"].n + 12 * n_pass + 16 * n_match
        avg_kernel_s = float(np.mean(kernel_ms)) / 1e3
        achieved_gbps = alg_bytes / avg_kernel_s / 1e9
        total_rows = (n_local * world if not use_exchange else n_local * world)
        out = {
            "metric": "rows/s",
            "value": total_rows * args.steps / elapsed,
            "unit": "rows/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1e3,
            "higher_is_better": True,
            "scaling": "strong" if use_exchange else "weak",
            "vs_baseline": None,
            "dtype": "f64",
            "data": "synthetic",
            "config": {
                "workload": "TPC-H SF%g Q3 (3-way hash join + group-by%s), "
                            "%d lineitem rows/GPU resident in HBM "
                            "(BASELINE.json configs[%d])"
                            % (args.sf,
                               ", RCCL all-to-all repartition" if use_exchange
                               else "", n_local, 3 if use_exchange else 2),
                "sf_per_gpu": args.sf if not use_exchange else args.sf / world,
                "rows_per_gpu": n_local,
                "query": "Q3",
                "exchange": "rccl" if use_exchange else "none",
                "overlap": (bool(overlap) if use_exchange else None),
                "exchange_chunks": (max(1, args.chunks) if overlap
                                    else None),
                "n_groups": int(n_groups),
                "orders_build_rows": int(fused.n_build),
                "lineitem_ship_pass": int(n_pass),
                "joined_rows": int(n_match),
            },
            "roofline": {
                "bound": "hbm",
                "achieved": achieved_gbps,
                "peak": HBM_PEAK_GBPS,
                "unit": "GB/s",
                "frac": achieved_gbps / HBM_PEAK_GBPS,
                "traffic": read_traffic(args.sf, "q3"),
                # in exchange mode the HIP-event window covers the
                # OVERLAPPED exchange+probe pipeline, not the bare probe
                # kernel: `achieved` is then a lower bound on kernel rate
                "window": ("exchange+probe overlapped" if use_exchange and
                           overlap else
                           "exchange+probe sequential" if use_exchange
                           else "probe kernel"),
            },
            "cpu_baseline": (None if args.skip_cpu_baseline or world > 1
                             else cpu_baseline_q3(args.cpu_sample_rows)),
        }
        if standalone:
            print(json.dumps(out))
    else:
        out = None
    timer.destroy()
    fused.free()
    # non-exchange mode aliases li_x/od_x to the base tables (free() is
    # idempotent, so the union below is safe either way)
    for c in (list(li.values()) + list(od.values()) + list(cu.values()) +
              list(li_x.values()) + list(od_x.values())):
        c.free()
    if comm_stream is not None:
        comm_stream.destroy()
    if comm is not None:
        comm.destroy()
    stream.destroy()
    return out


def main_q5(args, n, world, rank, dist, shim, DQ, standalone=True):
    """TPC-H Q5 on the fused device path (BASELINE.json configs[4] query
    shape; single-node). A step = rebuild the three key->nation tables +
    fused probe + per-nation result; weak scaling like Q3."""
    from quokka_amd import ops

    li, od, cu, su = gen_device_q5_tables(shim, n, rank)
    stream = shim.Stream()
    fused = DQ.Q5Fused(od, cu, su, stream)
    stream.sync()
    timer = shim.Timer()

    mc = ops._count_buf()
    fused.probe(li, mc)
    stream.sync()
    n_match = ops._read_u64(mc)
    mc.free()
    fused.reset_sums()

    def step(timed):
        fused.rebuild()
        if timed:
            timer.start(stream)
        fused.probe(li, nt=(4 if args.nt == 4 else bool(args.nt)))
        if timed:
            timer.stop(stream)
        res = fused.result()
        if dist is not None:
            import torch
            import torch.distributed as _d
            v = torch.tensor([r for _, r in res], dtype=torch.float64)
            _d.all_reduce(v, op=_d.ReduceOp.SUM)
            res = sorted(zip([nm for nm, _ in res], v.tolist()),
                         key=lambda t: -t[1])
        return res, (timer.elapsed_ms() if timed else None)

    for _ in range(args.warmup):
        step(False)
    if dist is not None:
        dist.barrier()
    stream.sync()
    t0 = time.time()
    kernel_ms = []
    res = None
    for _ in range(args.steps):
        res, kms = step(True)
        kernel_ms.append(kms)
    stream.sync()
    elapsed = time.time() - t0
    if dist is not None:
        import torch
        t = torch.tensor([elapsed], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t[0])
        dist.barrier()

    if args.verify:
        revs = np.array([r for _, r in res])
        assert np.isfinite(revs).all() and (revs >= 0).all()
        assert np.all(np.diff(revs) <= 0)           # sorted desc
        assert 0 < fused.n_build and n_match <= n   # sanity bounds
        if rank == 0:
            print("# verify ok: q5 matches=%d, 5 ASIA nations, revenues "
                  "sorted" % n_match, flush=True)

    if rank == 0:
        # probe algorithmic bytes: 16 B keys (orderkey+suppkey) per row +
        # one 12 B orders bucket per row + 12 B supplier bucket + 16 B
        # price/disc per matched row (matches dominate the second probe)
        alg_bytes = 28 * n + 28 * n_match
        avg_kernel_s = float(np.mean(kernel_ms)) / 1e3
        achieved_gbps = alg_bytes / avg_kernel_s / 1e9
        out = {
            "metric": "rows/s",
            "value": n * world * args.steps / elapsed,
            "unit": "rows/s",
            "n_gpus": world, "steps": args.steps, "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1e3,
            "higher_is_better": True, "scaling": "weak",
            "vs_baseline": None, "dtype": "f64", "data": "synthetic",
            "config": {
                "workload": "TPC-H SF%g Q5 (6-table join chain), %d "
                            "lineitem rows/GPU resident in HBM "
                            "(BASELINE.json configs[4] query on one node)"
                            % (args.sf, n),
                "sf_per_gpu": args.sf, "rows_per_gpu": n, "query": "Q5",
                "orders_build_rows": int(fused.n_build),
                "joined_rows": int(n_match),
                "revenue_top": res[0] if res else None,
            },
            "roofline": {
                "bound": "hbm", "achieved": achieved_gbps,
                "peak": HBM_PEAK_GBPS, "unit": "GB/s",
                "frac": achieved_gbps / HBM_PEAK_GBPS,
                "traffic": read_traffic(args.sf, "q5"),
            },
            "cpu_baseline": (None if args.skip_cpu_baseline or world > 1
                             else cpu_baseline_q5(args.cpu_sample_rows)),
        }
        if standalone:
            print(json.dumps(out))
    else:
        out = None
    timer.destroy()
    fused.free()
    for c in (list(li.values()) + list(od.values()) + list(cu.values()) +
              list(su.values())):
        c.free()
    stream.destroy()
    return out


def main_q6(args, n, world, rank, dist, shim, DQ, cols=None,
            standalone=True):
    """TPC-H Q6 entirely through the hiprtc JIT: the predicate AND the
    aggregate are runtime-compiled from the reference's SQL strings
    (tpch_ref.py:171-183) — no hand-written kernel on this path."""
    from quokka_amd import jit, ops
    import numpy as _np

    own_cols = cols is None
    if own_cols:
        cols = gen_device_lineitem(shim, n, rank)
    schema = {k: v.dtype for k, v in cols.items()}
    agg = jit.JitAggregate(
        schema, group_keys=[],
        aggs=["sum(l_extendedprice * l_discount) as revenue",
              "count(*) as rows_passed"],
        predicate="l_shipdate >= date '1994-01-01' and l_shipdate < "
                  "date '1994-01-01' + interval '1' year and l_discount "
                  "between 0.06 - 0.01 and 0.06 + 0.01 and "
                  "l_quantity < 24")
    stream = shim.Stream()
    timer = shim.Timer()
    acc = agg.make_acc()

    def step(timed):
        shim.call("qk_dmemset", acc.ptr, 0,
                  shim.c_u64(agg.ngroups * agg.naggs * 8))
        if timed:
            timer.start(stream)
        agg.run(cols, acc, stream)
        if timed:
            timer.stop(stream)
        stream.sync()
        return agg.read(acc), (timer.elapsed_ms() if timed else None)

    for _ in range(args.warmup):
        step(False)
    stream.sync()
    t0 = time.time()
    kernel_ms = []
    res = None
    for _ in range(args.steps):
        res, kms = step(True)
        kernel_ms.append(kms)
    stream.sync()
    elapsed = time.time() - t0

    if rank == 0:
        # Q6 reads 28 B/row (date + qty + price + disc), each once
        avg_kernel_s = float(_np.mean(kernel_ms)) / 1e3
        achieved = 28 * n / avg_kernel_s / 1e9
        if args.verify:
            assert res[0, 1] > 0 and _np.isfinite(res[0, 0])
            print("# verify ok: q6 revenue=%.2f over %d rows"
                  % (res[0, 0], int(res[0, 1])), flush=True)
        out = {
            "metric": "rows/s", "value": n * args.steps / elapsed,
            "unit": "rows/s", "n_gpus": world, "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1e3,
            "higher_is_better": True, "scaling": "weak",
            "vs_baseline": None, "dtype": "f64", "data": "synthetic",
            "config": {"workload": "TPC-H SF%g Q6 via the hiprtc-JIT "
                                   "fused scan (predicate+aggregate "
                                   "runtime-compiled), %d rows/GPU"
                                   % (args.sf, n),
                       "sf_per_gpu": args.sf, "rows_per_gpu": n,
                       "query": "Q6", "jit": True},
            "roofline": {"bound": "hbm", "achieved": achieved,
                         "peak": HBM_PEAK_GBPS, "unit": "GB/s",
                         "frac": achieved / HBM_PEAK_GBPS,
                         "traffic": None},
            "cpu_baseline": None,
        }
        if standalone:
            print(json.dumps(out))
    else:
        out = None
    acc.free()
    agg.free()
    if own_cols:
        for c in cols.values():
            c.free()
    timer.destroy()
    stream.destroy()
    return out


def main_q4(args, n, world, rank, dist, shim, DQ, standalone=True):
    """TPC-H Q4 at scale on the composed generic-operator pipeline
    (queries.q4: JIT col-vs-col filter -> dup-key build -> semi probe ->
    JIT grouped count). A step = the whole query; JIT programs are
    plan-time (cached). Weak scaling like Q1."""
    from quokka_amd.shim import DevColumn, c_u64, c_i64
    n_ord = max(1, n // 4)
    li = {k: DevColumn(dt, n) for k, dt in [
        ("l_orderkey", np.int64), ("l_commitdate", np.int32),
        ("l_receiptdate", np.int32)]}
    shim.call("qk_gen_lineitem", None, c_u64(n), c_u64(rank * n), c_u64(42),
              c_i64(20_000_000), c_i64(1_000_000), c_i64(n_ord),
              li["l_orderkey"].ptr, None, None, None, None, None,
              None, None, None, li["l_commitdate"].ptr,
              li["l_receiptdate"].ptr)
    od = {k: DevColumn(dt, n_ord) for k, dt in [
        ("o_orderkey", np.int64), ("o_orderdate", np.int32),
        ("o_orderpriority", np.uint8)]}
    shim.call("qk_gen_orders", None, c_u64(n_ord), c_u64(0), c_u64(42),
              c_i64(max(1, n_ord // 10)), od["o_orderkey"].ptr, None,
              od["o_orderdate"].ptr, None, od["o_orderpriority"].ptr,
              None, c_i64(1))
    res = DQ.q4(li, od)                 # warm (JIT compile, pool)
    t0 = time.time()
    for _ in range(args.steps):
        res = DQ.q4(li, od)
    elapsed = time.time() - t0
    out = None
    if rank == 0:
        out = {
            "metric": "rows/s", "value": n * args.steps / elapsed,
            "unit": "rows/s", "n_gpus": world, "steps": args.steps,
            "warmup": 1, "ms_per_step": elapsed / args.steps * 1e3,
            "higher_is_better": True, "scaling": "weak",
            "vs_baseline": None, "dtype": "f64", "data": "synthetic",
            "config": {"workload": "TPC-H SF%g Q4 (EXISTS semi-join + "
                                   "grouped count) via the generic "
                                   "operator pipeline, %d lineitem "
                                   "rows/GPU" % (args.sf, n),
                       "sf_per_gpu": args.sf, "rows_per_gpu": n,
                       "query": "Q4", "result": res},
        }
        if standalone:
            print(json.dumps(out))
    for c in list(li.values()) + list(od.values()):
        c.free()
    return out


def main_q18(args, n, world, rank, dist, shim, DQ, standalone=True):
    """TPC-H Q18 at scale: 600M-row group-by into ~n/4 groups on the
    growing device table, HAVING evaluated on device
    (qk_groupby_extract_gt), qualifying orders attached by device probe.
    The high-cardinality group-by benchmark."""
    from quokka_amd.shim import DevColumn, c_u64, c_i64
    n_ord = max(1, n // 4)
    li = {k: DevColumn(dt, n) for k, dt in [
        ("l_orderkey", np.int64), ("l_quantity", np.float64)]}
    shim.call("qk_gen_lineitem", None, c_u64(n), c_u64(rank * n), c_u64(42),
              c_i64(20_000_000), c_i64(1_000_000), c_i64(n_ord),
              li["l_orderkey"].ptr, None, li["l_quantity"].ptr, None,
              None, None, None, None, None, None, None)
    od = {k: DevColumn(dt, n_ord) for k, dt in [
        ("o_orderkey", np.int64), ("o_custkey", np.int64),
        ("o_orderdate", np.int32), ("o_totalprice", np.float64)]}
    shim.call("qk_gen_orders", None, c_u64(n_ord), c_u64(0), c_u64(42),
              c_i64(max(1, n_ord // 10)), od["o_orderkey"].ptr,
              od["o_custkey"].ptr, od["o_orderdate"].ptr, None, None,
              od["o_totalprice"].ptr, c_i64(20_000_000))
    # the device generator emits exactly 4 lines/order (max sum 200), so
    # the reference's 300 threshold would qualify nothing; bench at 150
    # (~5% of orders) to exercise the HAVING path at scale
    res = DQ.q18(li, od, threshold=150.0, n_groups_hint=n_ord)   # warm
    t0 = time.time()
    for _ in range(args.steps):
        res = DQ.q18(li, od, threshold=150.0, n_groups_hint=n_ord)
    elapsed = time.time() - t0
    out = None
    if rank == 0:
        out = {
            "metric": "rows/s", "value": n * args.steps / elapsed,
            "unit": "rows/s", "n_gpus": world, "steps": args.steps,
            "warmup": 1, "ms_per_step": elapsed / args.steps * 1e3,
            "higher_is_better": True, "scaling": "weak",
            "vs_baseline": None, "dtype": "f64", "data": "synthetic",
            "config": {"workload": "TPC-H SF%g Q18 (HAVING over a ~%dM-"
                                   "group device group-by), %d lineitem "
                                   "rows/GPU"
                                   % (args.sf, n_ord // 1_000_000, n),
                       "sf_per_gpu": args.sf, "rows_per_gpu": n,
                       "query": "Q18", "having_threshold": 150.0,
                       "qualifying_orders": len(res["o_orderkey"])},
        }
        if standalone:
            print(json.dumps(out))
    for c in list(li.values()) + list(od.values()):
        c.free()
    return out


def main():
    args = parse_args()
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    dist = None
    if world > 1:
        import torch.distributed as dist_mod
        dist = dist_mod
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        dist.init_process_group("gloo", rank=rank, world_size=world)

    from quokka_amd import shim, ops, queries as DQ
    # one rank per GPU; oversubscribe gracefully when ranks > devices
    # (lets the torchrun path be smoke-tested on a 1-GPU box)
    ndev = shim.device_count()
    shim.init(local_rank % max(1, ndev))

    global QUERY
    QUERY = args.query
    n = int(round(args.sf / 100.0 * SF100_LINEITEM_ROWS))
    n &= ~3  # multiple of 4 -> vectorized Q1 path, 4 lines/order for Q3
    if args.query in ("q3", "q4", "q5", "q6", "q18"):
        fn = {"q3": main_q3, "q4": main_q4, "q5": main_q5,
              "q6": main_q6, "q18": main_q18}[args.query]
        fn(args, n, world, rank, dist, shim, DQ)
        if dist is not None:
            dist.destroy_process_group()
        return
    cols = gen_device_lineitem(shim, n, rank)
    stream = shim.Stream()
    timer = shim.Timer()
    acc = None

    torch = None
    part_t = None
    if dist is not None:
        import torch  # only needed for the gloo combine; slow first import
        part_t = torch.zeros(48, dtype=torch.float64)

    def step(timed):
        nonlocal acc
        from quokka_amd.shim import DevBuffer, c_u64
        if acc is None:
            acc = DevBuffer(48 * 8)
        shim.call("qk_dmemset", acc.ptr, 0, c_u64(48 * 8))
        if timed:
            timer.start(stream)
        DQ.q1_partials_device(cols, stream=stream, acc=acc)
        if timed:
            timer.stop(stream)
        stream.sync()
        p = ops.q1_read_partials(acc)
        if dist is not None:
            part_t[:36] = torch.from_numpy(p.reshape(-1))
            dist.all_reduce(part_t, op=dist.ReduceOp.SUM)
            p = part_t[:36].numpy().reshape(6, 6)
        return DQ.q1_finalize(p), (timer.elapsed_ms() if timed else None)

    for _ in range(args.warmup):
        step(False)
    if dist is not None:
        dist.barrier()
    stream.sync()
    t0 = time.time()
    kernel_ms = []
    result = None
    for _ in range(args.steps):
        result, kms = step(True)
        kernel_ms.append(kms)
    stream.sync()
    elapsed = time.time() - t0
    if dist is not None:
        t = torch.tensor([elapsed], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t[0])
        dist.barrier()

    if args.verify:
        # independent-path cross-check at FULL size: the fused kernel's
        # total group count must equal the standalone filter kernel's
        # count of shipdate <= cutoff rows (different kernel, same data)
        import ctypes as _ct
        from quokka_amd.shim import DevColumn, c_u64
        p = ops.q1_read_partials(acc) if acc is not None else None
        idxbuf = DevColumn(np.uint32, n)
        cntbuf = ops._count_buf()
        shim.call("qk_filter_i32", stream.handle, c_u64(n),
                  cols["l_shipdate"].ptr, 1, _ct.c_int32(DQ.Q1_CUTOFF),
                  idxbuf.ptr, cntbuf.ptr)
        stream.sync()
        n_pass_indep = ops._read_u64(cntbuf)
        idxbuf.free(); cntbuf.free()
        if world == 1:
            got = int(result["count_order"].sum()) if result else 0
            assert got == n_pass_indep, (got, n_pass_indep)
        assert all(np.isfinite(result[c]).all() for c in
                   ("sum_qty", "sum_charge", "avg_disc")), "non-finite aggs"
        if rank == 0:
            print("# verify ok: q1 group-count sum == independent filter "
                  "count (%d)" % n_pass_indep, flush=True)

    # folded sub-measurements (driver-timed evidence for the other graded
    # configs in the SAME record — VERDICT r01 item 3): Q6 reuses the
    # resident Q1 columns; Q3 (and the RCCL exchange at world>1) and the
    # end-to-end Parquet leg run after the Q1 columns are freed.
    subs = {}
    if args.query == "q1" and not args.skip_subbench:
        import argparse as _ap
        sub = _ap.Namespace(**vars(args))
        sub.steps = max(3, min(10, args.steps))
        sub.warmup = max(1, min(3, args.warmup))
        sub.cpu_sample_rows = min(args.cpu_sample_rows, 6_000_000)
        r = main_q6(sub, n, world, rank, dist, shim, DQ, cols=cols,
                    standalone=False)
        if r:
            subs["q6"] = r
        for c in cols.values():
            c.free()
        r = main_q3(sub, n, world, rank, dist, shim, DQ, standalone=False)
        if r:
            subs["q3"] = r
        if world == 1:
            try:
                subs["e2e"] = run_e2e(sub, shim, DQ)
            except Exception as e:  # the headline line must still print
                subs["e2e"] = {"error": "%s: %s" % (type(e).__name__, e)}

    if rank == 0:
        total_rows = n * world * args.steps
        avg_kernel_s = float(np.mean(kernel_ms)) / 1e3
        achieved_gbps = n * Q1_BYTES_PER_ROW / avg_kernel_s / 1e9
        traffic = read_traffic(args.sf)
        out = {
            "metric": "rows/s",
            "value": total_rows / elapsed,
            "unit": "rows/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1e3,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "f64",
            "data": "synthetic",
            "config": {
                "workload": "TPC-H SF%g Q1, fused filter+group-by, "
                            "%d lineitem rows/GPU resident in HBM "
                            "(BASELINE.json configs[1])" % (args.sf, n),
                "sf_per_gpu": args.sf,
                "rows_per_gpu": n,
                "query": "Q1",
            },
            "roofline": {
                "bound": "hbm",
                "achieved": achieved_gbps,
                "peak": HBM_PEAK_GBPS,
                "unit": "GB/s",
                "frac": achieved_gbps / HBM_PEAK_GBPS,
                "traffic": traffic,
            },
            "cpu_baseline": (None if args.skip_cpu_baseline or world > 1
                             else cpu_baseline(args.cpu_sample_rows)),
            "q1_result_rows": len(result["count_order"]) if result else 0,
        }
        # folded legs: Q6 (JIT scan), Q3 (join+group-by, the north_star
        # core; RCCL-exchanged and overlapped at world>1), e2e Parquet —
        # each with its own HIP-event roofline, same record
        out.update(subs)
        print(json.dumps(out))

    timer.destroy()
    stream.destroy()
    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
