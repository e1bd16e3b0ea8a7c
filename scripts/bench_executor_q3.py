"""Integration-path TPC-H Q3: host Arrow batches -> the DROP-IN executor
boundary (gpu_partition_fn + GPUBuildProbeJoinExecutor + GPUAggExecutor
+ GPUTopKExecutor, exactly the objects a pyquokka lowering substitutes —
INTEGRATION.md) -> host Arrow result. This is the PCIe-inclusive number
for the plugin path, next to the HBM-resident fused-kernel headline.

Usage (GPU box): python scripts/bench_executor_q3.py [--sf 10]
"""
import argparse
import os
import sys
import time

import numpy as np

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

from oracle import tpch_gen as G, queries as OQ  # noqa: E402


def run_once(d, nbatches):
    import pyarrow as pa
    from quokka_amd import (GPUBuildProbeJoinExecutor, GPUAggExecutor,
                            GPUTopKExecutor, gpu_partition_fn)
    li, od, cu = d["lineitem"], d["orders"], d["customer"]

    # plan (mirrors logical.py:447-506): customer(BUILDING) semi orders,
    # then orders build / lineitem probe, group-by orderkey, top-10
    cust_t = pa.table({"c_custkey": cu["c_custkey"],
                       "c_mktsegment": cu["c_mktsegment"]})
    ord_t = pa.table({k: od[k] for k in ("o_orderkey", "o_custkey",
                                         "o_orderdate",
                                         "o_shippriority")})
    li_t = pa.table({k: li[k] for k in ("l_orderkey", "l_shipdate",
                                        "l_extendedprice",
                                        "l_discount")})
    join1 = GPUBuildProbeJoinExecutor(left_on="o_custkey",
                                      right_on="c_custkey", how="semi")
    join2 = GPUBuildProbeJoinExecutor(left_on="l_orderkey",
                                      right_on="o_orderkey", how="inner")
    agg = GPUAggExecutor(["l_orderkey"], [],
                         "sum(revenue) as revenue")
    topk = GPUTopKExecutor(["revenue"], 10, descending=[True])

    # map side: the partition functions the runtime registers
    seg = G.MKTSEGMENT.index("BUILDING")
    cust_f = cust_t.filter(
        pa.compute.equal(cust_t["c_mktsegment"], seg))
    join1.execute([cust_f.select(["c_custkey"])], 1, 0)
    n_ord = ord_t.num_rows
    step = (n_ord + nbatches - 1) // nbatches
    ord_semi = []
    for lo in range(0, n_ord, step):
        parts = gpu_partition_fn(ord_t.slice(lo, step), 0, 1,
                                 key="o_orderkey",
                                 predicate="o_orderdate < date "
                                           "'1995-03-15'")
        for t in parts.values():
            r = join1.execute([t], 0, 0)
            if r is not None and r.num_rows:
                ord_semi.append(r)
    for t in ord_semi:
        join2.execute([t], 1, 0)
    n_li = li_t.num_rows
    step = (n_li + nbatches - 1) // nbatches
    for lo in range(0, n_li, step):
        parts = gpu_partition_fn(
            li_t.slice(lo, step), 0, 1, key="l_orderkey",
            predicate="l_shipdate > date '1995-03-15'",
            transforms=[("revenue",
                         "l_extendedprice * (1 - l_discount)")])
        for t in parts.values():
            r = join2.execute([t], 0, 0)
            if r is not None and r.num_rows:
                agg.execute([r.select(["l_orderkey", "revenue"])], 0, 0)
    out = agg.done(0)
    topk.execute([out], 0, 0)
    return topk.done(0)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--sf", type=float, default=10.0)
    ap.add_argument("--batches", type=int, default=8)
    args = ap.parse_args()
    from quokka_amd import shim
    shim.init(0)
    d = G.gen_all(args.sf, 42)
    n = len(d["lineitem"]["l_orderkey"])
    res = run_once(d, args.batches)          # warm (JIT, pools)
    t0 = time.time()
    res = run_once(d, args.batches)
    dt = time.time() - t0
    # parity vs the oracle
    _, wtop = OQ.q3(d["lineitem"], d["orders"], d["customer"])
    got_k = np.asarray(res.column("l_orderkey"))
    got_r = np.asarray(res.column("revenue"))
    assert set(got_k) == set(wtop["l_orderkey"]), "top-10 keys differ"
    np.testing.assert_allclose(np.sort(got_r)[::-1],
                               np.sort(wtop["revenue"])[::-1], rtol=1e-9)
    print("executor-boundary Q3 SF%g: %.0f ms (%d lineitem rows, %d "
          "batches/table, host Arrow in -> host Arrow out, "
          "PCIe-inclusive) — top-10 == oracle"
          % (args.sf, dt * 1e3, n, args.batches))


if __name__ == "__main__":
    main()
