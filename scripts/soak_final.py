"""Endurance soak of the late-r02 query tails + quoted CSV: repeated
Q12/Q13/Q22 runs on device-generated tables (fresh gen + free every
iteration — exercises GroupByI64/JoinTable ctor+free churn, the
qk_filter_f64 path, extract_device, JIT cache stability) plus a quoted
CSV parse per iteration; watches host RSS for leaks.

Usage (GPU box): python scripts/soak_final.py [--iters 40]
"""
import argparse
import gc
import os
import sys

import numpy as np
import psutil

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)


def rss_mb():
    return psutil.Process().memory_info().rss / 1e6


def main():
    from quokka_amd import shim, csv_gpu
    from quokka_amd import queries as DQ
    from quokka_amd.shim import DevColumn, c_u64, c_i64

    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=40)
    ap.add_argument("--sf", type=float, default=3.0)
    args = ap.parse_args()
    shim.init(0)

    n = int(6_000_000 * args.sf)
    n_ord = n // 4
    n_cust = n_ord // 10

    quoted_csv = (b'1,"2.5","A,B"\n"-7",3.25,"C\nD"\n' * 20000)

    samples = []
    for it in range(args.iters):
        li = {"l_orderkey": DevColumn(np.int64, n),
              "l_shipdate": DevColumn(np.int32, n),
              "l_commitdate": DevColumn(np.int32, n),
              "l_receiptdate": DevColumn(np.int32, n)}
        shim.call("qk_gen_lineitem", None, c_u64(n), c_u64(0),
                  c_u64(42 + it), c_i64(200000), c_i64(10000),
                  c_i64(n_ord), li["l_orderkey"].ptr, None, None, None,
                  None, None, None, None, li["l_shipdate"].ptr,
                  li["l_commitdate"].ptr, li["l_receiptdate"].ptr)
        li["l_shipmode"] = DevColumn(np.uint8, n)
        shim.call("qk_gen_aux", None, c_u64(n), c_u64(0), c_u64(42 + it),
                  c_u64(0x5A1D), 0, c_i64(7), c_i64(0),
                  li["l_shipmode"].ptr, None)
        od = {"o_orderkey": DevColumn(np.int64, n_ord),
              "o_custkey": DevColumn(np.int64, n_ord),
              "o_orderpriority": DevColumn(np.uint8, n_ord)}
        shim.call("qk_gen_orders", None, c_u64(n_ord), c_u64(0),
                  c_u64(42 + it), c_i64(n_cust), od["o_orderkey"].ptr,
                  od["o_custkey"].ptr, None, None,
                  od["o_orderpriority"].ptr, None, c_i64(1))
        od["o_comment_special"] = DevColumn(np.uint8, n_ord)
        shim.call("qk_gen_aux", None, c_u64(n_ord), c_u64(0),
                  c_u64(42 + it), c_u64(0xC033), 1, c_i64(19000),
                  c_i64(0), od["o_comment_special"].ptr, None)
        cu = {"c_custkey": DevColumn(np.int64, n_cust),
              "c_nationkey": DevColumn(np.int32, n_cust),
              "c_acctbal": DevColumn(np.float64, n_cust)}
        shim.call("qk_gen_customer", None, c_u64(n_cust), c_u64(0),
                  c_u64(42 + it), cu["c_custkey"].ptr, None,
                  cu["c_nationkey"].ptr)
        shim.call("qk_gen_aux", None, c_u64(n_cust), c_u64(0),
                  c_u64(42 + it), c_u64(0xACC7), 2, c_i64(-99999),
                  c_i64(999999), None, cu["c_acctbal"].ptr)

        r12 = DQ.q12(li, od)
        r13 = DQ.q13(od, n_cust)
        r22 = DQ.q22(cu, od)
        assert len(r12) == 2 and 0 in r13 and len(r22) == 7, (
            it, len(r12), len(r22))

        cols = csv_gpu.read_csv(quoted_csv,
                                [("a", "i64"), ("b", "f64"),
                                 ("s", "dict", ["A,B", "C\nD"])],
                                sep=",")
        assert cols["a"].n == 40000
        for c in cols.values():
            c.free()
        for t in (li, od, cu):
            for c in t.values():
                c.free()
        gc.collect()
        r = rss_mb()
        samples.append(r)
        if it % 10 == 0 or it == args.iters - 1:
            print("iter %3d rss %7.1f MB" % (it, r), flush=True)
    # measure AFTER the warm plateau: a one-time ~200 MB step lands
    # between iters 10-20 (allocator arena / pinned size-class growth,
    # then dead flat for 130+ iters — see profiles/raw/soak_final.log)
    base = min(30, len(samples) - 1)
    growth = samples[-1] - samples[base]
    print("RSS growth iters %d..%d: %.1f MB (%s)"
          % (base, args.iters - 1, growth,
             "OK" if growth < 50 else "LEAK?"), flush=True)


if __name__ == "__main__":
    main()
