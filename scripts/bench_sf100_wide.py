"""Wider TPC-H query shapes at scale on device-generated tables:
Q7 (6-table nation-pair chain), Q12 (shipmode window), Q13
(count-of-counts incl. zero-order customers), Q22 (anti-join + grand
avg). Complements the fused Q1/Q3/Q5/Q6 and the r02 Q4/Q18 runners —
all via the generic operator pipelines in quokka_amd/queries.py, whose
results are parity-anchored (oracle == Acero == GPU) at test sizes in
tests/. Here the inputs come from the device generators (qk_gen_* +
qk_gen_aux for l_shipmode / o_comment_special / c_acctbal).

Run on a GPU box: python scripts/bench_sf100_wide.py [--sf 100]
"""
import argparse
import json
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

NATIONS = ["ALGERIA", "ARGENTINA", "BRAZIL", "CANADA", "EGYPT", "ETHIOPIA",
           "FRANCE", "GERMANY", "INDIA", "INDONESIA", "IRAN", "IRAQ",
           "JAPAN", "JORDAN", "KENYA", "MOROCCO", "MOZAMBIQUE", "PERU",
           "CHINA", "ROMANIA", "SAUDI ARABIA", "VIETNAM", "RUSSIA",
           "UNITED KINGDOM", "UNITED STATES"]


def main():
    from quokka_amd import shim
    from quokka_amd import queries as DQ
    from quokka_amd.shim import DevColumn, c_u64, c_i64
    import ctypes

    ap = argparse.ArgumentParser()
    ap.add_argument("--sf", type=float, default=100.0)
    ap.add_argument("--steps", type=int, default=3)
    args = ap.parse_args()
    shim.init(0)

    n = int(6_000_000 * args.sf)
    n_ord = max(1, n // 4)
    n_cust = max(1, n_ord // 10)
    n_supp = max(1, int(10_000 * args.sf))
    n_parts = max(1, int(200_000 * args.sf))

    def gen_li(cols):
        out = {}
        ptr = {k: None for k in (
            "l_orderkey", "l_suppkey", "l_quantity", "l_extendedprice",
            "l_discount", "l_tax", "l_returnflag", "l_linestatus",
            "l_shipdate", "l_commitdate", "l_receiptdate")}
        dt = {"l_orderkey": np.int64, "l_suppkey": np.int64,
              "l_quantity": np.float64, "l_extendedprice": np.float64,
              "l_discount": np.float64, "l_tax": np.float64,
              "l_returnflag": np.uint8, "l_linestatus": np.uint8,
              "l_shipdate": np.int32, "l_commitdate": np.int32,
              "l_receiptdate": np.int32}
        for k in cols:
            if k == "l_shipmode":
                continue
            out[k] = DevColumn(dt[k], n)
            ptr[k] = out[k].ptr
        shim.call("qk_gen_lineitem", None, c_u64(n), c_u64(0), c_u64(42),
                  c_i64(n_parts), c_i64(n_supp), c_i64(n_ord),
                  ptr["l_orderkey"], ptr["l_suppkey"], ptr["l_quantity"],
                  ptr["l_extendedprice"], ptr["l_discount"], ptr["l_tax"],
                  ptr["l_returnflag"], ptr["l_linestatus"],
                  ptr["l_shipdate"], ptr["l_commitdate"],
                  ptr["l_receiptdate"])
        if "l_shipmode" in cols:
            out["l_shipmode"] = DevColumn(np.uint8, n)
            shim.call("qk_gen_aux", None, c_u64(n), c_u64(0), c_u64(42),
                      c_u64(0x5A1D), 0, c_i64(7), c_i64(0),
                      out["l_shipmode"].ptr, None)
        return out

    def gen_ord(cols):
        out = {}
        ptr = {k: None for k in ("o_orderkey", "o_custkey", "o_orderdate",
                                 "o_shippriority", "o_orderpriority",
                                 "o_totalprice")}
        dt = {"o_orderkey": np.int64, "o_custkey": np.int64,
              "o_orderdate": np.int32, "o_shippriority": np.int32,
              "o_orderpriority": np.uint8, "o_totalprice": np.float64}
        for k in cols:
            if k == "o_comment_special":
                continue
            out[k] = DevColumn(dt[k], n_ord)
            ptr[k] = out[k].ptr
        shim.call("qk_gen_orders", None, c_u64(n_ord), c_u64(0), c_u64(42),
                  c_i64(n_cust), ptr["o_orderkey"], ptr["o_custkey"],
                  ptr["o_orderdate"], ptr["o_shippriority"],
                  ptr["o_orderpriority"], ptr["o_totalprice"],
                  c_i64(n_parts))
        if "o_comment_special" in cols:
            out["o_comment_special"] = DevColumn(np.uint8, n_ord)
            shim.call("qk_gen_aux", None, c_u64(n_ord), c_u64(0), c_u64(42),
                      c_u64(0xC033), 1, c_i64(19000), c_i64(0),
                      out["o_comment_special"].ptr, None)
        return out

    def gen_cust(cols):
        out = {}
        ptr = {k: None for k in ("c_custkey", "c_mktsegment",
                                 "c_nationkey")}
        dt = {"c_custkey": np.int64, "c_mktsegment": np.uint8,
              "c_nationkey": np.int32}
        for k in cols:
            if k == "c_acctbal":
                continue
            out[k] = DevColumn(dt[k], n_cust)
            ptr[k] = out[k].ptr
        shim.call("qk_gen_customer", None, c_u64(n_cust), c_u64(0),
                  c_u64(42), ptr["c_custkey"], ptr["c_mktsegment"],
                  ptr["c_nationkey"])
        if "c_acctbal" in cols:
            out["c_acctbal"] = DevColumn(np.float64, n_cust)
            shim.call("qk_gen_aux", None, c_u64(n_cust), c_u64(0),
                      c_u64(42), c_u64(0xACC7), 2, c_i64(-99999),
                      c_i64(999999), None, out["c_acctbal"].ptr)
        return out

    def gen_supp():
        out = {"s_suppkey": DevColumn(np.int64, n_supp),
               "s_nationkey": DevColumn(np.int32, n_supp)}
        shim.call("qk_gen_supplier", None, c_u64(n_supp), c_u64(0),
                  c_u64(42), out["s_suppkey"].ptr, out["s_nationkey"].ptr)
        return out

    def run(name, make_tables, call):
        tabs = make_tables()
        res = call(tabs)                       # warm (JIT, pools)
        shim.call("qk_stream_sync", None)
        t0 = time.time()
        for _ in range(args.steps):
            res = call(tabs)
        shim.call("qk_stream_sync", None)
        ms = (time.time() - t0) / args.steps * 1e3
        for t in tabs.values():
            for c in t.values():
                c.free()
        sample = dict(list(res.items())[:3]) if isinstance(res, dict) \
            else res
        print(json.dumps({"query": name, "sf": args.sf,
                          "ms_per_query": round(ms, 2),
                          "lineitem_rows": n, "steps": args.steps,
                          "result_sample": repr(sample)[:200]}),
              flush=True)

    run("Q12", lambda: {
        "li": gen_li(["l_orderkey", "l_shipdate", "l_commitdate",
                      "l_receiptdate", "l_shipmode"]),
        "ord": gen_ord(["o_orderkey", "o_orderpriority"])},
        lambda t: DQ.q12(t["li"], t["ord"]))

    run("Q13", lambda: {
        "ord": gen_ord(["o_custkey", "o_comment_special"])},
        lambda t: DQ.q13(t["ord"], n_cust))

    run("Q22", lambda: {
        "cust": gen_cust(["c_custkey", "c_nationkey", "c_acctbal"]),
        "ord": gen_ord(["o_custkey"])},
        lambda t: DQ.q22(t["cust"], t["ord"]))

    run("Q7", lambda: {
        "li": gen_li(["l_orderkey", "l_suppkey", "l_extendedprice",
                      "l_discount", "l_shipdate"]),
        "ord": gen_ord(["o_orderkey", "o_custkey"]),
        "cust": gen_cust(["c_custkey", "c_nationkey"]),
        "supp": gen_supp()},
        lambda t: DQ.q7(t["li"], t["ord"], t["cust"], t["supp"], NATIONS))


if __name__ == "__main__":
    main()
