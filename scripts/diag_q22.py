"""Per-step timing of q22 on device-generated inputs (find the ~230 ms)."""
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def t(label, fn, sync):
    t0 = time.time()
    r = fn()
    sync()
    print("%-28s %7.1f ms" % (label, (time.time() - t0) * 1e3), flush=True)
    return r


def main():
    from quokka_amd import shim, ops, jit
    from quokka_amd import queries as DQ
    from quokka_amd.shim import DevColumn, c_u64, c_i64
    import ctypes
    shim.init(0)
    sf = float(sys.argv[1]) if len(sys.argv) > 1 else 30.0
    n_ord = int(1_500_000 * sf)
    n_cust = n_ord // 10

    ocust = DevColumn(np.int64, n_ord)
    shim.call("qk_gen_orders", None, c_u64(n_ord), c_u64(0), c_u64(42),
              c_i64(n_cust), None, ocust.ptr, None, None, None, None,
              c_i64(1))
    ck = DevColumn(np.int64, n_cust)
    cnk = DevColumn(np.int32, n_cust)
    shim.call("qk_gen_customer", None, c_u64(n_cust), c_u64(0), c_u64(42),
              ck.ptr, None, cnk.ptr)
    bal = DevColumn(np.float64, n_cust)
    shim.call("qk_gen_aux", None, c_u64(n_cust), c_u64(0), c_u64(42),
              c_u64(0xACC7), 2, c_i64(-99999), c_i64(999999), None,
              bal.ptr)
    cust = {"c_custkey": ck, "c_nationkey": cnk, "c_acctbal": bal}
    ords = {"o_custkey": ocust}
    sync = lambda: shim.call("qk_stream_sync", None)
    sync()

    # warm then timed whole query
    DQ.q22(cust, ords)
    t("q22 whole (warm)", lambda: DQ.q22(cust, ords), sync)

    # step by step (mirrors q22's body)
    codes = [13, 31, 23, 29, 30, 18, 17]
    nats = [c - 10 for c in codes]
    in_list = " or ".join("c_nationkey = %d" % k for k in nats)
    csch = {"c_nationkey": np.dtype(np.int32),
            "c_acctbal": np.dtype(np.float64)}
    avg_agg = jit.JitAggregate(csch, [], ["SUM(c_acctbal) as s",
                                          "COUNT(*) as n"],
                               predicate="c_acctbal > 0 and (%s)" % in_list)
    acc = avg_agg.make_acc()
    t("grand avg agg", lambda: avg_agg.run(cust, acc, None), sync)
    s, npos = avg_agg.read(acc)[0]
    avg = s / npos
    acc.free()

    ones = DevColumn(np.float64, n_ord)
    shim.call("qk_fill_f64", None, ones.ptr, ctypes.c_double(1.0),
              c_u64(n_ord))
    gbd = ops.GroupByI64(expected_groups=max(1024, n_cust), nvals=1)
    t("dedup groupby 45M", lambda: gbd.update(ocust, [ones], n_ord), sync)
    r = t("extract_device", lambda: gbd.extract_device(), sync)
    dkeys, dsums, dk, _cap = r
    dsums.free()
    gbd.free()
    print("distinct custkeys:", dk)
    otab = ops.JoinTable(max(16, dk))
    t("build distinct", lambda: otab.build(dkeys, dk), sync)
    r = t("anti probe", lambda: otab.probe(cust["c_custkey"], mode=2),
          sync)
    apx, _, na = r
    print("no-order customers:", na)
    fin = {"c_nationkey": t("gather nk",
                            lambda: cust["c_nationkey"].gather(apx, na),
                            sync),
           "c_acctbal": cust["c_acctbal"].gather(apx, na)}
    sync()
    fagg = jit.JitAggregate(
        {"c_nationkey": np.dtype(np.int32),
         "c_acctbal": np.dtype(np.float64)}, [("c_nationkey", 25)],
        ["COUNT(*) as n", "SUM(c_acctbal) as s"],
        predicate="c_acctbal > %s and (%s)" % (repr(float(avg)), in_list))
    acc2 = fagg.make_acc()
    t("final grouped agg", lambda: fagg.run(fin, acc2, None), sync)
    t("read", lambda: fagg.read(acc2), sync)


def profile_whole():
    import cProfile
    import pstats
    from quokka_amd import shim
    from quokka_amd import queries as DQ
    from quokka_amd.shim import DevColumn, c_u64, c_i64
    shim.init(0)
    sf = 30.0
    n_ord = int(1_500_000 * sf)
    n_cust = n_ord // 10
    ocust = DevColumn(np.int64, n_ord)
    shim.call("qk_gen_orders", None, c_u64(n_ord), c_u64(0), c_u64(42),
              c_i64(n_cust), None, ocust.ptr, None, None, None, None,
              c_i64(1))
    ck = DevColumn(np.int64, n_cust)
    cnk = DevColumn(np.int32, n_cust)
    shim.call("qk_gen_customer", None, c_u64(n_cust), c_u64(0), c_u64(42),
              ck.ptr, None, cnk.ptr)
    bal = DevColumn(np.float64, n_cust)
    shim.call("qk_gen_aux", None, c_u64(n_cust), c_u64(0), c_u64(42),
              c_u64(0xACC7), 2, c_i64(-99999), c_i64(999999), None,
              bal.ptr)
    cust = {"c_custkey": ck, "c_nationkey": cnk, "c_acctbal": bal}
    ords = {"o_custkey": ocust}
    DQ.q22(cust, ords)
    pr = cProfile.Profile()
    pr.enable()
    DQ.q22(cust, ords)
    pr.disable()
    pstats.Stats(pr).sort_stats("cumtime").print_stats(18)


if __name__ == "__main__":
    if len(sys.argv) > 1 and sys.argv[1] == "profile":
        profile_whole()
    else:
        main()


