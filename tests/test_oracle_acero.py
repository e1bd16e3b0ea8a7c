"""External-engine anchor for the oracle.

The reference defines results via DuckDB SQL (apps/tpc-h/tpch_ref.py) and
computes with polars — neither is installable here, and dbgen's per-stream
seed table (dbgen rnd.c) is not vendored in /root/reference, so published
TPC-H answer sets cannot be asserted bit-for-bit (oracle/tpch_gen.py
docstring). The strongest independent engine REACHABLE in this container
is pyarrow Acero (C++ hash join / hash aggregation — the same Arrow
stack the reference itself hands its executors). These tests restate
Q1/Q3/Q5/Q6 clause-for-clause on Acero and assert the numpy oracle
matches it on the same generated inputs: two independent implementations
of the reference SQL agreeing, rather than the oracle checking itself.
"""
import numpy as np
import pyarrow as pa
import pyarrow.compute as pc
import pytest

from oracle import tpch_gen as G, queries as OQ

# Two (SF, seed) points: agreement at a second seed and size rules out
# a coincidental match tuned to one generated dataset.
@pytest.fixture(scope="module", params=[(0.05, 42), (0.08, 7)],
                ids=["sf.05-seed42", "sf.08-seed7"])
def data(request):
    sf, seed = request.param
    d = G.gen_all(sf, seed)
    d["_sf"], d["_seed"] = sf, seed
    return d


def _li_table(li):
    return pa.table({
        "l_orderkey": li["l_orderkey"],
        "l_suppkey": li["l_suppkey"],
        "l_quantity": li["l_quantity"],
        "l_extendedprice": li["l_extendedprice"],
        "l_discount": li["l_discount"],
        "l_tax": li["l_tax"],
        "l_returnflag": np.array(G.RETURNFLAG)[li["l_returnflag"]],
        "l_linestatus": np.array(G.LINESTATUS)[li["l_linestatus"]],
        "l_shipdate": li["l_shipdate"],
    })


def test_q1_oracle_equals_acero(data):
    """tpch_ref.py:16-38 on Acero: filter + grouped sums/means/count."""
    t = _li_table(data["lineitem"])
    t = t.filter(pc.less_equal(t["l_shipdate"], G.Q1_CUTOFF))
    disc_price = pc.multiply(t["l_extendedprice"],
                             pc.subtract(pa.scalar(1.0), t["l_discount"]))
    charge = pc.multiply(disc_price,
                         pc.add(pa.scalar(1.0), t["l_tax"]))
    t = t.append_column("disc_price", disc_price)
    t = t.append_column("charge", charge)
    g = t.group_by(["l_returnflag", "l_linestatus"]).aggregate([
        ("l_quantity", "sum"), ("l_extendedprice", "sum"),
        ("disc_price", "sum"), ("charge", "sum"),
        ("l_quantity", "mean"), ("l_extendedprice", "mean"),
        ("l_discount", "mean"), ("l_orderkey", "count"),
    ])
    g = g.sort_by([("l_returnflag", "ascending"),
                   ("l_linestatus", "ascending")])
    want = OQ.q1(data["lineitem"])
    assert g.num_rows == len(want["count_order"])
    assert g.column("l_returnflag").to_pylist() == \
        list(want["l_returnflag"])
    assert g.column("l_linestatus").to_pylist() == \
        list(want["l_linestatus"])
    assert np.array_equal(np.asarray(g.column("l_orderkey_count")),
                          want["count_order"])
    for acol, wcol in [("l_quantity_sum", "sum_qty"),
                       ("l_extendedprice_sum", "sum_base_price"),
                       ("disc_price_sum", "sum_disc_price"),
                       ("charge_sum", "sum_charge"),
                       ("l_quantity_mean", "avg_qty"),
                       ("l_extendedprice_mean", "avg_price"),
                       ("l_discount_mean", "avg_disc")]:
        np.testing.assert_allclose(np.asarray(g.column(acol)), want[wcol],
                                   rtol=1e-9, err_msg=acol)


def test_q6_oracle_equals_acero(data):
    li = data["lineitem"]
    t = pa.table({k: li[k] for k in ("l_shipdate", "l_quantity",
                                     "l_extendedprice", "l_discount")})
    lo, hi = 0.06 - 0.01, 0.06 + 0.01
    m = pc.and_(
        pc.and_(pc.greater_equal(t["l_shipdate"], G.Q5_LO),
                pc.less(t["l_shipdate"], G.Q5_HI)),
        pc.and_(pc.and_(pc.greater_equal(t["l_discount"], lo),
                        pc.less_equal(t["l_discount"], hi)),
                pc.less(t["l_quantity"], 24.0)))
    ft = t.filter(m)
    rev = pc.sum(pc.multiply(ft["l_extendedprice"],
                             ft["l_discount"])).as_py()
    want = OQ.q6(li)
    np.testing.assert_allclose(rev, want["revenue"], rtol=1e-9)
    assert ft.num_rows == want["rows_passed"]


def test_q3_oracle_equals_acero(data):
    """tpch_ref.py:89-115 on Acero: two hash joins + hash group-by."""
    li, orders, cust = data["lineitem"], data["orders"], data["customer"]
    c = pa.table({"c_custkey": cust["c_custkey"],
                  "seg": cust["c_mktsegment"]})
    c = c.filter(pc.equal(c["seg"], G.MKTSEGMENT.index("BUILDING")))
    o = pa.table({k: orders[k] for k in ("o_orderkey", "o_custkey",
                                         "o_orderdate", "o_shippriority")})
    o = o.filter(pc.less(o["o_orderdate"], G.Q3_DATE))
    o = o.join(c.select(["c_custkey"]), keys="o_custkey",
               right_keys="c_custkey", join_type="left semi")
    l = pa.table({k: li[k] for k in ("l_orderkey", "l_shipdate",
                                     "l_extendedprice", "l_discount")})
    l = l.filter(pc.greater(l["l_shipdate"], G.Q3_DATE))
    j = l.join(o, keys="l_orderkey", right_keys="o_orderkey",
               join_type="inner")
    rev = pc.multiply(j["l_extendedprice"],
                      pc.subtract(pa.scalar(1.0), j["l_discount"]))
    j = j.append_column("revenue", rev)
    g = j.group_by(["l_orderkey", "o_orderdate",
                    "o_shippriority"]).aggregate([("revenue", "sum")])
    g = g.sort_by([("l_orderkey", "ascending")])
    full, top10 = OQ.q3(li, orders, cust)
    order = np.argsort(full["l_orderkey"])
    assert g.num_rows == len(full["l_orderkey"])
    assert np.array_equal(np.asarray(g.column("l_orderkey")),
                          full["l_orderkey"][order])
    assert np.array_equal(np.asarray(g.column("o_orderdate")),
                          full["o_orderdate"][order])
    np.testing.assert_allclose(np.asarray(g.column("revenue_sum")),
                               full["revenue"][order], rtol=1e-9)
    # and the top-10 rule (revenue desc, orderdate asc) on Acero's result
    gs = g.sort_by([("revenue_sum", "descending"),
                    ("o_orderdate", "ascending"),
                    ("l_orderkey", "ascending")]).slice(0, 10)
    assert np.array_equal(np.asarray(gs.column("l_orderkey")),
                          top10["l_orderkey"])
    np.testing.assert_allclose(np.asarray(gs.column("revenue_sum")),
                               top10["revenue"], rtol=1e-9)


def test_q5_oracle_equals_acero(data):
    """tpch_ref.py:142-169 on Acero: 6-table chain with the extra
    c_nationkey = s_nationkey equi-predicate."""
    li, orders = data["lineitem"], data["orders"]
    cust, supp = data["customer"], data["supplier"]
    nat, reg = data["nation"], data["region"]
    asia = G.REGIONS.index("ASIA")
    n = pa.table({"n_nationkey": nat["n_nationkey"],
                  "n_regionkey": nat["n_regionkey"]})
    n = n.filter(pc.equal(n["n_regionkey"], asia))
    c = pa.table({"c_custkey": cust["c_custkey"],
                  "c_nationkey": cust["c_nationkey"]})
    c = c.join(n.select(["n_nationkey"]), keys="c_nationkey",
               right_keys="n_nationkey", join_type="left semi")
    o = pa.table({k: orders[k] for k in ("o_orderkey", "o_custkey",
                                         "o_orderdate")})
    o = o.filter(pc.and_(pc.greater_equal(o["o_orderdate"], G.Q5_LO),
                         pc.less(o["o_orderdate"], G.Q5_HI)))
    oc = o.join(c, keys="o_custkey", right_keys="c_custkey",
                join_type="inner")
    l = pa.table({k: li[k] for k in ("l_orderkey", "l_suppkey",
                                     "l_extendedprice", "l_discount")})
    j = l.join(oc.select(["o_orderkey", "c_nationkey"]),
               keys="l_orderkey", right_keys="o_orderkey",
               join_type="inner")
    s = pa.table({"s_suppkey": supp["s_suppkey"],
                  "s_nationkey": supp["s_nationkey"]})
    j = j.join(s, keys="l_suppkey", right_keys="s_suppkey",
               join_type="inner")
    j = j.filter(pc.equal(j["c_nationkey"], j["s_nationkey"]))
    rev = pc.multiply(j["l_extendedprice"],
                      pc.subtract(pa.scalar(1.0), j["l_discount"]))
    j = j.append_column("revenue", rev)
    g = j.group_by(["c_nationkey"]).aggregate([("revenue", "sum")])
    got = {int(k): v for k, v in zip(g.column("c_nationkey").to_pylist(),
                                     g.column("revenue_sum").to_pylist())}
    want = OQ.q5(li, orders, cust, supp, nat, reg)
    names = [nm for nm, _ in G.NATIONS]
    for nm, wrev in want:
        nk = names.index(nm)
        if wrev == 0.0:
            assert got.get(nk, 0.0) == 0.0
        else:
            np.testing.assert_allclose(got[nk], wrev, rtol=1e-9,
                                       err_msg=nm)
    # no non-ASIA nation appears
    asia_keys = {i for i in range(25) if nat["n_regionkey"][i] == asia}
    assert set(got) <= asia_keys


def test_q4_oracle_equals_acero(data):
    """tpch_ref.py:117-140 on Acero: EXISTS == left-semi join."""
    li, orders = data["lineitem"], data["orders"]
    l = pa.table({"l_orderkey": li["l_orderkey"],
                  "l_commitdate": li["l_commitdate"],
                  "l_receiptdate": li["l_receiptdate"]})
    l = l.filter(pc.less(l["l_commitdate"], l["l_receiptdate"]))
    o = pa.table({"o_orderkey": orders["o_orderkey"],
                  "o_orderdate": orders["o_orderdate"],
                  "o_orderpriority": orders["o_orderpriority"]})
    o = o.filter(pc.and_(pc.greater_equal(o["o_orderdate"], OQ.Q4_LO),
                         pc.less(o["o_orderdate"], OQ.Q4_HI)))
    o = o.join(l.select(["l_orderkey"]), keys="o_orderkey",
               right_keys="l_orderkey", join_type="left semi")
    g = o.group_by(["o_orderpriority"]).aggregate([("o_orderkey", "count")])
    got = {G.ORDERPRIORITY[int(k)]: int(v) for k, v in
           zip(g.column("o_orderpriority").to_pylist(),
               g.column("o_orderkey_count").to_pylist())}
    want = OQ.q4(li, orders)
    assert got == want


def test_q18_oracle_equals_acero(data):
    li, orders = data["lineitem"], data["orders"]
    cust = G.gen_customer(data["_sf"], data["_seed"], strings=True)
    l = pa.table({"l_orderkey": li["l_orderkey"],
                  "l_quantity": li["l_quantity"]})
    g = l.group_by("l_orderkey").aggregate([("l_quantity", "sum")])
    g = g.filter(pc.greater(g["l_quantity_sum"], 300.0))
    o = pa.table({k: orders[k] for k in ("o_orderkey", "o_custkey",
                                         "o_orderdate", "o_totalprice")})
    j = o.join(g, keys="o_orderkey", right_keys="l_orderkey",
               join_type="inner")
    c = pa.table({"c_custkey": cust["c_custkey"],
                  "c_name": pa.array(list(cust["c_name"]))})
    j = j.join(c, keys="o_custkey", right_keys="c_custkey",
               join_type="inner")
    j = j.sort_by([("o_totalprice", "descending"),
                   ("o_orderdate", "ascending"),
                   ("o_orderkey", "ascending")]).slice(0, 100)
    want = OQ.q18(li, orders, cust)
    assert j.num_rows == len(want["o_orderkey"])
    assert np.array_equal(np.asarray(j.column("o_orderkey")),
                          want["o_orderkey"])
    assert j.column("c_name").to_pylist() == list(want["c_name"])
    np.testing.assert_allclose(np.asarray(j.column("l_quantity_sum")),
                               want["sum_qty"], rtol=1e-12)
    np.testing.assert_allclose(np.asarray(j.column("o_totalprice")),
                               want["o_totalprice"], rtol=0)


def test_q10_oracle_equals_acero(data):
    li, orders, nat = data["lineitem"], data["orders"], data["nation"]
    cust = G.gen_customer(data["_sf"], data["_seed"], strings=True)
    rcode = G.RETURNFLAG.index("R")
    l = pa.table({k: li[k] for k in ("l_orderkey", "l_returnflag",
                                     "l_extendedprice", "l_discount")})
    l = l.filter(pc.equal(l["l_returnflag"], rcode))
    o = pa.table({k: orders[k] for k in ("o_orderkey", "o_custkey",
                                         "o_orderdate")})
    o = o.filter(pc.and_(pc.greater_equal(o["o_orderdate"], OQ.Q10_LO),
                         pc.less(o["o_orderdate"], OQ.Q10_HI)))
    j = l.join(o, keys="l_orderkey", right_keys="o_orderkey",
               join_type="inner")
    rev = pc.multiply(j["l_extendedprice"],
                      pc.subtract(pa.scalar(1.0), j["l_discount"]))
    j = j.append_column("revenue", rev)
    g = j.group_by("o_custkey").aggregate([("revenue", "sum")])
    g = g.sort_by([("revenue_sum", "descending"),
                   ("o_custkey", "ascending")]).slice(0, 20)
    want = OQ.q10(li, orders, cust, nat)
    assert np.array_equal(np.asarray(g.column("o_custkey")),
                          want["c_custkey"])
    np.testing.assert_allclose(np.asarray(g.column("revenue_sum")),
                               want["revenue"], rtol=1e-9)
    # attribute attachment (functionally dependent on c_custkey)
    row = want["c_custkey"] - 1
    assert list(want["c_name"]) == list(cust["c_name"][row])
    assert list(want["n_name"]) == \
        [nat["n_name"][k] for k in cust["c_nationkey"][row]]


def test_q12_oracle_equals_acero(data):
    li, orders = data["lineitem"], data["orders"]
    mail, shipm = G.SHIPMODE.index("MAIL"), G.SHIPMODE.index("SHIP")
    l = pa.table({k: li[k] for k in ("l_orderkey", "l_shipmode",
                                     "l_shipdate", "l_commitdate",
                                     "l_receiptdate")})
    m = pc.and_(
        pc.and_(pc.is_in(l["l_shipmode"],
                         value_set=pa.array([mail, shipm],
                                            type=pa.uint8())),
                pc.less(l["l_commitdate"], l["l_receiptdate"])),
        pc.and_(pc.less(l["l_shipdate"], l["l_commitdate"]),
                pc.and_(pc.greater_equal(l["l_receiptdate"], OQ.Q12_LO),
                        pc.less(l["l_receiptdate"], OQ.Q12_HI))))
    l = l.filter(m)
    o = pa.table({"o_orderkey": orders["o_orderkey"],
                  "o_orderpriority": orders["o_orderpriority"]})
    j = l.join(o, keys="l_orderkey", right_keys="o_orderkey",
               join_type="inner")
    high = pc.less_equal(j["o_orderpriority"], 1)
    want = OQ.q12(li, orders)
    for code, name in ((mail, "MAIL"), (shipm, "SHIP")):
        sel = pc.equal(j["l_shipmode"], code)
        h = j.filter(pc.and_(sel, high)).num_rows
        lo = j.filter(pc.and_(sel, pc.invert(high))).num_rows
        assert (h, lo) == want[name], name


def test_q14_oracle_equals_acero(data):
    li, part = data["lineitem"], data["part"]
    l = pa.table({k: li[k] for k in ("l_partkey", "l_shipdate",
                                     "l_extendedprice", "l_discount")})
    l = l.filter(pc.and_(pc.greater_equal(l["l_shipdate"], OQ.Q14_LO),
                         pc.less(l["l_shipdate"], OQ.Q14_HI)))
    p = pa.table({"p_partkey": part["p_partkey"],
                  "promo": (part["p_type"] // 25) == G.PTYPE_PROMO_SYL1})
    j = l.join(p, keys="l_partkey", right_keys="p_partkey",
               join_type="inner")
    rev = pc.multiply(j["l_extendedprice"],
                      pc.subtract(pa.scalar(1.0), j["l_discount"]))
    total = pc.sum(rev).as_py()
    promo = pc.sum(pc.if_else(j["promo"], rev, 0.0)).as_py()
    got = 100.0 * promo / total
    want = OQ.q14(li, part)
    np.testing.assert_allclose(got, want, rtol=1e-9)


def test_q7_oracle_equals_acero(data):
    li, orders = data["lineitem"], data["orders"]
    cust, supp, nat = data["customer"], data["supplier"], data["nation"]
    names = list(nat["n_name"])
    fr, de = names.index("FRANCE"), names.index("GERMANY")
    l = pa.table({k: li[k] for k in ("l_orderkey", "l_suppkey",
                                     "l_shipdate", "l_extendedprice",
                                     "l_discount")})
    l = l.filter(pc.and_(pc.greater_equal(l["l_shipdate"], OQ.Q7_LO),
                         pc.less_equal(l["l_shipdate"], OQ.Q7_HI)))
    o = pa.table({"o_orderkey": orders["o_orderkey"],
                  "o_custkey": orders["o_custkey"]})
    c = pa.table({"c_custkey": cust["c_custkey"],
                  "cn": cust["c_nationkey"]})
    s = pa.table({"s_suppkey": supp["s_suppkey"],
                  "sn": supp["s_nationkey"]})
    j = l.join(o, keys="l_orderkey", right_keys="o_orderkey")
    j = j.join(c, keys="o_custkey", right_keys="c_custkey")
    j = j.join(s, keys="l_suppkey", right_keys="s_suppkey")
    pair = pc.or_(pc.and_(pc.equal(j["sn"], fr), pc.equal(j["cn"], de)),
                  pc.and_(pc.equal(j["sn"], de), pc.equal(j["cn"], fr)))
    j = j.filter(pair)
    rev = pc.multiply(j["l_extendedprice"],
                      pc.subtract(pa.scalar(1.0), j["l_discount"]))
    j = j.append_column("rev", rev)
    yr = pc.if_else(pc.greater_equal(j["l_shipdate"], OQ.Y1996),
                    1996, 1995)
    j = j.append_column("yr", yr)
    g = j.group_by(["sn", "cn", "yr"]).aggregate([("rev", "sum")])
    got = {}
    for snv, cnv, yv, rv in zip(g.column("sn").to_pylist(),
                                g.column("cn").to_pylist(),
                                g.column("yr").to_pylist(),
                                g.column("rev_sum").to_pylist()):
        got[(names[snv], names[cnv], yv)] = rv
    want = OQ.q7(li, orders, cust, supp, nat)
    for k, v in want.items():
        np.testing.assert_allclose(got.get(k, 0.0), v, rtol=1e-9,
                                   err_msg=str(k))


def test_q8_oracle_equals_acero(data):
    li, orders = data["lineitem"], data["orders"]
    cust, supp = data["customer"], data["supplier"]
    part, nat, reg = data["part"], data["nation"], data["region"]
    america = G.REGIONS.index("AMERICA")
    code = ((G.PTYPE_SYL1.index("ECONOMY") * 5 +
             G.PTYPE_SYL2.index("ANODIZED")) * 5 +
            G.PTYPE_SYL3.index("STEEL"))
    p = pa.table({"p_partkey": part["p_partkey"],
                  "p_type": part["p_type"]})
    p = p.filter(pc.equal(p["p_type"], code))
    o = pa.table({k: orders[k] for k in ("o_orderkey", "o_custkey",
                                         "o_orderdate")})
    o = o.filter(pc.and_(pc.greater_equal(o["o_orderdate"], OQ.Q7_LO),
                         pc.less_equal(o["o_orderdate"], OQ.Q7_HI)))
    c = pa.table({"c_custkey": cust["c_custkey"],
                  "cn": cust["c_nationkey"]})
    amer = pa.table({"nk": np.nonzero(nat["n_regionkey"] == america)[0]
                     .astype(np.int32)})
    c = c.join(amer, keys="cn", right_keys="nk", join_type="left semi")
    o = o.join(c.select(["c_custkey"]), keys="o_custkey",
               right_keys="c_custkey", join_type="left semi")
    l = pa.table({k: li[k] for k in ("l_orderkey", "l_suppkey",
                                     "l_partkey", "l_extendedprice",
                                     "l_discount")})
    l = l.join(p.select(["p_partkey"]), keys="l_partkey",
               right_keys="p_partkey", join_type="left semi")
    j = l.join(o, keys="l_orderkey", right_keys="o_orderkey")
    s = pa.table({"s_suppkey": supp["s_suppkey"],
                  "sn": supp["s_nationkey"]})
    j = j.join(s, keys="l_suppkey", right_keys="s_suppkey")
    rev = pc.multiply(j["l_extendedprice"],
                      pc.subtract(pa.scalar(1.0), j["l_discount"]))
    j = j.append_column("rev", rev)
    brazil = list(nat["n_name"]).index("BRAZIL")
    want = OQ.q8(li, orders, cust, supp, part, nat, reg)
    for y, lo_ok in ((1995, True), (1996, True)):
        if y == 1995:
            m = pc.less(j["o_orderdate"], OQ.Y1996)
        else:
            m = pc.greater_equal(j["o_orderdate"], OQ.Y1996)
        sel = j.filter(m)
        tot = pc.sum(sel["rev"]).as_py() or 0.0
        br = pc.sum(sel.filter(pc.equal(sel["sn"],
                                        brazil))["rev"]).as_py() or 0.0
        got = br / tot if tot else 0.0
        np.testing.assert_allclose(got, want[y], rtol=1e-9, err_msg=y)


def test_q17_oracle_equals_acero(data):
    li, part = data["lineitem"], data["part"]
    g = pa.table({"l_partkey": li["l_partkey"],
                  "l_quantity": li["l_quantity"]}) \
        .group_by("l_partkey").aggregate([("l_quantity", "mean")])
    p = pa.table({"p_partkey": part["p_partkey"],
                  "b": part["p_brand"], "c": part["p_container"]})
    p = p.filter(pc.and_(pc.equal(p["b"], 12), pc.equal(p["c"], 17)))
    l = pa.table({k: li[k] for k in ("l_partkey", "l_quantity",
                                     "l_extendedprice")})
    j = l.join(p.select(["p_partkey"]), keys="l_partkey",
               right_keys="p_partkey", join_type="left semi")
    j = j.join(g, keys="l_partkey", right_keys="l_partkey",
               join_type="inner", right_suffix="_avg")
    j = j.filter(pc.less(j["l_quantity"],
                         pc.multiply(pa.scalar(0.2),
                                     j["l_quantity_mean"])))
    got = (pc.sum(j["l_extendedprice"]).as_py() or 0.0) / 7.0
    want = OQ.q17(li, part)
    np.testing.assert_allclose(got, want, rtol=1e-9)


def test_q15_oracle_equals_acero(data):
    li, supp = data["lineitem"], data["supplier"]
    l = pa.table({k: li[k] for k in ("l_suppkey", "l_shipdate",
                                     "l_extendedprice", "l_discount")})
    l = l.filter(pc.and_(pc.greater_equal(l["l_shipdate"], OQ.Q15_LO),
                         pc.less(l["l_shipdate"], OQ.Q15_HI)))
    rev = pc.multiply(l["l_extendedprice"],
                      pc.subtract(pa.scalar(1.0), l["l_discount"]))
    l = l.append_column("rev", rev)
    g = l.group_by("l_suppkey").aggregate([("rev", "sum")])
    mx = pc.max(g.column("rev_sum")).as_py()
    winners = sorted(g.filter(pc.equal(g["rev_sum"], mx))
                     .column("l_suppkey").to_pylist())
    wk, wmx = OQ.q15(li, supp)
    assert winners == list(wk)
    np.testing.assert_allclose(mx, wmx, rtol=1e-9)


def test_q19_oracle_equals_acero(data):
    li, part = data["lineitem"], data["part"]
    air = [G.SHIPMODE.index("AIR"), G.SHIPMODE.index("REG AIR")]
    deliver = G.SHIPINSTRUCT.index("DELIVER IN PERSON")
    l = pa.table({k: li[k] for k in ("l_partkey", "l_quantity",
                                     "l_extendedprice", "l_discount",
                                     "l_shipmode", "l_shipinstruct")})
    p = pa.table({k: part[k] for k in ("p_partkey", "p_brand",
                                       "p_container", "p_size")})
    j = l.join(p, keys="l_partkey", right_keys="p_partkey",
               join_type="inner")
    m = pa.array(np.zeros(j.num_rows, dtype=bool))
    for bname, conts, qlo, qhi, slo, shi in OQ.Q19_BRANCHES:
        bc = G.brand_code(bname)
        cc = [G.container_code(c) for c in conts]
        b = pc.and_(
            pc.and_(pc.equal(j["p_brand"], bc),
                    pc.is_in(j["p_container"],
                             value_set=pa.array(cc, type=pa.uint8()))),
            pc.and_(
                pc.and_(pc.greater_equal(j["l_quantity"], qlo),
                        pc.less_equal(j["l_quantity"], qhi)),
                pc.and_(pc.greater_equal(j["p_size"], slo),
                        pc.less_equal(j["p_size"], shi))))
        m = pc.or_(m, b)
    m = pc.and_(m, pc.and_(
        pc.is_in(j["l_shipmode"], value_set=pa.array(air,
                                                     type=pa.uint8())),
        pc.equal(j["l_shipinstruct"], deliver)))
    sel = j.filter(m)
    got = pc.sum(pc.multiply(sel["l_extendedprice"],
                             pc.subtract(pa.scalar(1.0),
                                         sel["l_discount"]))).as_py() or 0.0
    want = OQ.q19(li, part)
    np.testing.assert_allclose(got, want, rtol=1e-9)


def test_q2_oracle_equals_acero(data):
    part, supp = data["part"], data["supplier"]
    ps, nat = data["partsupp"], data["nation"]
    europe = G.REGIONS.index("EUROPE")
    eu = pa.table({"nk": np.nonzero(nat["n_regionkey"] == europe)[0]
                   .astype(np.int32)})
    s = pa.table({"s_suppkey": supp["s_suppkey"],
                  "sn": supp["s_nationkey"],
                  "s_acctbal": supp["s_acctbal"]})
    s_eu = s.join(eu, keys="sn", right_keys="nk", join_type="left semi")
    p = pa.table({k: ps[k] for k in ("ps_partkey", "ps_suppkey",
                                     "ps_supplycost")})
    p = p.join(s_eu.select(["s_suppkey"]), keys="ps_suppkey",
               right_keys="s_suppkey", join_type="left semi")
    mins = p.group_by("ps_partkey").aggregate([("ps_supplycost", "min")])
    pt = pa.table({"p_partkey": part["p_partkey"],
                   "p_size": part["p_size"], "p_type": part["p_type"]})
    pt = pt.filter(pc.and_(pc.equal(pt["p_size"], 15),
                           pc.equal(pc.bit_wise_and(pa.chunked_array(
                               [pa.array(np.asarray(part["p_type"]) % 5)]),
                               255), 0)))
    j = p.join(mins, keys="ps_partkey", right_keys="ps_partkey")
    j = j.filter(pc.equal(j["ps_supplycost"], j["ps_supplycost_min"]))
    j = j.join(pt.select(["p_partkey"]), keys="ps_partkey",
               right_keys="p_partkey", join_type="left semi")
    want = OQ.q2(part, supp, ps, nat, data["region"])
    got = set(zip(j.column("ps_partkey").to_pylist(),
                  j.column("ps_suppkey").to_pylist()))
    # the oracle's top-100 rows must all be Acero winners, counts match
    assert len(got) >= len(want["p_partkey"])
    for a, b in zip(want["p_partkey"], want["s_suppkey"]):
        assert (a, b) in got


def test_q11_oracle_equals_acero(data):
    ps, supp, nat = data["partsupp"], data["supplier"], data["nation"]
    germany = list(nat["n_name"]).index("GERMANY")
    s = pa.table({"s_suppkey": supp["s_suppkey"],
                  "sn": supp["s_nationkey"]})
    s = s.filter(pc.equal(s["sn"], germany))
    p = pa.table({k: ps[k] for k in ("ps_partkey", "ps_suppkey",
                                     "ps_supplycost", "ps_availqty")})
    p = p.join(s.select(["s_suppkey"]), keys="ps_suppkey",
               right_keys="s_suppkey", join_type="left semi")
    val = pc.multiply(p["ps_supplycost"],
                      pc.cast(p["ps_availqty"], pa.float64()))
    p = p.append_column("val", val)
    g = p.group_by("ps_partkey").aggregate([("val", "sum")])
    thr = pc.sum(p["val"]).as_py() * 0.0001
    g = g.filter(pc.greater(g["val_sum"], thr))
    g = g.sort_by([("val_sum", "descending"), ("ps_partkey", "ascending")])
    wk, wv = OQ.q11(ps, supp, nat)
    assert np.array_equal(np.asarray(g.column("ps_partkey")), wk)
    np.testing.assert_allclose(np.asarray(g.column("val_sum")), wv,
                               rtol=1e-9)


def test_q20_oracle_equals_acero(data):
    li, part = data["lineitem"], data["part"]
    ps, supp, nat = data["partsupp"], data["supplier"], data["nation"]
    forest = pa.table({"pk": part["p_partkey"][
        part["p_name1"] == G.P_NAME_FOREST]})
    l = pa.table({k: li[k] for k in ("l_partkey", "l_suppkey",
                                     "l_quantity", "l_shipdate")})
    l = l.filter(pc.and_(pc.greater_equal(l["l_shipdate"], G.Q5_LO),
                         pc.less(l["l_shipdate"], G.Q5_HI)))
    l = l.join(forest, keys="l_partkey", right_keys="pk",
               join_type="left semi")
    g = l.group_by(["l_partkey", "l_suppkey"]).aggregate(
        [("l_quantity", "sum")])
    p = pa.table({k: ps[k] for k in ("ps_partkey", "ps_suppkey",
                                     "ps_availqty")})
    p = p.join(forest, keys="ps_partkey", right_keys="pk",
               join_type="left semi")
    j = p.join(g, keys=["ps_partkey", "ps_suppkey"],
               right_keys=["l_partkey", "l_suppkey"], join_type="inner")
    j = j.filter(pc.greater(pc.cast(j["ps_availqty"], pa.float64()),
                            pc.multiply(pa.scalar(0.5),
                                        j["l_quantity_sum"])))
    canada = list(nat["n_name"]).index("CANADA")
    s = pa.table({"s_suppkey": supp["s_suppkey"],
                  "sn": supp["s_nationkey"]})
    s = s.filter(pc.equal(s["sn"], canada))
    winners = s.join(j.select(["ps_suppkey"]), keys="s_suppkey",
                     right_keys="ps_suppkey", join_type="left semi")
    got = sorted(winners.column("s_suppkey").to_pylist())
    want = OQ.q20(li, part, ps, supp, nat)
    assert got == list(want)


def test_q9_oracle_equals_acero(data):
    li, orders = data["lineitem"], data["orders"]
    supp, part, ps, nat = (data["supplier"], data["part"],
                           data["partsupp"], data["nation"])
    g = pa.table({"pk": part["p_partkey"][part["p_name_green"] == 1]})
    l = pa.table({k: li[k] for k in ("l_partkey", "l_suppkey",
                                     "l_orderkey", "l_quantity",
                                     "l_extendedprice", "l_discount")})
    l = l.join(g, keys="l_partkey", right_keys="pk", join_type="left semi")
    p = pa.table({k: ps[k] for k in ("ps_partkey", "ps_suppkey",
                                     "ps_supplycost")})
    j = l.join(p, keys=["l_partkey", "l_suppkey"],
               right_keys=["ps_partkey", "ps_suppkey"],
               join_type="inner")
    o = pa.table({"o_orderkey": orders["o_orderkey"],
                  "o_orderdate": orders["o_orderdate"]})
    j = j.join(o, keys="l_orderkey", right_keys="o_orderkey")
    s = pa.table({"s_suppkey": supp["s_suppkey"],
                  "sn": supp["s_nationkey"]})
    j = j.join(s, keys="l_suppkey", right_keys="s_suppkey")
    amount = pc.subtract(
        pc.multiply(j["l_extendedprice"],
                    pc.subtract(pa.scalar(1.0), j["l_discount"])),
        pc.multiply(j["ps_supplycost"], j["l_quantity"]))
    yr = pc.year(pc.cast(pc.multiply(pc.cast(j["o_orderdate"],
                                             pa.int64()),
                                     86400000), pa.timestamp("ms")))
    j = j.append_column("amount", amount).append_column("yr", yr)
    gr = j.group_by(["sn", "yr"]).aggregate([("amount", "sum")])
    names = list(nat["n_name"])
    got = {(names[s_], int(y)): v for s_, y, v in
           zip(gr.column("sn").to_pylist(), gr.column("yr").to_pylist(),
               gr.column("amount_sum").to_pylist())}
    want = OQ.q9(li, orders, supp, part, ps, nat)
    assert set(got) >= set(want)
    for k, v in want.items():
        np.testing.assert_allclose(got[k], v, rtol=1e-9, err_msg=str(k))


def test_q13_oracle_equals_acero(data):
    orders, cust = data["orders"], data["customer"]
    o = pa.table({"o_custkey": orders["o_custkey"],
                  "flag": orders["o_comment_special"]})
    o = o.filter(pc.equal(o["flag"], 0))
    g = o.group_by("o_custkey").aggregate([("flag", "count")])
    per = np.zeros(len(cust["c_custkey"]) + 2, dtype=np.int64)
    per[np.asarray(g.column("o_custkey"))] = \
        np.asarray(g.column("flag_count"))
    counts = np.bincount(per[1:len(cust["c_custkey"]) + 1])
    got = {int(c): int(v) for c, v in enumerate(counts) if v}
    assert got == OQ.q13(orders, cust)


def test_q16_oracle_equals_acero(data):
    part, ps, supp = data["part"], data["partsupp"], data["supplier"]
    med_pol = ((np.asarray(part["p_type"]) // 25 ==
                G.PTYPE_SYL1.index("MEDIUM")) &
               ((np.asarray(part["p_type"]) // 5) % 5 ==
                G.PTYPE_SYL2.index("POLISHED")))
    sel = ((part["p_brand"] != G.brand_code("Brand#45")) & ~med_pol &
           np.isin(part["p_size"], [49, 14, 23, 45, 19, 3, 36, 9]))
    p = pa.table({"p_partkey": part["p_partkey"][sel],
                  "b": part["p_brand"][sel], "t": part["p_type"][sel],
                  "z": part["p_size"][sel]})
    bad = pa.table({"sk": supp["s_suppkey"][
        supp["s_comment_complaints"] == 1]})
    j = pa.table({k: ps[k] for k in ("ps_partkey", "ps_suppkey")})
    j = j.join(bad, keys="ps_suppkey", right_keys="sk",
               join_type="left anti")
    j = j.join(p, keys="ps_partkey", right_keys="p_partkey",
               join_type="inner")
    dd = j.group_by(["b", "t", "z", "ps_suppkey"]).aggregate([])
    g = dd.group_by(["b", "t", "z"]).aggregate([("ps_suppkey", "count")])
    got = {(int(b), int(t), int(z)): int(c) for b, t, z, c in
           zip(g.column("b").to_pylist(), g.column("t").to_pylist(),
               g.column("z").to_pylist(),
               g.column("ps_suppkey_count").to_pylist())}
    assert got == dict(OQ.q16(part, ps, supp))


def test_q21_oracle_equals_acero(data):
    li, orders = data["lineitem"], data["orders"]
    supp, nat = data["supplier"], data["nation"]
    l = pa.table({k: li[k] for k in ("l_orderkey", "l_suppkey",
                                     "l_receiptdate", "l_commitdate")})
    dd = l.group_by(["l_orderkey", "l_suppkey"]).aggregate([])
    nall = dd.group_by("l_orderkey").aggregate([("l_suppkey", "count")])
    nall = nall.rename_columns(["l_orderkey", "n_all"])
    late = l.filter(pc.greater(l["l_receiptdate"], l["l_commitdate"]))
    dl = late.group_by(["l_orderkey", "l_suppkey"]).aggregate([])
    nlate = dl.group_by("l_orderkey").aggregate(
        [("l_suppkey", "count"), ("l_suppkey", "max")])
    nlate = nlate.rename_columns(["l_orderkey", "n_late",
                                  "l_suppkey_max"])
    o = pa.table({"o_orderkey": orders["o_orderkey"],
                  "st": orders["o_orderstatus"]})
    o = o.filter(pc.equal(o["st"], 0))
    j = o.join(nall, keys="o_orderkey", right_keys="l_orderkey")
    j = j.join(nlate, keys="o_orderkey", right_keys="l_orderkey")
    j = j.filter(pc.and_(pc.greater_equal(j["n_all"], 2),
                         pc.equal(j["n_late"], 1)))
    wait = np.asarray(j.column("l_suppkey_max"))
    saudi = list(nat["n_name"]).index("SAUDI ARABIA")
    sn = supp["s_nationkey"][wait - 1]
    cnt = np.bincount(wait[sn == saudi],
                      minlength=int(supp["s_suppkey"].max()) + 2)
    sk = np.nonzero(cnt)[0]
    order = np.lexsort((sk, -cnt[sk]))[:100]
    got = {int(sk[i]): int(cnt[sk[i]]) for i in order}
    assert got == OQ.q21(li, orders, supp, nat)


def test_q22_oracle_equals_acero(data):
    cust, orders = data["customer"], data["orders"]
    codes = np.array([13, 31, 23, 29, 30, 18, 17])
    cc = 10 + np.asarray(cust["c_nationkey"])
    c = pa.table({"c_custkey": cust["c_custkey"],
                  "cc": cc.astype(np.int32),
                  "c_acctbal": cust["c_acctbal"]})
    inl = c.filter(pc.is_in(c["cc"], value_set=pa.array(
        codes, type=pa.int32())))
    posv = inl.filter(pc.greater(inl["c_acctbal"], 0.0))
    avg = pc.mean(posv["c_acctbal"]).as_py()
    o = pa.table({"ck": np.unique(orders["o_custkey"])})
    sel = inl.filter(pc.greater(inl["c_acctbal"], avg))
    sel = sel.join(o, keys="c_custkey", right_keys="ck",
                   join_type="left anti")
    g = sel.group_by("cc").aggregate([("c_acctbal", "count"),
                                      ("c_acctbal", "sum")])
    got = {str(int(k)): (int(n), v) for k, n, v in
           zip(g.column("cc").to_pylist(),
               g.column("c_acctbal_count").to_pylist(),
               g.column("c_acctbal_sum").to_pylist())}
    want = OQ.q22(cust, orders)
    assert set(got) == set(want)
    for k in want:
        assert got[k][0] == want[k][0]
        np.testing.assert_allclose(got[k][1], want[k][1], rtol=1e-9)
