"""CPU oracle queries — TEST INFRASTRUCTURE ONLY.

Each function restates, on numpy, the result-defining SQL of the reference's
own golden oracle /root/reference/apps/tpc-h/tpch_ref.py (DuckDB), over the
columns produced by oracle.tpch_gen. All float arithmetic is f64, matching
the reference executors (DuckDB/polars compute these aggregates in double).

Used only by tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg.
"""
import numpy as np

from . import tpch_gen as G

N_Q1_GROUPS = 6  # returnflag in {A,N,R} x linestatus in {F,O}


def q1_group_id(returnflag_code, linestatus_code):
    """Group id = rflag*2 + lstatus; code order == lexicographic order, so
    ascending gid == 'order by l_returnflag, l_linestatus' (tpch_ref.py:34-37)."""
    return returnflag_code.astype(np.int64) * 2 + linestatus_code.astype(np.int64)


def q1_partials(li, cutoff=G.Q1_CUTOFF):
    """Q1 partial aggregates (the map-side / kernel-side contract).

    Restates tpch_ref.py:16-38 filter+aggregate in its two-phase form
    (sql_utils.py:299-413 rewrite: avg -> sum+count): returns a
    (6 groups x 6) f64 array with columns
    [sum_qty, sum_base_price, sum_disc_price, sum_charge, sum_disc, count].
    """
    mask = li["l_shipdate"] <= cutoff
    gid = q1_group_id(li["l_returnflag"], li["l_linestatus"])[mask]
    qty = li["l_quantity"][mask]
    price = li["l_extendedprice"][mask]
    disc = li["l_discount"][mask]
    tax = li["l_tax"][mask]
    disc_price = price * (1.0 - disc)
    charge = disc_price * (1.0 + tax)

    out = np.zeros((N_Q1_GROUPS, 6), dtype=np.float64)
    out[:, 0] = np.bincount(gid, weights=qty, minlength=N_Q1_GROUPS)
    out[:, 1] = np.bincount(gid, weights=price, minlength=N_Q1_GROUPS)
    out[:, 2] = np.bincount(gid, weights=disc_price, minlength=N_Q1_GROUPS)
    out[:, 3] = np.bincount(gid, weights=charge, minlength=N_Q1_GROUPS)
    out[:, 4] = np.bincount(gid, weights=disc, minlength=N_Q1_GROUPS)
    out[:, 5] = np.bincount(gid, minlength=N_Q1_GROUPS)
    return out


def q1_finalize(partials):
    """Final aggregate (SQLAggExecutor.done semantics, sql_executors.py:592-599):
    sum the partials, derive avgs as sum/sum per the rewrite, drop empty
    groups, order by (l_returnflag, l_linestatus).

    Returns dict of column name -> np array, columns as tpch_ref.py:16-38.
    """
    p = partials if partials.ndim == 2 else partials.reshape(-1, 6)
    nonempty = p[:, 5] > 0
    gids = np.nonzero(nonempty)[0]
    p = p[nonempty]
    cnt = p[:, 5]
    return {
        "l_returnflag": np.array([G.RETURNFLAG[g // 2] for g in gids]),
        "l_linestatus": np.array([G.LINESTATUS[g % 2] for g in gids]),
        "sum_qty": p[:, 0],
        "sum_base_price": p[:, 1],
        "sum_disc_price": p[:, 2],
        "sum_charge": p[:, 3],
        "avg_qty": p[:, 0] / cnt,
        "avg_price": p[:, 1] / cnt,
        "avg_disc": p[:, 4] / cnt,
        "count_order": cnt.astype(np.int64),
    }


def q1(li, cutoff=G.Q1_CUTOFF):
    return q1_finalize(q1_partials(li, cutoff))


def q6(li):
    """tpch_ref.py:171-183. The bounds are the literal fp64 expressions the
    reference SQL evaluates (0.06 - 0.01, 0.06 + 0.01): both oracle and GPU
    kernel must use these exact doubles."""
    lo, hi = 0.06 - 0.01, 0.06 + 0.01
    m = (
        (li["l_shipdate"] >= G.Q5_LO)
        & (li["l_shipdate"] < G.Q5_HI)
        & (li["l_discount"] >= lo)
        & (li["l_discount"] <= hi)
        & (li["l_quantity"] < 24.0)
    )
    return {
        "revenue": np.float64((li["l_extendedprice"][m] * li["l_discount"][m]).sum()),
        "rows_passed": int(m.sum()),
    }


def q3(li, orders, customer, limit=10):
    """tpch_ref.py:89-115: customer(BUILDING) x orders(<1995-03-15) x
    lineitem(ship>1995-03-15), group by (l_orderkey, o_orderdate,
    o_shippriority), revenue = sum(extendedprice*(1-discount)),
    order by revenue desc, o_orderdate asc, limit 10.

    Exploits the generator's dense o_orderkey/c_custkey for O(n) semi-join
    lookups; the general (key-agnostic) join restatement used for kernel
    parity is oracle.executors.build_probe_join.
    Returns (full_group_table_dict, top10_dict)."""
    seg_building = G.MKTSEGMENT.index("BUILDING")
    cust_ok = np.zeros(int(customer["c_custkey"].max()) + 2, dtype=bool)
    cust_ok[customer["c_custkey"][customer["c_mktsegment"] == seg_building]] = True

    omask = (orders["o_orderdate"] < G.Q3_DATE) & cust_ok[orders["o_custkey"]]
    okeys = orders["o_orderkey"][omask]
    odate = orders["o_orderdate"][omask]
    oprio = orders["o_shippriority"][omask]
    nkey = int(orders["o_orderkey"].max()) + 2
    order_ok = np.zeros(nkey, dtype=bool)
    order_ok[okeys] = True
    odate_by_key = np.zeros(nkey, dtype=np.int32)
    odate_by_key[okeys] = odate
    oprio_by_key = np.zeros(nkey, dtype=np.int32)
    oprio_by_key[okeys] = oprio

    lmask = (li["l_shipdate"] > G.Q3_DATE) & order_ok[li["l_orderkey"]]
    lkey = li["l_orderkey"][lmask]
    rev = li["l_extendedprice"][lmask] * (1.0 - li["l_discount"][lmask])

    sums = np.bincount(lkey, weights=rev, minlength=nkey)
    hit = np.zeros(nkey, dtype=bool)
    hit[lkey] = True
    gk = np.nonzero(hit)[0]
    full = {
        "l_orderkey": gk.astype(np.int64),
        "o_orderdate": odate_by_key[gk],
        "o_shippriority": oprio_by_key[gk],
        "revenue": sums[gk],
    }
    # order by revenue desc, o_orderdate asc, limit 10 (ties broken by
    # orderkey asc for determinism of the fixture; the reference SQL leaves
    # ties unordered)
    order = np.lexsort((full["l_orderkey"], full["o_orderdate"], -full["revenue"]))
    top = order[:limit]
    top10 = {k: v[top] for k, v in full.items()}
    return full, top10


Q4_LO = G.days(1993, 7, 1)
Q4_HI = G.days(1993, 10, 1)     # date + interval '3' month
Q10_LO = G.days(1993, 10, 1)
Q10_HI = G.days(1994, 1, 1)     # date + interval '3' month


def q4(li, orders):
    """tpch_ref.py:117-140: orders in [1993-07-01, +3mo) with EXISTS a
    line l_orderkey = o_orderkey and l_commitdate < l_receiptdate;
    count(*) group by o_orderpriority, order by o_orderpriority.
    Returns dict o_orderpriority(str) -> count."""
    lmask = li["l_commitdate"] < li["l_receiptdate"]
    nkey = int(orders["o_orderkey"].max()) + 2
    has_late = np.zeros(nkey, dtype=bool)
    has_late[li["l_orderkey"][lmask]] = True
    omask = ((orders["o_orderdate"] >= Q4_LO)
             & (orders["o_orderdate"] < Q4_HI)
             & has_late[orders["o_orderkey"]])
    pr = orders["o_orderpriority"][omask]
    counts = np.bincount(pr, minlength=len(G.ORDERPRIORITY))
    return {G.ORDERPRIORITY[i]: int(counts[i])
            for i in range(len(G.ORDERPRIORITY)) if counts[i]}


def q18(li, orders, customer, limit=100):
    """tpch_ref.py:544-580: orders whose lines sum(l_quantity) > 300,
    joined to customer; group by (c_name, c_custkey, o_orderkey,
    o_orderdate, o_totalprice) sum(l_quantity); order by o_totalprice
    desc, o_orderdate asc, limit 100. The group key includes o_orderkey,
    so each group IS one qualifying order. Ties beyond the reference's
    ordering broken by o_orderkey asc for fixture determinism."""
    nkey = int(orders["o_orderkey"].max()) + 2
    qty_by_key = np.bincount(li["l_orderkey"], weights=li["l_quantity"],
                             minlength=nkey)
    ok = orders["o_orderkey"]
    qual = qty_by_key[ok] > 300.0
    rows = np.nonzero(qual)[0]
    cust = orders["o_custkey"][rows]
    names = (customer["c_name"][cust - 1] if "c_name" in customer
             else np.array(["Customer#%09d" % c for c in cust],
                           dtype=object))
    out = {
        "c_name": names,
        "c_custkey": cust,
        "o_orderkey": ok[rows],
        "o_orderdate": orders["o_orderdate"][rows],
        "o_totalprice": orders["o_totalprice"][rows],
        "sum_qty": qty_by_key[ok[rows]],
    }
    order = np.lexsort((out["o_orderkey"], out["o_orderdate"],
                        -out["o_totalprice"]))
    top = order[:limit]
    return {k: v[top] for k, v in out.items()}


def q10(li, orders, customer, nation, limit=20):
    """tpch_ref.py:306-342: returned lines (l_returnflag = 'R') of orders
    in [1993-10-01, +3mo), revenue per customer with the customer's
    nation name and string attributes; order by revenue desc limit 20.
    The 7-column group key is functionally dependent on c_custkey.
    Ties broken by c_custkey asc for fixture determinism."""
    rflag_r = G.RETURNFLAG.index("R")
    omask = ((orders["o_orderdate"] >= Q10_LO)
             & (orders["o_orderdate"] < Q10_HI))
    nkey = int(orders["o_orderkey"].max()) + 2
    order_cust = np.zeros(nkey, dtype=np.int64)
    order_cust[orders["o_orderkey"][omask]] = orders["o_custkey"][omask]
    lmask = (li["l_returnflag"] == rflag_r) & \
        (order_cust[li["l_orderkey"]] > 0)
    cust = order_cust[li["l_orderkey"][lmask]]
    rev = li["l_extendedprice"][lmask] * (1.0 - li["l_discount"][lmask])
    ncust = int(customer["c_custkey"].max()) + 2
    rev_by_cust = np.bincount(cust, weights=rev, minlength=ncust)
    ck = np.nonzero(rev_by_cust > 0)[0]
    row = ck - 1                                  # custkey is dense 1..N
    out = {
        "c_custkey": ck.astype(np.int64),
        "revenue": rev_by_cust[ck],
        "c_acctbal": customer["c_acctbal"][row],
        "n_name": nation["n_name"][customer["c_nationkey"][row]],
    }
    for c in ("c_name", "c_address", "c_phone", "c_comment"):
        if c in customer:
            out[c] = customer[c][row]
    order = np.lexsort((out["c_custkey"], -out["revenue"]))
    top = order[:limit]
    return {k: v[top] for k, v in out.items()}


Q12_LO = G.days(1994, 1, 1)
Q12_HI = G.days(1995, 1, 1)     # + interval '1' year
Q14_LO = G.days(1995, 9, 1)
Q14_HI = G.days(1995, 10, 1)    # + interval '1' month


def q12(li, orders):
    """tpch_ref.py:376-407: lines with shipmode in (MAIL, SHIP),
    ship < commit < receipt, receipt in [1994-01-01, +1y), joined to
    orders; per shipmode count lines with HIGH (1-URGENT/2-HIGH) vs low
    priority. Returns dict shipmode(str) -> (high_count, low_count),
    shipmode ascending."""
    mail, ship_m = G.SHIPMODE.index("MAIL"), G.SHIPMODE.index("SHIP")
    m = (np.isin(li["l_shipmode"], [mail, ship_m])
         & (li["l_commitdate"] < li["l_receiptdate"])
         & (li["l_shipdate"] < li["l_commitdate"])
         & (li["l_receiptdate"] >= Q12_LO)
         & (li["l_receiptdate"] < Q12_HI))
    nkey = int(orders["o_orderkey"].max()) + 2
    prio_by_key = np.full(nkey, -1, dtype=np.int8)
    prio_by_key[orders["o_orderkey"]] = orders["o_orderpriority"]
    pr = prio_by_key[li["l_orderkey"][m]]
    sm = li["l_shipmode"][m]
    assert (pr >= 0).all(), "lineitem orderkey missing from orders"
    high = pr <= 1                  # codes 0/1 = 1-URGENT / 2-HIGH
    out = {}
    for code in sorted({mail, ship_m}):
        sel = sm == code
        out[G.SHIPMODE[code]] = (int((sel & high).sum()),
                                 int((sel & ~high).sum()))
    return out


def q14(li, part):
    """tpch_ref.py:434-450: promo revenue ratio over the shipdate month
    window, lineitem x part on partkey. Returns the percentage."""
    m = (li["l_shipdate"] >= Q14_LO) & (li["l_shipdate"] < Q14_HI)
    pk = li["l_partkey"][m]
    promo_flag = (part["p_type"] // 25) == G.PTYPE_PROMO_SYL1
    is_promo = promo_flag[pk - 1]        # partkey dense 1..N
    rev = li["l_extendedprice"][m] * (1.0 - li["l_discount"][m])
    total = rev.sum()
    return 100.0 * rev[is_promo].sum() / total if total else 0.0


Q7_LO = G.days(1995, 1, 1)
Q7_HI = G.days(1996, 12, 31)    # BETWEEN is inclusive
Y1996 = G.days(1996, 1, 1)


def q7(li, orders, customer, supplier, nation):
    """tpch_ref.py:185-227: FRANCE<->GERMANY shipping volume by
    (supp_nation, cust_nation, year of l_shipdate), shipdate BETWEEN
    1995-01-01 AND 1996-12-31 (inclusive). Returns dict
    (supp_nation, cust_nation, year) -> revenue, sorted key order."""
    names = list(nation["n_name"])
    fr, de = names.index("FRANCE"), names.index("GERMANY")
    cust_nat = np.full(int(customer["c_custkey"].max()) + 2, -1,
                       dtype=np.int32)
    cust_nat[customer["c_custkey"]] = customer["c_nationkey"]
    nkey = int(orders["o_orderkey"].max()) + 2
    order_cnat = np.full(nkey, -1, dtype=np.int32)
    order_cnat[orders["o_orderkey"]] = cust_nat[orders["o_custkey"]]
    supp_nat = np.full(int(supplier["s_suppkey"].max()) + 2, -1,
                       dtype=np.int32)
    supp_nat[supplier["s_suppkey"]] = supplier["s_nationkey"]

    m = (li["l_shipdate"] >= Q7_LO) & (li["l_shipdate"] <= Q7_HI)
    cn = order_cnat[li["l_orderkey"][m]]
    sn = supp_nat[li["l_suppkey"][m]]
    pair = ((sn == fr) & (cn == de)) | ((sn == de) & (cn == fr))
    rev = (li["l_extendedprice"][m] * (1.0 - li["l_discount"][m]))[pair]
    year = np.where(li["l_shipdate"][m][pair] >= Y1996, 1996, 1995)
    snp, cnp = sn[pair], cn[pair]
    out = {}
    for s, c in ((fr, de), (de, fr)):
        for y in (1995, 1996):
            v = rev[(snp == s) & (cnp == c) & (year == y)].sum()
            out[(names[s], names[c], y)] = float(v)
    return dict(sorted(out.items()))


def q8(li, orders, customer, supplier, part, nation, region):
    """tpch_ref.py:229-268: BRAZIL market share among AMERICA-region
    customers for p_type = 'ECONOMY ANODIZED STEEL', by order year.
    Returns dict year -> mkt_share."""
    america = G.REGIONS.index("AMERICA")
    nat_in_america = nation["n_regionkey"] == america
    # p_type code for 'ECONOMY ANODIZED STEEL'
    code = ((G.PTYPE_SYL1.index("ECONOMY") * 5 +
             G.PTYPE_SYL2.index("ANODIZED")) * 5 +
            G.PTYPE_SYL3.index("STEEL"))
    part_ok = np.zeros(int(part["p_partkey"].max()) + 2, dtype=bool)
    part_ok[part["p_partkey"][part["p_type"] == code]] = True

    cust_nat = np.full(int(customer["c_custkey"].max()) + 2, -1,
                       dtype=np.int32)
    cust_nat[customer["c_custkey"]] = customer["c_nationkey"]
    omask = ((orders["o_orderdate"] >= Q7_LO)
             & (orders["o_orderdate"] <= Q7_HI))
    ocn = cust_nat[orders["o_custkey"][omask]]
    okeep = nat_in_america[np.clip(ocn, 0, 24)] & (ocn >= 0)
    nkey = int(orders["o_orderkey"].max()) + 2
    order_year = np.zeros(nkey, dtype=np.int16)
    ok = orders["o_orderkey"][omask][okeep]
    order_year[ok] = np.where(
        orders["o_orderdate"][omask][okeep] >= Y1996, 1996, 1995)
    supp_nat = np.full(int(supplier["s_suppkey"].max()) + 2, -1,
                       dtype=np.int32)
    supp_nat[supplier["s_suppkey"]] = supplier["s_nationkey"]

    m = part_ok[li["l_partkey"]] & (order_year[li["l_orderkey"]] > 0)
    rev = li["l_extendedprice"][m] * (1.0 - li["l_discount"][m])
    yr = order_year[li["l_orderkey"][m]]
    sn = supp_nat[li["l_suppkey"][m]]
    brazil = list(nation["n_name"]).index("BRAZIL")
    out = {}
    for y in (1995, 1996):
        sel = yr == y
        tot = rev[sel].sum()
        br = rev[sel & (sn == brazil)].sum()
        out[y] = float(br / tot) if tot else 0.0
    return out


def q17(li, part, brand_code=12, container_code=17):
    """tpch_ref.py:522-542: small-quantity ('below 20% of that part's
    average quantity') revenue for one brand+container, / 7.0. Codes
    default to arbitrary members of the generated code spaces (the
    reference's 'Brand#23' / 'MED BOX' are dbgen strings; selection
    logic, not the label, is what is under test). Returns the scalar."""
    sel_part = (part["p_brand"] == brand_code) & \
        (part["p_container"] == container_code)
    nkey = int(part["p_partkey"].max()) + 2
    qty_sum = np.bincount(li["l_partkey"], weights=li["l_quantity"],
                          minlength=nkey)
    qty_cnt = np.bincount(li["l_partkey"], minlength=nkey)
    with np.errstate(invalid="ignore", divide="ignore"):
        thr = 0.2 * qty_sum / qty_cnt
    ok = np.zeros(nkey, dtype=bool)
    ok[part["p_partkey"][sel_part]] = True
    m = ok[li["l_partkey"]] & \
        (li["l_quantity"] < thr[li["l_partkey"]])
    return float(li["l_extendedprice"][m].sum() / 7.0)


Q15_LO = G.days(1996, 1, 1)
Q15_HI = G.days(1996, 4, 1)     # + interval '3' month


def q15(li, supplier):
    """tpch_ref.py:452-485: revenue per supplier over the 3-month window;
    suppliers whose revenue equals the maximum. Returns (suppkeys asc,
    max_revenue)."""
    m = (li["l_shipdate"] >= Q15_LO) & (li["l_shipdate"] < Q15_HI)
    nkey = int(supplier["s_suppkey"].max()) + 2
    rev = np.bincount(li["l_suppkey"][m],
                      weights=(li["l_extendedprice"][m]
                               * (1.0 - li["l_discount"][m])),
                      minlength=nkey)
    mx = rev.max()
    winners = np.nonzero(rev == mx)[0]
    return winners.astype(np.int64), float(mx)


# Q19 branch constants (tpch_ref.py:582-620), resolved to dictionary
# codes via the committed code tables (oracle.tpch_gen)
Q19_BRANCHES = [
    ("Brand#12", ["SM CASE", "SM BOX", "SM PACK", "SM PKG"], 1, 11, 1, 5),
    ("Brand#23", ["MED BAG", "MED BOX", "MED PKG", "MED PACK"], 10, 20, 1, 10),
    ("Brand#34", ["LG CASE", "LG BOX", "LG PACK", "LG PKG"], 20, 30, 1, 15),
]


def q19(li, part):
    """tpch_ref.py:582-620: sum of revenue over the OR of three
    brand/container/quantity/size/shipmode/shipinstruct branches."""
    air = [G.SHIPMODE.index("AIR"), G.SHIPMODE.index("REG AIR")]
    deliver = G.SHIPINSTRUCT.index("DELIVER IN PERSON")
    nkey = int(part["p_partkey"].max()) + 2
    brand = np.zeros(nkey, dtype=np.int16)
    cont = np.zeros(nkey, dtype=np.int16)
    size = np.zeros(nkey, dtype=np.int16)
    brand[part["p_partkey"]] = part["p_brand"]
    cont[part["p_partkey"]] = part["p_container"]
    size[part["p_partkey"]] = part["p_size"]
    lb = brand[li["l_partkey"]]
    lc = cont[li["l_partkey"]]
    ls = size[li["l_partkey"]]
    common = (np.isin(li["l_shipmode"], air)
              & (li["l_shipinstruct"] == deliver))
    m = np.zeros(len(lb), dtype=bool)
    for bname, conts, qlo, qhi, slo, shi in Q19_BRANCHES:
        bc = G.brand_code(bname)
        cc = [G.container_code(c) for c in conts]
        m |= ((lb == bc) & np.isin(lc, cc)
              & (li["l_quantity"] >= qlo) & (li["l_quantity"] <= qhi)
              & (ls >= slo) & (ls <= shi))
    m &= common
    return float((li["l_extendedprice"][m]
                  * (1.0 - li["l_discount"][m])).sum())


def q2(part, supplier, partsupp, nation, region, limit=100):
    """tpch_ref.py:40-86: EUROPE suppliers offering size-15 '%BRASS'
    parts at that part's EUROPE-minimum supply cost; order by s_acctbal
    desc, n_name, s_name (== s_suppkey order), p_partkey, limit 100.
    Returns dict of arrays (p_partkey, s_suppkey, s_acctbal, n_name,
    ps_supplycost)."""
    europe = G.REGIONS.index("EUROPE")
    nat_eu = nation["n_regionkey"] == europe
    supp_nat = np.full(int(supplier["s_suppkey"].max()) + 2, -1,
                       dtype=np.int32)
    supp_nat[supplier["s_suppkey"]] = supplier["s_nationkey"]
    ps_sn = supp_nat[partsupp["ps_suppkey"]]
    eu = (ps_sn >= 0) & nat_eu[np.clip(ps_sn, 0, 24)]
    pk = partsupp["ps_partkey"][eu]
    cost = partsupp["ps_supplycost"][eu]
    nkey = int(part["p_partkey"].max()) + 2
    min_cost = np.full(nkey, np.inf)
    np.minimum.at(min_cost, pk, cost)
    part_ok = np.zeros(nkey, dtype=bool)
    sel = (part["p_size"] == 15) & (part["p_type"] % 5 ==
                                    G.PTYPE_SYL3.index("BRASS"))
    part_ok[part["p_partkey"][sel]] = True
    win = part_ok[pk] & (cost == min_cost[pk])
    sk = partsupp["ps_suppkey"][eu][win]
    row = sk - 1
    names = list(nation["n_name"])
    out = {
        "p_partkey": pk[win],
        "s_suppkey": sk,
        "s_acctbal": supplier["s_acctbal"][row],
        "n_name": np.array([names[k] for k in
                            supplier["s_nationkey"][row]], dtype=object),
        "ps_supplycost": cost[win],
    }
    nrank = np.array([sorted(names).index(n) for n in out["n_name"]])
    order = np.lexsort((out["p_partkey"], out["s_suppkey"], nrank,
                        -out["s_acctbal"]))
    top = order[:limit]
    return {k: v[top] for k, v in out.items()}


def q11(partsupp, supplier, nation, fraction=0.0001):
    """tpch_ref.py:344-374: GERMANY suppliers' ps value per part,
    HAVING value > fraction * total; order by value desc. Returns
    (partkeys, values) sorted."""
    germany = list(nation["n_name"]).index("GERMANY")
    supp_nat = np.full(int(supplier["s_suppkey"].max()) + 2, -1,
                       dtype=np.int32)
    supp_nat[supplier["s_suppkey"]] = supplier["s_nationkey"]
    m = supp_nat[partsupp["ps_suppkey"]] == germany
    pk = partsupp["ps_partkey"][m]
    val = (partsupp["ps_supplycost"][m]
           * partsupp["ps_availqty"][m].astype(np.float64))
    nkey = int(partsupp["ps_partkey"].max()) + 2
    per = np.bincount(pk, weights=val, minlength=nkey)
    thr = val.sum() * fraction
    win = np.nonzero(per > thr)[0]
    order = np.lexsort((win, -per[win]))
    return win[order].astype(np.int64), per[win][order]


def q20(li, part, partsupp, supplier, nation):
    """tpch_ref.py:622-662: CANADA suppliers holding availqty > half the
    1994 shipped quantity of a 'forest%' part they supply. Returns
    sorted s_suppkey array (s_name order == suppkey order)."""
    forest = np.zeros(int(part["p_partkey"].max()) + 2, dtype=bool)
    forest[part["p_partkey"][part["p_name1"] == G.P_NAME_FOREST]] = True
    m = (forest[li["l_partkey"]]
         & (li["l_shipdate"] >= G.Q5_LO) & (li["l_shipdate"] < G.Q5_HI))
    S = int(max(li["l_suppkey"].max(), partsupp["ps_suppkey"].max())) + 1
    key = li["l_partkey"][m] * S + li["l_suppkey"][m]
    kk, inv = np.unique(key, return_inverse=True)
    qty = np.bincount(inv, weights=li["l_quantity"][m])
    pm = forest[partsupp["ps_partkey"]]
    pskey = partsupp["ps_partkey"][pm] * S + partsupp["ps_suppkey"][pm]
    pos = np.searchsorted(kk, pskey)
    have = (pos < len(kk)) & (kk[np.minimum(pos, len(kk) - 1)] == pskey)
    thr = np.zeros(len(pskey))
    thr[have] = 0.5 * qty[pos[have]]
    # SQL: availqty > (scalar subquery); an EMPTY subquery yields NULL
    # and the comparison is false — rows with no 1994 shipments do NOT
    # qualify (DuckDB semantics, tpch_ref.py:646-653)
    ok = have & (partsupp["ps_availqty"][pm] > thr)
    sk = np.unique(partsupp["ps_suppkey"][pm][ok])
    canada = list(nation["n_name"]).index("CANADA")
    sn = supplier["s_nationkey"][sk - 1]
    return sk[sn == canada].astype(np.int64)


def q9(li, orders, supplier, part, partsupp, nation):
    """tpch_ref.py (do_9): profit = revenue - supplycost*qty for 'green'
    parts, by (supplier nation, order year); order by nation asc, year
    desc. Returns dict (n_name, year) -> profit."""
    green = np.zeros(int(part["p_partkey"].max()) + 2, dtype=bool)
    green[part["p_partkey"][part["p_name_green"] == 1]] = True
    S = int(max(li["l_suppkey"].max(), partsupp["ps_suppkey"].max())) + 1
    pskey = partsupp["ps_partkey"] * S + partsupp["ps_suppkey"]
    order_ps = np.argsort(pskey)
    m = green[li["l_partkey"]]
    lkey = li["l_partkey"][m] * S + li["l_suppkey"][m]
    pos = np.searchsorted(pskey, lkey, sorter=order_ps)
    pos = np.minimum(pos, len(pskey) - 1)
    have = pskey[order_ps[pos]] == lkey
    # inner-join semantics: only (part, supp) pairs present in partsupp
    # (the generator maintains the spec invariant, so this keeps all)
    idx = np.nonzero(m)[0][have]
    m = np.zeros(len(li["l_partkey"]), dtype=bool)
    m[idx] = True
    cost = partsupp["ps_supplycost"][order_ps[pos][have]]
    amount = (li["l_extendedprice"][m] * (1.0 - li["l_discount"][m])
              - cost * li["l_quantity"][m])
    nkey = int(orders["o_orderkey"].max()) + 2
    oyear = np.zeros(nkey, dtype=np.int16)
    od = orders["o_orderdate"]
    oyear[orders["o_orderkey"]] = 1970 + (
        np.asarray(od, dtype="datetime64[D]").astype(
            "datetime64[Y]").astype(np.int64)).astype(np.int16)
    yr = oyear[li["l_orderkey"][m]]
    sn = supplier["s_nationkey"][li["l_suppkey"][m] - 1]
    names = list(nation["n_name"])
    out = {}
    for nk in range(25):
        for y in np.unique(yr):
            v = amount[(sn == nk) & (yr == y)].sum()
            if v != 0.0:
                out[(names[nk], int(y))] = float(v)
    return out


def q13(orders, customer):
    """tpch_ref.py:409-432: distribution of per-customer order counts,
    orders with the 'special requests' comment excluded (text-boundary
    flag), customers with zero orders included. Returns dict
    c_count -> custdist."""
    keep = orders["o_comment_special"] == 0
    ncust = len(customer["c_custkey"])
    cnt = np.bincount(orders["o_custkey"][keep], minlength=ncust + 2)
    per_cust = cnt[1:ncust + 1]
    counts = np.bincount(per_cust)
    return {int(c): int(v) for c, v in enumerate(counts) if v}


def q16(part, partsupp, supplier):
    """tpch_ref.py:487-520: DISTINCT supplier count per (brand, type,
    size) over filtered parts, complained suppliers excluded. Returns
    dict (brand, type, size) -> count, ordered by count desc then key."""
    med_pol = (part["p_type"] // 25 == G.PTYPE_SYL1.index("MEDIUM")) & \
        ((part["p_type"] // 5) % 5 == G.PTYPE_SYL2.index("POLISHED"))
    sel = ((part["p_brand"] != G.brand_code("Brand#45"))
           & ~med_pol
           & np.isin(part["p_size"], [49, 14, 23, 45, 19, 3, 36, 9]))
    ok = np.zeros(int(part["p_partkey"].max()) + 2, dtype=bool)
    ok[part["p_partkey"][sel]] = True
    attr = {}
    for k, b, t, z in zip(part["p_partkey"], part["p_brand"],
                          part["p_type"], part["p_size"]):
        attr[int(k)] = (int(b), int(t), int(z))
    bad_supp = np.zeros(int(supplier["s_suppkey"].max()) + 2, dtype=bool)
    bad_supp[supplier["s_suppkey"][
        supplier["s_comment_complaints"] == 1]] = True
    m = ok[partsupp["ps_partkey"]] & ~bad_supp[partsupp["ps_suppkey"]]
    seen = set()
    for pk, sk in zip(partsupp["ps_partkey"][m],
                      partsupp["ps_suppkey"][m]):
        seen.add(attr[int(pk)] + (int(sk),))
    out = {}
    for b, t, z, sk in seen:
        out[(b, t, z)] = out.get((b, t, z), 0) + 1
    return dict(sorted(out.items(), key=lambda kv: (-kv[1], kv[0])))


def q21(li, orders, supplier, nation, limit=100):
    """tpch_ref.py (do_21): suppliers (SAUDI ARABIA) that were the ONLY
    late supplier on a multi-supplier 'F' order. Returns dict
    s_suppkey -> numwait, ordered by numwait desc then suppkey."""
    nkey = int(orders["o_orderkey"].max()) + 2
    status_f = np.zeros(nkey, dtype=bool)
    status_f[orders["o_orderkey"][orders["o_orderstatus"] == 0]] = True
    S = int(li["l_suppkey"].max()) + 1
    # distinct suppliers per order (all lines)
    all_pairs = np.unique(li["l_orderkey"] * S + li["l_suppkey"])
    nsupp_all = np.bincount((all_pairs // S).astype(np.int64),
                            minlength=nkey)
    # distinct suppliers per order among LATE lines
    late = li["l_receiptdate"] > li["l_commitdate"]
    late_pairs = np.unique(li["l_orderkey"][late] * S
                           + li["l_suppkey"][late])
    lo = (late_pairs // S).astype(np.int64)
    nsupp_late = np.bincount(lo, minlength=nkey)
    # qualifying: F order, >=2 suppliers total, exactly 1 late supplier
    qual_orders = status_f & (nsupp_all >= 2) & (nsupp_late == 1)
    qual = qual_orders[lo]
    wait_supp = (late_pairs % S)[qual]
    saudi = list(nation["n_name"]).index("SAUDI ARABIA")
    sn = supplier["s_nationkey"][wait_supp - 1]
    wait_supp = wait_supp[sn == saudi]
    cnt = np.bincount(wait_supp, minlength=S + 1)
    sk = np.nonzero(cnt)[0]
    order = np.lexsort((sk, -cnt[sk]))
    top = order[:limit]
    return {int(sk[i]): int(cnt[sk[i]]) for i in top}


def q22(customer, orders):
    """tpch_ref.py (do_22): country code = first two phone digits
    (= 10 + c_nationkey in the generator, spec 4.2.2.9); codes
    {13,31,23,29,30,18,17}, acctbal above the positive average of those
    codes, and NO orders. Returns dict cntrycode -> (numcust, total)."""
    codes = np.array([13, 31, 23, 29, 30, 18, 17])
    cc = 10 + customer["c_nationkey"]
    in_list = np.isin(cc, codes)
    pos = in_list & (customer["c_acctbal"] > 0.0)
    avg = customer["c_acctbal"][pos].mean()
    ncust = int(customer["c_custkey"].max()) + 2
    has_order = np.zeros(ncust, dtype=bool)
    has_order[orders["o_custkey"]] = True
    m = in_list & (customer["c_acctbal"] > avg) & \
        ~has_order[customer["c_custkey"]]
    out = {}
    for code in sorted(codes):
        sel = m & (cc == code)
        if sel.any():
            out[str(code)] = (int(sel.sum()),
                              float(customer["c_acctbal"][sel].sum()))
    return out


def q5(li, orders, customer, supplier, nation, region):
    """tpch_ref.py:142-169: 6-table chain, r_name='ASIA',
    o_orderdate in [1994-01-01, 1995-01-01), extra equi-predicate
    c_nationkey = s_nationkey; group by n_name; order by revenue desc.
    Returns dict n_name -> revenue (sorted desc)."""
    asia = G.REGIONS.index("ASIA")
    nat_in_asia = nation["n_regionkey"] == asia  # indexed by nationkey 0..24

    cust_nat = np.full(int(customer["c_custkey"].max()) + 2, -1, dtype=np.int32)
    cust_nat[customer["c_custkey"]] = customer["c_nationkey"]

    omask = (orders["o_orderdate"] >= G.Q5_LO) & (orders["o_orderdate"] < G.Q5_HI)
    okeys = orders["o_orderkey"][omask]
    ocust = orders["o_custkey"][omask]
    nkey = int(orders["o_orderkey"].max()) + 2
    order_cnat = np.full(nkey, -1, dtype=np.int32)
    ocn = cust_nat[ocust]
    keep = (ocn >= 0) & nat_in_asia[np.clip(ocn, 0, 24)]
    order_cnat[okeys[keep]] = ocn[keep]

    supp_nat = np.full(int(supplier["s_suppkey"].max()) + 2, -1, dtype=np.int32)
    supp_nat[supplier["s_suppkey"]] = supplier["s_nationkey"]

    lcn = order_cnat[li["l_orderkey"]]
    lsn = supp_nat[li["l_suppkey"]]
    m = (lcn >= 0) & (lcn == lsn)
    rev = li["l_extendedprice"][m] * (1.0 - li["l_discount"][m])
    by_nat = np.bincount(lcn[m], weights=rev, minlength=25)
    names = [n for n, _ in G.NATIONS]
    out = [(names[i], by_nat[i]) for i in range(25) if nat_in_asia[i]]
    out.sort(key=lambda t: -t[1])
    return out
