"""Device query pipelines (TPC-H Q1/Q6/Q3) composed from ops.

These mirror how the reference lowers each query:
Q1 = scan -> fused filter+partial group-by (map side) -> tiny final agg;
Q3 = filter customer/orders -> hash-join build x2 -> probe lineitem ->
     group-by orderkey -> top-k on host;
Q6 = scan -> fused filter+sum.

Date constants are days-since-1970 (Arrow date32); the literals are the
reference SQL's (apps/tpc-h/tpch_ref.py)."""
import ctypes

import numpy as np

from . import ops
from .shim import DevBuffer, DevColumn, c_u64, c_vp, call

Q1_CUTOFF = 10471      # date '1998-12-01' - interval '90' day = 1998-09-02
Q3_DATE = 9204         # 1995-03-15
Q6_LO = 8766           # 1994-01-01
Q6_HI = 9131           # 1995-01-01

RETURNFLAG = ["A", "N", "R"]
LINESTATUS = ["F", "O"]
MKT_BUILDING = 1       # code of 'BUILDING' in sorted segment dictionary


def _zero48():
    b = DevBuffer(48 * 8)
    call("qk_dmemset", b.ptr, 0, c_u64(48 * 8))
    return b


def q1_partials_device(cols, stream=None, cutoff=Q1_CUTOFF, acc=None):
    """One fused pass over staged lineitem columns; accumulates into (and
    returns) `acc` (DevBuffer[48 f64])."""
    if acc is None:
        acc = _zero48()
    n = cols["l_shipdate"].n
    ops.q1_agg(n, cols["l_shipdate"], cols["l_quantity"],
               cols["l_extendedprice"], cols["l_discount"], cols["l_tax"],
               cols["l_returnflag"], cols["l_linestatus"], cutoff, acc,
               stream)
    return acc


def q1_finalize(partials):
    """partials: (6,6) [sum_qty,sum_base,sum_disc_price,sum_charge,sum_disc,
    count] -> final Q1 table (dict of numpy arrays), ordered by
    (l_returnflag, l_linestatus) = ascending group id."""
    p = np.asarray(partials, dtype=np.float64).reshape(6, 6)
    gids = np.nonzero(p[:, 5] > 0)[0]
    p = p[gids]
    cnt = p[:, 5]
    return {
        "l_returnflag": np.array([RETURNFLAG[g // 2] for g in gids]),
        "l_linestatus": np.array([LINESTATUS[g % 2] for g in gids]),
        "sum_qty": p[:, 0],
        "sum_base_price": p[:, 1],
        "sum_disc_price": p[:, 2],
        "sum_charge": p[:, 3],
        "avg_qty": p[:, 0] / cnt,
        "avg_price": p[:, 1] / cnt,
        "avg_disc": p[:, 4] / cnt,
        "count_order": cnt.astype(np.int64),
    }


def q1(cols, stream=None, cutoff=Q1_CUTOFF):
    acc = q1_partials_device(cols, stream, cutoff)
    if stream:
        stream.sync()
    out = q1_finalize(ops.q1_read_partials(acc))
    acc.free()
    return out


def q6(cols, stream=None):
    acc = DevBuffer(2 * 8)
    call("qk_dmemset", acc.ptr, 0, c_u64(16))
    n = cols["l_shipdate"].n
    # bounds are the reference SQL's literal fp64 expressions
    ops.q6_agg(n, cols["l_shipdate"], cols["l_quantity"],
               cols["l_extendedprice"], cols["l_discount"],
               Q6_LO, Q6_HI, 0.06 - 0.01, 0.06 + 0.01, 24.0, acc, stream)
    if stream:
        stream.sync()
    host = np.zeros(2, dtype=np.float64)
    call("qk_d2h", host.ctypes.data_as(c_vp), acc.ptr, c_u64(16))
    acc.free()
    return {"revenue": host[0], "rows_passed": int(host[1])}


def q3(li_cols, ord_cols, cust_cols, stream=None, limit=10):
    """Full device Q3: returns (full_groups dict, top10 dict).

    Stages (mirroring the reference plan, logical.py:447-506 with both
    build sides = the small filtered tables, as its join-order pass picks):
      1. customer filter mktsegment == BUILDING -> semi-join set
      2. orders filter orderdate < Q3_DATE, semi customer -> build table
      3. lineitem filter shipdate > Q3_DATE -> probe -> matched pairs
      4. group-by l_orderkey sum(revenue); attach o_orderdate/shippriority
      5. host: order by revenue desc, orderdate asc, limit 10
    """
    st = stream
    # 1. customer: filter to BUILDING, build table on c_custkey
    cidx, ncust = ops.filter_col(cust_cols["c_mktsegment"], ops.EQ,
                                 MKT_BUILDING, st)
    ckeys = cust_cols["c_custkey"].gather(cidx, ncust, st)
    cust_table = ops.JoinTable(max(16, ncust), st)
    if ncust:
        cust_table.build(ckeys)

    # 2. orders: filter date, then semi-probe against customers
    oidx, nord = ops.filter_col(ord_cols["o_orderdate"], ops.LT, Q3_DATE, st)
    o_custkey = ord_cols["o_custkey"].gather(oidx, nord, st)
    semi_idx, _, nsemi = cust_table.probe(o_custkey, mode=1, n=nord)
    # rows of the FILTERED orders that survive the semi join
    o_orderkey_f = ord_cols["o_orderkey"].gather(oidx, nord, st)
    o_orderdate_f = ord_cols["o_orderdate"].gather(oidx, nord, st)
    o_shipprio_f = ord_cols["o_shippriority"].gather(oidx, nord, st)
    b_orderkey = o_orderkey_f.gather(semi_idx, nsemi, st)
    b_orderdate = o_orderdate_f.gather(semi_idx, nsemi, st)
    b_shipprio = o_shipprio_f.gather(semi_idx, nsemi, st)

    ord_table = ops.JoinTable(max(16, nsemi), st)
    if nsemi:
        ord_table.build(b_orderkey)

    # 3. lineitem: filter shipdate, probe orders
    lidx, nli = ops.filter_col(li_cols["l_shipdate"], ops.GT, Q3_DATE, st)
    l_orderkey_f = li_cols["l_orderkey"].gather(lidx, nli, st)
    pidx, bidx, nmatch = ord_table.probe(l_orderkey_f, mode=0, n=nli)

    # 4. revenue per match -> group-by orderkey
    l_price_f = li_cols["l_extendedprice"].gather(lidx, nli, st)
    l_disc_f = li_cols["l_discount"].gather(lidx, nli, st)
    m_price = l_price_f.gather(pidx, nmatch, st)
    m_disc = l_disc_f.gather(pidx, nmatch, st)
    m_orderkey = l_orderkey_f.gather(pidx, nmatch, st)
    # revenue = price * (1 - disc) per matched row (the reference computes
    # the product per row before summing, tpch.py:151)
    rev = _mul_1md(m_price, m_disc, st)

    expected_groups = max(16, nmatch)
    gb = ops.GroupByI64(expected_groups, 1, st)
    gb.update(m_orderkey, [rev], nmatch)
    keys, sums = gb.extract()

    # attach o_orderdate/o_shippriority by probing group keys against the
    # orders build state (keys are unique there)
    gkeys = DevColumn.from_numpy(keys)
    gp, gbi, ng = ord_table.probe(gkeys, mode=0, n=len(keys))
    gp_h = gp.to_numpy(ng)
    gbi_h = gbi.to_numpy(ng)
    odate_h = b_orderdate.to_numpy(nsemi)[gbi_h] if nsemi else np.empty(0, np.int32)
    oprio_h = b_shipprio.to_numpy(nsemi)[gbi_h] if nsemi else np.empty(0, np.int32)
    # gp maps each matched pair back to position in `keys`
    odate_full = np.zeros(len(keys), dtype=np.int32)
    oprio_full = np.zeros(len(keys), dtype=np.int32)
    odate_full[gp_h] = odate_h
    oprio_full[gp_h] = oprio_h

    full = {
        "l_orderkey": keys,
        "o_orderdate": odate_full,
        "o_shippriority": oprio_full,
        "revenue": sums[0],
    }
    order = np.lexsort((full["l_orderkey"], full["o_orderdate"],
                        -full["revenue"]))
    top = order[:limit]
    top10 = {k: v[top] for k, v in full.items()}

    for obj in (cidx, ckeys, oidx, o_custkey, semi_idx, o_orderkey_f,
                o_orderdate_f, o_shipprio_f, b_orderkey, b_orderdate,
                b_shipprio, lidx, l_orderkey_f, pidx, bidx, l_price_f,
                l_disc_f, m_price, m_disc, m_orderkey, rev, gkeys, gp, gbi):
        try:
            obj.free()
        except AttributeError:
            pass
    cust_table.free()
    ord_table.free()
    gb.free()
    return full, top10


def _mul_1md(price_col, disc_col, stream):
    """revenue = price * (1 - disc), elementwise on device."""
    from . import shim
    n = price_col.n
    out = DevColumn(np.float64, max(1, n))
    sh = stream.handle if stream else None
    shim.call("qk_mul_1md", sh, c_u64(n), price_col.ptr, disc_col.ptr,
              out.ptr)
    out.n = n
    return out
