"""GPU parity tests: every HIP kernel against the CPU oracle on the SAME
seeded inputs. Gates (north_star): bit-exact row sets for filter/join/
partition; fp64 aggregates within 1e-9 relative."""
import numpy as np
import pytest

from oracle import tpch_gen as G, queries as OQ, executors as OE

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def gpu():
    from quokka_amd import shim
    shim.init(0)
    return shim


@pytest.fixture(scope="module")
def data():
    return G.gen_all(0.01, 42)


def _stage_li(li):
    from quokka_amd import staging
    return staging.stage_columns(li)


# ---------- Q1 ----------------------------------------------------------

def test_q1_parity_sf001(gpu, data):
    from quokka_amd import queries as DQ
    cols = _stage_li(data["lineitem"])
    got = DQ.q1(cols)
    want = OQ.q1(data["lineitem"])
    assert list(got["l_returnflag"]) == list(want["l_returnflag"])
    assert list(got["l_linestatus"]) == list(want["l_linestatus"])
    assert np.array_equal(got["count_order"], want["count_order"])
    for c in ("sum_qty", "sum_base_price", "sum_disc_price", "sum_charge",
              "avg_qty", "avg_price", "avg_disc"):
        np.testing.assert_allclose(got[c], want[c], rtol=1e-9, err_msg=c)
    for c in cols.values():
        c.free()


def test_q1_parity_sf03_and_batched(gpu):
    """Larger size + multi-batch accumulation (executor-state semantics,
    sql_executors.py:587-590): two chunked calls == one pass == oracle."""
    from quokka_amd import queries as DQ, staging, ops
    li = G.gen_lineitem(0.3, 11)
    n = len(li["l_shipdate"])
    half = n // 2
    acc = None
    for lo, hi in ((0, half), (half, n)):
        chunk = {k: v[lo:hi] for k, v in li.items()}
        cols = staging.stage_columns(chunk, names=[
            "l_shipdate", "l_quantity", "l_extendedprice", "l_discount",
            "l_tax", "l_returnflag", "l_linestatus"])
        acc = DQ.q1_partials_device(cols, acc=acc)
        for c in cols.values():
            c.free()
    got = DQ.q1_finalize(ops.q1_read_partials(acc))
    acc.free()
    want = OQ.q1(li)
    assert np.array_equal(got["count_order"], want["count_order"])
    for c in ("sum_qty", "sum_base_price", "sum_disc_price", "sum_charge",
              "avg_qty", "avg_price", "avg_disc"):
        np.testing.assert_allclose(got[c], want[c], rtol=1e-9, err_msg=c)


def test_q1_edge_cases(gpu):
    from quokka_amd import queries as DQ, staging
    # boundary date inclusive, single row, odd row counts
    for n in (1, 2, 3, 255, 257):
        li = {
            "l_shipdate": np.full(n, G.Q1_CUTOFF, np.int32),
            "l_quantity": np.ones(n),
            "l_extendedprice": np.full(n, 10.0),
            "l_discount": np.zeros(n),
            "l_tax": np.zeros(n),
            "l_returnflag": np.zeros(n, np.uint8),
            "l_linestatus": np.zeros(n, np.uint8),
        }
        li["l_shipdate"][n // 2] = G.Q1_CUTOFF + 1   # one excluded row
        cols = staging.stage_columns(li)
        got = DQ.q1(cols)
        want = OQ.q1(li)
        assert np.array_equal(got["count_order"], want["count_order"]), n
        for c in cols.values():
            c.free()
    # all rows filtered out -> empty result
    li = {k: v[:4] for k, v in li.items()}
    li["l_shipdate"] = np.full(4, G.Q1_CUTOFF + 9, np.int32)
    cols = staging.stage_columns(li)
    got = DQ.q1(cols)
    assert len(got["count_order"]) == 0
    for c in cols.values():
        c.free()


# ---------- Q6 ----------------------------------------------------------

def test_q6_parity(gpu, data):
    from quokka_amd import queries as DQ
    cols = _stage_li(data["lineitem"])
    got = DQ.q6(cols)
    want = OQ.q6(data["lineitem"])
    assert got["rows_passed"] == want["rows_passed"]
    np.testing.assert_allclose(got["revenue"], want["revenue"], rtol=1e-9)
    for c in cols.values():
        c.free()


# ---------- filter / gather --------------------------------------------

def test_filter_ordered_rowset_bitexact(gpu, data):
    from quokka_amd import ops, shim
    ship = data["lineitem"]["l_shipdate"]
    col = shim.DevColumn.from_numpy(ship)
    idx, n = ops.filter_col(col, ops.GT, G.Q3_DATE)
    got = idx.to_numpy(n)
    want = np.nonzero(ship > G.Q3_DATE)[0]
    assert np.array_equal(got, want)  # bit-exact AND ordered
    col.free()
    idx.free()


def test_filter_edges(gpu):
    from quokka_amd import ops, shim
    for arr, op, v, predfn in [
        (np.array([], np.int32), ops.LT, 5, None),
        (np.array([5], np.int32), ops.EQ, 5, lambda a: a == 5),
        (np.arange(1000, dtype=np.int32), ops.GE, 1000, lambda a: a >= 1000),
        (np.arange(100000, dtype=np.int32) % 7, ops.NE, 0, lambda a: a != 0),
    ]:
        col = shim.DevColumn.from_numpy(arr)
        idx, n = ops.filter_col(col, op, v)
        if predfn is None:
            assert n == 0
        else:
            assert np.array_equal(idx.to_numpy(n), np.nonzero(predfn(arr))[0])
        col.free()
        idx.free()


def test_gather_roundtrip(gpu):
    from quokka_amd import ops, shim
    rng = np.random.default_rng(5)
    src = rng.random(10000)
    sel = rng.integers(0, 10000, 3000).astype(np.uint32)
    scol = shim.DevColumn.from_numpy(src)
    icol = shim.DevColumn.from_numpy(sel)
    out = scol.gather(icol, len(sel))
    assert np.array_equal(out.to_numpy(), src[sel])
    for c in (scol, icol, out):
        c.free()


# ---------- join --------------------------------------------------------

@pytest.mark.parametrize("how,mode", [("inner", 0), ("semi", 1), ("anti", 2)])
def test_join_parity_multiset(gpu, how, mode):
    from quokka_amd import ops, shim
    rng = np.random.default_rng(9)
    bk = rng.integers(0, 5000, 20000).astype(np.int64)   # ~4x dup keys
    pk = rng.integers(0, 8000, 50000).astype(np.int64)   # ~40% miss
    bcol = shim.DevColumn.from_numpy(bk)
    pcol = shim.DevColumn.from_numpy(pk)
    table = ops.JoinTable(len(bk))
    # batched build (state accumulation across execute() calls)
    half = len(bk) // 2
    b1 = shim.DevColumn.from_numpy(bk[:half])
    b2 = shim.DevColumn.from_numpy(bk[half:])
    table.build(b1)
    table.build(b2)
    pidx, bidx, nm = table.probe(pcol, mode=mode)
    if how == "inner":
        want_p, want_b = OE.build_probe_join(bk, pk, "inner")
        got = set(zip(pidx.to_numpy(nm).tolist(), bidx.to_numpy(nm).tolist()))
        assert got == set(zip(want_p.tolist(), want_b.tolist()))
    else:
        want = set(OE.build_probe_join(bk, pk, how).tolist())
        assert set(pidx.to_numpy(nm).tolist()) == want
    for c in (bcol, pcol, b1, b2, pidx):
        c.free()
    if bidx:
        bidx.free()
    table.free()


def test_join_output_overflow_regrow(gpu):
    """First probe guess too small -> counted, re-run, never truncated."""
    from quokka_amd import ops, shim
    bk = np.zeros(100, np.int64)         # all same key
    pk = np.zeros(50, np.int64)          # 5000 output pairs from 50 probes
    table = ops.JoinTable(len(bk))
    bcol = shim.DevColumn.from_numpy(bk)
    table.build(bcol)
    pcol = shim.DevColumn.from_numpy(pk)
    pidx, bidx, nm = table.probe(pcol, mode=0, out_factor=0.1)
    assert nm == 5000
    got = list(zip(pidx.to_numpy(nm).tolist(), bidx.to_numpy(nm).tolist()))
    assert len(set(got)) == 5000
    for c in (bcol, pcol, pidx, bidx):
        c.free()
    table.free()


def test_join_empty_sides(gpu):
    from quokka_amd import ops, shim
    table = ops.JoinTable(16)
    pk = np.arange(10, dtype=np.int64)
    pcol = shim.DevColumn.from_numpy(pk)
    pidx, bidx, nm = table.probe(pcol, mode=0)
    assert nm == 0
    pidx2, _, nm2 = table.probe(pcol, mode=2)  # anti: all pass
    assert nm2 == 10
    for c in (pcol, pidx, pidx2):
        c.free()
    if bidx:
        bidx.free()
    table.free()


# ---------- group-by ----------------------------------------------------

def test_groupby_parity(gpu):
    from quokka_amd import ops, shim
    rng = np.random.default_rng(17)
    keys = rng.integers(0, 100000, 500000).astype(np.int64)
    v1 = rng.random(500000)
    v2 = rng.random(500000)
    gb = ops.GroupByI64(120000, 2)
    kcol = shim.DevColumn.from_numpy(keys)
    c1 = shim.DevColumn.from_numpy(v1)
    c2 = shim.DevColumn.from_numpy(v2)
    gb.update(kcol, [c1, c2])
    gk, gs = gb.extract()
    order = np.argsort(gk)
    uk, s1 = OE.groupby_sum_i64(keys, v1)
    _, s2 = OE.groupby_sum_i64(keys, v2)
    assert np.array_equal(gk[order], uk)
    np.testing.assert_allclose(gs[0][order], s1, rtol=1e-9)
    np.testing.assert_allclose(gs[1][order], s2, rtol=1e-9)
    for c in (kcol, c1, c2):
        c.free()
    gb.free()


# ---------- partition ---------------------------------------------------

@pytest.mark.parametrize("nparts", [1, 3, 8, 512])
def test_partition_parity_bitexact(gpu, nparts):
    from quokka_amd import ops, shim
    rng = np.random.default_rng(23)
    keys = rng.integers(0, 1 << 40, 300000).astype(np.int64)
    keys[::97] = -keys[::97]          # negative keys: no OOB, colocated
    kcol = shim.DevColumn.from_numpy(keys)
    offsets, idx = ops.partition_i64(kcol, nparts)
    sel = idx.to_numpy(len(keys))
    want = OE.partition_int(keys, nparts)
    hist = np.bincount(want, minlength=nparts)
    assert np.array_equal(np.diff(offsets.astype(np.int64)), hist)
    for p in range(nparts):
        rows = sel[int(offsets[p]):int(offsets[p + 1])]
        assert np.all(want[rows] == p)            # right bucket
    assert len(np.unique(sel)) == len(keys)       # a permutation
    kcol.free()
    idx.free()


# ---------- Q3 end-to-end ----------------------------------------------

def test_q3_parity(gpu, data):
    from quokka_amd import queries as DQ, staging
    li, orders, cust = data["lineitem"], data["orders"], data["customer"]
    lcols = staging.stage_columns(li, names=["l_orderkey", "l_shipdate",
                                             "l_extendedprice", "l_discount"])
    ocols = staging.stage_columns(orders)
    ccols = staging.stage_columns(cust, names=["c_custkey", "c_mktsegment"])
    full, top10 = DQ.q3(lcols, ocols, ccols)
    wfull, wtop = OQ.q3(li, orders, cust)
    # group KEY SET bit-exact; revenues within 1e-9
    order_g = np.argsort(full["l_orderkey"])
    order_w = np.argsort(wfull["l_orderkey"])
    assert np.array_equal(full["l_orderkey"][order_g],
                          wfull["l_orderkey"][order_w])
    np.testing.assert_allclose(full["revenue"][order_g],
                               wfull["revenue"][order_w], rtol=1e-9)
    assert np.array_equal(full["o_orderdate"][order_g],
                          wfull["o_orderdate"][order_w])
    assert np.array_equal(top10["l_orderkey"], wtop["l_orderkey"])
    np.testing.assert_allclose(top10["revenue"], wtop["revenue"], rtol=1e-9)
    for cs in (lcols, ocols, ccols):
        for c in cs.values():
            c.free()


# ---------- plugin-API executors (the drop-in boundary) -----------------

def test_executor_join_plugin_api(gpu):
    """GPUBuildProbeJoinExecutor through the reference's execute/done
    contract (batches = list[pyarrow.Table], stream 1 build then stream 0
    probe)."""
    import pyarrow as pa
    from quokka_amd import GPUBuildProbeJoinExecutor
    rng = np.random.default_rng(31)
    bk = rng.integers(0, 500, 2000).astype(np.int64)
    bval = rng.random(2000)
    pk = rng.integers(0, 800, 3000).astype(np.int64)
    pval = rng.integers(0, 100, 3000).astype(np.int64)
    ex = GPUBuildProbeJoinExecutor(left_on="pk", right_on="bk", how="inner")
    build = pa.table({"bk": bk, "bval": bval})
    probe = pa.table({"pk": pk, "pval": pval})
    assert ex.execute([build.slice(0, 1000), build.slice(1000)], 1, 0) is None
    out = ex.execute([probe], 0, 0)
    ex.done(0)
    want_p, want_b = OE.build_probe_join(bk, pk, "inner")
    got = set(zip(out.column("pk").to_pylist(),
                  out.column("pval").to_pylist(),
                  [round(v, 12) for v in out.column("bval").to_pylist()]))
    want = set(zip(pk[want_p].tolist(), pval[want_p].tolist(),
                   [round(v, 12) for v in bval[want_b].tolist()]))
    assert got == want


def test_executor_agg_plugin_api(gpu):
    """GPUAggExecutor with the post-rewrite final-agg SQL the reference
    produces (sql_utils.py:379-413), incl. avg as sum/sum."""
    import pyarrow as pa
    from quokka_amd import GPUAggExecutor
    rng = np.random.default_rng(37)
    g = rng.integers(0, 50, 5000).astype(np.int64)
    x = rng.random(5000)
    cnt = np.ones(5000)
    ex = GPUAggExecutor(["g"], [("g", "asc")],
                        "sum(e0_agg_0) as sx, sum(e0_agg_0) / sum(e0_agg_1) as ax")
    t = pa.table({"g": g, "e0_agg_0": x, "e0_agg_1": cnt})
    ex.execute([t.slice(0, 2500)], 0, 0)
    ex.execute([t.slice(2500)], 0, 0)
    out = ex.done(0)
    uk, sums = OE.groupby_sum_i64(g, x)
    _, cs = OE.groupby_sum_i64(g, cnt)
    assert out.column("g").to_pylist() == uk.tolist()
    np.testing.assert_allclose(out.column("sx").to_numpy(), sums, rtol=1e-9)
    np.testing.assert_allclose(out.column("ax").to_numpy(), sums / cs,
                               rtol=1e-9)


def test_gen_lineitem_device_vs_oracle_shape(gpu):
    """Device-generated bench data has the oracle generator's distributions
    (same filter selectivities within tolerance) and the Q1 kernel over it
    matches the oracle partials computed on the d2h copy of the SAME data."""
    from quokka_amd import shim, ops, queries as DQ
    from quokka_amd.shim import DevColumn, c_u64
    n = 1_000_000
    cols = {name: DevColumn(dt, n) for name, dt in [
        ("l_quantity", np.float64), ("l_extendedprice", np.float64),
        ("l_discount", np.float64), ("l_tax", np.float64),
        ("l_returnflag", np.uint8), ("l_linestatus", np.uint8),
        ("l_shipdate", np.int32)]}
    shim.call("qk_gen_lineitem", None, c_u64(n), c_u64(0), c_u64(42),
              shim.c_i64(2000), shim.c_i64(100), shim.c_i64(1000),
              None, None,
              cols["l_quantity"].ptr, cols["l_extendedprice"].ptr,
              cols["l_discount"].ptr, cols["l_tax"].ptr,
              cols["l_returnflag"].ptr, cols["l_linestatus"].ptr,
              cols["l_shipdate"].ptr, None, None)
    host = {k: v.to_numpy() for k, v in cols.items()}
    got = DQ.q1(cols)
    want = OQ.q1(host)
    assert np.array_equal(got["count_order"], want["count_order"])
    for c in ("sum_qty", "sum_charge", "avg_disc"):
        np.testing.assert_allclose(got[c], want[c], rtol=1e-9)
    # selectivity sanity vs TPC-H (~98.6%)
    frac = want["count_order"].sum() / n
    assert 0.975 < frac < 0.995
    for c in cols.values():
        c.free()


# ---------- fused Q3 path ----------------------------------------------

def test_q3_fused_parity(gpu, data):
    """Fused build/probe/agg kernels == oracle == generic pipeline."""
    from quokka_amd import queries as DQ, staging
    li, orders, cust = data["lineitem"], data["orders"], data["customer"]
    lcols = staging.stage_columns(li, names=["l_orderkey", "l_shipdate",
                                             "l_extendedprice", "l_discount"])
    ocols = staging.stage_columns(orders)
    ccols = staging.stage_columns(cust, names=["c_custkey", "c_mktsegment"])
    from quokka_amd import shim as _shim
    stream = _shim.Stream()   # exercise the non-default-stream path
    full, top10 = DQ.q3_fused(lcols, ocols, ccols, stream=stream)
    stream.destroy()
    wfull, wtop = OQ.q3(li, orders, cust)
    og = np.argsort(full["l_orderkey"])
    ow = np.argsort(wfull["l_orderkey"])
    assert np.array_equal(full["l_orderkey"][og], wfull["l_orderkey"][ow])
    np.testing.assert_allclose(full["revenue"][og], wfull["revenue"][ow],
                               rtol=1e-9)
    assert np.array_equal(full["o_orderdate"][og], wfull["o_orderdate"][ow])
    assert np.array_equal(top10["l_orderkey"], wtop["l_orderkey"])
    np.testing.assert_allclose(top10["revenue"], wtop["revenue"], rtol=1e-9)
    for cs in (lcols, ocols, ccols):
        for c in cs.values():
            c.free()


def test_q3_fused_rebuild_idempotent(gpu, data):
    from quokka_amd import queries as DQ, staging
    li, orders, cust = data["lineitem"], data["orders"], data["customer"]
    lcols = staging.stage_columns(li, names=["l_orderkey", "l_shipdate",
                                             "l_extendedprice", "l_discount"])
    ocols = staging.stage_columns(orders)
    ccols = staging.stage_columns(cust, names=["c_custkey", "c_mktsegment"])
    st = DQ.Q3Fused(ocols, ccols)
    st.probe(lcols, nt=False)     # plain-load variant
    f1, _ = st.extract()
    st.rebuild()
    st.probe(lcols, nt=True)      # non-temporal variant must agree
    f2, _ = st.extract()
    o1, o2 = np.argsort(f1["l_orderkey"]), np.argsort(f2["l_orderkey"])
    assert np.array_equal(f1["l_orderkey"][o1], f2["l_orderkey"][o2])
    np.testing.assert_allclose(f1["revenue"][o1], f2["revenue"][o2],
                               rtol=1e-12)
    st.free()
    for cs in (lcols, ocols, ccols):
        for c in cs.values():
            c.free()


def test_q3_fused_on_device_generated(gpu):
    """Device-generated lineitem/orders/customer -> fused Q3 == oracle on
    the d2h copies of the SAME data (validates the gen kernels' join-key
    and date-correlation structure)."""
    from quokka_amd import shim, queries as DQ
    from quokka_amd.shim import DevColumn, c_u64, c_i64
    n = 400_000
    n_ord, n_cust = n // 4, n // 40
    li = {k: DevColumn(dt, n) for k, dt in [
        ("l_orderkey", np.int64), ("l_shipdate", np.int32),
        ("l_extendedprice", np.float64), ("l_discount", np.float64)]}
    shim.call("qk_gen_lineitem", None, c_u64(n), c_u64(0), c_u64(5),
              c_i64(20000), c_i64(1000), c_i64(n_ord),
              li["l_orderkey"].ptr, None, None,
              li["l_extendedprice"].ptr, li["l_discount"].ptr, None,
              None, None, li["l_shipdate"].ptr, None, None)
    od = {k: DevColumn(dt, n_ord) for k, dt in [
        ("o_orderkey", np.int64), ("o_custkey", np.int64),
        ("o_orderdate", np.int32), ("o_shippriority", np.int32)]}
    shim.call("qk_gen_orders", None, c_u64(n_ord), c_u64(0), c_u64(5),
              c_i64(n_cust), od["o_orderkey"].ptr, od["o_custkey"].ptr,
              od["o_orderdate"].ptr, od["o_shippriority"].ptr,
              None, None, c_i64(1))
    cu = {"c_custkey": DevColumn(np.int64, n_cust),
          "c_mktsegment": DevColumn(np.uint8, n_cust)}
    shim.call("qk_gen_customer", None, c_u64(n_cust), c_u64(0), c_u64(5),
              cu["c_custkey"].ptr, cu["c_mktsegment"].ptr, None)
    li_h = {k: v.to_numpy() for k, v in li.items()}
    od_h = {k: v.to_numpy() for k, v in od.items()}
    cu_h = {k: v.to_numpy() for k, v in cu.items()}
    # shipdate must correlate with the order's orderdate (1..121 ahead);
    # invert the spec 4.2.3 sparse orderkey (8 keys per 32-key bucket)
    # back to the order ROW to index the generated orders table
    k0 = (li_h["l_orderkey"] - 1).astype(np.int64)
    assert np.all(k0 % 32 < 8), "orderkeys must be spec-sparse"
    orow = (k0 // 32) * 8 + (k0 % 32)
    omap = od_h["o_orderdate"][orow]
    delta = li_h["l_shipdate"] - omap
    assert delta.min() >= 1 and delta.max() <= 121
    # and the custkey mortality hole (spec: custkey % 3 != 0)
    assert np.all(od_h["o_custkey"] % 3 != 0)
    full, top10 = DQ.q3_fused(li, od, cu)
    wfull, wtop = OQ.q3(li_h, od_h, cu_h)
    og = np.argsort(full["l_orderkey"])
    ow = np.argsort(wfull["l_orderkey"])
    assert np.array_equal(full["l_orderkey"][og], wfull["l_orderkey"][ow])
    np.testing.assert_allclose(full["revenue"][og], wfull["revenue"][ow],
                               rtol=1e-9)
    assert np.array_equal(top10["l_orderkey"], wtop["l_orderkey"])
    for cs in (li, od, cu):
        for c in cs.values():
            c.free()


def test_q3_fused_edge_empty(gpu):
    """No BUILDING customers -> empty result; empty lineitem -> empty."""
    from quokka_amd import queries as DQ, staging
    od = staging.stage_columns({
        "o_orderkey": np.arange(1, 101, dtype=np.int64),
        "o_custkey": np.ones(100, dtype=np.int64),
        "o_orderdate": np.full(100, 9000, np.int32),
        "o_shippriority": np.zeros(100, np.int32)})
    cu = staging.stage_columns({
        "c_custkey": np.arange(1, 11, dtype=np.int64),
        "c_mktsegment": np.zeros(10, np.uint8)})  # none == BUILDING(1)
    li = staging.stage_columns({
        "l_orderkey": np.arange(1, 101, dtype=np.int64),
        "l_shipdate": np.full(100, 9500, np.int32),
        "l_extendedprice": np.full(100, 10.0),
        "l_discount": np.zeros(100)})
    full, top = DQ.q3_fused(li, od, cu)
    assert len(full["l_orderkey"]) == 0
    for cs in (od, cu, li):
        for c in cs.values():
            c.free()


# ---------- fused Q5 path ----------------------------------------------

def test_q5_fused_parity(gpu, data):
    from quokka_amd import queries as DQ, staging
    li, orders = data["lineitem"], data["orders"]
    cust, supp = data["customer"], data["supplier"]
    lcols = staging.stage_columns(li, names=["l_orderkey", "l_suppkey",
                                             "l_extendedprice", "l_discount"])
    ocols = staging.stage_columns(orders, names=["o_orderkey", "o_custkey",
                                                 "o_orderdate"])
    ccols = staging.stage_columns(cust, names=["c_custkey", "c_nationkey"])
    scols = staging.stage_columns(supp)
    got = DQ.q5_fused(lcols, ocols, ccols, scols)
    want = OQ.q5(li, orders, cust, supp, data["nation"], data["region"])
    assert [n for n, _ in got] == [n for n, _ in want]
    np.testing.assert_allclose([v for _, v in got], [v for _, v in want],
                               rtol=1e-9)
    for cs in (lcols, ocols, ccols, scols):
        for c in cs.values():
            c.free()


def test_q5_fused_on_device_generated(gpu):
    from quokka_amd import shim, queries as DQ
    from quokka_amd.shim import DevColumn, c_u64, c_i64
    n = 400_000
    n_ord, n_cust, n_supp = n // 4, n // 40, max(1, n // 600)
    li = {k: DevColumn(dt, n) for k, dt in [
        ("l_orderkey", np.int64), ("l_suppkey", np.int64),
        ("l_extendedprice", np.float64), ("l_discount", np.float64)]}
    shim.call("qk_gen_lineitem", None, c_u64(n), c_u64(0), c_u64(9),
              c_i64(20000), c_i64(n_supp), c_i64(n_ord),
              li["l_orderkey"].ptr, li["l_suppkey"].ptr, None,
              li["l_extendedprice"].ptr, li["l_discount"].ptr, None,
              None, None, None, None, None)
    od = {k: DevColumn(dt, n_ord) for k, dt in [
        ("o_orderkey", np.int64), ("o_custkey", np.int64),
        ("o_orderdate", np.int32)]}
    shim.call("qk_gen_orders", None, c_u64(n_ord), c_u64(0), c_u64(9),
              c_i64(n_cust), od["o_orderkey"].ptr, od["o_custkey"].ptr,
              od["o_orderdate"].ptr, None, None, None, c_i64(1))
    cu = {"c_custkey": DevColumn(np.int64, n_cust),
          "c_nationkey": DevColumn(np.int32, n_cust)}
    shim.call("qk_gen_customer", None, c_u64(n_cust), c_u64(0), c_u64(9),
              cu["c_custkey"].ptr, None, cu["c_nationkey"].ptr)
    su = {"s_suppkey": DevColumn(np.int64, n_supp),
          "s_nationkey": DevColumn(np.int32, n_supp)}
    shim.call("qk_gen_supplier", None, c_u64(n_supp), c_u64(0), c_u64(9),
              su["s_suppkey"].ptr, su["s_nationkey"].ptr)
    li_h = {k: v.to_numpy() for k, v in li.items()}
    od_h = {k: v.to_numpy() for k, v in od.items()}
    cu_h = {k: v.to_numpy() for k, v in cu.items()}
    su_h = {k: v.to_numpy() for k, v in su.items()}
    got = DQ.q5_fused(li, od, cu, su)
    want = OQ.q5(li_h, od_h, cu_h, su_h, G.gen_nation(), G.gen_region())
    assert [n_ for n_, _ in got] == [n_ for n_, _ in want]
    np.testing.assert_allclose([v for _, v in got], [v for _, v in want],
                               rtol=1e-9)
    for cs in (li, od, cu, su):
        for c in cs.values():
            c.free()


# ---------- RCCL exchange (world_size == 1 self-exchange) ---------------

def test_exchange_world1_roundtrip(gpu):
    """Comm init + alltoallv at world 1: the full repartition path runs on
    one GPU and must return exactly the rows whose key % 1 == 0 (all),
    partition-ordered == original multiset."""
    from quokka_amd import exchange, shim
    rng = np.random.default_rng(41)
    keys = rng.integers(0, 10_000, 100_000).astype(np.int64)
    vals = rng.random(100_000)
    kcol = shim.DevColumn.from_numpy(keys)
    vcol = shim.DevColumn.from_numpy(vals)
    comm = exchange.Comm(0, 1)
    rk, rp = exchange.repartition(comm, kcol, {"v": vcol})
    got_k = rk.to_numpy(rk.n)
    got_v = rp["v"].to_numpy(rp["v"].n)
    assert len(got_k) == len(keys)
    order_g = np.lexsort((got_v, got_k))
    order_w = np.lexsort((vals, keys))
    assert np.array_equal(got_k[order_g], keys[order_w])
    np.testing.assert_allclose(got_v[order_g], vals[order_w], rtol=0)
    rk.free(); rp["v"].free(); kcol.free(); vcol.free()
    comm.destroy()


def test_exchange_world1_over_1gib(gpu):
    """Regression: RCCL 2.27 p2p silently delivers only the FIRST HALF of
    any single send whose byte count exceeds 1 GiB (measured on MI355X,
    scripts/diag_exchange.py — exact at 2^30 B, half the rows above).
    qk_alltoallv must route the self-partition through a D2D copy and
    chunk peer pieces, so a >1 GiB column survives the exchange intact."""
    from quokka_amd import exchange, shim
    from quokka_amd.shim import DevColumn, c_u64
    n = 300_000_000                      # u32 -> 1.2 GB, over the 1 GiB cliff
    col = DevColumn(np.uint32, n)
    shim.call("qk_iota_u32", None, c_u64(n), col.ptr)
    comm = exchange.Comm(0, 1)
    so = np.zeros(1, dtype=np.uint64)
    sc = np.array([n], dtype=np.uint64)
    recv = comm.alltoallv_column(col, so, sc, sc)
    got = recv.to_numpy(n)
    bad = np.nonzero(got != np.arange(n, dtype=np.uint32))[0]
    assert bad.size == 0, "first bad idx %d of %d" % (bad[0], bad.size)
    recv.free()
    col.free()
    # and through the full repartition (partition -> gather -> exchange)
    # with device-generated i64 keys (1.28 GiB key column)
    from quokka_amd.shim import c_i64
    nk = 160_000_000
    kcol = DevColumn(np.int64, nk)
    dat = DevColumn(np.int32, nk)
    pr = DevColumn(np.float64, nk)
    di = DevColumn(np.float64, nk)
    shim.call("qk_gen_lineitem", None, c_u64(nk), c_u64(0), c_u64(42),
              c_i64(20_000_000), c_i64(1_000_000), c_i64(nk // 4),
              kcol.ptr, None, None, pr.ptr, di.ptr, None, None, None,
              dat.ptr, None, None)
    pr.free()
    di.free()
    want_k = kcol.to_numpy(nk)
    want_d = dat.to_numpy(nk)
    rk, rp = exchange.repartition(comm, kcol, {"d": dat})
    hk = rk.to_numpy(nk)
    hd = rp["d"].to_numpy(nk)
    # permutation-invariant checks (sum/xor/min/max) on both columns
    for got, want in ((hk, want_k), (hd, want_d)):
        assert int(np.sum(got, dtype=np.uint64)) == \
            int(np.sum(want, dtype=np.uint64))
        assert int(got.min()) == int(want.min())
        assert int(got.max()) == int(want.max())
    rk.free()
    rp["d"].free()
    kcol.free()
    dat.free()
    comm.destroy()


def test_exchange_allreduce_world1(gpu):
    from quokka_amd import exchange, shim
    from quokka_amd.shim import DevColumn
    comm = exchange.Comm(0, 1)
    arr = np.arange(48, dtype=np.float64)
    col = DevColumn.from_numpy(arr)
    comm.allreduce_f64(col.ptr, 48)
    got = col.to_numpy()
    np.testing.assert_allclose(got, arr)  # world 1: identity
    col.free()
    comm.destroy()


def test_extract_top10_matches_full_extract(gpu, data):
    from quokka_amd import queries as DQ, staging
    li, orders, cust = data["lineitem"], data["orders"], data["customer"]
    lcols = staging.stage_columns(li, names=["l_orderkey", "l_shipdate",
                                             "l_extendedprice", "l_discount"])
    ocols = staging.stage_columns(orders)
    ccols = staging.stage_columns(cust, names=["c_custkey", "c_mktsegment"])
    st = DQ.Q3Fused(ocols, ccols)
    st.probe(lcols)
    full, top_a = st.extract(10)
    k, top_b = st.extract_top10(10)
    assert k == len(full["l_orderkey"])
    assert np.array_equal(top_a["l_orderkey"], top_b["l_orderkey"])
    np.testing.assert_allclose(top_a["revenue"], top_b["revenue"], rtol=0)
    assert np.array_equal(top_a["o_orderdate"], top_b["o_orderdate"])
    st.free()
    for cs in (lcols, ocols, ccols):
        for c in cs.values():
            c.free()


def test_q3_after_exchange_world1(gpu, data):
    """repartition (world 1 self-exchange) -> fused Q3 on the received
    partition == oracle (the configs[3] path on one GPU)."""
    from quokka_amd import queries as DQ, staging, exchange
    li, orders, cust = data["lineitem"], data["orders"], data["customer"]
    lcols = staging.stage_columns(li, names=["l_orderkey", "l_shipdate",
                                             "l_extendedprice", "l_discount"])
    ocols = staging.stage_columns(orders)
    ccols = staging.stage_columns(cust, names=["c_custkey", "c_mktsegment"])
    comm = exchange.Comm(0, 1)
    rk, rp = exchange.repartition(
        comm, lcols["l_orderkey"],
        {k: v for k, v in lcols.items() if k != "l_orderkey"})
    li_x = {"l_orderkey": rk, **rp}
    ok, op = exchange.repartition(
        comm, ocols["o_orderkey"],
        {k: v for k, v in ocols.items() if k != "o_orderkey"})
    od_x = {"o_orderkey": ok, **op}
    full, top10 = DQ.q3_fused(li_x, od_x, ccols)
    wfull, wtop = OQ.q3(li, orders, cust)
    og, ow = np.argsort(full["l_orderkey"]), np.argsort(wfull["l_orderkey"])
    assert np.array_equal(full["l_orderkey"][og], wfull["l_orderkey"][ow])
    np.testing.assert_allclose(full["revenue"][og], wfull["revenue"][ow],
                               rtol=1e-9)
    assert np.array_equal(top10["l_orderkey"], wtop["l_orderkey"])
    comm.destroy()
    for cs in (lcols, ocols, ccols, li_x, od_x):
        for c in cs.values():
            c.free()


def test_executor_agg_min_max(gpu):
    """GPUAggExecutor with MIN/MAX partials (the other distributive ops
    the two-phase rewrite emits, sql_utils.py:299-413)."""
    import pyarrow as pa
    from quokka_amd import GPUAggExecutor
    rng = np.random.default_rng(51)
    g = rng.integers(0, 40, 4000).astype(np.int64)
    x = rng.random(4000)
    y = rng.random(4000)
    # distinct partial columns per aggregate, as the reference's rewrite
    # emits (sql_utils.py:379-413 assigns each partial a unique alias)
    ex = GPUAggExecutor(["g"], [("g", "asc")],
                        "min(e0_agg_0) as mn, max(e1_agg_0) as mx, "
                        "sum(e2_agg_0) as sx")
    t = pa.table({"g": g, "e0_agg_0": x, "e1_agg_0": y, "e2_agg_0": x})
    ex.execute([t.slice(0, 2000)], 0, 0)
    ex.execute([t.slice(2000)], 0, 0)
    out = ex.done(0)
    uk = np.unique(g)
    assert out.column("g").to_pylist() == uk.tolist()
    want_mn = np.array([x[g == k].min() for k in uk])
    want_mx = np.array([y[g == k].max() for k in uk])
    want_sx = np.array([x[g == k].sum() for k in uk])
    np.testing.assert_allclose(out.column("mn").to_numpy(), want_mn, rtol=0)
    np.testing.assert_allclose(out.column("mx").to_numpy(), want_mx, rtol=0)
    np.testing.assert_allclose(out.column("sx").to_numpy(), want_sx,
                               rtol=1e-9)


def test_executor_agg_q1_composite_keys(gpu, data):
    """Q1's final aggregate through the GENERIC GPUAggExecutor with the
    reference's two-string group keys and rewritten final-agg SQL
    (datastream.py:1819-1856 lowering; avg -> sum/sum per
    sql_utils.py:299-413) — composite key encode/decode path."""
    import pyarrow as pa
    from quokka_amd import GPUAggExecutor
    li = data["lineitem"]
    # map-side partials per chunk, as the folded batch_funcs produce them
    n = len(li["l_orderkey"])
    rows = {"l_returnflag": [], "l_linestatus": [], "e0": [], "e1": [],
            "e2": [], "e3": [], "cnt": []}
    for lo in range(0, n, 9973):
        chunk = {k: v[lo:lo + 9973] for k, v in li.items()}
        p = Q_partials(chunk)
        for g in range(6):
            if p[g, 5] > 0:
                rows["l_returnflag"].append(G.RETURNFLAG[g // 2])
                rows["l_linestatus"].append(G.LINESTATUS[g % 2])
                rows["e0"].append(p[g, 0])
                rows["e1"].append(p[g, 1])
                rows["e2"].append(p[g, 2])
                rows["e3"].append(p[g, 3])
                rows["cnt"].append(p[g, 5])
    t = pa.table({k: (np.array(v) if k not in ("l_returnflag",
                                               "l_linestatus")
                      else np.array(v, dtype=object).astype(str))
                  for k, v in rows.items()})
    ex = GPUAggExecutor(
        ["l_returnflag", "l_linestatus"],
        [("l_returnflag", "asc"), ("l_linestatus", "asc")],
        "sum(e0) as sum_qty, sum(e1) as sum_base_price, "
        "sum(e2) as sum_disc_price, sum(e3) as sum_charge, "
        "sum(e0) / sum(cnt) as avg_qty, sum(cnt) as count_order")
    half = len(t) // 2
    ex.execute([t.slice(0, half)], 0, 0)
    ex.execute([t.slice(half)], 0, 0)
    out = ex.done(0)
    want = OQ.q1(li)
    assert out.column("l_returnflag").to_pylist() == \
        list(want["l_returnflag"])
    assert out.column("l_linestatus").to_pylist() == \
        list(want["l_linestatus"])
    np.testing.assert_allclose(out.column("sum_qty").to_numpy(),
                               want["sum_qty"], rtol=1e-9)
    np.testing.assert_allclose(out.column("avg_qty").to_numpy(),
                               want["avg_qty"], rtol=1e-9)
    np.testing.assert_allclose(out.column("count_order").to_numpy(),
                               want["count_order"].astype(float), rtol=0)


def Q_partials(chunk):
    return OQ.q1_partials(chunk)


def test_gpu_partition_fn_plugin_api(gpu):
    """gpu_partition_fn through the registration-time contract
    (core.py:152-206): pyarrow table in, dict channel -> table out,
    int-key buckets bit-exact with quokka_runtime.py:222."""
    import pyarrow as pa
    from quokka_amd import gpu_partition_fn
    rng = np.random.default_rng(61)
    keys = rng.integers(0, 100000, 20000).astype(np.int64)
    vals = rng.random(20000)
    t = pa.table({"k": keys, "v": vals})
    out = gpu_partition_fn(t, source_channel=0, num_target_channels=4,
                           key="k")
    want = OE.partition_int(keys, 4)
    total = 0
    for ch, tbl in out.items():
        got_k = np.asarray(tbl.column("k").to_numpy())
        assert np.all(got_k % 4 == ch)
        total += len(tbl)
        # payload rows stay attached to their keys
        got_v = np.asarray(tbl.column("v").to_numpy())
        lookup = {(k_, round(v_, 12)) for k_, v_ in zip(keys, vals)}
        assert all((k_, round(v_, 12)) in lookup
                   for k_, v_ in zip(got_k[:50], got_v[:50]))
    assert total == len(keys)


def test_gpu_partition_fn_batch_agg_fusion(gpu):
    """Map-side partial-agg fusion (the reference
    folds batch partial aggs into partition_fn, core.py:173-176 +
    df.py:1354-1394): predicate + group-by partial aggregate run as one
    fused device pass and the PARTIAL table (one row per observed
    group), not rows, is what gets partitioned — by group id % N."""
    import pyarrow as pa
    from quokka_amd import gpu_partition_fn
    rng = np.random.default_rng(63)
    n = 200_000
    g1 = rng.integers(0, 3, n).astype(np.uint8)
    g2 = rng.integers(0, 2, n).astype(np.uint8)
    x = np.round(rng.random(n), 6)
    d = rng.integers(0, 1000, n).astype(np.int32)
    t = pa.table({"flag": g1, "status": g2, "x": x, "d": d})
    out = gpu_partition_fn(
        t, source_channel=0, num_target_channels=4, key=None,
        predicate="d < 700",
        batch_agg=([("flag", 3), ("status", 2)],
                   ["SUM(x) as sx", "COUNT(*) as cnt"]))
    mask = d < 700
    seen_groups = 0
    for ch, tbl in out.items():
        gf = np.asarray(tbl.column("flag"))
        gs = np.asarray(tbl.column("status"))
        sx = np.asarray(tbl.column("sx"))
        cnt = np.asarray(tbl.column("cnt"))
        for i in range(len(tbl)):
            gid = int(gf[i]) * 2 + int(gs[i])
            assert gid % 4 == ch                     # int-key bucketing
            m = mask & (g1 == gf[i]) & (g2 == gs[i])
            np.testing.assert_allclose(sx[i], x[m].sum(), rtol=1e-9)
            assert cnt[i] == m.sum()
            seen_groups += 1
    assert seen_groups == 6                          # all groups observed
    # shuffle payload is O(groups), not O(rows)
    assert sum(len(tbl) for tbl in out.values()) == 6


def test_executor_distinct_and_broadcast_and_count(gpu):
    """The remaining hot-path executor family (sql_executors.py:69
    CountExecutor, :275 BroadcastJoinExecutor, :517 DistinctExecutor)
    through the plugin contract."""
    import pyarrow as pa
    from quokka_amd import (GPUCountExecutor, GPUBroadcastJoinExecutor,
                            GPUDistinctExecutor)
    rng = np.random.default_rng(71)
    # count
    ce = GPUCountExecutor()
    t = pa.table({"x": np.arange(1000)})
    ce.execute([t.slice(0, 300), t.slice(300)], 0, 0)
    ce.execute([t], 0, 0)
    assert ce.done(0).column("count").to_pylist() == [2000]
    # broadcast join (small side at construction)
    small = pa.table({"k": np.arange(0, 50, dtype=np.int64),
                      "tag": np.arange(0, 50) * 10})
    bj = GPUBroadcastJoinExecutor(small, small_on="k", big_on="bk",
                                  how="inner")
    big_k = rng.integers(0, 80, 500).astype(np.int64)
    big = pa.table({"bk": big_k, "v": rng.random(500)})
    out = bj.execute([big], 0, 0)
    want_p, want_b = OE.build_probe_join(np.arange(0, 50, dtype=np.int64),
                                         big_k, "inner")
    got = sorted(zip(out.column("bk").to_pylist(),
                     out.column("tag").to_pylist()))
    want = sorted(zip(big_k[want_p].tolist(),
                      (want_b * 10).tolist()))
    assert got == want
    # distinct: per-batch contributions are globally unique & complete,
    # incl. growth past the initial table size
    de = GPUDistinctExecutor("k")
    seen = set()
    all_keys = rng.integers(0, 5000, 100_000).astype(np.int64)
    for lo in range(0, len(all_keys), 10_000):
        chunk = all_keys[lo:lo + 10_000]
        tb = pa.table({"k": chunk, "p": chunk * 2})
        contrib = de.execute([tb], 0, 0)
        if contrib is None:
            continue
        ck = contrib.column("k").to_pylist()
        assert len(set(ck)) == len(ck)          # unique within batch
        assert not (set(ck) & seen)             # never seen before
        # payload stays attached
        assert contrib.column("p").to_pylist() == [k * 2 for k in ck]
        seen |= set(ck)
    assert seen == set(all_keys.tolist())
    assert de.done(0) is None


def test_jit_filter_parity(gpu, data):
    """hiprtc-JIT'd predicates == oracle row sets, bit-exact & ordered:
    Q6's compound predicate, a string-dict equality, OR/NOT forms."""
    from quokka_amd import jit, staging
    li = data["lineitem"]
    cols = staging.stage_columns(li)
    schema = {k: v.dtype for k, v in cols.items()}

    f = jit.JitFilter(
        "l_shipdate >= date '1994-01-01' and l_shipdate < "
        "date '1994-01-01' + interval '1' year and l_discount between "
        "0.06 - 0.01 and 0.06 + 0.01 and l_quantity < 24", schema)
    idx, k = f.run(cols)
    lo, hi = 0.06 - 0.01, 0.06 + 0.01
    want = np.nonzero((li["l_shipdate"] >= G.Q5_LO)
                      & (li["l_shipdate"] < G.Q5_HI)
                      & (li["l_discount"] >= lo) & (li["l_discount"] <= hi)
                      & (li["l_quantity"] < 24))[0]
    assert np.array_equal(idx.to_numpy(k), want)
    f.free(); idx.free()

    f2 = jit.JitFilter(
        "not (l_returnflag = 'A' or l_returnflag = 'R') and "
        "l_quantity >= 25", schema,
        string_dicts={"l_returnflag": type("SD", (), {
            "codes": {"A": 0, "N": 1, "R": 2}})()})
    idx2, k2 = f2.run(cols)
    want2 = np.nonzero(~((li["l_returnflag"] == 0)
                         | (li["l_returnflag"] == 2))
                       & (li["l_quantity"] >= 25))[0]
    assert np.array_equal(idx2.to_numpy(k2), want2)
    f2.free(); idx2.free()

    # empty result
    f3 = jit.JitFilter("l_quantity > 1000", schema)
    idx3, k3 = f3.run(cols)
    assert k3 == 0
    f3.free(); idx3.free()
    for c in cols.values():
        c.free()


def test_jit_aggregate_q1_equals_handwritten(gpu, data):
    """The JIT-generated fused scan+group-by (from Q1's SQL strings) must
    reproduce BOTH the oracle and the hand-written k_q1_agg kernel —
    the hand-written kernel is the template the codegen generalizes."""
    from quokka_amd import jit, staging, queries as DQ
    li = data["lineitem"]
    cols = staging.stage_columns(li)
    schema = {k: v.dtype for k, v in cols.items()}
    agg = jit.JitAggregate(
        schema,
        group_keys=[("l_returnflag", 3), ("l_linestatus", 2)],
        aggs=["sum(l_quantity) as sum_qty",
              "sum(l_extendedprice) as sum_base_price",
              "sum(l_extendedprice * (1 - l_discount)) as sum_disc_price",
              "sum(l_extendedprice * (1 - l_discount) * (1 + l_tax)) "
              "as sum_charge",
              "sum(l_discount) as sum_disc",
              "count(*) as count_order"],
        predicate="l_shipdate <= date '1998-12-01' - interval '90' day")
    acc = agg.make_acc()
    agg.run(cols, acc)
    got = agg.read(acc)                      # (6 groups, 6 aggs)
    want = OQ.q1_partials(li)                # oracle, same layout
    np.testing.assert_allclose(got, want, rtol=1e-9)
    # and against the hand-written kernel's partials
    hand = DQ.q1_partials_device(cols)
    from quokka_amd import ops
    hw = ops.q1_read_partials(hand)
    np.testing.assert_allclose(got, hw, rtol=1e-12)
    hand.free()
    # batched accumulation (executor-state semantics)
    acc2 = agg.make_acc()
    half = len(li["l_shipdate"]) // 2
    for lo, hi in ((0, half), (half, len(li["l_shipdate"]))):
        chunk = {k: v[lo:hi] for k, v in li.items()}
        ccols = staging.stage_columns(chunk)
        agg.run(ccols, acc2)
        for c in ccols.values():
            c.free()
    np.testing.assert_allclose(agg.read(acc2), want, rtol=1e-9)
    acc.free(); acc2.free(); agg.free()
    for c in cols.values():
        c.free()


def test_jit_aggregate_adhoc(gpu, data):
    """An aggregate shape with no hand-written kernel at all: group by
    linestatus only, revenue-weighted sums under a compound predicate."""
    from quokka_amd import jit, staging
    li = data["lineitem"]
    cols = staging.stage_columns(li)
    schema = {k: v.dtype for k, v in cols.items()}
    agg = jit.JitAggregate(
        schema, group_keys=[("l_linestatus", 2)],
        aggs=["sum(l_extendedprice * l_discount) as rev",
              "count(*) as n"],
        predicate="l_quantity between 10 and 20 and "
                  "l_shipdate > date '1995-03-15'")
    acc = agg.make_acc()
    agg.run(cols, acc)
    got = agg.read(acc)
    m = ((li["l_quantity"] >= 10) & (li["l_quantity"] <= 20)
         & (li["l_shipdate"] > G.Q3_DATE))
    for ls in (0, 1):
        mm = m & (li["l_linestatus"] == ls)
        np.testing.assert_allclose(
            got[ls, 0], (li["l_extendedprice"][mm]
                         * li["l_discount"][mm]).sum(), rtol=1e-9)
        assert got[ls, 1] == mm.sum()
    acc.free(); agg.free()
    for c in cols.values():
        c.free()


def test_jit_map_parity(gpu, data):
    """JIT transform == numpy on the revenue expression and a compound
    arithmetic expression; equals the static qk_mul_1md kernel exactly."""
    from quokka_amd import jit, staging, queries as DQ
    li = data["lineitem"]
    cols = staging.stage_columns(li, names=["l_extendedprice",
                                            "l_discount", "l_tax",
                                            "l_quantity"])
    schema = {k: v.dtype for k, v in cols.items()}
    m = jit.JitMap("l_extendedprice * (1 - l_discount)", schema)
    out = m.run(cols)
    want = li["l_extendedprice"] * (1.0 - li["l_discount"])
    got = out.to_numpy(out.n)
    np.testing.assert_array_equal(got, want)      # identical f64 ops
    # same as the static kernel
    rev2 = DQ._mul_1md(cols["l_extendedprice"], cols["l_discount"], None)
    np.testing.assert_array_equal(got, rev2.to_numpy(rev2.n))
    m.free(); out.free(); rev2.free()
    m2 = jit.JitMap(
        "l_extendedprice * (1 - l_discount) * (1 + l_tax) / l_quantity",
        schema)
    out2 = m2.run(cols)
    want2 = (li["l_extendedprice"] * (1.0 - li["l_discount"])
             * (1.0 + li["l_tax"]) / li["l_quantity"])
    np.testing.assert_array_equal(out2.to_numpy(out2.n), want2)
    m2.free(); out2.free()
    for c in cols.values():
        c.free()


# ---------- device radix sort -------------------------------------------

def test_sort_permutation_stable(gpu):
    """Stable ascending/descending permutation vs numpy kind='stable' on
    i64 (with heavy duplicates) and f64 (negatives, zeros, duplicates)."""
    from quokka_amd import ops, shim
    rng = np.random.default_rng(81)
    ki = rng.integers(-1000, 1000, 300_000).astype(np.int64)
    col = shim.DevColumn.from_numpy(ki)
    perm = ops.sort_permutation(col)
    got = perm.to_numpy(perm.n)
    want = np.argsort(ki, kind="stable")
    assert np.array_equal(got, want)           # stability: exact match
    perm.free()
    permd = ops.sort_permutation(col, descending=True)
    gotd = permd.to_numpy(permd.n)
    wantd = np.argsort(-ki, kind="stable")
    assert np.array_equal(gotd, wantd)
    permd.free(); col.free()

    kf = np.round(rng.standard_normal(200_000), 2)  # dups, negatives, 0s
    kf[::1000] = 0.0
    kf[1::1000] = -0.0
    colf = shim.DevColumn.from_numpy(kf)
    permf = ops.sort_permutation(colf)
    gotf = permf.to_numpy(permf.n)
    # -0.0 < 0.0 in the total order map; numpy treats them equal -> compare
    # sorted VALUES exactly and stability among exactly-equal bit patterns
    assert np.array_equal(kf[gotf], np.sort(kf, kind="stable"))
    permf.free(); colf.free()


def test_sort_edge_cases(gpu):
    from quokka_amd import ops, shim
    for arr in (np.array([], np.int64), np.array([7], np.int64),
                np.zeros(1000, np.int64),
                np.arange(513, dtype=np.int64)[::-1].copy()):
        col = shim.DevColumn.from_numpy(arr)
        perm = ops.sort_permutation(col)
        got = perm.to_numpy(perm.n)
        assert np.array_equal(got, np.argsort(arr, kind="stable"))
        perm.free(); col.free()


def test_sort_executor_plugin_api(gpu):
    """GPUSortExecutor through the execute/done contract == polars-style
    stable sort of the accumulated stream (SuperFastSortExecutor's output
    contract, sql_executors.py:88-187)."""
    import pyarrow as pa
    from quokka_amd import GPUSortExecutor
    rng = np.random.default_rng(91)
    k = rng.integers(0, 500, 20_000).astype(np.int64)
    v = np.arange(20_000)
    t = pa.table({"k": k, "v": v})
    ex = GPUSortExecutor("k")
    ex.execute([t.slice(0, 12_000)], 0, 0)
    ex.execute([t.slice(12_000)], 0, 0)
    out = ex.done(0)
    order = np.argsort(k, kind="stable")
    assert out.column("k").to_pylist() == k[order].tolist()
    assert out.column("v").to_pylist() == v[order].tolist()


def test_executor_join_i32_keys_widened(gpu):
    """Integer keys narrower than i64 are accepted (the reference joins on
    any int dtype); widened at the boundary."""
    import pyarrow as pa
    from quokka_amd import GPUBuildProbeJoinExecutor
    bk = np.arange(0, 50, dtype=np.int32)
    pk = np.array([1, 7, 7, 99, 3], dtype=np.int32)
    ex = GPUBuildProbeJoinExecutor(left_on="pk", right_on="bk", how="inner")
    ex.execute([pa.table({"bk": bk, "t": bk * 2})], 1, 0)
    out = ex.execute([pa.table({"pk": pk})], 0, 0)
    got = sorted(zip(out.column("pk").to_pylist(),
                     out.column("t").to_pylist()))
    assert got == [(1, 2), (3, 6), (7, 14), (7, 14)]


def test_parquet_scan_to_gpu_q1(gpu, tmp_path, data):
    """End-to-end scan side: lineitem written to
    Parquet, read back with column selection (as the reference's
    InputParquetDataset does, unordered_readers.py:73-99), staged to HBM,
    fused Q1 == oracle."""
    import pyarrow as pa
    import pyarrow.parquet as pq
    from quokka_amd import staging, queries as DQ
    li = data["lineitem"]
    path = str(tmp_path / "lineitem.parquet")
    pq.write_table(pa.table({k: v for k, v in li.items()}), path)
    cols_needed = ["l_shipdate", "l_quantity", "l_extendedprice",
                   "l_discount", "l_tax", "l_returnflag", "l_linestatus"]
    t = pq.read_table(path, columns=cols_needed)   # column pushdown
    dcols = staging.stage_columns(t)
    got = DQ.q1(dcols)
    want = OQ.q1(li)
    assert np.array_equal(got["count_order"], want["count_order"])
    for c in ("sum_qty", "sum_charge", "avg_disc"):
        np.testing.assert_allclose(got[c], want[c], rtol=1e-9)
    for c in dcols.values():
        c.free()


def test_executor_left_join_plugin_api(gpu):
    """LEFT join through the executor: matched pairs + unmatched probe
    rows with null build payload (sql_executors.py:341 how set)."""
    import pyarrow as pa
    from quokka_amd import GPUBuildProbeJoinExecutor
    bk = np.array([1, 2, 2, 5], dtype=np.int64)
    bv = np.array([10.0, 20.0, 21.0, 50.0])
    pk = np.array([2, 3, 1, 7, 2], dtype=np.int64)
    ex = GPUBuildProbeJoinExecutor(left_on="pk", right_on="bk", how="left")
    ex.execute([pa.table({"bk": bk, "bv": bv})], 1, 0)
    out = ex.execute([pa.table({"pk": pk, "row": np.arange(5)})], 0, 0)
    got = sorted(
        (r["pk"], r["row"], r["bv"])
        for r in out.to_pylist())
    want = sorted([
        (2, 0, 20.0), (2, 0, 21.0), (1, 2, 10.0),
        (2, 4, 20.0), (2, 4, 21.0),
        (3, 1, None), (7, 3, None),
    ])
    assert got == want


def test_jit_aggregate_min_max(gpu, data):
    """JIT aggregate with MIN/MAX ops (generalizing GroupByI64's op set
    to the fused register-accumulator kernel): mixed
    sum/min/max/count per group vs numpy on the same rows, including a
    group the predicate filters out entirely (keeps the identity)."""
    from quokka_amd import jit, staging
    li = data["lineitem"]
    cols = staging.stage_columns(li)
    schema = {k: v.dtype for k, v in cols.items()}
    agg = jit.JitAggregate(
        schema,
        group_keys=[("l_returnflag", 3)],
        aggs=["min(l_extendedprice) as mn",
              "max(l_extendedprice * (1 - l_discount)) as mx",
              "sum(l_quantity) as s",
              "count(*) as n"],
        predicate="l_shipdate > date '1995-06-17'")
    acc = agg.make_acc()
    agg.run(cols, acc)
    got = agg.read(acc)                      # (3 groups, 4 aggs)
    mask = li["l_shipdate"] > (np.datetime64("1995-06-17")
                               - np.datetime64("1970-01-01")).astype(int)
    for g in range(3):
        m = mask & (li["l_returnflag"] == g)
        if m.sum() == 0:
            assert got[g, 0] == np.inf and got[g, 1] == -np.inf
            assert got[g, 2] == 0 and got[g, 3] == 0
            continue
        np.testing.assert_allclose(got[g, 0], li["l_extendedprice"][m].min(),
                                   rtol=1e-12)
        rev = li["l_extendedprice"][m] * (1 - li["l_discount"][m])
        np.testing.assert_allclose(got[g, 1], rev.max(), rtol=1e-12)
        np.testing.assert_allclose(got[g, 2], li["l_quantity"][m].sum(),
                                   rtol=1e-9)
        assert got[g, 3] == m.sum()
    # batched accumulation preserves min/max identity semantics
    acc2 = agg.make_acc()
    half = len(li["l_shipdate"]) // 2
    for lo, hi in ((0, half), (half, len(li["l_shipdate"]))):
        chunk = {k: v[lo:hi] for k, v in li.items()}
        ccols = staging.stage_columns(chunk)
        agg.run(ccols, acc2)
        for c in ccols.values():
            c.free()
    np.testing.assert_allclose(agg.read(acc2), got, rtol=1e-12)
    acc.free(); acc2.free(); agg.free()
    for c in cols.values():
        c.free()


def test_q10_shaped_composed_plan(gpu, data):
    """A query we did NOT hand-fuse, composed exactly as pyquokka lowers
    TPC-H Q10 (apps/tpc-h/tpch.py Q10 plan shape: filter -> join chain ->
    group-by -> top-k): returned-lineitem revenue per customer inside an
    order-date window, through gpu_partition_fn (JIT predicate) ->
    GPUBuildProbeJoinExecutor x2 -> GPUAggExecutor -> GPUTopKExecutor,
    all via the execute/done plugin contract, vs a numpy restatement."""
    import pyarrow as pa
    from quokka_amd import (GPUAggExecutor, GPUBuildProbeJoinExecutor,
                            GPUTopKExecutor, gpu_partition_fn)
    li, od, cu = data["lineitem"], data["orders"], data["customer"]
    D1, D2 = 8800, 9100                       # order-date window (date32)
    R = 2                                     # l_returnflag == 'R' code

    # --- numpy restatement (same dense-key layout as the oracle) ---
    om = (od["o_orderdate"] >= D1) & (od["o_orderdate"] < D2)
    okeys = od["o_orderkey"][om]
    ocust = od["o_custkey"][om]
    key2cust = dict(zip(okeys.tolist(), ocust.tolist()))
    lm = li["l_returnflag"] == R
    rev = {}
    for k, p, d in zip(li["l_orderkey"][lm].tolist(),
                       li["l_extendedprice"][lm].tolist(),
                       li["l_discount"][lm].tolist()):
        c = key2cust.get(k)
        if c is not None:
            rev[c] = rev.get(c, 0.0) + p * (1 - d)
    want = sorted(rev.items(), key=lambda kv: (-kv[1], kv[0]))[:20]

    # --- GPU plan ---
    t_li = pa.table({k: li[k] for k in ("l_orderkey", "l_extendedprice",
                                        "l_discount", "l_returnflag")})
    fl = gpu_partition_fn(t_li, 0, 1, key="l_orderkey",
                          predicate="l_returnflag = %d" % R)[0]
    t_od = pa.table({k: od[k] for k in ("o_orderkey", "o_custkey",
                                        "o_orderdate")})
    fo = gpu_partition_fn(
        t_od, 0, 1, key="o_orderkey",
        predicate="o_orderdate >= %d and o_orderdate < %d" % (D1, D2))[0]
    t_cu = pa.table({"c_custkey": cu["c_custkey"]})
    j1 = GPUBuildProbeJoinExecutor(left_on="o_custkey",
                                   right_on="c_custkey", how="inner")
    j1.execute([t_cu], 1, 0)
    r1 = j1.execute([fo], 0, 0)
    j1.done(0)
    j2 = GPUBuildProbeJoinExecutor(left_on="l_orderkey",
                                   right_on="o_orderkey", how="inner")
    j2.execute([r1], 1, 0)
    r2 = j2.execute([fl], 0, 0)
    j2.done(0)
    # map-side partial (the two-phase rewrite's e0_agg_0 column)
    partial = pa.table({
        "c_custkey": r2.column("o_custkey"),
        "e0_agg_0": np.asarray(r2.column("l_extendedprice")) *
        (1 - np.asarray(r2.column("l_discount")))})
    agg = GPUAggExecutor(["c_custkey"], [], "sum(e0_agg_0) as revenue")
    agg.execute([partial], 0, 0)
    groups = agg.done(0)
    topk = GPUTopKExecutor(["revenue", "c_custkey"], 20,
                           descending=[True, False])
    topk.execute([groups], 0, 0)
    out = topk.done(0)
    got = list(zip(out.column("c_custkey").to_pylist(),
                   out.column("revenue").to_pylist()))
    assert len(got) == len(want)
    for (gc, gr), (wc, wr) in zip(got, want):
        assert gc == wc
        np.testing.assert_allclose(gr, wr, rtol=1e-9)


def test_gpu_partition_fn_range_and_broadcast(gpu):
    """The reference's other two partitioners (quokka_runtime.py:234-246):
    range = (key-1) // (total_range // N) (clamped at the edges where the
    reference would misroute out-of-range keys), broadcast = full table
    to every channel."""
    import pyarrow as pa
    from quokka_amd import gpu_partition_fn
    rng = np.random.default_rng(65)
    keys = rng.integers(1, 10_001, 30_000).astype(np.int64)
    vals = rng.random(30_000)
    t = pa.table({"k": keys, "v": vals})
    nch = 4
    out = gpu_partition_fn(t, 0, nch, key="k", partitioner="range",
                           total_range=10_000)
    per = 10_000 // nch
    want = np.clip((keys - 1) // per, 0, nch - 1)
    total = 0
    for ch, tbl in out.items():
        gk = np.asarray(tbl.column("k"))
        assert np.all(np.clip((gk - 1) // per, 0, nch - 1) == ch)
        total += len(tbl)
    assert total == len(keys)
    assert sorted(out) == sorted(set(want.tolist()))
    # edge clamping: keys outside [1, total_range]
    t2 = pa.table({"k": np.array([-5, 0, 1, 10_000, 10_001, 99_999],
                                 dtype=np.int64)})
    out2 = gpu_partition_fn(t2, 0, nch, key="k", partitioner="range",
                            total_range=10_000)
    assert np.asarray(out2[0].column("k")).tolist() == [-5, 0, 1]
    assert np.asarray(out2[nch - 1].column("k")).tolist() == [
        10_000, 10_001, 99_999]
    # broadcast
    outb = gpu_partition_fn(t, 0, 3, partitioner="broadcast")
    assert sorted(outb) == [0, 1, 2]
    for ch in range(3):
        assert outb[ch] is t


def test_distributed_sort_via_range_partition(gpu):
    """The reference's global-sort recipe (sort_info 'range' mode,
    unordered_readers.py:301-309 + SuperFastSortExecutor
    sql_executors.py:88): range-partition rows by key so channel i holds
    keys below channel i+1, sort each channel on device, concatenate in
    channel order -> globally sorted."""
    import pyarrow as pa
    from quokka_amd import gpu_partition_fn, ops, shim
    rng = np.random.default_rng(67)
    n = 200_000
    keys = rng.integers(1, 1_000_001, n).astype(np.int64)
    vals = rng.random(n)
    t = pa.table({"k": keys, "v": vals})
    nch = 4
    parts = gpu_partition_fn(t, 0, nch, key="k", partitioner="range",
                             total_range=1_000_000)
    out_k, out_v = [], []
    for ch in range(nch):
        if ch not in parts:
            continue
        ck = np.asarray(parts[ch].column("k"))
        cv = np.asarray(parts[ch].column("v"))
        kcol = shim.DevColumn.from_numpy(ck)
        perm = ops.sort_permutation(kcol)
        sel = perm.to_numpy(perm.n)
        out_k.append(ck[sel])
        out_v.append(cv[sel])
        perm.free()
        kcol.free()
    gk = np.concatenate(out_k)
    gv = np.concatenate(out_v)
    order = np.argsort(keys, kind="stable")
    assert np.array_equal(gk, keys[order])        # globally sorted
    # stability within equal keys is preserved per channel; values must
    # be a permutation attached to the right keys
    assert np.array_equal(np.sort(gv), np.sort(vals))
    lookup = {}
    for k_, v_ in zip(keys.tolist(), vals.tolist()):
        lookup.setdefault(k_, set()).add(round(v_, 12))
    for k_, v_ in zip(gk[:200].tolist(), gv[:200].tolist()):
        assert round(v_, 12) in lookup[k_]


def test_gpu_partition_fn_transforms(gpu):
    """Folded map transforms inside partition_fn (the reference's
    transform_sql batch_funcs, core.py:173-176 + datastream.py:652-815):
    predicate -> JIT elementwise transform -> partition; the computed
    column rides along like any input column."""
    import pyarrow as pa
    from quokka_amd import gpu_partition_fn
    rng = np.random.default_rng(69)
    n = 40_000
    k = rng.integers(0, 10_000, n).astype(np.int64)
    p = np.round(rng.uniform(1, 100, n), 2)
    d = np.round(rng.uniform(0, 0.1, n), 2)
    t = pa.table({"k": k, "p": p, "d": d})
    out = gpu_partition_fn(
        t, 0, 3, key="k", predicate="d < 0.05",
        transforms=[("revenue", "p * (1 - d)")],
        projection=["k", "revenue"])
    mask = d < 0.05
    want = {}
    for kk, rr in zip(k[mask].tolist(), (p[mask] * (1 - d[mask])).tolist()):
        want.setdefault(kk % 3, []).append((kk, round(rr, 9)))
    total = 0
    for ch, tbl in out.items():
        assert tbl.column_names == ["k", "revenue"]
        gk = np.asarray(tbl.column("k"))
        gr = np.asarray(tbl.column("revenue"))
        assert np.all(gk % 3 == ch)
        got = sorted(zip(gk.tolist(), [round(v, 9) for v in gr.tolist()]))
        assert got == sorted(want.get(ch, [])), ch
        total += len(tbl)
    assert total == int(mask.sum())


def test_repartition_overlapped_world1_multiset(gpu):
    """Chunked overlapped self-exchange (exchange.repartition_overlapped):
    every row delivered exactly once across the chunk-major layout, and
    each consume() chunk view holds exactly its slice."""
    from quokka_amd import exchange, shim
    rng = np.random.default_rng(43)
    n = 123_457
    keys = rng.integers(0, 10_000, n).astype(np.int64)
    vals = rng.random(n)
    kcol = shim.DevColumn.from_numpy(keys)
    vcol = shim.DevColumn.from_numpy(vals)
    comm = exchange.Comm(0, 1)
    comp, cstr = shim.Stream(), shim.Stream()
    seen_chunks = []

    def consume(views, start, nrows, j):
        comp.sync()                     # test-only: read views host-side
        seen_chunks.append((start, nrows, views["__key__"].to_numpy(),
                            views["v"].to_numpy()))

    rk, rp, chunk_start, chunk_rows = exchange.repartition_overlapped(
        comm, kcol, {"v": vcol}, comp, cstr, consume, nchunks=5)
    got_k = rk.to_numpy(rk.n)
    got_v = rp["v"].to_numpy(rp["v"].n)
    # full multiset preserved
    og, ow = np.lexsort((got_v, got_k)), np.lexsort((vals, keys))
    assert np.array_equal(got_k[og], keys[ow])
    np.testing.assert_allclose(got_v[og], vals[ow], rtol=0)
    # chunks tile the receive buffer and the views match it
    assert len(seen_chunks) == 5
    cursor = 0
    for (start, nrows, ck, cv), cs, cr in zip(seen_chunks, chunk_start,
                                              chunk_rows):
        assert start == int(cs) == cursor and nrows == int(cr)
        assert np.array_equal(ck, got_k[start:start + nrows])
        np.testing.assert_allclose(cv, got_v[start:start + nrows], rtol=0)
        cursor += nrows
    assert cursor == n
    rk.free(); rp["v"].free(); kcol.free(); vcol.free()
    comp.destroy(); cstr.destroy()
    comm.destroy()


def test_q3_overlapped_world1_parity(gpu, data):
    """The bench's overlapped exchange step at world 1: orders exchanged
    plain + rebuilt, lineitem repartition overlapped with the fused probe
    chunk pipeline == oracle Q3 (the configs[3] overlap path on one GPU)."""
    from quokka_amd import queries as DQ, staging, exchange, shim
    li, orders, cust = data["lineitem"], data["orders"], data["customer"]
    lcols = staging.stage_columns(li, names=["l_orderkey", "l_shipdate",
                                             "l_extendedprice", "l_discount"])
    ocols = staging.stage_columns(orders)
    ccols = staging.stage_columns(cust, names=["c_custkey", "c_mktsegment"])
    comm = exchange.Comm(0, 1)
    comp, cstr = shim.Stream(), shim.Stream()
    ok, op = exchange.repartition(
        comm, ocols["o_orderkey"],
        {k: v for k, v in ocols.items() if k != "o_orderkey"}, comp)
    od_x = {"o_orderkey": ok, **op}
    fused = DQ.Q3Fused(od_x, ccols, comp)

    def consume(views, start, nrows, j):
        fused.probe({"l_orderkey": views["__key__"],
                     "l_shipdate": views["l_shipdate"],
                     "l_extendedprice": views["l_extendedprice"],
                     "l_discount": views["l_discount"]})

    rk, rp, _, _ = exchange.repartition_overlapped(
        comm, lcols["l_orderkey"],
        {k: v for k, v in lcols.items() if k != "l_orderkey"},
        comp, cstr, consume, nchunks=4)
    full, top10 = fused.extract()
    wfull, wtop = OQ.q3(li, orders, cust)
    og = np.argsort(full["l_orderkey"])
    ow = np.argsort(wfull["l_orderkey"])
    assert np.array_equal(full["l_orderkey"][og], wfull["l_orderkey"][ow])
    np.testing.assert_allclose(full["revenue"][og], wfull["revenue"][ow],
                               rtol=1e-9)
    assert np.array_equal(top10["l_orderkey"], wtop["l_orderkey"])
    np.testing.assert_allclose(top10["revenue"], wtop["revenue"], rtol=1e-9)
    fused.free()
    for c in ([rk] + list(rp.values()) + list(od_x.values()) +
              list(lcols.values()) + list(ocols.values()) +
              list(ccols.values())):
        c.free()
    comp.destroy(); cstr.destroy()
    comm.destroy()


def test_groupby_extract_where_gt(gpu):
    """Thresholded (HAVING) extract: only groups with sums[col] > t come
    back, values exact, rerun-on-undershoot counted not truncated."""
    from quokka_amd import ops, shim
    rng = np.random.default_rng(61)
    keys = rng.integers(0, 5000, 100_000).astype(np.int64)
    vals = rng.random(100_000)
    gb = ops.GroupByI64(expected_groups=5000, nvals=2)
    kc = shim.DevColumn.from_numpy(keys)
    vc = shim.DevColumn.from_numpy(vals)
    gb.update(kc, [vc, vc])
    want_k, want_s = gb.extract()
    t = float(np.median(want_s[0]))
    # tiny out_guess forces the rerun path
    got_k, got_s = gb.extract_where_gt(0, t, out_guess=16)
    m = want_s[0] > t
    ow, og = np.argsort(want_k[m]), np.argsort(got_k)
    assert np.array_equal(got_k[og], want_k[m][ow])
    np.testing.assert_allclose(got_s[0][og], want_s[0][m][ow], rtol=0)
    np.testing.assert_allclose(got_s[1][og], want_s[1][m][ow], rtol=0)
    kc.free(); vc.free()
    gb.free()


def test_gen_aux_distributions(gpu):
    """qk_gen_aux device mirrors of the oracle's independent draws:
    uniform shipmode codes, bernoulli comment flag, uniform acctbal
    (tpch_gen.py:231/:165/:250) — distributional match at n=2M."""
    from quokka_amd.shim import DevColumn, c_u64, c_i64
    shim = gpu
    n = 2_000_000
    mode = DevColumn(np.uint8, n)
    shim.call("qk_gen_aux", None, c_u64(n), c_u64(0), c_u64(42),
              c_u64(0x5A1D), 0, c_i64(7), c_i64(0), mode.ptr, None)
    counts = np.bincount(mode.to_numpy(n), minlength=7)
    assert len(counts) == 7
    assert abs(counts / n - 1 / 7).max() < 0.002       # uniform 0..6
    flag = DevColumn(np.uint8, n)
    shim.call("qk_gen_aux", None, c_u64(n), c_u64(0), c_u64(42),
              c_u64(0xC033), 1, c_i64(19000), c_i64(0), flag.ptr, None)
    fl = flag.to_numpy(n)
    assert set(np.unique(fl)) <= {0, 1}
    assert abs(fl.mean() - 0.019) < 0.001              # bernoulli p=.019
    bal = DevColumn(np.float64, n)
    shim.call("qk_gen_aux", None, c_u64(n), c_u64(0), c_u64(42),
              c_u64(0xACC7), 2, c_i64(-99999), c_i64(999999), None,
              bal.ptr)
    b = bal.to_numpy(n)
    assert b.min() >= -999.99 and b.max() <= 9999.99
    assert abs(b.mean() - 4500.0) < 25                 # U[-999.99,9999.99]
    assert np.allclose(b * 100, np.round(b * 100))     # exact cents
    # determinism + row_offset consistency: second half regenerated at
    # an offset equals the tail of the full draw
    half = DevColumn(np.uint8, n // 2)
    shim.call("qk_gen_aux", None, c_u64(n // 2), c_u64(n // 2), c_u64(42),
              c_u64(0x5A1D), 0, c_i64(7), c_i64(0), half.ptr, None)
    assert np.array_equal(half.to_numpy(n // 2),
                          mode.to_numpy(n)[n // 2:])
    for c in (mode, flag, bal, half):
        c.free()


def test_filter_f64_threshold(gpu):
    """qk_filter_f64: runtime-threshold f64 compaction (Q22's '> avg'
    cut) — row set and order vs numpy on all six ops."""
    from quokka_amd import ops
    from quokka_amd.shim import DevColumn
    rng = np.random.default_rng(31)
    a = np.round(rng.uniform(-100, 100, 100_000), 2)
    a[rng.integers(0, len(a), 500)] = 3.5          # exact-equality hits
    col = DevColumn.from_numpy(a)
    npop = {ops.LT: np.less, ops.LE: np.less_equal, ops.GT: np.greater,
            ops.GE: np.greater_equal, ops.EQ: np.equal,
            ops.NE: np.not_equal}
    for op, f in npop.items():
        idx, n = ops.filter_col(col, op, 3.5)
        want = np.nonzero(f(a, 3.5))[0]
        assert n == len(want)
        assert np.array_equal(idx.to_numpy(n), want)
        idx.free()
    col.free()
