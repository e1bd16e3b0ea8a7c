"""CPU tests: oracle vs committed golden fixtures + semantics properties."""
import json
import os

import numpy as np
import pytest

from oracle import tpch_gen as G, queries as Q, executors as E

GOLDEN = os.path.join(os.path.dirname(__file__), "golden", "golden.json")


@pytest.fixture(scope="module")
def data():
    return G.gen_all(0.01, 42)


@pytest.fixture(scope="module")
def golden():
    with open(GOLDEN) as f:
        return json.load(f)


def _floats(lst):
    return np.array([float(x) for x in lst])


def test_generator_deterministic(data, golden):
    assert len(data["lineitem"]["l_orderkey"]) == golden["lineitem_rows"]
    d2 = G.gen_lineitem(0.01, 42)
    for k in d2:
        assert np.array_equal(d2[k], data["lineitem"][k]), k


def test_q1_matches_golden(data, golden):
    r = Q.q1(data["lineitem"])
    g = golden["q1"]
    assert list(r["l_returnflag"]) == g["l_returnflag"]
    assert list(r["l_linestatus"]) == g["l_linestatus"]
    for c in ("sum_qty", "sum_base_price", "sum_disc_price", "sum_charge",
              "avg_qty", "avg_price", "avg_disc"):
        np.testing.assert_allclose(r[c], _floats(g[c]), rtol=1e-12)
    assert list(r["count_order"]) == g["count_order"]


def test_q6_matches_golden(data, golden):
    r = Q.q6(data["lineitem"])
    assert r["rows_passed"] == golden["q6"]["rows_passed"]
    np.testing.assert_allclose(r["revenue"], float(golden["q6"]["revenue"]),
                               rtol=1e-12)


def test_q3_matches_golden(data, golden):
    full, top10 = Q.q3(data["lineitem"], data["orders"], data["customer"])
    g = golden["q3_top10"]
    assert len(full["l_orderkey"]) == golden["q3_n_groups"]
    assert list(top10["l_orderkey"]) == g["l_orderkey"]
    np.testing.assert_allclose(top10["revenue"], _floats(g["revenue"]),
                               rtol=1e-12)


def test_q5_matches_golden(data, golden):
    r = Q.q5(data["lineitem"], data["orders"], data["customer"],
             data["supplier"], data["nation"], data["region"])
    g = golden["q5"]
    assert [n for n, _ in r] == [n for n, _ in g]
    np.testing.assert_allclose([v for _, v in r],
                               [float(v) for _, v in g], rtol=1e-12)


def test_q1_two_phase_equals_single_pass(data):
    """Partial-agg-per-batch then final == one pass (the executor
    decomposition, sql_executors.py:587-599 + sql_utils rewrite)."""
    li = data["lineitem"]
    n = len(li["l_orderkey"])
    acc = np.zeros((6, 6))
    for lo in range(0, n, 7919):
        chunk = {k: v[lo:lo + 7919] for k, v in li.items()}
        acc += Q.q1_partials(chunk)
    r1 = Q.q1_finalize(acc)
    r2 = Q.q1(li)
    for c in ("sum_qty", "sum_disc_price", "avg_disc"):
        np.testing.assert_allclose(r1[c], r2[c], rtol=1e-12)
    assert np.array_equal(r1["count_order"], r2["count_order"])


def test_q1_cutoff_boundary():
    """<= is inclusive: a row exactly at the cutoff is counted."""
    li = {
        "l_shipdate": np.array([G.Q1_CUTOFF, G.Q1_CUTOFF + 1], np.int32),
        "l_quantity": np.array([1.0, 1.0]),
        "l_extendedprice": np.array([10.0, 10.0]),
        "l_discount": np.array([0.0, 0.0]),
        "l_tax": np.array([0.0, 0.0]),
        "l_returnflag": np.array([0, 0], np.uint8),
        "l_linestatus": np.array([0, 0], np.uint8),
    }
    r = Q.q1(li)
    assert list(r["count_order"]) == [1]


def test_q1_empty_and_all_filtered():
    li = {k: np.empty(0, dt) for k, dt in [
        ("l_shipdate", np.int32), ("l_quantity", np.float64),
        ("l_extendedprice", np.float64), ("l_discount", np.float64),
        ("l_tax", np.float64), ("l_returnflag", np.uint8),
        ("l_linestatus", np.uint8)]}
    r = Q.q1(li)
    assert len(r["count_order"]) == 0
    li2 = {**li, "l_shipdate": np.array([G.Q1_CUTOFF + 5], np.int32)}
    for k in li:
        if k != "l_shipdate":
            li2[k] = np.array([1], li[k].dtype) if li[k].dtype == np.uint8 \
                else np.array([1.0], li[k].dtype)
    r2 = Q.q1(li2)
    assert len(r2["count_order"]) == 0


def test_join_oracle_against_bruteforce():
    rng = np.random.default_rng(0)
    bk = rng.integers(0, 50, 200).astype(np.int64)   # dup build keys
    pk = rng.integers(0, 80, 300).astype(np.int64)   # some missing
    pi, bi = E.build_probe_join(bk, pk, "inner")
    # brute force
    want = set()
    for i, k in enumerate(pk):
        for j, kb in enumerate(bk):
            if k == kb:
                want.add((i, j))
    assert set(zip(pi.tolist(), bi.tolist())) == want
    semi = E.build_probe_join(bk, pk, "semi")
    assert set(semi.tolist()) == {i for i, _ in want}
    anti = E.build_probe_join(bk, pk, "anti")
    assert set(anti.tolist()) == set(range(300)) - {i for i, _ in want}
    # left: matched pairs + unmatched with -1
    pl, bl = E.build_probe_join(bk, pk, "left")
    got_pairs = {(p, b) for p, b in zip(pl.tolist(), bl.tolist()) if b >= 0}
    assert got_pairs == want
    assert {p for p, b in zip(pl.tolist(), bl.tolist()) if b < 0} == \
        set(anti.tolist())


def test_partition_int_semantics():
    keys = np.array([0, 1, 7, 8, 15, 16, 123456789], np.int64)
    assert list(E.partition_int(keys, 8)) == [0, 1, 7, 0, 7, 0,
                                              123456789 % 8]
