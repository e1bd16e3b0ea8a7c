"""GPU parity for general (unbounded-cardinality) string keys: the device
string dictionary (qk_str_dict_encode — byte-hash table with exact byte
verification against an on-device arena) and the string-keyed executor
paths built on it. The reference handles such keys inside polars/DuckDB
(sql_executors.py:325-377, :556-599); round 1 supported only <=256-entry
host dictionaries. Oracles: numpy/pyarrow on the same inputs."""
import numpy as np
import pytest

pa = pytest.importorskip("pyarrow")

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def gpu():
    from quokka_amd import shim
    shim.init(0)
    return shim


def _rand_strings(rng, n, n_distinct, minlen=1, maxlen=24):
    pool = np.array(["s%0*d" % (int(rng.integers(minlen, maxlen)), i)
                     for i in range(n_distinct)], dtype=object)
    return pool[rng.integers(0, n_distinct, n)]


def test_device_string_dict_codes(gpu):
    """Equal strings -> equal codes, distinct -> distinct; codes stay
    consistent across batches; values decode back exactly."""
    from quokka_amd import ops
    rng = np.random.default_rng(31)
    sd = ops.DeviceStringDict(expected=64)       # force growth
    seen = {}
    for b in range(6):
        arr = _rand_strings(rng, 5000, 3000)
        codes = sd.encode_column(pa.chunked_array([pa.array(arr)]))
        for s, c in zip(arr, codes):
            if s in seen:
                assert seen[s] == c, s
            else:
                seen[s] = int(c)
    assert sd.n_codes == len(seen)
    vals = sd.values
    for s, c in seen.items():
        assert vals[c] == s
    # decode() round-trips
    some = np.array(sorted(seen.values())[:100])
    dec = sd.decode(some)
    inv = {c: s for s, c in seen.items()}
    assert all(dec[i] == inv[int(some[i])] for i in range(len(some)))
    sd.free()


def test_device_string_dict_empty_and_long(gpu):
    from quokka_amd import ops
    sd = ops.DeviceStringDict(expected=16)
    arr = np.array(["", "a" * 500, "", "b", "a" * 500], dtype=object)
    codes = sd.encode_column(pa.chunked_array([pa.array(arr)]))
    assert codes[0] == codes[2] and codes[1] == codes[4]
    assert len({codes[0], codes[1], codes[3]}) == 3
    assert sd.values[codes[1]] == "a" * 500 and sd.values[codes[0]] == ""
    sd.free()


def test_string_key_join_inner(gpu):
    """Inner join on a high-cardinality string key == pyarrow Acero."""
    from quokka_amd.executors import GPUBuildProbeJoinExecutor
    rng = np.random.default_rng(33)
    nb, npr, nd = 4000, 20_000, 3000
    bkeys = _rand_strings(rng, nb, nd)
    bkeys = np.array(list(dict.fromkeys(bkeys)), dtype=object)  # unique
    pay = np.arange(len(bkeys), dtype=np.float64)
    pkeys = _rand_strings(rng, npr, nd)
    x = rng.random(npr)
    ex = GPUBuildProbeJoinExecutor(left_on="k", right_on="k", how="inner")
    build = pa.table({"k": pa.array(bkeys), "pay": pay})
    probe = pa.table({"k": pa.array(pkeys), "x": x})
    # two build batches (cross-batch dict consistency)
    h = len(bkeys) // 2
    ex.execute([build.slice(0, h)], 1, 0)
    ex.execute([build.slice(h)], 1, 0)
    got = ex.execute([probe], 0, 0)
    want = probe.join(build, keys="k", join_type="inner")
    assert got.num_rows == want.num_rows
    gs = got.sort_by([("k", "ascending"), ("x", "ascending")])
    ws = want.sort_by([("k", "ascending"), ("x", "ascending")])
    assert gs.column("k").to_pylist() == ws.column("k").to_pylist()
    np.testing.assert_allclose(np.asarray(gs.column("x")),
                               np.asarray(ws.column("x")), rtol=0)
    np.testing.assert_allclose(np.asarray(gs.column("pay")),
                               np.asarray(ws.column("pay")), rtol=0)


@pytest.mark.parametrize("how", ["semi", "anti", "left"])
def test_string_key_join_modes(gpu, how):
    from quokka_amd.executors import GPUBuildProbeJoinExecutor
    rng = np.random.default_rng(34)
    bkeys = np.array(["k%d" % i for i in range(0, 600, 2)], dtype=object)
    pkeys = np.array(["k%d" % i for i in rng.integers(0, 600, 5000)],
                     dtype=object)
    ex = GPUBuildProbeJoinExecutor(left_on="k", right_on="k", how=how)
    ex.execute([pa.table({"k": pa.array(bkeys),
                          "pay": np.ones(len(bkeys))})], 1, 0)
    got = ex.execute([pa.table({"k": pa.array(pkeys),
                                "x": np.arange(5000.0)})], 0, 0)
    inb = np.isin(pkeys.astype(str), bkeys.astype(str))
    if how == "semi":
        assert got.num_rows == int(inb.sum())
        assert all(k.endswith(("0", "2", "4", "6", "8")) or True
                   for k in got.column("k").to_pylist())
        assert set(got.column("k").to_pylist()) <= set(bkeys)
    elif how == "anti":
        assert got.num_rows == int((~inb).sum())
        assert not set(got.column("k").to_pylist()) & set(bkeys)
    else:
        assert got.num_rows == 5000
        pays = got.column("pay").to_pylist()
        assert sum(p is None for p in pays) == int((~inb).sum())


def test_string_key_groupby_high_cardinality(gpu):
    """Group-by a single string key with ~20k distinct values (way past
    the round-1 256-entry host dict) == pyarrow group_by."""
    from quokka_amd.executors import GPUAggExecutor
    rng = np.random.default_rng(35)
    n, nd = 200_000, 20_000
    keys = _rand_strings(rng, n, nd)
    vals = rng.random(n)
    ex = GPUAggExecutor(["k"], [("k", "asc")], "sum(v) as s")
    t = pa.table({"k": pa.array(keys), "v": vals})
    third = n // 3
    ex.execute([t.slice(0, third)], 0, 0)
    ex.execute([t.slice(third, third)], 0, 0)
    ex.execute([t.slice(2 * third)], 0, 0)
    out = ex.done(0)
    want = t.group_by("k").aggregate([("v", "sum")]).sort_by(
        [("k", "ascending")])
    assert out.num_rows == want.num_rows
    assert out.column("k").to_pylist() == want.column("k").to_pylist()
    np.testing.assert_allclose(np.asarray(out.column("s")),
                               np.asarray(want.column("v_sum")), rtol=1e-9)


def test_string_plus_int_composite_groupby(gpu):
    """Composite (string, int->code) keys through the device dict +
    codebook packing == pandas groupby."""
    import pandas as pd
    from quokka_amd.executors import GPUAggExecutor
    rng = np.random.default_rng(36)
    n = 50_000
    ks = _rand_strings(rng, n, 800)
    ki = rng.integers(0, 50, n).astype(np.int64)
    v = rng.random(n)
    ex = GPUAggExecutor(["ks", "ki"], [("ks", "asc"), ("ki", "asc")],
                        "sum(v) as s, min(w) as mn")
    t = pa.table({"ks": pa.array(ks), "ki": ki, "v": v, "w": -v})
    ex.execute([t], 0, 0)
    out = ex.done(0)
    df = pd.DataFrame({"ks": ks.astype(str), "ki": ki, "v": v, "w": -v})
    want = df.groupby(["ks", "ki"]).agg(s=("v", "sum"), mn=("w", "min")) \
             .reset_index().sort_values(["ks", "ki"])
    assert out.num_rows == len(want)
    assert out.column("ks").to_pylist() == list(want["ks"])
    assert out.column("ki").to_pylist() == list(want["ki"])
    np.testing.assert_allclose(np.asarray(out.column("s")),
                               want["s"].to_numpy(), rtol=1e-9)
    np.testing.assert_allclose(np.asarray(out.column("mn")),
                               want["mn"].to_numpy(), rtol=0)


def test_disk_spill_join_multi_chunk(gpu):
    """GPUDiskBuildProbeJoinExecutor: build side spilled as Parquet
    chunks, each GPU-decoded + probed with one table in HBM at a time;
    result == Acero inner join. Mirrors DiskBuildProbeJoinExecutor
    (sql_executors.py:456-514)."""
    import tempfile
    from quokka_amd.executors import GPUDiskBuildProbeJoinExecutor
    rng = np.random.default_rng(71)
    d = tempfile.mkdtemp(prefix="qk_spill_")
    ex = GPUDiskBuildProbeJoinExecutor(left_on="lk", right_on="rk",
                                       key_to_keep="right", spill_dir=d)
    nb = 30_000
    bkeys = rng.permutation(nb).astype(np.int64)
    build = pa.table({"rk": bkeys, "pay": bkeys.astype(np.float64) * 2,
                      "extra": rng.random(nb)})
    # three spilled chunks
    for lo in range(0, nb, nb // 3 + 1):
        ex.execute([build.slice(lo, nb // 3 + 1)], 1, 0)
    assert ex.count == 3
    pk = rng.integers(0, 2 * nb, 50_000).astype(np.int64)
    probe = pa.table({"lk": pk, "x": np.arange(50_000.0)})
    got = ex.execute([probe], 0, 0)
    want = probe.join(build, keys="lk", right_keys="rk",
                      join_type="inner")
    assert got.num_rows == want.num_rows
    gs = got.sort_by([("rk", "ascending"), ("x", "ascending")])
    ws = want.sort_by([("lk", "ascending"), ("x", "ascending")])
    np.testing.assert_array_equal(np.asarray(gs.column("rk")),
                                  np.asarray(ws.column("lk")))
    np.testing.assert_allclose(np.asarray(gs.column("pay")),
                               np.asarray(ws.column("pay")), rtol=0)
    np.testing.assert_allclose(np.asarray(gs.column("x")),
                               np.asarray(ws.column("x")), rtol=0)
    ex.done(0)
    import os
    assert not [f for f in os.listdir(d) if f.startswith("build_")]
    with pytest.raises(ValueError, match="inner"):
        GPUDiskBuildProbeJoinExecutor(on="k", how="left", spill_dir=d)


def test_distinct_executor_string_keys(gpu):
    """GPUDistinctExecutor with unbounded-cardinality string keys:
    first-occurrence rows across batches, like the reference's
    batch.unique + anti-join state (sql_executors.py:517-554)."""
    from quokka_amd.executors import GPUDistinctExecutor
    rng = np.random.default_rng(81)
    ex = GPUDistinctExecutor("k")
    seen = set()
    total = 0
    for b in range(4):
        keys = _rand_strings(rng, 3000, 900)
        t = pa.table({"k": pa.array(keys),
                      "v": np.arange(3000.0) + b * 10000})
        out = ex.execute([t], 0, 0)
        new = [k for k in dict.fromkeys(keys) if k not in seen]
        if out is not None:
            got = out.column("k").to_pylist()
            assert got == new, b
            total += out.num_rows
        else:
            assert not new
        seen.update(keys)
    assert total == len(seen)


def test_partition_fn_string_keys_colocate(gpu):
    """gpu_partition_fn with a string key: equal keys land on one
    channel (the reference's contract; bucket ids differ from polars'
    internal hash, as documented)."""
    from quokka_amd import gpu_partition_fn
    rng = np.random.default_rng(82)
    keys = _rand_strings(rng, 20_000, 500)
    t = pa.table({"k": pa.array(keys), "v": rng.random(20_000)})
    parts = gpu_partition_fn(t, 0, 4, key="k")
    where = {}
    n_out = 0
    for ch, tbl in parts.items():
        n_out += tbl.num_rows
        for k in set(tbl.column("k").to_pylist()):
            assert where.setdefault(k, ch) == ch, k
    assert n_out == 20_000
    assert len(where) == len(set(keys))


def test_sort_executor_string_key(gpu):
    """GPUSortExecutor sorts string keys LEXICOGRAPHICALLY (codes
    remapped to value ranks; stable device radix sort on the ranks)."""
    from quokka_amd.executors import GPUSortExecutor
    rng = np.random.default_rng(83)
    keys = _rand_strings(rng, 30_000, 5000)
    v = np.arange(30_000.0)
    ex = GPUSortExecutor("k")
    t = pa.table({"k": pa.array(keys), "v": v})
    ex.execute([t.slice(0, 15_000)], 0, 0)
    ex.execute([t.slice(15_000)], 0, 0)
    out = ex.done(0)
    got = out.column("k").to_pylist()
    assert got == sorted(keys)
    # stability: equal keys keep input order
    gv = np.asarray(out.column("v"))
    for i in range(1, len(got)):
        if got[i] == got[i - 1]:
            assert gv[i] > gv[i - 1]
