"""GPU parity for the Parquet decode kernels: the same file shapes the
CPU plan tests pin (tests/test_parquet_cpu.py) decoded by the real
qk_pq_plain_copy / qk_pq_rle_expand kernels via parquet_gpu.read_table,
compared against pyarrow's decode (the reference's reader,
pyquokka/dataset.py). Ends with Parquet -> HBM -> Q1 vs the oracle."""
import io

import numpy as np
import pytest

pa = pytest.importorskip("pyarrow")
import pyarrow.parquet as pq  # noqa: E402

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def gpu():
    from quokka_amd import shim
    shim.init(0)
    return shim


def write(table, **kw):
    buf = io.BytesIO()
    kw.setdefault("compression", "NONE")
    kw.setdefault("data_page_version", "1.0")
    pq.write_table(table, buf, **kw)
    return buf.getvalue()


def roundtrip(gpu, table, **kw):
    from quokka_amd import parquet_gpu as P
    raw = write(table, **kw)
    cols = P.read_table(raw)
    try:
        for name in table.schema.names:
            want = table.column(name).to_numpy()
            got = cols[name]
            if isinstance(got, tuple):
                codes, cb = got
                vals = np.asarray(cb)[codes.to_numpy(codes.n)]
                np.testing.assert_array_equal(vals, want, err_msg=name)
            else:
                np.testing.assert_array_equal(got.to_numpy(got.n), want,
                                              err_msg=name)
    finally:
        for c in cols.values():
            (c[0] if isinstance(c, tuple) else c).free()


def test_plain_mixed_types(gpu):
    rng = np.random.default_rng(0)
    n = 300_000
    roundtrip(gpu, pa.table({
        "i64": rng.integers(-1 << 40, 1 << 40, n),
        "f64": rng.random(n),
        "i32": rng.integers(-1 << 30, 1 << 30, n).astype(np.int32)}),
        use_dictionary=False)


def test_dict_small_and_wide_bitwidths(gpu):
    rng = np.random.default_rng(1)
    n = 400_000
    roundtrip(gpu, pa.table({
        "narrow": rng.integers(0, 7, n),          # bw 3
        "mid": rng.integers(0, 300, n),           # bw 9
        "wide": rng.integers(0, 20_000, n)}),     # bw 15
        use_dictionary=True)


def test_dict_fallback_mid_chunk(gpu):
    rng = np.random.default_rng(2)
    roundtrip(gpu, pa.table({"k": rng.integers(0, 1 << 30, 150_000)}),
              use_dictionary=True, dictionary_pagesize_limit=4096)


def test_multi_row_group_and_v2(gpu):
    rng = np.random.default_rng(3)
    t = pa.table({"a": rng.integers(0, 1 << 40, 250_000),
                  "k": rng.integers(0, 37, 250_000)})
    roundtrip(gpu, t, use_dictionary=["k"], row_group_size=40_000)
    roundtrip(gpu, t, use_dictionary=["k"], data_page_version="2.0")


def test_string_dict_codes(gpu):
    rng = np.random.default_rng(4)
    vals = np.array(["BUILDING", "AUTOMOBILE", "MACHINERY", "HOUSEHOLD",
                     "FURNITURE"])
    s = vals[rng.integers(0, 5, 200_000)]
    roundtrip(gpu, pa.table({"seg": s}), use_dictionary=True)


def test_parquet_to_q1_vs_oracle(gpu):
    """End-to-end scan row: TPC-H lineitem written as
    Parquet, decoded by the GPU kernels straight into HBM columns, fused
    Q1 on them, parity vs the CPU oracle on the same host data."""
    from quokka_amd import parquet_gpu as P, queries as DQ
    from oracle import tpch_gen as G, queries as OQ
    li = G.gen_lineitem(0.01, seed=11)
    table = pa.table({
        "l_shipdate": li["l_shipdate"],
        "l_quantity": li["l_quantity"],
        "l_extendedprice": li["l_extendedprice"],
        "l_discount": li["l_discount"],
        "l_tax": li["l_tax"],
        "l_returnflag": li["l_returnflag"].astype(np.int32),
        "l_linestatus": li["l_linestatus"].astype(np.int32)})
    raw = write(table, use_dictionary=False)
    cols = P.read_table(raw)
    # Q1 wants u8 flag/status codes; decode returns i32 — narrow on host
    # copy for the two tiny dict-code columns (u8 staging is the staging
    # module's job; this test pins the decode+compute path)
    from quokka_amd import staging
    dev = {k: cols[k] for k in ("l_shipdate", "l_quantity",
                                "l_extendedprice", "l_discount", "l_tax")}
    for k in ("l_returnflag", "l_linestatus"):
        narrowed = cols[k].to_numpy(cols[k].n).astype(np.uint8)
        dev[k] = gpu.DevColumn.from_numpy(narrowed)
        cols[k].free()
    got = DQ.q1(dev)
    want = OQ.q1(li)
    assert list(got["l_returnflag"]) == list(want["l_returnflag"])
    np.testing.assert_array_equal(got["count_order"], want["count_order"])
    for c in ("sum_qty", "sum_base_price", "sum_disc_price", "sum_charge",
              "avg_qty", "avg_price", "avg_disc"):
        np.testing.assert_allclose(got[c], want[c], rtol=1e-9)
    for c in dev.values():
        c.free()


def test_fuzz_random_files_gpu(gpu):
    """Same seeded random-file family as the CPU plan fuzz
    (tests/test_parquet_cpu.py::test_fuzz_random_files), decoded by the
    real kernels."""
    from quokka_amd import parquet_gpu as P
    for case in range(12):
        rng = np.random.default_rng(7000 + case)
        n = int(rng.integers(1, 60_000))
        cols = {}
        for c in range(int(rng.integers(1, 4))):
            kind = int(rng.integers(0, 4))
            name = "c%d" % c
            if kind == 0:
                cols[name] = rng.integers(-1 << 50, 1 << 50, n)
            elif kind == 1:
                cols[name] = rng.random(n)
            elif kind == 2:
                cols[name] = rng.integers(-1 << 20, 1 << 20,
                                          n).astype(np.int32)
            else:
                card = int(rng.integers(1, 200))
                cols[name] = rng.integers(0, card, n)
        t = pa.table(cols)
        kw = {}
        if rng.integers(0, 2):
            kw["use_dictionary"] = bool(rng.integers(0, 2))
        if rng.integers(0, 2):
            kw["row_group_size"] = int(rng.integers(1, n + 1))
        if rng.integers(0, 2):
            kw["data_page_size"] = int(rng.integers(256, 1 << 16))
        if rng.integers(0, 2):
            kw["data_page_version"] = "2.0"
        raw = write(t, **kw)
        got = P.read_table(raw)
        try:
            for name in t.schema.names:
                want = t.column(name).to_numpy()
                col = got[name]
                np.testing.assert_array_equal(
                    col.to_numpy(col.n), want,
                    err_msg="case %d col %s kw %r" % (case, name, kw))
        finally:
            for c in got.values():
                (c[0] if isinstance(c, tuple) else c).free()


def test_column_selection(gpu):
    from quokka_amd import parquet_gpu as P
    rng = np.random.default_rng(9)
    t = pa.table({"a": rng.integers(0, 1000, 5000),
                  "b": rng.random(5000),
                  "c": rng.integers(0, 9, 5000)})
    raw = write(t, use_dictionary=False)
    cols = P.read_table(raw, columns=["b", "c"])
    assert set(cols) == {"b", "c"}
    np.testing.assert_array_equal(cols["b"].to_numpy(5000),
                                  t.column("b").to_numpy())
    for c in cols.values():
        c.free()


# ---------- GPU snappy decompression ------------------------------------

def _mix_table(n, rng):
    return pa.table({
        "f": rng.random(n),
        "i": rng.integers(-1 << 40, 1 << 40, n).astype(np.int64),
        "d": rng.integers(8000, 11000, n).astype(np.int32),
        "k": rng.integers(0, 37, n).astype(np.int64),    # dict-friendly
        "s": pa.array(np.array(["MAIL", "SHIP", "RAIL", "AIR"])[
            rng.integers(0, 4, n)]).dictionary_encode(),
    })


def test_snappy_roundtrip_v1(gpu):
    rng = np.random.default_rng(21)
    t = _mix_table(150_000, rng)
    roundtrip(gpu, t, compression="SNAPPY", use_dictionary=["k", "s"])


def test_snappy_roundtrip_v1_nonnullable(gpu):
    """max_def == 0: no levels block inside the compressed pages."""
    rng = np.random.default_rng(22)
    t = _mix_table(100_000, rng)
    t = t.cast(pa.schema([pa.field(f.name, f.type, nullable=False)
                          for f in t.schema]))
    roundtrip(gpu, t, compression="SNAPPY", use_dictionary=["k", "s"])


def test_snappy_roundtrip_v2(gpu):
    """v2 pages: levels uncompressed in-file, data snappy-compressed."""
    rng = np.random.default_rng(23)
    t = _mix_table(120_000, rng)
    roundtrip(gpu, t, compression="SNAPPY", use_dictionary=["k", "s"],
              data_page_version="2.0")


def test_snappy_multi_row_group_and_dict_fallback(gpu):
    """Multiple row groups + tiny dictionary-page limit (forces the
    mid-chunk dictionary -> PLAIN fallback inside compressed chunks)."""
    rng = np.random.default_rng(24)
    t = _mix_table(200_000, rng)
    roundtrip(gpu, t, compression="SNAPPY", use_dictionary=True,
              row_group_size=60_000, dictionary_pagesize_limit=4096)


def test_snappy_highly_compressible(gpu):
    """Long runs -> snappy copy elements with small offsets (the
    overlapped-match path) and large back-references."""
    n = 300_000
    rep = np.tile(np.arange(50, dtype=np.int64), n // 50)
    const = np.full(n, 3.14159)
    txt = pa.array(np.array(["AAAA"] * n)).dictionary_encode()
    t = pa.table({"rep": rep, "const": const, "txt": txt})
    roundtrip(gpu, t, compression="SNAPPY", use_dictionary=["txt"])


def test_snappy_lineitem_q1_vs_oracle(gpu):
    """Compressed lineitem end-to-end: snappy parquet -> GPU decompress +
    decode -> fused Q1 == oracle on the same rows."""
    from oracle import tpch_gen as G, queries as OQ
    from quokka_amd import parquet_gpu as P, queries as DQ, shim
    from quokka_amd.shim import DevColumn, c_u64
    li = G.gen_lineitem(0.03, seed=77)
    t = pa.table({
        "l_quantity": li["l_quantity"],
        "l_extendedprice": li["l_extendedprice"],
        "l_discount": li["l_discount"],
        "l_tax": li["l_tax"],
        "l_shipdate": pa.array(li["l_shipdate"], type=pa.int32()),
        "l_returnflag": pa.array(
            np.array(G.RETURNFLAG)[li["l_returnflag"]]).dictionary_encode(),
        "l_linestatus": pa.array(
            np.array(G.LINESTATUS)[li["l_linestatus"]]).dictionary_encode(),
    })
    raw = write(t, compression="SNAPPY",
                use_dictionary=["l_returnflag", "l_linestatus"])
    dec = P.read_table(raw)

    def canon(name, order):
        codes_u32, values = dec[name]
        m = np.zeros(max(1, len(values)), dtype=np.uint8)
        for i, v in enumerate(values):
            m[i] = order.index(v)
        mcol = DevColumn.from_numpy(m)
        out = DevColumn(np.uint8, codes_u32.n)
        shim.call("qk_gather_u8", None, c_u64(codes_u32.n), codes_u32.ptr,
                  mcol.ptr, out.ptr)
        out.n = codes_u32.n
        mcol.free(); codes_u32.free()
        return out
    dec["l_returnflag"] = canon("l_returnflag", G.RETURNFLAG)
    dec["l_linestatus"] = canon("l_linestatus", G.LINESTATUS)
    got = DQ.q1(dec)
    want = OQ.q1(li)
    assert np.array_equal(got["count_order"], want["count_order"])
    for c in ("sum_qty", "sum_charge", "avg_disc"):
        np.testing.assert_allclose(got[c], want[c], rtol=1e-9)
    for c in dec.values():
        c.free()


# ---------- GPU gzip (DEFLATE) decompression ----------------------------

def _nn(t):
    return t.cast(pa.schema([pa.field(f.name, f.type, nullable=False)
                             for f in t.schema]))


def test_gzip_roundtrip_v1_nonnullable(gpu):
    rng = np.random.default_rng(26)
    t = _nn(_mix_table(150_000, rng))
    roundtrip(gpu, t, compression="GZIP", use_dictionary=["k", "s"])


def test_gzip_roundtrip_v2(gpu):
    """v2: levels uncompressed in-file; nullable schema allowed."""
    rng = np.random.default_rng(27)
    t = _mix_table(120_000, rng)
    roundtrip(gpu, t, compression="GZIP", use_dictionary=["k", "s"],
              data_page_version="2.0")


def test_gzip_multi_row_group_and_dict_fallback(gpu):
    rng = np.random.default_rng(28)
    t = _nn(_mix_table(200_000, rng))
    roundtrip(gpu, t, compression="GZIP", use_dictionary=True,
              row_group_size=60_000, dictionary_pagesize_limit=4096)


def test_gzip_highly_compressible(gpu):
    """Long runs exercise long matches and dynamic Huffman tables."""
    n = 300_000
    rep = np.tile(np.arange(50, dtype=np.int64), n // 50)
    const = np.full(n, 2.5)
    t = _nn(pa.table({"rep": rep, "const": const}))
    roundtrip(gpu, t, compression="GZIP", use_dictionary=False)


def test_gzip_nullable_v1_raises(gpu):
    """Nullable v1 + GZIP keeps its levels inside the compressed stream:
    refused with an actionable message (write non-nullable or v2)."""
    from quokka_amd import parquet_gpu as P
    rng = np.random.default_rng(29)
    t = _mix_table(10_000, rng)          # nullable schema
    raw = write(t, compression="GZIP", use_dictionary=["k", "s"])
    with pytest.raises(P.QkParquetError, match="non-nullable|2.0"):
        P.read_table(raw)


def test_gzip_lineitem_q6(gpu):
    """GZIP lineitem -> GPU inflate + decode -> Q6 == oracle."""
    from oracle import tpch_gen as G, queries as OQ
    from quokka_amd import parquet_gpu as P, queries as DQ
    li = G.gen_lineitem(0.02, seed=91)
    t = _nn(pa.table({
        "l_quantity": li["l_quantity"],
        "l_extendedprice": li["l_extendedprice"],
        "l_discount": li["l_discount"],
        "l_shipdate": pa.array(li["l_shipdate"], type=pa.int32())}))
    raw = write(t, compression="GZIP", use_dictionary=False)
    dec = P.read_table(raw)
    got = DQ.q6(dec)
    want = OQ.q6(li)
    np.testing.assert_allclose(got["revenue"], want["revenue"], rtol=1e-9)
    assert got["rows_passed"] == want["rows_passed"]
    for c in dec.values():
        c.free()


@pytest.mark.parametrize("codec", ["NONE", "SNAPPY", "GZIP"])
def test_single_row_and_tiny_tables(gpu, codec):
    """Degenerate sizes through every codec path."""
    t = pa.table({"a": np.array([3.25]), "k": np.array([7], np.int64)})
    t = t.cast(pa.schema([pa.field(f.name, f.type, nullable=False)
                          for f in t.schema]))
    roundtrip(gpu, t, compression=codec, use_dictionary=["k"])
    t2 = pa.table({"x": np.arange(3, dtype=np.int64)})
    t2 = t2.cast(pa.schema([pa.field("x", pa.int64(), nullable=False)]))
    roundtrip(gpu, t2, compression=codec, use_dictionary=False)
