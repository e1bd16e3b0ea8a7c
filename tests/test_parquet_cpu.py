"""CPU tests of the GPU-Parquet decode PLAN (thrift page walk,
definition-level no-null verification, PLAIN tiles, RLE/bit-packed run
entries): the plan is executed by a numpy simulator of the qk_pq_*
kernels and compared against pyarrow's own decode of the same file —
pyarrow IS the reference's Parquet reader (pyquokka/dataset.py), so this
pins plan-level parity without a GPU. The GPU suite runs the same files
through the real kernels."""
import io

import numpy as np
import pytest

pa = pytest.importorskip("pyarrow")
import pyarrow.parquet as pq  # noqa: E402

from quokka_amd import parquet_gpu as P  # noqa: E402
from quokka_amd import parquet_thrift as T  # noqa: E402


def write(table, **kw):
    buf = io.BytesIO()
    kw.setdefault("compression", "NONE")
    kw.setdefault("data_page_version", "1.0")
    pq.write_table(table, buf, **kw)
    return buf.getvalue()


def sim_column(raw, ci):
    """Numpy simulation of read_table for one column index."""
    f = pq.ParquetFile(io.BytesIO(raw))
    md = f.metadata
    max_def = md.schema.column(ci).max_definition_level
    total = md.num_rows
    chunks = []
    row = 0
    for rg in range(md.num_row_groups):
        col = md.row_group(rg).column(ci)
        ch = P._Chunk(raw, col, max_def, row)
        row += ch.n
        chunks.append(ch)
    assert row == total
    if chunks[0].is_ba:
        glob = {}
        for ch in chunks:
            for v in ch.dict_vals or []:
                glob.setdefault(v, len(glob))
        out = np.empty(total, dtype=np.uint32)
        for ch in chunks:
            idx, base = sim_rle_pages(raw, ch.rle_pages)
            remap = np.asarray([glob[v] for v in ch.dict_vals],
                               dtype=np.uint32)
            out[base:base + len(idx)] = remap[idx]
        return out, sorted(glob, key=glob.get)
    dt = chunks[0].dtype
    out = np.empty(total, dtype=dt)
    for ch in chunks:
        # plain_tiles holds vectorized per-page blocks (ntiles x 3 u64)
        for blk in ch.plain_tiles:
            for src, dst, cnt in np.atleast_2d(
                    np.asarray(blk, dtype=np.int64)):
                src, dst, cnt = int(src), int(dst), int(cnt)
                out[dst:dst + cnt] = np.frombuffer(raw, dtype=dt,
                                                   count=cnt, offset=src)
        if ch.rle_pages:
            idx, base = sim_rle_pages(raw, ch.rle_pages)
            out[base:base + len(idx)] = ch.dict_vals[idx]
    return out


def sim_rle_pages(raw, pages):
    """Walk each page's run stream with the (test-retained) host walker,
    then simulate the expansion — pins the per-page descriptors the GPU
    kernel consumes against run-level semantics."""
    ents = []
    for src, end, dst, cnt, bw in pages:
        if bw == 0:
            ents.append((0, dst, cnt, 0, 0))
        else:
            P._walk_rle(raw, src, end, bw, cnt, dst, ents)
    return sim_rle(raw, ents)


def sim_rle(raw, ents):
    """Simulate qk_pq_rle_expand (including the 8-byte unaligned load)."""
    e = np.asarray(ents, dtype=np.uint64)
    base = int(e[:, 1].min())
    nv = int(e[:, 2].sum())
    assert int((e[:, 1] + e[:, 2]).max()) - base == nv
    out = np.empty(nv, dtype=np.uint32)
    pad = raw + b"\0" * 8
    for kind, dst, cnt, a, b in e.tolist():
        dst -= base
        if kind == 0:
            out[dst:dst + cnt] = a
        else:
            for i in range(cnt):
                bit = a + i * b
                w = int.from_bytes(pad[bit >> 3:(bit >> 3) + 8], "little")
                out[dst + i] = (w >> (bit & 7)) & ((1 << b) - 1)
    return out, base


def check_fixed(table, name, **kw):
    raw = write(table, **kw)
    ci = table.schema.names.index(name)
    got = sim_column(raw, ci)
    want = table.column(name).to_numpy()
    np.testing.assert_array_equal(got, want)


def test_plain_int64():
    check_fixed(pa.table({"a": np.arange(100_000, dtype=np.int64)}), "a",
                use_dictionary=False)


def test_plain_three_types():
    rng = np.random.default_rng(0)
    t = pa.table({"i": rng.integers(-1 << 40, 1 << 40, 30_000),
                  "d": rng.random(30_000),
                  "s": rng.integers(0, 20_000, 30_000).astype(np.int32)})
    for c in ("i", "d", "s"):
        check_fixed(t, c, use_dictionary=False)


def test_dict_int64():
    rng = np.random.default_rng(1)
    check_fixed(pa.table({"k": rng.integers(0, 50, 100_000)}), "k",
                use_dictionary=True)


def test_dict_high_cardinality_multibyte_bitwidth():
    rng = np.random.default_rng(2)
    check_fixed(pa.table({"k": rng.integers(0, 3000, 200_000)}), "k",
                use_dictionary=True)


def test_dict_fallback_to_plain_mid_chunk():
    """Force pyarrow's dictionary->PLAIN fallback (tiny dict page size):
    the chunk mixes RLE_DICTIONARY and PLAIN pages; the RLE entries must
    be a contiguous prefix and the simulated decode must still match."""
    rng = np.random.default_rng(3)
    t = pa.table({"k": rng.integers(0, 1 << 30, 120_000)})
    check_fixed(t, "k", use_dictionary=True, dictionary_pagesize_limit=4096)


def test_multi_row_group():
    rng = np.random.default_rng(4)
    t = pa.table({"a": rng.integers(0, 1 << 40, 250_000)})
    check_fixed(t, "a", use_dictionary=False, row_group_size=40_000)


def test_data_page_v2():
    rng = np.random.default_rng(5)
    t = pa.table({"a": rng.integers(0, 1 << 40, 80_000),
                  "k": rng.integers(0, 37, 80_000)})
    check_fixed(t, "a", use_dictionary=False, data_page_version="2.0")
    check_fixed(t, "k", use_dictionary=True, data_page_version="2.0")


def test_non_nullable_schema():
    a = pa.array(np.arange(10_000, dtype=np.int64))
    t = pa.Table.from_arrays([a], schema=pa.schema(
        [pa.field("a", pa.int64(), nullable=False)]))
    check_fixed(t, "a", use_dictionary=False)


def test_string_dictionary_codes():
    rng = np.random.default_rng(6)
    vals = np.array(["BUILDING", "AUTOMOBILE", "MACHINERY", "HOUSEHOLD",
                     "FURNITURE"])
    s = vals[rng.integers(0, 5, 50_000)]
    t = pa.table({"seg": s})
    raw = write(t, use_dictionary=True)
    codes, cb = sim_column(raw, 0)
    got = np.asarray(cb)[codes]
    np.testing.assert_array_equal(got, s)


def test_nulls_raise():
    t = pa.table({"a": pa.array([1, None, 3], type=pa.int64())})
    raw = write(t, use_dictionary=False)
    with pytest.raises(P.QkParquetError):
        sim_column(raw, 0)


def test_compressed_raises():
    t = pa.table({"a": np.arange(1000, dtype=np.int64)})
    raw = write(t, use_dictionary=False, compression="snappy")
    with pytest.raises(P.QkParquetError):
        sim_column(raw, 0)


def test_page_walk_counts():
    t = pa.table({"a": np.arange(100_000, dtype=np.int64)})
    raw = write(t, use_dictionary=False)
    md = pq.ParquetFile(io.BytesIO(raw)).metadata
    c = md.row_group(0).column(0)
    pages = T.walk_pages(raw, c.data_page_offset, c.total_compressed_size,
                         c.num_values)
    assert sum(p.num_values for p in pages if p.kind != T.PAGE_DICT) \
        == c.num_values


def test_fuzz_random_files():
    """Seeded fuzz: random schemas, row counts, row-group/page sizes,
    dictionary toggles — plan + simulated kernels vs pyarrow decode."""
    for case in range(30):
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
        for ci, name in enumerate(t.schema.names):
            got = sim_column(raw, ci)
            want = t.column(name).to_numpy()
            np.testing.assert_array_equal(
                got, want, err_msg="case %d col %s kw %r" % (case, name, kw))


def test_thrift_primitives_handcrafted():
    """Compact-protocol primitives against handcrafted byte sequences
    (pyarrow only ever exercises a writer-specific subset)."""
    from quokka_amd.parquet_thrift import (_varint, _zigzag, _skip,
                                           _read_struct)
    assert _varint(b"\x00", 0) == (0, 1)
    assert _varint(b"\x7f", 0) == (127, 1)
    assert _varint(b"\x80\x01", 0) == (128, 2)
    assert _varint(b"\xff\xff\x03", 0) == (0xFFFF, 3)
    assert _zigzag(b"\x00", 0) == (0, 1)
    assert _zigzag(b"\x01", 0) == (-1, 1)
    assert _zigzag(b"\x02", 0) == (1, 1)
    assert _zigzag(b"\x03", 0) == (-2, 1)
    # skip: bool(no payload), byte, i32 varint, double, binary, list
    assert _skip(b"", 0, 1) == 0
    assert _skip(b"\x42", 0, 3) == 1
    assert _skip(b"\x80\x01", 0, 5) == 2
    assert _skip(b"\x00" * 8, 0, 7) == 8
    assert _skip(b"\x03abc", 0, 8) == 4
    assert _skip(b"\x25\x02\x04", 0, 9) == 3   # list: 2 elems of i32
    # struct with field ids via delta and long-form, nested struct skip
    # field 1 (i32 = 5), field 3 (struct{field 1 bool true}), stop
    buf = bytes([0x15, 0x0A,                   # delta1 type5 -> zigzag 10
                 0x2C, 0x11, 0x00,             # delta2 type12: {f1 booltrue}
                 0x00])
    out, pos = _read_struct(buf, 0, {1: "i", 3: {1: "i"}})
    assert out == {1: 5, 3: {1: True}} and pos == len(buf)
    # unknown field types are skipped without corrupting position
    buf2 = bytes([0x17, 0, 0, 0, 0, 0, 0, 0, 0,   # f1 double, skipped
                  0x15, 0x06,                      # f2 i32 = 3
                  0x00])
    out2, _ = _read_struct(buf2, 0, {2: "i"})
    assert out2 == {2: 3}


def test_c_page_walker_matches_python_reference():
    """qk_pq_walk_pages (host C Thrift parser) == _walk_pages_py on every
    page of files covering v1/v2 pages, dictionary + PLAIN fallback,
    multiple row groups."""
    import io
    import numpy as np
    import pyarrow as pa
    import pyarrow.parquet as pq
    from quokka_amd import parquet_thrift as T

    rng = np.random.default_rng(9)
    n = 120_000
    t = pa.table({
        "a": rng.random(n),
        "k": rng.integers(0, 50, n).astype(np.int64),
        "s": pa.array(np.array(["x", "y", "z"])[rng.integers(0, 3, n)]
                      ).dictionary_encode(),
    })
    for kw in (dict(use_dictionary=False),
               dict(use_dictionary=True, row_group_size=30_000),
               dict(use_dictionary=["k"], data_page_version="2.0"),
               dict(use_dictionary=True, dictionary_pagesize_limit=4096)):
        buf = io.BytesIO()
        pq.write_table(t, buf, compression="NONE", **kw)
        raw = buf.getvalue()
        md = pq.ParquetFile(io.BytesIO(raw)).metadata
        for ci in range(md.num_columns):
            for rg in range(md.num_row_groups):
                col = md.row_group(rg).column(ci)
                start = col.data_page_offset
                if col.dictionary_page_offset is not None:
                    start = min(start, col.dictionary_page_offset)
                want = T._walk_pages_py(raw, start,
                                        col.total_compressed_size,
                                        col.num_values)
                got = T.walk_pages(raw, start, col.total_compressed_size,
                                   col.num_values)
                assert len(got) == len(want)
                for g, w in zip(got, want):
                    for f in ("kind", "num_values", "encoding", "def_enc",
                              "data_off", "data_len", "v2_levels_len",
                              "num_nulls", "uncompressed_len"):
                        assert getattr(g, f) == getattr(w, f), (f, g, w)
