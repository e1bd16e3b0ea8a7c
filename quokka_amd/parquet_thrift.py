"""Minimal Thrift compact-protocol reader for Parquet page headers.

The reference scans Parquet through pyarrow's CPU reader
(pyquokka/dataset.py InputParquetDataset / InputS3FilesDataset read via
pyarrow.parquet). Our GPU decode path keeps pyarrow for FILE metadata
(footer) but needs the per-page headers, which pyarrow does not expose:
each column chunk is a byte range `[PageHeader][page bytes]...` and the
PageHeader is a Thrift compact-protocol struct (parquet-format
PageHeader, parquet.thrift). This module parses exactly that — a few
hundred bytes per page, host-side; all VALUE decoding happens on the
GPU (csrc qk_pq_* kernels).

Thrift compact protocol essentials (thrift compact_protocol.md):
- struct = field headers until STOP(0x00); header byte hi-nibble =
  field-id delta (0 => explicit zigzag-varint id follows), lo-nibble =
  type (1/2=bool true/false, 3=byte, 4=i16, 5=i32, 6=i64, 7=double,
  8=binary, 9=list, 12=struct)
- integers are zigzag varints; binary is varint length + bytes;
  list header = (size<<4)|elem_type with size==0xF => varint size
"""


def _varint(buf, pos):
    x = 0
    shift = 0
    while True:
        b = buf[pos]
        pos += 1
        x |= (b & 0x7F) << shift
        if not b & 0x80:
            return x, pos
        shift += 7


def _zigzag(buf, pos):
    x, pos = _varint(buf, pos)
    return (x >> 1) ^ -(x & 1), pos


def _skip(buf, pos, ftype):
    if ftype in (1, 2):
        return pos
    if ftype == 3:
        return pos + 1
    if ftype in (4, 5, 6):
        return _varint(buf, pos)[1]
    if ftype == 7:
        return pos + 8
    if ftype == 8:
        n, pos = _varint(buf, pos)
        return pos + n
    if ftype in (9, 10):
        h = buf[pos]
        pos += 1
        size = h >> 4
        if size == 0xF:
            size, pos = _varint(buf, pos)
        for _ in range(size):
            pos = _skip(buf, pos, h & 0xF)
        return pos
    if ftype == 12:
        return _skip_struct(buf, pos)
    raise ValueError("thrift: cannot skip type %d" % ftype)


def _skip_struct(buf, pos):
    fid = 0
    while True:
        b = buf[pos]
        pos += 1
        if b == 0:
            return pos
        delta = b >> 4
        if delta == 0:
            fid, pos = _zigzag(buf, pos)
        else:
            fid += delta
        pos = _skip(buf, pos, b & 0xF)


def _read_struct(buf, pos, want):
    """Parse a struct, returning {field_id: value} for field ids in
    `want` ({fid: 'i'|'struct:<subwant>'}); everything else skipped."""
    out = {}
    fid = 0
    while True:
        b = buf[pos]
        pos += 1
        if b == 0:
            return out, pos
        delta = b >> 4
        ftype = b & 0xF
        if delta == 0:
            fid, pos = _zigzag(buf, pos)
        else:
            fid += delta
        spec = want.get(fid)
        if spec is None:
            pos = _skip(buf, pos, ftype)
        elif spec == "i":
            if ftype in (1, 2):
                out[fid] = ftype == 1
            else:
                out[fid], pos = _zigzag(buf, pos)
        else:                                   # nested struct spec
            out[fid], pos = _read_struct(buf, pos, spec)


# PageHeader (parquet.thrift): 1 type, 2 uncompressed_page_size,
# 3 compressed_page_size, 5 data_page_header{1 num_values, 2 encoding,
# 3 definition_level_encoding, 4 repetition_level_encoding},
# 7 dictionary_page_header{1 num_values, 2 encoding},
# 8 data_page_header_v2{1 num_values, 2 num_nulls, 3 num_rows,
# 4 encoding, 5 def_levels_len, 6 rep_levels_len, 7 is_compressed}
_PAGE_WANT = {
    1: "i", 2: "i", 3: "i",
    5: {1: "i", 2: "i", 3: "i", 4: "i"},
    7: {1: "i", 2: "i"},
    8: {1: "i", 2: "i", 3: "i", 4: "i", 5: "i", 6: "i", 7: "i"},
}

PAGE_DATA = 0
PAGE_DICT = 2
PAGE_DATA_V2 = 3


class PageInfo:
    __slots__ = ("kind", "num_values", "encoding", "def_enc",
                 "data_off", "data_len", "v2_levels_len", "num_nulls",
                 "uncompressed_len", "is_compressed")

    def __repr__(self):
        return ("PageInfo(kind=%d n=%d enc=%d off=%d len=%d)"
                % (self.kind, self.num_values, self.encoding,
                   self.data_off, self.data_len))


def walk_pages(buf, start, total_len, num_values):
    """Walk a column chunk's byte range; yield PageInfo per page until
    `num_values` data values are covered. `buf` is the whole file (or
    chunk) as bytes/memoryview; offsets in the returned PageInfo are
    absolute into `buf`.

    The walk itself runs in host C (qk_pq_walk_pages in the .so) — the
    pure-Python walk below (_walk_pages_py) is the documented reference
    of the format subset and the parity pin for the C parser
    (tests/test_parquet_cpu.py)."""
    import ctypes
    import numpy as np
    from . import shim

    # zero-copy pointer for bytes / mmap / any buffer-protocol object
    barr = np.frombuffer(buf, dtype=np.uint8)
    bptr = barr.ctypes.data_as(shim.c_vp)
    cap = 4096
    while True:
        out = np.empty((cap, 10), dtype=np.int64)
        n_out = ctypes.c_int64(0)
        try:
            shim.call(
                "qk_pq_walk_pages", bptr,
                shim.c_u64(start), shim.c_u64(total_len),
                shim.c_i64(num_values),
                out.ctypes.data_as(shim.c_vp), shim.c_i64(cap),
                ctypes.byref(n_out))
        except shim.QkError as e:
            if "max_pages" in str(e) and cap < (1 << 24):
                cap *= 8
                continue
            raise
        break
    pages = []
    for row in out[: n_out.value]:
        p = PageInfo()
        (p.kind, p.num_values, p.encoding, p.def_enc, p.data_off,
         p.data_len, p.v2_levels_len, p.num_nulls, p.uncompressed_len,
         p.is_compressed) = (int(row[0]), int(row[1]), int(row[2]),
                             int(row[3]), int(row[4]), int(row[5]),
                             int(row[6]), int(row[7]), int(row[8]),
                             bool(row[9]))
        pages.append(p)
    return pages


def _walk_pages_py(buf, start, total_len, num_values):
    """Pure-Python reference walk (see walk_pages)."""
    pages = []
    pos = start
    end = start + total_len
    seen = 0
    while seen < num_values and pos < end:
        h, after = _read_struct(buf, pos, _PAGE_WANT)
        p = PageInfo()
        p.kind = h[1]
        p.data_off = after
        p.data_len = h[3]                      # compressed size == plain
        p.uncompressed_len = h.get(2, h[3])
        p.is_compressed = True
        p.v2_levels_len = 0
        p.num_nulls = 0
        if p.kind == PAGE_DATA:
            p.num_values = h[5][1]
            p.encoding = h[5][2]
            p.def_enc = h[5][3]
            seen += p.num_values
        elif p.kind == PAGE_DICT:
            p.num_values = h[7][1]
            p.encoding = h[7][2]
            p.def_enc = -1
        elif p.kind == PAGE_DATA_V2:
            p.num_values = h[8][1]
            p.num_nulls = h[8].get(2, 0)
            p.encoding = h[8][4]
            p.def_enc = 3
            p.v2_levels_len = h[8].get(5, 0) + h[8].get(6, 0)
            # optional bool, default true; writers clear it on pages they
            # chose to leave uncompressed
            p.is_compressed = bool(h[8].get(7, True))
            seen += p.num_values
        else:
            raise ValueError("unsupported page type %d" % p.kind)
        pages.append(p)
        pos = after + p.data_len
    return pages
