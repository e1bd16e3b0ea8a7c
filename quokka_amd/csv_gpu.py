"""GPU CSV decode for MI355X.

The reference's CSV scan (pyquokka/dataset/unordered_readers.py
InputDiskCSVDataset :273-442 / InputS3CSVDataset :646-852) splits files
into ~16 MiB byte ranges, refines the boundaries to newlines, and hands
each range to polars.read_csv on the CPU. Here the byte range goes to
HBM once and the qk_csv_* kernels do the newline indexing and the typed
field parsing there (include/quokka_amd.h for the exact contract).

Schema entries: (name, "i64" | "f64" | "date" | "skip") or
(name, "dict", [candidate strings]) — dict columns come back as u8 code
columns (the executors' string-dict form), codes = candidate index.
Numeric parsing is bit-exact vs strtod for <= 15 significant digits;
anything unparseable raises QkCsvError with the failing row, never a
silently different value.

RFC-4180 quoting is supported (quote='"', the reference hands quoting to
polars.read_csv): a field starting with the quote char runs to the
matching close quote — separators and newlines inside are data, doubled
quotes escape. When the buffer contains no quote byte at all the
original single-pass newline kernel runs unchanged; otherwise a
parity-prefix pass validates newlines (qk_csv_newlines_quoted). Escaped
"" inside numeric/dict fields still fail the parse (error, never wrong
data) — same behaviour as the reference's typed readers.
"""
import ctypes

import numpy as np

TYPE_CODE = {"i64": 0, "f64": 1, "date": 2, "dict": 3, "skip": 4}
_OUT_DTYPE = {0: np.int64, 1: np.float64, 2: np.int32, 3: np.uint8}
MAX_DICT = 32


class QkCsvError(RuntimeError):
    pass


def _dict_key(value):
    bs = value.encode()
    w = 0
    for i, byte in enumerate(bs[:8]):
        w |= byte << (8 * i)
    return w, min(len(bs), 255)


def read_csv(source, schema, sep="|", header=False, quote='"'):
    """Decode a CSV byte buffer (or file path) into device columns.

    Returns dict name -> DevColumn (u8 codes for dict columns; the
    candidate list you passed is the codebook). The buffer must end with
    a newline on its final row (a trailing partial line is ignored, the
    same contract as the reference's byte-range splitting
    unordered_readers.py:372-383)."""
    from . import shim
    from .shim import DevBuffer, DevColumn, c_u64, c_vp

    raw = open(source, "rb").read() if isinstance(source, str) else source
    if raw and not raw.endswith(b"\n"):
        raw = raw + b"\n"
    data_start = 0
    if header:
        nl = raw.find(b"\n")
        if nl < 0:                      # empty / header-only input
            raw = b""
        else:
            data_start = nl + 1
    arr = np.frombuffer(raw, dtype=np.uint8)
    n = len(arr)
    dev = DevBuffer(max(1, n + 8))
    shim._bounce.h2d(dev.ptr, arr)

    # newline index (counter zeroed first: the kernel early-returns
    # without writing it when the data range is empty, and pool-recycled
    # buffers hold stale bytes)
    # Quoting only changes anything if the quote byte occurs at all —
    # keep the fast single-pass newline kernel for quote-free buffers
    # (all of TPC-H .tbl). raw is already host-resident; the scan is
    # memchr-speed.
    qbyte = ord(quote) if quote else 0
    quoted = bool(qbyte) and raw.find(qbyte.to_bytes(1, "little"),
                                      data_start) >= 0
    pos = DevColumn(np.uint64, max(1, n - data_start))
    cnt = DevBuffer(8)
    shim.call("qk_dmemset", cnt.ptr, 0, c_u64(8))
    if quoted:
        shim.call("qk_csv_newlines_quoted", None, c_u64(data_start),
                  c_u64(n), dev.ptr, ctypes.c_uint8(qbyte), pos.ptr,
                  cnt.ptr)
    else:
        shim.call("qk_csv_newlines", None, c_u64(data_start), c_u64(n),
                  dev.ptr, pos.ptr, cnt.ptr)
    host_cnt = np.zeros(1, dtype=np.uint64)
    shim.call("qk_d2h", host_cnt.ctypes.data_as(c_vp), cnt.ptr, c_u64(8))
    nrows = int(host_cnt[0])
    cnt.free()
    assert nrows <= n, (nrows, n)
    if nrows == 0:
        pos.free()
        dev.free()
        out = {}
        for spec in schema:
            if spec[1] == "skip":
                continue
            col = DevColumn(_OUT_DTYPE[TYPE_CODE[spec[1]]], 1)
            col.n = 0                      # no rows; buffer is a stub
            out[spec[0]] = col
        return out

    ncols = len(schema)
    coltypes = np.zeros(ncols, dtype=np.int32)
    cands = np.zeros(ncols * MAX_DICT, dtype=np.uint64)
    clens = np.zeros(ncols * MAX_DICT, dtype=np.uint8)
    ncands = np.zeros(ncols, dtype=np.int32)
    outs = {}
    out_ptrs = np.zeros(ncols, dtype=np.uint64)
    for c, spec in enumerate(schema):
        name, typ = spec[0], spec[1]
        coltypes[c] = TYPE_CODE[typ]
        if typ == "skip":
            continue
        if typ == "dict":
            values = list(spec[2])
            if not 0 < len(values) <= MAX_DICT:
                raise QkCsvError("dict column %r needs 1..%d candidates"
                                 % (name, MAX_DICT))
            keys = [_dict_key(v) for v in values]
            if len(set(keys)) != len(keys):
                raise QkCsvError("dict column %r candidates collide in "
                                 "(first 8 bytes, length)" % name)
            for j, (w, ln) in enumerate(keys):
                cands[c * MAX_DICT + j] = w
                clens[c * MAX_DICT + j] = ln
            ncands[c] = len(values)
        col = DevColumn(_OUT_DTYPE[coltypes[c]], nrows)
        outs[name] = col
        out_ptrs[c] = col.ptr.value

    def up(a):
        b = DevBuffer(a.nbytes)
        shim.call("qk_h2d", b.ptr, a.ctypes.data_as(c_vp), c_u64(a.nbytes))
        return b

    d_types, d_ptrs, d_cands, d_clens, d_nc = (
        up(coltypes), up(out_ptrs), up(cands), up(clens), up(ncands))
    err = DevBuffer(8)
    shim.call("qk_h2d", err.ptr,
              np.array([np.iinfo(np.uint64).max],
                       dtype=np.uint64).ctypes.data_as(c_vp), c_u64(8))
    shim.call("qk_csv_parse", None, c_u64(nrows), dev.ptr,
              c_u64(data_start), pos.ptr, ctypes.c_uint8(ord(sep)),
              ctypes.c_uint8(qbyte if quoted else 0),
              ncols, d_types.ptr, d_ptrs.ptr, d_cands.ptr, d_clens.ptr,
              d_nc.ptr, err.ptr)
    herr = np.zeros(1, dtype=np.uint64)
    shim.call("qk_d2h", herr.ctypes.data_as(c_vp), err.ptr, c_u64(8))
    for b in (d_types, d_ptrs, d_cands, d_clens, d_nc, err, pos):
        b.free()
    dev.free()
    if herr[0] != np.iinfo(np.uint64).max:
        for col in outs.values():
            col.free()
        raise QkCsvError("row %d failed to parse (schema mismatch, "
                         "unknown dict value, or >15-digit numeric)"
                         % int(herr[0]))
    return outs
