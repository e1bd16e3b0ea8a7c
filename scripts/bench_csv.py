"""Measure the GPU CSV parse against pyarrow's multithreaded C++ reader
(the reference decodes CSV on CPU via polars.read_csv,
unordered_readers.py:438 — pyarrow.csv is the comparable in-image
multicore baseline). Times e2e (upload + newline index + parse) and
kernels-only on a lineitem-shaped table.
Run on a GPU box: python scripts/bench_csv.py [rows]
"""
import io
import sys
import time

import numpy as np
import pyarrow as pa
import pyarrow.csv as pacsv

sys.path.insert(0, __import__("os").path.dirname(
    __import__("os").path.dirname(__import__("os").path.abspath(__file__))))
from quokka_amd import csv_gpu, shim                  # noqa: E402
from quokka_amd.shim import DevBuffer, DevColumn, c_u64, c_vp, Timer  # noqa


def main():
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 8_000_000
    rng = np.random.default_rng(7)
    keys = rng.integers(0, 1 << 40, n)
    price = np.round(rng.uniform(900, 105000, n), 2)
    days = rng.integers(8000, 11000, n)
    segs = np.array(["BUILDING", "AUTOMOBILE", "MACHINERY", "HOUSEHOLD",
                     "FURNITURE"])
    seg = segs[rng.integers(0, 5, n)]
    dates = days.astype("datetime64[D]").astype(str)
    t = pa.table({"k": keys, "p": price, "d": dates, "s": seg})
    buf = io.BytesIO()
    pacsv.write_csv(t, buf, write_options=pacsv.WriteOptions(
        include_header=False, quoting_style="none"))
    raw = buf.getvalue()
    print("rows=%d text=%.2f GB" % (n, len(raw) / 1e9), flush=True)

    co = pacsv.ConvertOptions(column_types={
        "k": pa.int64(), "p": pa.float64(), "d": pa.date32(),
        "s": pa.string()})
    ro = pacsv.ReadOptions(column_names=["k", "p", "d", "s"])
    best = min(_t(lambda: pacsv.read_csv(io.BytesIO(raw),
                                         read_options=ro,
                                         convert_options=co))
               for _ in range(3))
    print("pyarrow.csv read:  %.3f s  %.2f GB/s  %.1f M rows/s"
          % (best, len(raw) / best / 1e9, n / best / 1e6), flush=True)

    shim.init(0)
    schema = [("k", "i64"), ("p", "f64"), ("d", "date"),
              ("s", "dict", list(segs))]

    def gpu_once():
        cols = csv_gpu.read_csv(raw, schema, sep=",")
        for c in cols.values():
            c.free()
    gpu_once()
    best_g = min(_t(gpu_once) for _ in range(3))
    print("gpu read_csv e2e:  %.3f s  %.2f GB/s  %.1f M rows/s"
          % (best_g, len(raw) / best_g / 1e9, n / best_g / 1e6),
          flush=True)

    # quoted path: every string value holds a comma, so pyarrow's
    # needed-style writer quotes each one and the parity-prefix newline
    # pass + quote-aware field scan run (RFC-4180)
    segs_q = np.array([s + ",Q" for s in segs], dtype=object)
    tq = pa.table({"k": keys, "p": price, "d": dates,
                   "s": segs_q[rng.integers(0, 5, n)]})
    bufq = io.BytesIO()
    pacsv.write_csv(tq, bufq, write_options=pacsv.WriteOptions(
        include_header=False))
    rawq = bufq.getvalue()
    assert rawq.count(b'"') >= 2 * n
    best_pq = min(_t(lambda: pacsv.read_csv(
        io.BytesIO(rawq), read_options=ro,
        convert_options=pacsv.ConvertOptions(column_types={
            "k": pa.int64(), "p": pa.float64(), "d": pa.date32(),
            "s": pa.string()}))) for _ in range(3))
    print("pyarrow quoted:    %.3f s  %.2f GB/s  %.1f M rows/s"
          % (best_pq, len(rawq) / best_pq / 1e9, n / best_pq / 1e6),
          flush=True)
    schema_q = [("k", "i64"), ("p", "f64"), ("d", "date"),
                ("s", "dict", [str(v) for v in segs_q])]

    def gpu_quoted():
        cols = csv_gpu.read_csv(rawq, schema_q, sep=",")
        for c in cols.values():
            c.free()
    gpu_quoted()
    best_q = min(_t(gpu_quoted) for _ in range(3))
    print("gpu quoted e2e:    %.3f s  %.2f GB/s  %.1f M rows/s"
          % (best_q, len(rawq) / best_q / 1e9, n / best_q / 1e6),
          flush=True)

    # kernels only (bytes resident in HBM)
    arr = np.frombuffer(raw, dtype=np.uint8)
    dev = DevBuffer(len(arr) + 8)
    shim._bounce.h2d(dev.ptr, arr)
    timer = Timer()

    def kernels_once():
        pos = DevColumn(np.uint64, len(arr))
        cnt = DevBuffer(8)
        shim.call("qk_csv_newlines", None, c_u64(0), c_u64(len(arr)),
                  dev.ptr, pos.ptr, cnt.ptr)
        import ctypes
        outs = [DevColumn(np.int64, n), DevColumn(np.float64, n),
                DevColumn(np.int32, n), DevColumn(np.uint8, n)]
        coltypes = np.array([0, 1, 2, 3], dtype=np.int32)
        cands = np.zeros(4 * csv_gpu.MAX_DICT, dtype=np.uint64)
        clens = np.zeros(4 * csv_gpu.MAX_DICT, dtype=np.uint8)
        nc = np.array([0, 0, 0, 5], dtype=np.int32)
        for j, v in enumerate(segs):
            w, ln = csv_gpu._dict_key(str(v))
            cands[3 * csv_gpu.MAX_DICT + j] = w
            clens[3 * csv_gpu.MAX_DICT + j] = ln
        ptrs = np.array([o.ptr.value for o in outs], dtype=np.uint64)
        ups = []
        for a in (coltypes, ptrs, cands, clens, nc):
            b = DevBuffer(a.nbytes)
            shim.call("qk_h2d", b.ptr, a.ctypes.data_as(c_vp),
                      c_u64(a.nbytes))
            ups.append(b)
        err = DevBuffer(8)
        shim.call("qk_h2d", err.ptr,
                  np.array([np.iinfo(np.uint64).max], dtype=np.uint64)
                  .ctypes.data_as(c_vp), c_u64(8))
        shim.call("qk_csv_parse", None, c_u64(n), dev.ptr, c_u64(0),
                  pos.ptr, ctypes.c_uint8(ord(",")), ctypes.c_uint8(0),
                  4, ups[0].ptr, ups[1].ptr, ups[2].ptr, ups[3].ptr,
                  ups[4].ptr, err.ptr)
        shim.call("qk_stream_sync", None)
        for b in ups + [err, cnt, pos] + outs:
            b.free()

    kernels_once()
    ts = []
    for _ in range(5):
        timer.start(None)
        kernels_once()
        timer.stop(None)
        shim.call("qk_stream_sync", None)
        ts.append(timer.elapsed_ms() / 1e3)
    tk = min(ts)
    print("gpu kernels only:  %.4f s  %.2f GB/s  %.1f M rows/s"
          % (tk, len(raw) / tk / 1e9, n / tk / 1e6), flush=True)
    dev.free()


def _t(fn):
    t0 = time.time()
    fn()
    return time.time() - t0


if __name__ == "__main__":
    main()
