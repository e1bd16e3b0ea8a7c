"""GPU snappy decode rate: snappy parquet lineitem -> qk_snappy_pages +
decode kernels, vs pyarrow's multicore decode of the same file.

Usage (on a GPU box): python scripts/bench_snappy.py [--sf 3]
"""
import argparse
import io
import sys
import time
import os

import numpy as np

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)


def main():
    import pyarrow as pa
    import pyarrow.parquet as pq
    from oracle import tpch_gen as G
    from quokka_amd import parquet_gpu as P

    ap = argparse.ArgumentParser()
    ap.add_argument("--sf", type=float, default=3.0)
    ap.add_argument("--passes", type=int, default=3)
    ap.add_argument("--codec", default="SNAPPY", choices=["SNAPPY", "GZIP"])
    args = ap.parse_args()

    li = G.gen_lineitem(args.sf, seed=5)
    n = len(li["l_orderkey"])
    t = pa.table({
        "l_orderkey": li["l_orderkey"],
        "l_quantity": li["l_quantity"],
        "l_extendedprice": li["l_extendedprice"],
        "l_discount": li["l_discount"],
        "l_shipdate": pa.array(li["l_shipdate"], type=pa.int32()),
        "l_returnflag": pa.array(
            np.array(G.RETURNFLAG)[li["l_returnflag"]]).dictionary_encode(),
    })
    t = t.cast(pa.schema([pa.field(f.name, f.type, nullable=False)
                          for f in t.schema]))
    buf = io.BytesIO()
    pq.write_table(t, buf, compression=args.codec,
                   use_dictionary=["l_returnflag"])
    raw = buf.getvalue()
    unc = sum(t.column(c).nbytes for c in t.column_names)
    print("rows=%d file=%.2f GB (%s), uncompressed columns=%.2f GB"
          % (n, len(raw) / 1e9, args.codec, unc / 1e9))

    cols = P.read_table(raw)          # warm (pool, hiprtc none needed)
    for c in cols.values():
        (c[0] if isinstance(c, tuple) else c).free()
    t0 = time.time()
    for _ in range(args.passes):
        cols = P.read_table(raw)
        for c in cols.values():
            (c[0] if isinstance(c, tuple) else c).free()
    dt = (time.time() - t0) / args.passes
    print("GPU: %.3f s/pass = %.1f GB/s file, %.1f GB/s decoded, "
          "%.1f M rows/s (incl. upload + host planning)"
          % (dt, len(raw) / dt / 1e9, unc / dt / 1e9, n / dt / 1e6))

    t0 = time.time()
    for _ in range(args.passes):
        pq.read_table(io.BytesIO(raw))
    dt = (time.time() - t0) / args.passes
    print("pyarrow multicore: %.3f s/pass = %.1f GB/s file" %
          (dt, len(raw) / dt / 1e9))


if __name__ == "__main__":
    main()
