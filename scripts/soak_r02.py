"""Endurance soak of the round-2 paths: repeated join emits through the
pinned-buffer pool (finalizer recycling), device string dictionary
growth, GPU snappy decode, and the overlapped exchange step — watching
host RSS and device pool stability for leaks.

Usage (GPU box): python scripts/soak_r02.py [--iters 60]
"""
import argparse
import gc
import io
import os
import sys

import numpy as np
import psutil

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)


def rss_mb():
    return psutil.Process().memory_info().rss / 1e6


def main():
    import pyarrow as pa
    import pyarrow.parquet as pq
    from quokka_amd import shim, ops, exchange, parquet_gpu, bridge
    from quokka_amd.executors import GPUBuildProbeJoinExecutor

    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=60)
    args = ap.parse_args()
    shim.init(0)
    rng = np.random.default_rng(0)

    # snappy source file (once)
    n = 2_000_000
    t = pa.table({"a": rng.random(n),
                  "k": rng.integers(0, 97, n).astype(np.int64)})
    t = t.cast(pa.schema([pa.field(f.name, f.type, nullable=False)
                          for f in t.schema]))
    buf = io.BytesIO()
    pq.write_table(t, buf, compression="SNAPPY", use_dictionary=["k"])
    raw = buf.getvalue()

    # overlapped-exchange fixtures (world 1 self-exchange)
    comm = exchange.Comm(0, 1)
    comp, cstr = shim.Stream(), shim.Stream()
    kcol = shim.DevColumn.from_numpy(
        rng.integers(0, 1 << 40, 3_000_000).astype(np.int64))
    vcol = shim.DevColumn.from_numpy(rng.random(3_000_000))

    base = None
    for it in range(args.iters):
        # 1. join emit through the pinned pool (large result each iter)
        ex = GPUBuildProbeJoinExecutor(on="k", how="inner")
        bk = rng.permutation(200_000).astype(np.int64)
        ex.execute([pa.table({"k": bk, "pay": bk * 1.5})], 1, 0)
        pk = rng.integers(0, 200_000, 1_500_000).astype(np.int64)
        out = ex.execute([pa.table({"k": pk,
                                    "x": rng.random(1_500_000)})], 0, 0)
        assert out.num_rows == 1_500_000
        del out, ex
        # 2. string dictionary growth
        sd = ops.DeviceStringDict(expected=64)
        arr = np.array(["it%d_%d" % (it, i) for i in
                        rng.integers(0, 30_000, 40_000)], dtype=object)
        codes = sd.encode_column(pa.chunked_array([pa.array(arr)]))
        assert len(codes) == 40_000
        sd.free()
        # 3. snappy decode
        cols = parquet_gpu.read_table(raw)
        for c in cols.values():
            (c[0] if isinstance(c, tuple) else c).free()
        # 4. overlapped exchange (chunked self-exchange + consume)
        got = []

        def consume(views, start, nrows, j):
            got.append(nrows)

        rk, rp, _, _ = exchange.repartition_overlapped(
            comm, kcol, {"v": vcol}, comp, cstr, consume, nchunks=3)
        assert sum(got) == 3_000_000
        rk.free()
        rp["v"].free()
        gc.collect()
        r = rss_mb()
        if it == 4:
            base = r
        if it % 10 == 4:
            print("iter %3d rss %7.1f MB (pinned pool %.0f MB, dev pool "
                  "%.0f MB cached)"
                  % (it, r, bridge._pinned_pool.cached / 1e6,
                     shim._pool.cached / 1e6), flush=True)
    growth = r - base
    print("RSS growth iters 5..%d: %.1f MB (%s)"
          % (args.iters - 1, growth,
             "OK" if growth < 400 else "SUSPICIOUS"))
    assert growth < 800, "leak suspected: RSS grew %.0f MB" % growth
    comp.destroy()
    cstr.destroy()
    comm.destroy()
    print("soak ok")


if __name__ == "__main__":
    main()
