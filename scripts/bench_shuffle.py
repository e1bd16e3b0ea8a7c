"""Measure the shuffle map-side kernels (the
reference's partition_fn + Flight push data plane) at SF100 scale:
qk_partition_hist / qk_partition_scatter over 600M i64 keys and the
per-column payload gather, reported as achieved GB/s against their
algorithmic bytes. Run: python scripts/bench_shuffle.py [rows] [nparts]
"""
import sys

import numpy as np

sys.path.insert(0, __import__("os").path.dirname(
    __import__("os").path.dirname(__import__("os").path.abspath(__file__))))
from quokka_amd import ops, shim                       # noqa: E402
from quokka_amd.shim import DevColumn, c_u64, c_i64, Timer  # noqa: E402


def main():
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 600_037_900
    nparts = int(sys.argv[2]) if len(sys.argv) > 2 else 8
    shim.init(0)
    keys = DevColumn(np.int64, n)
    pay = DevColumn(np.float64, n)
    d32 = DevColumn(np.int32, n)
    disc = DevColumn(np.float64, n)
    shim.call("qk_gen_lineitem", None, c_u64(n), c_u64(0), c_u64(42),
              c_i64(20_000_000), c_i64(1_000_000), c_i64(n // 4),
              keys.ptr, None, None, pay.ptr, disc.ptr, None, None, None,
              d32.ptr)
    timer = Timer()

    def timed(fn, reps=5):
        fn()
        best = None
        for _ in range(reps):
            timer.start(None)
            fn()
            timer.stop(None)
            shim.call("qk_stream_sync", None)
            ms = timer.elapsed_ms()
            best = ms if best is None else min(best, ms)
        return best

    # partition (hist+scan+scatter): reads keys twice + writes u32 idx
    state = {}

    def part():
        if "idx" in state:
            state["idx"].free()
        offsets, idx = ops.partition_i64(keys, nparts)
        state["idx"] = idx
        state["offsets"] = offsets
    ms = timed(part)
    alg = 8 * n * 2 + 4 * n
    print("partition_i64 nparts=%d: %.2f ms  %.0f GB/s algorithmic "
          "(2x key read + idx write)" % (nparts, ms, alg / ms / 1e6),
          flush=True)

    idx = state["idx"]

    def gath():
        out = pay.gather(idx, n)
        out.free()
    ms = timed(gath)
    alg = 4 * n + 8 * n + 8 * n     # idx read + random gather + write
    print("gather f64 by u32 perm: %.2f ms  %.0f GB/s algorithmic"
          % (ms, alg / ms / 1e6), flush=True)

    def gath32():
        out = d32.gather(idx, n)
        out.free()
    ms = timed(gath32)
    alg = 4 * n + 4 * n + 4 * n
    print("gather i32 by u32 perm: %.2f ms  %.0f GB/s algorithmic"
          % (ms, alg / ms / 1e6), flush=True)


if __name__ == "__main__":
    main()
