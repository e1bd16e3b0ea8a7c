"""Measure the executor-boundary staging rate (pageable numpy <-> HBM
through the double-buffered pinned bounce, shim._PinnedBounce) — the
PCIe-inclusive rate that applies when host Arrow batches cross the
drop-in boundary. Run: python scripts/bench_staging.py [GB]
"""
import sys
import time

import numpy as np

sys.path.insert(0, __import__("os").path.dirname(
    __import__("os").path.dirname(__import__("os").path.abspath(__file__))))
from quokka_amd import shim                             # noqa: E402
from quokka_amd.shim import DevColumn                   # noqa: E402


def main():
    gb = float(sys.argv[1]) if len(sys.argv) > 1 else 8.0
    shim.init(0)
    n = int(gb * 1e9 / 8)
    host = np.random.default_rng(1).random(n)
    col = DevColumn.from_numpy(host[:1024])   # warm pool/bounce/streams
    col.free()
    best_up = min(_t(lambda: DevColumn.from_numpy(host).free())
                  for _ in range(3))
    col = DevColumn.from_numpy(host)
    best_down = min(_t(lambda: col.to_numpy(n)) for _ in range(3))
    back = col.to_numpy(n)
    assert np.array_equal(back, host)         # staging is lossless
    col.free()
    nbytes = host.nbytes
    print("staging h2d (pageable->HBM): %.3f s  %.2f GB/s"
          % (best_up, nbytes / best_up / 1e9))
    print("staging d2h (HBM->pageable): %.3f s  %.2f GB/s"
          % (best_down, nbytes / best_down / 1e9))


def _t(fn):
    t0 = time.time()
    fn()
    return time.time() - t0


if __name__ == "__main__":
    main()
