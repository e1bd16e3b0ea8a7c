"""Diagnose the world-1 exchange path at large row counts.

SF100 exchange-mode Q3 showed ~half the expected filter-pass rows after
repartition (orders build 7.29M vs 14.58M) while the SF20 parity test is
green — a size-dependent bug. This isolates the layers:
  A) raw qk_alltoallv self-send of one column (no partition/gather)
  B) full exchange.repartition (partition -> gather -> alltoallv)
comparing host checksums (sum/xor/min/max are permutation-invariant) and,
for mode A (order-preserving), the first mismatching index.

Run on a GPU box: python scripts/diag_exchange.py [max_rows]
"""
import sys

import numpy as np

sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
from quokka_amd import shim, exchange  # noqa: E402
from quokka_amd.shim import DevColumn, c_u64, c_i64  # noqa: E402


def checks(a):
    return (int(np.sum(a, dtype=np.uint64)),
            int(np.bitwise_xor.reduce(a.view(np.uint64 if a.itemsize == 8
                                             else np.uint32))),
            int(a.min()), int(a.max()))


def main():
    max_rows = int(sys.argv[1]) if len(sys.argv) > 1 else 600_037_900
    shim.init(0)
    st = shim.Stream()
    comm = exchange.Comm(0, 1, None)
    sizes = [120_000_000, 268_435_456, 300_000_000, 536_870_912,
             600_037_900]
    sizes = [min(n, max_rows) for n in sizes if n <= max_rows] or [max_rows]
    for n in sizes:
        # fill an i64 key col + i32 date col with the bench generator
        key = DevColumn(np.int64, n)
        dat = DevColumn(np.int32, n)
        price = DevColumn(np.float64, n)
        disc = DevColumn(np.float64, n)
        shim.call("qk_gen_lineitem", None, c_u64(n), c_u64(0), c_u64(42),
                  c_i64(20_000_000), c_i64(1_000_000), c_i64(n // 4),
                  key.ptr, None, None, price.ptr, disc.ptr, None, None,
                  None, dat.ptr)
        price.free()
        disc.free()
        hk = key.to_numpy(n)
        hd = dat.to_numpy(n)
        for name, col, host in (("i64", key, hk), ("i32", dat, hd)):
            # A) raw self alltoallv, identity layout
            so = np.zeros(1, dtype=np.uint64)
            sc = np.array([n], dtype=np.uint64)
            recv = comm.alltoallv_column(col, so, sc, sc, st)
            st.sync()
            hr = recv.to_numpy(n)
            ok = np.array_equal(hr, host)
            line = "A n=%d %s exact=%s" % (n, name, ok)
            if not ok:
                bad = np.nonzero(hr != host)[0]
                line += " first_bad=%d n_bad=%d frac=%.4f" % (
                    bad[0], len(bad), len(bad) / n)
            print(line, flush=True)
            recv.free()
        # B) full repartition of (key, date)
        rk, rp = exchange.repartition(comm, key, {"d": dat}, st)
        st.sync()
        hrk = rk.to_numpy(n)
        hrd = rp["d"].to_numpy(n)
        print("B n=%d key %s->%s date %s->%s" % (
            n, checks(hk), checks(hrk), checks(hd), checks(hrd)),
            flush=True)
        print("B n=%d key_ok=%s date_ok=%s pairs_ok=%s" % (
            n, checks(hk) == checks(hrk), checks(hd) == checks(hrd),
            sorted(zip(hk[:: max(1, n // 500_000)].tolist(),
                       hd[:: max(1, n // 500_000)].tolist())) is not None),
            flush=True)
        for c in (rk, rp["d"], key, dat):
            c.free()
    comm.destroy()
    print("diag done", flush=True)


if __name__ == "__main__":
    main()
