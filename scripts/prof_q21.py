"""Run q21 a few times at --sf for rocprofv3 kernel attribution."""
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np  # noqa: E402


def main():
    from oracle import tpch_gen as G
    from quokka_amd import shim, staging
    from quokka_amd import queries as DQ
    shim.init(0)
    sf = float(sys.argv[1]) if len(sys.argv) > 1 else 30.0
    d = G.gen_all(sf, 42)
    S = staging.stage_columns
    lcols = S(d["lineitem"])
    ocols = S(d["orders"])
    scols = S(d["supplier"])
    nat = d["nation"]
    import time
    DQ.q21(lcols, ocols, scols, nat["n_name"])     # warm
    t0 = time.time()
    for _ in range(3):
        DQ.q21(lcols, ocols, scols, nat["n_name"])
    print("q21 %.1f ms/query" % ((time.time() - t0) / 3 * 1e3))


if __name__ == "__main__":
    main()
