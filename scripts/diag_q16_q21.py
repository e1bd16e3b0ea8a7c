"""cProfile q16 and q21 at SF30 on oracle-generated host data (same
staging as run_all_queries) to find host vs device time."""
import cProfile
import os
import pstats
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


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
    pcols = S(d["part"])
    pscols = S(d["partsupp"])
    nat = d["nation"]
    part_host = {k: d["part"][k] for k in ("p_partkey", "p_brand",
                                           "p_type", "p_size")}

    for name, fn in [
        ("q16", lambda: DQ.q16(pcols, pscols, scols, part_host)),
        ("q21", lambda: DQ.q21(lcols, ocols, scols, nat["n_name"])),
    ]:
        fn()                                   # warm
        pr = cProfile.Profile()
        pr.enable()
        fn()
        pr.disable()
        print("==== %s ====" % name, flush=True)
        pstats.Stats(pr).sort_stats("cumtime").print_stats(14)


if __name__ == "__main__":
    main()
