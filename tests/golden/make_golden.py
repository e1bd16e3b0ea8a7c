"""Generate the committed golden fixtures (tests/golden/golden.json).

Run from the repo root: python tests/golden/make_golden.py

The fixtures are the CPU oracle's results on the seeded SF0.01 dataset
(seed 42). They pin the oracle against accidental regression; the oracle
itself is pinned to the reference's own golden SQL (tpch_ref.py) — see
oracle/__init__.py for the parity-anchoring statement."""
import json
import os
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, ROOT)

import numpy as np  # noqa: E402
from oracle import tpch_gen as G, queries as Q  # noqa: E402

SF = 0.01
SEED = 42


def jsonable(d):
    out = {}
    for k, v in d.items():
        v = np.asarray(v)
        if v.dtype.kind == "f":
            out[k] = [repr(float(x)) for x in v]  # full-precision decimal
        elif v.dtype.kind in "iu":
            out[k] = [int(x) for x in v]
        else:
            out[k] = [str(x) for x in v]
    return out


def main():
    d = G.gen_all(SF, SEED)
    li = d["lineitem"]
    q1 = Q.q1(li)
    q6 = Q.q6(li)
    full, top10 = Q.q3(li, d["orders"], d["customer"])
    q5 = Q.q5(li, d["orders"], d["customer"], d["supplier"], d["nation"],
              d["region"])
    golden = {
        "sf": SF,
        "seed": SEED,
        "lineitem_rows": int(len(li["l_orderkey"])),
        "q1": jsonable(q1),
        "q6": {"revenue": repr(float(q6["revenue"])),
               "rows_passed": q6["rows_passed"]},
        "q3_top10": jsonable(top10),
        "q3_n_groups": int(len(full["l_orderkey"])),
        "q5": [[n, repr(float(r))] for n, r in q5],
    }
    path = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                        "golden.json")
    with open(path, "w") as f:
        json.dump(golden, f, indent=1)
    print("wrote", path)


if __name__ == "__main__":
    main()
