"""Seeded randomized GPU fuzz: many small random instances of every
operator against the CPU oracle. One test per operator family keeps the
suite time bounded (~1 min on an MI355X); seeds are fixed so failures
reproduce."""
import numpy as np
import pytest

from oracle import executors as OE

pytestmark = pytest.mark.gpu

N_CASES = 25


@pytest.fixture(scope="module")
def gpu():
    from quokka_amd import shim
    shim.init(0)
    return shim


def test_fuzz_join(gpu):
    from quokka_amd import ops, shim
    for case in range(N_CASES):
        rng = np.random.default_rng(1000 + case)
        nb = int(rng.integers(1, 5000))
        npr = int(rng.integers(1, 8000))
        key_space = int(rng.integers(1, 4 * nb + 2))
        bk = rng.integers(0, key_space, nb).astype(np.int64)
        pk = rng.integers(0, 2 * key_space, npr).astype(np.int64)
        mode = int(rng.integers(0, 3))
        table = ops.JoinTable(nb)
        bcol = shim.DevColumn.from_numpy(bk)
        table.build(bcol)
        pcol = shim.DevColumn.from_numpy(pk)
        pidx, bidx, nm = table.probe(pcol, mode=mode,
                                     out_factor=float(rng.uniform(0.1, 2)))
        if mode == 0:
            wp, wb = OE.build_probe_join(bk, pk, "inner")
            got = set(zip(pidx.to_numpy(nm).tolist(),
                          bidx.to_numpy(nm).tolist()))
            assert got == set(zip(wp.tolist(), wb.tolist())), case
        else:
            how = "semi" if mode == 1 else "anti"
            want = set(OE.build_probe_join(bk, pk, how).tolist())
            assert set(pidx.to_numpy(nm).tolist()) == want, case
        for c in (bcol, pcol, pidx):
            c.free()
        if bidx:
            bidx.free()
        table.free()


def test_fuzz_groupby(gpu):
    from quokka_amd import ops, shim
    for case in range(N_CASES):
        rng = np.random.default_rng(2000 + case)
        n = int(rng.integers(1, 60_000))
        ngroups = int(rng.integers(1, max(2, n)))
        keys = rng.integers(-ngroups, ngroups, n).astype(np.int64)
        nvals = int(rng.integers(1, 4))
        agg_ops = [int(o) for o in rng.integers(0, 3, nvals)]
        vals = [np.round(rng.standard_normal(n), 4) for _ in range(nvals)]
        gb = ops.GroupByI64(len(np.unique(keys)), nvals, agg_ops=agg_ops)
        kcol = shim.DevColumn.from_numpy(keys)
        vcols = [shim.DevColumn.from_numpy(v) for v in vals]
        gb.update(kcol, vcols)
        gk, gs = gb.extract()
        order = np.argsort(gk)
        uk = np.unique(keys)
        assert np.array_equal(gk[order], uk), case
        for c, (op, v) in enumerate(zip(agg_ops, vals)):
            fn = {0: np.sum, 1: np.min, 2: np.max}[op]
            want = np.array([fn(v[keys == k]) for k in uk])
            np.testing.assert_allclose(gs[c][order], want, rtol=1e-9,
                                       err_msg="case %d col %d op %d"
                                       % (case, c, op))
        kcol.free()
        for c in vcols:
            c.free()
        gb.free()


def test_fuzz_partition_and_sort(gpu):
    from quokka_amd import ops, shim
    for case in range(N_CASES):
        rng = np.random.default_rng(3000 + case)
        n = int(rng.integers(1, 100_000))
        nparts = int(rng.integers(1, 65))
        keys = rng.integers(0, 1 << int(rng.integers(4, 62)), n)
        keys = keys.astype(np.int64)
        kcol = shim.DevColumn.from_numpy(keys)
        offsets, idx = ops.partition_i64(kcol, nparts)
        sel = idx.to_numpy(n)
        want = OE.partition_int(keys, nparts)
        assert np.array_equal(np.diff(offsets.astype(np.int64)),
                              np.bincount(want, minlength=nparts)), case
        assert len(np.unique(sel)) == n, case
        for p in range(nparts):
            rows = sel[int(offsets[p]):int(offsets[p + 1])]
            assert np.all(want[rows] == p), case
        # sort the same keys (ascending + descending, stability)
        perm = ops.sort_permutation(kcol)
        assert np.array_equal(perm.to_numpy(perm.n),
                              np.argsort(keys, kind="stable")), case
        perm.free()
        kcol.free()
        idx.free()


def test_fuzz_jit_filter(gpu):
    from quokka_amd import jit, shim
    schema = {"a": np.dtype(np.int32), "b": np.dtype(np.float64),
              "k": np.dtype(np.int64)}
    ops_sql = ["<", "<=", ">", ">=", "=", "<>"]
    for case in range(12):   # compiles are ~1s each; keep bounded
        rng = np.random.default_rng(4000 + case)
        n = int(rng.integers(1, 50_000))
        cols_np = {"a": rng.integers(-100, 100, n).astype(np.int32),
                   "b": np.round(rng.uniform(-10, 10, n), 3),
                   "k": rng.integers(-1000, 1000, n).astype(np.int64)}
        t1 = "a %s %d" % (ops_sql[rng.integers(0, 6)],
                          int(rng.integers(-120, 120)))
        t2 = "b between %r and %r" % tuple(
            float(x) for x in sorted(np.round(rng.uniform(-12, 12, 2), 3)))
        t3 = "k %s %d" % (ops_sql[rng.integers(0, 6)],
                          int(rng.integers(-1200, 1200)))
        conn = [" and ", " or "][rng.integers(0, 2)]
        pred = ("not (%s)%s(%s) and %s" % (t1, conn, t2, t3)
                if rng.integers(0, 2) else "(%s)%s(%s)" % (t1, conn, t3))
        f = jit.JitFilter(pred, schema)
        dcols = {c: shim.DevColumn.from_numpy(v)
                 for c, v in cols_np.items()}
        idx, k = f.run(dcols)
        env = {"v%d" % i: cols_np[name] for i, name in enumerate(f.cols)}
        py = (f.expr.replace("&&", "&").replace("||", "|")
              .replace("!(", "~("))
        want = np.nonzero(eval(py, {}, env))[0]
        assert np.array_equal(idx.to_numpy(k), want), (case, pred)
        f.free()
        idx.free()
        for c in dcols.values():
            c.free()


def test_jit_filter_in_lists(gpu):
    """Compiled IN / NOT IN predicates on device vs numpy."""
    from quokka_amd import jit, shim
    rng = np.random.default_rng(4100)
    n = 40_000
    a = rng.integers(-50, 50, n).astype(np.int32)
    k = rng.integers(-500, 500, n).astype(np.int64)
    schema = {"a": np.dtype(np.int32), "k": np.dtype(np.int64)}
    for pred, want in [
        ("a in (3, -7, 11, 42)", np.isin(a, [3, -7, 11, 42])),
        ("k not in (0, 250, -250)", ~np.isin(k, [0, 250, -250])),
        ("a in (1,2,3) and k > 0", np.isin(a, [1, 2, 3]) & (k > 0)),
    ]:
        f = jit.JitFilter(pred, schema)
        dcols = {"a": shim.DevColumn.from_numpy(a),
                 "k": shim.DevColumn.from_numpy(k)}
        idx, m = f.run(dcols)
        assert np.array_equal(idx.to_numpy(m), np.nonzero(want)[0]), pred
        f.free()
        idx.free()
        for c in dcols.values():
            c.free()
