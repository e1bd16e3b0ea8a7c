"""Randomized CPU tests of the JIT predicate translator: generated
predicates are rendered to SQL text, translated to C, and the C
expression (evaluated via numpy with C semantics) must match a direct
numpy evaluation of the predicate AST on random column data."""
import datetime

import numpy as np
from hypothesis import given, settings, strategies as st

from quokka_amd import jit

SCHEMA = {
    "a": np.dtype(np.int32),
    "b": np.dtype(np.float64),
    "c": np.dtype(np.float64),
    "k": np.dtype(np.int64),
}

_num = st.one_of(
    st.integers(min_value=-1000, max_value=1000),
    st.floats(min_value=-100, max_value=100, allow_nan=False,
              allow_infinity=False).map(lambda x: round(x, 3)),
)
_col = st.sampled_from(list(SCHEMA))
_cmp = st.sampled_from(["<", "<=", ">", ">=", "=", "<>"])


@st.composite
def comparison(draw):
    col = draw(_col)
    kind = draw(st.integers(0, 3))
    if kind == 0:
        return ("cmp", col, draw(_cmp), draw(_num))
    if kind == 1:
        lo, hi = sorted([draw(_num), draw(_num)])
        return ("between", col, lo, hi)
    els = draw(st.lists(_num, min_size=1, max_size=5))
    return ("in" if kind == 2 else "notin", col, els)


@st.composite
def predicate(draw, depth=0):
    if depth >= 2 or draw(st.integers(0, 2)) == 0:
        return draw(comparison())
    kind = draw(st.sampled_from(["and", "or", "not"]))
    if kind == "not":
        return ("not", draw(predicate(depth=depth + 1)))
    return (kind, draw(predicate(depth=depth + 1)),
            draw(predicate(depth=depth + 1)))


def to_sql(p):
    k = p[0]
    if k == "cmp":
        _, col, op, lit = p
        return "%s %s %r" % (col, op, lit)
    if k == "between":
        _, col, lo, hi = p
        return "%s between %r and %r" % (col, lo, hi)
    if k in ("in", "notin"):
        _, col, els = p
        return "%s %sin (%s)" % (col, "not " if k == "notin" else "",
                                 ", ".join(repr(e) for e in els))
    if k == "not":
        return "not (%s)" % to_sql(p[1])
    return "(%s) %s (%s)" % (to_sql(p[1]), p[0], to_sql(p[2]))


def np_eval(p, cols):
    k = p[0]
    if k == "cmp":
        _, col, op, lit = p
        v = cols[col]
        return {"<": v < lit, "<=": v <= lit, ">": v > lit,
                ">=": v >= lit, "=": v == lit, "<>": v != lit}[op]
    if k == "between":
        _, col, lo, hi = p
        return (cols[col] >= lo) & (cols[col] <= hi)
    if k in ("in", "notin"):
        _, col, els = p
        m = np.zeros(len(cols[col]), dtype=bool)
        for e in els:
            m |= cols[col] == e
        return ~m if k == "notin" else m
    if k == "not":
        return ~np_eval(p[1], cols)
    a, b = np_eval(p[1], cols), np_eval(p[2], cols)
    return a & b if k == "and" else a | b


def c_eval(expr, order, cols):
    """Evaluate the translated C expression with numpy (same semantics
    for the generated comparison/boolean subset)."""
    env = {"v%d" % i: cols[name] for i, name in enumerate(order)}
    py = expr.replace("&&", "&").replace("||", "|").replace("!(", "~(")
    # wrap comparisons: C precedence was fully parenthesized by the
    # translator, so & / | / ~ bind correctly over the parenthesized terms
    return eval(py, {"np": np}, env) != 0


@settings(max_examples=200, deadline=None)
@given(predicate())
def test_translate_matches_numpy(p):
    sql = to_sql(p)
    expr, order = jit.translate(sql, SCHEMA)
    rng = np.random.default_rng(abs(hash(sql)) % (2 ** 32))
    cols = {
        "a": rng.integers(-1200, 1200, 500).astype(np.int32),
        "b": np.round(rng.uniform(-120, 120, 500), 3),
        "c": np.round(rng.uniform(-120, 120, 500), 3),
        "k": rng.integers(-1200, 1200, 500).astype(np.int64),
    }
    want = np_eval(p, cols)
    got = c_eval(expr, order, cols)
    assert np.array_equal(got, want), sql


def test_translate_date_intervals_exhaustive():
    """Date +/- interval folding vs Python date math across units."""
    schema = {"d": np.dtype(np.int32)}
    epoch = datetime.date(1970, 1, 1)
    for y, m, dd in [(1994, 1, 1), (1998, 12, 1), (1996, 2, 29),
                     (1995, 3, 15)]:
        for n, unit in [(90, "day"), (3, "month"), (1, "year"),
                        (14, "month")]:
            for sign, sgn in [("+", 1), ("-", -1)]:
                sql = ("d < date '%04d-%02d-%02d' %s interval '%d' %s"
                       % (y, m, dd, sign, n, unit))
                expr, _ = jit.translate(sql, schema)
                base = datetime.date(y, m, dd)
                if unit == "day":
                    want = base + datetime.timedelta(days=sgn * n)
                else:
                    months = sgn * n * (12 if unit == "year" else 1)
                    mo = base.month - 1 + months
                    wy, wm = base.year + mo // 12, mo % 12 + 1
                    if wm == 12:
                        ld = 31
                    else:
                        ld = (datetime.date(wy, wm + 1, 1)
                              - datetime.timedelta(days=1)).day
                    want = datetime.date(wy, wm, min(base.day, ld))
                days = (want - epoch).days
                assert expr == "(v0) < (%d)" % days, (sql, expr)


def test_in_lists():
    """IN / NOT IN (numeric and dict-string lists) against numpy."""
    import numpy as np
    from quokka_amd import jit
    schema = {"a": np.dtype(np.int32), "b": np.dtype(np.float64)}
    rng = np.random.default_rng(5)
    data = {"a": rng.integers(-10, 10, 2000).astype(np.int32),
            "b": np.round(rng.uniform(-2, 2, 2000), 3)}
    cases = [
        ("a in (1, 2, 5)", np.isin(data["a"], [1, 2, 5])),
        ("a not in (0, -3)", ~np.isin(data["a"], [0, -3])),
        ("b in (0.5, -1.25)", np.isin(data["b"], [0.5, -1.25])),
        ("a in (1,2) and b < 0.5",
         np.isin(data["a"], [1, 2]) & (data["b"] < 0.5)),
        ("not (a in (7))", ~np.isin(data["a"], [7])),
    ]
    for pred, want in cases:
        e, cols = jit.translate(pred, schema)
        env = {"v%d" % i: data[c] for i, c in enumerate(cols)}
        py = e.replace("&&", "&").replace("||", "|").replace("!(", "~(")
        got = eval(py, {}, env)
        assert np.array_equal(got, want), pred
    for bad in ("a in ()", "a in (1,'x')", "a in (1", "a in 1"):
        try:
            jit.translate(bad, schema)
            raise AssertionError("accepted %r" % bad)
        except ValueError:
            pass


def test_in_lists_dict_strings():
    import numpy as np
    from quokka_amd import jit

    class SD:
        codes = {"MAIL": 0, "SHIP": 1, "RAIL": 2}
    tr = jit.Translator({"m": np.dtype(np.uint8)}, {"m": SD()})
    e = tr.translate("m in ('MAIL', 'RAIL')")
    v0 = np.array([0, 1, 2, 2, 0], dtype=np.uint8)
    got = eval(e.replace("||", "|"), {}, {"v0": v0})
    assert np.array_equal(got, np.isin(v0, [0, 2]))
    # absent value never matches: folds to the constant-false expression
    # (never a sentinel code, which a later dictionary entry could take)
    e2 = jit.Translator({"m": np.dtype(np.uint8)}, {"m": SD()}) \
        .translate("m in ('TRUCK')")
    assert not np.any(eval(e2.replace("||", "|"), {}, {"v0": v0}))


def test_like_on_dict_columns():
    """LIKE/NOT LIKE resolve the SQL pattern against the dictionary
    VALUES host-side (low-cardinality dicts make LIKE a code-set test)."""
    import numpy as np
    from quokka_amd import jit

    class SD:
        codes = {"MAIL": 0, "SHIP": 1, "RAIL": 2, "REG AIR": 3, "AIR": 4}

    vals = np.array(["MAIL", "SHIP", "RAIL", "REG AIR", "AIR"])
    v0 = np.array([0, 1, 2, 3, 4, 3, 0], dtype=np.uint8)

    def run(pred):
        tr = jit.Translator({"m": np.dtype(np.uint8)}, {"m": SD()})
        e = tr.translate(pred)
        r = eval(e.replace("||", "|").replace("!(", "~("), {},
                 {"v0": v0})
        # "(0)" / "(1)" constant folds broadcast like the C scalar would
        return np.broadcast_to(np.asarray(r, dtype=bool), v0.shape)

    import fnmatch
    for pat, sql in [("*AIR", "%AIR"), ("R*", "R%"), ("?AIL", "_AIL"),
                     ("*A*", "%A%"), ("TRUCK", "TRUCK")]:
        want = np.array([fnmatch.fnmatchcase(vals[c], pat) for c in v0])
        assert np.array_equal(run("m like '%s'" % sql), want), sql
        assert np.array_equal(run("m not like '%s'" % sql), ~want), sql
    try:
        jit.Translator({"x": np.dtype(np.int64)}, {}).translate(
            "x like '5'")
        raise AssertionError("LIKE accepted on non-dict column")
    except ValueError:
        pass


def test_in_list_date_literals():
    import numpy as np
    from quokka_amd import jit
    schema = {"d": np.dtype(np.int32)}
    e, cols = jit.translate(
        "d in (date '1994-01-01', date '1995-06-17')", schema)
    d0 = (np.datetime64("1994-01-01") -
          np.datetime64("1970-01-01")).astype(int)
    d1 = (np.datetime64("1995-06-17") -
          np.datetime64("1970-01-01")).astype(int)
    v0 = np.array([d0, d1, d0 + 1, 0], dtype=np.int32)
    got = eval(e.replace("||", "|"), {}, {"v0": v0})
    assert np.array_equal(got, np.isin(v0, [d0, d1]))
