"""SQL-predicate -> JIT-fused filter kernels (hiprtc).

The reference's partition_fn accepts arbitrary predicates: polars
expressions compiled by sql_utils.evaluate (:86-223) or raw DuckDB SQL
(core.py:157-163). This module covers the same predicate grammar the
reference's TPC-H workload uses (apps/tpc-h/tpch.py filter_sql calls):

  comparisons  = == != <> < <= > >=
  connectives  AND OR NOT, parentheses
  BETWEEN x AND y (inclusive, as DuckDB)
  expr IN (v1, v2, ...) / expr NOT IN (...) — numeric lists, or string
  lists on dict-coded columns (each element resolved through the
  column's StringDict; absent values never match)
  col LIKE 'pat' / NOT LIKE — dict-coded columns only: the SQL pattern
  (% = any run, _ = any char) is matched against the dictionary VALUES
  host-side and becomes a code-set test (no per-row string work on
  device — low-cardinality dictionaries make LIKE a set membership)
  arithmetic   + - * / on columns and literals
  literals     ints, floats, date 'YYYY-MM-DD' (+/- interval 'N'
               day/month/year, evaluated host-side to date32 days),
               single-quoted strings (equality on dictionary-coded
               columns via the caller's StringDict)

translate() lowers a predicate to a C expression over typed per-row
values v0..vN; JitFilter compiles it with hiprtc into a fused
count+scan+scatter kernel triple for gfx950 (csrc/qk_jit.cpp) and runs
it like ops.filter_col. Literal arithmetic is evaluated in f64 exactly
as the reference SQL evaluates it (e.g. 0.06 - 0.01), preserving the
bit-level comparison semantics.
"""
import ctypes
import datetime
import re

import numpy as np

from . import shim
from .shim import DevColumn, c_u64, c_vp

_EPOCH = datetime.date(1970, 1, 1)
_TYPE_CODE = {np.dtype(np.int32): 0, np.dtype(np.float64): 1,
              np.dtype(np.uint8): 2, np.dtype(np.int64): 3}

_TOKEN = re.compile(r"""
    \s*(?:
      (?P<date>date\s*'(\d{4})-(\d{2})-(\d{2})')
    | (?P<interval>interval\s*'(\d+)'\s*(day|month|year)s?)
    | (?P<num>\d+\.\d+|\.\d+|\d+)
    | (?P<str>'[^']*')
    | (?P<op><>|!=|>=|<=|==|=|<|>|\+|-|\*|/|\(|\)|,)
    | (?P<word>[A-Za-z_][A-Za-z0-9_]*)
    )""", re.X | re.I)


def _days(y, m, d):
    return (datetime.date(y, m, d) - _EPOCH).days


class _Tok:
    def __init__(self, kind, val):
        self.kind = kind
        self.val = val

    def __repr__(self):
        return "%s(%r)" % (self.kind, self.val)


def _tokenize(s):
    out, pos = [], 0
    while pos < len(s):
        m = _TOKEN.match(s, pos)
        if not m or m.end() == pos:
            if s[pos:].strip() == "":
                break
            raise ValueError("cannot tokenize %r" % s[pos:pos + 20])
        pos = m.end()
        if m.group("date"):
            out.append(_Tok("date", _days(int(m.group(2)), int(m.group(3)),
                                          int(m.group(4)))))
        elif m.group("interval"):
            n = int(m.group(6))
            unit = m.group(7).lower()
            out.append(_Tok("interval", (n, unit)))
        elif m.group("num"):
            t = m.group("num")
            out.append(_Tok("num", float(t) if "." in t else int(t)))
        elif m.group("str"):
            out.append(_Tok("str", m.group("str")[1:-1]))
        elif m.group("op"):
            out.append(_Tok("op", m.group("op")))
        else:
            w = m.group("word").upper()
            if w in ("AND", "OR", "NOT", "BETWEEN", "IN", "LIKE"):
                out.append(_Tok(w, w))
            else:
                out.append(_Tok("ident", m.group("word")))
    return out


def _last_dom(y, m):
    if m == 12:
        return 31
    return (datetime.date(y, m + 1, 1) - datetime.timedelta(days=1)).day


def _date_add(days, n, unit):
    d = _EPOCH + datetime.timedelta(days=days)
    if unit == "day":
        d = d + datetime.timedelta(days=n)
    else:
        months = n if unit == "month" else 12 * n
        mo = d.month - 1 + months
        y, m = d.year + mo // 12, mo % 12 + 1
        # SQL interval semantics: clamp the day to the target month's end
        d = datetime.date(y, m, min(d.day, _last_dom(y, m)))
    return (d - _EPOCH).days


class Translator:
    """Recursive-descent predicate -> C expression over v0..vN."""

    def __init__(self, schema, string_dicts=None):
        self.schema = schema            # name -> np dtype
        self.string_dicts = string_dicts or {}
        self.cols = []                  # referenced column names, in order

    def col_ref(self, name):
        if name not in self.schema:
            raise ValueError("unknown column %r" % name)
        if name not in self.cols:
            self.cols.append(name)
        return "v%d" % self.cols.index(name), self.schema[name]

    def translate(self, s):
        self.toks = _tokenize(s)
        self.i = 0
        expr = self.p_or()
        if self.i != len(self.toks):
            raise ValueError("trailing tokens: %r" % self.toks[self.i:])
        return expr

    def peek(self):
        return self.toks[self.i] if self.i < len(self.toks) else None

    def take(self, kind=None):
        t = self.peek()
        if t is None or (kind and t.kind != kind):
            raise ValueError("expected %s at %r" % (kind, t))
        self.i += 1
        return t

    def p_or(self):
        left = self.p_and()
        while self.peek() and self.peek().kind == "OR":
            self.take()
            left = "(%s) || (%s)" % (left, self.p_and())
        return left

    def p_and(self):
        left = self.p_not()
        while self.peek() and self.peek().kind == "AND":
            self.take()
            left = "(%s) && (%s)" % (left, self.p_not())
        return left

    def p_not(self):
        if self.peek() and self.peek().kind == "NOT":
            self.take()
            return "!(%s)" % self.p_not()
        return self.p_cmp()

    def p_cmp(self):
        try:
            return self._p_cmp_inner()
        except _Folded as f:
            return f.expr

    def _p_cmp_inner(self):
        t = self.peek()
        if t and t.kind == "op" and t.val == "(":
            # lookahead: parenthesized boolean vs arithmetic group
            save = self.i
            try:
                self.take()
                inner = self.p_or()
                close = self.take("op")
                if close.val != ")":
                    raise ValueError("expected )")
                nxt = self.peek()
                if nxt is None or nxt.kind in ("AND", "OR") or \
                        (nxt.kind == "op" and nxt.val == ")"):
                    return "(%s)" % inner
                raise ValueError("arith reparse")
            except ValueError:
                self.i = save
        left, _ = self.p_arith()
        t = self.peek()
        if t and (t.kind == "IN" or
                  (t.kind == "NOT" and self.i + 1 < len(self.toks)
                   and self.toks[self.i + 1].kind == "IN")):
            neg = t.kind == "NOT"
            self.take()
            if neg:
                self.take("IN")
            els = self._p_in_list(allow_str=False)
            body = " || ".join("((%s) == (%s))" % (left, repr(e))
                               for e in els)
            return ("!(%s)" if neg else "(%s)") % body
        if t and t.kind == "BETWEEN":
            self.take()
            lo, _ = self.p_arith()
            self.take("AND")
            hi, _ = self.p_arith()
            return "(((%s) >= (%s)) && ((%s) <= (%s)))" % (left, lo,
                                                            left, hi)
        op = self.take("op").val
        if op not in ("<", "<=", ">", ">=", "=", "==", "<>", "!="):
            raise ValueError("expected comparison operator, got %r" % op)
        cop = {"=": "==", "<>": "!="}.get(op, op)
        right, _ = self.p_arith()
        return "(%s) %s (%s)" % (left, cop, right)

    def _p_in_list(self, allow_str):
        """Parse IN's '(v1, v2, ...)' — literal numbers (and strings when
        allow_str) only."""
        p = self.take("op")   # caller already consumed IN (and NOT)
        if p.val != "(":
            raise ValueError("expected ( after IN")
        out = []
        while True:
            t = self.take()
            neg = False
            if t.kind == "op" and t.val in ("-", "+"):
                neg = t.val == "-"
                t = self.take()
            if t.kind == "num":
                out.append(-t.val if neg else t.val)
            elif t.kind == "date" and not neg:
                out.append(t.val)
            elif t.kind == "str" and allow_str and not neg:
                out.append(t.val)
            else:
                raise ValueError("bad IN list element %r" % (t,))
            t = self.take("op")
            if t.val == ")":
                break
            if t.val != ",":
                raise ValueError("expected , or ) in IN list")
        if not out:
            raise ValueError("empty IN list")
        return out

    def p_arith(self):
        left, lt = self.p_term()
        while self.peek() and self.peek().kind == "op" and \
                self.peek().val in "+-":
            op = self.take().val
            right, rt = self.p_term()
            left, lt = self._fold(left, lt, op, right, rt)
        return left, lt

    def p_term(self):
        left, lt = self.p_atom()
        while self.peek() and self.peek().kind == "op" and \
                self.peek().val in "*/":
            op = self.take().val
            right, rt = self.p_atom()
            left, lt = self._fold(left, lt, op, right, rt)
        return left, lt

    @staticmethod
    def _fold(l, lt, op, r, rt):
        # host-fold literal arithmetic in f64, exactly as the reference
        # SQL engine evaluates constant expressions (bit-identical bounds)
        if lt == "lit" and rt == "lit":
            v = eval("(%s) %s (%s)" % (l, "/" if op == "/" else op, r))
            return repr(float(v)) if isinstance(v, float) else repr(v), "lit"
        if lt == "date" and rt == "interval":
            raise ValueError("interval arithmetic handled in p_atom")
        return "(%s) %s (%s)" % (l, op, r), "mixed"

    def p_atom(self):
        t = self.take()
        if t.kind == "op" and t.val in ("-", "+"):
            e, et = self.p_atom()
            if t.val == "-":
                if et == "lit":
                    v = eval("-(%s)" % e)
                    return (repr(float(v)) if isinstance(v, float)
                            else repr(v)), "lit"
                return "(-(%s))" % e, "mixed"
            return e, et
        if t.kind == "op" and t.val == "(":
            e, et = self.p_arith()
            c = self.take("op")
            if c.val != ")":
                raise ValueError("expected )")
            return "(%s)" % e, et
        if t.kind == "num":
            return repr(t.val), "lit"
        if t.kind == "date":
            days = t.val
            # date +/- interval chains, folded host-side
            while self.peek() and self.peek().kind == "op" and \
                    self.peek().val in "+-" and self.i + 1 < len(self.toks) \
                    and self.toks[self.i + 1].kind == "interval":
                op = self.take().val
                n, unit = self.take("interval").val
                days = _date_add(days, n if op == "+" else -n, unit)
            return repr(days), "lit"
        if t.kind == "str":
            return t.val, "strlit"   # resolved by comparison partner
        if t.kind == "ident":
            ref, dtype = self.col_ref(t.val)
            # string equality: partner literal becomes the dict code
            nxt = self.peek()
            if dtype == np.dtype(np.uint8) and nxt and (
                    nxt.kind == "LIKE" or
                    (nxt.kind == "NOT" and self.i + 1 < len(self.toks)
                     and self.toks[self.i + 1].kind == "LIKE")):
                neg = nxt.kind == "NOT"
                self.take()
                if neg:
                    self.take("LIKE")
                p = self.take("str")
                sd = self.string_dicts.get(t.val)
                if sd is None:
                    raise ValueError("LIKE on %r needs its StringDict"
                                     % t.val)
                import re as _re
                rx = _re.compile(
                    "^" + "".join(
                        ".*" if ch == "%" else "." if ch == "_"
                        else _re.escape(ch) for ch in p.val) + "$", _re.S)
                codes = [c for v, c in sd.codes.items() if rx.match(v)]
                if not codes:
                    raise _Folded("(0)" if not neg else "(1)")
                body = " || ".join("((%s) == (%d))" % (ref, c)
                                   for c in sorted(codes))
                raise _Folded(("!(%s)" if neg else "(%s)") % body)
            if dtype == np.dtype(np.uint8) and nxt and (
                    nxt.kind == "IN" or
                    (nxt.kind == "NOT" and self.i + 1 < len(self.toks)
                     and self.toks[self.i + 1].kind == "IN")):
                neg = nxt.kind == "NOT"
                self.take()
                if neg:
                    self.take("IN")
                codes = []
                for el in self._p_in_list(allow_str=True):
                    if isinstance(el, str):
                        sd = self.string_dicts.get(t.val)
                        if sd is None:
                            raise ValueError(
                                "string IN list on %r needs its "
                                "StringDict" % t.val)
                        # absent literal: no code can ever equal it — drop
                        # it (a sentinel code could collide with a later
                        # legitimate dictionary entry)
                        c = sd.codes.get(el)
                        if c is not None:
                            codes.append(c)
                    else:
                        codes.append(int(el))
                if not codes:
                    raise _Folded("(1)" if neg else "(0)")
                body = " || ".join("((%s) == (%d))" % (ref, c)
                                   for c in codes)
                raise _Folded(("!(%s)" if neg else "(%s)") % body)
            if dtype == np.dtype(np.uint8) and nxt and nxt.kind == "op" \
                    and nxt.val in ("=", "==", "!=", "<>"):
                save = self.i
                op = self.take().val
                p = self.peek()
                if p is not None and p.kind == "str":
                    self.take()
                    sd = self.string_dicts.get(t.val)
                    if sd is None:
                        raise ValueError(
                            "string literal compare on %r needs its "
                            "StringDict" % t.val)
                    code = sd.codes.get(p.val)
                    if code is None:
                        # absent value: constant false for =, true for != —
                        # never a sentinel code (it could collide with a
                        # later legitimate dictionary entry)
                        raise _Folded("(1)" if op in ("!=", "<>") else "(0)")
                    cop = {"=": "==", "<>": "!="}.get(op, op)
                    raise _Folded("(%s) %s (%d)" % (ref, cop, code))
                self.i = save
            return ref, dtype
        raise ValueError("unexpected token %r" % t)


class _Folded(Exception):
    """Early-return carrier for already-complete comparisons."""

    def __init__(self, expr):
        self.expr = expr


def translate(predicate, schema, string_dicts=None):
    """-> (c_expr, ordered column names). schema: name -> np dtype."""
    tr = Translator(schema, string_dicts)
    return tr.translate(predicate), tr.cols


def translate_arith(expr, schema, translator=None):
    """Arithmetic expression (no comparisons) -> (c_expr, columns)."""
    tr = translator or Translator(schema)
    tr.toks = _tokenize(expr)
    tr.i = 0
    e, _ = tr.p_arith()
    if tr.i != len(tr.toks):
        raise ValueError("trailing tokens in %r" % expr)
    return e, tr.cols


_AGG_SQL = re.compile(r"^\s*(sum|count|min|max)\s*\(\s*(.*?)\s*\)\s*"
                      r"(?:as\s+([A-Za-z_][A-Za-z0-9_]*))?\s*$",
                      re.I | re.S)


class JitAggregate:
    """JIT-fused scan + small-cardinality group-by partial aggregate:
    the generalization of the hand-written Q1 kernel (csrc qk_jit_agg_*).

    group_keys: list of (column_name, cardinality) over u8 code columns;
    aggs: list of 'SUM(expr)' / 'COUNT(*)' strings (the map-side partial
    forms the two-phase rewrite emits, sql_utils.py:299-413);
    predicate: optional filter_sql string fused into the same pass.
    run() ACCUMULATES into a DevBuffer of ngroups*naggs f64 (executor
    state semantics, like qk_q1_agg)."""

    def __init__(self, schema, group_keys, aggs, predicate=None,
                 string_dicts=None):
        tr = Translator(schema, string_dicts)
        pred_expr = ""
        if predicate:
            pred_expr = tr.translate(predicate)
        self.group_keys = list(group_keys or [])
        self.ngroups = 1
        gparts = []
        for name, card in self.group_keys:
            ref, dt = tr.col_ref(name)
            if dt not in (np.dtype(np.uint8), np.dtype(np.int32)):
                raise TypeError("group keys must be u8 code or small-"
                                "cardinality i32 columns (values in "
                                "[0, cardinality), e.g. nation keys)")
            gparts.append((ref, card))
            self.ngroups *= int(card)
        gexpr = "0"   # [] group keys -> one grand-aggregate group
        for ref, card in gparts:
            gexpr = "(%s) * %d + (int)%s" % (gexpr, card, ref)
        self.agg_names = []
        self.agg_ops = []            # 0=SUM, 1=MIN, 2=MAX (COUNT -> SUM 1)
        agg_exprs = []
        for a in aggs:
            m = _AGG_SQL.match(a)
            if not m:
                raise ValueError("unsupported aggregate %r (SUM/MIN/MAX"
                                 "(expr) / COUNT(*))" % a)
            fn, inner, alias = m.group(1).lower(), m.group(2), m.group(3)
            if fn == "count":
                agg_exprs.append("1.0")
                self.agg_ops.append(0)
            else:
                e, _ = translate_arith(inner, schema, tr)
                agg_exprs.append(e)
                self.agg_ops.append({"sum": 0, "min": 1, "max": 2}[fn])
            self.agg_names.append(alias or a.strip())
        self.naggs = len(agg_exprs)
        self.cols = tr.cols
        self.dtypes = [np.dtype(schema[c]) for c in self.cols]

        lib = shim._lib
        lib.qk_jit_agg_build.argtypes = [
            ctypes.c_char_p, ctypes.c_char_p, ctypes.c_int, ctypes.c_int,
            ctypes.POINTER(ctypes.c_char_p), ctypes.POINTER(ctypes.c_int),
            ctypes.c_int, ctypes.POINTER(ctypes.c_int), c_vp]
        lib.qk_jit_last_error.restype = ctypes.c_char_p
        exprs_c = (ctypes.c_char_p * self.naggs)(
            *[e.encode() for e in agg_exprs])
        types_c = (ctypes.c_int * len(self.cols))(
            *[_TYPE_CODE[d] for d in self.dtypes])
        ops_c = (ctypes.c_int * self.naggs)(*self.agg_ops)
        prog = c_vp(0)
        rc = lib.qk_jit_agg_build(pred_expr.encode(), gexpr.encode(),
                                  self.ngroups, self.naggs, exprs_c,
                                  ops_c, len(self.cols), types_c,
                                  ctypes.byref(prog))
        if rc != 0:
            raise shim.QkError("jit agg build failed: %s"
                               % lib.qk_jit_last_error().decode())
        self.prog = prog

    def make_acc(self):
        """Accumulator initialized to each aggregate's identity
        (0 / +inf / -inf) — MIN/MAX groups never touched keep it."""
        from .shim import DevBuffer, c_vp as _vp
        b = DevBuffer(self.ngroups * self.naggs * 8)
        init = np.zeros((self.ngroups, self.naggs), dtype=np.float64)
        for a, o in enumerate(self.agg_ops):
            if o == 1:
                init[:, a] = np.inf
            elif o == 2:
                init[:, a] = -np.inf
        flat = np.ascontiguousarray(init.reshape(-1))
        shim.call("qk_h2d", b.ptr, flat.ctypes.data_as(_vp),
                  c_u64(flat.nbytes))
        return b

    def run(self, cols, acc, stream=None):
        lib = shim._lib
        lib.qk_jit_agg_run.argtypes = [c_vp, c_vp, c_u64,
                                       ctypes.POINTER(c_vp), c_vp]
        n = cols[self.cols[0]].n
        ptrs = (c_vp * len(self.cols))(*[cols[c].ptr for c in self.cols])
        sh = stream.handle if stream else None
        rc = lib.qk_jit_agg_run(self.prog, sh, c_u64(n), ptrs, acc.ptr)
        if rc != 0:
            raise shim.QkError("jit agg run failed: %s"
                               % lib.qk_jit_last_error().decode())

    def read(self, acc):
        """d2h -> (ngroups, naggs) f64."""
        host = np.zeros(self.ngroups * self.naggs, dtype=np.float64)
        shim.call("qk_d2h", host.ctypes.data_as(c_vp), acc.ptr,
                  c_u64(host.nbytes))
        return host.reshape(self.ngroups, self.naggs)

    def free(self):
        if self.prog:
            shim._lib.qk_jit_filter_free(self.prog)
            self.prog = None


class JitMap:
    """JIT elementwise transform: out f64 column = expr(columns) per row
    (the reference's transform_sql / with_columns_sql batch expressions,
    datastream.py:652-815; generalizes the static qk_mul_1md)."""

    def __init__(self, expr, schema):
        self.expr, self.cols = translate_arith(expr, schema)
        self.dtypes = [np.dtype(schema[c]) for c in self.cols]
        lib = shim._lib
        lib.qk_jit_map_build.argtypes = [ctypes.c_char_p, ctypes.c_int,
                                         ctypes.POINTER(ctypes.c_int), c_vp]
        lib.qk_jit_last_error.restype = ctypes.c_char_p
        types_c = (ctypes.c_int * len(self.cols))(
            *[_TYPE_CODE[d] for d in self.dtypes])
        prog = c_vp(0)
        rc = lib.qk_jit_map_build(self.expr.encode(), len(self.cols),
                                  types_c, ctypes.byref(prog))
        if rc != 0:
            raise shim.QkError("jit map build failed: %s"
                               % lib.qk_jit_last_error().decode())
        self.prog = prog

    def run(self, cols, stream=None, out=None):
        lib = shim._lib
        lib.qk_jit_map_run.argtypes = [c_vp, c_vp, c_u64,
                                       ctypes.POINTER(c_vp), c_vp]
        n = cols[self.cols[0]].n
        if out is None:
            out = DevColumn(np.float64, max(1, n))
            out.n = n
        ptrs = (c_vp * len(self.cols))(*[cols[c].ptr for c in self.cols])
        sh = stream.handle if stream else None
        rc = lib.qk_jit_map_run(self.prog, sh, c_u64(n), ptrs, out.ptr)
        if rc != 0:
            raise shim.QkError("jit map run failed: %s"
                               % lib.qk_jit_last_error().decode())
        return out

    def free(self):
        if self.prog:
            shim._lib.qk_jit_filter_free(self.prog)
            self.prog = None


class JitFilter:
    """Compiled fused filter for one predicate over a fixed schema."""

    def __init__(self, predicate, schema, string_dicts=None):
        self.expr, self.cols = translate(predicate, schema, string_dicts)
        self.dtypes = [np.dtype(schema[c]) for c in self.cols]
        types = (ctypes.c_int * len(self.cols))(
            *[_TYPE_CODE[d] for d in self.dtypes])
        prog = c_vp(0)
        lib = shim._lib
        lib.qk_jit_filter_build.argtypes = [ctypes.c_char_p, ctypes.c_int,
                                            ctypes.POINTER(ctypes.c_int),
                                            c_vp]
        lib.qk_jit_last_error.restype = ctypes.c_char_p
        rc = lib.qk_jit_filter_build(self.expr.encode(), len(self.cols),
                                     types, ctypes.byref(prog))
        if rc != 0:
            raise shim.QkError("jit build failed: %s"
                               % lib.qk_jit_last_error().decode())
        self.prog = prog

    def run(self, cols, stream=None):
        """cols: dict name -> DevColumn (must include every referenced
        column, all same length). Returns (idx DevColumn u32, count)."""
        from . import ops
        n = cols[self.cols[0]].n
        ptrs = (c_vp * len(self.cols))(
            *[cols[c].ptr for c in self.cols])
        idx = DevColumn(np.uint32, max(1, n))
        cnt = ops._count_buf()
        lib = shim._lib
        lib.qk_jit_filter_run.argtypes = [c_vp, c_vp, c_u64,
                                          ctypes.POINTER(c_vp), c_vp, c_vp]
        sh = stream.handle if stream else None
        rc = lib.qk_jit_filter_run(self.prog, sh, c_u64(n), ptrs, idx.ptr,
                                   cnt.ptr)
        if rc != 0:
            raise shim.QkError("jit run failed: %s"
                               % lib.qk_jit_last_error().decode())
        if stream:
            stream.sync()
        k = ops._read_u64(cnt)
        cnt.free()
        idx.n = k
        return idx, k

    def free(self):
        if self.prog:
            shim._lib.qk_jit_filter_free(self.prog)
            self.prog = None
