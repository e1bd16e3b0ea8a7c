"""Drop-in Executor subclasses and partition function for pyquokka.

These mirror the reference's plugin boundary exactly
(pyquokka/executors/base_executor.py:26-32, call sites core.py:632/:667,
task.py:131-134): stateful objects whose `execute(batches, stream_id,
channel_id)` consumes `list[pyarrow.Table]` and whose `done(channel_id)`
flushes. Stream mapping for joins: stream 1 = build, stream 0 = probe
(logical.py:459-471). Objects must be picklable at registration time
(quokka_runtime.py:325) — all GPU state is created lazily on first
execute, never in __init__.

A pyquokka deployment drops these in at lowering time in place of
BuildProbeJoinExecutor / SQLAggExecutor."""
import re

import numpy as np


class Executor:
    """= pyquokka.executors.base_executor.Executor (:26-32)."""

    def __init__(self):
        raise NotImplementedError

    def execute(self, batches, stream_id, executor_id):
        raise NotImplementedError

    def done(self, executor_id):
        raise NotImplementedError


def _lazy_gpu():
    """Import the HIP shim at first use (fails loudly without the .so /
    a GPU; keeps executors picklable before first execute)."""
    from . import ops, shim, staging
    return ops, shim, staging


def _to_table(d):
    import pyarrow as pa
    return pa.table({k: v for k, v in d.items()})


def _is_stringish(col):
    """pyarrow string / large_string / dictionary-of-string column."""
    import pyarrow as pa
    t = col.type
    if pa.types.is_dictionary(t):
        t = t.value_type
    return pa.types.is_string(t) or pa.types.is_large_string(t)


def _concat_col(parts):
    import pyarrow as pa
    chunks = []
    for p in parts:
        chunks.extend(p.chunks if isinstance(p, pa.ChunkedArray) else [p])
    return pa.chunked_array(chunks)


class GPUBuildProbeJoinExecutor(Executor):
    """GPU replacement for BuildProbeJoinExecutor (sql_executors.py:325-377).

    Same constructor signature and argument meaning; same stream contract
    (1 = build, vstacked into state; 0 = probe, one result per probe batch);
    same `how` set {inner, left, semi, anti}; key_to_keep renames the key
    column like :372-373. Differences documented in DESIGN.md: the build
    state is a device hash table (the reference's vstack+sort+polars-join is
    an implementation detail, :358/:369); result row ORDER within a batch is
    unspecified (as is polars'). i64 keys supported."""

    def __init__(self, on=None, left_on=None, right_on=None, how="inner",
                 key_to_keep="left"):
        if on is not None:
            if left_on is not None or right_on is not None:
                raise ValueError("pass either on= or left_on=/right_on=, "
                                 "not both (sql_executors.py:327-343)")
            self.left_on = on
            self.right_on = on
        else:
            if left_on is None or right_on is None:
                raise ValueError("join needs on= or both left_on= and "
                                 "right_on=")
            self.left_on = left_on
            self.right_on = right_on
        if how not in {"inner", "left", "semi", "anti"}:
            raise ValueError("unsupported join how=%r (inner/left/semi/"
                             "anti, as the reference)" % (how,))
        self.how = how
        self.key_to_keep = key_to_keep
        self.phase = "build"
        # lazy GPU state (created on first execute, never in __init__)
        self._table = None
        self._build_cols = None     # dict name -> list of numpy arrays
        self._build_names = None
        self._key_dict = None       # DeviceStringDict for string keys
        self._build_payload_host = {}

    def __getstate__(self):
        assert self._table is None, "executor must be pickled before first execute"
        d = dict(self.__dict__)
        return d

    def execute(self, batches, stream_id, executor_id):
        import pyarrow as pa
        ops, shim, staging = _lazy_gpu()
        batches = [b for b in batches if b is not None and len(b) > 0]
        if len(batches) == 0:
            return
        batch = pa.concat_tables(batches)

        if stream_id == 1:                       # build (state vstack)
            assert self.phase == "build"
            if self._build_cols is None:
                self._build_names = [c for c in batch.column_names]
                self._build_cols = {c: [] for c in self._build_names}
            for c in self._build_names:
                col = batch.column(c)
                if _is_stringish(col):
                    # string columns stay Arrow until _finish_build:
                    # KEYS go through the device string dictionary,
                    # payloads are host-gathered at emit
                    self._build_cols[c].append(col)
                else:
                    self._build_cols[c].append(
                        staging.column_to_numpy(col))
            return

        # probe
        if self._table is None and self._build_cols is None:
            if self.how == "anti":
                return batch
            return
        if self.phase == "build":
            self._finish_build()
        self.phase = "probe"
        return self._probe(batch)

    def _finish_build(self):
        ops, shim, staging = _lazy_gpu()
        self._host_build = {}
        self._build_payload_host = {}     # string payloads, host-gathered
        for c, v in self._build_cols.items():
            if v and not isinstance(v[0], np.ndarray):
                col = _concat_col(v)
                if c == self.right_on:
                    # string join key: device string dictionary -> dense
                    # codes; probe side encodes into the SAME dict so
                    # equal strings get equal codes (bytes verified)
                    self._key_dict = ops.DeviceStringDict(
                        expected=max(1024, col.length()))
                    self._host_build[c] = self._key_dict.encode_column(
                        col).astype(np.int64)
                else:
                    self._build_payload_host[c] = np.asarray(
                        col.to_pylist(), dtype=object)
            else:
                arr = np.concatenate(v)
                if arr.dtype == object:   # broadcast-join host strings
                    import pyarrow as pa
                    if c == self.right_on:
                        self._key_dict = ops.DeviceStringDict(
                            expected=max(1024, len(arr)))
                        self._host_build[c] = self._key_dict.encode_column(
                            pa.chunked_array([pa.array(arr)])
                        ).astype(np.int64)
                    else:
                        self._build_payload_host[c] = arr
                else:
                    self._host_build[c] = arr
        keys = self._host_build[self.right_on]
        if keys.dtype != np.int64:
            if keys.dtype.kind in "iu":   # any int key, as the reference
                keys = keys.astype(np.int64)
            else:
                raise TypeError("GPU join requires integer or string keys,"
                                " got %s" % keys.dtype)
        n = len(keys)
        self._table = ops.JoinTable(max(16, n))
        if n:
            kcol = shim.DevColumn.from_numpy(keys)
            self._table.build(kcol)
            kcol.free()
        self._build_payload_dev = {}
        for c in self._build_names:
            if c == self.right_on or c in self._build_payload_host:
                continue
            self._build_payload_dev[c] = shim.DevColumn.from_numpy(
                self._host_build[c])
        self._build_cols = None

    def _probe(self, batch):
        ops, shim, staging = _lazy_gpu()
        key_col = batch.column(self.left_on)
        if _is_stringish(key_col):
            if self._key_dict is None:
                raise TypeError("string probe keys against a non-string "
                                "build side")
            # same device dictionary as the build side: equal strings ==
            # equal codes (bytes verified in-kernel)
            probe_keys = self._key_dict.encode_column(
                key_col).astype(np.int64)
        else:
            probe_keys = staging.column_to_numpy(key_col)
        if probe_keys.dtype != np.int64:
            if probe_keys.dtype.kind in "iu":
                probe_keys = probe_keys.astype(np.int64)
            else:
                raise TypeError("GPU join requires integer or string keys,"
                                " got %s" % probe_keys.dtype)
        kcol = shim.DevColumn.from_numpy(probe_keys)
        mode = {"inner": 0, "left": 0, "semi": 1, "anti": 2}[self.how]
        pidx, bidx, nm = self._table.probe(kcol, mode=mode)

        def dev_gather(host_arr, idx_col):
            """stage -> device gather by idx -> one DMA into a PINNED
            buffer (the payload gather is the polars-join emit step,
            sql_executors.py:371; pyarrow wraps the pinned array
            zero-copy — quokka_amd.bridge, §8f row 3)."""
            from . import bridge
            src = shim.DevColumn.from_numpy(host_arr)
            g = src.gather(idx_col, nm)
            res = bridge.to_pinned_numpy(g, nm)
            src.free()
            g.free()
            return res

        # the reference renames the kept key column after EVERY join emit
        # (sql_executors.py:372-373), regardless of how/path
        def rename_key(cols):
            if self.key_to_keep == "right" and self.left_on != self.right_on \
                    and self.left_on in cols:
                cols[self.right_on] = cols.pop(self.left_on)
            return cols

        sel = pidx.to_numpy(nm)

        def probe_col_rows(c, rows_dev, rows_host):
            """One probe-side column at the selected rows: strings gather
            host-side (emit is host Arrow anyway), numerics through the
            device gather like the reference's polars emit."""
            col = batch.column(c)
            if _is_stringish(col):
                return np.asarray(col.to_pylist(), dtype=object)[rows_host]
            return dev_gather(staging.column_to_numpy(col), rows_dev)

        out = {}
        if self.how in ("semi", "anti"):
            for c in batch.column_names:
                out[c] = probe_col_rows(c, pidx, sel)
        else:
            for c in batch.column_names:
                out[c] = probe_col_rows(c, pidx, sel)
            bsel = bidx.to_numpy(nm)
            from . import bridge
            for c, dev in self._build_payload_dev.items():
                g = dev.gather(bidx, nm)
                out[c] = bridge.to_pinned_numpy(g, nm)
                g.free()
            for c, arr in self._build_payload_host.items():
                out[c] = arr[bsel]
            if self.how == "left":
                # unmatched probe rows appended with null build payload
                matched = np.zeros(len(probe_keys), dtype=bool)
                matched[sel] = True
                un = np.nonzero(~matched)[0]
                if len(un):
                    import pyarrow as pa
                    for c in batch.column_names:
                        col = batch.column(c)
                        if _is_stringish(col):
                            tailv = np.asarray(col.to_pylist(),
                                               dtype=object)[un]
                        else:
                            tailv = staging.column_to_numpy(col)[un]
                        out[c] = np.concatenate([out[c], tailv])
                    probe_cols = rename_key(
                        {k: v for k, v in out.items()
                         if k in batch.column_names})
                    tbl = _to_table(probe_cols)
                    pay = {}
                    for c in (list(self._build_payload_dev) +
                              list(self._build_payload_host)):
                        a = pa.array(out[c])
                        pay[c] = pa.chunked_array([
                            a, pa.nulls(len(un), a.type)])
                    for c, v in pay.items():
                        tbl = tbl.append_column(c, v)
                    pidx.free()
                    if bidx:
                        bidx.free()
                    kcol.free()
                    return tbl
        rename_key(out)
        pidx.free()
        if bidx:
            bidx.free()
        kcol.free()
        return _to_table(out)

    def done(self, executor_id):
        pass


_AGG_RE = re.compile(r"(sum|min|max)\s*\(\s*([A-Za-z_][A-Za-z0-9_]*)\s*\)",
                     re.I)
_AGG_OP = {"sum": 0, "min": 1, "max": 2}


class GPUAggExecutor(Executor):
    """GPU replacement for SQLAggExecutor (sql_executors.py:556-599).

    Same constructor contract: groupby_keys (list), orderby_keys (list of
    (key, 'asc'|'desc')), sql_statement = the FINAL aggregate clause the
    two-phase rewrite produces (sql_utils.py:379-413) — expressions over
    SUM(partial_col) terms with aliases, e.g.
    "sum(e0_agg_0) as revenue" or "sum(e3_agg_0) / sum(e3_agg_1) as avg_x";
    MIN(col)/MAX(col) partials are supported alongside SUM.

    execute() accumulates partial rows into a device group-by hash table
    (single i64 group key, or multiple keys composite-encoded — DESIGN.md);
    done() extracts groups, evaluates the final expressions host-side over
    the per-group sums (tiny), applies order-by, returns a pyarrow Table.
    """

    def __init__(self, groupby_keys, orderby_keys, sql_statement):
        if not isinstance(groupby_keys, list):
            raise TypeError("groupby_keys must be a list (may be empty "
                            "for a grand aggregate)")
        self.groupby_keys = groupby_keys
        self.orderby_keys = orderby_keys or []
        self.sql_statement = sql_statement
        self.sum_cols = []          # partial columns referenced by aggs
        self.agg_ops = []           # per column: 0 SUM / 1 MIN / 2 MAX
        self.exprs = []             # (alias, python expr over s['col'])
        for part in self._split_top(sql_statement):
            alias = None
            m = re.search(r"\s+as\s+([A-Za-z_][A-Za-z0-9_]*)\s*$", part,
                          re.I)
            if m:
                alias = m.group(1)
                part = part[: m.start()]
            low = part.lower()
            if re.search(r"count\s*\(\s*distinct", low):
                raise ValueError(
                    "count(distinct ...) is not distributive and cannot "
                    "appear in the FINAL aggregate this executor computes; "
                    "the reference's two-phase rewrite (sql_utils.py:299-"
                    "413) does not emit it either — pre-aggregate "
                    "distinct keys upstream (e.g. GPUDistinctExecutor)")
            if re.search(r"\bavg\s*\(", low):
                raise ValueError(
                    "avg(...) must arrive REWRITTEN as sum(partial_sum)/"
                    "sum(partial_count) (the two-phase rewrite, "
                    "sql_utils.py:379-413); avg over partials would be "
                    "wrong: %r" % part)
            hits = _AGG_RE.findall(part)
            if not hits:
                raise ValueError("unsupported aggregate (supported: "
                                 "expressions over SUM/MIN/MAX(col) of "
                                 "partial columns): %r" % part)
            for fn, c in hits:
                if c not in self.sum_cols:
                    self.sum_cols.append(c)
                    self.agg_ops.append(_AGG_OP[fn.lower()])
                elif self.agg_ops[self.sum_cols.index(c)] != _AGG_OP[fn.lower()]:
                    raise ValueError("column %r used with two different "
                                     "aggregate functions" % c)
            expr = _AGG_RE.sub(lambda m: "s[%r]" % m.group(2), part)
            self.exprs.append((alias or part.strip(), expr))
        self._gb = None
        self._key_state = None

    @staticmethod
    def _split_top(s):
        parts, depth, cur = [], 0, []
        for ch in s:
            if ch == "(":
                depth += 1
            elif ch == ")":
                depth -= 1
            if ch == "," and depth == 0:
                parts.append("".join(cur))
                cur = []
            else:
                cur.append(ch)
        if cur:
            parts.append("".join(cur))
        return [p.strip() for p in parts if p.strip()]

    def __getstate__(self):
        assert self._gb is None, "executor must be pickled before first execute"
        return dict(self.__dict__)

    def _encode_keys(self, batch):
        """Multiple group keys -> one i64 composite via accumulated host
        codebooks (DESIGN.md §Group keys). Single i64 key passes through."""
        ops, shim, staging = _lazy_gpu()
        if len(self.groupby_keys) > 3:
            raise ValueError(
                "composite group keys: at most 3 keys fit 63 bits at 21 "
                "bits/key (got %d)" % len(self.groupby_keys))
        if self._key_state is None:
            self._key_state = {"dicts": {}, "decode": []}
        arrs = []
        for k in self.groupby_keys:
            col = batch.column(k)
            if _is_stringish(col):
                # device string dictionary: unbounded cardinality, codes
                # consistent across batches (replaces the <=256-entry
                # host StringDict of round 1)
                sd = self._key_state["dicts"].get(k)
                if sd is None:
                    sd = ops.DeviceStringDict(expected=4096)
                    self._key_state["dicts"][k] = sd
                arr = sd.encode_column(col).astype(np.int64)
            else:
                arr = staging.column_to_numpy(col)
            arrs.append(arr)
        if len(arrs) == 1 and arrs[0].dtype == np.int64:
            self._key_state["mode"] = "passthrough"
            self._key_state["pass_dict"] = self._key_state["dicts"].get(
                self.groupby_keys[0])
            return arrs[0]
        # composite: each non-i64 key coded via np codebook, packed base-N
        self._key_state["mode"] = "composite"
        packed = np.zeros(len(arrs[0]), dtype=np.int64)
        comps = []
        for k, arr in zip(self.groupby_keys, arrs):
            cb = self._key_state.setdefault("codebooks", {}).setdefault(k, {})
            vals, inv = np.unique(arr, return_inverse=True)
            codes = np.empty(len(vals), dtype=np.int64)
            for i, v in enumerate(vals):
                key = v.item() if hasattr(v, "item") else v
                if key not in cb:
                    if len(cb) >= (1 << 21):
                        raise ValueError(
                            "composite group key %r exceeds 2**21 distinct "
                            "values; exceeding the 21-bit field would "
                            "silently merge distinct groups" % k)
                    cb[key] = len(cb)
                codes[i] = cb[key]
            comps.append(codes[inv])
        for c in comps:
            packed = packed * (1 << 21) + c  # 21 bits per key, <= 3 keys
        self._key_state["ncomps"] = len(comps)
        return packed

    def _decode_keys(self, packed):
        ncomps = self._key_state.get("ncomps", 1)
        outs = []
        rem = packed.astype(np.int64)
        for _ in range(ncomps):
            outs.append(rem % (1 << 21))
            rem = rem // (1 << 21)
        outs.reverse()
        cols = {}
        for k, codes in zip(self.groupby_keys, outs):
            cb = self._key_state["codebooks"][k]
            inv = {v: kk for kk, v in cb.items()}
            vals = [inv[c] for c in codes.astype(np.int64)]
            # string columns were StringDict-coded before the codebook:
            # unwind that layer too so callers get the original strings
            sd = self._key_state["dicts"].get(k)
            if sd is not None and sd.values:
                vals = [sd.values[v] for v in vals]
            cols[k] = np.array(vals)
        return cols

    def execute(self, batches, stream_id, executor_id):
        import pyarrow as pa
        ops, shim, staging = _lazy_gpu()
        batches = [b for b in batches if b is not None and len(b) > 0]
        if not batches:
            return
        batch = pa.concat_tables(batches)
        if self.groupby_keys:
            keys = self._encode_keys(batch)
        else:
            keys = np.zeros(len(batch), dtype=np.int64)
        if self._gb is None:
            self._gb = ops.GroupByI64(
                expected_groups=max(1024, len(batch)),
                nvals=len(self.sum_cols), agg_ops=self.agg_ops)
        kcol = shim.DevColumn.from_numpy(keys)
        vcols = [shim.DevColumn.from_numpy(
            staging.column_to_numpy(batch.column(c)).astype(np.float64))
            for c in self.sum_cols]
        self._gb.update(kcol, vcols)
        kcol.free()
        for v in vcols:
            v.free()

    def done(self, executor_id):
        if self._gb is None:
            return None
        keys, sums = self._gb.extract()
        s = {c: sums[i] for i, c in enumerate(self.sum_cols)}
        out = {}
        if self.groupby_keys:
            if self._key_state and self._key_state.get("mode") == "composite":
                out.update(self._decode_keys(keys))
            else:
                sd = (self._key_state or {}).get("pass_dict")
                out[self.groupby_keys[0]] = (sd.decode(keys) if sd
                                             else keys)
        for alias, expr in self.exprs:
            out[alias] = eval(expr, {"s": s})  # noqa: S307 — expr built above
        if self.orderby_keys:
            cols = []
            for item in reversed(self.orderby_keys):
                k, d = item if isinstance(item, (tuple, list)) else (item, "asc")
                v = np.asarray(out[k])
                if d == "desc":
                    if v.dtype.kind in "biuf":
                        v = -v
                    else:
                        # non-numeric desc (e.g. decoded string group keys):
                        # invert factorized rank instead of numeric negation
                        _, inv = np.unique(v, return_inverse=True)
                        v = inv.max() - inv
                cols.append(v)
            order = np.lexsort(cols)
            out = {k: np.asarray(v)[order] for k, v in out.items()}
        self._gb.free()
        self._gb = None
        return _to_table(out)


class GPUCountExecutor(Executor):
    """= CountExecutor (sql_executors.py:69-86): running row count,
    done() returns a one-row table."""

    def __init__(self):
        self.state = 0

    def execute(self, batches, stream_id, executor_id):
        self.state += sum(len(b) for b in batches
                          if b is not None)

    def done(self, executor_id):
        import pyarrow as pa
        return pa.table({"count": np.array([self.state], dtype=np.int64)})


class GPUBroadcastJoinExecutor(GPUBuildProbeJoinExecutor):
    """= BroadcastJoinExecutor (sql_executors.py:275-319): the small build
    side is given at CONSTRUCTION (broadcast to every channel) instead of
    arriving as stream 1; every execute() batch is a probe. Same device
    hash-table machinery as GPUBuildProbeJoinExecutor; picklable because
    the table is built lazily from the stored host columns."""

    def __init__(self, small_table, on=None, small_on=None, big_on=None,
                 suffix="_small", how="inner"):
        if on is not None:
            assert small_on is None and big_on is None
            small_on = big_on = on
        super().__init__(left_on=big_on, right_on=small_on, how=how)
        self.suffix = suffix
        import pyarrow as pa
        if not isinstance(small_table, pa.Table):
            small_table = pa.table(small_table)
        self._small_host = {c: small_table.column(c).to_numpy(
            zero_copy_only=False) for c in small_table.column_names}
        assert self.right_on in self._small_host

    def execute(self, batches, stream_id, executor_id):
        import pyarrow as pa
        batches = [b for b in batches if b is not None and len(b) > 0]
        if not batches:
            return
        if self._table is None:
            self._build_names = list(self._small_host.keys())
            self._build_cols = {c: [np.asarray(v)]
                                for c, v in self._small_host.items()}
            self._finish_build()
        self.phase = "probe"
        return self._probe(pa.concat_tables(batches))


class GPUDiskBuildProbeJoinExecutor(Executor):
    """GPU mirror of DiskBuildProbeJoinExecutor (sql_executors.py:456-514):
    build batches SPILL to Parquet chunks in `spill_dir` instead of
    accumulating in memory, and each probe batch joins against every
    spilled chunk with at most ONE chunk's hash table resident in HBM at
    a time — the out-of-HBM build-side path. The chunks round-trip
    through our own GPU Parquet decode (parquet_gpu), so the read-back
    is device-side like everything else.

    Same constructor signature as the reference (+ its spill_dir). Like
    the reference's per-chunk join-then-concat plan (its :505-507
    polars.concat of per-chunk lazy joins), the semantics are exact for
    INNER joins; other `how` values are refused with a typed error (the
    reference's concat silently duplicates unmatched rows for them).
    Integer keys, numeric payload columns (the spill subset)."""

    def __init__(self, on=None, left_on=None, right_on=None, how="inner",
                 key_to_keep="left", spill_dir=None):
        import secrets
        if on is not None:
            self.left_on = self.right_on = on
        else:
            if left_on is None or right_on is None:
                raise ValueError("join needs on= or both left_on= and "
                                 "right_on=")
            self.left_on, self.right_on = left_on, right_on
        if how != "inner":
            raise ValueError(
                "spill join supports how='inner' only: per-chunk "
                "join-then-concat is incorrect for left/semi/anti (the "
                "reference's DiskBuildProbeJoinExecutor concat has the "
                "same flaw); use GPUBuildProbeJoinExecutor for those")
        self.how = how
        self.key_to_keep = key_to_keep
        self.spill_dir = spill_dir or "/tmp"
        self.prefix = secrets.token_hex(6)
        self.count = 0
        self.phase = "build"

    def _path(self, executor_id, i):
        import os
        return os.path.join(self.spill_dir, "build_%s_%s_%d.parquet"
                            % (self.prefix, executor_id, i))

    def execute(self, batches, stream_id, executor_id):
        import pyarrow as pa
        import pyarrow.parquet as pq
        batches = [b for b in batches if b is not None and len(b) > 0]
        if not batches:
            return
        batch = pa.concat_tables(batches)
        if stream_id == 1:                       # build: spill the chunk
            assert self.phase == "build"
            t = batch.cast(pa.schema([pa.field(f.name, f.type,
                                               nullable=False)
                                      for f in batch.schema]))
            pq.write_table(t, self._path(executor_id, self.count),
                           compression="NONE", data_page_version="1.0",
                           use_dictionary=False)
            self.count += 1
            return
        # probe
        if self.count == 0:
            return
        self.phase = "probe"
        ops, shim, staging = _lazy_gpu()
        from . import parquet_gpu, bridge
        probe_keys = staging.column_to_numpy(batch.column(self.left_on))
        if probe_keys.dtype.kind not in "iu":
            raise TypeError("spill join requires integer keys")
        kcol = shim.DevColumn.from_numpy(probe_keys.astype(np.int64))
        probe_dev = {c: shim.DevColumn.from_numpy(
            staging.column_to_numpy(batch.column(c)))
            for c in batch.column_names}
        outs = []
        for i in range(self.count):
            cols = parquet_gpu.read_table(self._path(executor_id, i))
            bkeys = cols[self.right_on]
            if bkeys.dtype != np.dtype(np.int64):
                tmp = bkeys.to_numpy(bkeys.n).astype(np.int64)
                bkeys.free()
                bkeys = shim.DevColumn.from_numpy(tmp)
                cols[self.right_on] = bkeys
            table = ops.JoinTable(max(16, bkeys.n))
            table.build(bkeys)
            pidx, bidx, nm = table.probe(kcol)
            if nm:
                out = {}
                for c, dev in probe_dev.items():
                    g = dev.gather(pidx, nm)
                    out[c] = bridge.to_pinned_numpy(g, nm)
                    g.free()
                for c, dev in cols.items():
                    if c == self.right_on:
                        continue
                    g = dev.gather(bidx, nm)
                    out[c] = bridge.to_pinned_numpy(g, nm)
                    g.free()
                if self.key_to_keep == "right" and \
                        self.left_on != self.right_on:
                    out[self.right_on] = out.pop(self.left_on)
                outs.append(_to_table(out))
            pidx.free()
            if bidx:
                bidx.free()
            table.free()
            for dev in cols.values():
                (dev[0] if isinstance(dev, tuple) else dev).free()
        kcol.free()
        for dev in probe_dev.values():
            dev.free()
        if not outs:
            return None
        import pyarrow as pa
        return pa.concat_tables(outs)

    def done(self, executor_id):
        import os
        for i in range(self.count):
            try:
                os.unlink(self._path(executor_id, i))
            except OSError:
                pass
        return None


class GPUDistinctExecutor(Executor):
    """= DistinctExecutor (sql_executors.py:517-554): per batch, emit the
    rows whose key was never seen before (batch.unique() then anti-join
    against the accumulated state, :529-543). Device hash table keyed i64;
    done() returns None like the reference (state already emitted)."""

    def __init__(self, keys):
        self.keys = keys if isinstance(keys, str) else keys[0]
        if not isinstance(self.keys, str):
            raise TypeError("single distinct key column expected")
        self._table = None
        self._n_rows = 0
        self._key_dict = None       # DeviceStringDict for string keys

    def __getstate__(self):
        assert self._table is None, "pickle before first execute"
        return dict(self.__dict__)

    def execute(self, batches, stream_id, executor_id):
        import pyarrow as pa
        ops, shim, staging = _lazy_gpu()
        batches = [b for b in batches if b is not None and len(b) > 0]
        if not batches:
            return
        batch = pa.concat_tables(batches)
        key_col = batch.column(self.keys)
        if _is_stringish(key_col):
            # device string dictionary -> dense codes (consistent across
            # batches), distinct-on-codes == distinct-on-strings
            if self._key_dict is None:
                self._key_dict = ops.DeviceStringDict(
                    expected=max(1024, len(batch)))
            keys = self._key_dict.encode_column(key_col).astype(np.int64)
        else:
            keys = staging.column_to_numpy(key_col)
        if keys.dtype != np.int64:
            if keys.dtype.kind in "iu":
                keys = keys.astype(np.int64)
            else:
                raise TypeError("GPUDistinctExecutor requires integer or "
                                "string keys, got %s" % keys.dtype)
        # within-batch unique (first occurrence), as batch.unique() does
        _, first_idx = np.unique(keys, return_index=True)
        first_idx.sort()
        batch = batch.take(first_idx)
        keys = keys[first_idx]
        if self._table is None:
            self._table = ops.JoinTable(max(1 << 16, 4 * len(keys)))
            self._seen = []
        if self._n_rows + len(keys) > min(self._table.chain_cap,
                                          self._table.cap // 2):
            self._grow(len(keys))
        kcol = shim.DevColumn.from_numpy(keys)
        pidx, _, nm = self._table.probe(kcol, mode=2)   # anti: unseen rows
        sel = np.sort(pidx.to_numpy(nm))
        contribution = batch.take(sel)
        new_keys = keys[sel]
        if len(new_keys):
            ncol = shim.DevColumn.from_numpy(new_keys)
            self._table.build(ncol)
            ncol.free()
            self._seen.append(new_keys)
            self._n_rows += len(new_keys)
        kcol.free()
        pidx.free()
        return contribution if len(contribution) else None

    def _grow(self, incoming):
        """Rebuild a larger table from the accumulated distinct keys
        (the reference's vstack state just grows, :542)."""
        ops, shim, staging = _lazy_gpu()
        self._table.free()
        allk = np.concatenate(self._seen) if self._seen else             np.empty(0, np.int64)
        self._table = ops.JoinTable(max(1 << 16,
                                        4 * (self._n_rows + incoming)))
        if len(allk):
            kcol = shim.DevColumn.from_numpy(allk)
            self._table.build(kcol)
            kcol.free()

    def done(self, executor_id):
        return None


class GPUSortExecutor(Executor):
    """Mirror of SuperFastSortExecutor (sql_executors.py:88-187): the
    operator that makes its output stream SORTED by `key`. The reference
    implements it as a disk-backed external merge sort (spill files +
    k-way merge) because its build ran on 64 GB hosts; with 288 GB of HBM
    the state fits device memory, so this mirror accumulates batches and
    done() returns the fully sorted table via the device radix sort
    (stable, ascending — the reference's output contract)."""

    def __init__(self, key, record_batch_rows=None, output_batch_rows=None,
                 file_prefix=None):
        # extra args accepted for signature compatibility (:89); the
        # spill tuning knobs have no meaning on the device path
        self.key = key
        self._batches = None

    def __getstate__(self):
        assert self._batches is None, "pickle before first execute"
        return dict(self.__dict__)

    def execute(self, batches, stream_id, executor_id):
        import pyarrow as pa
        batches = [b for b in batches if b is not None and len(b) > 0]
        if not batches:
            return
        b = pa.concat_tables(batches)
        self._batches = b if self._batches is None             else pa.concat_tables([self._batches, b])
        return None

    def done(self, executor_id):
        if self._batches is None:
            return None
        ops, shim, staging = _lazy_gpu()
        t = self._batches
        key_col = t.column(self.key)
        if _is_stringish(key_col):
            # lexicographic string sort: dictionary codes are
            # arrival-ordered, so remap code -> lexicographic rank of
            # its value (host sorts the DISTINCT values only), then the
            # stable device radix sort runs on the ranks
            sd = ops.DeviceStringDict(expected=max(1024, t.num_rows))
            codes = sd.encode_column(key_col)
            vals = sd.values
            rank_of = np.empty(len(vals), dtype=np.int64)
            rank_of[np.argsort(np.asarray(vals, dtype=object))] = \
                np.arange(len(vals))
            keys = rank_of[codes]
            sd.free()
        else:
            keys = staging.column_to_numpy(key_col)
        kcol = shim.DevColumn.from_numpy(keys)
        perm = ops.sort_permutation(kcol)
        order = perm.to_numpy(perm.n)
        kcol.free()
        perm.free()
        return t.take(order)


class GPUTopKExecutor(Executor):
    """Mirror of ConcatThenSQLExecutor as lowered by DataStream.top_k
    (sql_executors.py:45-67; datastream.py:1746-1767): concatenate incoming
    batches, done() returns the top-k rows by `sort_keys`/`descending`.

    Like the reference (one DuckDB 'order by ... limit k' over the
    concatenated partials in done()), the final selection runs host-side —
    the inputs at this operator are the already-aggregated partials, tiny
    relative to the scan the GPU kernels consumed."""

    def __init__(self, sort_keys, k, descending=None):
        assert isinstance(sort_keys, list) and len(sort_keys) >= 1
        self.sort_keys = sort_keys
        self.k = int(k)
        self.descending = descending or [False] * len(sort_keys)
        self.state = None

    def execute(self, batches, stream_id, executor_id):
        import pyarrow as pa
        batches = [b for b in batches if b is not None and len(b) > 0]
        if not batches:
            return
        batch = pa.concat_tables(batches)
        self.state = batch if self.state is None \
            else pa.concat_tables([self.state, batch])
        return None

    def done(self, executor_id):
        if self.state is None:
            return None
        cols = []
        for key, desc in zip(reversed(self.sort_keys),
                             reversed(self.descending)):
            v = np.asarray(self.state.column(key).to_numpy(
                zero_copy_only=False))
            cols.append(-v if desc else v)
        order = np.lexsort(cols)[: self.k]
        return self.state.take(order)


def gpu_partition_fn(data, source_channel, num_target_channels, key=None,
                     predicate=None, string_dicts=None, projection=None,
                     batch_agg=None, partitioner="hash", total_range=None,
                     transforms=None):
    """GPU partition function mirroring the reference's full partition_fn
    (core.py:152-195): optional PREDICATE (filter_sql grammar, JIT-fused
    on device via quokka_amd.jit) -> optional map-side PARTIAL AGGREGATE
    (`batch_agg`, the reference's folded batch_funcs, core.py:173-176 +
    df.py:1354-1394) -> hash partition (int key -> key % N, bit-exact
    with quokka_runtime.py:222) -> optional column projection (sorted
    order, core.py:191-193).

    batch_agg = (group_keys, aggs) with group_keys a list of
    (u8_code_column, cardinality) and aggs the two-phase PARTIAL forms
    ('SUM(expr) as x' / 'COUNT(*) as n', sql_utils.py:299-413). The
    predicate and the aggregate run as ONE fused device pass
    (jit.JitAggregate — the Q1-shaped kernel), and what gets partitioned
    (by group id % N) is the tiny partial table, not rows: the map-side
    fusion that shrinks shuffle bytes from O(rows) to O(groups).
    transforms: list of (out_name, arith_expr) applied per batch AFTER
    the predicate (the reference's folded transform_sql maps inside
    batch_funcs, core.py:173-176 + datastream.py:652-815): each
    expression JIT-compiles to a device elementwise kernel (quokka_amd
    .jit.JitMap, f64 out) and the result column joins the partition +
    projection like any input column.
    Row ORDER within a partition is unspecified (the device scatter
    assigns block-chunk ranges; the reference's polars partition_by is
    input-ordered) — every consumer on this path is order-insensitive
    (joins/aggregates; sorts order explicitly).
    partitioner: "hash" (key % N, quokka_runtime.py:222), "range"
    (id = (key-1) // (total_range // N), quokka_runtime.py:234-243 —
    keys outside [1, total_range] are CLAMPED to the edge channels where
    the reference would misroute them; same key -> same channel either
    way), or "broadcast" (every channel gets the full table,
    quokka_runtime.py:245-246).
    `data`: pyarrow Table; returns dict target_channel -> pyarrow Table."""
    import pyarrow as pa
    ops, shim, staging = _lazy_gpu()
    if partitioner == "broadcast":
        return {i: data for i in range(num_target_channels)}
    assert partitioner in ("hash", "range"), partitioner
    def _col(c):
        try:
            return staging.column_to_numpy(data.column(c),
                                           (string_dicts or {}).get(c))
        except TypeError:
            col = data.column(c)
            if _is_stringish(col):       # general strings: host object
                return np.asarray(col.to_pylist(), dtype=object)
            raise

    host_cols = {c: _col(c) for c in data.column_names}

    if batch_agg is not None:
        if transforms:
            raise ValueError("batch_agg already aggregates arbitrary "
                             "arithmetic expressions; transforms cannot "
                             "be combined with it")
        from . import jit
        group_keys, aggs = batch_agg
        # COUNT(*) identifies which groups were actually observed (the
        # reference's partial tables contain only observed groups);
        # append one internally if the caller didn't ask for it
        have_count = any(a.lower().replace(" ", "").startswith("count(")
                         for a in aggs)
        run_aggs = list(aggs) + ([] if have_count
                                 else ["COUNT(*) as __qk_presence"])
        dcols = {c: shim.DevColumn.from_numpy(v)
                 for c, v in host_cols.items() if v.dtype != object}
        schema = {c: v.dtype for c, v in dcols.items()}
        agg = jit.JitAggregate(schema, group_keys, run_aggs, predicate,
                               string_dicts)
        acc = agg.make_acc()
        agg.run(dcols, acc)
        partials = agg.read(acc)               # (ngroups, naggs)
        acc.free()
        agg.free()
        for c in dcols.values():
            c.free()
        gid = np.arange(partials.shape[0])
        count_col = (len(aggs) if not have_count else
                     next(i for i, a in enumerate(aggs)
                          if a.lower().replace(" ", "")
                          .startswith("count(")))
        observed = partials[:, count_col] > 0
        cols_out = {}
        rem = gid
        for name, card in reversed(list(group_keys)):
            cols_out[name] = (rem % card).astype(np.uint8)
            rem = rem // card
        for j, aname in enumerate(agg.agg_names):
            if aname != "__qk_presence":
                cols_out[aname] = partials[:, j]
        parts = gid % num_target_channels      # group id is the int key
        out = {}
        for p in range(num_target_channels):
            m = (parts == p) & observed
            if m.any():
                out[p] = pa.table({c: v[m] for c, v in cols_out.items()})
        return out

    if predicate is not None:
        from . import jit
        dcols = {c: shim.DevColumn.from_numpy(v)
                 for c, v in host_cols.items() if v.dtype != object}
        schema = {c: v.dtype for c, v in dcols.items()}
        f = jit.JitFilter(predicate, schema, string_dicts)
        fidx, k = f.run(dcols)
        sel0 = fidx.to_numpy(k)
        host_cols = {c: v[sel0] for c, v in host_cols.items()}
        f.free()
        fidx.free()
        for c in dcols.values():
            c.free()

    if transforms:
        from . import jit
        dcols = {c: shim.DevColumn.from_numpy(v)
                 for c, v in host_cols.items() if v.dtype != object}
        schema = {c: v.dtype for c, v in dcols.items()}
        for out_name, expr in transforms:
            m = jit.JitMap(expr, schema)
            o = m.run(dcols)
            host_cols[out_name] = o.to_numpy(o.n)
            o.free()
            m.free()
        for c in dcols.values():
            c.free()

    keys = host_cols[key]
    if keys.dtype == object:
        # string partition keys: device string dictionary -> dense codes
        # (the reference hashes strings inside polars,
        # quokka_runtime.py:224 — an un-vendored implementation detail;
        # the preserved contract is equal keys -> same channel, which
        # code % N gives. Range partitioning needs an ordered key.)
        if partitioner != "hash":
            raise TypeError("string keys support hash partitioning only")
        import pyarrow as _pa
        sd = ops.DeviceStringDict(expected=max(1024, len(keys)))
        keys = sd.encode_column(
            _pa.chunked_array([_pa.array(keys)])).astype(np.int64)
        sd.free()
    if keys.dtype != np.int64:
        if keys.dtype.kind in "iu":
            keys = keys.astype(np.int64)
        else:
            raise TypeError("gpu_partition_fn: integer or string keys, "
                            "got %s" % keys.dtype)
    kcol = shim.DevColumn.from_numpy(keys)
    if partitioner == "range":
        assert total_range, "range partitioner needs total_range"
        from .shim import DevColumn, c_u64, c_i64, c_u32
        per = max(1, int(total_range) // num_target_channels)
        ids = DevColumn(np.int64, max(1, len(keys)))
        shim.call("qk_range_part_ids", None, c_u64(len(keys)), kcol.ptr,
                  c_i64(per), c_u32(num_target_channels), ids.ptr)
        kcol.free()
        kcol = ids
    offsets, idx = ops.partition_i64(kcol, num_target_channels)
    sel = idx.to_numpy(len(keys))
    names = sorted(projection) if projection else sorted(host_cols)
    out = {}
    for p in range(num_target_channels):
        lo, hi = int(offsets[p]), int(offsets[p + 1])
        if hi == lo:
            continue
        rows = sel[lo:hi]
        out[p] = pa.table({c: host_cols[c][rows] for c in names})
    kcol.free()
    idx.free()
    return out
