"""Mid-level device operators over DevColumns.

Each operator maps 1:1 onto a reference delegate (SURVEY.md §2 native-
accounting table): filter -> polars DataFrame.filter (core.py:170);
join build/probe -> polars join (sql_executors.py:371); groupby -> DuckDB
group-by (sql_executors.py:595); partition -> partition_key_str
(quokka_runtime.py:217-231).
"""
import ctypes

import numpy as np

from . import shim
from .shim import DevBuffer, DevColumn, c_u64, c_u32, c_i64, c_i32, c_vp

# comparison ops for filter kernels
LT, LE, GT, GE, EQ, NE = range(6)


def _count_buf():
    b = DevBuffer(8)
    shim.call("qk_dmemset", b.ptr, 0, c_u64(8))
    return b


def _read_u64(buf):
    out = np.zeros(1, dtype=np.uint64)
    shim.call("qk_d2h", out.ctypes.data_as(c_vp), buf.ptr, c_u64(8))
    return int(out[0])


def filter_col(col, op, value, stream=None):
    """Ordered compaction: returns (idx DevColumn u32, count). Row order of
    passing rows is preserved (polars filter semantics)."""
    sh = stream.handle if stream else None
    idx = DevColumn(np.uint32, col.n)
    cnt = _count_buf()
    if col.dtype == np.dtype(np.int32):
        shim.call("qk_filter_i32", sh, c_u64(col.n), col.ptr, op,
                  c_i32(int(value)), idx.ptr, cnt.ptr)
    elif col.dtype == np.dtype(np.uint8):
        shim.call("qk_filter_u8", sh, c_u64(col.n), col.ptr, op,
                  ctypes.c_uint8(int(value)), idx.ptr, cnt.ptr)
    elif col.dtype == np.dtype(np.float64):
        shim.call("qk_filter_f64", sh, c_u64(col.n), col.ptr, op,
                  ctypes.c_double(float(value)), idx.ptr, cnt.ptr)
    else:
        raise TypeError("filter_col: unsupported dtype %s" % col.dtype)
    if stream:
        stream.sync()
    n = _read_u64(cnt)
    cnt.free()
    idx.n = n  # logical length; allocation stays col.n
    return idx, n


def q1_agg(n, shipdate, qty, price, disc, tax, rflag, lstat, cutoff,
           out_buf, stream=None):
    """Accumulate Q1 partials into out_buf (DevBuffer of 48 f64, zeroed by
    caller). See include/quokka_amd.h qk_q1_agg."""
    sh = stream.handle if stream else None
    shim.call("qk_q1_agg", sh, c_u64(n), shipdate.ptr, qty.ptr, price.ptr,
              disc.ptr, tax.ptr, rflag.ptr, lstat.ptr, c_i32(cutoff),
              out_buf.ptr)


def q1_read_partials(out_buf):
    """d2h the 6x8 accumulator block -> (6,6) float array
    [sum_qty,sum_base,sum_disc_price,sum_charge,sum_disc,count]."""
    host = np.zeros(48, dtype=np.float64)
    shim.call("qk_d2h", host.ctypes.data_as(c_vp), out_buf.ptr, c_u64(48 * 8))
    return host.reshape(6, 8)[:, :6].copy()


def q6_agg(n, shipdate, qty, price, disc, date_lo, date_hi, disc_lo, disc_hi,
           qty_hi, out_buf, stream=None):
    sh = stream.handle if stream else None
    shim.call("qk_q6_agg", sh, c_u64(n), shipdate.ptr, qty.ptr, price.ptr,
              disc.ptr, c_i32(date_lo), c_i32(date_hi),
              ctypes.c_double(disc_lo), ctypes.c_double(disc_hi),
              ctypes.c_double(qty_hi), out_buf.ptr)


def _pow2_at_least(n):
    c = 1
    while c < n:
        c <<= 1
    return c


class JoinTable:
    """Device hash-join state = BuildProbeJoinExecutor's vstacked build side
    (sql_executors.py:346-374). build() may be called per build batch;
    probe() per probe batch."""

    def __init__(self, expected_build_rows, stream=None):
        self.cap = _pow2_at_least(max(16, 2 * int(expected_build_rows)))
        self.stream = stream
        self.slot_keys = DevColumn(np.int64, self.cap)
        self.slot_head = DevColumn(np.int32, self.cap)
        self.chain_cap = max(16, int(expected_build_rows))
        self.chain_next = DevColumn(np.int32, self.chain_cap)
        self.n_build = 0
        sh = stream.handle if stream else None
        shim.call("qk_fill_i64", sh, self.slot_keys.ptr,
                  c_i64(int(shim.JOIN_EMPTY)), c_u64(self.cap))
        shim.call("qk_dmemset", self.slot_head.ptr, 0xFF,
                  c_u64(self.cap * 4))

    def build(self, keys_col, n=None):
        n = keys_col.n if n is None else n
        if self.n_build + n > self.chain_cap or self.n_build + n > self.cap // 2:
            raise RuntimeError(
                "JoinTable overflow: built %d + %d > capacity %d; size the "
                "table from the build side's total rows"
                % (self.n_build, n, min(self.chain_cap, self.cap // 2)))
        sh = self.stream.handle if self.stream else None
        shim.call("qk_join_build", sh, c_u64(n), keys_col.ptr,
                  c_u32(self.n_build), self.slot_keys.ptr,
                  self.slot_head.ptr, self.chain_next.ptr, c_u64(self.cap))
        self.n_build += n

    def probe(self, keys_col, mode=0, out_factor=1.5, n=None):
        """mode 0 inner / 1 semi / 2 anti. Returns (probe_idx, build_idx,
        count) DevColumns (build_idx None for semi/anti). Grows the output
        and re-probes when the first guess undershoots (counted, never
        truncated)."""
        n = keys_col.n if n is None else n
        sh = self.stream.handle if self.stream else None
        out_cap = max(16, int(n * out_factor))
        while True:
            probe_idx = DevColumn(np.uint32, out_cap)
            build_idx = DevColumn(np.uint32, out_cap) if mode == 0 else None
            cur = _count_buf()
            shim.call("qk_join_probe", sh, c_u64(n), keys_col.ptr,
                      self.slot_keys.ptr, self.slot_head.ptr,
                      self.chain_next.ptr, c_u64(self.cap), mode,
                      probe_idx.ptr, build_idx.ptr if build_idx else None,
                      c_u64(out_cap), cur.ptr)
            if self.stream:
                self.stream.sync()
            total = _read_u64(cur)
            cur.free()
            if total <= out_cap:
                probe_idx.n = total
                if build_idx is not None:
                    build_idx.n = total
                return probe_idx, build_idx, total
            probe_idx.free()
            if build_idx is not None:
                build_idx.free()
            out_cap = int(total)

    def free(self):
        self.slot_keys.free()
        self.slot_head.free()
        self.chain_next.free()


class GroupByI64:
    """Device group-by accumulate table over an i64 key with nvals f64 SUM
    columns (SQLAggExecutor's post-rewrite distributive form,
    sql_utils.py:299-413). update() per batch, extract() at done()."""

    AGG_INIT = {0: 0.0, 1: float("inf"), 2: float("-inf")}  # SUM/MIN/MAX

    def __init__(self, expected_groups, nvals, stream=None, agg_ops=None):
        """agg_ops: per value column 0=SUM, 1=MIN, 2=MAX (default all SUM),
        the distributive ops of the two-phase rewrite (sql_utils.py)."""
        self.cap = _pow2_at_least(max(16, 2 * int(expected_groups)))
        self.nvals = nvals
        self.agg_ops = list(agg_ops) if agg_ops else [0] * nvals
        assert len(self.agg_ops) == nvals
        self.stream = stream
        # INTERLEAVED slot records [key | v0..v(nvals-1) | pad], rstride
        # pow2 8-byte words: find-or-insert touches ONE random HBM line
        # per row (key and values share the line) instead of one line in
        # the key array plus one per value array — the bound the
        # high-cardinality dedups (Q13/Q16/Q18/Q21) sit on
        self.rstride = 1
        while self.rstride < 1 + nvals:
            self.rstride <<= 1
        self.table = DevColumn(np.int64, self.cap * self.rstride)
        self._inits_dev = DevColumn.from_numpy(
            np.asarray([self.AGG_INIT[op] for op in self.agg_ops],
                       dtype=np.float64))
        sh = stream.handle if stream else None
        shim.call("qk_groupby_init", sh, self.table.ptr, c_u64(self.cap),
                  self.rstride, self.nvals, self._inits_dev.ptr)
        self._ops_dev = None
        if any(self.agg_ops):
            self._ops_dev = DevColumn.from_numpy(
                np.asarray(self.agg_ops, dtype=np.int32))
        self.n_groups = 0           # exact, from the device insert counter
        self._nins = _count_buf()

    def update(self, keys_col, val_cols, n=None, max_new_groups=None):
        """max_new_groups: optional caller-known bound on DISTINCT keys
        this batch can introduce (e.g. the key domain size) — without it
        the growth check conservatively assumes every row is new."""
        n = keys_col.n if n is None else n
        if not n:
            return
        assert len(val_cols) == self.nvals
        # cumulative distinct keys can exceed the first batch's sizing: a
        # full table makes the kernel's find-or-insert loop spin forever,
        # so grow FIRST whenever this batch could push the exact group
        # count (device insert counter) past half capacity
        bound = n if max_new_groups is None else min(n, int(max_new_groups))
        if 2 * (self.n_groups + bound) > self.cap:
            self._grow(self.n_groups + bound)
        sh = self.stream.handle if self.stream else None
        ptrs = np.array([c.ptr.value if hasattr(c.ptr, "value") else c.ptr
                         for c in val_cols], dtype=np.uint64)
        dptrs = DevBuffer(ptrs.nbytes)
        shim.call("qk_h2d", dptrs.ptr, ptrs.ctypes.data_as(c_vp),
                  c_u64(ptrs.nbytes))
        shim.call("qk_groupby_i64_sum", sh, c_u64(n), keys_col.ptr,
                  dptrs.ptr, self._ops_dev.ptr if self._ops_dev else None,
                  self.nvals, self.rstride, self.table.ptr,
                  c_u64(self.cap), self._nins.ptr)
        if self.stream:
            self.stream.sync()
        self.n_groups = _read_u64(self._nins)
        dptrs.free()

    def _grow(self, need_groups):
        """Rebuild into a larger table (>= 4x the needed group count):
        extract the accumulated (keys, sums) and re-insert them — correct
        for SUM (adds into the zero identity) and MIN/MAX (single value vs
        the op identity) alike. Mirrors GPUDistinctExecutor._grow."""
        keys, sums = self.extract()
        old_groups = len(keys)
        self.table.free()
        self.cap = _pow2_at_least(max(16, 4 * int(need_groups)))
        self.table = DevColumn(np.int64, self.cap * self.rstride)
        sh = self.stream.handle if self.stream else None
        shim.call("qk_groupby_init", sh, self.table.ptr, c_u64(self.cap),
                  self.rstride, self.nvals, self._inits_dev.ptr)
        shim.call("qk_dmemset", self._nins.ptr, 0, c_u64(8))
        self.n_groups = 0
        if old_groups:
            kcol = DevColumn.from_numpy(keys)
            vcols = [DevColumn.from_numpy(np.ascontiguousarray(sums[i]))
                     for i in range(self.nvals)]
            self.update(kcol, vcols, old_groups)
            kcol.free()
            for v in vcols:
                v.free()

    def extract(self):
        """-> (keys np.int64[K], sums np.float64[nvals, K]); unordered."""
        sh = self.stream.handle if self.stream else None
        out_cap = self.cap
        out_keys = DevColumn(np.int64, out_cap)
        out_sums = DevColumn(np.float64, out_cap * self.nvals)
        cur = _count_buf()
        shim.call("qk_groupby_extract", sh, self.table.ptr,
                  self.rstride, self.nvals, c_u64(self.cap),
                  out_keys.ptr, out_sums.ptr, c_u64(out_cap), cur.ptr)
        if self.stream:
            self.stream.sync()
        k = _read_u64(cur)
        cur.free()
        keys = out_keys.to_numpy(k)
        # d2h only the K occupied entries of each value column (the
        # compaction writes column c at [c*out_cap, c*out_cap + K) —
        # a full-capacity copy measured 100s of ms on 2^27-slot tables)
        sums = np.empty((self.nvals, k), dtype=np.float64)
        for c in range(self.nvals):
            if k:
                shim.call("qk_d2h", sums[c].ctypes.data_as(c_vp),
                          shim.c_vp(out_sums.ptr.value + c * out_cap * 8),
                          c_u64(k * 8))
        out_keys.free()
        out_sums.free()
        return keys, sums

    def extract_device(self):
        """Like extract() but the compacted (keys, sums) STAY ON DEVICE:
        returns (keys DevColumn i64 [n], sums DevColumn f64
        [nvals x out_cap, column-major with stride out_cap], n, out_cap).
        Caller frees both columns. Feeds device-side second-level
        aggregation (e.g. Q21's per-order distinct-supplier counts)."""
        sh = self.stream.handle if self.stream else None
        out_cap = self.cap
        out_keys = DevColumn(np.int64, out_cap)
        out_sums = DevColumn(np.float64, out_cap * self.nvals)
        cur = _count_buf()
        shim.call("qk_groupby_extract", sh, self.table.ptr,
                  self.rstride, self.nvals, c_u64(self.cap),
                  out_keys.ptr, out_sums.ptr, c_u64(out_cap), cur.ptr)
        if self.stream:
            self.stream.sync()
        k = _read_u64(cur)
        cur.free()
        out_keys.n = k
        return out_keys, out_sums, k, out_cap

    def extract_where_gt(self, col, threshold, out_guess=1 << 20):
        """Extract only groups whose sums[col] > threshold (HAVING — e.g.
        Q18's sum(l_quantity) > 300): the d2h stays proportional to the
        QUALIFYING groups, not the table. Returns (keys, sums) like
        extract(); reruns with a larger buffer if the guess undershoots
        (counted, never truncated)."""
        import ctypes as _ct
        sh = self.stream.handle if self.stream else None
        out_cap = min(self.cap, max(16, int(out_guess)))
        while True:
            out_keys = DevColumn(np.int64, out_cap)
            out_sums = DevColumn(np.float64, out_cap * self.nvals)
            cur = _count_buf()
            shim.call("qk_groupby_extract_gt", sh, self.table.ptr,
                      self.rstride, self.nvals, c_u64(self.cap),
                      int(col), _ct.c_double(float(threshold)),
                      out_keys.ptr, out_sums.ptr, c_u64(out_cap), cur.ptr)
            if self.stream:
                self.stream.sync()
            k = _read_u64(cur)
            cur.free()
            if k <= out_cap:
                keys = out_keys.to_numpy(k)
                sums = out_sums.to_numpy(out_cap * self.nvals).reshape(
                    self.nvals, out_cap)[:, :k].copy()
                out_keys.free()
                out_sums.free()
                return keys, sums
            out_keys.free()
            out_sums.free()
            out_cap = int(k)

    def free(self):
        self.table.free()
        self._inits_dev.free()
        self._nins.free()
        if self._ops_dev is not None:
            self._ops_dev.free()


class DeviceStringDict:
    """Device-resident accumulating string dictionary: arbitrary-width,
    unbounded-cardinality string keys -> dense u32 codes, consistent
    across batches (the table and byte arena persist; growth re-seats
    hashes without changing codes). Replaces the round-1 host
    `staging.StringDict` (<=256 values) for join/group-by keys — the
    reference handles such keys inside polars (sql_executors.py:325-377,
    :556-599).

    encode(offsets, bytes) takes Arrow-layout string data (int64 offsets
    of n+1 entries + the byte buffer) and returns np.uint32 codes[n].
    `values` materializes decoded strings lazily (host list, index ==
    code)."""

    def __init__(self, expected=1024, stream=None):
        self.stream = stream
        self.cap = _pow2_at_least(max(64, 4 * int(expected)))
        self._alloc_table(self.cap)
        self.code_cap = max(1024, self.cap // 2)
        self.code_off = DevColumn(np.uint64, self.code_cap)
        self.code_len = DevColumn(np.uint32, self.code_cap)
        self.arena_cap = 1 << 20
        self.arena = DevBuffer(self.arena_cap)
        self._ctr = DevBuffer(16)           # [arena_cursor u64, counter u32]
        shim.call("qk_dmemset", self._ctr.ptr, 0, c_u64(16))
        self.n_codes = 0
        self.arena_used = 0
        self._values = []                   # host cache, len == decoded so far

    def _alloc_table(self, cap):
        self.slot_hash = DevColumn(np.uint64, cap)
        self.slot_code = DevColumn(np.int32, cap)
        shim.call("qk_dmemset", self.slot_hash.ptr, 0, c_u64(cap * 8))
        shim.call("qk_dmemset", self.slot_code.ptr, 0xFF, c_u64(cap * 4))

    @property
    def _cursor_ptrs(self):
        import ctypes as _ct
        return (self._ctr.ptr,
                shim.c_vp(self._ctr.ptr.value + 8))

    def encode(self, offsets, data, stream=None):
        """offsets: np.int64[n+1] (or any int np array), data: np.uint8
        bytes buffer. Returns np.uint32[n] codes."""
        st = stream or self.stream
        sh = st.handle if st else None
        offsets = np.ascontiguousarray(offsets, dtype=np.int64)
        n = len(offsets) - 1
        if n <= 0:
            return np.empty(0, dtype=np.uint32)
        data = np.ascontiguousarray(data, dtype=np.uint8)
        # worst case: every row is a new distinct string
        self._ensure(n, int(offsets[-1] - offsets[0]), sh)
        doff = DevColumn.from_numpy(offsets - offsets[0])
        dbytes = DevColumn.from_numpy(
            data[int(offsets[0]):int(offsets[-1])] if offsets[0] else
            data[: int(offsets[-1])])
        out = DevColumn(np.uint32, n)
        acur, cnt = self._cursor_ptrs
        shim.call("qk_str_dict_encode", sh, c_u64(n), doff.ptr, dbytes.ptr,
                  self.slot_hash.ptr, self.slot_code.ptr, c_u64(self.cap),
                  self.code_off.ptr, self.code_len.ptr, self.arena.ptr,
                  acur, cnt, out.ptr)
        if st:
            st.sync()
        else:
            shim.call("qk_stream_sync", None)
        host = np.zeros(2, dtype=np.uint64)
        shim.call("qk_d2h", host.ctypes.data_as(c_vp), self._ctr.ptr,
                  c_u64(16))
        self.arena_used = int(host[0])
        self.n_codes = int(np.uint64(host[1]) & np.uint64(0xFFFFFFFF))
        codes = out.to_numpy(n)
        doff.free(); dbytes.free(); out.free()
        return codes

    def encode_column(self, col, stream=None):
        """pyarrow string/large_string column -> np.uint32 codes."""
        import pyarrow as pa
        if isinstance(col, pa.ChunkedArray):
            col = col.combine_chunks()
        if pa.types.is_dictionary(col.type):
            col = col.dictionary_decode() if hasattr(col, "dictionary_decode") \
                else col.cast(col.type.value_type)
        if col.null_count:
            raise TypeError("null string keys unsupported")
        bufs = col.buffers()
        width = 8 if pa.types.is_large_string(col.type) else 4
        odt = np.int64 if width == 8 else np.int32
        off = np.frombuffer(bufs[1], dtype=odt,
                            count=len(col) + 1 + col.offset)[col.offset:]
        data = np.frombuffer(bufs[2], dtype=np.uint8)
        return self.encode(off.astype(np.int64), data, stream)

    def _ensure(self, n_new, nbytes_new, sh):
        need_codes = self.n_codes + n_new
        if 2 * need_codes > self.cap:
            new_cap = _pow2_at_least(4 * need_codes)
            self.slot_hash.free(); self.slot_code.free()
            self._alloc_table(new_cap)
            self.cap = new_cap
            shim.call("qk_str_dict_rehash", sh, c_u32(self.n_codes),
                      self.code_off.ptr, self.code_len.ptr, self.arena.ptr,
                      self.slot_hash.ptr, self.slot_code.ptr,
                      c_u64(self.cap))
        if need_codes > self.code_cap:
            new_cc = _pow2_at_least(2 * need_codes)
            for name in ("code_off", "code_len"):
                old = getattr(self, name)
                new = DevColumn(old.dtype, new_cc)
                if self.n_codes:
                    shim.call("qk_d2d", new.ptr, old.ptr,
                              c_u64(self.n_codes * old.dtype.itemsize))
                old.free()
                setattr(self, name, new)
            self.code_cap = new_cc
        if self.arena_used + nbytes_new > self.arena_cap:
            new_ac = max(2 * self.arena_cap,
                         self.arena_used + nbytes_new)
            new_ar = DevBuffer(new_ac)
            if self.arena_used:
                shim.call("qk_d2d", new_ar.ptr, self.arena.ptr,
                          c_u64(self.arena_used))
            self.arena.free()
            self.arena = new_ar
            self.arena_cap = new_ac

    @property
    def values(self):
        """Decoded strings, index == code (lazily materialized)."""
        if len(self._values) < self.n_codes:
            k0 = len(self._values)
            k = self.n_codes - k0
            offs = self.code_off.to_numpy(self.n_codes)[k0:]
            lens = self.code_len.to_numpy(self.n_codes)[k0:]
            lo = int(offs.min()) if k else 0
            hi = int((offs + lens).max()) if k else 0
            blob = np.empty(max(1, hi - lo), dtype=np.uint8)
            if hi > lo:
                shim.call("qk_d2h", blob.ctypes.data_as(c_vp),
                          shim.c_vp(self.arena.ptr.value + lo),
                          c_u64(hi - lo))
            raw = blob.tobytes()
            for o, ln in zip(offs, lens):
                s = int(o) - lo
                self._values.append(raw[s:s + int(ln)].decode())
        return self._values

    def decode(self, codes):
        vals = np.asarray(self.values, dtype=object)
        return vals[np.asarray(codes, dtype=np.int64)]

    def free(self):
        for c in (self.slot_hash, self.slot_code, self.code_off,
                  self.code_len):
            c.free()
        self.arena.free()
        self._ctr.free()


def sort_permutation(col, stream=None, descending=False):
    """Stable sort permutation of a DevColumn (i64/f64/i32/u32 keys):
    returns a u32 DevColumn `perm` with rows in ascending (or descending)
    key order; equal keys keep their original relative order (stable,
    like polars sort — sql_executors.py:369 build-side sort semantics).
    Device LSD radix sort (qk_sort_pairs_u64)."""
    n = col.n
    sh = stream.handle if stream else None
    keys = DevColumn(np.uint64, max(1, n))
    if col.dtype == np.dtype(np.float64):
        shim.call("qk_map_f64_u64", sh, c_u64(n), col.ptr, keys.ptr)
    elif col.dtype == np.dtype(np.int64):
        shim.call("qk_map_i64_u64", sh, c_u64(n), col.ptr, keys.ptr)
    elif col.dtype in (np.dtype(np.int32), np.dtype(np.uint32)):
        tmp = DevColumn(np.int64, max(1, n))
        host = col.to_numpy(n).astype(np.int64)   # widen via host (i32 cols are small in this path)
        tmp2 = DevColumn.from_numpy(host)
        shim.call("qk_map_i64_u64", sh, c_u64(n), tmp2.ptr, keys.ptr)
        tmp.free(); tmp2.free()
    else:
        raise TypeError("sort_permutation: unsupported dtype %s" % col.dtype)
    if descending:
        # stable descending == stable ascending on complemented key image
        shim.call("qk_bnot_u64", sh, c_u64(n), keys.ptr)
    perm = DevColumn(np.uint32, max(1, n))
    shim.call("qk_iota_u32", sh, c_u64(n), perm.ptr)
    ktmp = DevColumn(np.uint64, max(1, n))
    ptmp = DevColumn(np.uint32, max(1, n))
    shim.call("qk_sort_pairs_u64", sh, c_u64(n), keys.ptr, perm.ptr,
              ktmp.ptr, ptmp.ptr, -1)
    if stream:
        stream.sync()
    keys.free(); ktmp.free(); ptmp.free()
    perm.n = n
    return perm


def partition_i64(keys_col, nparts, stream=None, n=None):
    """Hash partition, int-key semantics key % nparts
    (quokka_runtime.py:222). Returns (offsets np.uint64[nparts+1],
    idx DevColumn u32 with rows grouped by partition)."""
    n = keys_col.n if n is None else n
    sh = stream.handle if stream else None
    hist = DevBuffer(nparts * 8)
    shim.call("qk_dmemset", hist.ptr, 0, c_u64(nparts * 8))
    shim.call("qk_partition_hist", sh, c_u64(n), keys_col.ptr,
              c_u32(nparts), hist.ptr)
    if stream:
        stream.sync()
    h = np.zeros(nparts, dtype=np.uint64)
    shim.call("qk_d2h", h.ctypes.data_as(c_vp), hist.ptr, c_u64(nparts * 8))
    offsets = np.zeros(nparts + 1, dtype=np.uint64)
    np.cumsum(h, out=offsets[1:])
    cursors = DevBuffer(nparts * 8)
    shim.call("qk_h2d", cursors.ptr,
              np.ascontiguousarray(offsets[:nparts]).ctypes.data_as(c_vp),
              c_u64(nparts * 8))
    idx = DevColumn(np.uint32, max(1, n))
    shim.call("qk_partition_scatter", sh, c_u64(n), keys_col.ptr,
              c_u32(nparts), cursors.ptr, idx.ptr)
    if stream:
        stream.sync()
    hist.free()
    cursors.free()
    idx.n = n
    return offsets, idx
