"""ctypes binding of the quokka_amd C ABI (include/quokka_amd.h).

This is the ONLY place the product path touches the HIP library. If the
library is missing or was built for the wrong arch, import fails loudly —
there is no CPU fallback anywhere in quokka_amd (the CPU restatement under
oracle/ is test infrastructure and is never imported from here).
"""
import ctypes
import os

import numpy as np

_LIB_PATH = os.path.join(os.path.dirname(__file__), "libquokka_amd.so")

if not os.path.exists(_LIB_PATH):
    raise ImportError(
        "quokka_amd: HIP extension libquokka_amd.so not found at %s. "
        "Build it with `make -C quokka_amd/csrc` (hipcc --offload-arch=gfx950). "
        "quokka_amd has no CPU fallback." % _LIB_PATH
    )

_lib = ctypes.CDLL(_LIB_PATH)

c_u64 = ctypes.c_uint64
c_i64 = ctypes.c_int64
c_i32 = ctypes.c_int32
c_u32 = ctypes.c_uint32
c_u8 = ctypes.c_uint8
c_f64 = ctypes.c_double
c_vp = ctypes.c_void_p

_lib.qk_last_error.restype = ctypes.c_char_p
_lib.qk_build_arch.restype = ctypes.c_char_p

_SIGS = {
    "qk_init": [ctypes.c_int],
    "qk_device_count": [c_vp],
    "qk_dmalloc": [c_u64, c_vp],
    "qk_hmalloc_impl": [c_u64, c_vp],
    "qk_hfree": [c_vp],
    "qk_dfree": [c_vp],
    "qk_h2d": [c_vp, c_vp, c_u64],
    "qk_d2h": [c_vp, c_vp, c_u64],
    "qk_h2d_async": [c_vp, c_vp, c_vp, c_u64],
    "qk_d2h_async": [c_vp, c_vp, c_vp, c_u64],
    "qk_dmemset": [c_vp, ctypes.c_int, c_u64],
    "qk_fill_i64": [c_vp, c_vp, c_i64, c_u64],
    "qk_stream_create": [c_vp],
    "qk_stream_destroy": [c_vp],
    "qk_stream_sync": [c_vp],
    "qk_event_create": [c_vp],
    "qk_event_destroy": [c_vp],
    "qk_event_record": [c_vp, c_vp],
    "qk_stream_wait_event": [c_vp, c_vp],
    "qk_timer_create": [c_vp],
    "qk_timer_destroy": [c_vp],
    "qk_timer_start": [c_vp, c_vp],
    "qk_timer_stop": [c_vp, c_vp],
    "qk_timer_elapsed_ms": [c_vp, c_vp],
    "qk_gen_lineitem": [c_vp, c_u64, c_u64, c_u64, c_i64, c_i64, c_i64,
                        c_vp, c_vp, c_vp, c_vp, c_vp, c_vp, c_vp, c_vp, c_vp,
                        c_vp, c_vp],
    "qk_q1_agg": [c_vp, c_u64, c_vp, c_vp, c_vp, c_vp, c_vp, c_vp, c_vp,
                  c_i32, c_vp],
    "qk_q6_agg": [c_vp, c_u64, c_vp, c_vp, c_vp, c_vp, c_i32, c_i32,
                  c_f64, c_f64, c_f64, c_vp],
    "qk_filter_i32": [c_vp, c_u64, c_vp, ctypes.c_int, c_i32, c_vp, c_vp],
    "qk_filter_u8": [c_vp, c_u64, c_vp, ctypes.c_int, c_u8, c_vp, c_vp],
    "qk_filter_f64": [c_vp, c_u64, c_vp, ctypes.c_int, c_f64, c_vp,
                      c_vp],
    "qk_flag_gt_i32": [c_vp, c_u64, c_vp, c_vp, c_vp],
    "qk_mul_1md": [c_vp, c_u64, c_vp, c_vp, c_vp],
    "qk_gather_i64": [c_vp, c_u64, c_vp, c_vp, c_vp],
    "qk_gather_f64": [c_vp, c_u64, c_vp, c_vp, c_vp],
    "qk_gather_i32": [c_vp, c_u64, c_vp, c_vp, c_vp],
    "qk_gather_u8": [c_vp, c_u64, c_vp, c_vp, c_vp],
    "qk_join_build": [c_vp, c_u64, c_vp, c_u32, c_vp, c_vp, c_vp, c_u64],
    "qk_join_probe": [c_vp, c_u64, c_vp, c_vp, c_vp, c_vp, c_u64,
                      ctypes.c_int, c_vp, c_vp, c_u64, c_vp],
    "qk_gen_orders": [c_vp, c_u64, c_u64, c_u64, c_i64, c_vp, c_vp, c_vp,
                      c_vp, c_vp, c_vp, c_i64],
    "qk_gen_aux": [c_vp, c_u64, c_u64, c_u64, c_u64, ctypes.c_int,
                   c_i64, c_i64, c_vp, c_vp],
    "qk_gen_customer": [c_vp, c_u64, c_u64, c_u64, c_vp, c_vp, c_vp],
    "qk_gen_supplier": [c_vp, c_u64, c_u64, c_u64, c_vp, c_vp],
    "qk_build_keyval_i32": [c_vp, c_u64, c_vp, c_vp, c_u32, c_vp, c_vp,
                            c_u64, c_vp, c_u64],
    "qk_q5_build_orders": [c_vp, c_u64, c_vp, c_vp, c_vp, c_i32, c_i32,
                           c_vp, c_vp, c_u64, c_vp, c_vp, c_u64, c_vp, c_vp,
                           c_u64, c_vp, c_u64],
    "qk_q5_probe_agg": [c_vp, c_u64, c_vp, c_vp, c_vp, c_vp, c_vp, c_vp,
                        c_u64, c_vp, c_vp, c_u64, c_vp, c_vp],
    "qk_build_u8eq": [c_vp, c_u64, c_vp, c_vp, c_u8, c_vp, c_vp, c_u64,
                      c_vp, c_u64],
    "qk_q3_build_orders": [c_vp, c_u64, c_vp, c_vp, c_vp, c_i32, c_vp, c_vp,
                           c_u64, c_vp, c_vp, c_u64, c_vp, c_u64, c_vp,
                           c_u64],
    "qk_q3_count_orders": [c_vp, c_u64, c_vp, c_vp, c_i32, c_vp, c_vp,
                           c_u64, c_vp, c_vp, c_u64],
    "qk_q3_probe_agg": [c_vp, c_u64, c_vp, c_vp, c_vp, c_vp, c_i32, c_vp,
                        c_vp, c_u64, c_vp, c_vp],
    "qk_q3_probe_agg_nt": [c_vp, c_u64, c_vp, c_vp, c_vp, c_vp, c_i32, c_vp,
                           c_vp, c_u64, c_vp, c_vp, c_vp, c_u64],
    "qk_q3_probe_agg_nt4": [c_vp, c_u64, c_vp, c_vp, c_vp, c_vp, c_i32,
                            c_vp, c_vp, c_u64, c_vp, c_vp, c_vp, c_u64],
    "qk_q5_probe_agg_nt": [c_vp, c_u64, c_vp, c_vp, c_vp, c_vp, c_vp, c_vp,
                           c_u64, c_vp, c_vp, c_u64, c_vp, c_vp, c_vp,
                           c_u64],
    "qk_q3_extract": [c_vp, c_vp, c_vp, c_vp, c_u64, c_vp, c_vp, c_vp,
                      c_u64, c_vp],
    "qk_groupby_i64_sum": [c_vp, c_u64, c_vp, c_vp, c_vp, ctypes.c_int,
                           ctypes.c_int, c_vp, c_u64, c_vp],
    "qk_groupby_init": [c_vp, c_vp, c_u64, ctypes.c_int, ctypes.c_int,
                        c_vp],
    "qk_fill_f64": [c_vp, c_vp, c_f64, c_u64],
    "qk_groupby_extract": [c_vp, c_vp, ctypes.c_int, ctypes.c_int, c_u64,
                           c_vp, c_vp, c_u64, c_vp],
    "qk_groupby_extract_gt": [c_vp, c_vp, ctypes.c_int, ctypes.c_int,
                              c_u64, ctypes.c_int, c_f64, c_vp, c_vp,
                              c_u64, c_vp],
    "qk_bloom_count": [c_vp, c_u64, c_vp, c_vp, c_u64, c_vp],
    "qk_sort_pairs_u64": [c_vp, c_u64, c_vp, c_vp, c_vp, c_vp,
                          ctypes.c_int],
    "qk_map_f64_u64": [c_vp, c_u64, c_vp, c_vp],
    "qk_map_i64_u64": [c_vp, c_u64, c_vp, c_vp],
    "qk_iota_u32": [c_vp, c_u64, c_vp],
    "qk_bnot_u64": [c_vp, c_u64, c_vp],
    "qk_partition_hist": [c_vp, c_u64, c_vp, c_u32, c_vp],
    "qk_range_part_ids": [c_vp, c_u64, c_vp, c_i64, c_u32, c_vp],
    "qk_csv_newlines": [c_vp, c_u64, c_u64, c_vp, c_vp, c_vp],
    "qk_csv_newlines_quoted": [c_vp, c_u64, c_u64, c_vp, c_u8, c_vp, c_vp],
    "qk_csv_parse": [c_vp, c_u64, c_vp, c_u64, c_vp, c_u8, c_u8,
                     ctypes.c_int, c_vp, c_vp, c_vp, c_vp, c_vp, c_vp],
    "qk_pq_plain_copy": [c_vp, c_u64, c_vp, c_vp, c_vp, c_u32],
    "qk_pq_rle_pages": [c_vp, c_u64, c_vp, c_vp, c_vp],
    "qk_pq_walk_pages": [c_vp, c_u64, c_u64, c_i64, c_vp, c_i64, c_vp],
    "qk_snappy_pages": [c_vp, c_u64, c_vp, c_vp, c_vp, c_vp],
    "qk_gzip_pages": [c_vp, c_u64, c_vp, c_vp, c_vp, c_vp],
    "qk_str_dict_encode": [c_vp, c_u64, c_vp, c_vp, c_vp, c_vp, c_u64,
                           c_vp, c_vp, c_vp, c_vp, c_vp, c_vp],
    "qk_str_dict_rehash": [c_vp, c_u32, c_vp, c_vp, c_vp, c_vp, c_vp,
                           c_u64],
    "qk_d2d": [c_vp, c_vp, c_u64],
    "qk_i64_combine": [c_vp, c_u64, c_vp, c_vp, c_i64, c_vp],
    "qk_i64_shr": [c_vp, c_u64, c_vp, ctypes.c_int, c_vp],
    "qk_partition_scatter": [c_vp, c_u64, c_vp, c_u32, c_vp, c_vp],
}
for name, argtypes in _SIGS.items():
    fn = getattr(_lib, name)
    fn.argtypes = argtypes
    fn.restype = ctypes.c_int


class QkError(RuntimeError):
    pass


def _check(rc, name):
    if rc != 0:
        raise QkError("%s failed (rc=%d): %s"
                      % (name, rc, _lib.qk_last_error().decode()))


def call(name, *args):
    _check(getattr(_lib, name)(*args), name)


def build_arch():
    return _lib.qk_build_arch().decode()


def init(device=0):
    call("qk_init", device)


def device_count():
    n = ctypes.c_int(0)
    rc = _lib.qk_device_count(ctypes.byref(n))
    if rc == 100:  # hipErrorNoDevice: a GPU-less host has 0 devices
        return 0
    _check(rc, "qk_device_count")
    return n.value


JOIN_EMPTY = np.int64(-(2 ** 63))  # QK_JOIN_EMPTY

class _PinnedBounce:
    """Double-buffered pinned bounce for pageable h2d/d2h copies (PCIe
    runs ~2-3x faster from pinned memory; the executor boundary hands us
    pageable numpy arrays). Two 64 MB pinned chunks on two copy streams:
    the host memmove of chunk k+1 overlaps the DMA of chunk k, so the
    staging rate approaches max(memmove, DMA) instead of their sum."""

    CHUNK = 64 << 20

    MEMMOVE_THREADS = int(os.environ.get("QK_MEMMOVE_THREADS", "4"))

    def __init__(self):
        self.bufs = None
        self.streams = None
        self.pool = None

    def _mm(self, dst, src, n):
        """Parallel memmove: ctypes.memmove releases the GIL, and one
        core moves only ~10 GB/s — slice across a small thread pool."""
        if n < (8 << 20):
            ctypes.memmove(dst, src, n)
            return
        if self.pool is None:
            from concurrent.futures import ThreadPoolExecutor
            self.pool = ThreadPoolExecutor(self.MEMMOVE_THREADS)
        t = self.MEMMOVE_THREADS
        step = (n + t - 1) // t
        futs = [self.pool.submit(ctypes.memmove, dst + i * step,
                                 src + i * step,
                                 min(step, n - i * step))
                for i in range(t) if i * step < n]
        for f in futs:
            f.result()

    def _ensure(self):
        if self.bufs is None:
            bufs = []
            for _ in range(2):
                p = c_vp(0)
                rc = _lib.qk_hmalloc_impl(c_u64(self.CHUNK),
                                          ctypes.byref(p))
                if rc != 0:
                    return None
                bufs.append(p)
            streams = []
            for _ in range(2):
                s = c_vp(0)
                call("qk_stream_create", ctypes.byref(s))
                streams.append(s)
            self.bufs = bufs
            self.streams = streams
        return self.bufs

    def h2d(self, dst_dev, src_arr):
        nbytes = src_arr.nbytes
        if nbytes < (8 << 20) or self._ensure() is None:
            call("qk_h2d", dst_dev, src_arr.ctypes.data_as(c_vp),
                 c_u64(nbytes))
            return
        src = src_arr.ctypes.data_as(c_vp).value
        # blocking hipMemcpy synchronized with the null stream; keep that
        # ordering contract for the async path
        call("qk_stream_sync", None)
        nchunks = (nbytes + self.CHUNK - 1) // self.CHUNK
        for k in range(nchunks):
            off = k * self.CHUNK
            m = min(self.CHUNK, nbytes - off)
            i = k & 1
            # wait for the DMA issued from this buffer two chunks ago
            call("qk_stream_sync", self.streams[i])
            self._mm(self.bufs[i].value, src + off, m)
            call("qk_h2d_async", self.streams[i],
                 c_vp(dst_dev.value + off), self.bufs[i], c_u64(m))
        call("qk_stream_sync", self.streams[0])
        call("qk_stream_sync", self.streams[1])

    def d2h(self, dst_arr, src_dev):
        nbytes = dst_arr.nbytes
        if nbytes < (8 << 20) or self._ensure() is None:
            call("qk_d2h", dst_arr.ctypes.data_as(c_vp), src_dev,
                 c_u64(nbytes))
            return
        dst = dst_arr.ctypes.data_as(c_vp).value
        call("qk_stream_sync", None)
        nchunks = (nbytes + self.CHUNK - 1) // self.CHUNK
        sizes = [min(self.CHUNK, nbytes - k * self.CHUNK)
                 for k in range(nchunks)]
        call("qk_d2h_async", self.streams[0], self.bufs[0], src_dev,
             c_u64(sizes[0]))
        for k in range(nchunks):
            if k + 1 < nchunks:
                j = (k + 1) & 1
                call("qk_d2h_async", self.streams[j], self.bufs[j],
                     c_vp(src_dev.value + (k + 1) * self.CHUNK),
                     c_u64(sizes[k + 1]))
            i = k & 1
            call("qk_stream_sync", self.streams[i])
            self._mm(dst + k * self.CHUNK, self.bufs[i].value, sizes[k])


_bounce = _PinnedBounce()

_DTYPE_GATHER = {
    np.dtype(np.int64): "qk_gather_i64",
    np.dtype(np.float64): "qk_gather_f64",
    np.dtype(np.int32): "qk_gather_i32",
    np.dtype(np.uint8): "qk_gather_u8",
    np.dtype(np.uint32): "qk_gather_i32",  # same width
}


class _DevPool:
    """Size-exact free-list for device buffers. Allocation patterns here
    repeat identically across bench steps (same column sizes every step),
    so recycling by exact size removes the hipMalloc/hipFree churn that
    dominated the per-step exchange path (~17 GB realloc/step). Capped;
    flushed on OOM so a fresh hipMalloc can succeed."""

    def __init__(self, cap_bytes=None):
        if cap_bytes is None:
            # Cap bounds CACHED FREE bytes only (live allocations are not
            # counted). 288 GB HBM per GPU, one process per GPU: a large
            # cap is safe, and it matters — the SF100 exchange step frees/
            # reallocs ~20 GB; with a small cap those become real
            # hipFree/hipMalloc calls, which on some boxes cost ~50 ms per
            # multi-GB buffer (measured 298 vs 40 ms/step box-to-box until
            # the working set fit the pool).
            cap_bytes = int(os.environ.get("QK_POOL_CAP_GB", "128")) << 30
        self.cap = cap_bytes
        self.cached = 0
        self.buckets = {}

    def get(self, nbytes):
        lst = self.buckets.get(nbytes)
        if lst:
            self.cached -= nbytes
            return lst.pop()
        return None

    def put(self, nbytes, ptr):
        if self.cached + nbytes > self.cap:
            return False
        self.buckets.setdefault(nbytes, []).append(ptr)
        self.cached += nbytes
        return True

    def flush(self):
        for nb, lst in self.buckets.items():
            for p in lst:
                _lib.qk_dfree(p)
        self.buckets.clear()
        self.cached = 0


_pool = _DevPool()


class DevBuffer:
    """Owning device allocation (recycled through _pool)."""

    def __init__(self, nbytes):
        nbytes = max(1, nbytes)
        p = _pool.get(nbytes)
        if p is None:
            p = c_vp(0)
            rc = _lib.qk_dmalloc(c_u64(nbytes), ctypes.byref(p))
            if rc != 0:          # OOM: flush the pool and retry once
                _pool.flush()
                call("qk_dmalloc", c_u64(nbytes), ctypes.byref(p))
        self.ptr = p
        self.nbytes = nbytes

    def free(self):
        if self.ptr is not None and self.ptr.value:
            if not _pool.put(self.nbytes, self.ptr):
                call("qk_dfree", self.ptr)
            self.ptr = None

    def __del__(self):
        try:
            self.free()
        except Exception:
            pass


class DevColumn:
    """A dense device column of numpy dtype `dtype` with `n` rows."""

    def __init__(self, dtype, n):
        self.dtype = np.dtype(dtype)
        self.n = int(n)
        self.buf = DevBuffer(self.n * self.dtype.itemsize)

    @property
    def ptr(self):
        return self.buf.ptr

    @classmethod
    def from_numpy(cls, arr):
        arr = np.ascontiguousarray(arr)
        col = cls(arr.dtype, len(arr))
        if len(arr):
            _bounce.h2d(col.ptr, arr)
        return col

    def to_numpy(self, n=None):
        n = self.n if n is None else int(n)
        out = np.empty(n, dtype=self.dtype)
        if n:
            _bounce.d2h(out, self.ptr)
        return out

    def gather(self, idx_col, n_idx, stream=None):
        """New column: self[idx] for a device u32 index column. `stream`
        may be a Stream object, a raw handle, or None."""
        fn = _DTYPE_GATHER[self.dtype]
        sh = stream.handle if isinstance(stream, Stream) else stream
        out = DevColumn(self.dtype, n_idx)
        if n_idx:
            call(fn, sh, c_u64(n_idx), idx_col.ptr, self.ptr, out.ptr)
        return out

    def view(self, offset_rows, n_rows):
        """Non-owning row-range view (the overlapped exchange probes a
        received CHUNK of a column while the next chunk is in flight)."""
        return DevColumnView(self, offset_rows, n_rows)

    def free(self):
        self.buf.free()


class DevColumnView:
    """Non-owning slice of a DevColumn: same duck type (ptr/dtype/n) for
    kernel calls; free() is a no-op (the base column owns the memory)."""

    def __init__(self, base, offset_rows, n_rows):
        self.dtype = base.dtype
        self.n = int(n_rows)
        self._base = base
        self.ptr = c_vp(base.ptr.value + int(offset_rows) *
                        base.dtype.itemsize)

    def gather(self, idx_col, n_idx, stream=None):
        fn = _DTYPE_GATHER[self.dtype]
        sh = stream.handle if isinstance(stream, Stream) else stream
        out = DevColumn(self.dtype, n_idx)
        if n_idx:
            call(fn, sh, c_u64(n_idx), idx_col.ptr, self.ptr, out.ptr)
        return out

    def to_numpy(self, n=None):
        n = self.n if n is None else int(n)
        out = np.empty(n, dtype=self.dtype)
        if n:
            _bounce.d2h(out, self.ptr)
        return out

    def free(self):
        pass


class Stream:
    def __init__(self):
        p = c_vp(0)
        call("qk_stream_create", ctypes.byref(p))
        self.handle = p

    def sync(self):
        call("qk_stream_sync", self.handle)

    def destroy(self):
        if self.handle is not None:
            call("qk_stream_destroy", self.handle)
            self.handle = None


class Event:
    """Bare HIP event (no timing) for cross-stream ordering."""

    def __init__(self):
        p = c_vp(0)
        call("qk_event_create", ctypes.byref(p))
        self.handle = p

    def record(self, stream):
        call("qk_event_record", self.handle,
             stream.handle if stream else None)

    def wait(self, stream):
        """Make `stream` wait until this event's recorded point."""
        call("qk_stream_wait_event",
             stream.handle if stream else None, self.handle)

    def destroy(self):
        if self.handle is not None:
            call("qk_event_destroy", self.handle)
            self.handle = None


class Timer:
    """HIP-event timer pair on a given stream (roofline measurement)."""

    def __init__(self):
        p = c_vp(0)
        call("qk_timer_create", ctypes.byref(p))
        self.handle = p

    def start(self, stream):
        call("qk_timer_start", self.handle, stream.handle if stream else None)

    def stop(self, stream):
        call("qk_timer_stop", self.handle, stream.handle if stream else None)

    def elapsed_ms(self):
        out = ctypes.c_float(0)
        call("qk_timer_elapsed_ms", self.handle, ctypes.byref(out))
        return float(out.value)

    def destroy(self):
        if self.handle is not None:
            call("qk_timer_destroy", self.handle)
            self.handle = None
