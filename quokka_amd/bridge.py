"""Device -> Arrow result bridge (the blocking-node
result path core.py:455-482 / quokka_dataset.py).

Round 1 returned results through the generic pageable bounce
(DevColumn.to_numpy: DMA into a pinned staging chunk, then a host memmove
into the pageable numpy array Arrow wraps — measured ~17 GB/s d2h). This
module removes the second copy: the DESTINATION buffer itself is pinned
host memory (hipHostMalloc), the DMA engine writes straight into it, and
pyarrow wraps it zero-copy (the numpy array IS the Arrow buffer; a
finalizer frees the pinned allocation when the last reference drops).
One DMA, no memmove — d2h at full PCIe rate.
"""
import ctypes
import weakref

import numpy as np

from . import shim
from .shim import c_u64, c_vp


class PinnedArray:
    """numpy array over pinned (hipHostMalloc) memory; freed via weakref
    finalizer when the array (and anything wrapping it, e.g. an Arrow
    buffer) is garbage-collected."""

    def __init__(self, dtype, n):
        dtype = np.dtype(dtype)
        nbytes = max(1, int(n) * dtype.itemsize)
        p = c_vp(0)
        shim.call("qk_hmalloc_impl", c_u64(nbytes), ctypes.byref(p))
        self.ptr = p
        buf = (ctypes.c_byte * nbytes).from_address(p.value)
        self.arr = np.frombuffer(buf, dtype=dtype, count=int(n))
        # the view chain (arrow Buffer -> numpy -> ctypes buf) keeps `buf`
        # alive; free the pinned block only when that chain is gone
        weakref.finalize(buf, shim._lib.qk_hfree, p)


def to_pinned_numpy(col, n=None, stream=None):
    """DevColumn -> numpy over a pinned buffer (one DMA, no memmove).
    Drop-in for DevColumn.to_numpy on result paths: pyarrow wraps the
    returned array zero-copy, so a pa.table() built from it references
    the DMA-written pinned memory directly."""
    n = col.n if n is None else int(n)
    pin = PinnedArray(col.dtype, n)
    if n:
        sh = stream.handle if stream else None
        shim.call("qk_d2h_async", sh, pin.ptr, col.ptr,
                  c_u64(n * col.dtype.itemsize))
        shim.call("qk_stream_sync", sh)
    return pin.arr


def column_to_arrow(col, n=None, stream=None):
    """One DevColumn -> pyarrow.Array over a pinned buffer (single DMA,
    zero host copies)."""
    import pyarrow as pa
    n = col.n if n is None else int(n)
    pin = PinnedArray(col.dtype, n)
    if n:
        sh = stream.handle if stream else None
        shim.call("qk_d2h_async", sh, pin.ptr, col.ptr,
                  c_u64(n * col.dtype.itemsize))
        shim.call("qk_stream_sync", sh)
    return pa.Array.from_buffers(
        pa.from_numpy_dtype(col.dtype), n,
        [None, pa.py_buffer(pin.arr)])


def table_to_arrow(cols, stream=None):
    """dict name -> DevColumn  =>  pyarrow.Table (each column one direct
    DMA into its own pinned Arrow buffer). The reference's executors hand
    host Arrow back to the runtime (core.py:455-482); this is that
    boundary at PCIe rate."""
    import pyarrow as pa
    arrays = {name: column_to_arrow(c, stream=stream)
              for name, c in cols.items()}
    return pa.table(arrays)
