"""Host->HBM staging of Arrow/numpy columns.

The reference hands executors `list[pyarrow.Table]` (core.py:627-632); this
module turns those into dense device columns. Parquet/CSV/Arrow decode stays
on the host for now (SURVEY.md §8f row 1 is the planned GPU decode);
columns are copied over PCIe once and stay resident.

Supported physical types: int64, float64, int32/date32, uint8, and
strings/dictionaries (dictionary-encoded to u8 codes with a per-Dictionary
mapping maintained by the caller — low-cardinality TPC-H flags/segments)."""
import numpy as np

from .shim import DevColumn


class StringDict:
    """Accumulating host-side dictionary str -> u8 code, consistent across
    batches of one executor/channel."""

    def __init__(self):
        self.codes = {}
        self.values = []

    def encode(self, arr):
        out = np.empty(len(arr), dtype=np.uint8)
        codes = self.codes
        for i, v in enumerate(arr):
            c = codes.get(v)
            if c is None:
                c = len(self.values)
                if c > 255:
                    raise ValueError("StringDict overflow (>256 distinct)")
                codes[v] = c
                self.values.append(v)
            out[i] = c
        return out

    def decode(self, codes):
        vals = np.asarray(self.values, dtype=object)
        return vals[np.asarray(codes)]


def column_to_numpy(col, string_dict=None):
    """pyarrow ChunkedArray/Array (or numpy) -> dense numpy array of a
    device-supported dtype."""
    import pyarrow as pa

    if isinstance(col, np.ndarray):
        arr = col
    else:
        if isinstance(col, pa.ChunkedArray):
            col = col.combine_chunks()
        t = col.type
        if pa.types.is_dictionary(t):
            col = col.dictionary_decode() if hasattr(col, "dictionary_decode") \
                else col.cast(t.value_type)
        if pa.types.is_string(col.type) or pa.types.is_large_string(col.type):
            if string_dict is None:
                raise TypeError("string column needs a StringDict")
            return string_dict.encode(col.to_pylist())
        arr = col.to_numpy(zero_copy_only=False)
    if arr.dtype == np.dtype("datetime64[D]"):
        arr = arr.astype(np.int32)
    if arr.dtype.kind == "M":  # other datetime64 units -> days
        arr = arr.astype("datetime64[D]").astype(np.int32)
    if arr.dtype == np.bool_:
        arr = arr.astype(np.uint8)
    if arr.dtype not in (np.dtype(np.int64), np.dtype(np.float64),
                         np.dtype(np.int32), np.dtype(np.uint8),
                         np.dtype(np.uint32)):
        if arr.dtype.kind == "i":
            arr = arr.astype(np.int64)
        elif arr.dtype.kind == "f":
            arr = arr.astype(np.float64)
        else:
            raise TypeError("unsupported column dtype %s" % arr.dtype)
    return arr


def stage_columns(table, names=None, string_dicts=None):
    """pyarrow Table or dict-of-numpy -> dict name -> DevColumn."""
    out = {}
    if isinstance(table, dict):
        items = table.items() if names is None else [(n, table[n]) for n in names]
        for name, arr in items:
            out[name] = DevColumn.from_numpy(
                column_to_numpy(arr, (string_dicts or {}).get(name)))
        return out
    names = names or table.column_names
    for name in names:
        arr = column_to_numpy(table.column(name),
                              (string_dicts or {}).get(name))
        out[name] = DevColumn.from_numpy(arr)
    return out
