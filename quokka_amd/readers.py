"""Scan-reader mirrors of the reference's dataset API
(pyquokka/dataset/unordered_readers.py): `get_own_state(num_channels)`
plans per-channel work, `execute(channel, payload)` produces one batch —
the contract the IOTaskManager drives (core.py:825-965). These wrap the
on-device decoders (parquet_gpu / csv_gpu), so what a channel returns is
device columns, not host Arrow.

GPUParquetReader  = InputParquetDataset   (unordered_readers.py:73-99)
GPUCSVReader      = InputDiskCSVDataset   (unordered_readers.py:273-442):
  files are split into ~stride-byte ranges; a range owns every row that
  STARTS inside it (the reference refines boundaries to newlines with a
  small read window, :372-383 — same ownership rule here, implemented by
  skipping to the first newline after `start` and parsing through the
  first newline at/after `end`).

Quoting caveat (shared with the reference's range refinement): boundary
refinement treats every newline as a row end, so a QUOTED newline that
straddles a range boundary mis-splits the row — with multi-range
strides, keep embedded newlines out of the data or use one range per
file (stride >= file size). Within a single range, csv_gpu handles
RFC-4180 quoting fully.
"""
import os


class GPUParquetReader:
    def __init__(self, filename, columns=None):
        self.filename = filename
        self.columns = columns

    def get_own_state(self, num_channels):
        if os.path.isdir(self.filename):
            files = sorted(os.path.join(self.filename, f)
                           for f in os.listdir(self.filename))
        else:
            files = [self.filename]
        state = {}
        for i, f in enumerate(files):
            state.setdefault(i % num_channels, []).append(f)
        return state

    def execute(self, mapper_id, filename=None):
        from . import parquet_gpu
        return None, parquet_gpu.read_table(filename or self.filename,
                                            self.columns)


class GPUCSVReader:
    def __init__(self, filepath, schema, sep="|", stride=16 << 20,
                 header=False, window=4096):
        self.filepath = filepath
        self.schema = schema
        self.sep = sep
        self.stride = stride
        self.header = header
        self.window = window

    def get_own_state(self, num_channels):
        if os.path.isdir(self.filepath):
            files = sorted(os.path.join(self.filepath, f)
                           for f in os.listdir(self.filepath))
        else:
            files = [self.filepath]
        chunks = []
        for f in files:
            size = os.path.getsize(f)
            start = 0
            while start < size:
                chunks.append((f, start, min(size, start + self.stride)))
                start += self.stride
        state = {}
        for i, ch in enumerate(chunks):
            state.setdefault(i % num_channels, []).append(ch)
        return state

    def execute(self, mapper_id, chunk):
        """chunk = (file, start, end): parse every row starting in
        [start, end) — rows may extend past `end` by up to `window`
        bytes (the reference's newline-refinement window)."""
        from . import csv_gpu
        f, start, end = chunk
        size = os.path.getsize(f)
        # one byte of lookback so a row starting EXACTLY at `start` is
        # recognized as ours (byte start-1 is its preceding newline) and
        # not dropped by the skip-partial-first-line rule
        base = start if start == 0 else start - 1
        with open(f, "rb") as fh:
            fh.seek(base)
            buf = fh.read(end - base + self.window)
        if start == 0:
            skip = buf.index(b"\n") + 1 if self.header else 0
        else:
            nl = buf.find(b"\n")
            if nl < 0:
                return None, {}
            skip = nl + 1
        # keep rows whose START is < end: cut after the first newline at
        # absolute position >= end-1 (the row after it starts >= end)
        limit = end - base
        if base + skip >= end:      # first complete row starts past end
            return None, {}
        if end >= size:
            buf = buf[skip:]           # the final range: runs to EOF
        else:
            tail = buf.find(b"\n", max(skip, limit - 1))
            if tail < 0:
                if base + len(buf) >= size:
                    buf = buf[skip:]   # straddling last row, no final \n
                else:
                    raise csv_gpu.QkCsvError(
                        "row longer than window=%d at %s:%d"
                        % (self.window, f, end))
            else:
                buf = buf[skip:tail + 1]
        if not buf:
            return None, {}
        return None, csv_gpu.read_csv(buf, self.schema, sep=self.sep)
