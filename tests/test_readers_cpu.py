"""CPU tests of GPUCSVReader's byte-range splitting (host logic only —
csv_gpu.read_csv is stubbed with a host parser, so the ownership rules
are pinned without a GPU; the GPU suite runs the same sweep through the
real kernels). Mirrors the reference's newline-refined range splitting
(unordered_readers.py:273-442): a range owns every row whose first byte
falls inside it, exactly once, across any stride/channel choice."""
import itertools
from unittest import mock

import numpy as np
import pytest


def fake_read_csv(buf, schema, sep="|"):
    return {"a": [int(line.split(sep)[0])
                  for line in buf.decode().splitlines() if line]}


@pytest.mark.parametrize("trailing_newline", [True, False])
def test_chunk_ownership_sweep(tmp_path, trailing_newline):
    from quokka_amd import readers
    import quokka_amd.csv_gpu as cg
    rng = np.random.default_rng(17)
    vals = rng.integers(-10**9, 10**9, 400)
    text = "".join("%d|\n" % v for v in vals).encode()
    if not trailing_newline:
        text = text[:-1]
    p = tmp_path / "t.tbl"
    p.write_bytes(text)
    with mock.patch.object(cg, "read_csv", fake_read_csv):
        for stride in itertools.chain(range(3, 40),
                                      (64, 997, len(text), len(text) + 5)):
            for nch in (1, 2, 5):
                r = readers.GPUCSVReader(str(p), [("a", "i64")], sep="|",
                                         stride=stride, window=96)
                got = []
                for ch, chunks in r.get_own_state(nch).items():
                    for chunk in chunks:
                        _, cols = r.execute(ch, chunk)
                        got.extend((cols or {}).get("a", []))
                assert sorted(got) == sorted(vals.tolist()), \
                    (stride, nch, len(got))


def test_window_too_small_raises(tmp_path):
    from quokka_amd import readers
    import quokka_amd.csv_gpu as cg
    p = tmp_path / "long.csv"
    p.write_bytes(b"x" * 500 + b"|\n" + b"y" * 500 + b"|\n")
    with mock.patch.object(cg, "read_csv", fake_read_csv):
        r = readers.GPUCSVReader(str(p), [("a", "i64")], sep="|",
                                 stride=100, window=8)
        with pytest.raises(cg.QkCsvError):
            for ch, chunks in r.get_own_state(1).items():
                for chunk in chunks:
                    r.execute(ch, chunk)


def test_header_skipped_only_in_first_chunk(tmp_path):
    from quokka_amd import readers
    import quokka_amd.csv_gpu as cg
    p = tmp_path / "h.csv"
    p.write_bytes(b"colname|\n" + b"".join(b"%d|\n" % i for i in range(50)))
    with mock.patch.object(cg, "read_csv", fake_read_csv):
        for stride in (7, 16, 1000):
            r = readers.GPUCSVReader(str(p), [("a", "i64")], sep="|",
                                     stride=stride, header=True, window=64)
            got = []
            for ch, chunks in r.get_own_state(2).items():
                for chunk in chunks:
                    _, cols = r.execute(ch, chunk)
                    got.extend((cols or {}).get("a", []))
            assert sorted(got) == list(range(50)), stride
