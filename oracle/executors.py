"""CPU restatement of the reference executors' semantics — TEST INFRA ONLY.

Key-agnostic restatements (no density assumptions) used as the parity oracle
for the HIP kernels, following:

- BuildProbeJoinExecutor, /root/reference/pyquokka/executors/sql_executors.py:325-377
  (build side = stream 1 vstacked; probe batches joined how in
  {inner,left,semi,anti}; result row ORDER is not part of the contract —
  polars' hash join emits an unspecified order — so parity compares
  multisets of rows).
- partition_key_str, /root/reference/pyquokka/quokka_runtime.py:217-231:
  int keys -> key % num_target_channels (:222). For str/float keys the
  reference uses polars' internal 64-bit hash (:224) whose exact value is a
  polars implementation detail; the semantic contract is only that equal
  keys land on the same channel. We restate the int path bit-exactly and
  document our own hash for non-int keys.
- SQLAggExecutor two-phase group-by (sql_executors.py:556-599 +
  sql_utils.py:299-413): distributive partials (sum/count/min/max), final
  agg over concatenated partials.
"""
import numpy as np


def build_probe_join(build_keys, probe_keys, how="inner"):
    """Return match indices for probe (stream 0) against build (stream 1).

    inner -> (probe_idx, build_idx) int64 arrays, one entry per match pair.
    semi  -> probe_idx of probe rows with >=1 match (each row once).
    anti  -> probe_idx of probe rows with 0 matches.
    left  -> (probe_idx, build_idx) with build_idx == -1 for unmatched.
    Duplicate build keys produce one pair per (probe row, matching build row),
    as polars join does.
    """
    build_keys = np.asarray(build_keys)
    probe_keys = np.asarray(probe_keys)
    order = np.argsort(build_keys, kind="stable")
    sk = build_keys[order]
    lo = np.searchsorted(sk, probe_keys, side="left")
    hi = np.searchsorted(sk, probe_keys, side="right")
    cnt = hi - lo
    if how == "semi":
        return np.nonzero(cnt > 0)[0].astype(np.int64)
    if how == "anti":
        return np.nonzero(cnt == 0)[0].astype(np.int64)
    if how == "inner":
        probe_idx = np.repeat(np.arange(len(probe_keys), dtype=np.int64), cnt)
        ranges = [np.arange(l, h, dtype=np.int64) for l, h in zip(lo, hi) if h > l]
        build_idx = order[np.concatenate(ranges)] if ranges else np.empty(0, np.int64)
        return probe_idx, build_idx
    if how == "left":
        cnt2 = np.maximum(cnt, 1)
        probe_idx = np.repeat(np.arange(len(probe_keys), dtype=np.int64), cnt2)
        parts = []
        for l, h in zip(lo, hi):
            if h > l:
                parts.append(order[np.arange(l, h, dtype=np.int64)])
            else:
                parts.append(np.array([-1], dtype=np.int64))
        build_idx = np.concatenate(parts) if parts else np.empty(0, np.int64)
        return probe_idx, build_idx
    raise ValueError(how)


def partition_int(keys, num_target_channels):
    """quokka_runtime.py:222: int key -> key % num_target_channels.
    numpy % is the mathematical (non-negative) mod, matching the GPU
    kernels for negative keys too (the reference's polars % yields
    negative partition ids there and its runtime breaks — DESIGN.md)."""
    return np.asarray(keys, dtype=np.int64) % num_target_channels


def splitmix64(x):
    """The 64-bit finalizer quokka_amd uses for non-int partition keys and
    for hash-table slots (kernel parity reference; device twin in csrc/quokka_amd.hip)."""
    x = np.asarray(x, dtype=np.uint64).copy()
    x += np.uint64(0x9E3779B97F4A7C15)
    x ^= x >> np.uint64(30)
    x *= np.uint64(0xBF58476D1CE4E5B9)
    x ^= x >> np.uint64(27)
    x *= np.uint64(0x94D049BB133111EB)
    x ^= x >> np.uint64(31)
    return x


def groupby_sum_i64(keys, values):
    """Distributive group-by partial: sum `values` per distinct i64 key.
    Returns (unique_keys_sorted, sums)."""
    keys = np.asarray(keys, dtype=np.int64)
    uk, inv = np.unique(keys, return_inverse=True)
    sums = np.bincount(inv, weights=np.asarray(values, dtype=np.float64),
                       minlength=len(uk))
    return uk, sums
