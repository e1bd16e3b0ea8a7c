"""quokka_amd — MI355X-native columnar execution path for Quokka's hot path.

Replaces the polars/duckdb-backed operators of marsupialtail/quokka's
executors (scan+filter/project, BuildProbeJoinExecutor, SQLAggExecutor, hash
partition) with hand-written HIP/CDNA4 kernels behind the reference's own
Executor / partition-function plugin API. See DESIGN.md and INTEGRATION.md.

Importing the compute modules requires the in-tree HIP library
(quokka_amd/libquokka_amd.so, built by `make -C quokka_amd/csrc` or
__graft_entry__.build()); there is no CPU fallback.
"""
__version__ = "0.1.0"

# Executor classes are importable without the .so (picklable before first
# execute, per the registration contract quokka_runtime.py:325); the shim
# import inside them fails loudly at first execute if the library is absent.
from .executors import (Executor, GPUAggExecutor,  # noqa: F401
                        GPUBroadcastJoinExecutor,
                        GPUBuildProbeJoinExecutor, GPUCountExecutor,
                        GPUDiskBuildProbeJoinExecutor,
                        GPUDistinctExecutor, GPUSortExecutor,
                        GPUTopKExecutor,
                        gpu_partition_fn)

# Scan-reader mirrors of the reference dataset API (readers.py imports
# the decoders lazily inside execute(), so this too is .so-free at
# registration time).
from .readers import GPUParquetReader, GPUCSVReader  # noqa: F401
