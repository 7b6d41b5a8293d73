"""Device ops: the gfx950 snapshot engine and elementwise reductions.

Kernels live in ``cpp/hip/snapshot_kernels.hip`` (dirty-page tracking,
sparse XOR page diff + compaction, merge apply, typed elementwise ops);
host wrappers in ``cpp/src/ops.cpp``. These replace the reference's CPU
byte-crunching loops (faabric ``src/util/snapshot.cpp``,
``src/util/dirty.cpp``) with HBM3E-speed equivalents.
"""

from faabric_amd._core import (  # noqa: F401
    DeviceSnapshot,
    bench_snapshot_pipeline,
    bench_copy,
    delta_apply,
    delta_encode,
    device_elementwise_op,
    gpu_available,
    gpu_count,
)

# Elementwise op codes (match MpiOp + extensions)
OP_SUM = 0
OP_MAX = 1
OP_MIN = 2
OP_PROD = 3
OP_SUB = 4
OP_XOR = 5

# dtype codes (match MpiDataType)
DTYPE_INT32 = 0
DTYPE_INT64 = 1
DTYPE_UINT64 = 2
DTYPE_FLOAT = 3
DTYPE_DOUBLE = 4
DTYPE_BYTE = 5
