"""faabric-mi355x: an MI355X-native distributed-function runtime.

A brand-new implementation of faasm/faabric's capabilities (scheduling,
messaging, state, snapshots and MPI worlds for distributed runtimes),
designed for AMD Instinct MI355X nodes: one executor slot per GPU, MPI
collectives on RCCL over xGMI, snapshot diff/merge as hand-written CDNA4
HIP kernels, and distributed state resident in HBM3E.

The runtime core is native C++20 (see ``cpp/``); this package is the thin
Python driver used by benchmarks, tests and deployment glue.
"""

import os

# torch (PyTorch-ROCm) bundles its own copies of libamdhip64/libhsa-runtime
# and dlopens them by absolute path. If _core loads first, its DT_NEEDED
# pulls /opt/rocm's runtime and the process ends up with TWO HSA runtimes —
# whichever initialises second fails device discovery intermittently.
# Importing torch first makes its libs the canonical ones (SONAME dedup),
# so _core and torch share a single HIP runtime.
try:  # pragma: no cover - environment dependent
    import torch  # noqa: F401
except Exception:  # torch genuinely absent: /opt/rocm runtime is used
    pass

# HIP runtime import is safe without a GPU; extensions are built for gfx950.
from faabric_amd._core import (  # noqa: F401
    BatchExecuteRequest,
    BatchExecuteRequestStatus,
    BatchExecuteType,
    Host,
    Message,
    MessageType,
    PointToPointMapping,
    PointToPointMappings,
    batch_exec_factory,
    generate_gid,
    get_main_thread_snapshot_key,
    get_primary_ip,
    get_usable_cores,
    is_batch_exec_request_valid,
    message_factory,
    set_log_level,
    set_mock_mode,
    set_port_offset,
)

__version__ = "0.1.0"
