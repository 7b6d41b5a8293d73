"""MPI worlds on RCCL over xGMI: one rank per GPU.

Functions running inside executors call :func:`mpi_init` and then the
collectives; device pointers (e.g. ``tensor.data_ptr()``) go through the
world's RCCL communicator, host byte-strings ride the ordered
point-to-point broker (faabric ``src/mpi/MpiWorld.cpp`` re-designed —
see docs/ARCHITECTURE.md "MPI data planes").
"""

from faabric_amd._core import (  # noqa: F401
    MpiDataType,
    MpiOp,
    mpi_allgather_bytes,
    mpi_allgather_ptr,
    mpi_allreduce_bytes,
    mpi_allreduce_ptr,
    mpi_alltoall_bytes,
    mpi_alltoall_ptr,
    mpi_barrier,
    mpi_bcast_bytes,
    mpi_bcast_ptr,
    mpi_finalize,
    mpi_gather_bytes,
    mpi_get_host_for_rank,
    mpi_init,
    mpi_recv_bytes,
    mpi_recv_ptr,
    mpi_reduce_bytes,
    mpi_reduce_ptr,
    mpi_reducescatter_ptr,
    mpi_scan_bytes,
    mpi_scatter_bytes,
    mpi_send_bytes,
    mpi_send_ptr,
    mpi_sendrecv_bytes,
    ptp_group_barrier,
    ptp_group_lock,
    ptp_group_unlock,
    ptp_recv,
    ptp_send,
)
