"""MPI world tests on the host data plane (CPU): world creation through the
planner's gang scheduling, point-to-point and every collective
(reference coverage: tests/test/mpi/* and tests/dist/mpi/*, single-host)."""

import struct

import pytest

from faabric_amd import _core
from faabric_amd.runtime import LocalRuntime, wait_for_batch

WORLD_SIZE = 4


def ints(*vals):
    return struct.pack(f"<{len(vals)}i", *vals)


def unints(data):
    return list(struct.unpack(f"<{len(data) // 4}i", data))


def mpi_payload(msg):
    world_id, rank, size = _core.mpi_init()
    assert size == WORLD_SIZE

    # barrier
    _core.mpi_barrier(rank)

    # send/recv ring: rank r sends (r+100) to r+1
    if rank < size - 1:
        _core.mpi_send_bytes(rank, rank + 1, ints(rank + 100))
    if rank > 0:
        got = unints(_core.mpi_recv_bytes(rank - 1, rank, 4))
        assert got == [rank - 1 + 100], f"ring recv wrong: {got}"

    # allreduce SUM of [rank, 2*rank]
    out = unints(
        _core.mpi_allreduce_bytes(
            rank, ints(rank, 2 * rank), _core.MpiDataType.INT32, _core.MpiOp.SUM
        )
    )
    expect_sum = sum(range(size))
    assert out == [expect_sum, 2 * expect_sum], f"allreduce wrong: {out}"

    # reduce MAX to root 0
    out = unints(
        _core.mpi_reduce_bytes(
            rank, 0, ints(rank * 7), _core.MpiDataType.INT32, _core.MpiOp.MAX
        )
    )
    if rank == 0:
        assert out == [7 * (size - 1)]

    # broadcast from root 2
    data = ints(999, 888) if rank == 2 else b""
    got = _core.mpi_bcast_bytes(2, rank, data, 8)
    assert unints(got) == [999, 888]

    # scatter from root 0
    send = ints(*range(size)) if rank == 0 else b""
    got = unints(_core.mpi_scatter_bytes(0, rank, send, 4))
    assert got == [rank]

    # gather to root 1
    got = _core.mpi_gather_bytes(rank, 1, ints(rank * 3), size)
    if rank == 1:
        assert unints(got) == [3 * i for i in range(size)]

    # allgather
    got = unints(_core.mpi_allgather_bytes(rank, ints(rank + 50), size))
    assert got == [50 + i for i in range(size)]

    # alltoall: rank r sends value r*10+j to rank j
    send = ints(*[rank * 10 + j for j in range(size)])
    got = unints(_core.mpi_alltoall_bytes(rank, send, size))
    assert got == [i * 10 + rank for i in range(size)]

    # scan (inclusive prefix sum)
    got = unints(
        _core.mpi_scan_bytes(
            rank, ints(rank + 1), _core.MpiDataType.INT32, _core.MpiOp.SUM
        )
    )
    assert got == [sum(range(1, rank + 2))]

    # reduce_scatter: each rank contributes [r*size+j for j], gets the
    # SUM over ranks of its own slice
    rs_send = ints(*[rank * size + j for j in range(size)])
    got = unints(
        _core.mpi_reducescatter_bytes(
            rank, rs_send, _core.MpiDataType.INT32, 1, _core.MpiOp.SUM
        )
    )
    assert got == [sum(r * size + rank for r in range(size))], got

    # sendrecv around the ring
    got = _core.mpi_sendrecv_bytes(
        rank, (rank + 1) % size, (rank - 1 + size) % size, ints(rank)
    )
    assert unints(got) == [(rank - 1 + size) % size]

    _core.mpi_barrier(rank)
    msg.output_data = f"rank {rank} ok"
    return 0


@pytest.fixture(scope="module")
def runtime():
    rt = LocalRuntime(slots=WORLD_SIZE)
    rt.start_planner(with_snapshot_server=False)
    rt.start_worker()
    _core.register_function("mpi", "alltests", mpi_payload)
    yield rt
    rt.stop()


def submit_mpi_batch(user, func, world_size, timeout_ms=60_000):
    ber = _core.batch_exec_factory(user, func, 1)
    msgs = ber.messages
    msgs[0].is_mpi = True
    msgs[0].mpi_world_size = world_size
    ber.messages = msgs
    decision = _core.call_functions(ber)
    assert decision.app_id == ber.app_id, f"schedule failed: {decision.app_id}"
    return wait_for_batch(ber.app_id, world_size, timeout_ms)


def _dtype_matrix_payload(msg):
    """Allreduce every datatype x op against a numpy model (reference:
    op_reduce executes max/min/sum over int, uint64, double, long long —
    src/mpi/MpiWorld.cpp:1266-1389; PROD comes free on this rebuild)."""
    import struct as st

    import numpy as np

    from faabric_amd import _core as c

    world_id, rank, size = c.mpi_init()
    c.mpi_barrier(rank)

    cases = [
        (c.MpiDataType.INT32, "<4i", np.int32,
         [rank + 1, -rank, rank * 7, 2]),
        (c.MpiDataType.INT64, "<4q", np.int64,
         [2**40 + rank, -(rank + 1), rank, 3]),
        (c.MpiDataType.UINT64, "<4Q", np.uint64,
         [2**50 + rank, rank + 1, 5, rank * 11]),
        (c.MpiDataType.DOUBLE, "<4d", np.float64,
         [rank + 0.5, -1.25 * rank, 3.0, rank * 0.125]),
        (c.MpiDataType.FLOAT, "<4f", np.float32,
         [rank + 0.5, 2.0, -float(rank), 1.5]),
    ]
    ops = [(c.MpiOp.SUM, np.add), (c.MpiOp.MAX, np.maximum),
           (c.MpiOp.MIN, np.minimum), (c.MpiOp.PROD, np.multiply)]

    for dtype, fmt, npt, mine in cases:
        for op, npop in ops:
            got = c.mpi_allreduce_bytes(
                rank, st.pack(fmt, *[npt(v).item() for v in mine]),
                dtype, op)
            got_vals = np.array(st.unpack(fmt, got), dtype=npt)
            # Model: fold every rank's contribution
            acc = None
            for r in range(size):
                contrib = np.array(
                    [npt(v).item() for v in [
                        {0: r + 1, 1: -r, 2: r * 7, 3: 2}[i]
                        if dtype == c.MpiDataType.INT32 else
                        {0: 2**40 + r, 1: -(r + 1), 2: r, 3: 3}[i]
                        if dtype == c.MpiDataType.INT64 else
                        {0: 2**50 + r, 1: r + 1, 2: 5, 3: r * 11}[i]
                        if dtype == c.MpiDataType.UINT64 else
                        {0: r + 0.5, 1: -1.25 * r, 2: 3.0,
                         3: r * 0.125}[i]
                        if dtype == c.MpiDataType.DOUBLE else
                        {0: r + 0.5, 1: 2.0, 2: -float(r), 3: 1.5}[i]
                        for i in range(4)
                    ]], dtype=npt)
                acc = contrib if acc is None else npop(acc, contrib)
            if not np.allclose(got_vals.astype(np.float64),
                               acc.astype(np.float64)):
                return 10 + int(dtype)
    c.mpi_barrier(rank)
    return 0


def test_mpi_datatype_op_matrix(runtime):
    _core.register_function("mpi", "dtypematrix", _dtype_matrix_payload)
    results = submit_mpi_batch("mpi", "dtypematrix", WORLD_SIZE)
    assert all(r.return_value == 0 for r in results), [
        (r.mpi_rank, r.return_value) for r in results
    ]


def test_mpi_world_all_collectives(runtime):
    results = submit_mpi_batch("mpi", "alltests", WORLD_SIZE)
    assert len(results) == WORLD_SIZE
    for r in results:
        assert r.return_value == 0, r.output_data
    outputs = sorted(r.output_data for r in results)
    assert outputs == sorted(f"rank {i} ok" for i in range(WORLD_SIZE))


def test_mpi_world_reuse(runtime):
    # A second world after the first finished: ids and state must not leak
    results = submit_mpi_batch("mpi", "alltests", WORLD_SIZE)
    assert all(r.return_value == 0 for r in results)


def test_cpp_mpi_examples(runtime):
    """C++ MPI programs written against the MPI_* shim (Appendix A
    surface) run as native functions."""
    _core.register_mpi_example_functions()
    for func in ("allreduce", "ring", "async", "allreduce-bench",
                 "allreduce-small-bench", "vcollectives", "cartesian",
                 "inplace"):
        results = submit_mpi_batch("mpi-cpp", func, WORLD_SIZE)
        assert len(results) == WORLD_SIZE
        for r in results:
            assert r.return_value == 0, (func, r.output_data)


def test_mpi_exec_graph_details(runtime):
    """Per-rank MPI message counters ride the exec graph (reference:
    tests/test/mpi/test_mpi_exec_graph.cpp)."""

    def traced(msg):
        world_id, rank, size = _core.mpi_init()
        _core.mpi_send_bytes(rank, (rank + 1) % size, b"x")
        _core.mpi_recv_bytes((rank - 1 + size) % size, rank, 1)
        return 0

    _core.register_function("mpi", "traced", traced)
    ber = _core.batch_exec_factory("mpi", "traced", 1)
    msgs = ber.messages
    msgs[0].is_mpi = True
    msgs[0].mpi_world_size = WORLD_SIZE
    msgs[0].record_exec_graph = True
    ber.messages = msgs
    _core.call_functions(ber)
    results = wait_for_batch(ber.app_id, WORLD_SIZE, 60_000)
    rank0 = [r for r in results if r.mpi_rank == 0][0]
    details = rank0.int_exec_graph_details
    assert any(k.startswith("mpi-msgcount-torank-") for k in details), details


def test_concurrent_mpi_worlds(runtime):
    """Two MPI worlds run concurrently in one worker without channel or
    registry interference (reference: tests/test/mpi multi-world cases)."""
    bers = []
    for _ in range(2):
        ber = _core.batch_exec_factory("mpi-cpp", "vcollectives", 1)
        msgs = ber.messages
        msgs[0].is_mpi = True
        msgs[0].mpi_world_size = 2
        ber.messages = msgs
        d = _core.call_functions(ber)
        assert d.app_id == ber.app_id, f"schedule failed: {d.app_id}"
        bers.append(ber)
    for ber in bers:
        results = wait_for_batch(ber.app_id, 2, 60_000)
        assert all(r.return_value == 0 for r in results), [
            (r.mpi_rank, r.return_value, r.output_data) for r in results
        ]
