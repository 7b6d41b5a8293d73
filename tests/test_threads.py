"""THREADS fork-join over a shared memory snapshot with typed merge
regions (the reference's OpenMP-style path: SURVEY §3.4,
tests/test/executor/test_executor.cpp threads cases)."""

import struct

import pytest

from faabric_amd import _core
from faabric_amd.runtime import LocalRuntime, wait_for_batch

SLOTS = 8


def thread_adder(msg):
    # Each thread increments its own int slot by (group_idx), and scribbles
    # into the bytewise area at a thread-specific offset
    idx = msg.group_idx
    raw = _core.executor_read_memory(0, 64)
    vals = list(struct.unpack("<16i", raw))
    _core.executor_write_memory(
        (idx - 1) * 4, struct.pack("<i", vals[idx - 1] + idx)
    )
    _core.executor_write_memory(4096 + idx, bytes([0x40 + idx]))
    return 0


def fork_parent(msg):
    _core.executor_set_memory_size(8192)
    _core.executor_write_memory(0, struct.pack("<16i", *([10] * 16)))
    _core.executor_write_memory(4096, bytes(16))

    results = _core.execute_threads(
        "threads",
        "adder",
        3,
        merge_regions=[
            (0, 64, int(_core.SnapshotDataType.Int.value),
             int(_core.SnapshotMergeOperation.Sum.value)),
        ],
    )
    if len(results) != 3 or any(rv != 0 for _, rv in results):
        msg.output_data = f"thread failures: {results}"
        return 1

    vals = struct.unpack("<16i", _core.executor_read_memory(0, 64))
    if vals[:3] != (11, 12, 13) or any(v != 10 for v in vals[3:]):
        msg.output_data = f"bad merged ints: {vals}"
        return 2

    tail = _core.executor_read_memory(4096, 16)
    if tail[1:4] != bytes([0x41, 0x42, 0x43]):
        msg.output_data = f"bad bytewise merge: {tail!r}"
        return 3
    msg.output_data = "fork-join ok"
    return 0


@pytest.fixture(scope="module")
def runtime():
    rt = LocalRuntime(slots=SLOTS, port_offset=400, planner_port_offset=400)
    rt.start_planner(with_snapshot_server=False)
    rt.start_worker()
    _core.register_function("threads", "adder", thread_adder)
    _core.register_function("threads", "parent", fork_parent)
    yield rt
    rt.stop()


def test_threads_fork_join_sum_merge(runtime):
    ber = _core.batch_exec_factory("threads", "parent", 1)
    decision = _core.call_functions(ber)
    assert decision.app_id == ber.app_id
    results = wait_for_batch(ber.app_id, 1, timeout_ms=30_000)
    assert results[0].return_value == 0, results[0].output_data
    assert results[0].output_data == "fork-join ok"


def test_threads_repeat_fork(runtime):
    # Repeated fork-join from the same executor must not leak snapshots
    for _ in range(2):
        ber = _core.batch_exec_factory("threads", "parent", 1)
        _core.call_functions(ber)
        results = wait_for_batch(ber.app_id, 1, timeout_ms=30_000)
        assert results[0].return_value == 0, results[0].output_data


def elastic_thread(msg):
    idx = msg.group_idx
    raw = _core.executor_read_memory(0, 4)
    _core.executor_write_memory(
        64 + idx * 4, struct.pack("<i", idx * 100)
    )
    return 0


def elastic_parent(msg):
    _core.executor_set_memory_size(4096)
    _core.executor_write_memory(0, struct.pack("<i", 7))
    results = _core.execute_threads(
        "threads", "elastic_thread", 1, elastic=True
    )
    if any(rv != 0 for _, rv in results):
        msg.output_data = f"failures: {results}"
        return 1
    msg.output_data = f"nthreads={len(results)}"
    return 0


def test_elastic_scale_up(runtime):
    """elasticScaleHint grows a 1-thread fork to every free slot on the
    main host (reference: src/planner/Planner.cpp:832-891)."""
    _core.register_function("threads", "elastic_thread", elastic_thread)
    _core.register_function("threads", "elastic_parent", elastic_parent)
    ber = _core.batch_exec_factory("threads", "elastic_parent", 1)
    _core.call_functions(ber)
    results = wait_for_batch(ber.app_id, 2, timeout_ms=30_000)
    parent = [r for r in results if r.output_data.startswith("nthreads")][0]
    assert parent.return_value == 0
    n = int(parent.output_data.split("=")[1])
    # Parent occupies 1 of SLOTS slots; the fork should expand beyond the
    # single requested thread
    assert n > 1, parent.output_data
