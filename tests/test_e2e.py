"""End-to-end single-host tests: planner + worker in one process, batches
through the full RPC path (the reference's examples/check.cpp +
tests/dist/scheduler/test_funcs.cpp coverage, single-host part)."""

import threading
import time

import pytest

import faabric_amd as fa
from faabric_amd import _core
from faabric_amd.runtime import LocalRuntime, execute_batch

SLOTS = 8


@pytest.fixture(scope="module")
def runtime():
    rt = LocalRuntime(slots=SLOTS)
    # Colocated planner+worker share the process snapshot registry, so the
    # planner does not need its own snapshot server (port clash otherwise)
    rt.start_planner(with_snapshot_server=False)
    rt.start_worker()
    _core.register_native_echo("demo", "echo")
    _core.register_native_noop("demo", "noop")
    yield rt
    rt.stop()


def test_single_function(runtime):
    results = execute_batch("demo", "echo", 1, input_data=b"hello world")
    assert len(results) == 1
    assert results[0].return_value == 0
    assert results[0].output_data == "hello world"
    assert results[0].executed_host == runtime.identity


def test_batch_of_functions(runtime):
    results = execute_batch("demo", "noop", SLOTS)
    assert len(results) == SLOTS
    assert all(r.return_value == 0 for r in results)
    assert sorted(r.app_idx for r in results) == list(range(SLOTS))


def test_python_function(runtime):
    calls = []

    def handler(msg):
        calls.append(msg.input_data)
        msg.output_data = "py:" + msg.input_data.decode()
        return 0

    _core.register_function("demo", "pyfunc", handler)
    results = execute_batch("demo", "pyfunc", 2, input_data=b"x")
    assert len(results) == 2
    assert all(r.output_data == "py:x" for r in results)
    assert calls == [b"x", b"x"]


def test_failing_function(runtime):
    def bad(msg):
        raise ValueError("boom")

    _core.register_function("demo", "bad", bad)
    results = execute_batch("demo", "bad", 1)
    assert results[0].return_value == 1
    assert "boom" in results[0].output_data


def test_processes_batch_type(runtime):
    """PROCESSES batches take the same path as FUNCTIONS (reference:
    proto BatchExecuteType PROCESSES, src/proto/faabric.proto:35)."""
    results = execute_batch("demo", "echo", 2, input_data=b"proc",
                            batch_type=_core.BatchExecuteType.PROCESSES)
    assert len(results) == 2
    assert all(r.output_data == "proc" for r in results)


def test_not_enough_slots(runtime):
    ber = _core.batch_exec_factory("demo", "noop", SLOTS + 1)
    decision = _core.call_functions(ber)
    assert decision.app_id == _core.NOT_ENOUGH_SLOTS()


def test_sequential_batches_reuse_executors(runtime):
    results = execute_batch("demo", "noop", 4)
    assert len(results) == 4
    count_after_first = _core.get_executor_count()
    for _ in range(3):
        results = execute_batch("demo", "noop", 4)
        assert len(results) == 4
    # Warm executors are reused, not recreated per batch
    assert _core.get_executor_count() == count_after_first


def test_get_message_result(runtime):
    ber = _core.batch_exec_factory("demo", "echo", 1)
    msgs = ber.messages
    msgs[0].input_data = b"direct"
    ber.messages = msgs
    msg_id = ber.messages[0].id
    _core.call_functions(ber)
    result = _core.get_message_result(ber.app_id, msg_id, 10_000)
    assert result.output_data == "direct"


def test_available_hosts(runtime):
    hosts = _core.get_available_hosts()
    assert len(hosts) == 1
    assert hosts[0].ip == runtime.identity
    assert hosts[0].slots == SLOTS


def test_ptp_local_messaging(runtime):
    decision = _core.SchedulingDecision()
    decision.app_id = 424242
    decision.group_id = 424243
    decision.hosts = [runtime.identity, runtime.identity]
    decision.message_ids = [1, 2]
    decision.app_idxs = [0, 1]
    decision.group_idxs = [0, 1]
    decision.mpi_ports = [0, 0]
    decision.n_functions = 2
    _core.ptp_setup_local_mappings(decision)

    _core.ptp_send(424242, 424243, 0, 1, b"ping", True)
    _core.ptp_send(424242, 424243, 0, 1, b"pong", True)
    assert _core.ptp_recv(424243, 0, 1, True) == b"ping"
    assert _core.ptp_recv(424243, 0, 1, True) == b"pong"


def test_ptp_group_barrier_and_lock(runtime):
    decision = _core.SchedulingDecision()
    decision.app_id = 555000
    decision.group_id = 555001
    decision.hosts = [runtime.identity] * 3
    decision.message_ids = [1, 2, 3]
    decision.app_idxs = [0, 1, 2]
    decision.group_idxs = [0, 1, 2]
    decision.mpi_ports = [0, 0, 0]
    decision.n_functions = 3
    _core.ptp_setup_local_mappings(decision)

    passed = []
    counter = {"v": 0}

    def member(idx):
        _core.ptp_group_barrier(555001, idx)
        # notify: non-main members signal, main waits for all
        _core.ptp_group_notify(555001, idx)
        _core.ptp_group_lock(555001, idx)
        v = counter["v"]
        counter["v"] = v + 1
        _core.ptp_group_unlock(555001, idx)
        _core.ptp_group_barrier(555001, idx)
        passed.append(idx)

    threads = [threading.Thread(target=member, args=(i,)) for i in range(3)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=15)
    assert sorted(passed) == [0, 1, 2]
    assert counter["v"] == 3


def test_ptp_group_recursive_lock(runtime):
    """Recursive group lock: re-acquisition by the holder nests; a
    waiter gets it only after full unwind (reference:
    PointToPointBroker LOCK_GROUP_RECURSIVE handling)."""
    decision = _core.SchedulingDecision()
    decision.app_id = 556000
    decision.group_id = 556001
    decision.hosts = [runtime.identity] * 2
    decision.message_ids = [1, 2]
    decision.app_idxs = [0, 1]
    decision.group_idxs = [0, 1]
    decision.mpi_ports = [0, 0]
    decision.n_functions = 2
    _core.ptp_setup_local_mappings(decision)

    order = []
    held = threading.Event()
    waiting = threading.Event()

    def holder():
        _core.ptp_group_lock(556001, 0, True)
        _core.ptp_group_lock(556001, 0, True)  # nested re-acquire
        order.append("held")
        held.set()
        waiting.wait(10)  # let the waiter block on the lock
        time.sleep(0.2)
        _core.ptp_group_unlock(556001, 0, True)
        order.append("partial-unwind")
        time.sleep(0.2)
        _core.ptp_group_unlock(556001, 0, True)

    def waiter():
        held.wait(10)
        waiting.set()
        _core.ptp_group_lock(556001, 1, True)
        order.append("waiter-got-it")
        _core.ptp_group_unlock(556001, 1, True)

    t1 = threading.Thread(target=holder)
    t2 = threading.Thread(target=waiter)
    t1.start()
    t2.start()
    t1.join(timeout=15)
    t2.join(timeout=15)
    assert order == ["held", "partial-unwind", "waiter-got-it"]


def test_state_local(runtime):
    kv = _core.state_get_kv("demo", "statekey", 128)
    assert kv.is_master
    kv.set(b"a" * 128)
    assert kv.get() == b"a" * 128
    kv.set_chunk(10, b"BBBB")
    data = kv.get()
    assert data[10:14] == b"BBBB"
    kv.append(b"v1")
    kv.append(b"v2")
    assert kv.get_appended(2) == [b"v1", b"v2"]


def test_executor_reaper(runtime):
    """Idle warm executors are reaped after the bound timeout (reference:
    Scheduler::reapStaleExecutors, src/scheduler/Scheduler.cpp:166)."""
    import time as _t

    from faabric_amd import _core

    from faabric_amd.runtime import wait_for_batch

    ber = _core.batch_exec_factory("demo", "echo", 4)
    _core.call_functions(ber)
    wait_for_batch(ber.app_id, 4, 30_000)
    assert _core.get_executor_count() >= 4

    # Not yet stale
    assert _core.reap_stale_executors() == 0

    # The reaper reads the live config; shrink the bound and retry
    _core.set_bound_timeout(50)
    try:
        _t.sleep(0.3)
        reaped = _core.reap_stale_executors()
        assert reaped >= 4, reaped
        assert _core.get_executor_count() == 0
    finally:
        _core.set_bound_timeout(30_000)


def test_wait_batch_done_event_path(runtime):
    """Event-driven batch completion: the planner pushes BATCH_DONE to
    the registered host's function-call server when an app's last result
    lands (replaces status polling; see PlannerCalls::WaitBatchDone)."""
    import time

    ber = _core.batch_exec_factory("demo", "noop", SLOTS)
    _core.call_functions(ber)
    t0 = time.perf_counter()
    assert _core.wait_batch_done(ber.app_id, 10_000)
    elapsed = time.perf_counter() - t0
    # An event (or the immediate already-done answer) must beat the
    # 500 ms fallback poll by a wide margin
    assert elapsed < 0.45, f"wait_batch_done fell back to polling: {elapsed}"
    status = _core.get_batch_results(ber.app_id)
    assert status.finished

    # Already-completed app: returns immediately with True
    t0 = time.perf_counter()
    assert _core.wait_batch_done(ber.app_id, 10_000)
    assert time.perf_counter() - t0 < 0.2

    # Unknown app: the planner answers "not in flight" (treated as done,
    # caller verifies via status — which reports unknown)
    assert _core.wait_batch_done(987654321, 2_000)
    unknown = _core.get_batch_results(987654321)
    assert unknown.expected_num_messages == -1


def test_runtime_state_bounded_after_batches(runtime):
    """Completed apps must not accumulate runtime state: PTP group
    mappings are retired by the planner's GROUP_CLEAR broadcast and
    appResults is bounded by FAABRIC_RESULT_TTL_MS/FAABRIC_MAX_DONE_APPS
    GC (a long-running planner previously leaked both)."""
    before = _core._debug_runtime_sizes()
    for _ in range(20):
        ber = _core.batch_exec_factory("demo", "noop", SLOTS)
        _core.call_functions(ber)
        assert _core.wait_batch_done(ber.app_id, 10_000)
    sizes = _core._debug_runtime_sizes()
    assert sizes["planner_in_flight"] == 0
    # Other tests in this module leave long-lived groups behind; the 20
    # completed batches here must not add to the mapping/seq tables
    assert sizes["broker_mappings"] <= before["broker_mappings"], (
        before, sizes)
    assert sizes["broker_send_seqs"] <= before["broker_send_seqs"], (
        before, sizes)
    # bounded, not necessarily zero (completed results kept for late fetch)
    assert sizes["planner_done_apps"] <= 4096


def test_concurrent_apps_interleaved(runtime):
    """Many apps in flight at once: concurrent submitters + event waits
    must not deadlock the dispatch pool (fan-outs serialize per batch)
    or cross-wire results between apps."""
    import threading

    errors = []

    def submitter(tid):
        try:
            for r in range(10):
                # 4 submitters x 2 messages = SLOTS: fits capacity, so
                # NOT_ENOUGH_SLOTS cannot occur
                ber = _core.batch_exec_factory("demo", "echo", 2)
                msgs = ber.messages
                for m in msgs:
                    m.input_data = f"t{tid}r{r}".encode()
                ber.messages = msgs
                d = _core.call_functions(ber)
                assert d.app_id == ber.app_id, f"not scheduled: {d.app_id}"
                assert _core.wait_batch_done(ber.app_id, 30_000)
                status = _core.get_batch_results(ber.app_id)
                assert status.finished
                outs = {m.output_data for m in status.message_results}
                assert outs == {f"t{tid}r{r}"}, outs
        except Exception as e:  # pragma: no cover
            errors.append(f"t{tid}: {e!r}")

    threads = [threading.Thread(target=submitter, args=(t,)) for t in range(4)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=120)
    assert not errors, errors
