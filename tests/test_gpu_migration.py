"""Migration of a GPU-resident function between two workers (one physical
MI355X): the HBM arena is shipped as a device snapshot and restored into
the destination executor's arena before re-entry (reference analog:
mpi_native.cpp mpiMigrationPoint + snapshot push, but with device memory
— cpp/src/migration.cpp device branches)."""

import multiprocessing as mp
import os
import sys
import time

import pytest

pytestmark = pytest.mark.gpu

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

WORKER_SLOTS = 2
OFFSETS = (5300, 5400)
PATTERN = bytes(range(256)) * 16  # 4096 bytes


def _gpu_mig_fn(msg):
    import time as _t

    from faabric_amd import _core

    if msg.input_data == b"resume":
        got = _core.executor_device_read_memory(0, len(PATTERN))
        if got != PATTERN:
            msg.output_data = "arena lost after migration"
            return 9
        msg.output_data = "resumed-with-arena"
        return 0

    _core.executor_set_device_memory_size(64 * 1024)
    _core.executor_device_write_memory(0, PATTERN)
    for i in range(14):
        _t.sleep(0.25)
        if i == 7:
            rc = _core.migration_point(b"resume")
            if rc != 0:
                return rc
    msg.output_data = "stayed"
    return 0


def _hbm_state_owner_fn(msg):
    from faabric_amd import _core

    kv = _core.state_get_kv_device("gpu", "xhost", 128 * 1024)
    kv.set(bytes([0xC3]) * (128 * 1024))
    kv.set_chunk(8192, b"OWNED")
    msg.output_data = "owner ok"
    return 0


def _hbm_state_reader_fn(msg):
    from faabric_amd import _core

    master = msg.input_data.decode()
    _core.state_set_master_host("gpu", "xhost", master)
    kv = _core.state_get_kv_device("gpu", "xhost", 128 * 1024)
    if kv.is_master:
        msg.output_data = "reader unexpectedly master"
        return 1
    chunk = kv.get_chunk(8190, 16)
    if chunk[2:7] != b"OWNED" or chunk[0] != 0xC3:
        msg.output_data = f"bad chunk: {chunk!r}"
        return 2
    # Push a write back to the owner
    kv.set_chunk(4096, b"FROMB")
    kv.push_partial()
    msg.output_data = "reader ok"
    return 0


def _worker_main(port_offset, stop_event, ready_event):
    sys.path.insert(0, REPO_ROOT)
    from faabric_amd import _core
    from faabric_amd.runtime import LocalRuntime

    _core.set_log_level("error")
    rt = LocalRuntime(port_offset=port_offset, planner_port_offset=5200,
                      slots=WORKER_SLOTS)
    rt.start_worker()
    _core.register_native_sleep("gmig", "blocker", 600)
    _core.register_function("gmig", "worker", _gpu_mig_fn)
    _core.register_function("gmig", "stateowner", _hbm_state_owner_fn)
    _core.register_function("gmig", "statereader", _hbm_state_reader_fn)
    ready_event.set()
    stop_event.wait(180)
    rt.stop()


@pytest.fixture(scope="module")
def cluster():
    from faabric_amd import _core
    from faabric_amd.runtime import LocalRuntime

    rt = LocalRuntime(port_offset=5200, planner_port_offset=5200)
    rt.start_planner()
    ctx = mp.get_context("spawn")
    stop = ctx.Event()
    procs = []
    for off in OFFSETS:
        ready = ctx.Event()
        p = ctx.Process(target=_worker_main, args=(off, stop, ready))
        p.start()
        procs.append(p)
        assert ready.wait(120)
    deadline = time.monotonic() + 20
    while time.monotonic() < deadline:
        if len(_core.get_available_hosts()) == 2:
            break
        time.sleep(0.05)
    yield rt
    stop.set()
    for p in procs:
        p.join(timeout=30)
        if p.is_alive():
            p.terminate()
    rt.stop()


def test_gpu_arena_migrates(cluster):
    from faabric_amd import _core
    from faabric_amd.runtime import wait_for_batch

    w1, w2 = [f"127.0.0.1@{o}" for o in OFFSETS]

    # Occupy one slot on each host so the 2-message app splits 1+1, then
    # bin-pack's DIST_CHANGE consolidates it and migrates one function
    n = 2
    blockers = _core.batch_exec_factory("gmig", "blocker", n)
    d = _core.SchedulingDecision()
    d.app_id = blockers.app_id
    for i, h in enumerate([w1, w2]):
        d.hosts = d.hosts + [h]
        d.message_ids = d.message_ids + [0]
        d.app_idxs = d.app_idxs + [i]
        d.group_idxs = d.group_idxs + [i]
        d.mpi_ports = d.mpi_ports + [0]
    d.n_functions = n
    _core.preload_scheduling_decision(blockers.app_id, d)
    _core.call_functions(blockers)

    app = _core.batch_exec_factory("gmig", "worker", n)
    msgs = app.messages
    for i, m in enumerate(msgs):
        m.group_idx = i
        m.group_size = n
    app.messages = msgs
    d = _core.SchedulingDecision()
    d.app_id = app.app_id
    for i, h in enumerate([w1, w2]):
        d.hosts = d.hosts + [h]
        d.message_ids = d.message_ids + [0]
        d.app_idxs = d.app_idxs + [i]
        d.group_idxs = d.group_idxs + [i]
        d.mpi_ports = d.mpi_ports + [0]
    d.n_functions = n
    _core.preload_scheduling_decision(app.app_id, d)
    _core.call_functions(app)

    results = wait_for_batch(app.app_id, n, timeout_ms=90_000)
    assert all(r.return_value == 0 for r in results), [
        (r.return_value, r.output_data) for r in results
    ]
    outputs = sorted(r.output_data for r in results)
    assert outputs == ["resumed-with-arena", "stayed"], outputs
    assert len({r.executed_host for r in results}) == 1

    wait_for_batch(blockers.app_id, n, timeout_ms=30_000)


def _submit_one(user, func, host, input_data=b""):
    from faabric_amd import _core

    ber = _core.batch_exec_factory(user, func, 1)
    msgs = ber.messages
    msgs[0].input_data = input_data
    ber.messages = msgs
    d = _core.SchedulingDecision()
    d.app_id = ber.app_id
    d.hosts = [host]
    d.message_ids = [0]
    d.app_idxs = [0]
    d.group_idxs = [0]
    d.mpi_ports = [0]
    d.n_functions = 1
    _core.preload_scheduling_decision(ber.app_id, d)
    _core.call_functions(ber)
    return ber


def test_hbm_state_cross_worker(cluster):
    """Distributed state with the value resident in the owner worker's
    HBM: a second worker lazily pulls chunks over the state RPC into its
    own GPU and pushes partial writes back (reference: two-backend
    StateKeyValue pull/push, src/state/StateKeyValue.cpp — HBM-resident
    re-design)."""
    from faabric_amd import _core
    from faabric_amd.runtime import wait_for_batch

    w1, w2 = [f"127.0.0.1@{o}" for o in OFFSETS]
    owner = _submit_one("gmig", "stateowner", w1)
    r = wait_for_batch(owner.app_id, 1, timeout_ms=60_000)
    assert r[0].return_value == 0, r[0].output_data

    reader = _submit_one("gmig", "statereader", w2, w1.encode())
    r = wait_for_batch(reader.app_id, 1, timeout_ms=60_000)
    assert r[0].return_value == 0, r[0].output_data
    assert r[0].output_data == "reader ok"
