"""Cross-process device payloads over HIP IPC (same-node workers).

The deployment shape is one worker process per GPU; device payloads
between two such workers must ride hipIpc arenas + peer copies, never
D2H -> TCP -> H2D. These tests run two real processes against one GPU
(both on device 0 — the IPC path is identical to the peer-GPU case up
to the link the copy crosses).

Reference analog: the PTP broker local fast path
(/root/reference/src/transport/PointToPointBroker.cpp:637-764).
"""

import multiprocessing as mp
import os
import sys

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

torch = pytest.importorskip("torch")

requires_gpu = pytest.mark.gpu

APP_ID = 881_000
GROUP_ID = 881_001
HOST_A = "127.0.0.1@5100"
HOST_B = "127.0.0.1@5200"


def _mk_decision(core):
    decision = core.SchedulingDecision()
    decision.app_id = APP_ID
    decision.group_id = GROUP_ID
    decision.hosts = [HOST_A, HOST_B]
    decision.message_ids = [1, 2]
    decision.app_idxs = [0, 1]
    decision.group_idxs = [0, 1]
    decision.mpi_ports = [0, 0]
    decision.n_functions = 2
    return decision


def _receiver_proc(ready, done, result_q, n_msgs, msg_elems, arena_mb):
    sys.path.insert(0, REPO_ROOT)
    os.environ["FAABRIC_IPC_ARENA_MB"] = str(arena_mb)
    import torch as t

    from faabric_amd import _core

    try:
        _core.set_log_level("error")
        _core.set_port_offset(5200)
        _core.set_endpoint_host(HOST_B)
        server = _core.PointToPointServerHandle()
        server.start()
        _core.ptp_setup_local_mappings(_mk_decision(_core))
        ready.set()

        dst = t.zeros(msg_elems, dtype=t.float32, device="cuda")
        ok = True
        detail = ""
        for i in range(n_msgs):
            got = _core.ptp_recv_device(
                GROUP_ID, 0, 1, dst.data_ptr(), msg_elems * 4, True, 60_000
            )
            t.cuda.synchronize()
            expect = t.full(
                (msg_elems,), float(i + 1), dtype=t.float32, device="cuda"
            )
            if got != msg_elems * 4 or not t.equal(dst, expect):
                ok = False
                detail = f"msg {i}: got {got} bytes, first={dst[0].item()}"
                break
        result_q.put((ok, detail))
        done.wait(60)
        server.stop()
    except Exception as e:  # surface the failure to the parent
        result_q.put((False, repr(e)))
        ready.set()


def _sender_proc(ready, done, result_q, n_msgs, msg_elems, arena_mb):
    sys.path.insert(0, REPO_ROOT)
    os.environ["FAABRIC_IPC_ARENA_MB"] = str(arena_mb)
    import torch as t

    from faabric_amd import _core

    try:
        _core.set_log_level("error")
        _core.set_port_offset(5100)
        _core.set_endpoint_host(HOST_A)
        server = _core.PointToPointServerHandle()
        server.start()
        _core.ptp_setup_local_mappings(_mk_decision(_core))
        assert ready.wait(60), "receiver did not come up"

        if not _core.ipc_available(HOST_B):
            result_q.put((False, "ipc arena to receiver unavailable"))
            return

        src = t.zeros(msg_elems, dtype=t.float32, device="cuda")
        for i in range(n_msgs):
            src.fill_(float(i + 1))
            t.cuda.synchronize()
            _core.ptp_send_device(
                APP_ID, GROUP_ID, 0, 1, src.data_ptr(), msg_elems * 4, True
            )
        segs, nbytes = _core.ipc_shipped()
        ok = segs == n_msgs and nbytes == n_msgs * msg_elems * 4
        result_q.put((ok, f"shipped segs={segs} bytes={nbytes}"))
        done.wait(60)
        server.stop()
    except Exception as e:
        result_q.put((False, repr(e)))


def _run_pair(n_msgs, msg_elems, arena_mb):
    ctx = mp.get_context("spawn")
    ready = ctx.Event()
    done = ctx.Event()
    recv_q = ctx.Queue()
    send_q = ctx.Queue()
    recv = ctx.Process(
        target=_receiver_proc,
        args=(ready, done, recv_q, n_msgs, msg_elems, arena_mb),
    )
    send = ctx.Process(
        target=_sender_proc,
        args=(ready, done, send_q, n_msgs, msg_elems, arena_mb),
    )
    recv.start()
    send.start()
    try:
        recv_ok, recv_detail = recv_q.get(timeout=120)
        send_ok, send_detail = send_q.get(timeout=120)
    finally:
        done.set()
        send.join(timeout=30)
        recv.join(timeout=30)
        for p in (send, recv):
            if p.is_alive():
                p.terminate()
    assert recv_ok, f"receiver: {recv_detail}"
    assert send_ok, f"sender: {send_detail}"


@requires_gpu
def test_cross_process_device_ptp_over_ipc():
    """Ordered device payloads between two worker processes: the sender's
    ipc_shipped counter proves no message fell back to the D2H path."""
    _run_pair(n_msgs=4, msg_elems=1 << 16, arena_mb=32)


@requires_gpu
def test_ipc_ring_recycles_under_pressure():
    """Total shipped bytes (24 x 1 MiB) far exceed a 4 MiB arena: acks
    must recycle segments or the sender would stall and time out."""
    _run_pair(n_msgs=24, msg_elems=(1 << 20) // 4, arena_mb=4)


# ---------------------------------------------------------------------------
# State KV: device values pulled/pushed between same-node workers over IPC
# ---------------------------------------------------------------------------

KV_BYTES = 8 << 20  # spans multiple IPC chunks at a 4 MiB arena


def _state_master_proc(ready, done, result_q):
    sys.path.insert(0, REPO_ROOT)
    os.environ["FAABRIC_IPC_ARENA_MB"] = "4"
    import torch as t  # noqa: F401 (brings up the GPU runtime)

    from faabric_amd import _core

    try:
        _core.set_log_level("error")
        _core.set_port_offset(5100)
        _core.set_endpoint_host(HOST_A)
        ptp = _core.PointToPointServerHandle()
        ptp.start()
        state = _core.StateServerHandle()
        state.start()
        _core.state_set_master_host("ipc", "kv", HOST_A)
        kv = _core.state_get_kv_device("ipc", "kv", KV_BYTES)
        kv.set(bytes([0xA5]) * KV_BYTES)
        ready.set()
        assert done.wait(120), "worker never finished"
        # Worker pushed a modified middle range; verify it landed in HBM
        mid = kv.get_chunk(KV_BYTES // 2, 1 << 20)
        ok = mid == bytes([0x3C]) * (1 << 20)
        head = kv.get_chunk(0, 64)
        ok = ok and head == bytes([0xA5]) * 64
        result_q.put((ok, "master verify"))
        state.stop()
        ptp.stop()
    except Exception as e:
        result_q.put((False, repr(e)))
        ready.set()


def _state_worker_proc(ready, done, result_q):
    sys.path.insert(0, REPO_ROOT)
    os.environ["FAABRIC_IPC_ARENA_MB"] = "4"
    import torch as t  # noqa: F401

    from faabric_amd import _core

    try:
        _core.set_log_level("error")
        _core.set_port_offset(5200)
        _core.set_endpoint_host(HOST_B)
        ptp = _core.PointToPointServerHandle()
        ptp.start()
        state = _core.StateServerHandle()
        state.start()
        assert ready.wait(120), "master never came up"
        _core.state_set_master_host("ipc", "kv", HOST_A)
        kv = _core.state_get_kv_device("ipc", "kv", KV_BYTES)
        kv.pull()  # whole 8 MiB through the 4 MiB arena in chunks
        got = kv.get_chunk(0, KV_BYTES)
        ok = got == bytes([0xA5]) * KV_BYTES
        segs0, _ = _core.ipc_shipped()
        # Modify the middle 1 MiB and push just that range
        kv.set_chunk(KV_BYTES // 2, bytes([0x3C]) * (1 << 20))
        result_q.put((ok, f"pull verify, pushed; local segs={segs0}"))
        done.set()
        state.stop()
        ptp.stop()
    except Exception as e:
        result_q.put((False, repr(e)))
        done.set()


@requires_gpu
def test_state_device_kv_pull_push_over_ipc():
    """A device KV owned by worker A is pulled by worker B (8 MiB value
    through a 4 MiB arena) and a dirty range pushed back, all over HIP
    IPC with RPC only carrying segment descriptors."""
    ctx = mp.get_context("spawn")
    ready = ctx.Event()
    done = ctx.Event()
    master_q = ctx.Queue()
    worker_q = ctx.Queue()
    master = ctx.Process(
        target=_state_master_proc, args=(ready, done, master_q)
    )
    worker = ctx.Process(
        target=_state_worker_proc, args=(ready, done, worker_q)
    )
    master.start()
    worker.start()
    try:
        worker_ok, worker_detail = worker_q.get(timeout=180)
        master_ok, master_detail = master_q.get(timeout=180)
    finally:
        done.set()
        worker.join(timeout=30)
        master.join(timeout=30)
        for p in (worker, master):
            if p.is_alive():
                p.terminate()
    assert worker_ok, f"worker: {worker_detail}"
    assert master_ok, f"master: {master_detail}"


# ---------------------------------------------------------------------------
# Device snapshots streamed between workers over IPC
# ---------------------------------------------------------------------------

SNAP_BYTES = 8 << 20


def _snap_receiver_proc(ready, done, result_q):
    sys.path.insert(0, REPO_ROOT)
    os.environ["FAABRIC_IPC_ARENA_MB"] = "4"
    import torch as t  # noqa: F401

    from faabric_amd import _core

    try:
        _core.set_log_level("error")
        _core.set_port_offset(5200)
        _core.set_endpoint_host(HOST_B)
        ptp = _core.PointToPointServerHandle()
        ptp.start()
        snap = _core.SnapshotServerHandle()
        snap.start()
        ready.set()
        assert done.wait(120), "sender never finished"
        ok = _core.device_snapshot_exists("ipcsnap")
        detail = "snapshot missing"
        if ok:
            head = _core.device_snapshot_read("ipcsnap", 0, 256)
            tail = _core.device_snapshot_read(
                "ipcsnap", SNAP_BYTES - 256, 256
            )
            ok = head == bytes(range(256)) and tail == bytes(
                reversed(range(256))
            )
            detail = "content mismatch" if not ok else ""
        result_q.put((ok, detail))
        snap.stop()
        ptp.stop()
    except Exception as e:
        result_q.put((False, repr(e)))
        ready.set()


def _snap_sender_proc(ready, done, result_q):
    sys.path.insert(0, REPO_ROOT)
    os.environ["FAABRIC_IPC_ARENA_MB"] = "4"
    import torch as t

    from faabric_amd import _core

    try:
        _core.set_log_level("error")
        _core.set_port_offset(5100)
        _core.set_endpoint_host(HOST_A)
        ptp = _core.PointToPointServerHandle()
        ptp.start()
        assert ready.wait(120), "receiver never came up"

        buf = t.zeros(SNAP_BYTES, dtype=t.uint8, device="cuda")
        buf[:256] = t.arange(256, dtype=t.uint8)
        buf[-256:] = t.arange(255, -1, -1, dtype=t.uint8)
        t.cuda.synchronize()
        _core.snapshot_push_device_from_ptr(
            HOST_B, "ipcsnap", buf.data_ptr(), SNAP_BYTES
        )
        segs, nbytes = _core.ipc_shipped()
        ok = segs >= 2 and nbytes == SNAP_BYTES  # chunked through arena
        result_q.put((ok, f"segs={segs} bytes={nbytes}"))
        done.set()
        ptp.stop()
    except Exception as e:
        result_q.put((False, repr(e)))
        done.set()


def _planner_store_proc(stop, ready):
    sys.path.insert(0, REPO_ROOT)
    from faabric_amd import _core

    _core.set_log_level("error")
    _core.set_port_offset(5700)
    _core.set_endpoint_host("127.0.0.1@5700")
    srv = _core.StateServerHandle()
    srv.start()
    ready.set()
    stop.wait(120)
    srv.stop()


@requires_gpu
def test_planner_backed_device_kv():
    """STATE_MODE=planner with an HBM-resident value: the worker's
    device KV pushes/pulls against the planner's global store (the
    Redis-service analog holds host bytes, like Redis did)."""
    ctx = mp.get_context("spawn")
    stop = ctx.Event()
    ready = ctx.Event()
    p = ctx.Process(target=_planner_store_proc, args=(stop, ready))
    p.start()
    try:
        assert ready.wait(60)
        from faabric_amd import _core

        prev = _core.get_endpoint_host()
        _core.set_state_mode("planner")
        _core.set_planner_host("127.0.0.1@5700")
        _core.set_endpoint_host("127.0.0.1@5800")
        _core.state_clear_all()
        kv = _core.state_get_kv_device("gk", "devval", 256 * 1024)
        assert kv.on_device and not kv.is_master
        kv.set(b"\x77" * (256 * 1024))  # lands in HBM AND pushes
        # Drop the local replica; re-pull from the planner store into HBM
        _core.state_clear_all()
        kv2 = _core.state_get_kv_device("gk", "devval", 256 * 1024)
        kv2.pull()
        assert kv2.get_chunk(0, 64) == b"\x77" * 64
        assert kv2.get_chunk(256 * 1024 - 64, 64) == b"\x77" * 64
        # Partial push of a modified HBM range
        kv2.set_chunk(1000, b"\x88" * 500)
        kv2.push_partial()
        _core.state_clear_all()
        kv3 = _core.state_get_kv_device("gk", "devval", 256 * 1024)
        assert kv3.get_chunk(990, 20) == (
            b"\x77" * 10 + b"\x88" * 10
        )
    finally:
        stop.set()
        p.join(timeout=30)
        if p.is_alive():
            p.terminate()
        _core.state_clear_all()
        _core.set_state_mode("inmemory")
        _core.set_planner_host("127.0.0.1")
        _core.set_endpoint_host(prev)


@requires_gpu
def test_device_snapshot_streams_over_ipc():
    """An 8 MiB HBM snapshot ships worker-to-worker through a 4 MiB IPC
    arena in acked chunks; the receiver's registry holds the bytes."""
    ctx = mp.get_context("spawn")
    ready = ctx.Event()
    done = ctx.Event()
    recv_q = ctx.Queue()
    send_q = ctx.Queue()
    recv = ctx.Process(target=_snap_receiver_proc, args=(ready, done, recv_q))
    send = ctx.Process(target=_snap_sender_proc, args=(ready, done, send_q))
    recv.start()
    send.start()
    try:
        send_ok, send_detail = send_q.get(timeout=180)
        recv_ok, recv_detail = recv_q.get(timeout=180)
    finally:
        done.set()
        send.join(timeout=30)
        recv.join(timeout=30)
        for p in (send, recv):
            if p.is_alive():
                p.terminate()
    assert send_ok, f"sender: {send_detail}"
    assert recv_ok, f"receiver: {recv_detail}"
