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
