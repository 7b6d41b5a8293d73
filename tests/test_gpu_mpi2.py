"""2-rank MPI world with DEVICE buffers across two worker processes on
one GPU.

RCCL refuses two ranks on one device, so this exercises the PTP/HIP-IPC
device fallback plane end-to-end: planner gang-scheduling, cross-process
PTP mappings, device payloads over IPC arenas, and gfx950 elementwise
kernels for the reductions. On a multi-GPU node the same collectives
take the RCCL path (each rank gets its own device); the driver's SCALE
run covers that.

Reference shapes: tests/dist/mpi/test_mpi_functions.cpp.
"""

import multiprocessing as mp
import os
import sys
import time

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

torch = pytest.importorskip("torch")

requires_gpu = pytest.mark.gpu

WORLD = 2
N = 1 << 18  # floats per rank


def _device_collectives_fn(msg):
    import torch as t

    from faabric_amd import _core

    world_id, rank, size = _core.mpi_init()
    try:
        dev = t.device("cuda")
        F = _core.MpiDataType.FLOAT
        SUM = _core.MpiOp.SUM

        # allReduce: SUM of (rank+1)-filled buffers = 3.0 at size 2
        send = t.full((N,), float(rank + 1), device=dev)
        recv = t.zeros(N, device=dev)
        t.cuda.synchronize()
        _core.mpi_allreduce_ptr(rank, send.data_ptr(), recv.data_ptr(), N,
                                F, SUM)
        t.cuda.synchronize()
        expect = float(size * (size + 1) // 2)
        if not t.allclose(recv, t.full((N,), expect, device=dev)):
            msg.output_data = "allreduce mismatch"
            return 1

        # send/recv ring: rank 0 -> 1 -> 0
        buf = t.full((N,), float(100 + rank), device=dev)
        got = t.zeros(N, device=dev)
        t.cuda.synchronize()
        nxt = (rank + 1) % size
        prv = (rank - 1 + size) % size
        _core.mpi_send_ptr(rank, nxt, buf.data_ptr(), N, F)
        _core.mpi_recv_ptr(prv, rank, got.data_ptr(), N, F)
        t.cuda.synchronize()
        if not t.allclose(got, t.full((N,), float(100 + prv), device=dev)):
            msg.output_data = "send/recv mismatch"
            return 1

        # broadcast from root 1
        b = t.full((N,), float(7 * (rank + 1)), device=dev)
        t.cuda.synchronize()
        _core.mpi_bcast_ptr(1, rank, b.data_ptr(), N, F)
        t.cuda.synchronize()
        if not t.allclose(b, t.full((N,), 14.0, device=dev)):
            msg.output_data = "bcast mismatch"
            return 1

        # allGather
        ag_send = t.full((N,), float(rank + 50), device=dev)
        ag_recv = t.zeros(N * size, device=dev)
        t.cuda.synchronize()
        _core.mpi_allgather_ptr(rank, ag_send.data_ptr(),
                                ag_recv.data_ptr(), N, F)
        t.cuda.synchronize()
        for r in range(size):
            seg = ag_recv[r * N:(r + 1) * N]
            if not t.allclose(seg, t.full((N,), float(r + 50),
                                          device=dev)):
                msg.output_data = f"allgather mismatch seg {r}"
                return 1

        # allToAll: chunk j of rank r = r*10 + j
        a2a_send = t.zeros(N * size, device=dev)
        for j in range(size):
            a2a_send[j * N:(j + 1) * N] = float(rank * 10 + j)
        a2a_recv = t.zeros(N * size, device=dev)
        t.cuda.synchronize()
        _core.mpi_alltoall_ptr(rank, a2a_send.data_ptr(),
                               a2a_recv.data_ptr(), N, F)
        t.cuda.synchronize()
        for j in range(size):
            seg = a2a_recv[j * N:(j + 1) * N]
            if not t.allclose(seg, t.full((N,), float(j * 10 + rank),
                                          device=dev)):
                msg.output_data = f"alltoall mismatch seg {j}"
                return 1

        # scan (inclusive prefix sum)
        sc_send = t.full((N,), float(rank + 1), device=dev)
        sc_recv = t.zeros(N, device=dev)
        t.cuda.synchronize()
        _core.mpi_scan_ptr(rank, sc_send.data_ptr(), sc_recv.data_ptr(),
                           N, F, SUM)
        t.cuda.synchronize()
        prefix = float((rank + 1) * (rank + 2) // 2)
        if not t.allclose(sc_recv, t.full((N,), prefix, device=dev)):
            msg.output_data = "scan mismatch"
            return 1

        # reduceScatter: each rank gets its slice of the elementwise sum
        rs_send = t.zeros(N * size, device=dev)
        for j in range(size):
            rs_send[j * N:(j + 1) * N] = float(rank + 1 + j)
        rs_recv = t.zeros(N, device=dev)
        t.cuda.synchronize()
        _core.mpi_reducescatter_ptr(rank, rs_send.data_ptr(),
                                    rs_recv.data_ptr(), N, F, SUM)
        t.cuda.synchronize()
        # sum over ranks r of (r+1+slice) where slice = this rank
        expect_rs = float(sum(r + 1 + rank for r in range(size)))
        if not t.allclose(rs_recv, t.full((N,), expect_rs, device=dev)):
            msg.output_data = "reducescatter mismatch"
            return 1

        # Async isend/irecv overlapping GPU compute: post the request,
        # run a kernel on other data, then await and verify
        ab = t.full((N,), float(rank + 200), device=dev)
        ag = t.zeros(N, device=dev)
        work = t.rand(1 << 22, device=dev)
        t.cuda.synchronize()
        if rank == 0:
            req = _core.mpi_isend_ptr(0, 1, ab.data_ptr(), N, F)
            for _ in range(4):
                work = work * 1.0001 + 0.5  # overlapped compute
            _core.mpi_await(req)
            _core.mpi_recv_ptr(1, 0, ag.data_ptr(), N, F)
        else:
            req = _core.mpi_irecv_ptr(0, 1, ag.data_ptr(), N, F)
            for _ in range(4):
                work = work * 1.0001 + 0.5
            _core.mpi_await(req)
            _core.mpi_send_ptr(1, 0, ab.data_ptr(), N, F)
        t.cuda.synchronize()
        peer = 1 - rank
        if not t.allclose(ag, t.full((N,), float(peer + 200),
                                     device=dev)):
            msg.output_data = "async isend/irecv mismatch"
            return 1

        _core.mpi_barrier(rank)
        msg.output_data = f"device collectives ok rank {rank}"
        return 0
    finally:
        _core.mpi_finalize()


def _worker_proc(rank, stop, ready, result_q):
    sys.path.insert(0, REPO_ROOT)
    os.environ["FAABRIC_IPC_ARENA_MB"] = "64"
    import torch as t  # noqa: F401

    from faabric_amd import _core
    from faabric_amd.runtime import LocalRuntime, wait_for_batch

    base = 6100
    off = base + rank * 300
    try:
        _core.set_log_level("error")
        rt = LocalRuntime(port_offset=off, planner_port_offset=base,
                          slots=1)
        if rank == 0:
            # The worker's own snapshot server shares this offset
            rt.start_planner(with_snapshot_server=False)
        rt.start_worker()
        _core.register_function("gpumpi2", "devcoll",
                                _device_collectives_fn)
        ready.set()

        if rank == 0:
            deadline = time.monotonic() + 60
            while (time.monotonic() < deadline
                   and len(_core.get_available_hosts()) < 2):
                time.sleep(0.05)
            assert len(_core.get_available_hosts()) == 2

            ber = _core.batch_exec_factory("gpumpi2", "devcoll", 1)
            msgs = ber.messages
            msgs[0].is_mpi = True
            msgs[0].mpi_world_size = WORLD
            ber.messages = msgs
            # Gang-place rank r on worker r (preloaded decision)
            d = _core.SchedulingDecision()
            d.app_id = ber.app_id
            d.group_id = 0
            d.hosts = [f"127.0.0.1@{base + r * 300}" for r in range(WORLD)]
            d.message_ids = [0] * WORLD
            d.app_idxs = list(range(WORLD))
            d.group_idxs = list(range(WORLD))
            d.mpi_ports = [0] * WORLD
            d.n_functions = WORLD
            _core.preload_scheduling_decision(ber.app_id, d)
            _core.call_functions(ber)
            results = wait_for_batch(ber.app_id, WORLD, 180_000)
            result_q.put([
                (r.mpi_rank, r.return_value, r.output_data[:200],
                 r.executed_host) for r in results
            ])
        stop.wait(240)
        rt.stop()
    except Exception as e:
        result_q.put(("error", repr(e)))
        ready.set()


@requires_gpu
def test_two_rank_device_collectives_cross_process():
    """Every device collective verified on a real 2-rank world spanning
    two worker processes (PTP/IPC device plane; RCCL refuses shared
    devices and the world falls back automatically)."""
    ctx = mp.get_context("spawn")
    stop = ctx.Event()
    result_q = ctx.Queue()
    readies = []
    procs = []
    for r in range(WORLD):
        ready = ctx.Event()
        p = ctx.Process(target=_worker_proc,
                        args=(r, stop, ready, result_q))
        p.start()
        procs.append(p)
        readies.append(ready)
    try:
        for ready in readies:
            assert ready.wait(120), "worker failed to start"
        results = result_q.get(timeout=240)
        assert results and results[0] != "error", results
        assert len(results) == WORLD
        for rank, rc, out, host in results:
            assert rc == 0, (rank, out)
        hosts = {host for _, _, _, host in results}
        assert len(hosts) == WORLD, f"ranks did not span workers: {hosts}"
    finally:
        stop.set()
        for p in procs:
            p.join(timeout=30)
            if p.is_alive():
                p.terminate()
