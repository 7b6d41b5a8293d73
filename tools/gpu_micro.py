"""Micro-benchmarks for runtime primitives unique to the MI355X design:
- PTP broker device-payload bandwidth (ordered D2D staging on the side
  stream — the plane a ring/sequence-parallel layer would ride)
- GPU THREADS fork-join round trip (HBM arena snapshot + N threads +
  sparse XOR diff merge-back)

Run on an MI355X box: python tools/gpu_micro.py
"""

import json
import sys
import time

sys.path.insert(0, __file__.rsplit("/", 2)[0])

import torch  # noqa: E402

from faabric_amd import _core  # noqa: E402
from faabric_amd.runtime import LocalRuntime, wait_for_batch  # noqa: E402

ARENA_MB = 64


def _micro_fork_parent(msg):
    _core.executor_set_device_memory_size(ARENA_MB << 20)
    results = _core.execute_threads("gmicro", "forkchild", 4)
    return 0 if all(rv == 0 for _, rv in results) else 1


def _micro_fork_child(msg):
    idx = msg.group_idx
    cur = _core.executor_device_read_memory((idx - 1) * 4096, 4096)
    _core.executor_device_write_memory(
        (idx - 1) * 4096, bytes(b ^ 0xA5 for b in cur)
    )
    return 0


def bench_ptp_device(rt):
    decision = _core.SchedulingDecision()
    decision.app_id = 888000
    decision.group_id = 888001
    decision.hosts = [rt.identity, rt.identity]
    decision.message_ids = [1, 2]
    decision.app_idxs = [0, 1]
    decision.group_idxs = [0, 1]
    decision.mpi_ports = [0, 0]
    decision.n_functions = 2
    _core.ptp_setup_local_mappings(decision)

    out = {}
    for mb in (1, 16, 64):
        n = mb << 20
        src = torch.full((n,), 7, dtype=torch.uint8, device="cuda")
        dst = torch.zeros(n, dtype=torch.uint8, device="cuda")
        torch.cuda.synchronize()
        reps = 20
        t0 = time.perf_counter()
        for _ in range(reps):
            _core.ptp_send_device(
                888000, 888001, 0, 1, src.data_ptr(), n, True
            )
            got = _core.ptp_recv_device(
                888001, 0, 1, dst.data_ptr(), n, True, 10_000
            )
            assert got == n
        el = time.perf_counter() - t0
        out[f"ptp_device_{mb}mb_gbps"] = round(n * reps / el / 1e9, 2)
    return out


def bench_fork_join():
    _core.register_function("gmicro", "forkparent", _micro_fork_parent)
    _core.register_function("gmicro", "forkchild", _micro_fork_child)
    # warm
    for _ in range(2):
        ber = _core.batch_exec_factory("gmicro", "forkparent", 1)
        _core.call_functions(ber)
        wait_for_batch(ber.app_id, 1, 60_000)
    reps = 10
    t0 = time.perf_counter()
    for _ in range(reps):
        ber = _core.batch_exec_factory("gmicro", "forkparent", 1)
        _core.call_functions(ber)
        rs = wait_for_batch(ber.app_id, 1, 60_000)
        assert rs[0].return_value == 0
    el = time.perf_counter() - t0
    return {
        "gpu_forkjoin_ms": round(el / reps * 1000, 2),
        "gpu_forkjoin_arena_mb": ARENA_MB,
        "gpu_forkjoin_threads": 4,
    }


def main():
    assert torch.cuda.is_available()
    rt = LocalRuntime(slots=8, port_offset=880, planner_port_offset=880)
    rt.start_planner(with_snapshot_server=False)
    rt.start_worker()
    res = {}
    res.update(bench_ptp_device(rt))
    res.update(bench_fork_join())
    print(json.dumps(res))
    rt.stop()


if __name__ == "__main__":
    main()
