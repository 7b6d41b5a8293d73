"""HBM leak soak for the GPU paths: repeated device fork-join, HBM state
KV churn and DeviceSnapshot create/diff/destroy cycles, asserting free
HBM does not drift (long-running production-worthiness check).

Run on an MI355X box: python tools/gpu_soak.py [cycles]
"""

import sys

sys.path.insert(0, __file__.rsplit("/", 2)[0])

import torch  # noqa: E402

from faabric_amd import _core  # noqa: E402
from faabric_amd.runtime import LocalRuntime, wait_for_batch  # noqa: E402


def _rccl_world_fn(msg):
    world_id, rank, size = _core.mpi_init()
    n = 1 << 16
    send = torch.full((n,), 3.0, dtype=torch.float32, device="cuda")
    recv = torch.zeros(n, dtype=torch.float32, device="cuda")
    torch.cuda.synchronize()
    _core.mpi_allreduce_ptr(rank, send.data_ptr(), recv.data_ptr(), n,
                            _core.MpiDataType.FLOAT, _core.MpiOp.SUM)
    torch.cuda.synchronize()
    return 0 if torch.equal(recv, send) else 1


def _fork_parent(msg):
    _core.executor_set_device_memory_size(1 << 20)
    _core.executor_device_write_memory(0, bytes(64))
    results = _core.execute_threads("gsoak", "forkchild", 4)
    return 0 if all(rv == 0 for _, rv in results) else 1


def _fork_child(msg):
    idx = msg.group_idx
    cur = _core.executor_device_read_memory((idx - 1) * 4096, 4096)
    _core.executor_device_write_memory(
        (idx - 1) * 4096, bytes(b ^ 0x55 for b in cur)
    )
    return 0


def main(cycles=25):
    assert torch.cuda.is_available(), "needs an MI355X"
    rt = LocalRuntime(slots=16, port_offset=860, planner_port_offset=860)
    rt.start_planner(with_snapshot_server=False)
    rt.start_worker()
    _core.register_function("gsoak", "forkparent", _fork_parent)
    _core.register_function("gsoak", "forkchild", _fork_child)
    _core.register_function("gsoak", "rcclworld", _rccl_world_fn)

    # Warm up allocators/pools before baselining free HBM
    for _ in range(3):
        ber = _core.batch_exec_factory("gsoak", "forkparent", 1)
        _core.call_functions(ber)
        wait_for_batch(ber.app_id, 1, 60_000)
    torch.cuda.synchronize()
    free0, total = torch.cuda.mem_get_info()
    mid_drift_mb = 0.0

    for i in range(cycles):
        # 1. Device fork-join (DeviceSnapshot + diff buffers per fork)
        ber = _core.batch_exec_factory("gsoak", "forkparent", 1)
        _core.call_functions(ber)
        rs = wait_for_batch(ber.app_id, 1, 60_000)
        assert rs[0].return_value == 0, rs[0].output_data

        # 2. HBM KV churn (pinned-mirror write-through path: partial
        # writes, mirror fills, group-commit sync)
        kv = _core.state_get_kv_device("gsoak", f"k{i % 3}", 1 << 20)
        kv.set(bytes([i % 251]) * (1 << 20))
        assert kv.get_chunk(0, 1) == bytes([i % 251])
        kv.set_chunk(777, bytes([(i + 1) % 251]) * 100)
        kv.sync()
        got = kv.get_chunk(770, 10)
        assert got[:7] == bytes([i % 251]) * 7

        # 3. RCCL world churn: a fresh device-plane world per cycle; its
        # communicator and stream must be reclaimed when the rank finishes
        ber = _core.batch_exec_factory("gsoak", "rcclworld", 1)
        msgs = ber.messages
        msgs[0].is_mpi = True
        msgs[0].mpi_world_size = 1
        ber.messages = msgs
        _core.call_functions(ber)
        rs = wait_for_batch(ber.app_id, 1, 60_000)
        assert rs[0].return_value == 0, rs[0].output_data

        # 4. Raw snapshot cycle with the scattered-random kernels
        snap = _core.DeviceSnapshot(1 << 22, 0)
        t = torch.empty(1 << 22, dtype=torch.uint8, device="cuda")
        _core.fam_fill_random(t.data_ptr(), 1 << 22, i)
        snap.capture_from_ptr(t.data_ptr())
        _core.fam_touch_pages(t.data_ptr(), [3, 77, 500, 1001], i)
        nd = snap.diff_xor(t.data_ptr())
        assert nd == 4, nd
        del snap, t

        if (i + 1) % 10 == 0:
            torch.cuda.synchronize()
            free_now, _ = torch.cuda.mem_get_info()
            drift = (free0 - free_now) / (1 << 20)
            if i + 1 == (cycles // 2 // 10) * 10:
                mid_drift_mb = drift
            rss = 0
            try:
                import psutil

                rss = psutil.Process().memory_info().rss / (1 << 20)
            except Exception:
                pass
            print(f"cycle {i+1}/{cycles}: HBM drift {drift:+.1f} MiB, "
                  f"RSS {rss:.0f} MiB", flush=True)

    torch.cuda.synchronize()
    free1, _ = torch.cuda.mem_get_info()
    drift_mb = (free0 - free1) / (1 << 20)
    # Pools (state KVs, stripe streams, RCCL init, executor arenas)
    # settle in the first half; a REAL leak shows as continued growth
    # in the second half of the run
    growth_mb = drift_mb - mid_drift_mb
    assert growth_mb < 16, (
        f"HBM leak: {growth_mb:.0f} MiB grown over the second half "
        f"(total drift {drift_mb:.0f} MiB)"
    )
    assert drift_mb < 256, f"HBM settle too large: {drift_mb:.0f} MiB"
    print(f"GPU SOAK OK: {cycles} cycles, HBM drift {drift_mb:+.1f} MiB "
          f"(second-half growth {growth_mb:+.1f} MiB)")
    rt.stop()


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 25)
