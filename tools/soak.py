"""Churn soak: repeated batch/MPI-world/THREADS/migration cycles against
one planner to surface leaks, port exhaustion, stale-state bugs. Run
manually: python tools/soak.py [cycles]."""

import sys
import time

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from faabric_amd import _core  # noqa: E402
from faabric_amd.runtime import LocalRuntime, wait_for_batch  # noqa: E402


def main(cycles=15):
    rt = LocalRuntime(slots=32, port_offset=820, planner_port_offset=820)
    rt.start_planner(with_snapshot_server=False)
    rt.start_worker()
    _core.register_bench_functions()
    _core.register_mpi_example_functions()
    _core.register_function("soak", "noop", lambda msg: 0)

    t0 = time.monotonic()
    for i in range(cycles):
        # 1. Plain batch
        ber = _core.batch_exec_factory("soak", "noop", 16)
        _core.call_functions(ber)
        rs = wait_for_batch(ber.app_id, 16, 30_000)
        assert all(r.return_value == 0 for r in rs)

        # 2. MPI world churn (create/run/destroy a 4-rank world)
        ber = _core.batch_exec_factory("mpi-cpp", "vcollectives", 1)
        msgs = ber.messages
        msgs[0].is_mpi = True
        msgs[0].mpi_world_size = 4
        ber.messages = msgs
        _core.call_functions(ber)
        rs = wait_for_batch(ber.app_id, 4, 60_000)
        assert all(r.return_value == 0 for r in rs), [
            (r.mpi_rank, r.return_value, r.output_data) for r in rs
        ]

        # 3. State KV churn
        kv = _core.state_get_kv("soak", f"key{i % 4}", 64 * 1024)
        kv.set(bytes([i % 256]) * 64 * 1024)
        got = kv.get()
        assert got[:1] == bytes([i % 256])
        kv.append(b"x" * 128)

        if i % 5 == 4:
            print(f"cycle {i+1}/{cycles} ok "
                  f"({time.monotonic()-t0:.1f}s)", flush=True)

    print(f"SOAK OK: {cycles} cycles in {time.monotonic()-t0:.1f}s")
    rt.stop()


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 15)
