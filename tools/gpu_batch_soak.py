"""Sustained batch-execution soak on one GPU: thousands of 128-function
state-KV batches through the planner/scheduler path with the device-KV
pinned mirror, asserting steady-state RSS and bounded runtime tables
(result GC + GROUP_CLEAR retirement). Catches per-batch leaks the
functional tests can't see.

Run on an MI355X box: python tools/gpu_batch_soak.py [seconds] [batch]
"""

import os
import sys
import time

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from faabric_amd import _core
from faabric_amd.runtime import LocalRuntime, wait_for_batch


def rss_mb():
    with open("/proc/self/statm") as f:
        return int(f.read().split()[1]) * 4096 / 1e6


def main():
    seconds = int(sys.argv[1]) if len(sys.argv) > 1 else 120
    batch_n = int(sys.argv[2]) if len(sys.argv) > 2 else 128
    _core.set_log_level("error")
    rt = LocalRuntime(port_offset=7700, planner_port_offset=7700,
                      slots=batch_n + 1)
    rt.start_planner(with_snapshot_server=False)
    rt.start_worker()
    _core.register_bench_functions()

    batch = batch_n
    for _ in range(10):
        ber = _core.batch_exec_factory("bench", "kvtouch", batch)
        _core.call_functions(ber)
        wait_for_batch(ber.app_id, batch, 30_000)

    r0 = rss_mb()
    t0 = time.time()
    n = 0
    rmid = None
    while time.time() - t0 < seconds:
        ber = _core.batch_exec_factory("bench", "kvtouch", batch)
        _core.call_functions(ber)
        wait_for_batch(ber.app_id, batch, 30_000)
        n += 1
        if rmid is None and time.time() - t0 > 0.6 * seconds:
            rmid = rss_mb()
    r1 = rss_mb()
    sizes = dict(_core._debug_runtime_sizes())
    rate = n * batch / (time.time() - t0)
    print(f"GPU BATCH SOAK: {n} batches, {rate:.0f} msgs/s, "
          f"rss {r0:.0f} -> {rmid:.0f} -> {r1:.0f} MB "
          f"(2nd-half growth {r1 - rmid:.1f}), sizes {sizes}")
    # Allocator arenas still expand slowly early on; the bound is loose
    # enough to ignore that and tight enough to catch per-batch leaks
    # (a 1 KB/batch leak at this rate is >100 MB)
    assert r1 - rmid < 64, f"rss grew {r1 - rmid:.1f} MB in steady state"
    assert sizes["broker_mappings"] == 0, sizes
    assert sizes["planner_in_flight"] == 0, sizes
    print("GPU BATCH SOAK OK")
    rt.stop()


if __name__ == "__main__":
    main()
