"""True 2-rank 256 MB fp32 MPI_Allreduce on ONE GPU: two worker
processes share device 0, so the world takes the PTP/HIP-IPC device
fallback plane (RCCL refuses duplicate devices) — reduce-to-0 with the
gfx950 elementwise kernel + broadcast, all payload moving through IPC
arenas. Reports algorithmic GB/s per op. On a multi-GPU node the same
collective runs on RCCL over xGMI (driver SCALE run).

Run on an MI355X box: python tools/gpu_allreduce2.py
"""

import json
import multiprocessing as mp
import os
import re
import sys
import time

sys.path.insert(0, __file__.rsplit("/", 2)[0])

BASE = 6700
BYTES = 256 * 1024 * 1024
STEPS = 8
WARMUP = 2


def rank_proc(rank, q):
    os.environ["FAABRIC_IPC_ARENA_MB"] = "768"
    from faabric_amd import _core
    from faabric_amd.runtime import LocalRuntime, wait_for_batch

    _core.set_log_level("error")
    off = BASE + rank * 200
    rt = LocalRuntime(port_offset=off, planner_port_offset=BASE, slots=2)
    if rank == 0:
        rt.start_planner(with_snapshot_server=False)
    rt.start_worker()
    _core.register_bench_functions()

    if rank == 0:
        deadline = time.monotonic() + 60
        while (time.monotonic() < deadline
               and len(_core.get_available_hosts()) < 2):
            time.sleep(0.05)
        assert len(_core.get_available_hosts()) == 2

        ber = _core.batch_exec_factory("bench", "rankstep", 1)
        msgs = ber.messages
        msgs[0].is_mpi = True
        msgs[0].mpi_world_size = 2
        msgs[0].input_data = (
            f"steps={STEPS};warmup={WARMUP};bytes={BYTES};batch=0;"
            f"kvbytes=0;a2abytes=0;ppbytes=0;snapbytes=0"
        ).encode()
        ber.messages = msgs
        d = _core.SchedulingDecision()
        d.app_id = ber.app_id
        d.group_id = 0
        d.hosts = [f"127.0.0.1@{BASE + r * 200}" for r in range(2)]
        d.message_ids = [0, 0]
        d.app_idxs = [0, 1]
        d.group_idxs = [0, 1]
        d.mpi_ports = [0, 0]
        d.n_functions = 2
        _core.preload_scheduling_decision(ber.app_id, d)
        _core.call_functions(ber)
        try:
            rs = wait_for_batch(ber.app_id, 2, 300_000)
            out = [r.output_data for r in rs if r.mpi_rank == 0][0]
            # rankstep output: "step:...;ar:a,b,c;..." (ms per op)
            m = re.search(r"ar:([0-9.,]+)", out)
            ar_ms = [float(x) for x in m.group(1).split(",") if x]
            mean_s = sum(ar_ms) / len(ar_ms) / 1e3
            algbw = BYTES / mean_s / 1e9
            busbw = algbw * 2 * (2 - 1) / 2
            q.put(("ok", json.dumps({
                "allreduce2_shared_gpu_algbw_gbps": round(algbw, 2),
                "allreduce2_shared_gpu_busbw_gbps": round(busbw, 2),
                "ar_ms": [round(v, 2) for v in ar_ms],
            })))
        except Exception as e:
            q.put(("err", repr(e)))
    else:
        time.sleep(120)
    rt.stop()


def main():
    mp.set_start_method("spawn")
    q = mp.Queue()
    ps = [mp.Process(target=rank_proc, args=(r, q)) for r in range(2)]
    for p in ps:
        p.start()
    try:
        status, out = q.get(timeout=300)
    finally:
        for p in ps:
            p.terminate()
            p.join(timeout=10)
    assert status == "ok", out
    print(out)
    outdir = os.environ.get("GPU_MICRO_OUT", "gpurun_out")
    os.makedirs(outdir, exist_ok=True)
    with open(os.path.join(outdir, "allreduce2.json"), "w") as f:
        f.write(out)


if __name__ == "__main__":
    main()
