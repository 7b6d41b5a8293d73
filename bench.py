#!/usr/bin/env python3
"""Flagship benchmark: the BASELINE.json composite metric —
`MPI_Allreduce` GB/s + batch-exec msgs/sec on an N-GPU MI355X node.

One *step* = one 256 MB fp32 `MPI_Allreduce` over the N-rank world (one
rank per GPU, RCCL over xGMI through faabric_amd's MpiWorld) + one small
all-to-all + one EXECUTE_BATCH of `--batch`×N state-KV functions through
the planner/scheduler path. The headline `value` is whole-job batch
throughput (messages/sec aggregated over the node); the measured allreduce
bus-bandwidth is reported in `config.allreduce_busbw_gbps`.

Launch (driver contract):
  python bench.py --gpus 1 --steps K --warmup W                  # 1 GPU
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N ...              # N GPUs

Each torchrun rank hosts one single-GPU faabric_amd worker process; rank 0
also runs the planner and submits the gang-scheduled MPI batch. Rank
functions are native C++ (cpp/src/bench_funcs.cpp) — Python never touches
the timed path.
"""

import argparse
import json
import os
import sys

REPO_ROOT = os.path.dirname(os.path.abspath(__file__))
if REPO_ROOT not in sys.path:
    sys.path.insert(0, REPO_ROOT)

RANK = int(os.environ.get("RANK", "0"))
LOCAL_RANK = int(os.environ.get("LOCAL_RANK", str(RANK)))
WORLD_SIZE = int(os.environ.get("WORLD_SIZE", "1"))

# Pin this process to its GPU before any HIP initialisation
if "HIP_VISIBLE_DEVICES" not in os.environ and WORLD_SIZE > 1:
    os.environ["HIP_VISIBLE_DEVICES"] = str(LOCAL_RANK)

BASE_OFFSET = int(os.environ.get("FAABRIC_BENCH_BASE_OFFSET", "9000"))
ALLREDUCE_BYTES = 256 * 1024 * 1024


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--batch", type=int, default=128,
                   help="kvtouch functions per host per step")
    p.add_argument("--bytes", type=int, default=ALLREDUCE_BYTES)
    p.add_argument("--a2a-bytes", type=int, default=1024 * 1024,
                   help="alltoall chunk bytes per rank pair")
    p.add_argument("--kv-bytes", type=int, default=4096)
    p.add_argument("--pp-bytes", type=int, default=64 * 1024 * 1024,
                   help="rank0<->rank1 ping-pong bytes (config 2)")
    p.add_argument("--snap-bytes", type=int, default=4 * 1024 * 1024 * 1024,
                   help="per-rank GPU snapshot diff+merge region (config 4: "
                        "4 GB random-byte region per GPU)")
    return p.parse_args()


def main():
    args = parse_args()
    n = WORLD_SIZE if WORLD_SIZE > 1 else args.gpus
    if WORLD_SIZE > 1 and args.gpus != WORLD_SIZE:
        n = WORLD_SIZE

    import faabric_amd  # noqa: F401
    from faabric_amd import _core
    from faabric_amd.runtime import LocalRuntime

    _core.set_log_level(os.environ.get("LOG_LEVEL", "warn"))

    have_gpu = False
    try:
        import torch

        have_gpu = torch.cuda.is_available()
    except Exception:
        pass
    if not have_gpu:
        # CPU fallback: shrink the host-path allreduce so the run stays fast
        args.bytes = min(args.bytes, 8 * 1024 * 1024)
        args.a2a_bytes = min(args.a2a_bytes, 64 * 1024)

    dist = None
    if WORLD_SIZE > 1:
        import torch.distributed as dist_mod

        dist = dist_mod
        dist.init_process_group(backend="gloo", rank=RANK,
                                world_size=WORLD_SIZE)

    # --- bring up the runtime: one worker per rank, planner on rank 0 ---
    my_offset = BASE_OFFSET + RANK * 200
    slots = args.batch + 1  # one gang slot + the batch-bench slots
    rt = LocalRuntime(
        port_offset=my_offset,
        planner_port_offset=BASE_OFFSET,
        slots=slots,
    )
    if RANK == 0:
        rt.start_planner(with_snapshot_server=False)
    if dist is not None:
        dist.barrier()
    rt.start_worker()
    _core.register_bench_functions()
    if dist is not None:
        dist.barrier()

    result = None
    if RANK == 0:
        # Wait for every worker to register
        import time

        deadline = time.monotonic() + 60
        while time.monotonic() < deadline:
            if len(_core.get_available_hosts()) == n:
                break
            time.sleep(0.05)
        hosts = sorted(h.ip for h in _core.get_available_hosts())
        assert len(hosts) == n, f"only {hosts} registered"

        # Gang placement: rank i on worker i (preloaded decision)
        # ping-pong reuses the allreduce buffers: clamp to their size
        pp = min(args.pp_bytes, args.bytes) if n >= 2 else 0
        params = (
            f"steps={args.steps};warmup={args.warmup};bytes={args.bytes};"
            f"batch={args.batch};kvbytes={args.kv_bytes};"
            f"a2abytes={args.a2a_bytes};ppbytes={pp};"
            f"snapbytes={args.snap_bytes if have_gpu else 0}"
        )
        ber = _core.batch_exec_factory("bench", "rankstep", 1)
        msgs = ber.messages
        msgs[0].is_mpi = True
        msgs[0].mpi_world_size = n
        msgs[0].input_data = params.encode()
        ber.messages = msgs

        ident_of_rank = [f"127.0.0.1@{BASE_OFFSET + r * 200}"
                         for r in range(n)]
        decision = _core.SchedulingDecision()
        decision.app_id = ber.app_id
        decision.group_id = 0
        for i in range(n):
            decision.hosts = decision.hosts + [ident_of_rank[i]]
            decision.message_ids = decision.message_ids + [0]
            decision.app_idxs = decision.app_idxs + [i]
            decision.group_idxs = decision.group_idxs + [i]
            decision.mpi_ports = decision.mpi_ports + [0]
        decision.n_functions = n
        _core.preload_scheduling_decision(ber.app_id, decision)

        sched = _core.call_functions(ber)
        assert sched.app_id == ber.app_id, f"schedule failed: {sched.app_id}"

        from faabric_amd.runtime import wait_for_batch

        timeout_ms = 120_000 + (args.steps + args.warmup) * 30_000
        results = wait_for_batch(ber.app_id, n, timeout_ms)
        assert all(r.return_value == 0 for r in results), [
            (r.mpi_rank, r.return_value, r.output_data) for r in results
        ]

        # Parse per-rank per-step times; take the MAX over ranks per step
        def parse_times(out, key):
            for part in out.split(";"):
                if part.startswith(key + ":"):
                    vals = part[len(key) + 1:]
                    return [float(x) for x in vals.split(",") if x]
            return []

        per_rank_step = [parse_times(r.output_data, "step") for r in results]
        per_rank_ar = [parse_times(r.output_data, "ar") for r in results]
        per_rank_batch = [parse_times(r.output_data, "batch")
                          for r in results]
        rank0_pp = parse_times(results[0].output_data, "pp")
        per_rank_ring = [parse_times(r.output_data, "ring")
                         for r in results]
        # Per-rank snapshot diff/apply GB/s (config 4; 0 on CPU)
        snap_diff = [parse_times(r.output_data, "snapdiff") for r in results]
        snap_apply = [parse_times(r.output_data, "snapapply") for r in results]
        snap_region = [parse_times(r.output_data, "snapdiffregion")
                       for r in results]
        snap_pct = parse_times(results[0].output_data, "snapdirtypct")
        snap_diff_total = sum(v[0] for v in snap_diff if v)
        snap_apply_total = sum(v[0] for v in snap_apply if v)
        snap_region_total = sum(v[0] for v in snap_region if v)
        k = min(len(s) for s in per_rank_step)
        step_ms = [max(s[i] for s in per_rank_step) for i in range(k)]
        ar_ms = [max(s[i] for s in per_rank_ar) for i in range(k)]
        batch_ms = [max(s[i] for s in per_rank_batch) for i in range(k)]
        batch_mean_ms = sum(batch_ms) / len(batch_ms) if batch_ms else 0.0

        total_s = sum(step_ms) / 1000.0
        ms_per_step = sum(step_ms) / k
        msgs_per_step = args.batch * n
        msgs_per_sec = msgs_per_step * k / total_s if total_s > 0 else 0.0

        ar_mean_s = (sum(ar_ms) / len(ar_ms)) / 1000.0 if ar_ms else 0.0
        algbw = args.bytes / ar_mean_s / 1e9 if ar_mean_s > 0 else 0.0
        busbw = algbw * 2 * (n - 1) / n if n > 1 else 0.0
        pp_mean_s = (
            (sum(rank0_pp) / len(rank0_pp)) / 1000.0
            if rank0_pp and n >= 2
            else 0.0
        )
        # Round trip moves 2 x pp_bytes through the link
        pp_gbps = 2 * pp / pp_mean_s / 1e9 if pp_mean_s > 0 else 0.0
        # Ring step: slowest rank's sendRecv of pp bytes each way
        ring_ms = [max(s[i] for s in per_rank_ring if s)
                   for i in range(k)] if n >= 2 and pp > 0 else []
        ring_mean_s = (sum(ring_ms) / len(ring_ms)) / 1000.0 if ring_ms \
            else 0.0
        ring_gbps = 2 * pp / ring_mean_s / 1e9 if ring_mean_s > 0 else 0.0

        result = {
            "metric": "MPI_Allreduce GB/s + batch-exec msgs/sec, "
                      "8-rank world at 1/2/4/8 MI355X",
            "value": round(msgs_per_sec, 2),
            "unit": "msgs/sec",
            "n_gpus": n,
            "steps": k,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic",
            "config": {
                "model": "faabric composite: 256MB fp32 allreduce + "
                         "1MB alltoall + kvtouch batch",
                "global_batch": msgs_per_step,
                "seq_len": None,
                "parallelism": f"mpi{n}",
                "allreduce_bytes": args.bytes,
                "allreduce_algbw_gbps": round(algbw, 2),
                "allreduce_busbw_gbps": round(busbw, 2),
                "allreduce_ms": round(ar_mean_s * 1000.0, 3),
                "pingpong_bytes": pp,
                "pingpong_gbps": round(pp_gbps, 2),
                "ring_exchange_gbps_per_rank": round(ring_gbps, 2),
                "batch_per_host": args.batch,
                "batch_msgs_per_sec": round(msgs_per_sec, 2),
                "batch_ms": round(batch_mean_ms, 3),
                "kv_bytes": args.kv_bytes,
                "snapshot_bytes_per_rank": args.snap_bytes if have_gpu else 0,
                "snapshot_diff_gbps_total": round(snap_diff_total, 2),
                "snapshot_apply_gbps_total": round(snap_apply_total, 2),
                "snapshot_diff_region_gbps_total":
                    round(snap_region_total, 2),
                "snapshot_dirty": "random-scattered "
                    + (str(int(snap_pct[0])) if snap_pct else "25")
                    + "pct, random-byte region",
                "gpu": have_gpu,
            },
        }

    if dist is not None:
        dist.barrier()
    if RANK == 0 and result is not None:
        print(json.dumps(result), flush=True)
        if os.environ.get("FAABRIC_PROF"):
            print(_core.prof_summary(), file=sys.stderr, flush=True)
    if dist is not None:
        dist.barrier()
        dist.destroy_process_group()
    rt.stop()


if __name__ == "__main__":
    main()
