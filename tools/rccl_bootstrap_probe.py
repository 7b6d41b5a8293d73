"""RCCL bootstrap probe: 2 MPI ranks on ONE GPU. Expected outcome on a
1-GPU box is a clean `RCCL error: invalid usage` (duplicate device) from
both ranks, which proves the cross-process ncclUniqueId handshake and
RCCL rendezvous work and that comm-init failures propagate as loud task
failures instead of hangs. On a multi-GPU node each rank binds its own
device and init succeeds (driver scale run)."""
import os, sys, multiprocessing as mp
sys.path.insert(0, "/root/repo")

def rank_proc(rank, q):
    os.environ["HIP_VISIBLE_DEVICES"] = "0"
    os.environ["FAABRIC_PORT_OFFSET"] = str(9400 + rank * 200)
    import faabric_amd
    from faabric_amd import _core
    from faabric_amd.runtime import LocalRuntime
    _core.set_log_level("warn")
    rt = LocalRuntime(port_offset=9400 + rank * 200, planner_port_offset=9400, slots=4)
    if rank == 0:
        rt.start_planner(with_snapshot_server=False)
    rt.start_worker()
    _core.register_bench_functions()
    import time
    if rank == 0:
        deadline = time.monotonic() + 30
        while time.monotonic() < deadline and len(_core.get_available_hosts()) < 2:
            time.sleep(0.05)
        ber = _core.batch_exec_factory("bench", "rankstep", 1)
        msgs = ber.messages
        msgs[0].is_mpi = True
        msgs[0].mpi_world_size = 2
        msgs[0].input_data = b"steps=1;warmup=0;bytes=1048576;batch=0;kvbytes=0;a2abytes=0;ppbytes=0;snapbytes=0"
        ber.messages = msgs
        d = _core.SchedulingDecision()
        d.app_id = ber.app_id; d.group_id = 0
        for i in range(2):
            d.hosts = d.hosts + [f"127.0.0.1@{9400 + i*200}"]
            d.message_ids = d.message_ids + [0]
            d.app_idxs = d.app_idxs + [i]
            d.group_idxs = d.group_idxs + [i]
            d.mpi_ports = d.mpi_ports + [0]
        d.n_functions = 2
        _core.preload_scheduling_decision(ber.app_id, d)
        _core.call_functions(ber)
        from faabric_amd.runtime import wait_for_batch
        try:
            results = wait_for_batch(ber.app_id, 2, 60_000)
            q.put(("done", [(r.mpi_rank, r.return_value, r.output_data[:200]) for r in results]))
        except Exception as e:
            q.put(("error", str(e)[:300]))
    else:
        time.sleep(70)
    rt.stop()

if __name__ == "__main__":
    mp.set_start_method("spawn")
    q = mp.Queue()
    ps = [mp.Process(target=rank_proc, args=(r, q)) for r in range(2)]
    [p.start() for p in ps]
    try:
        print(q.get(timeout=80))
    except Exception as e:
        print("NO RESULT:", e)
    [p.terminate() for p in ps]
    [p.join() for p in ps]
