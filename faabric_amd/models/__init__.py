"""Benchmark workload definitions ("models" of this runtime are the
distributed function workloads, not neural networks — the reference has
no tensor dimension).

Each workload mirrors a BASELINE.json config:
  - ``composite_step``   — 256 MB fp32 allreduce + alltoall + state-KV
                           batch (bench.py's flagship step)
  - ``snapshot_pipeline``— 4 GiB diff+merge on the gfx950 kernels
  - ``mpi_examples``     — the C++ MPI programs (allreduce/ring/async)
"""

from faabric_amd import _core


def register_all():
    """Register every native benchmark payload in this process."""
    _core.register_bench_functions()
    _core.register_mpi_example_functions()


def composite_step_params(
    steps: int,
    warmup: int,
    allreduce_bytes: int = 256 * 1024 * 1024,
    batch_per_host: int = 128,
    kv_bytes: int = 4096,
    a2a_bytes: int = 1024 * 1024,
) -> bytes:
    return (
        f"steps={steps};warmup={warmup};bytes={allreduce_bytes};"
        f"batch={batch_per_host};kvbytes={kv_bytes};a2abytes={a2a_bytes}"
    ).encode()


def run_snapshot_pipeline(gib: float = 4.0, dirty_pct: float = 25.0,
                          iters: int = 5):
    return _core.bench_snapshot_pipeline(
        int(gib * (1 << 30)), iters=iters, warmup=2, dirty_pct=dirty_pct
    )
