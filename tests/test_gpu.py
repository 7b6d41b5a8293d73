"""GPU tests (MI355X): RCCL data plane, device buffers, bench smoke.
Run via gpurun: python -m pytest tests -m gpu -x -q"""

import struct

import pytest

import faabric_amd as fa
from faabric_amd import _core
from faabric_amd.runtime import LocalRuntime, wait_for_batch

pytestmark = pytest.mark.gpu

torch = pytest.importorskip("torch")

requires_gpu = pytest.mark.skipif(
    not torch.cuda.is_available(), reason="needs an MI355X"
)


@pytest.fixture(scope="module")
def runtime():
    rt = LocalRuntime(slots=8, port_offset=13000, planner_port_offset=13000)
    rt.start_planner(with_snapshot_server=False)
    rt.start_worker()
    yield rt
    rt.stop()


def submit_mpi(user, func, world_size, input_data=b"", timeout_ms=120_000):
    ber = _core.batch_exec_factory(user, func, 1)
    msgs = ber.messages
    msgs[0].is_mpi = True
    msgs[0].mpi_world_size = world_size
    msgs[0].input_data = input_data
    ber.messages = msgs
    decision = _core.call_functions(ber)
    assert decision.app_id == ber.app_id
    return wait_for_batch(ber.app_id, world_size, timeout_ms)


def _rccl_1rank_fn(msg):
    """1-rank world: device allreduce is a D2D copy through RCCL; checks
    the whole GPU plane (comm bootstrap, stream, pointer probing)."""
    world_id, rank, size = _core.mpi_init()
    n = 1 << 20
    send = torch.arange(n, dtype=torch.float32, device="cuda")
    recv = torch.zeros(n, dtype=torch.float32, device="cuda")
    _core.mpi_allreduce_ptr(
        rank,
        send.data_ptr(),
        recv.data_ptr(),
        n,
        _core.MpiDataType.FLOAT,
        _core.MpiOp.SUM,
    )
    torch.cuda.synchronize()
    if not torch.equal(send, recv):
        return 1
    # alltoall with 1 rank = local copy
    recv2 = torch.zeros(n, dtype=torch.float32, device="cuda")
    _core.mpi_alltoall_ptr(
        rank, send.data_ptr(), recv2.data_ptr(), n, _core.MpiDataType.FLOAT
    )
    torch.cuda.synchronize()
    if not torch.equal(send, recv2):
        return 2
    msg.output_data = "rccl 1-rank ok"
    return 0


@requires_gpu
def test_rccl_single_rank_world(runtime):
    _core.register_function("gpu", "rccl1", _rccl_1rank_fn)
    results = submit_mpi("gpu", "rccl1", 1)
    assert results[0].return_value == 0, results[0].output_data


@requires_gpu
def test_device_pointer_probe(runtime):
    # AUTO location must route host pointers through the host plane even
    # when a GPU is present
    def fn(msg):
        world_id, rank, size = _core.mpi_init()
        out = _core.mpi_allreduce_bytes(
            rank,
            struct.pack("<i", 21),
            _core.MpiDataType.INT32,
            _core.MpiOp.SUM,
        )
        (v,) = struct.unpack("<i", out)
        return 0 if v == 21 else 1

    _core.register_function("gpu", "hostptr", fn)
    results = submit_mpi("gpu", "hostptr", 1)
    assert results[0].return_value == 0


@requires_gpu
def test_bench_functions_on_gpu(runtime):
    _core.register_bench_functions()
    params = b"steps=2;warmup=1;bytes=16777216;batch=8;kvbytes=4096;a2abytes=65536"
    results = submit_mpi("bench", "rankstep", 1, input_data=params)
    assert results[0].return_value == 0, results[0].output_data
    assert "step:" in results[0].output_data
