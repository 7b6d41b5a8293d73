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


# ---------------------------------------------------------------------------
# gfx950 snapshot-engine kernels: numerics vs a plain torch reference
# ---------------------------------------------------------------------------


@requires_gpu
def test_dirty_pages_kernel_vs_torch():
    n_pages = 4096  # 16 MiB
    size = n_pages * 4096
    torch.manual_seed(0)
    base = torch.randint(0, 256, (size,), dtype=torch.uint8, device="cuda")
    snap = _core.DeviceSnapshot(size)
    snap.capture_from_ptr(base.data_ptr())

    updated = base.clone()
    dirty = [0, 1, 17, 100, 1000, 4095]
    for p in dirty:
        updated[p * 4096 + 7] ^= 0xFF
    torch.cuda.synchronize()

    flags = snap.dirty_pages(updated.data_ptr())
    got = [i for i, f in enumerate(flags) if f]
    assert got == dirty

    # torch reference: pages where any byte differs
    ref = (
        (base.view(n_pages, 4096) != updated.view(n_pages, 4096))
        .any(dim=1)
        .nonzero()
        .flatten()
        .tolist()
    )
    assert got == ref


@requires_gpu
def test_diff_apply_roundtrip_vs_torch():
    n_pages = 1024
    size = n_pages * 4096
    torch.manual_seed(1)
    base = torch.randint(0, 256, (size,), dtype=torch.uint8, device="cuda")
    snap = _core.DeviceSnapshot(size)
    snap.capture_from_ptr(base.data_ptr())

    updated = base.clone()
    updated[5 * 4096 : 6 * 4096] = 0x42
    updated[900 * 4096 + 100] = 0x00
    torch.cuda.synchronize()

    nd = snap.diff_xor(updated.data_ptr())
    assert 1 <= nd <= 3
    snap.apply_last_diff()

    out = snap.copy_out_host(size)
    ref = bytes(updated.cpu().numpy().tobytes())
    assert out == ref


@requires_gpu
@pytest.mark.parametrize(
    "dtype,op,torch_fn",
    [
        (3, 0, lambda a, b: a + b),  # float sum
        (3, 1, torch.maximum),       # float max
        (3, 2, torch.minimum),       # float min
        (3, 3, lambda a, b: a * b),  # float prod
        (0, 0, lambda a, b: a + b),  # int32 sum
        (4, 0, lambda a, b: a + b),  # double sum
    ],
)
@pytest.mark.parametrize("n", [1 << 20, 1_000_003, 513])
def test_elementwise_op_vs_torch(dtype, op, torch_fn, n):
    """Odd sizes exercise the unrolled kernels' tail loops."""
    torch_dtype = {0: torch.int32, 3: torch.float32, 4: torch.float64}[dtype]
    torch.manual_seed(2)
    if torch_dtype == torch.int32:
        a = torch.randint(-1000, 1000, (n,), dtype=torch_dtype, device="cuda")
        b = torch.randint(-1000, 1000, (n,), dtype=torch_dtype, device="cuda")
    else:
        a = torch.randn(n, dtype=torch_dtype, device="cuda")
        b = torch.randn(n, dtype=torch_dtype, device="cuda")
    ref = torch_fn(a.clone(), b)
    torch.cuda.synchronize()
    _core.device_elementwise_op(a.data_ptr(), b.data_ptr(), n, dtype, op)
    torch.cuda.synchronize()
    assert torch.equal(a, ref)


@requires_gpu
def test_snapshot_pipeline_bench_small():
    res = _core.bench_snapshot_pipeline(
        256 * 1024 * 1024, iters=2, warmup=1, dirty_pct=25.0
    )
    assert res["dirty_pages"] == int(res["n_pages"] * 0.25)
    assert res["diff_gbps"] > 10  # sanity floor; target is TB/s-class


@requires_gpu
def test_state_kv_in_hbm(runtime):
    kv = _core.state_get_kv_device("gpu", "hbmkey", 256 * 1024)
    assert kv.on_device
    assert kv.is_master
    kv.set(b"\x07" * (256 * 1024))
    assert kv.get() == b"\x07" * (256 * 1024)
    kv.set_chunk(4096, b"CHUNK")
    data = kv.get_chunk(4090, 16)
    assert data[6:11] == b"CHUNK"
    # The HBM pointer is directly usable by kernels
    n = 256 * 1024
    t = torch.zeros(n, dtype=torch.uint8, device="cuda")
    torch.cuda.synchronize()
    _core.device_elementwise_op(
        t.data_ptr(), kv.data_ptr, n, 5, 0
    )  # byte sum: t += kv
    torch.cuda.synchronize()
    assert int(t[0]) == 7 and int(t[5000]) == 7


@requires_gpu
def test_snapshot_scattered_random_diff(runtime):
    """Random-byte region + randomly scattered dirty pages: the diff
    kernels must find exactly the touched set and merge back to equal
    buffers (the BASELINE config-4 shape, not a contiguous memset)."""
    import random as pyrandom

    n_pages = 512
    nbytes = n_pages * 4096
    buf = torch.empty(nbytes, dtype=torch.uint8, device="cuda")
    _core.fam_fill_random(buf.data_ptr(), nbytes, 99)
    snap = _core.DeviceSnapshot(nbytes)
    snap.capture_from_ptr(buf.data_ptr())

    pyrandom.seed(5)
    dirty = sorted(pyrandom.sample(range(n_pages), 128))
    _core.fam_touch_pages(buf.data_ptr(), dirty, 1234)

    nd = snap.diff_xor(buf.data_ptr())
    assert nd == 128
    pages, payload = snap.gather_last_diff()
    assert sorted(pages) == dirty

    snap.apply_last_diff()
    # Snapshot now equals the updated buffer: a second diff is empty
    assert snap.diff_xor(buf.data_ptr()) == 0
    got = snap.copy_out_host(nbytes)
    expect = buf.cpu().numpy().tobytes()
    assert got == expect


@requires_gpu
def test_state_kv_mirror_coherence(runtime):
    """The pinned write-through mirror stays coherent with HBM across
    direct device-pointer writes (data_ptr access invalidates the
    mirror) and chunk writes land in HBM after sync()."""
    n = 64 * 1024
    kv = _core.state_get_kv_device("gpu", "mirrorkey", n)
    kv.set(b"\x11" * n)
    kv.sync()

    # Bypass write: kernel adds 0x22 to every byte straight in HBM
    add = torch.full((n,), 0x22, dtype=torch.uint8, device="cuda")
    torch.cuda.synchronize()
    _core.device_elementwise_op(kv.data_ptr, add.data_ptr(), n, 5, 0)
    torch.cuda.synchronize()
    # Chunk reads must observe the kernel's bytes, not a stale mirror
    assert kv.get_chunk(0, 16) == bytes([0x33]) * 16
    assert kv.get_chunk(n - 16, 16) == bytes([0x33]) * 16

    # Partial write into an invalidated page keeps its neighbours
    kv.set_chunk(100, b"\x44" * 8)
    got = kv.get_chunk(96, 16)
    assert got == bytes([0x33]) * 4 + bytes([0x44]) * 8 + bytes([0x33]) * 4

    # After sync, HBM holds the merged content (read via a fresh kernel)
    kv.sync()
    out = torch.zeros(n, dtype=torch.uint8, device="cuda")
    torch.cuda.synchronize()
    _core.device_elementwise_op(out.data_ptr(), kv.data_ptr, n, 5, 0)
    torch.cuda.synchronize()
    assert int(out[100]) == 0x44 and int(out[99]) == 0x33
    assert int(out[50000]) == 0x33


def _gpu_thread_body(msg):
    # Flip one device page per thread: thread i XORs page (i) with 0x0F
    idx = msg.group_idx
    page = bytes([0x0F]) * 4096
    cur = _core.executor_device_read_memory((idx - 1) * 4096, 4096)
    mixed = bytes(a ^ b for a, b in zip(cur, page))
    _core.executor_device_write_memory((idx - 1) * 4096, mixed)
    return 0


def _gpu_fork_parent(msg):
    n_pages = 64
    _core.executor_set_device_memory_size(n_pages * 4096)
    base = bytes([0x30]) * (n_pages * 4096)
    _core.executor_device_write_memory(0, base)

    results = _core.execute_threads("gputhreads", "body", 3)
    if len(results) != 3 or any(rv != 0 for _, rv in results):
        msg.output_data = f"thread failures: {results}"
        return 1

    data = _core.executor_device_read_memory(0, n_pages * 4096)
    for p in range(3):
        if data[p * 4096] != 0x3F:  # 0x30 ^ 0x0F
            msg.output_data = f"page {p} wrong: {data[p * 4096]:#x}"
            return 2
    if data[3 * 4096] != 0x30:
        msg.output_data = "untouched page modified"
        return 3
    msg.output_data = "gpu fork-join ok"
    return 0


@requires_gpu
def test_gpu_threads_fork_join(runtime):
    """THREADS fork-join over an HBM-resident snapshot: capture, restore,
    XOR page diff on the gfx950 kernels, packed merge-back."""
    _core.register_function("gputhreads", "body", _gpu_thread_body)
    _core.register_function("gputhreads", "parent", _gpu_fork_parent)
    ber = _core.batch_exec_factory("gputhreads", "parent", 1)
    decision = _core.call_functions(ber)
    assert decision.app_id == ber.app_id
    results = wait_for_batch(ber.app_id, 1, timeout_ms=60_000)
    assert results[0].return_value == 0, results[0].output_data
    assert results[0].output_data == "gpu fork-join ok"


@requires_gpu
def test_ptp_device_payloads(runtime):
    """PTP broker device payloads: same-process delivery via D2D staging
    on the broker's side stream, ordered."""
    decision = _core.SchedulingDecision()
    decision.app_id = 777000
    decision.group_id = 777001
    decision.hosts = [runtime.identity, runtime.identity]
    decision.message_ids = [1, 2]
    decision.app_idxs = [0, 1]
    decision.group_idxs = [0, 1]
    decision.mpi_ports = [0, 0]
    decision.n_functions = 2
    _core.ptp_setup_local_mappings(decision)

    n = 1 << 16
    src = torch.arange(n, dtype=torch.float32, device="cuda")
    dst = torch.zeros(n, dtype=torch.float32, device="cuda")
    torch.cuda.synchronize()
    _core.ptp_send_device(
        777000, 777001, 0, 1, src.data_ptr(), n * 4, True
    )
    src.fill_(0)  # sender buffer reusable immediately (staged copy)
    torch.cuda.synchronize()
    got = _core.ptp_recv_device(777001, 0, 1, dst.data_ptr(), n * 4, True)
    assert got == n * 4
    torch.cuda.synchronize()
    assert torch.equal(dst, torch.arange(n, dtype=torch.float32,
                                         device="cuda"))

    # Host-sent payload received into a device buffer
    _core.ptp_send(777000, 777001, 0, 1, b"\x42" * 1024, False)
    buf = torch.zeros(1024, dtype=torch.uint8, device="cuda")
    got = _core.ptp_recv_device(777001, 0, 1, buf.data_ptr(), 1024)
    assert got == 1024
    torch.cuda.synchronize()
    assert int(buf.sum()) == 0x42 * 1024


@pytest.mark.gpu
def test_device_snapshot_wire_diff_ship():
    """Peer-GPU snapshot shipping contract: diff on snapshot A, gather the
    compact wire form ({pages, slot-compacted 4KiB payloads}) to the host,
    apply it onto an identical snapshot B (reference parity:
    SnapshotClient::pushSnapshotUpdate shipping diffs between hosts,
    src/snapshot/SnapshotClient.cpp)."""
    import torch

    from faabric_amd import _core

    n = 64 * 4096
    base = torch.randint(0, 256, (n,), dtype=torch.uint8, device="cuda")
    snap_a = _core.DeviceSnapshot(n, 0)
    snap_b = _core.DeviceSnapshot(n, 0)
    snap_a.capture_from_ptr(base.data_ptr())
    snap_b.capture_from_ptr(base.data_ptr())

    # Mutate a few pages
    updated = base.clone()
    updated[5 * 4096 + 3] ^= 0xFF
    updated[17 * 4096 : 18 * 4096] = 0xAB
    updated[63 * 4096 + 4095] ^= 0x01
    torch.cuda.synchronize()

    nd = snap_a.diff_xor(updated.data_ptr())
    assert nd == 3, nd
    pages, payload = snap_a.gather_last_diff()
    assert sorted(pages) == [5, 17, 63]
    assert len(payload) == 3 * 4096

    snap_b.apply_compact_diff(pages, payload)
    out = snap_b.copy_out_host(n)
    assert out == updated.cpu().numpy().tobytes()


def _scan_1rank_fn(msg):
    world_id, rank, size = _core.mpi_init()
    n = 1 << 18
    send = torch.arange(n, dtype=torch.float32, device="cuda")
    recv = torch.zeros(n, dtype=torch.float32, device="cuda")
    torch.cuda.synchronize()
    _core.mpi_scan_ptr(rank, send.data_ptr(), recv.data_ptr(), n,
                       _core.MpiDataType.FLOAT, _core.MpiOp.SUM)
    torch.cuda.synchronize()
    if not torch.equal(recv, send):
        return 1
    msg.output_data = "scan ok"
    return 0


@requires_gpu
def test_device_scan_single_rank(runtime):
    """Device-buffer MPI_Scan (RCCL chain + fused elementwise combine);
    the 1-rank world exercises the NT device copy path."""
    _core.register_function("gpu", "scan1", _scan_1rank_fn)
    results = submit_mpi("gpu", "scan1", 1)
    assert results[0].return_value == 0, results[0].output_data
    assert results[0].output_data == "scan ok"


@pytest.mark.gpu
def test_state_kv_mirror_coherence_writeback():
    """Same mirror-coherence semantics under FAABRIC_KV_WRITEBACK=1
    (dirty pages flushed coalesced at sync); run in a subprocess because
    the policy is a process-wide static."""
    import os
    import subprocess
    import sys as _sys

    REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    code = """
import sys
sys.path.insert(0, %r)
from faabric_amd import _core
import torch

_core.set_log_level("error")
n = 64 * 1024
kv = _core.state_get_kv_device("gpu", "wbkey", n)
kv.set(b"\\x11" * n)
kv.sync()
add = torch.full((n,), 0x22, dtype=torch.uint8, device="cuda")
torch.cuda.synchronize()
_core.device_elementwise_op(kv.data_ptr, add.data_ptr(), n, 5, 0)
torch.cuda.synchronize()
assert kv.get_chunk(0, 16) == bytes([0x33]) * 16
kv.set_chunk(100, b"\\x44" * 8)
got = kv.get_chunk(96, 16)
assert got == bytes([0x33]) * 4 + bytes([0x44]) * 8 + bytes([0x33]) * 4
kv.sync()
out = torch.zeros(n, dtype=torch.uint8, device="cuda")
torch.cuda.synchronize()
_core.device_elementwise_op(out.data_ptr(), kv.data_ptr, n, 5, 0)
torch.cuda.synchronize()
assert int(out[100]) == 0x44 and int(out[99]) == 0x33 and int(out[50000]) == 0x33
print("WB_COHERENCE_OK")
""" % REPO_ROOT
    env = dict(os.environ, FAABRIC_KV_WRITEBACK="1")
    r = subprocess.run([_sys.executable, "-c", code], env=env,
                       capture_output=True, text=True, timeout=180)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "WB_COHERENCE_OK" in r.stdout
