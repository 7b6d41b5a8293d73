"""Both state backends behind one State API, like the reference's
InMemory/Redis pair behind StateKeyValue (reference:
src/state/InMemoryStateKeyValue.cpp, src/state/RedisStateKeyValue.cpp,
tests/utils/fixtures.h:57-103 StateFixture runs suites against both).

Backend "inmemory": master-per-key, first-toucher owns (default).
Backend "planner": the planner process hosts a global StateServer and
owns every key — the Redis-service role — including scripted locks
(redis/Redis.h:154-168: acquire with expiry, release only if the token
matches).
"""

import multiprocessing as mp
import os
import sys
import time

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

PLANNER_OFF = 7000
WORKER_OFF = 7100


def _planner_proc(stop, ready):
    sys.path.insert(0, REPO_ROOT)
    from faabric_amd import _core
    from faabric_amd.runtime import LocalRuntime

    _core.set_log_level("error")
    rt = LocalRuntime(port_offset=PLANNER_OFF)
    rt.start_planner(with_snapshot_server=False, with_state_server=True)
    ready.set()
    stop.wait(120)
    rt.stop()


@pytest.fixture(scope="module")
def planner_store():
    ctx = mp.get_context("spawn")
    stop = ctx.Event()
    ready = ctx.Event()
    p = ctx.Process(target=_planner_proc, args=(stop, ready))
    p.start()
    assert ready.wait(60), "planner failed to start"
    yield
    stop.set()
    p.join(timeout=30)
    if p.is_alive():
        p.terminate()


@pytest.fixture(params=["inmemory", "planner"])
def backend(request, planner_store):
    from faabric_amd import _core

    mode = request.param
    prev_off = _core.get_port_offset()
    prev_host = _core.get_endpoint_host()
    _core.set_state_mode(mode)
    _core.set_port_offset(WORKER_OFF)
    _core.set_endpoint_host(f"127.0.0.1@{WORKER_OFF}")
    if mode == "planner":
        _core.set_planner_host(f"127.0.0.1@{PLANNER_OFF}")
    _core.state_clear_all()
    yield mode
    _core.state_clear_all()
    _core.set_state_mode("inmemory")
    _core.set_planner_host("127.0.0.1")
    _core.set_port_offset(prev_off)
    _core.set_endpoint_host(prev_host)


def test_set_get_roundtrip(backend):
    from faabric_amd import _core

    kv = _core.state_get_kv("bk", f"val-{backend}", 1024)
    if backend == "planner":
        assert not kv.is_master
    else:
        assert kv.is_master
    kv.set(b"\xab" * 1024)
    assert kv.get() == b"\xab" * 1024


def test_chunked_ops(backend):
    from faabric_amd import _core

    kv = _core.state_get_kv("bk", f"chunk-{backend}", 256 * 1024)
    kv.set(b"\x00" * (256 * 1024))
    kv.set_chunk(70_000, b"HELLO")
    got = kv.get_chunk(69_998, 9)
    assert got == b"\x00\x00HELLO\x00\x00"
    kv.set_chunk(256 * 1024 - 4, b"TAIL")
    assert kv.get_chunk(256 * 1024 - 4, 4) == b"TAIL"


def test_push_pull_partial(backend):
    from faabric_amd import _core

    kv = _core.state_get_kv("bk", f"pp-{backend}", 128 * 1024)
    kv.set(b"\x01" * (128 * 1024))
    kv.push_full()
    kv.set_chunk(65_536, b"\x02" * 100)
    kv.push_partial()
    kv.pull()
    assert kv.get_chunk(65_536, 100) == b"\x02" * 100
    assert kv.get_chunk(0, 16) == b"\x01" * 16


def test_append_channel(backend):
    from faabric_amd import _core

    kv = _core.state_get_kv("bk", f"app-{backend}", 1)
    kv.append(b"one")
    kv.append(b"two")
    kv.append(b"three")
    vals = kv.get_appended(3)
    assert [bytes(v) for v in vals] == [b"one", b"two", b"three"]
    kv.clear_appended()


def test_scripted_locks(backend):
    from faabric_amd import _core

    user, key = "bk", f"lock-{backend}"
    token = _core.state_acquire_lock(user, key, 10_000)
    assert token != 0
    # Contention: held lock cannot be re-acquired
    assert _core.state_acquire_lock(user, key, 10_000) == 0
    # Scripted release: a wrong token must NOT free it
    _core.state_release_lock(user, key, token + 1)
    assert _core.state_acquire_lock(user, key, 10_000) == 0
    # Matching token frees it
    _core.state_release_lock(user, key, token)
    token2 = _core.state_acquire_lock(user, key, 10_000)
    assert token2 != 0 and token2 != token
    _core.state_release_lock(user, key, token2)


def test_lock_expiry(backend):
    from faabric_amd import _core

    user, key = "bk", f"exp-{backend}"
    token = _core.state_acquire_lock(user, key, 400)
    assert token != 0
    # Immediately contended (the 400 ms window gives slack on a loaded
    # machine between these two calls)
    assert _core.state_acquire_lock(user, key, 400) == 0
    time.sleep(1.0)
    # Expired: a new holder can take it (reference Redis expiry semantics)
    token2 = _core.state_acquire_lock(user, key, 10_000)
    assert token2 != 0
    # The stale holder's release must not free the new holder's lock
    _core.state_release_lock(user, key, token)
    assert _core.state_acquire_lock(user, key, 10_000) == 0
    _core.state_release_lock(user, key, token2)


def test_planner_store_is_shared(planner_store):
    """Two 'worker' identities see the same authoritative value through
    the planner store (the global-KV property the Redis backend gives
    the reference)."""
    from faabric_amd import _core

    _core.set_state_mode("planner")
    _core.set_planner_host(f"127.0.0.1@{PLANNER_OFF}")
    _core.set_port_offset(WORKER_OFF)
    _core.set_endpoint_host(f"127.0.0.1@{WORKER_OFF}")
    _core.state_clear_all()
    try:
        kv = _core.state_get_kv("bk", "shared", 4096)
        kv.set(b"\x5a" * 4096)
        # A "different worker": drop local caches, re-resolve from the
        # planner store
        _core.state_clear_all()
        _core.set_endpoint_host("127.0.0.1@7200")
        kv2 = _core.state_get_kv("bk", "shared", 4096)
        assert not kv2.is_master
        kv2.pull()
        assert kv2.get_chunk(0, 4096) == b"\x5a" * 4096
    finally:
        _core.state_clear_all()
        _core.set_endpoint_host(f"127.0.0.1@{WORKER_OFF}")
        _core.set_state_mode("inmemory")
