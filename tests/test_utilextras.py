"""Util parity extras: page-delta codec (xor+zlib), PROF timers, segfault
dirty tracker (reference: src/util/delta.cpp, util/timing.h,
src/util/dirty.cpp)."""

import os
import pytest
import random

from faabric_amd import _core


def test_delta_roundtrip_xor_zlib():
    random.seed(3)
    old = bytes(random.getrandbits(8) for _ in range(64 * 1024))
    new = bytearray(old)
    # Mutate a few scattered pages + extend
    for page in (0, 5, 11):
        new[page * 4096 + 100] ^= 0xFF
    new.extend(b"tail-extension" * 10)
    new = bytes(new)

    delta = _core.delta_encode(old, new, "pages=4096;xor;zlib=1")
    assert len(delta) < len(new) // 4  # sparse change compresses well
    restored = _core.delta_apply(old, delta)
    assert restored == new


def test_delta_roundtrip_xor_zstd():
    """Default codec is the reference's pages=4096;xor;zstd=1
    (reference: src/util/delta.cpp:15-57, config.cpp:27)."""
    random.seed(7)
    old = bytes(random.getrandbits(8) for _ in range(64 * 1024))
    new = bytearray(old)
    for page in (1, 7, 13):
        new[page * 4096 + 50] ^= 0xAA
    new = bytes(new)

    delta = _core.delta_encode(old, new, "pages=4096;xor;zstd=1")
    assert len(delta) < len(new) // 4
    assert _core.delta_apply(old, delta) == new

    # Default config string spells zstd like the reference
    assert "zstd=1" in _core.delta_default_config()


def test_delta_no_compress_no_xor():
    old = b"a" * 8192
    new = b"a" * 4096 + b"b" * 4096
    delta = _core.delta_encode(old, new, "pages=4096")
    assert _core.delta_apply(old, delta) == new


def test_delta_shrink_grow():
    old = b"x" * 10000
    new = b"y" * 3000
    delta = _core.delta_encode(old, new)
    assert _core.delta_apply(old, delta) == new


def test_segfault_dirty_tracker():
    assert _core._selftest_segfault_tracker()


def test_prof_timers():
    _core.prof_clear()
    summary = _core.prof_summary()
    assert "PROF totals" in summary


def test_pin_thread():
    core = _core.pin_thread_to_free_cpu()
    assert core >= 0 or os.cpu_count() is None


def test_uffd_dirty_tracker():
    """userfaultfd write-protect tracking (reference: src/util/dirty.cpp
    uffd modes); skipped where the kernel lacks uffd-wp."""
    result = _core._selftest_uffd_tracker()
    if result is None:
        pytest.skip("kernel lacks uffd write-protect")
    assert result


def test_softpte_dirty_tracker():
    """Soft-dirty PTE tracking via /proc/self/clear_refs + pagemap bit 55
    (reference: util/dirty.h:58-90 SoftPTEDirtyTracker); skipped where the
    kernel lacks CONFIG_MEM_SOFT_DIRTY."""
    result = _core._selftest_softpte_tracker()
    if result is None:
        pytest.skip("kernel lacks soft-dirty PTE support")
    assert result
