"""Placement-policy tables (reference coverage:
tests/test/batch-scheduler per-policy placement tests)."""

import pytest

from faabric_amd import _core

A = "10.0.0.1"
B = "10.0.0.2"
C = "10.0.0.3"
EVICT = "E.VI.CT.ME"


def decide(mode, hosts, n, **kw):
    return _core.test_make_scheduling_decision(mode, hosts, n, **kw)


def test_binpack_prefers_most_free():
    d = decide("bin-pack", [(A, 4, 2), (B, 4, 0)], 3)
    assert d.hosts == [B, B, B]


def test_binpack_overflows_in_capacity_order():
    d = decide("bin-pack", [(A, 4, 3), (B, 4, 1), (C, 2, 0)], 5)
    # B has 3 free, C has 2 free, A has 1 free
    assert d.hosts == [B, B, B, C, C]


def test_binpack_tie_breaks_on_larger_host():
    d = decide("bin-pack", [(A, 2, 0), (B, 4, 2)], 2)
    # Both have 2 free; the larger host wins
    assert d.hosts == [B, B]


def test_binpack_not_enough_slots():
    d = decide("bin-pack", [(A, 2, 1), (B, 2, 2)], 3)
    assert d.app_id == _core.NOT_ENOUGH_SLOTS()


def test_binpack_scale_change_colocates():
    # App 7 already runs 2 messages on A; a scale-change prefers A even
    # though B has more free slots
    d = decide(
        "bin-pack",
        [(A, 4, 2), (B, 8, 0)],
        1,
        app_id=7,
        in_flight=[(7, [A, A])],
    )
    assert d.hosts == [A]


def test_binpack_migration_consolidates():
    # App 9 split across A and B; after returning its own slots everything
    # fits on one host (ties break to the larger ip, like the reference)
    d = decide(
        "bin-pack",
        [(A, 4, 1), (B, 4, 1)],
        2,
        app_id=9,
        migration=True,
        in_flight=[(9, [A, B])],
    )
    assert sorted(d.hosts) == [B, B]
    # The message already on B stays put (minimised migrations)
    assert d.hosts[1] == B


def test_binpack_migration_declined_when_no_better():
    # One slot per host: the re-scheduled placement still spans two hosts,
    # so the locality does not improve
    d = decide(
        "bin-pack",
        [(A, 1, 1), (B, 1, 1)],
        2,
        app_id=9,
        migration=True,
        in_flight=[(9, [A, B])],
    )
    assert d.app_id == _core.DO_NOT_MIGRATE()


def test_spot_avoids_evicted_vm():
    # B is doomed: despite having the most slots it must not be used
    d = decide("spot", [(A, 2, 0), (EVICT, 8, 0)], 2)
    assert d.hosts == [A, A]


def test_spot_freezes_when_no_capacity():
    # App 5 runs on the doomed VM and nothing else fits
    d = decide(
        "spot",
        [(EVICT, 4, 2), (A, 1, 1)],
        2,
        app_id=5,
        migration=True,
        in_flight=[(5, [EVICT, EVICT])],
    )
    assert d.app_id == _core.MUST_FREEZE()


def test_spot_migrates_off_evicted_vm():
    d = decide(
        "spot",
        [(EVICT, 4, 2), (A, 4, 0)],
        2,
        app_id=5,
        migration=True,
        in_flight=[(5, [EVICT, EVICT])],
    )
    assert sorted(d.hosts) == [A, A]


def test_spot_no_migration_when_not_on_evicted_vm():
    d = decide(
        "spot",
        [(EVICT, 4, 0), (A, 4, 2), (B, 4, 0)],
        2,
        app_id=5,
        migration=True,
        in_flight=[(5, [A, A])],
    )
    assert d.app_id == _core.DO_NOT_MIGRATE()


def test_compact_multi_tenant_filter():
    # Another user's app (different subType simulation via in_flight of a
    # different app) occupies B; compact still packs onto the emptiest of
    # the remaining hosts
    d = decide("compact", [(A, 4, 0), (B, 4, 1)], 2)
    assert d.hosts == [A, A]


def test_compact_migration_frees_hosts():
    # App 3 split 1+1; consolidating onto one host frees the other
    d = decide(
        "compact",
        [(A, 4, 1), (B, 4, 1)],
        2,
        app_id=3,
        migration=True,
        in_flight=[(3, [A, B])],
    )
    assert len(set(d.hosts)) == 1
