"""Snapshot typed-merge semantics on the host path (reference coverage:
tests/test/util/test_snapshot.cpp — diffs must match bit-for-bit)."""

import struct

from faabric_amd import _core

Int = _core.SnapshotDataType.Int
Long = _core.SnapshotDataType.Long
Float = _core.SnapshotDataType.Float
Double = _core.SnapshotDataType.Double
Raw = _core.SnapshotDataType.Raw

Sum = _core.SnapshotMergeOperation.Sum
Product = _core.SnapshotMergeOperation.Product
Subtract = _core.SnapshotMergeOperation.Subtract
Max = _core.SnapshotMergeOperation.Max
Min = _core.SnapshotMergeOperation.Min
XOR = _core.SnapshotMergeOperation.XOR
Bytewise = _core.SnapshotMergeOperation.Bytewise


def make_snap(data, max_size=0):
    return _core.SnapshotData(data, max_size)


def test_bytewise_diff_chunking():
    base = bytes(8192)
    snap = make_snap(base)
    snap.add_merge_region(0, 0, Raw, Bytewise)
    updated = bytearray(base)
    updated[100:104] = b"abcd"
    updated[5000] = 0xFF
    diffs = snap.diff_with_memory(bytes(updated))
    assert len(diffs) == 2
    assert diffs[0].offset == 100 and diffs[0].data == b"abcd"
    assert diffs[1].offset == 5000 and diffs[1].data == b"\xff"
    snap.apply_diffs(diffs)
    assert snap.get_data() == bytes(updated)


def test_sum_sends_delta_and_merges():
    base = struct.pack("<4i", 10, 20, 30, 40)
    snap = make_snap(base)
    snap.add_merge_region(0, 16, Int, Sum)

    # Two "threads" diff against the same baseline
    up1 = struct.pack("<4i", 15, 20, 30, 40)  # +5 on [0]
    up2 = struct.pack("<4i", 10, 22, 30, 47)  # +2 on [1], +7 on [3]
    diffs1 = snap.diff_with_memory(up1)
    diffs2 = snap.diff_with_memory(up2)
    assert len(diffs1) == 1 and len(diffs2) == 2
    assert struct.unpack("<i", diffs1[0].data)[0] == 5  # the DELTA

    snap.queue_diffs(diffs1)
    snap.queue_diffs(diffs2)
    n = snap.write_queued_diffs()
    assert n == 3
    assert struct.unpack("<4i", snap.get_data()) == (15, 22, 30, 47)


def test_subtract_product_max_min():
    base = struct.pack("<4i", 100, 3, 50, 50)
    snap = make_snap(base)
    snap.add_merge_region(0, 4, Int, Subtract)
    snap.add_merge_region(4, 4, Int, Product)
    snap.add_merge_region(8, 4, Int, Max)
    snap.add_merge_region(12, 4, Int, Min)

    updated = struct.pack("<4i", 90, 12, 70, 30)
    diffs = snap.diff_with_memory(updated)
    by_off = {d.offset: d for d in diffs}
    assert struct.unpack("<i", by_off[0].data)[0] == 10   # 100-90 delta
    assert struct.unpack("<i", by_off[4].data)[0] == 4    # 12/3 factor
    assert struct.unpack("<i", by_off[8].data)[0] == 70   # max VALUE
    assert struct.unpack("<i", by_off[12].data)[0] == 30  # min VALUE

    snap.apply_diffs(diffs)
    assert struct.unpack("<4i", snap.get_data()) == (90, 12, 70, 30)

    # Applying a max diff lower than current is a no-op
    worse = by_off[8]
    worse.data = struct.pack("<i", 60)
    snap.apply_diffs([worse])
    assert struct.unpack("<4i", snap.get_data())[2] == 70


def test_float_double_sum():
    base = struct.pack("<2f", 1.5, 2.5) + struct.pack("<d", 10.0)
    snap = make_snap(base)
    snap.add_merge_region(0, 8, Float, Sum)
    snap.add_merge_region(8, 8, Double, Sum)
    updated = struct.pack("<2f", 2.0, 2.5) + struct.pack("<d", 12.25)
    diffs = snap.diff_with_memory(updated)
    snap.apply_diffs(diffs)
    f0, f1 = struct.unpack("<2f", snap.get_data()[:8])
    (d0,) = struct.unpack("<d", snap.get_data()[8:])
    assert abs(f0 - 2.0) < 1e-6 and f1 == 2.5 and d0 == 12.25


def test_xor_region_roundtrip():
    base = bytes(range(256)) * 32  # 8192 bytes
    snap = make_snap(base)
    snap.add_merge_region(0, 0, Raw, XOR)
    updated = bytearray(base)
    updated[4097] ^= 0x55
    diffs = snap.diff_with_memory(bytes(updated))
    assert len(diffs) == 1
    assert diffs[0].operation == XOR
    snap.apply_diffs(diffs)
    assert snap.get_data() == bytes(updated)


def test_fill_gaps_with_bytewise():
    base = bytes(4 * 4096)
    snap = make_snap(base)
    snap.add_merge_region(0, 16, Int, Sum)
    snap.fill_gaps_with_bytewise_regions()
    updated = bytearray(base)
    updated[9000] = 1  # outside the typed region
    diffs = snap.diff_with_memory(bytes(updated))
    # Gap regions default to XOR (DIFFING_MODE=xor): page-granular diffs
    assert any(
        d.offset <= 9000 < d.offset + len(d.data) for d in diffs
    )
    snap.apply_diffs(diffs)
    assert snap.get_data() == bytes(updated)


def test_extension_ships_bytewise():
    snap = make_snap(bytes(4096), max_size=8192)
    updated = bytes(4096) + b"Z" * 100
    diffs = snap.diff_with_memory(updated)
    assert diffs[0].offset == 4096
    assert diffs[0].data == b"Z" * 100
    snap.apply_diffs(diffs)
    assert snap.size == 4196


def test_registry_lifecycle():
    snap = make_snap(b"x" * 64)
    _core.snapshot_register("t/reg_1", snap)
    assert _core.snapshot_exists("t/reg_1")
    got = _core.snapshot_get("t/reg_1")
    assert got.get_data() == b"x" * 64
    _core.snapshot_delete("t/reg_1")
    assert not _core.snapshot_exists("t/reg_1")


# ---- property test: typed merges vs a pure-python model (reference
# semantics: util/snapshot.h calculateDiffValue/applyDiffValue — Sum/
# Subtract/Product ship the delta, Max/Min ship the value) ----

from hypothesis import given, settings, strategies as st  # noqa: E402

_vals = st.lists(
    st.integers(min_value=-(2**30), max_value=2**30 - 1),
    min_size=8, max_size=8,
)


@settings(max_examples=60, deadline=None)
@given(base=_vals, up1=_vals, up2=_vals,
       op=st.sampled_from(["sum", "max", "min"]))
def test_int_merge_matrix_vs_model(base, up1, up2, op):
    opmap = {"sum": Sum, "max": Max, "min": Min}
    raw = struct.pack("<8i", *base)
    snap = make_snap(raw)
    snap.add_merge_region(0, 32, Int, opmap[op])

    diffs = [snap.diff_with_memory(struct.pack("<8i", *u))
             for u in (up1, up2)]
    for d in diffs:
        snap.queue_diffs(d)
    snap.write_queued_diffs()
    got = list(struct.unpack("<8i", snap.get_data()))

    # Pure-python model of the reference semantics
    expect = list(base)
    for i in range(8):
        if op == "sum":
            # Each thread contributes its delta; overflow wraps at i32
            v = base[i] + (up1[i] - base[i]) + (up2[i] - base[i])
            v = (v + 2**31) % 2**32 - 2**31
            expect[i] = v
        elif op == "max":
            # Only values differing from base ship; merge keeps the max
            for u in (up1, up2):
                if u[i] != base[i] and u[i] > expect[i]:
                    expect[i] = u[i]
        elif op == "min":
            for u in (up1, up2):
                if u[i] != base[i] and u[i] < expect[i]:
                    expect[i] = u[i]
    assert got == expect, (op, base, up1, up2)
