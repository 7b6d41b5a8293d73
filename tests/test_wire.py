"""Wire-format and message-model tests.

Mirrors the reference's proto round-trip coverage (tests/test/proto) and
additionally pins the protobuf wire encoding to known bytes so the format
stays interoperable with a real protobuf decoder
(reference schema: src/proto/faabric.proto).
"""

import faabric_amd as fa


def test_message_roundtrip():
    m = fa.message_factory("alice", "echo")
    m.input_data = b"\x00\x01binary\xff"
    m.output_data = "result"
    m.return_value = -7
    m.group_id = 42
    m.group_idx = 3
    m.is_mpi = True
    m.mpi_world_id = 123
    m.mpi_rank = 5
    m.mpi_world_size = 8
    m.chained_msg_ids = [11, 22, 33]
    m.int_exec_graph_details = {"mpi-msgcount-torank-0": 9}
    m.exec_graph_details = {"kind": "test"}

    dec = fa.Message.decode(m.encode())
    assert dec.id == m.id
    assert dec.app_id == m.app_id
    assert dec.user == "alice"
    assert dec.function == "echo"
    assert dec.input_data == b"\x00\x01binary\xff"
    assert dec.output_data == "result"
    assert dec.return_value == -7
    assert dec.group_id == 42
    assert dec.group_idx == 3
    assert dec.is_mpi is True
    assert dec.mpi_world_size == 8
    assert dec.chained_msg_ids == [11, 22, 33]
    assert dec.int_exec_graph_details == {"mpi-msgcount-torank-0": 9}
    assert dec.exec_graph_details == {"kind": "test"}


def test_batch_roundtrip():
    ber = fa.batch_exec_factory("bob", "work", 4)
    ber.type = fa.BatchExecuteType.THREADS
    ber.snapshot_key = "bob/work_123"
    ber.single_host_hint = True
    enc = ber.encode()
    dec = fa.BatchExecuteRequest.decode(enc)
    assert dec.app_id == ber.app_id
    assert dec.type == fa.BatchExecuteType.THREADS
    assert dec.snapshot_key == "bob/work_123"
    assert dec.single_host_hint is True
    assert len(dec.messages) == 4
    assert [m.app_idx for m in dec.messages] == [0, 1, 2, 3]
    assert fa.is_batch_exec_request_valid(dec)


def test_known_protobuf_bytes():
    # Field 1 (id) varint, field 6 (user) string: protobuf wire format
    m = fa.Message()
    m.id = 150  # classic protobuf example: 0x08 0x96 0x01
    m.user = "hi"  # tag 6<<3|2 = 0x32, len 2
    enc = m.encode()
    assert enc == bytes([0x08, 0x96, 0x01, 0x32, 0x02]) + b"hi"


def test_proto3_defaults_skipped():
    m = fa.Message()
    assert m.encode() == b""


def test_negative_int32_sign_extended():
    # protobuf encodes negative int32 as 10-byte sign-extended varint
    m = fa.Message()
    m.return_value = -1
    enc = m.encode()
    # tag for field 11 varint = 0x58
    assert enc[0] == 0x58
    assert enc[1:] == b"\xff\xff\xff\xff\xff\xff\xff\xff\xff\x01"
    dec = fa.Message.decode(enc)
    assert dec.return_value == -1


def test_gid_uniqueness():
    gids = {fa.generate_gid() for _ in range(10000)}
    assert len(gids) == 10000
    assert all(g > 0 for g in gids)


def test_ptp_mappings_roundtrip():
    pm = fa.PointToPointMappings()
    pm.app_id = 7
    pm.group_id = 9
    for i in range(3):
        entry = fa.PointToPointMapping()
        entry.host = f"10.0.0.{i}"
        entry.message_id = 100 + i
        entry.app_idx = i
        entry.group_idx = i
        entry.mpi_port = 8020 + i
        pm.mappings = pm.mappings + [entry]
    dec = fa.PointToPointMappings.decode(pm.encode())
    assert dec.app_id == 7
    assert len(dec.mappings) == 3
    assert dec.mappings[2].host == "10.0.0.2"
    assert dec.mappings[2].mpi_port == 8022


# ---- property-based fuzz (hypothesis): arbitrary field values must
# round-trip through the hand-written protobuf codec ----

from hypothesis import given, settings, strategies as st  # noqa: E402

_i32 = st.integers(min_value=-(2**31), max_value=2**31 - 1)
_u31 = st.integers(min_value=0, max_value=2**31 - 1)
_i64 = st.integers(min_value=-(2**63), max_value=2**63 - 1)
_text = st.text(max_size=40)
_blob = st.binary(max_size=256)


@settings(max_examples=80, deadline=None)
@given(
    user=_text,
    function=_text,
    input_data=_blob,
    output_data=_text,
    return_value=_i32,
    group_id=_i32,
    group_idx=_i32,
    mpi_rank=_i32,
    timestamp=_i64,
    chained=st.lists(_u31, max_size=6),
    int_details=st.dictionaries(_text, _i32, max_size=4),
    str_details=st.dictionaries(_text, _text, max_size=4),
)
def test_message_fuzz_roundtrip(user, function, input_data, output_data,
                                return_value, group_id, group_idx,
                                mpi_rank, timestamp, chained, int_details,
                                str_details):
    m = fa.message_factory(user, function)
    m.input_data = input_data
    m.output_data = output_data
    m.return_value = return_value
    m.group_id = group_id
    m.group_idx = group_idx
    m.mpi_rank = mpi_rank
    m.finish_timestamp = timestamp
    m.chained_msg_ids = chained
    m.int_exec_graph_details = int_details
    m.exec_graph_details = str_details

    dec = fa.Message.decode(m.encode())
    assert dec.user == user
    assert dec.function == function
    assert dec.input_data == input_data
    assert dec.output_data == output_data
    assert dec.return_value == return_value
    assert dec.group_id == group_id
    assert dec.group_idx == group_idx
    assert dec.mpi_rank == mpi_rank
    assert dec.finish_timestamp == timestamp
    assert dec.chained_msg_ids == chained
    assert dec.int_exec_graph_details == int_details
    assert dec.exec_graph_details == str_details


@settings(max_examples=40, deadline=None)
@given(old=st.binary(max_size=20000), new=st.binary(max_size=20000))
def test_delta_codec_fuzz(old, new):
    from faabric_amd import _core

    delta = _core.delta_encode(old, new)
    assert _core.delta_apply(old, delta) == new
