"""GPU table compiler tests (CPU-only): layout, determinism, indices."""

import struct

from examples.protos import ALL_FDPS
from ggrmcp_amd.descriptors.loader import build_pool, extract_method_infos
from ggrmcp_amd.engine import tables as T


def _tools():
    pool = build_pool(ALL_FDPS)
    infos = extract_method_infos(ALL_FDPS, pool, compat_names=False)
    return {mi.tool_name(): mi for mi in infos}


def test_compile_basic():
    ct = T.compile_tables(_tools())
    assert ct.n_tools == 5
    assert ct.n_msgs >= 7  # request/response messages + nested + Timestamp
    assert len(ct.tool_table) == ct.n_tools * T.TOOL_ENTRY_SIZE
    assert len(ct.msg_table) == ct.n_msgs * T.MSG_ENTRY_SIZE
    assert "hello.HelloRequest" in ct.msg_index
    assert "google.protobuf.Timestamp" in ct.msg_index
    assert ct.tool_index["hello_helloservice_sayhello"] >= 0
    # tool order is sorted
    assert ct.tool_order == sorted(ct.tool_order)


def test_deterministic():
    a, b = T.compile_tables(_tools()), T.compile_tables(_tools())
    assert a.checksum() == b.checksum()
    assert a.blobs() == b.blobs()


def test_field_entry_contents():
    ct = T.compile_tables(_tools())
    mi = ct.msg_index["complex.UserProfile"]
    field_start, field_count, wkt, _ = struct.unpack_from(
        T.MSG_ENTRY_FMT, ct.msg_table, mi * T.MSG_ENTRY_SIZE
    )
    assert wkt == T.WKT_NONE
    assert field_count == 8
    fields = {}
    for i in range(field_count):
        rec = struct.unpack_from(
            T.FIELD_ENTRY_FMT, ct.field_table, (field_start + i) * T.FIELD_ENTRY_SIZE
        )
        (hash_json, hash_orig, number, name_off, json_off, name_len, json_len,
         sub, kind, flags, oneof_id, _pad) = rec
        name = ct.name_blob[name_off : name_off + name_len].decode()
        fields[name] = dict(number=number, kind=kind, flags=flags, sub=sub,
                            json=ct.name_blob[json_off : json_off + json_len].decode(),
                            hash_json=hash_json)
    assert fields["user_id"]["json"] == "userId"
    assert fields["user_id"]["hash_json"] == T.fnv1a64(b"userId")
    assert fields["score"]["kind"] == 3  # TYPE_INT64
    assert fields["tags"]["flags"] & T.F_REPEATED
    assert fields["status"]["kind"] == 14  # TYPE_ENUM
    ts_idx = fields["created_at"]["sub"]
    ts = struct.unpack_from(T.MSG_ENTRY_FMT, ct.msg_table, ts_idx * T.MSG_ENTRY_SIZE)
    assert ts[2] == T.WKT_TIMESTAMP


def test_oneof_and_map_flags():
    ct = T.compile_tables(_tools())
    mi = ct.msg_index["complex.Document"]
    field_start, field_count, _, _ = struct.unpack_from(
        T.MSG_ENTRY_FMT, ct.msg_table, mi * T.MSG_ENTRY_SIZE
    )
    flags_by_name = {}
    for i in range(field_count):
        rec = struct.unpack_from(
            T.FIELD_ENTRY_FMT, ct.field_table, (field_start + i) * T.FIELD_ENTRY_SIZE
        )
        name = ct.name_blob[rec[3] : rec[3] + rec[5]].decode()
        flags_by_name[name] = rec[9]
    assert flags_by_name["text"] & T.F_ONEOF
    assert flags_by_name["binary"] & T.F_ONEOF
    assert flags_by_name["metadata"] & T.F_MAP
