"""Raw columnar wire-frame codec parity (CPU-only).

The reference pins this format with its ONE frozen-bytes golden
(pkg/query/vectorized/frame/golden_test.go): the golden batch below is the
reference's buildGoldenBatch (:129-160), and the expected byte string is
hand-derived, byte by byte, from encode.go:42-170 + the measure wire
numbering (golden_test.go:38-121) + the TagValue proto layout
(api/proto/banyandb/model/v1/common.proto:48-58)."""
import struct

from banyandb_amd.frame import (FrameBuilder, FrameReader, ROLE_TIMESTAMP,
                                ROLE_TAG, ROLE_FIELD, TYPE_I64)


def build_golden():
    fb = FrameBuilder(3)
    fb.add_i64(ROLE_TIMESTAMP, "ts", "", [1000, -2000, 3000])
    fb.add_str(ROLE_TAG, "svc", "meta", [b"checkout", None, b""])
    fb.add_bytes(ROLE_TAG, "raw", "meta", [b"\xDE\xAD", b"", None])
    fb.add_tagvalue_str(ROLE_TAG, "tv", "meta", [b"v0", None, b"v2"])
    return fb.finish()


def expected_golden() -> bytes:
    out = bytearray()
    out += b"\x00VFR"          # magic (leading 0x00 + 'VFR')
    out += b"\x03"             # wire version 3
    out += b"\x03\x04"         # uvarint nrows=3, ncols=4
    # col 0: ts — role 1, type 1 (i64)
    out += b"\x01\x01" + b"\x02ts" + b"\x00"
    out += b"\x00"             # bitmap: no nulls
    out += struct.pack("<q", 1000) + struct.pack("<q", -2000) + struct.pack("<q", 3000)
    # col 1: svc — role 5 (tag), type 3 (str)
    out += b"\x05\x03" + b"\x03svc" + b"\x04meta"
    out += b"\x02"             # bitmap: row 1 null
    out += b"\x08checkout" + b"\x00" + b"\x00"
    # col 2: raw — role 5, type 4 (bytes)
    out += b"\x05\x04" + b"\x03raw" + b"\x04meta"
    out += b"\x04"             # bitmap: row 2 null
    out += b"\x02\xDE\xAD" + b"\x00" + b"\x00"
    # col 3: tv — role 5, type 5 (TagValue)
    out += b"\x05\x05" + b"\x02tv" + b"\x04meta"
    out += b"\x02"             # bitmap: row 1 null
    # TagValue{str:{value:"v0"}} proto: field2 LEN(0x12) len4 [Str: field1
    # LEN(0x0A) len2 "v0"]
    out += b"\x06\x12\x04\x0a\x02v0" + b"\x00" + b"\x06\x12\x04\x0a\x02v2"
    return bytes(out)


def test_golden_bytes():
    assert build_golden() == expected_golden()


def test_golden_committed_fixture():
    import os
    path = os.path.join(os.path.dirname(__file__), "golden", "frame_golden.hex")
    want = bytes.fromhex(open(path).read().strip())
    assert build_golden() == want


def test_roundtrip_decode():
    data = build_golden()
    r = FrameReader(data)
    assert r.nrows == 3 and r.ncols == 4
    assert r.col_info(0) == (1, 1, "ts", "")
    assert r.col_i64(0) == [1000, -2000, 3000]
    assert r.col_nulls(0) == [False, False, False]
    assert r.col_info(1) == (5, 3, "svc", "meta")
    assert r.col_var(1) == [b"checkout", None, b""]
    assert r.col_var(2) == [b"\xDE\xAD", b"", None]
    assert r.col_nulls(3) == [False, True, False]


def test_numeric_and_fieldvalue_frames():
    fb = FrameBuilder(4)
    fb.add_i64(4, "shard_id", "", [0, 1, 2, 3])                 # RoleShardID
    fb.add_f64(ROLE_FIELD, "latency", "", [1.5, -0.25, 0.0, 2.0 ** 53])
    fb.add_tagvalue_int(ROLE_TAG, "code", "meta", [200, -1, 404, 500],
                        nulls=[0, 1, 0, 0])
    fb.add_tagvalue_str(ROLE_FIELD, "svc", "", [b"a", b"bb", None, b""],
                        field_value=True)
    data = fb.finish()
    r = FrameReader(data)
    assert r.nrows == 4 and r.ncols == 4
    assert r.col_i64(0) == [0, 1, 2, 3]
    f = r.col_i64(1)
    assert struct.unpack("<d", struct.pack("<q", f[0]))[0] == 1.5
    # TagValue int cells: field 4 LEN [Int field1 varint]
    cells = r.col_var(2)
    assert cells[0] == b"\x22\x03\x08\xc8\x01"  # TagValue{int:{value:200}}
    assert cells[1] is None
    # FieldValue str cell: field 2 LEN [Str field1 LEN bytes]
    cells = r.col_var(3)
    assert cells[0] == b"\x12\x03\x0a\x01a"


def test_bad_magic_fails_loud():
    data = bytearray(build_golden())
    data[0] = 0xFF
    try:
        FrameReader(bytes(data))
        raise AssertionError("bad magic must fail")
    except ValueError as e:
        assert "magic" in str(e)


def test_emit_group_results_as_frame():
    """The raw_emit slot: aggregation group rows emitted straight into the
    wire frame (shard_id + tag + value columns, AggModeMap shape —
    vectorized/measure/raw_emit.go)."""
    groups = [("svc_a", 100, 7), ("svc_b", -5, 3)]
    fb = FrameBuilder(len(groups))
    fb.add_i64(4, "shard_id", "", [0] * len(groups))
    fb.add_str(ROLE_TAG, "service_id", "meta", [g[0].encode() for g in groups])
    fb.add_i64(ROLE_FIELD, "value", "", [g[1] for g in groups])
    fb.add_i64(ROLE_FIELD, "value__agg_count", "", [g[2] for g in groups])
    data = fb.finish()
    r = FrameReader(data)
    assert r.ncols == 4
    assert r.col_var(1) == [b"svc_a", b"svc_b"]
    assert r.col_i64(2) == [100, -5]
    assert r.col_i64(3) == [7, 3]
