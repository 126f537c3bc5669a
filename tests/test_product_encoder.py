"""Product fixture-writer vs CPU oracle: byte-equality of encoded streams.

The product's part builder (banyandb_amd/csrc/encode.cpp) and the oracle
(oracle/bydb_oracle.c) are independent restatements of the reference write
path; their outputs must be byte-identical, and the oracle must decode the
product's streams back to the inputs.  CPU-only."""
import random

import oracle as o
from banyandb_amd import PartBuilder, VT_INT64, VT_FLOAT64


def _desc_to_oracle(d):
    """Convert a product bydb_block_desc to the oracle's bo_block_desc dict.

    The product descriptor stores header-free streams + parsed header
    fields; the oracle scan expects the column payload WITH header.  For
    byte-level checks we reconstruct the oracle column payload from the
    parsed fields (the reverse of the host-side header parse)."""
    return d


def _extract(payload, d):
    ts_stream = payload[d.ts_off - 0: d.ts_off + d.ts_len]
    f_stream = payload[d.field_off: d.field_off + d.field_len]
    return ts_stream, f_stream


def test_i64_block_streams_match_oracle():
    random.seed(42)
    for trial in range(20):
        n = random.randint(1, 700)
        style = trial % 4
        if style == 0:
            vals = [random.randint(-10 ** 12, 10 ** 12) for _ in range(n)]
        elif style == 1:
            base = random.randint(0, 10 ** 9)
            vals = [base + i * 7 + random.randint(-3, 3) for i in range(n)]
        elif style == 2:
            vals = [12345] * n
        else:
            vals = [100 + i * 50 for i in range(n)]
        ts = [1_700_000_000_000_000_000 + i * 10 ** 6 for i in range(n)]
        vers = [1] * n

        b = PartBuilder()
        b.add_block_i64(7, ts, vers, vals, 0)
        d = b.blocks()[0]
        payload = b.payload

        # oracle encodes the same values
        enc, et, first = o.int64_list_encode(vals)
        assert d.field_enc == et
        assert d.field_first == first
        ts_stream, f_stream = _extract(payload, d)
        assert f_stream == enc, f"style={style} n={n}"

        om = o.timestamps_encode(ts, vers)
        assert d.ts_enc_with_version == om["enc"]
        assert d.ts_min == om["ts_min"]
        assert d.ts_max == om["ts_max"]
        assert d.version_enc == om["version_enc"]
        assert d.version_first == om["version_first"]
        # product ts stream = ts part of the oracle payload
        assert ts_stream == om["payload"][: om["version_offset"]]

        # oracle decodes the product stream back to the inputs
        assert o.int64_list_decode(f_stream, d.field_enc, d.field_first, n) == vals


def test_f64_block_streams_match_oracle():
    random.seed(43)
    for _ in range(10):
        n = random.randint(1, 500)
        cents = [random.randint(-10 ** 7, 10 ** 7) for _ in range(n)]
        vals = [c / 100.0 for c in cents]
        ts = [1_700_000_000_000_000_000 + i * 10 ** 6 for i in range(n)]
        b = PartBuilder()
        b.add_block_f64(3, ts, [1] * n, vals, 0)
        d = b.blocks()[0]
        assert d.field_vtype == VT_FLOAT64
        ints, exp = o.float_to_decimal(vals)
        assert d.exp == exp
        enc, et, first = o.int64_list_encode(ints)
        assert d.field_enc == et
        assert d.field_first == first
        _, f_stream = _extract(b.payload, d)
        assert f_stream == enc
        # decode through the oracle float path: rebuild the column payload
        col_payload = bytes([et]) + d.exp.to_bytes(2, "big", signed=True) + \
            o.cell_encode(first) + f_stream
        assert o.column_f64_decode(col_payload, n) == vals


def test_tag_column_matches_oracle():
    values = [b"prod", b"dev", b"prod", b"staging"] * 100
    n = len(values)
    ts = [1_700_000_000_000_000_000 + i * 10 ** 6 for i in range(n)]
    b = PartBuilder()
    b.add_block_i64(1, ts, [1] * n, list(range(n)), 0)
    b.set_block_tag(values)
    d = b.blocks()[0]
    payload = b.payload
    tag_payload = payload[d.tag_off: d.tag_off + d.tag_len]
    assert tag_payload[0] == 10  # EncodeTypeDictionary
    oracle_dict = o.dictionary_encode(values)
    assert tag_payload[1:] == oracle_dict
    assert o.dictionary_decode(tag_payload[1:], n) == values
    assert o.dictionary_decode_codes(tag_payload[1:], n)[:4] == [0, 1, 0, 2]


def test_generator_matches_oracle_scan():
    """Full pipeline on CPU: product generator -> oracle scan+aggregate ==
    direct recomputation from the generator's definition."""
    b = PartBuilder()
    n_series, n_dp = 5, 3000
    t0, stride = 1_700_000_000_000_000_000, 10 ** 6
    seed = 0xB4DB
    for s in range(n_series):
        b.gen_series_i64(s, n_dp, t0, stride, base=s * 1000, ramp=1, seed=seed)
    payload = b.payload
    descs = b.blocks()
    assert len(descs) == n_series  # 3000 rows -> 1 block per series
    blocks = []
    for d in descs:
        # oracle desc: column payload includes the header bytes -> rebuild
        col = bytes([d.field_enc]) + o.cell_encode(d.field_first) + \
            payload[d.field_off: d.field_off + d.field_len]
        blocks.append(dict(
            series_id=d.series_id, count=d.count,
            ts_enc_with_version=d.ts_enc_with_version, version_enc=d.version_enc,
            ts_min=d.ts_min, ts_max=d.ts_max, version_first=d.version_first,
            ts_off=d.ts_off, ts_len=d.ts_len, ver_len=0,
            col_off=len(payload), col_len=len(col), group_code=0))
        payload = payload + col
    res = o.scan_agg(payload, blocks, VT_INT64)[0]

    # recompute from the generator definition (splitmix64 noise)
    def splitmix_stream(state):
        while True:
            state = (state + 0x9E3779B97F4A7C15) % 2 ** 64
            z = state
            z = ((z ^ (z >> 30)) * 0xBF58476D1CE4E5B9) % 2 ** 64
            z = ((z ^ (z >> 27)) * 0x94D049BB133111EB) % 2 ** 64
            yield z ^ (z >> 31)

    total = cnt = 0
    mn, mx = 2 ** 63, -2 ** 63
    for s in range(n_series):
        g = splitmix_stream(seed ^ ((s * 0x9E3779B97F4A7C15 + 1) % 2 ** 64))
        for i in range(n_dp):
            v = s * 1000 + i + (next(g) % 7) - 3
            total += v
            cnt += 1
            mn = min(mn, v)
            mx = max(mx, v)
    assert res.count == cnt
    assert res.sum_i == total
    assert res.min_i == mn
    assert res.max_i == mx


def test_time_clamp_through_oracle():
    b = PartBuilder()
    n = 1000
    t0, stride = 10 ** 18, 10 ** 6
    b.gen_series_i64(0, n, t0, stride, base=0, ramp=0, seed=1)
    payload = b.payload
    d = b.blocks()[0]
    col = bytes([d.field_enc]) + o.cell_encode(d.field_first) + \
        payload[d.field_off: d.field_off + d.field_len]
    blocks = [dict(series_id=d.series_id, count=d.count,
                   ts_enc_with_version=d.ts_enc_with_version,
                   version_enc=d.version_enc, ts_min=d.ts_min, ts_max=d.ts_max,
                   version_first=d.version_first, ts_off=d.ts_off,
                   ts_len=d.ts_len, ver_len=0, col_off=len(payload),
                   col_len=len(col), group_code=0)]
    payload += col
    # clamp to rows [100, 899]
    res = o.scan_agg(payload, blocks, VT_INT64,
                     min_ts=t0 + 100 * stride, max_ts=t0 + 899 * stride)[0]
    assert res.count == 800
