"""Loud error paths: corrupt or unsupported inputs must surface as device
errors at finalize, never silently wrong results."""
import ctypes as C

import pytest

import banyandb_amd as ba
from banyandb_amd import (PartBuilder, Session, VT_INT64, VT_FLOAT64,
                          AGG_SUM, AGG_COUNT)

pytestmark = pytest.mark.gpu

T0 = 1_700_000_000_000_000_000
MS = 10 ** 6


def _desc_copy(d):
    new = ba.BlockDesc()
    C.memmove(C.byref(new), C.byref(d), C.sizeof(ba.BlockDesc))
    return new


def _upload_raw(payload: bytes, descs):
    s = Session(0)
    lib = ba.lib()
    s._ck(lib.bydb_part_reserve(s._h, len(payload), len(descs)))
    buf = (C.c_uint8 * len(payload)).from_buffer_copy(payload)
    arr = (ba.BlockDesc * len(descs))(*descs)
    s._ck(lib.bydb_part_append(s._h, buf, len(payload), arr, len(descs)))
    return s


def test_unsupported_encode_type_fails_loud():
    """Unknown encode byte reaches the device and is flagged at finalize."""
    b = PartBuilder()
    ts = [T0 + i * MS for i in range(64)]
    b.add_block_i64(1, ts, [1] * 64, list(range(0, 6400, 100)))
    d = _desc_copy(b.blocks()[0])
    d.field_enc = 77  # not an encode type
    s = _upload_raw(b.payload, [d])
    s.configure(VT_INT64, [AGG_SUM, AGG_COUNT])
    s.consume()
    with pytest.raises(RuntimeError, match="decode error"):
        s.finalize()
    s.close()


def test_fake_plain_field_fails_loud_at_upload():
    """A descriptor claiming Plain over a non-Plain stream is rejected by
    the host normalization at part registration."""
    b = PartBuilder()
    ts = [T0 + i * MS for i in range(64)]
    b.add_block_i64(1, ts, [1] * 64, list(range(0, 6400, 100)))
    d = _desc_copy(b.blocks()[0])
    d.field_enc = 9  # Plain claimed, stream is a delta varint stream
    with pytest.raises(RuntimeError, match="normalization failed"):
        _upload_raw(b.payload, [d])


def test_descending_timestamps_fail_loud():
    b = PartBuilder()
    ts = [T0 + i * MS for i in range(64)]
    b.add_block_i64(1, ts, [1] * 64, list(range(64)))
    d = _desc_copy(b.blocks()[0])
    d.ts_min, d.ts_max = d.ts_max, d.ts_min  # descending metadata
    s = _upload_raw(b.payload, [d])
    s.configure(VT_INT64, [AGG_SUM, AGG_COUNT])
    s.consume()
    with pytest.raises(RuntimeError, match="decode error"):
        s.finalize()
    s.close()


def test_corrupt_varint_stream_fails_loud():
    b = PartBuilder()
    import random
    rng = random.Random(3)
    ts = [T0 + i * MS for i in range(300)]
    b.add_block_i64(1, ts, [1] * 300,
                    [rng.randint(-2**61, 2**61) for _ in range(300)])
    d = b.blocks()[0]
    payload = bytearray(b.payload)
    # make a 64-byte run of continuation bytes inside the field stream
    for k in range(70):
        payload[d.field_off + 16 + k] |= 0x80
    s = _upload_raw(bytes(payload), [_desc_copy(d)])
    s.configure(VT_INT64, [AGG_SUM, AGG_COUNT])
    s.consume()
    with pytest.raises(RuntimeError, match="decode error"):
        s.finalize()
    s.close()


def test_sticky_error_then_reset_recovers():
    b = PartBuilder()
    ts = [T0 + i * MS for i in range(64)]
    b.add_block_i64(1, ts, [1] * 64, [7] * 64)
    d_bad = _desc_copy(b.blocks()[0])
    d_bad.field_enc = 77
    s = _upload_raw(b.payload, [d_bad])
    s.configure(VT_INT64, [AGG_SUM, AGG_COUNT])
    s.consume()
    with pytest.raises(RuntimeError):
        s.finalize()
    # a reset + good part recovers the session
    s._ck(ba.lib().bydb_part_reserve(s._h, b.payload_len, 1))
    buf = (C.c_uint8 * b.payload_len).from_buffer_copy(b.payload)
    arr = (ba.BlockDesc * 1)(_desc_copy(b.blocks()[0]))
    s._ck(ba.lib().bydb_part_append(s._h, buf, b.payload_len, arr, 1))
    s.reset()
    s.consume()
    g = s.finalize()[0]
    assert g.count == 64 and g.sum_i == 7 * 64
    s.close()


def test_float_exp_below_session_fails_loud():
    """A block whose decimal exponent is BELOW the session's cannot be
    represented in the session's mantissa domain — loud device error.
    (Blocks ABOVE the session exponent rescale correctly; see
    test_gpu_parity-style mixed-exponent coverage below.)"""
    from banyandb_amd import VT_FLOAT64
    b = PartBuilder()
    n = 128
    ts = [T0 + i * MS for i in range(n)]
    b.add_block_f64(1, ts, [1] * n,
                    [i / 100.0 for i in range(1, 2 * n, 2)])  # exp -2
    d = b.blocks()[0]
    assert d.exp == -2
    s = Session(0)
    s.upload_part(b)
    s.configure(VT_FLOAT64, [AGG_SUM, AGG_COUNT], float_exp=-1)
    s.consume()
    with pytest.raises(RuntimeError, match="decode error 5"):
        s.finalize()
    s.close()


def test_float_mixed_exponent_blocks_rescale():
    """Blocks at different decimal exponents fold correctly against a
    session configured at the part's minimum exponent: the kernel
    rescales each block's mantissa partials by 10^(block_exp - cfg_exp)
    (the oracle restores per block natively)."""
    import math
    from banyandb_amd import VT_FLOAT64, AGG_MIN, AGG_MAX
    from helpers import oracle_scan
    b = PartBuilder()
    n = 200
    ts = [T0 + i * MS for i in range(n)]
    b.add_block_f64(1, ts, [1] * n,
                    [i / 100.0 for i in range(1, 2 * n, 2)])   # exp -2
    b.add_block_f64(2, ts, [1] * n,
                    [i / 10.0 for i in range(1, n + 1)])       # exp -1
    b.add_block_f64(3, ts, [1] * n,
                    [float(i + 5) for i in range(n)])          # exp 0
    exps = sorted(d.exp for d in b.blocks())
    assert exps == [-2, -1, 0]
    orc = oracle_scan(b, VT_FLOAT64)[0]
    s = Session(0)
    s.upload_part(b)
    s.configure(VT_FLOAT64, [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX],
                float_exp=-2)
    s.consume()
    g = s.finalize()[0]
    s.close()
    assert g.count == orc.count == 3 * n
    assert math.isclose(g.sum_f, orc.sum_f, rel_tol=1e-9)
    assert math.isclose(g.min_f, orc.min_f, rel_tol=1e-12)
    assert math.isclose(g.max_f, orc.max_f, rel_tol=1e-12)


def test_f64_mantissa_sum_overflow_is_loud():
    """A float64 block whose mantissa sum would exceed int64 errs loudly
    (DERR_F64_SUM_OVF) instead of returning a silently wrapped sum
    (the reference's per-row float adds never wrap).  Large
    17-digit mantissas x 8192 rows overflow 2^63."""
    n = 8192
    ts = [10 ** 18 + i * 10 ** 6 for i in range(n)]
    base = 1234567890123456.0   # mantissa 1.23e15 (no trailing zeros)
    b = PartBuilder()
    b.add_block_f64(1, ts, [1] * n, [base + i for i in range(n)])
    s = Session(0)
    s.upload_part(b)
    s.configure(VT_FLOAT64, [AGG_SUM, AGG_COUNT], float_exp=0)
    s.consume()
    with pytest.raises(RuntimeError, match="device decode error 6"):
        s.finalize()
    s.close()
