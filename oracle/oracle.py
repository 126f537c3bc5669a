"""ctypes wrapper over the CPU oracle (oracle/libbydb_oracle.so).

TEST INFRASTRUCTURE ONLY — see oracle/bydb_oracle.h.  Importable only from
tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg.
"""
import ctypes as C
import os
import subprocess

INT64_MIN = -(2 ** 63)
INT64_MAX = 2 ** 63 - 1

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "libbydb_oracle.so")


def _ensure_built():
    if not os.path.exists(_SO) or os.path.getmtime(_SO) < os.path.getmtime(
        os.path.join(_DIR, "bydb_oracle.c")
    ):
        subprocess.run(["make", "-C", _DIR], check=True, capture_output=True)


_ensure_built()
_L = C.CDLL(_SO)

i64p = C.POINTER(C.c_int64)
u8p = C.POINTER(C.c_uint8)
u32p = C.POINTER(C.c_uint32)
f64p = C.POINTER(C.c_double)

_L.bo_varint64_list_encode.restype = C.c_size_t
_L.bo_varint64_list_encode.argtypes = [u8p, i64p, C.c_int64]
_L.bo_varint64_list_decode.restype = C.c_int
_L.bo_varint64_list_decode.argtypes = [i64p, C.c_int64, u8p, C.c_size_t, C.POINTER(C.c_size_t)]
_L.bo_int64_list_encode.restype = C.c_int
_L.bo_int64_list_encode.argtypes = [u8p, C.c_size_t, i64p, C.c_int64,
                                    C.POINTER(C.c_size_t), C.POINTER(C.c_uint8), i64p]
_L.bo_int64_list_decode.restype = C.c_int
_L.bo_int64_list_decode.argtypes = [i64p, u8p, C.c_size_t, C.c_uint8, C.c_int64, C.c_int64]
_L.bo_float_list_to_decimal.restype = C.c_int
_L.bo_float_list_to_decimal.argtypes = [f64p, C.c_int64, i64p, C.POINTER(C.c_int16)]
_L.bo_decimal_to_float_list.restype = C.c_int
_L.bo_decimal_to_float_list.argtypes = [f64p, i64p, C.c_int16, C.c_int64]
_L.bo_go_pow10.restype = C.c_double
_L.bo_go_pow10.argtypes = [C.c_int]
_L.bo_cell_i64_to_bytes.argtypes = [u8p, C.c_int64]
_L.bo_cell_bytes_to_i64.restype = C.c_int64
_L.bo_cell_bytes_to_i64.argtypes = [u8p]
_L.bo_column_i64_encode.restype = C.c_int
_L.bo_column_i64_encode.argtypes = [u8p, C.c_size_t, i64p, C.c_int64, C.POINTER(C.c_size_t)]
_L.bo_column_i64_decode.restype = C.c_int
_L.bo_column_i64_decode.argtypes = [i64p, u8p, C.c_size_t, C.c_int64]
_L.bo_column_f64_encode.restype = C.c_int
_L.bo_column_f64_encode.argtypes = [u8p, C.c_size_t, f64p, C.c_int64, C.POINTER(C.c_size_t)]
_L.bo_column_f64_decode.restype = C.c_int
_L.bo_column_f64_decode.argtypes = [f64p, u8p, C.c_size_t, C.c_int64]
_L.bo_timestamps_encode.restype = C.c_int
_L.bo_timestamps_encode.argtypes = [u8p, C.c_size_t, i64p, i64p, C.c_int64,
                                    C.POINTER(C.c_size_t), C.POINTER(C.c_uint8),
                                    i64p, i64p, C.POINTER(C.c_uint64),
                                    C.POINTER(C.c_uint8), i64p]
_L.bo_timestamps_decode.restype = C.c_int
_L.bo_timestamps_decode.argtypes = [i64p, i64p, u8p, C.c_size_t, C.c_uint8,
                                    C.c_int64, C.c_uint64, C.c_uint8, C.c_int64, C.c_int64]
_L.bo_find_range.restype = C.c_int
_L.bo_find_range.argtypes = [i64p, C.c_int64, C.c_int64, C.c_int64, i64p, i64p]
_L.bo_dictionary_encode.restype = C.c_int
_L.bo_dictionary_encode.argtypes = [u8p, C.c_size_t, u8p, i64p, C.c_int64, C.POINTER(C.c_size_t)]
_L.bo_dictionary_decode.restype = C.c_int
_L.bo_dictionary_decode.argtypes = [u8p, C.c_size_t, i64p, u8p, C.c_size_t, C.c_int64,
                                    C.POINTER(C.c_size_t)]
_L.bo_dictionary_decode_codes.restype = C.c_int
_L.bo_dictionary_decode_codes.argtypes = [u32p, u8p, C.c_size_t, C.c_int64]
_L.bo_bytes_block_encode.restype = C.c_int
_L.bo_bytes_block_encode.argtypes = [u8p, C.c_size_t, u8p, i64p, C.c_int64, C.POINTER(C.c_size_t)]
_L.bo_bytes_block_decode.restype = C.c_int
_L.bo_bytes_block_decode.argtypes = [u8p, C.c_size_t, i64p, u8p, C.c_size_t, C.c_int64,
                                     C.POINTER(C.c_size_t)]
_L.bo_mean_val_i64.restype = C.c_int64
_L.bo_mean_val_i64.argtypes = [C.c_int64, C.c_int64]
_L.bo_mean_val_f64.restype = C.c_double
_L.bo_mean_val_f64.argtypes = [C.c_double, C.c_double]
_L.bo_xxhash64.restype = C.c_uint64
_L.bo_xxhash64.argtypes = [u8p, C.c_size_t]


class BlockDesc(C.Structure):
    _fields_ = [
        ("series_id", C.c_uint64),
        ("count", C.c_uint32),
        ("ts_enc_with_version", C.c_uint8),
        ("version_enc", C.c_uint8),
        ("_pad", C.c_uint8 * 2),
        ("ts_min", C.c_int64),
        ("ts_max", C.c_int64),
        ("version_first", C.c_int64),
        ("ts_off", C.c_uint64),
        ("ts_len", C.c_uint64),
        ("ver_len", C.c_uint64),
        ("col_off", C.c_uint64),
        ("col_len", C.c_uint64),
        ("tag_off", C.c_uint64),
        ("tag_len", C.c_uint64),
        ("tag2_off", C.c_uint64),
        ("tag2_len", C.c_uint64),
        ("tag3_off", C.c_uint64),
        ("tag3_len", C.c_uint64),
        ("group_code", C.c_uint32),
        ("_pad2", C.c_uint32),
    ]


class AggResult(C.Structure):
    _fields_ = [
        ("sum_i", C.c_int64),
        ("sum_f", C.c_double),
        ("count", C.c_int64),
        ("min_i", C.c_int64),
        ("max_i", C.c_int64),
        ("min_f", C.c_double),
        ("max_f", C.c_double),
    ]


_L.bo_scan_agg.restype = C.c_int
_L.bo_scan_agg.argtypes = [u8p, C.POINTER(BlockDesc), C.c_int64, C.c_int,
                           C.c_int64, C.c_int64, u8p, C.c_int64, C.POINTER(AggResult)]
_L.bo_scan_agg_grouped.restype = C.c_int
_L.bo_scan_agg_grouped.argtypes = [u8p, C.POINTER(BlockDesc), C.c_int64, C.c_int,
                                   C.c_int64, C.c_int64, u8p, C.c_int64,
                                   C.POINTER(AggResult), C.c_int64]
_L.bo_scan_agg_bytags.restype = C.c_int
_L.bo_scan_agg_bytags.argtypes = [u8p, C.POINTER(BlockDesc), C.c_int64, C.c_int,
                                  C.c_int64, C.c_int64, C.POINTER(C.c_int),
                                  C.c_int, C.POINTER(u8p), C.POINTER(i64p),
                                  i64p, u8p, i64p, C.POINTER(AggResult)]
_L.bo_scan_agg_bytag.restype = C.c_int
_L.bo_scan_agg_bytag.argtypes = [u8p, C.POINTER(BlockDesc), C.c_int64, C.c_int,
                                 C.c_int64, C.c_int64, C.c_int, u8p, i64p,
                                 C.c_int64, C.POINTER(AggResult)]
_L.bo_scan_agg_multi.restype = C.c_int
_L.bo_scan_agg_multi.argtypes = [u8p, C.POINTER(BlockDesc), C.c_int64, C.c_int,
                                 C.c_int64, C.c_int64, u8p, i64p,
                                 C.POINTER(AggResult), C.c_int64]

VT_INT64 = 2
VT_FLOAT64 = 3

ENC_NAMES = {0: "Unknown", 1: "Const", 2: "DeltaConst", 3: "Delta", 4: "DeltaOfDelta",
             5: "ConstWV", 6: "DeltaConstWV", 7: "DeltaWV", 8: "DeltaOfDeltaWV",
             9: "Plain", 10: "Dictionary"}


def _i64arr(vals):
    return (C.c_int64 * len(vals))(*vals)


def _f64arr(vals):
    return (C.c_double * len(vals))(*vals)


def _u8buf(n):
    return (C.c_uint8 * n)()


def varint_encode(vals):
    buf = _u8buf(10 * len(vals) + 1)
    n = _L.bo_varint64_list_encode(buf, _i64arr(vals), len(vals))
    return bytes(bytearray(buf)[:n])


def varint_decode(data, n):
    out = (C.c_int64 * max(n, 1))()
    consumed = C.c_size_t()
    src = (C.c_uint8 * max(len(data), 1)).from_buffer_copy(data + b"\0" * (not data))
    rc = _L.bo_varint64_list_decode(out, n, src, len(data), C.byref(consumed))
    if rc != 0:
        raise ValueError(f"varint decode failed rc={rc}")
    return list(out[:n]), consumed.value


def int64_list_encode(vals):
    buf = _u8buf(10 * len(vals) + 16)
    out_len = C.c_size_t()
    enc = C.c_uint8()
    first = C.c_int64()
    rc = _L.bo_int64_list_encode(buf, len(buf), _i64arr(vals), len(vals),
                                 C.byref(out_len), C.byref(enc), C.byref(first))
    if rc != 0:
        raise ValueError(f"encode failed rc={rc}")
    return bytes(bytearray(buf)[: out_len.value]), enc.value, first.value


def int64_list_decode(data, enc, first, n):
    out = (C.c_int64 * max(n, 1))()
    src = (C.c_uint8 * max(len(data), 1)).from_buffer_copy(data or b"\0")
    rc = _L.bo_int64_list_decode(out, src, len(data), enc, first, n)
    if rc != 0:
        raise ValueError(f"decode failed rc={rc}")
    return list(out[:n])


def float_to_decimal(vals):
    ints = (C.c_int64 * max(len(vals), 1))()
    exp = C.c_int16()
    rc = _L.bo_float_list_to_decimal(_f64arr(vals), len(vals), ints, C.byref(exp))
    if rc != 0:
        raise ValueError(f"float encode failed rc={rc}")
    return list(ints[: len(vals)]), exp.value


def decimal_to_float(ints, exp, n=None):
    n = n if n is not None else len(ints)
    out = (C.c_double * max(n, 1))()
    rc = _L.bo_decimal_to_float_list(out, _i64arr(ints), exp, n)
    if rc != 0:
        raise ValueError(f"float decode failed rc={rc}")
    return list(out[:n])


def cell_encode(v):
    buf = _u8buf(8)
    _L.bo_cell_i64_to_bytes(buf, v)
    return bytes(bytearray(buf))


def cell_decode(b):
    src = (C.c_uint8 * 8).from_buffer_copy(b)
    return _L.bo_cell_bytes_to_i64(src)


def column_i64_encode(vals):
    buf = _u8buf(10 * len(vals) + 32)
    out_len = C.c_size_t()
    rc = _L.bo_column_i64_encode(buf, len(buf), _i64arr(vals), len(vals), C.byref(out_len))
    if rc != 0:
        raise ValueError(f"column encode failed rc={rc}")
    return bytes(bytearray(buf)[: out_len.value])


def column_i64_decode(payload, n):
    out = (C.c_int64 * n)()
    src = (C.c_uint8 * len(payload)).from_buffer_copy(payload)
    rc = _L.bo_column_i64_decode(out, src, len(payload), n)
    if rc != 0:
        raise ValueError(f"column decode failed rc={rc}")
    return list(out)


def column_f64_encode(vals):
    buf = _u8buf(10 * len(vals) + 32)
    out_len = C.c_size_t()
    rc = _L.bo_column_f64_encode(buf, len(buf), _f64arr(vals), len(vals), C.byref(out_len))
    if rc != 0:
        raise ValueError(f"column f64 encode failed rc={rc}")
    return bytes(bytearray(buf)[: out_len.value])


def column_f64_decode(payload, n):
    out = (C.c_double * n)()
    src = (C.c_uint8 * len(payload)).from_buffer_copy(payload)
    rc = _L.bo_column_f64_decode(out, src, len(payload), n)
    if rc != 0:
        raise ValueError(f"column f64 decode failed rc={rc}")
    return list(out)


def timestamps_encode(ts, versions):
    n = len(ts)
    buf = _u8buf(20 * n + 32)
    out_len = C.c_size_t()
    enc = C.c_uint8()
    tmin = C.c_int64()
    tmax = C.c_int64()
    voff = C.c_uint64()
    venc = C.c_uint8()
    vfirst = C.c_int64()
    rc = _L.bo_timestamps_encode(buf, len(buf), _i64arr(ts), _i64arr(versions), n,
                                 C.byref(out_len), C.byref(enc), C.byref(tmin),
                                 C.byref(tmax), C.byref(voff), C.byref(venc),
                                 C.byref(vfirst))
    if rc != 0:
        raise ValueError(f"ts encode failed rc={rc}")
    return dict(payload=bytes(bytearray(buf)[: out_len.value]), enc=enc.value,
                ts_min=tmin.value, ts_max=tmax.value, version_offset=voff.value,
                version_enc=venc.value, version_first=vfirst.value)


def timestamps_decode(meta, n):
    ts = (C.c_int64 * n)()
    vers = (C.c_int64 * n)()
    payload = meta["payload"]
    src = (C.c_uint8 * len(payload)).from_buffer_copy(payload)
    rc = _L.bo_timestamps_decode(ts, vers, src, len(payload), meta["enc"],
                                 meta["ts_min"], meta["version_offset"],
                                 meta["version_enc"], meta["version_first"], n)
    if rc != 0:
        raise ValueError(f"ts decode failed rc={rc}")
    return list(ts), list(vers)


def find_range(ts, min_val, max_val):
    s = C.c_int64()
    e = C.c_int64()
    found = _L.bo_find_range(_i64arr(ts), len(ts), min_val, max_val, C.byref(s), C.byref(e))
    return s.value, e.value, bool(found)


def _pack_values(values):
    """values: list of bytes-or-None -> (concat_data, lens)."""
    data = b"".join(v for v in values if v is not None)
    lens = [(-1 if v is None else len(v)) for v in values]
    return data, lens


def dictionary_encode(values):
    data, lens = _pack_values(values)
    buf = _u8buf(len(data) + 64 * len(values) + 1024)
    out_len = C.c_size_t()
    src = (C.c_uint8 * max(len(data), 1)).from_buffer_copy(data or b"\0")
    rc = _L.bo_dictionary_encode(buf, len(buf), src, _i64arr(lens), len(values),
                                 C.byref(out_len))
    if rc != 0:
        raise ValueError(f"dict encode failed rc={rc}")
    return bytes(bytearray(buf)[: out_len.value])


def dictionary_decode(payload, n):
    data_cap = 1 << 24
    data_out = _u8buf(data_cap)
    lens_out = (C.c_int64 * n)()
    dlen = C.c_size_t()
    src = (C.c_uint8 * len(payload)).from_buffer_copy(payload)
    rc = _L.bo_dictionary_decode(data_out, data_cap, lens_out, src, len(payload), n,
                                 C.byref(dlen))
    if rc != 0:
        raise ValueError(f"dict decode failed rc={rc}")
    out = []
    raw = bytes(bytearray(data_out)[: dlen.value])
    pos = 0
    for i in range(n):
        ln = lens_out[i]
        if ln < 0:
            out.append(None)
        else:
            out.append(raw[pos: pos + ln])
            pos += ln
    return out


def dictionary_decode_codes(payload, n):
    codes = (C.c_uint32 * n)()
    src = (C.c_uint8 * len(payload)).from_buffer_copy(payload)
    rc = _L.bo_dictionary_decode_codes(codes, src, len(payload), n)
    if rc != 0:
        raise ValueError(f"dict codes decode failed rc={rc}")
    return list(codes)


def bytes_block_encode(values):
    data, lens = _pack_values(values)
    buf = _u8buf(len(data) + 16 * len(values) + 1024)
    out_len = C.c_size_t()
    src = (C.c_uint8 * max(len(data), 1)).from_buffer_copy(data or b"\0")
    rc = _L.bo_bytes_block_encode(buf, len(buf), src, _i64arr(lens), len(values),
                                  C.byref(out_len))
    if rc != 0:
        raise ValueError(f"bytes block encode failed rc={rc}")
    return bytes(bytearray(buf)[: out_len.value])


def bytes_block_decode(payload, n):
    data_cap = 1 << 24
    data_out = _u8buf(data_cap)
    lens_out = (C.c_int64 * n)()
    dlen = C.c_size_t()
    src = (C.c_uint8 * len(payload)).from_buffer_copy(payload)
    rc = _L.bo_bytes_block_decode(data_out, data_cap, lens_out, src, len(payload), n,
                                  C.byref(dlen))
    if rc != 0:
        raise ValueError(f"bytes block decode failed rc={rc}")
    out = []
    raw = bytes(bytearray(data_out)[: dlen.value])
    pos = 0
    for i in range(n):
        ln = lens_out[i]
        if ln < 0:
            out.append(None)
        else:
            out.append(raw[pos: pos + ln])
            pos += ln
    return out


def scan_agg_bytag(payload: bytes, blocks, field_vtype, slot, domain,
                   min_ts=INT64_MIN, max_ts=INT64_MAX):
    """Per-row group-by on a dictionary tag; gid = index in domain."""
    descs = (BlockDesc * len(blocks))()
    for i, b in enumerate(blocks):
        for k, v in b.items():
            setattr(descs[i], k, v)
    blob = b"".join(v for v in domain if v is not None)
    lens = [(-1 if v is None else len(v)) for v in domain]
    out = (AggResult * len(domain))()
    src = (C.c_uint8 * max(len(payload), 1)).from_buffer_copy(payload or b"\0")
    bb = (C.c_uint8 * max(len(blob), 1)).from_buffer_copy(blob or b"\0")
    rc = _L.bo_scan_agg_bytag(src, descs, len(blocks), field_vtype, min_ts,
                              max_ts, slot, bb, (C.c_int64 * len(domain))(*lens),
                              len(domain), out)
    if rc != 0:
        raise ValueError(f"scan_agg_bytag rc={rc}")
    return list(out)


def scan_agg_bytags(payload: bytes, blocks, field_vtype, slots, domains,
                    min_ts=INT64_MIN, max_ts=INT64_MAX, preds=None):
    """Composite group-by over multiple tag slots; gid = g0 + n0*g1 + ..."""
    descs = (BlockDesc * len(blocks))()
    for i, b in enumerate(blocks):
        for k, v in b.items():
            setattr(descs[i], k, v)
    total = 1
    blobs, lens_arrs, nds = [], [], []
    bufs = []
    for dom in domains:
        blob = b"".join(v for v in dom if v is not None)
        lens = [(-1 if v is None else len(v)) for v in dom]
        buf = (C.c_uint8 * max(len(blob), 1)).from_buffer_copy(blob or b"\0")
        la = (C.c_int64 * len(dom))(*lens)
        bufs.append((buf, la))
        blobs.append(C.cast(buf, u8p))
        lens_arrs.append(C.cast(la, i64p))
        nds.append(len(dom))
        total *= len(dom)
    out = (AggResult * total)()
    src = (C.c_uint8 * max(len(payload), 1)).from_buffer_copy(payload or b"\0")
    predbuf = None
    plens = None
    if preds is not None:
        concat = b"".join(p or b"" for p in preds)
        lens3 = [len(p or b"") for p in preds] + [0, 0, 0]
        predbuf = (C.c_uint8 * max(len(concat), 1)).from_buffer_copy(
            concat or b"\0")
        plens = (C.c_int64 * 3)(*lens3[:3])
    rc = _L.bo_scan_agg_bytags(
        src, descs, len(blocks), field_vtype, min_ts, max_ts,
        (C.c_int * len(slots))(*slots), len(slots),
        (u8p * len(domains))(*blobs), (i64p * len(domains))(*lens_arrs),
        (C.c_int64 * len(domains))(*nds),
        C.cast(predbuf, u8p) if predbuf else None,
        C.cast(plens, i64p) if plens else None, out)
    if rc != 0:
        raise ValueError(f"scan_agg_bytags rc={rc}")
    return list(out)


def mean_val_i64(s, c):
    return _L.bo_mean_val_i64(s, c)


def mean_val_f64(s, c):
    return _L.bo_mean_val_f64(s, c)


def xxhash64(data: bytes) -> int:
    src = (C.c_uint8 * max(len(data), 1)).from_buffer_copy(data or b"\0")
    return _L.bo_xxhash64(src, len(data))


INT64_MIN = -(2 ** 63)
INT64_MAX = 2 ** 63 - 1


def scan_agg(payload: bytes, blocks, field_vtype, min_ts=INT64_MIN,
             max_ts=INT64_MAX, pred: bytes = b"", n_groups=1, preds=None):
    """Run the oracle scan+aggregate over a part payload + block directory.

    blocks: list of dicts with BlockDesc fields.  preds: optional list of
    up to 3 predicate byte strings (conjunctive, one per tag slot);
    overrides pred.  Returns list of AggResult (length n_groups)."""
    descs = (BlockDesc * len(blocks))()
    for i, b in enumerate(blocks):
        for k, v in b.items():
            setattr(descs[i], k, v)
    out = (AggResult * n_groups)()
    src = (C.c_uint8 * max(len(payload), 1)).from_buffer_copy(payload or b"\0")
    if preds is not None:
        concat = b"".join(p or b"" for p in preds)
        lens = [len(p or b"") for p in preds] + [0, 0, 0]
        predbuf = (C.c_uint8 * max(len(concat), 1)).from_buffer_copy(concat or b"\0")
        rc = _L.bo_scan_agg_multi(src, descs, len(blocks), field_vtype, min_ts,
                                  max_ts, predbuf, (C.c_int64 * 3)(*lens[:3]),
                                  out, n_groups)
    else:
        predbuf = (C.c_uint8 * max(len(pred), 1)).from_buffer_copy(pred or b"\0")
        rc = _L.bo_scan_agg_grouped(src, descs, len(blocks), field_vtype, min_ts,
                                    max_ts, predbuf, len(pred), out, n_groups)
    if rc != 0:
        raise ValueError(f"scan_agg failed rc={rc}")
    return list(out)


def make_descs(blocks):
    """Build the oracle-side (BlockDesc * n) array once (zero-copy scans
    slice it per thread)."""
    descs = (BlockDesc * len(blocks))()
    for i, b in enumerate(blocks):
        for k, v in b.items():
            setattr(descs[i], k, v)
    return descs


def scan_agg_raw(src_ptr, descs, i0, n_blocks, field_vtype):
    """Zero-copy whole-range scan for the bench's CPU-baseline leg: the
    caller owns the payload buffer (the part builder's arena) and a
    make_descs array; [i0, i0+n_blocks) is this call's slice.  Returns a
    single AggResult.  The underlying C call releases the GIL, so
    threads over disjoint slices run on separate cores."""
    out = (AggResult * 1)()
    dp = C.cast(C.byref(descs, i0 * C.sizeof(BlockDesc)),
                C.POINTER(BlockDesc))
    rc = _L.bo_scan_agg_grouped(src_ptr, dp, n_blocks, field_vtype,
                                INT64_MIN, INT64_MAX,
                                C.cast(b"\0", u8p), 0, out, 1)
    if rc != 0:
        raise ValueError(f"scan_agg_raw rc={rc}")
    return out[0]
