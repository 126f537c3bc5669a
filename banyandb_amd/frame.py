"""ctypes wrapper for the raw columnar wire-frame codec (csrc/frame.cpp) —
the egress / Map-partial transport slot (SURVEY section 2 'Columnar wire
frame'; pkg/query/vectorized/frame)."""
import ctypes as C

from banyandb_amd import lib

ROLE_TIMESTAMP, ROLE_VERSION, ROLE_SERIES, ROLE_SHARD, ROLE_TAG, ROLE_FIELD = \
    1, 2, 3, 4, 5, 6
TYPE_I64, TYPE_F64, TYPE_STR, TYPE_BYTES, TYPE_TAGVALUE, TYPE_FIELDVALUE = \
    1, 2, 3, 4, 5, 6

_l = lib()
u8p = C.POINTER(C.c_uint8)
i64p = C.POINTER(C.c_int64)
f64p = C.POINTER(C.c_double)

_l.bydb_frame_builder_create.restype = C.c_void_p
_l.bydb_frame_builder_create.argtypes = [C.c_uint64]
_l.bydb_frame_builder_destroy.argtypes = [C.c_void_p]
_l.bydb_frame_add_i64.restype = C.c_int
_l.bydb_frame_add_i64.argtypes = [C.c_void_p, C.c_uint8, C.c_char_p, C.c_char_p, i64p, u8p]
_l.bydb_frame_add_f64.restype = C.c_int
_l.bydb_frame_add_f64.argtypes = [C.c_void_p, C.c_uint8, C.c_char_p, C.c_char_p, f64p, u8p]
_l.bydb_frame_add_str.restype = C.c_int
_l.bydb_frame_add_str.argtypes = [C.c_void_p, C.c_uint8, C.c_char_p, C.c_char_p, u8p, i64p]
_l.bydb_frame_add_bytes.restype = C.c_int
_l.bydb_frame_add_bytes.argtypes = [C.c_void_p, C.c_uint8, C.c_char_p, C.c_char_p, u8p, i64p]
_l.bydb_frame_add_tagvalue_str.restype = C.c_int
_l.bydb_frame_add_tagvalue_str.argtypes = [C.c_void_p, C.c_uint8, C.c_char_p,
                                           C.c_char_p, u8p, i64p, C.c_int]
_l.bydb_frame_add_tagvalue_int.restype = C.c_int
_l.bydb_frame_add_tagvalue_int.argtypes = [C.c_void_p, C.c_uint8, C.c_char_p,
                                           C.c_char_p, i64p, u8p, C.c_int]
_l.bydb_frame_finish.restype = C.c_int
_l.bydb_frame_finish.argtypes = [C.c_void_p]
_l.bydb_frame_len.restype = C.c_uint64
_l.bydb_frame_len.argtypes = [C.c_void_p]
_l.bydb_frame_data.restype = u8p
_l.bydb_frame_data.argtypes = [C.c_void_p]

_l.bydb_frame_open.restype = C.c_void_p
_l.bydb_frame_open.argtypes = [u8p, C.c_uint64]
_l.bydb_frame_close.argtypes = [C.c_void_p]
_l.bydb_frame_reader_error.restype = C.c_char_p
_l.bydb_frame_reader_error.argtypes = [C.c_void_p]
_l.bydb_frame_nrows.restype = C.c_uint64
_l.bydb_frame_nrows.argtypes = [C.c_void_p]
_l.bydb_frame_ncols.restype = C.c_uint64
_l.bydb_frame_ncols.argtypes = [C.c_void_p]
_l.bydb_frame_col_info.restype = C.c_int
_l.bydb_frame_col_info.argtypes = [C.c_void_p, C.c_uint64, C.POINTER(C.c_uint8),
                                   C.POINTER(C.c_uint8), C.c_char_p, C.c_uint64,
                                   C.c_char_p, C.c_uint64]
_l.bydb_frame_col_null.restype = C.c_int
_l.bydb_frame_col_null.argtypes = [C.c_void_p, C.c_uint64, C.c_uint64]
_l.bydb_frame_col_i64.restype = C.c_int
_l.bydb_frame_col_i64.argtypes = [C.c_void_p, C.c_uint64, i64p]
_l.bydb_frame_col_var.restype = C.c_int
_l.bydb_frame_col_var.argtypes = [C.c_void_p, C.c_uint64, u8p, C.c_uint64, i64p,
                                  C.POINTER(C.c_uint64)]


def _pack(values):
    data = b"".join(v for v in values if v is not None)
    lens = [(-1 if v is None else len(v)) for v in values]
    return data, lens


class FrameBuilder:
    def __init__(self, nrows):
        self.nrows = nrows
        self._h = _l.bydb_frame_builder_create(nrows)

    def add_i64(self, role, name, family, vals, nulls=None):
        nb = (C.c_uint8 * self.nrows)(*(nulls or [0] * self.nrows))
        _l.bydb_frame_add_i64(self._h, role, name.encode(), family.encode(),
                              (C.c_int64 * self.nrows)(*vals), nb)

    def add_f64(self, role, name, family, vals, nulls=None):
        nb = (C.c_uint8 * self.nrows)(*(nulls or [0] * self.nrows))
        _l.bydb_frame_add_f64(self._h, role, name.encode(), family.encode(),
                              (C.c_double * self.nrows)(*vals), nb)

    def _var(self, fn, role, name, family, values):
        data, lens = _pack(values)
        buf = (C.c_uint8 * max(len(data), 1)).from_buffer_copy(data or b"\0")
        fn(self._h, role, name.encode(), family.encode(), buf,
           (C.c_int64 * self.nrows)(*lens))

    def add_str(self, role, name, family, values):
        self._var(_l.bydb_frame_add_str, role, name, family, values)

    def add_bytes(self, role, name, family, values):
        self._var(_l.bydb_frame_add_bytes, role, name, family, values)

    def add_tagvalue_str(self, role, name, family, values, field_value=False):
        data, lens = _pack(values)
        buf = (C.c_uint8 * max(len(data), 1)).from_buffer_copy(data or b"\0")
        _l.bydb_frame_add_tagvalue_str(self._h, role, name.encode(),
                                       family.encode(), buf,
                                       (C.c_int64 * self.nrows)(*lens),
                                       1 if field_value else 0)

    def add_tagvalue_int(self, role, name, family, vals, nulls=None,
                         field_value=False):
        nb = (C.c_uint8 * self.nrows)(*(nulls or [0] * self.nrows))
        _l.bydb_frame_add_tagvalue_int(self._h, role, name.encode(),
                                       family.encode(),
                                       (C.c_int64 * self.nrows)(*vals), nb,
                                       1 if field_value else 0)

    def finish(self) -> bytes:
        _l.bydb_frame_finish(self._h)
        n = _l.bydb_frame_len(self._h)
        return C.string_at(_l.bydb_frame_data(self._h), n)

    def __del__(self):
        if getattr(self, "_h", None):
            _l.bydb_frame_builder_destroy(self._h)
            self._h = None


class FrameReader:
    def __init__(self, data: bytes):
        buf = (C.c_uint8 * len(data)).from_buffer_copy(data)
        self._h = _l.bydb_frame_open(buf, len(data))
        err = _l.bydb_frame_reader_error(self._h).decode()
        if err:
            raise ValueError(f"frame decode: {err}")
        self.nrows = _l.bydb_frame_nrows(self._h)
        self.ncols = _l.bydb_frame_ncols(self._h)

    def col_info(self, ci):
        role = C.c_uint8()
        typ = C.c_uint8()
        name = C.create_string_buffer(256)
        fam = C.create_string_buffer(256)
        _l.bydb_frame_col_info(self._h, ci, C.byref(role), C.byref(typ), name,
                               256, fam, 256)
        return role.value, typ.value, name.value.decode(), fam.value.decode()

    def col_nulls(self, ci):
        return [bool(_l.bydb_frame_col_null(self._h, ci, r))
                for r in range(self.nrows)]

    def col_i64(self, ci):
        out = (C.c_int64 * self.nrows)()
        rc = _l.bydb_frame_col_i64(self._h, ci, out)
        assert rc == 0
        return list(out)

    def col_var(self, ci):
        cap = 1 << 22
        data = (C.c_uint8 * cap)()
        lens = (C.c_int64 * self.nrows)()
        dl = C.c_uint64()
        rc = _l.bydb_frame_col_var(self._h, ci, data, cap, lens, C.byref(dl))
        assert rc == 0
        raw = bytes(bytearray(data)[: dl.value])
        out = []
        pos = 0
        for i in range(self.nrows):
            ln = lens[i]
            if ln < 0:
                out.append(None)
            else:
                out.append(raw[pos: pos + ln])
                pos += ln
        return out

    def __del__(self):
        if getattr(self, "_h", None):
            _l.bydb_frame_close(self._h)
            self._h = None


class ReduceSpec(C.Structure):
    # mirrors bydb_reduce_spec (include/bydb_gpu.h)
    _fields_ = [("input_col", C.c_int32), ("func", C.c_int32)]


_l.bydb_reduce_frames.restype = C.c_int
_l.bydb_reduce_frames.argtypes = [
    C.POINTER(u8p), C.POINTER(C.c_uint64), C.c_int,
    C.c_int, C.POINTER(C.c_int32), C.c_int,
    C.POINTER(ReduceSpec), C.c_int,
    C.c_int64, i64p, f64p,
    u8p, C.c_uint64, C.POINTER(C.c_uint64), i64p]


def reduce_frames(frames, specs, key_cols, shard_col=-1, out_cap=65536,
                  key_buf_cap=1 << 22):
    """Product-side AggModeReduce with replica dedup
    (aggregation_reduce.go:83-138) over raw map frames.

    frames: list of frame bytes; specs: list of (input_col, func);
    key_cols: group-key column indices; shard_col: RoleShardID column
    (or -1 to dedup on the group key alone).  Returns a list of
    (key_bytes, [per-spec (int_final, float_final)]) in first-seen group
    order."""
    n = len(frames)
    bufs = [(C.c_uint8 * max(len(f), 1)).from_buffer_copy(f or b"\0")
            for f in frames]
    fptrs = (u8p * n)(*[C.cast(b, u8p) for b in bufs])
    flens = (C.c_uint64 * n)(*[len(f) for f in frames])
    sarr = (ReduceSpec * len(specs))(*[ReduceSpec(c, f) for c, f in specs])
    karr = (C.c_int32 * len(key_cols))(*key_cols)
    out_i = (C.c_int64 * (out_cap * len(specs)))()
    out_f = (C.c_double * (out_cap * len(specs)))()
    kbuf = (C.c_uint8 * key_buf_cap)()
    koffs = (C.c_uint64 * (out_cap + 1))()
    ng = C.c_int64()
    rc = _l.bydb_reduce_frames(fptrs, flens, n, shard_col, karr,
                               len(key_cols), sarr, len(specs), out_cap,
                               out_i, out_f, kbuf, key_buf_cap, koffs,
                               C.byref(ng))
    if rc != 0:
        raise RuntimeError(f"bydb_reduce_frames rc={rc}")
    raw = bytes(bytearray(kbuf))
    out = []
    for g in range(ng.value):
        key = raw[koffs[g]: koffs[g + 1]]
        vals = [(out_i[g * len(specs) + s], out_f[g * len(specs) + s])
                for s in range(len(specs))]
        out.append((key, vals))
    return out
