"""banyandb_amd — MI355X-native BanyanDB measure scan+aggregate engine.

Python here is plumbing only (device setup, torch.distributed/RCCL merge,
test drivers).  The product is libbydb_gpu.so: C++ host + HIP/CDNA4 kernels
behind the C-ABI in include/bydb_gpu.h.  On a GPU box the HIP extension is
REQUIRED — if the .so is missing or a GPU call fails, errors are raised
loudly; there is no CPU fallback anywhere in this package.
"""
import ctypes as C
import os

_PKG = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_PKG, "libbydb_gpu.so")

VT_INT64 = 2
VT_FLOAT64 = 3
AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX, AGG_MEAN = range(5)
MODE_ALL, MODE_MAP, MODE_REDUCE = range(3)
INT64_MIN = -(2 ** 63)
FLOAT_RAW_EXP = -32768  # sessions over nullable float64 columns
INT64_MAX = 2 ** 63 - 1


class BlockDesc(C.Structure):
    # must mirror bydb_block_desc in include/bydb_gpu.h exactly
    _fields_ = [
        ("series_id", C.c_uint64),
        ("count", C.c_uint32),
        ("ts_enc_with_version", C.c_uint8),
        ("version_enc", C.c_uint8),
        ("field_enc", C.c_uint8),
        ("field_vtype", C.c_uint8),
        ("ts_min", C.c_int64),
        ("ts_max", C.c_int64),
        ("version_first", C.c_int64),
        ("field_first", C.c_int64),
        ("exp", C.c_int16),
        ("_pad", C.c_uint8 * 6),
        ("ts_off", C.c_uint64),
        ("ts_len", C.c_uint64),
        ("field_off", C.c_uint64),
        ("field_len", C.c_uint64),
        ("tag_off", C.c_uint64),
        ("tag_len", C.c_uint64),
        ("tag2_off", C.c_uint64),
        ("tag2_len", C.c_uint64),
        ("tag3_off", C.c_uint64),
        ("tag3_len", C.c_uint64),
        ("group_code", C.c_uint32),
        ("_pad2", C.c_uint32),
    ]


class Partial(C.Structure):
    _fields_ = [
        ("sum_i", C.c_int64),
        ("count", C.c_int64),
        ("min_i", C.c_int64),
        ("max_i", C.c_int64),
        ("sum_f", C.c_double),
        ("_pad", C.c_double),
    ]


class Result(C.Structure):
    _fields_ = [
        ("sum_i", C.c_int64),
        ("sum_f", C.c_double),
        ("count", C.c_int64),
        ("min_i", C.c_int64),
        ("max_i", C.c_int64),
        ("min_f", C.c_double),
        ("max_f", C.c_double),
        ("mean_i", C.c_int64),
        ("mean_f", C.c_double),
    ]


def _build_if_possible():
    from banyandb_amd.build import build
    return build(verbose=False)


def _load():
    # Load torch's bundled HIP runtime FIRST: our .so and torch both carry
    # libamdhip64 with the same soname; whichever loads first is reused by
    # the other.  The engine runs fine on torch's runtime, but torch breaks
    # on the system one — so pin the order here (observed on the MI355X
    # box: torch-first OK both ways, engine-first torch.cuda dead).
    try:
        import torch  # noqa: F401
    except Exception:
        pass
    if not os.path.exists(_SO):
        try:
            _build_if_possible()
        except Exception as e:  # loud: no silent fallback
            raise RuntimeError(
                f"libbydb_gpu.so missing and build failed: {e}. "
                "Run `python banyandb_amd/build.py`."
            ) from e
    lib = C.CDLL(_SO)
    u8p = C.POINTER(C.c_uint8)
    i64p = C.POINTER(C.c_int64)
    f64p = C.POINTER(C.c_double)
    bp = C.POINTER(BlockDesc)

    lib.bydb_session_create.restype = C.c_void_p
    lib.bydb_session_create.argtypes = [C.c_int]
    lib.bydb_session_destroy.argtypes = [C.c_void_p]
    lib.bydb_last_error.restype = C.c_char_p
    lib.bydb_last_error.argtypes = [C.c_void_p]
    lib.bydb_part_reserve.restype = C.c_int
    lib.bydb_part_reserve.argtypes = [C.c_void_p, C.c_uint64, C.c_int64]
    lib.bydb_part_append.restype = C.c_int
    lib.bydb_part_append.argtypes = [C.c_void_p, u8p, C.c_uint64, bp, C.c_int64]
    lib.bydb_agg_configure.restype = C.c_int
    lib.bydb_agg_configure.argtypes = [C.c_void_p, C.c_int, C.c_uint32, C.c_uint32, C.c_int]
    lib.bydb_agg_configure_by_tags.restype = C.c_int
    lib.bydb_agg_configure_by_tags.argtypes = [C.c_void_p, C.c_int, C.c_uint32,
                                               C.POINTER(C.c_int), C.c_int,
                                               C.POINTER(u8p),
                                               C.POINTER(C.POINTER(C.c_uint64)),
                                               C.POINTER(C.c_uint32), C.c_int]
    lib.bydb_agg_configure_by_tag.restype = C.c_int
    lib.bydb_agg_configure_by_tag.argtypes = [C.c_void_p, C.c_int, C.c_uint32,
                                              C.c_int, u8p,
                                              C.POINTER(C.c_uint64), C.c_uint32,
                                              C.c_int]
    lib.bydb_set_partials_buffer.restype = C.c_int
    lib.bydb_set_partials_buffer.argtypes = [C.c_void_p, C.c_void_p, C.c_uint64]
    lib.bydb_reset.restype = C.c_int
    lib.bydb_reset.argtypes = [C.c_void_p]
    lib.bydb_consume.restype = C.c_int
    lib.bydb_consume.argtypes = [C.c_void_p, C.c_int64, C.c_int64, u8p, C.c_uint64]
    lib.bydb_consume_multi.restype = C.c_int
    lib.bydb_consume_multi.argtypes = [C.c_void_p, C.c_int64, C.c_int64,
                                       C.POINTER(u8p), C.POINTER(C.c_uint64),
                                       C.c_int]
    lib.bydb_finalize.restype = C.c_int
    lib.bydb_finalize.argtypes = [C.c_void_p, C.POINTER(Result), C.c_int64]
    lib.bydb_finalize_partials.restype = C.c_int
    lib.bydb_finalize_partials.argtypes = [C.c_void_p, C.POINTER(Partial), C.c_int64]
    lib.bydb_set_float_exp.restype = C.c_int
    lib.bydb_set_float_exp.argtypes = [C.c_void_p, C.c_int16]
    lib.bydb_last_consume_ms.restype = C.c_double
    lib.bydb_last_consume_ms.argtypes = [C.c_void_p]
    lib.bydb_group_first_seen.restype = C.c_int
    lib.bydb_group_first_seen.argtypes = [C.c_void_p, C.POINTER(C.c_uint64),
                                          C.c_int64]
    lib.bydb_reduce_partials2.restype = C.c_int
    lib.bydb_reduce_partials2.argtypes = [C.POINTER(Partial), C.c_int64, C.c_int64,
                                          C.c_int, C.c_int16, C.POINTER(Result)]

    lib.bydb_part_builder_create.restype = C.c_void_p
    lib.bydb_part_builder_destroy.argtypes = [C.c_void_p]
    lib.bydb_part_builder_error.restype = C.c_char_p
    lib.bydb_part_builder_error.argtypes = [C.c_void_p]
    lib.bydb_part_builder_add_block_f64_nullable.restype = C.c_int
    lib.bydb_part_builder_add_block_f64_nullable.argtypes = [
        C.c_void_p, C.c_uint64, i64p, i64p, C.POINTER(C.c_double), u8p,
        C.c_int64, C.c_uint32]
    lib.bydb_part_builder_add_block_i64_nullable.restype = C.c_int
    lib.bydb_part_builder_add_block_i64_nullable.argtypes = [
        C.c_void_p, C.c_uint64, i64p, i64p, i64p, u8p, C.c_int64, C.c_uint32]
    lib.bydb_part_builder_add_block_i64.restype = C.c_int
    lib.bydb_part_builder_add_block_i64.argtypes = [C.c_void_p, C.c_uint64, i64p,
                                                    i64p, i64p, C.c_int64, C.c_uint32]
    lib.bydb_part_builder_add_block_f64.restype = C.c_int
    lib.bydb_part_builder_add_block_f64.argtypes = [C.c_void_p, C.c_uint64, i64p,
                                                    i64p, f64p, C.c_int64, C.c_uint32]
    lib.bydb_part_builder_set_block_tag.restype = C.c_int
    lib.bydb_part_builder_set_block_tag.argtypes = [C.c_void_p, u8p, i64p, C.c_int64]
    lib.bydb_part_builder_set_tag_table.restype = C.c_int
    lib.bydb_part_builder_set_tag_table.argtypes = [C.c_void_p, C.c_int, u8p, i64p,
                                                    C.c_int64]
    lib.bydb_part_builder_payload_len.restype = C.c_uint64
    lib.bydb_part_builder_payload_len.argtypes = [C.c_void_p]
    lib.bydb_part_builder_payload.restype = u8p
    lib.bydb_part_builder_payload.argtypes = [C.c_void_p]
    lib.bydb_part_builder_n_blocks.restype = C.c_int64
    lib.bydb_part_builder_n_blocks.argtypes = [C.c_void_p]
    lib.bydb_part_builder_blocks.restype = bp
    lib.bydb_part_builder_blocks.argtypes = [C.c_void_p]
    lib.bydb_part_builder_drain.restype = C.c_int
    lib.bydb_part_builder_drain.argtypes = [C.c_void_p]
    lib.bydb_gen_series_i64.restype = C.c_int
    lib.bydb_gen_series_i64.argtypes = [C.c_void_p, C.c_uint64, C.c_int64, C.c_int64,
                                        C.c_int64, C.c_int64, C.c_int64, C.c_uint64,
                                        C.c_uint32]
    lib.bydb_gen_series_f64.restype = C.c_int
    lib.bydb_gen_series_f64.argtypes = [C.c_void_p, C.c_uint64, C.c_int64, C.c_int64,
                                        C.c_int64, C.c_double, C.c_double, C.c_uint64,
                                        C.c_uint32]
    lib.bydb_gen_series_bulk_i64.restype = C.c_int
    lib.bydb_gen_series_bulk_i64.argtypes = [C.c_void_p, C.c_uint64, C.c_int64,
                                             C.c_int64, C.c_int64, C.c_int64,
                                             C.c_int64, C.c_int64, C.c_uint64,
                                             C.c_uint32, C.c_int]
    lib.bydb_part_write_dir.restype = C.c_int
    lib.bydb_part_write_dir.argtypes = [C.c_void_p, C.c_char_p, C.c_char_p,
                                        C.c_char_p, C.POINTER(C.c_char_p), C.c_int]
    lib.bydb_part_read_dir.restype = C.c_int
    lib.bydb_part_read_dir.argtypes = [C.c_void_p, C.c_char_p]
    lib.bydb_gen_series_bulk_f64.restype = C.c_int
    lib.bydb_gen_series_bulk_f64.argtypes = [C.c_void_p, C.c_uint64, C.c_int64,
                                             C.c_int64, C.c_int64, C.c_int64,
                                             C.c_double, C.c_double, C.c_uint64,
                                             C.c_uint32, C.c_int]
    return lib


_lib = _load()


def lib():
    return _lib


class PartBuilder:
    """Host-side fixture writer (mirrors the reference block write path)."""

    def __init__(self):
        self._h = _lib.bydb_part_builder_create()
        if not self._h:
            raise RuntimeError("part builder create failed")

    def _ck(self, rc):
        if rc != 0:
            raise RuntimeError(
                f"part builder error rc={rc}: "
                f"{_lib.bydb_part_builder_error(self._h).decode()}")

    def add_block_i64(self, series_id, ts, versions, vals, group_code=0):
        n = len(ts)
        self._ck(_lib.bydb_part_builder_add_block_i64(
            self._h, series_id, (C.c_int64 * n)(*ts), (C.c_int64 * n)(*versions),
            (C.c_int64 * n)(*vals), n, group_code))

    def add_block_i64_nullable(self, series_id, ts, versions, vals,
                               group_code=0):
        """Null-bearing int64 field column: vals[i] is None for null rows
        (stored as the reference's Plain cell block; the fold skips
        nulls)."""
        n = len(ts)
        valid = (C.c_uint8 * n)(*[0 if v is None else 1 for v in vals])
        vv = (C.c_int64 * n)(*[0 if v is None else v for v in vals])
        self._ck(_lib.bydb_part_builder_add_block_i64_nullable(
            self._h, series_id, (C.c_int64 * n)(*ts),
            (C.c_int64 * n)(*versions), vv, valid, n, group_code))

    def add_block_f64_nullable(self, series_id, ts, versions, vals,
                               group_code=0):
        """Null-bearing float64 column (raw IEEE-754 cells; configure the
        session with float_exp=FLOAT_RAW_EXP)."""
        n = len(ts)
        valid = (C.c_uint8 * n)(*[0 if v is None else 1 for v in vals])
        vv = (C.c_double * n)(*[0.0 if v is None else v for v in vals])
        self._ck(_lib.bydb_part_builder_add_block_f64_nullable(
            self._h, series_id, (C.c_int64 * n)(*ts),
            (C.c_int64 * n)(*versions), vv, valid, n, group_code))

    def add_block_f64(self, series_id, ts, versions, vals, group_code=0):
        n = len(ts)
        self._ck(_lib.bydb_part_builder_add_block_f64(
            self._h, series_id, (C.c_int64 * n)(*ts), (C.c_int64 * n)(*versions),
            (C.c_double * n)(*vals), n, group_code))

    def set_tag_table(self, slot, values):
        """Entity-tag table: generators attach values[series %% len] as a
        constant tag column (slot) to every generated block."""
        data = b"".join(v for v in values if v is not None)
        lens = [(-1 if v is None else len(v)) for v in values]
        buf = (C.c_uint8 * max(len(data), 1)).from_buffer_copy(data or b"\0")
        self._ck(_lib.bydb_part_builder_set_tag_table(
            self._h, slot, buf, (C.c_int64 * len(values))(*lens), len(values)))

    def set_block_tag(self, values):
        """values: list of bytes-or-None, one per row of the last block."""
        data = b"".join(v for v in values if v is not None)
        lens = [(-1 if v is None else len(v)) for v in values]
        n = len(values)
        buf = (C.c_uint8 * max(len(data), 1)).from_buffer_copy(data or b"\0")
        self._ck(_lib.bydb_part_builder_set_block_tag(
            self._h, buf, (C.c_int64 * n)(*lens), n))

    def gen_series_i64(self, series_index, n_dp, t0, stride_ns, base, ramp,
                       seed, group_code=0):
        self._ck(_lib.bydb_gen_series_i64(self._h, series_index, n_dp, t0,
                                          stride_ns, base, ramp, seed, group_code))

    def gen_series_f64(self, series_index, n_dp, t0, stride_ns, base, ramp,
                       seed, group_code=0):
        self._ck(_lib.bydb_gen_series_f64(self._h, series_index, n_dp, t0,
                                          stride_ns, base, ramp, seed, group_code))

    def gen_bulk_i64(self, first_index, n_series, n_dp, t0, stride_ns,
                     base_step, ramp, seed, group_mod=0, threads=None):
        threads = threads or os.cpu_count() or 8
        self._ck(_lib.bydb_gen_series_bulk_i64(
            self._h, first_index, n_series, n_dp, t0, stride_ns, base_step,
            ramp, seed, group_mod, threads))

    def gen_bulk_f64(self, first_index, n_series, n_dp, t0, stride_ns,
                     base_step, ramp, seed, group_mod=0, threads=None):
        threads = threads or os.cpu_count() or 8
        self._ck(_lib.bydb_gen_series_bulk_f64(
            self._h, first_index, n_series, n_dp, t0, stride_ns, base_step,
            ramp, seed, group_mod, threads))

    @property
    def payload(self) -> bytes:
        n = _lib.bydb_part_builder_payload_len(self._h)
        if n == 0:
            return b""
        ptr = _lib.bydb_part_builder_payload(self._h)
        # NOT string_at: its size argument is a C int, so payloads past
        # 2 GiB would truncate (mod 2^32) silently
        return bytes((C.c_ubyte * n).from_address(
            C.addressof(ptr.contents)))

    @property
    def payload_len(self) -> int:
        return _lib.bydb_part_builder_payload_len(self._h)

    def raw_payload_ptr(self):
        return _lib.bydb_part_builder_payload(self._h)

    @property
    def n_blocks(self) -> int:
        return _lib.bydb_part_builder_n_blocks(self._h)

    def blocks_ptr(self):
        return _lib.bydb_part_builder_blocks(self._h)

    def blocks(self):
        ptr = _lib.bydb_part_builder_blocks(self._h)
        return [ptr[i] for i in range(self.n_blocks)]

    def write_dir(self, path, field_name="value", tag_family="default",
                  tag_names=()):
        """Write the part as a reference-layout on-disk part directory."""
        arr = (C.c_char_p * max(len(tag_names), 1))(
            *[t.encode() for t in tag_names] or [b""])
        self._ck(_lib.bydb_part_write_dir(self._h, path.encode(),
                                          field_name.encode(),
                                          tag_family.encode(), arr,
                                          len(tag_names)))

    def read_dir(self, path):
        """Load a part directory written by write_dir (or the reference)."""
        self._ck(_lib.bydb_part_read_dir(self._h, path.encode()))

    def drain(self):
        self._ck(_lib.bydb_part_builder_drain(self._h))

    def __del__(self):
        try:
            if getattr(self, "_h", None):
                _lib.bydb_part_builder_destroy(self._h)
        except Exception:
            pass  # interpreter shutdown
        self._h = None


class Session:
    """GPU session over the C-ABI (BreakerOperator lifecycle)."""

    def __init__(self, device=0):
        self._h = _lib.bydb_session_create(device)
        if not self._h:
            raise RuntimeError(
                "bydb_session_create failed — no GPU visible? The HIP engine "
                "has no CPU fallback.")
        self.n_groups = 1
        self.field_vtype = VT_INT64

    def _ck(self, rc):
        if rc != 0:
            raise RuntimeError(
                f"bydb error rc={rc}: {_lib.bydb_last_error(self._h).decode()}")

    def upload_part(self, builder: PartBuilder):
        self._ck(_lib.bydb_part_reserve(self._h, builder.payload_len,
                                        builder.n_blocks))
        self._ck(_lib.bydb_part_append(self._h, builder.raw_payload_ptr(),
                                       builder.payload_len, builder.blocks_ptr(),
                                       builder.n_blocks))

    def reserve(self, payload_bytes, n_blocks):
        self._ck(_lib.bydb_part_reserve(self._h, payload_bytes, n_blocks))

    def append(self, builder: PartBuilder):
        self._ck(_lib.bydb_part_append(self._h, builder.raw_payload_ptr(),
                                       builder.payload_len, builder.blocks_ptr(),
                                       builder.n_blocks))

    def configure(self, field_vtype, funcs, n_groups=1, mode=MODE_ALL,
                  float_exp=0):
        mask = 0
        for f in funcs:
            mask |= 1 << f
        self.n_groups = n_groups
        self.field_vtype = field_vtype
        self._ck(_lib.bydb_agg_configure(self._h, field_vtype, mask, n_groups, mode))
        self._ck(_lib.bydb_set_float_exp(self._h, float_exp))

    def configure_by_tag(self, field_vtype, funcs, tag_slot, domain,
                         mode=MODE_ALL, float_exp=0):
        """Per-row group-by on the dictionary tag in tag_slot; group id =
        index into domain (list of bytes values)."""
        mask = 0
        for f in funcs:
            mask |= 1 << f
        blob = b"".join(domain)
        offs = [0]
        for v in domain:
            offs.append(offs[-1] + len(v))
        self.n_groups = len(domain)
        self.field_vtype = field_vtype
        buf = (C.c_uint8 * max(len(blob), 1)).from_buffer_copy(blob or b"\0")
        self._ck(_lib.bydb_agg_configure_by_tag(
            self._h, field_vtype, mask, tag_slot, buf,
            (C.c_uint64 * len(offs))(*offs), len(domain), mode))
        self._ck(_lib.bydb_set_float_exp(self._h, float_exp))

    def configure_by_tags(self, field_vtype, funcs, slots, domains,
                          mode=MODE_ALL, float_exp=0):
        """Composite group-by over multiple dictionary tags; group id =
        g0 + n0*g1 + n0*n1*g2 (domains in slot-list order)."""
        mask = 0
        for f in funcs:
            mask |= 1 << f
        u8p = C.POINTER(C.c_uint8)
        bufs = []
        blob_ptrs = (u8p * len(domains))()
        off_ptrs = (C.POINTER(C.c_uint64) * len(domains))()
        nvals = (C.c_uint32 * len(domains))()
        total = 1
        for i, dom in enumerate(domains):
            blob = b"".join(dom)
            offs = [0]
            for v in dom:
                offs.append(offs[-1] + len(v))
            buf = (C.c_uint8 * max(len(blob), 1)).from_buffer_copy(blob or b"\0")
            oa = (C.c_uint64 * len(offs))(*offs)
            bufs.append((buf, oa))
            blob_ptrs[i] = C.cast(buf, u8p)
            off_ptrs[i] = oa
            nvals[i] = len(dom)
            total *= len(dom)
        self.n_groups = total
        self.field_vtype = field_vtype
        self._ck(_lib.bydb_agg_configure_by_tags(
            self._h, field_vtype, mask, (C.c_int * len(slots))(*slots),
            len(slots), blob_ptrs, off_ptrs, nvals, mode))
        self._ck(_lib.bydb_set_float_exp(self._h, float_exp))

    def set_partials_buffer(self, dev_ptr, nbytes):
        self._ck(_lib.bydb_set_partials_buffer(self._h, dev_ptr, nbytes))

    def reset(self):
        self._ck(_lib.bydb_reset(self._h))

    def consume(self, min_ts=INT64_MIN, max_ts=INT64_MAX, pred=b"", preds=None):
        if preds is not None:
            bufs = [(C.c_uint8 * max(len(p), 1)).from_buffer_copy(p or b"\0")
                    for p in preds]
            arr = (C.POINTER(C.c_uint8) * len(preds))(
                *[C.cast(b, C.POINTER(C.c_uint8)) for b in bufs])
            lens = (C.c_uint64 * len(preds))(*[len(p) for p in preds])
            self._ck(_lib.bydb_consume_multi(self._h, min_ts, max_ts, arr,
                                             lens, len(preds)))
            return
        buf = (C.c_uint8 * max(len(pred), 1)).from_buffer_copy(pred or b"\0")
        self._ck(_lib.bydb_consume(self._h, min_ts, max_ts, buf, len(pred)))

    def finalize(self):
        out = (Result * self.n_groups)()
        self._ck(_lib.bydb_finalize(self._h, out, self.n_groups))
        return list(out)

    def finalize_partials(self):
        out = (Partial * self.n_groups)()
        self._ck(_lib.bydb_finalize_partials(self._h, out, self.n_groups))
        return list(out)

    def last_consume_ms(self):
        return _lib.bydb_last_consume_ms(self._h)

    def group_first_seen(self):
        """Per-group first-seen keys; sorting group ids by them gives the
        reference first-seen materialisation order (2^64-1 = group never
        entered)."""
        out = (C.c_uint64 * self.n_groups)()
        self._ck(_lib.bydb_group_first_seen(self._h, out, self.n_groups))
        return list(out)

    def groups_in_first_seen_order(self):
        ks = self.group_first_seen()
        return [g for g, k in sorted(enumerate(ks), key=lambda t: t[1])
                if ks[g] != (1 << 64) - 1]

    def close(self):
        if self._h:
            _lib.bydb_session_destroy(self._h)
            self._h = None

    def __del__(self):
        self.close()


def reduce_partials(parts_flat, n_parts_per_group, n_groups, field_vtype,
                    float_exp=0):
    """Host AggModeReduce combine (aggregation_reduce.go semantics)."""
    arr = (Partial * (n_parts_per_group * n_groups))(*parts_flat)
    out = (Result * n_groups)()
    rc = _lib.bydb_reduce_partials2(arr, n_parts_per_group, n_groups,
                                    field_vtype, float_exp, out)
    if rc != 0:
        raise RuntimeError(f"reduce_partials rc={rc}")
    return list(out)


def i64_tag_cell(v: int) -> bytes:
    """Stored bytes of an int64 tag value: the reference's order-preserving
    sign-flip 8-byte BE cell (convert/number.go:33-46 Int64ToBytes) — the
    raw value bytes measure tag columns hold for ValueTypeInt64
    (batch_decode.go:48-53 decodes with convert.BytesToInt64).  Use these
    as tag values / group domains for numeric group keys."""
    u = (v | (1 << 63)) if v >= 0 else ((1 << 63) - (-v)) % (1 << 64)
    return u.to_bytes(8, "big")


def f64_tag_cell(v: float) -> bytes:
    """Stored bytes of a float64 tag value: IEEE-754 big-endian
    (convert/number.go:128-132 Float64ToBytes)."""
    import struct
    return struct.pack(">d", v)
