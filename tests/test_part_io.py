"""On-disk part directory round-trip (Appendix A layout) — CPU-only.

write_dir emits the reference's part files (metadata.json, meta.bin,
primary.bin, timestamps.bin, fv.bin, <family>.tfm/.tf); read_dir walks
them back.  Parity: the reloaded part scans (oracle) identically to the
original."""
import json
import os
import random
import tempfile

import pytest

import banyandb_amd as ba
from helpers import oracle_scan

T0 = 1_700_000_000_000_000_000
MS = 10 ** 6


def build_part(with_tags=True, float_vals=False):
    rng = random.Random(77)
    b = ba.PartBuilder()
    if with_tags:
        b.set_tag_table(0, [b"prod", b"dev", b"staging", b"qa"])
        b.set_tag_table(1, [f"r{i}".encode() for i in range(8)])
    for s in range(6):
        if float_vals:
            b.gen_series_f64(s, 20000, T0, MS, 10.0 + s, 0.01, 9)
        else:
            b.gen_series_i64(s, 20000, T0, MS, s * 1000, 1, 9)
    return b


def test_part_dir_files_and_metadata():
    b = build_part()
    with tempfile.TemporaryDirectory() as td:
        p = os.path.join(td, "0000000000000001")
        b.write_dir(p, field_name="value", tag_family="default",
                    tag_names=["env", "region"])
        for f in ("metadata.json", "meta.bin", "primary.bin",
                  "timestamps.bin", "fv.bin", "default.tfm", "default.tf"):
            assert os.path.exists(os.path.join(p, f)), f
        pm = json.load(open(os.path.join(p, "metadata.json")))
        assert pm["totalCount"] == 6 * 20000
        assert pm["blocksCount"] == b.n_blocks
        assert pm["minTimestamp"] == T0
        assert pm["maxTimestamp"] == T0 + 19999 * MS


def test_part_dir_roundtrip_scan_i64():
    b = build_part()
    orc0 = oracle_scan(b, ba.VT_INT64)[0]
    orc0p = oracle_scan(b, ba.VT_INT64, pred=b"dev")[0]
    with tempfile.TemporaryDirectory() as td:
        p = os.path.join(td, "0000000000000001")
        b.write_dir(p, tag_names=["env", "region"])
        b2 = ba.PartBuilder()
        b2.read_dir(p)
        assert b2.n_blocks == b.n_blocks
        orc1 = oracle_scan(b2, ba.VT_INT64)[0]
        assert (orc1.count, orc1.sum_i, orc1.min_i, orc1.max_i) == \
            (orc0.count, orc0.sum_i, orc0.min_i, orc0.max_i)
        # tags survive the round trip (predicate parity)
        orc1p = oracle_scan(b2, ba.VT_INT64, pred=b"dev")[0]
        assert (orc1p.count, orc1p.sum_i) == (orc0p.count, orc0p.sum_i)


def test_part_dir_roundtrip_scan_f64():
    b = build_part(with_tags=False, float_vals=True)
    orc0 = oracle_scan(b, ba.VT_FLOAT64)[0]
    with tempfile.TemporaryDirectory() as td:
        p = os.path.join(td, "0000000000000001")
        b.write_dir(p)
        b2 = ba.PartBuilder()
        b2.read_dir(p)
        orc1 = oracle_scan(b2, ba.VT_FLOAT64)[0]
        assert orc1.count == orc0.count
        assert orc1.sum_f == orc0.sum_f
        assert orc1.min_f == orc0.min_f and orc1.max_f == orc0.max_f
        # exponent survived
        assert b2.blocks()[0].exp == b.blocks()[0].exp


def test_part_dir_clamped_scan_after_reload():
    b = build_part(with_tags=False)
    lo, hi = T0 + 101 * MS, T0 + 9999 * MS
    orc0 = oracle_scan(b, ba.VT_INT64, min_ts=lo, max_ts=hi)[0]
    with tempfile.TemporaryDirectory() as td:
        p = os.path.join(td, "p")
        b.write_dir(p)
        b2 = ba.PartBuilder()
        b2.read_dir(p)
        orc1 = oracle_scan(b2, ba.VT_INT64, min_ts=lo, max_ts=hi)[0]
        assert (orc1.count, orc1.sum_i) == (orc0.count, orc0.sum_i)


def test_part_dir_roundtrip_nullable_and_plain_tag():
    """Nullable field columns and plain (>256-distinct) tag columns
    survive the on-disk round trip byte-identically (their streams are
    the reference formats, written/read verbatim)."""
    rng = random.Random(88)
    b = ba.PartBuilder()
    for sid in range(4):
        n = 2000
        ts = [T0 + i * MS for i in range(n)]
        vals = [None if rng.random() < 0.3
                else rng.randint(-10**10, 10**10) for _ in range(n)]
        b.add_block_i64_nullable(sid + 1, ts, [1] * n, vals)
        b.set_block_tag([b"user_%03d" % rng.randrange(300)
                         for _ in range(n)])  # plain (card > 256)
    ref = oracle_scan(b, ba.VT_INT64, pred=b"user_042")[0]
    ref_nopred = oracle_scan(b, ba.VT_INT64)[0]
    with tempfile.TemporaryDirectory() as td:
        p = os.path.join(td, "000000000000000a")
        b.write_dir(p, tag_names=["user"])
        b2 = ba.PartBuilder()
        b2.read_dir(p)
        got = oracle_scan(b2, ba.VT_INT64, pred=b"user_042")[0]
        got_nopred = oracle_scan(b2, ba.VT_INT64)[0]
    assert ref_nopred.count > 0
    assert got_nopred.count == ref_nopred.count
    assert got_nopred.sum_i == ref_nopred.sum_i
    assert got_nopred.min_i == ref_nopred.min_i
    assert got.count == ref.count
    assert got.sum_i == ref.sum_i


def test_part_dir_roundtrip_custom_family_name():
    """A part written with a non-"default" tag family name round-trips
    with its tags intact: the reader opens <fam>.tfm/.tf by the family
    name each blockMetadata record carries (block_metadata.go:129-147),
    not a hard-coded probe."""
    b = build_part()
    orc0p = oracle_scan(b, ba.VT_INT64, pred=b"dev")[0]
    with tempfile.TemporaryDirectory() as td:
        p = os.path.join(td, "0000000000000002")
        b.write_dir(p, tag_family="searchable", tag_names=["env", "region"])
        assert os.path.exists(os.path.join(p, "searchable.tfm"))
        assert not os.path.exists(os.path.join(p, "default.tfm"))
        b2 = ba.PartBuilder()
        b2.read_dir(p)
        orc1p = oracle_scan(b2, ba.VT_INT64, pred=b"dev")[0]
        assert orc1p.count > 0
        assert (orc1p.count, orc1p.sum_i) == (orc0p.count, orc0p.sum_i)


def test_part_dir_missing_family_files_is_loud():
    """A block that references a tag family whose .tfm/.tf files are
    absent must error (BYDB_ERR_BAD_DATA), never silently drop the tag
    columns."""
    b = build_part()
    with tempfile.TemporaryDirectory() as td:
        p = os.path.join(td, "0000000000000003")
        b.write_dir(p, tag_family="searchable", tag_names=["env", "region"])
        os.remove(os.path.join(p, "searchable.tfm"))
        b2 = ba.PartBuilder()
        with pytest.raises(RuntimeError):
            b2.read_dir(p)
