"""Duplicate-(sid,ts) fold at part build (part.go:178,192-198 +
datapoints.go:189-197): rows order by (ts asc, version desc); only the
first row of each timestamp — the highest version — survives, and
per-row tag lists fold the same way.  CPU-only (oracle scan parity)."""
import banyandb_amd as ba
from helpers import oracle_scan

T0 = 1_700_000_000_000_000_000
MS = 10 ** 6


def test_dedup_keeps_highest_version():
    b = ba.PartBuilder()
    ts = [T0, T0 + MS, T0 + MS, T0 + 2 * MS, T0 + 2 * MS, T0 + 3 * MS]
    ver = [1, 3, 7, 2, 2, 1]
    vals = [10, 20, 99, 31, 30, 40]
    b.add_block_i64(5, ts, ver, vals)
    d = b.blocks()[0]
    assert d.count == 4
    r = oracle_scan(b, ba.VT_INT64)[0]
    # ts T0+MS keeps version 7 (value 99); T0+2MS has tied versions ->
    # stable order keeps the FIRST input row (value 31)
    assert r.count == 4
    assert r.sum_i == 10 + 99 + 31 + 40
    assert r.min_i == 10 and r.max_i == 99


def test_dedup_unsorted_input_sorts_by_ts():
    b = ba.PartBuilder()
    ts = [T0 + 2 * MS, T0, T0 + MS, T0]
    ver = [1, 5, 1, 9]
    vals = [3, 1, 2, 100]
    b.add_block_i64(1, ts, ver, vals)
    d = b.blocks()[0]
    assert d.count == 3
    assert d.ts_min == T0 and d.ts_max == T0 + 2 * MS
    r = oracle_scan(b, ba.VT_INT64)[0]
    assert r.sum_i == 100 + 2 + 3  # T0 keeps version 9


def test_dedup_folds_tag_rows_like_the_reference():
    b = ba.PartBuilder()
    ts = [T0, T0 + MS, T0 + MS, T0 + 2 * MS]
    ver = [1, 9, 2, 1]
    vals = [1, 2, 22, 3]
    b.add_block_i64(7, ts, ver, vals)
    # caller passes the ORIGINAL per-row tag list; the surviving rows'
    # tags are kept (row with version 2 at T0+MS drops, so its tag does)
    b.set_block_tag([b"a", b"b", b"x", b"a"])
    r = oracle_scan(b, ba.VT_INT64, pred=b"a")[0]
    assert (r.count, r.sum_i) == (2, 4)
    r = oracle_scan(b, ba.VT_INT64, pred=b"b")[0]
    assert (r.count, r.sum_i) == (1, 2)
    r = oracle_scan(b, ba.VT_INT64, pred=b"x")[0]
    assert r.count == 0


def test_dedup_nullable_and_f64():
    b = ba.PartBuilder()
    ts = [T0, T0, T0 + MS]
    ver = [2, 8, 1]
    b.add_block_i64_nullable(3, ts, ver, [5, None, 6])
    d = b.blocks()[0]
    assert d.count == 2
    r = oracle_scan(b, ba.VT_INT64)[0]
    assert r.count == 1  # surviving T0 row is null (version 8)
    assert r.sum_i == 6

    b2 = ba.PartBuilder()
    b2.add_block_f64(4, ts, ver, [1.25, 7.5, 2.25])
    assert b2.blocks()[0].count == 2
    r2 = oracle_scan(b2, ba.VT_FLOAT64)[0]
    assert r2.count == 2
    assert r2.sum_f == 7.5 + 2.25


def test_dedup_property_random():
    """Property: for random (ts, version) multisets the surviving rows
    equal the reference model — sort by (ts asc, version desc) stably,
    keep the first row per timestamp (part.go:178 + datapoints.go:189-197
    + the :192-198 skip loop)."""
    import random as _r
    for seed in range(40):
        rng = _r.Random(1000 + seed)
        n = rng.randint(1, 60)
        ts = [T0 + rng.randint(0, 9) * MS for _ in range(n)]
        ver = [rng.randint(0, 4) for _ in range(n)]
        vals = list(range(n))
        # reference model
        order = sorted(range(n), key=lambda i: (ts[i], -ver[i]))
        seen, keep = set(), []
        for i in order:
            if ts[i] not in seen:
                seen.add(ts[i])
                keep.append(i)
        want = sorted((ts[i], vals[i]) for i in keep)

        b = ba.PartBuilder()
        b.add_block_i64(1, ts, ver, vals)
        d = b.blocks()[0]
        assert d.count == len(keep), (seed, d.count, len(keep))
        r = oracle_scan(b, ba.VT_INT64)[0]
        assert r.count == len(keep)
        assert r.sum_i == sum(v for _, v in want), seed
        assert r.min_i == min(v for _, v in want)
        assert r.max_i == max(v for _, v in want)
