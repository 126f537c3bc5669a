"""Group-by on plain (>256-distinct) tag columns and on numeric key
columns — the remaining computeKey semantics
(pkg/query/vectorized/measure/groupby.go:287-364: key components over ANY
key column; banyand/measure/column.go:266-278: the >256-card dictionary
bail stores a plain bytes block).

CPU tests pin the oracle; the GPU tests are the engine-vs-oracle parity
proper (k_resolve_plain_groups (gid,count) runs + the run-merge fold).
Numeric keys use the reference's stored tag cell bytes
(convert/number.go:33-46 sign-flip int64 / :128-132 IEEE-754 BE float64);
group identity is byte equality, exactly appendKeyComponent's contract.
"""
import random

import pytest

import oracle as o
from banyandb_amd import (PartBuilder, Session, VT_INT64, VT_FLOAT64,
                          AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX,
                          i64_tag_cell, f64_tag_cell, FLOAT_RAW_EXP)
from helpers import oracle_blocks

T0 = 1_700_000_000_000_000_000
MS = 10 ** 6
FUNCS = [AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX]


def build_plain_part(n_sids=6, n=3000, card=300, seed=31, runs=True,
                     nullable_f64=False):
    """Blocks whose tag 0 is a PLAIN column (card > 256 forces the
    dictionary bail).  runs=True gives run-shaped tags (telemetry-like),
    else i.i.d. per row."""
    rng = random.Random(seed)
    b = PartBuilder()
    values = [b"user_%03d" % i for i in range(card)]
    for sid in range(n_sids):
        ts = [T0 + i * MS for i in range(n)]
        if nullable_f64:
            vals = [None if rng.random() < 0.2 else
                    rng.uniform(-1e6, 1e6) for _ in range(n)]
            b.add_block_f64_nullable(sid + 1, ts, [1] * n, vals)
        else:
            b.add_block_i64(sid + 1, ts, [1] * n,
                            [rng.randint(-10**9, 10**9) for _ in range(n)])
        tags = []
        i = 0
        while len(tags) < n:
            v = values[rng.randrange(card)]
            if rng.random() < 0.02:
                v = None  # nil rows drop from grouped output
            rl = rng.randint(1, 40) if runs else 1
            tags.extend([v] * rl)
            i += 1
        b.set_block_tag(tags[:n])
    return b


def oracle_bytag(b, vtype, slot, domain, preds=None):
    payload, blocks = oracle_blocks(b)
    kw = {"preds": preds} if preds else {}
    return o.scan_agg_bytags(payload, blocks, vtype, [slot], [domain], **kw)


def cmp_groups(gs, orc, float_mode=False):
    assert len(gs) == len(orc)
    for g, oc in zip(gs, orc):
        assert g.count == oc.count
        if float_mode:
            if oc.count:
                assert g.min_f == oc.min_f and g.max_f == oc.max_f
                # fold order differs; tolerance scaled to the magnitude
                # of the addends (cancellation-aware)
                assert abs(g.sum_f - oc.sum_f) <= 1e-9 * max(
                    1.0, oc.count * 1e6)
        else:
            assert g.sum_i == oc.sum_i
            if oc.count:
                assert g.min_i == oc.min_i and g.max_i == oc.max_i


def test_oracle_groups_plain_tag_cpu():
    """The oracle itself groups on plain tag columns (pins the GPU
    parity target)."""
    b = build_plain_part(n_sids=2, n=500, card=270)
    domain = [b"user_%03d" % i for i in range(270)]
    orc = oracle_bytag(b, VT_INT64, 0, domain)
    total = sum(r.count for r in orc)
    assert 0 < total <= 2 * 500  # nil rows dropped


def test_oracle_groups_numeric_key_cpu():
    b = PartBuilder()
    n = 400
    keys = [i64_tag_cell(k) for k in (-5, 0, 7, 10**12)]
    rng = random.Random(3)
    ts = [T0 + i * MS for i in range(n)]
    b.add_block_i64(1, ts, [1] * n, list(range(n)))
    b.set_block_tag([keys[rng.randrange(4)] for _ in range(n)])
    orc = oracle_bytag(b, VT_INT64, 0, keys)
    assert sum(r.count for r in orc) == n


@pytest.mark.gpu
class TestGpuPlainGroupBy:
    def run_both(self, b, vtype, domain, slot=0, preds=None, float_exp=0):
        orc = oracle_bytag(b, vtype, slot, domain, preds=preds)
        s = Session(0)
        s.upload_part(b)
        s.configure_by_tag(vtype, FUNCS, slot, domain, float_exp=float_exp)
        s.consume(preds=preds) if preds else s.consume()
        gs = s.finalize()
        s.close()
        return gs, orc

    def test_plain_tag_runs(self):
        b = build_plain_part(card=300, runs=True)
        domain = [b"user_%03d" % i for i in range(300)]
        gs, orc = self.run_both(b, VT_INT64, domain)
        cmp_groups(gs, orc)
        assert sum(g.count for g in gs) > 0

    def test_plain_tag_iid_rows(self):
        """i.i.d. tags: worst-case run compression (every row a run)."""
        b = build_plain_part(n_sids=3, n=2000, card=280, runs=False, seed=7)
        domain = [b"user_%03d" % i for i in range(280)]
        gs, orc = self.run_both(b, VT_INT64, domain)
        cmp_groups(gs, orc)

    def test_plain_tag_partial_domain(self):
        """Domain covering a subset: out-of-domain rows drop (nil and
        unmapped drop per the host-domain contract)."""
        b = build_plain_part(card=300, seed=8)
        domain = [b"user_%03d" % i for i in range(0, 300, 3)]
        gs, orc = self.run_both(b, VT_INT64, domain)
        cmp_groups(gs, orc)

    def test_plain_uniform_block_fast_path(self):
        """A plain column whose block is one run collapses to the uniform
        gid (fast fold path)."""
        b = PartBuilder()
        n = 2000
        for sid in range(4):
            ts = [T0 + i * MS for i in range(n)]
            b.add_block_i64(sid + 1, ts, [1] * n, list(range(n)))
            b.set_block_tag([b"user_%03d" % (sid * 70)] * n)  # uniform
        domain = [b"user_%03d" % i for i in range(290)]
        gs, orc = self.run_both(b, VT_INT64, domain)
        cmp_groups(gs, orc)
        assert sum(g.count for g in gs) == 4 * n

    def test_composite_dict_x_plain(self):
        """Composite key: slot 0 dictionary (env), slot 1 plain user."""
        rng = random.Random(9)
        ENVS = [b"prod", b"dev"]
        users = [b"user_%03d" % i for i in range(270)]
        b = PartBuilder()
        n = 2000
        for sid in range(4):
            ts = [T0 + i * MS for i in range(n)]
            b.add_block_i64(sid + 1, ts, [1] * n,
                            [rng.randint(-10**6, 10**6) for _ in range(n)])
            b.set_block_tag([ENVS[rng.randrange(2)] for _ in range(n)])
            tags = []
            while len(tags) < n:
                tags.extend([users[rng.randrange(270)]] *
                            rng.randint(1, 30))
            b.set_block_tag(tags[:n])
        payload, blocks = oracle_blocks(b)
        orc = o.scan_agg_bytags(payload, blocks, VT_INT64, [0, 1],
                                [ENVS, users])
        s = Session(0)
        s.upload_part(b)
        s.configure_by_tags(VT_INT64, FUNCS, [0, 1], [ENVS, users])
        s.consume()
        gs = s.finalize()
        s.close()
        cmp_groups(gs, orc)

    def test_plain_group_with_dict_predicate(self):
        """Plain group slot combined with a dictionary predicate on
        another slot (run-merge + pred cursors)."""
        rng = random.Random(11)
        ENVS = [b"prod", b"dev", b"qa"]
        users = [b"user_%03d" % i for i in range(260)]
        b = PartBuilder()
        n = 2000
        for sid in range(4):
            ts = [T0 + i * MS for i in range(n)]
            b.add_block_i64(sid + 1, ts, [1] * n,
                            [rng.randint(-10**6, 10**6) for _ in range(n)])
            tags = []
            while len(tags) < n:
                tags.extend([ENVS[rng.randrange(3)]] * rng.randint(1, 50))
            b.set_block_tag(tags[:n])
            tags2 = []
            while len(tags2) < n:
                tags2.extend([users[rng.randrange(260)]] *
                             rng.randint(1, 30))
            b.set_block_tag(tags2[:n])
        domain = users
        preds = [b"prod", b""]
        payload, blocks = oracle_blocks(b)
        orc = o.scan_agg_bytags(payload, blocks, VT_INT64, [1], [domain],
                                preds=[b"prod", b"", b""])
        s = Session(0)
        s.upload_part(b)
        s.configure_by_tag(VT_INT64, FUNCS, 1, domain)
        s.consume(preds=[b"prod"])
        gs = s.finalize()
        s.close()
        cmp_groups(gs, orc)

    def test_numeric_i64_keys_dictionary(self):
        """int64 group keys (sign-flip cells) through the dictionary
        path (card <= 256)."""
        rng = random.Random(12)
        keys = [i64_tag_cell(k) for k in
                [-10**15, -7, -1, 0, 1, 42, 10**9, 10**18]]
        b = PartBuilder()
        n = 3000
        for sid in range(4):
            ts = [T0 + i * MS for i in range(n)]
            b.add_block_i64(sid + 1, ts, [1] * n,
                            [rng.randint(-10**9, 10**9) for _ in range(n)])
            tags = []
            while len(tags) < n:
                tags.extend([keys[rng.randrange(8)]] * rng.randint(1, 60))
            b.set_block_tag(tags[:n])
        gs, orc = self.run_both(b, VT_INT64, keys)
        cmp_groups(gs, orc)
        assert sum(g.count for g in gs) == 4 * n

    def test_numeric_i64_keys_plain(self):
        """int64 group keys at cardinality > 256 (plain fallback)."""
        rng = random.Random(13)
        keys = [i64_tag_cell(k * 37 - 5000) for k in range(300)]
        b = PartBuilder()
        n = 2000
        for sid in range(3):
            ts = [T0 + i * MS for i in range(n)]
            b.add_block_i64(sid + 1, ts, [1] * n, list(range(n)))
            tags = []
            while len(tags) < n:
                tags.extend([keys[rng.randrange(300)]] * rng.randint(1, 20))
            b.set_block_tag(tags[:n])
        gs, orc = self.run_both(b, VT_INT64, keys)
        cmp_groups(gs, orc)

    def test_f64_keys_dictionary(self):
        """float64 group keys (IEEE-754 BE cells)."""
        rng = random.Random(14)
        keys = [f64_tag_cell(v) for v in
                [-2.5, 7.5, 0.0, 1.25, 3.14159, 1e300]]
        b = PartBuilder()
        n = 2000
        ts = [T0 + i * MS for i in range(n)]
        b.add_block_i64(9, ts, [1] * n, list(range(n)))
        tags = []
        while len(tags) < n:
            tags.extend([keys[rng.randrange(6)]] * rng.randint(1, 25))
        b.set_block_tag(tags[:n])
        gs, orc = self.run_both(b, VT_INT64, keys)
        cmp_groups(gs, orc)

    def test_nullable_f64_field_under_plain_group(self):
        """Null-bearing float64 field folded under a plain-tag group-by:
        nulls drop from count and every aggregate (aggregation.go:310)."""
        b = build_plain_part(n_sids=3, n=1500, card=280, seed=15,
                             nullable_f64=True)
        domain = [b"user_%03d" % i for i in range(280)]
        gs, orc = self.run_both(b, VT_FLOAT64, domain,
                                float_exp=FLOAT_RAW_EXP)
        cmp_groups(gs, orc, float_mode=True)


@pytest.mark.gpu
class TestBitmapPredInGroupMerge:
    """Bitmap-mode (plain >256-card tag) predicates combined with
    row-varying group-by — previously a loud v1 error, now filtered
    per-row inside fold_range (the reference applies predicates to any
    column uniformly)."""

    def _part(self, seed, nullable=False):
        rng = random.Random(seed)
        ENVS = [b"prod", b"dev", b"qa"]
        users = [b"user_%03d" % i for i in range(280)]
        b = PartBuilder()
        n = 2000
        for sid in range(4):
            ts = [T0 + i * MS for i in range(n)]
            if nullable:
                b.add_block_i64_nullable(
                    sid + 1, ts, [1] * n,
                    [None if rng.random() < 0.15 else
                     rng.randint(-10**6, 10**6) for _ in range(n)])
            else:
                b.add_block_i64(sid + 1, ts, [1] * n,
                                [rng.randint(-10**6, 10**6)
                                 for _ in range(n)])
            tags = []
            while len(tags) < n:
                tags.extend([ENVS[rng.randrange(3)]] * rng.randint(1, 40))
            b.set_block_tag(tags[:n])
            b.set_block_tag([users[rng.randrange(280)]
                             for _ in range(n)])  # plain, i.i.d.
        return b, ENVS, users

    def test_plain_pred_with_dict_group(self):
        b, ENVS, users = self._part(21)
        payload, blocks = oracle_blocks(b)
        orc = o.scan_agg_bytags(payload, blocks, VT_INT64, [0], [ENVS],
                                preds=[b"", b"user_007", b""])
        s = Session(0)
        s.upload_part(b)
        s.configure_by_tag(VT_INT64, FUNCS, 0, ENVS)
        s.consume(preds=[b"", b"user_007"])
        gs = s.finalize()
        s.close()
        assert sum(oc.count for oc in orc) > 0
        cmp_groups(gs, orc)

    def test_plain_pred_with_plain_group(self):
        """Both the group slot and the predicate slot are plain."""
        b, ENVS, users = self._part(22)
        domain = users[:270]
        payload, blocks = oracle_blocks(b)
        orc = o.scan_agg_bytags(payload, blocks, VT_INT64, [1], [domain],
                                preds=[b"prod", b"", b""])
        s = Session(0)
        s.upload_part(b)
        s.configure_by_tag(VT_INT64, FUNCS, 1, domain)
        s.consume(preds=[b"prod"])
        gs = s.finalize()
        s.close()
        cmp_groups(gs, orc)

    def test_plain_pred_dict_group_nullable_field(self):
        b, ENVS, users = self._part(23, nullable=True)
        payload, blocks = oracle_blocks(b)
        orc = o.scan_agg_bytags(payload, blocks, VT_INT64, [0], [ENVS],
                                preds=[b"", b"user_012", b""])
        s = Session(0)
        s.upload_part(b)
        s.configure_by_tag(VT_INT64, FUNCS, 0, ENVS)
        s.consume(preds=[b"", b"user_012"])
        gs = s.finalize()
        s.close()
        cmp_groups(gs, orc)
