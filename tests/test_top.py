"""Top-N over group results — BatchTop ordering contract (top.go:31-121,
ApplyTopToReduce reduce.go:290).  CPU-only."""
import ctypes as C

import banyandb_amd as ba

_l = ba.lib()
_l.bydb_top_groups.restype = C.c_int
_l.bydb_top_groups.argtypes = [C.POINTER(ba.Result), C.c_int64, C.c_int,
                               C.c_int64, C.c_int, C.POINTER(C.c_int64),
                               C.POINTER(C.c_int64)]

SEL_SUM_I, SEL_COUNT, SEL_MIN_I, SEL_MAX_I, SEL_MEAN_I = range(5)
SEL_SUM_F, SEL_MIN_F, SEL_MAX_F, SEL_MEAN_F = range(5, 9)


def top(results_vals, sel, k, asc, counts=None):
    n = len(results_vals)
    arr = (ba.Result * n)()
    for i, v in enumerate(results_vals):
        arr[i].count = counts[i] if counts else 1
        if sel < 5:
            setattr(arr[i], ["sum_i", "count", "min_i", "max_i", "mean_i"][sel], v)
            if sel == 1:
                arr[i].count = v
        else:
            setattr(arr[i], ["sum_f", "min_f", "max_f", "mean_f"][sel - 5], v)
    out = (C.c_int64 * n)()
    out_n = C.c_int64()
    rc = _l.bydb_top_groups(arr, n, sel, k, 1 if asc else 0, out, C.byref(out_n))
    assert rc == 0
    return list(out[: out_n.value])


def test_desc_keeps_highest_largest_first():
    assert top([5, 1, 9, 7, 3], SEL_SUM_I, 3, asc=False) == [2, 3, 0]


def test_asc_keeps_lowest_smallest_first():
    assert top([5, 1, 9, 7, 3], SEL_SUM_I, 3, asc=True) == [1, 4, 0]


def test_ties_earlier_insertion_wins():
    # top.go:78-81: ties surface in insertion order
    assert top([7, 7, 7, 1], SEL_SUM_I, 2, asc=False) == [0, 1]
    assert top([7, 7, 7, 9], SEL_SUM_I, 2, asc=True) == [0, 1]


def test_nulls_sort_lowest():
    # count==0 groups are nulls (top.go:34): kept first in asc, evicted in desc
    vals = [5, 0, 9]
    counts = [1, 0, 1]
    assert top(vals, SEL_SUM_I, 2, asc=True, counts=counts) == [1, 0]
    assert top(vals, SEL_SUM_I, 2, asc=False, counts=counts) == [2, 0]


def test_float_selector():
    assert top([1.5, -2.0, 3.25], SEL_MAX_F, 2, asc=False) == [2, 0]


def test_k_larger_than_groups():
    assert top([2, 1], SEL_SUM_I, 10, asc=True) == [1, 0]
