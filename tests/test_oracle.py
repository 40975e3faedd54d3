"""Oracle vs the committed golden vectors (CPU only).

The golden vectors under tests/golden/ were extracted from the reference's
own in-tree tests by tests/golden/generate_golden.py (see its docstring for
file:line citations).  These tests pin the oracle BEFORE it is trusted as the
parity checker for the HIP kernels.
"""
import json
import os

import numpy as np
import pytest

import oracle

HERE = os.path.dirname(os.path.abspath(__file__))
GOLD = os.path.join(HERE, "golden")


def load_json(name):
    with open(os.path.join(GOLD, name)) as f:
        return json.load(f)


@pytest.fixture(scope="module")
def lineitem():
    return np.load(os.path.join(GOLD, "lineitem_slice.npz"))


@pytest.fixture(scope="module")
def alltypes():
    return np.load(os.path.join(GOLD, "alltypes_plain.npz"))


def test_oracle_q6_golden(lineitem):
    e = load_json("lineitem_expected.json")
    li = lineitem
    n = len(li["l_shipdate"])
    disc16 = oracle.dec128_from_ints(li["l_discount"])
    qty16 = oracle.dec128_from_ints(li["l_quantity"])
    price16 = oracle.dec128_from_ints(li["l_extendedprice"])
    lo, hi = e["q6_1996_window"]
    cnt, total = oracle.q6(li["l_shipdate"], disc16, qty16, price16,
                           lo, hi, e["q6_1996_disc"][0], e["q6_1996_disc"][1],
                           e["q6_1996_qty_lt"])
    assert cnt == e["q6_1996_count"]
    assert total == e["q6_1996_sum_scale4"]
    # the canonical 1994 parameters give an empty result on this slice
    cnt2, total2 = oracle.q6(li["l_shipdate"], disc16, qty16, price16,
                             e["q6_window"][0], e["q6_window"][1], 5, 7, 2400)
    assert cnt2 == e["q6_count"] and total2 == e["q6_sum_scale4"]


def test_oracle_q1_golden(lineitem):
    e = load_json("lineitem_expected.json")
    li = lineitem
    # dictionary-encode returnflag/linestatus to u8 codes < 16
    rf_vals = sorted(set(li["l_returnflag"]))
    ls_vals = sorted(set(li["l_linestatus"]))
    rf = np.array([rf_vals.index(v) for v in li["l_returnflag"]], dtype=np.uint8)
    ls = np.array([ls_vals.index(v) for v in li["l_linestatus"]], dtype=np.uint8)
    out = oracle.q1(rf, ls,
                    oracle.dec128_from_ints(li["l_quantity"]),
                    oracle.dec128_from_ints(li["l_extendedprice"]),
                    oracle.dec128_from_ints(li["l_discount"]),
                    oracle.dec128_from_ints(li["l_tax"]),
                    li["l_shipdate"], e["q1_cutoff_date32"])
    got = {}
    for g, (count, sums) in out.items():
        key = f"{rf_vals[g >> 4]}|{ls_vals[g & 15]}"
        got[key] = {"count": count, "sum_qty": sums[0], "sum_price": sums[1],
                    "sum_disc_price": sums[2], "sum_charge": sums[3],
                    "sum_disc": sums[4]}
    assert got == e["q1_groups"]


def test_oracle_filter_alltypes_golden(alltypes):
    """Pins FilterExec semantics against context_checks.rs:63-77."""
    e = load_json("alltypes_expected.json")
    ids = alltypes["id"].astype(np.int32)
    n = len(ids)
    mask = oracle.filter_mask([("i32", ids, None, 4, 4, 0)], n)  # op 4: x > lo
    idx = oracle.mask_to_indices(mask, n)
    got_str = [str(alltypes["string_col"][i]) for i in idx]
    assert got_str == e["filter_id_gt4"]["string_col"]
    got_ts = alltypes["timestamp_ns"][idx]
    want_ts = np.array(e["filter_id_gt4"]["timestamp_iso"],
                       dtype="datetime64[s]").astype("datetime64[ns]").astype(np.int64)
    assert np.array_equal(got_ts, want_ts)


def test_oracle_groupby_count_alltypes_golden(alltypes):
    """Pins the filtered group-by count against context_checks.rs:810-827."""
    e = load_json("alltypes_expected.json")
    ids = alltypes["id"].astype(np.int32)
    n = len(ids)
    mask = oracle.filter_mask([("i32", ids, None, 4, 4, 0)], n)
    idx = oracle.mask_to_indices(mask, n)
    counts = {}
    for i in idx:
        counts[str(alltypes["string_col"][i])] = \
            counts.get(str(alltypes["string_col"][i]), 0) + 1
    assert counts == e["groupby_count_id_gt4"]


def test_hash_regression():
    """Freezes the bg_ahash.h restatement (NOT DataFusion-pinned; see
    bg_ahash.h PARITY STATUS)."""
    e = load_json("hash_regression.json")
    vals = np.array(e["i64_inputs"], dtype=np.int64)
    h = oracle.hash_columns([("i64", vals)], len(vals))
    assert [int(x) for x in h] == e["i64_hashes"]


def test_partition_indices_row_conservation():
    """Mirrors the reference row-conservation writer test
    (sort_shuffle/writer.rs round-trip tests): every row lands in exactly one
    partition, ascending row order inside each partition
    (writer.rs:1273-1277)."""
    rng = np.random.default_rng(7)
    n, k = 10_000, 16
    keys = rng.integers(-2**62, 2**62, size=n, dtype=np.int64)
    h = oracle.hash_columns([("i64", keys)], n)
    pids = oracle.partition_ids(h, k)
    idx, offs = oracle.partition_indices(pids, k)
    assert offs[0] == 0 and offs[-1] == n
    seen = np.sort(idx)
    assert np.array_equal(seen, np.arange(n, dtype=np.uint32))
    for p in range(k):
        part = idx[offs[p]:offs[p + 1]]
        assert np.all(np.diff(part.astype(np.int64)) > 0) if len(part) > 1 else True
        assert np.all(pids[part] == p)
    # non-degenerate spread
    sizes = np.diff(offs)
    assert (sizes > 0).sum() == k


def test_multi_column_hash_and_utf8():
    """create_hashes multi-column combine + Utf8 hashing are deterministic
    and sensitive to each column (combine_hashes restatement)."""
    n = 8
    a = np.arange(n, dtype=np.int64)
    strings = [b"A", b"BB", b"x" * 20, b"", b"A", b"BB", b"yy", b"zzz"]
    data = np.frombuffer(b"".join(strings), dtype=np.uint8)
    offs = np.zeros(n + 1, dtype=np.int32)
    for i, s in enumerate(strings):
        offs[i + 1] = offs[i] + len(s)
    h1 = oracle.hash_columns([("utf8", data, offs), ("i64", a)], n)
    h2 = oracle.hash_columns([("utf8", data, offs), ("i64", a + 1)], n)
    assert not np.array_equal(h1, h2)
    # identical (string, int) pairs hash identically
    assert h1[0] != h1[4]  # differing second column
    h3 = oracle.hash_columns([("utf8", data, offs)], n)
    assert h3[0] == h3[4] and h3[1] == h3[5]  # same strings, single column


def test_nulls_leave_hash_untouched():
    """create_hashes skips null slots (hash_utils semantics): a null row in
    the second key column keeps the first column's hash."""
    n = 4
    a = np.arange(n, dtype=np.int64)
    b = np.arange(n, dtype=np.int64) + 100
    valid = np.array([0b1011], dtype=np.uint8)  # row 2 null
    h_ab = oracle.hash_columns([("i64", a), ("i64", b, valid)], n)
    h_a = oracle.hash_columns([("i64", a)], n)
    assert h_ab[2] == h_a[2]
    assert all(h_ab[i] != h_a[i] for i in (0, 1, 3))


def test_gather_roundtrip():
    rng = np.random.default_rng(3)
    src = rng.integers(0, 2**40, size=100, dtype=np.int64)
    idx = np.array([5, 1, 99, 0, 5], dtype=np.uint32)
    out = oracle.gather(src, 8, idx).view(np.int64)
    assert np.array_equal(out, src[idx])


def test_hashagg_nulls_restatement():
    """Pin the null-aware group-by restatement on a hand-computed case."""
    import numpy as np
    import oracle
    keys = np.array([1, 1, 2, 2, 3], dtype=np.int64)
    kvalid = np.array([True, True, True, False, False])
    vals = np.array([10, 20, 30, 40, 50], dtype=np.int64)
    vvalid = np.array([True, False, False, True, True])
    got = oracle.hashagg_nulls([(keys, kvalid)],
                               [("sum", vals, vvalid),
                                ("min", vals, vvalid)], 5)
    # rows: (1,10v) (1,20n) (2,30n) (None,40v) (None,50v)
    assert got == {
        (1,): (2, [10, 10], [1, 1]),
        (2,): (1, [None, None], [0, 0]),
        (None,): (2, [90, 40], [2, 2]),
    }
