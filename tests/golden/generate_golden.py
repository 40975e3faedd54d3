#!/usr/bin/env python3
"""Generate committed golden fixtures from the reference tree.

Run IN THE SURVEY CONTAINER ONLY (where /root/reference is mounted); the GPU
box never sees /root/reference — tests read the committed .npz/.json outputs.

Sources (all in-tree reference test vectors):
  1. ballista/client/testdata/alltypes_plain.parquet — the fixture under the
     reference's own literal `assert_batches_eq!` assertions:
       - filter:   "select string_col, timestamp_col from test where id > 4"
                   -> [("31", 2009-03-01T00:01:00), ("30", 2009-04-01T00:00:00),
                       ("31", 2009-04-01T00:01:00)]
                   (ballista/client/tests/context_checks.rs:63-77; binary
                    rendered as hex by arrow pretty-print: b"1"=31, b"0"=30)
       - groupby:  "select string_col, count(*) ... where id > 4 group by
                    string_col order by string_col" -> {30:1, 31:2}
                   (context_checks.rs:810-827)
       - join:     "select t0.id from t0 join t1 on t0.id = t1.id order by
                    t0.id desc limit 5" -> [7,6,5,4,3]
                   (context_checks.rs:986-1003, :1046-1063)
  2. ballista/scheduler/testdata/lineitem/partition0.tbl — 10 real TPC-H
     lineitem rows (SF1 head).  Used as INPUT for q1/q6-shaped semantics
     cross-checks; expected aggregates are computed here with pure-Python
     exact integer arithmetic (independent of the C oracle), per the TPC-H
     text in benchmarks/queries/q1.sql, q6.sql.  This leg is a restatement
     cross-check, not reference-pinned output (the reference ships no answer
     files in-tree; benchmarks/src/bin/tpch.rs:677-687 loads them at runtime).
"""
import json
import os

import numpy as np
import pyarrow.parquet as pq

HERE = os.path.dirname(os.path.abspath(__file__))
REF = "/root/reference"


def gen_alltypes():
    t = pq.read_table(os.path.join(REF, "ballista/client/testdata/alltypes_plain.parquet"))
    d = t.to_pydict()
    ids = np.array(d["id"], dtype=np.int32)
    string_col = [v.decode() for v in d["string_col"]]  # binary, ASCII digits
    ts_ns = np.array([v.value for v in t.column("timestamp_col").to_pylist()],
                     dtype="datetime64[ns]").astype(np.int64)
    np.savez(os.path.join(HERE, "alltypes_plain.npz"),
             id=ids, string_col=np.array(string_col),
             timestamp_ns=ts_ns)
    expected = {
        # context_checks.rs:63-77 — rows where id > 4, file order preserved
        "filter_id_gt4": {
            "string_col": ["1", "0", "1"],
            "timestamp_iso": ["2009-03-01T00:01:00", "2009-04-01T00:00:00",
                              "2009-04-01T00:01:00"],
            "pretty_hex": ["31", "30", "31"],
        },
        # context_checks.rs:810-827
        "groupby_count_id_gt4": {"0": 1, "1": 2},
        # context_checks.rs:986-1003 (self equi-join on id, desc limit 5)
        "join_ids_desc5": [7, 6, 5, 4, 3],
        "source": "ballista/client/tests/context_checks.rs:63-77,810-827,986-1003",
    }
    with open(os.path.join(HERE, "alltypes_expected.json"), "w") as f:
        json.dump(expected, f, indent=1)


LINEITEM_COLS = [
    "l_orderkey", "l_partkey", "l_suppkey", "l_linenumber", "l_quantity",
    "l_extendedprice", "l_discount", "l_tax", "l_returnflag", "l_linestatus",
    "l_shipdate", "l_commitdate", "l_receiptdate", "l_shipinstruct",
    "l_shipmode", "l_comment",
]


def date32(s):
    return (np.datetime64(s, "D") - np.datetime64("1970-01-01", "D")).astype(int)


def dec2(s):
    """'21168.23' -> scaled int at scale 2 (exact)."""
    neg = s.startswith("-")
    if neg:
        s = s[1:]
    if "." in s:
        a, b = s.split(".")
        b = (b + "00")[:2]
    else:
        a, b = s, "00"
    v = int(a) * 100 + int(b)
    return -v if neg else v


def gen_lineitem():
    rows = []
    with open(os.path.join(REF, "ballista/scheduler/testdata/lineitem/partition0.tbl")) as f:
        for line in f:
            line = line.rstrip("\n")
            if not line:
                continue
            parts = line.split("|")[:-1]
            rows.append(dict(zip(LINEITEM_COLS, parts)))

    n = len(rows)
    orderkey = np.array([int(r["l_orderkey"]) for r in rows], dtype=np.int64)
    partkey = np.array([int(r["l_partkey"]) for r in rows], dtype=np.int64)
    qty = np.array([dec2(r["l_quantity"]) for r in rows], dtype=np.int64)
    price = np.array([dec2(r["l_extendedprice"]) for r in rows], dtype=np.int64)
    disc = np.array([dec2(r["l_discount"]) for r in rows], dtype=np.int64)
    tax = np.array([dec2(r["l_tax"]) for r in rows], dtype=np.int64)
    rf = np.array([r["l_returnflag"] for r in rows])
    ls = np.array([r["l_linestatus"] for r in rows])
    shipdate = np.array([date32(r["l_shipdate"]) for r in rows], dtype=np.int32)

    np.savez(os.path.join(HERE, "lineitem_slice.npz"),
             l_orderkey=orderkey, l_partkey=partkey, l_quantity=qty,
             l_extendedprice=price, l_discount=disc, l_tax=tax,
             l_returnflag=rf, l_linestatus=ls, l_shipdate=shipdate)

    # --- q1-shaped expected (pure-Python exact ints; date <= 1998-09-02) ---
    cutoff = int(date32("1998-09-02"))
    groups = {}
    for i in range(n):
        if int(shipdate[i]) > cutoff:
            continue
        g = (rf[i], ls[i])
        e = groups.setdefault(g, {"count": 0, "sum_qty": 0, "sum_price": 0,
                                  "sum_disc_price": 0, "sum_charge": 0,
                                  "sum_disc": 0})
        p, dsc, tx, q = int(price[i]), int(disc[i]), int(tax[i]), int(qty[i])
        e["count"] += 1
        e["sum_qty"] += q
        e["sum_price"] += p
        e["sum_disc_price"] += p * (100 - dsc)       # scale 4
        e["sum_charge"] += p * (100 - dsc) * (100 + tx)  # scale 6
        e["sum_disc"] += dsc

    # --- q6-shaped expected (1994 window, disc in [5,7], qty < 24.00) ---
    d_lo, d_hi = int(date32("1994-01-01")), int(date32("1995-01-01"))
    q6_count, q6_sum = 0, 0
    for i in range(n):
        if d_lo <= int(shipdate[i]) < d_hi and 5 <= int(disc[i]) <= 7 \
                and int(qty[i]) < 2400:
            q6_count += 1
            q6_sum += int(price[i]) * int(disc[i])
    # same predicate shape over the 1996 window so the slice (all-1996 rows)
    # produces a non-empty result too
    e_lo, e_hi = int(date32("1996-01-01")), int(date32("1997-01-01"))
    q6b_count, q6b_sum = 0, 0
    for i in range(n):
        if e_lo <= int(shipdate[i]) < e_hi and 2 <= int(disc[i]) <= 7 \
                and int(qty[i]) < 4000:
            q6b_count += 1
            q6b_sum += int(price[i]) * int(disc[i])

    expected = {
        "q6_1996_window": [e_lo, e_hi],
        "q6_1996_disc": [2, 7],
        "q6_1996_qty_lt": 4000,
        "q6_1996_count": q6b_count,
        "q6_1996_sum_scale4": q6b_sum,
        "q1_cutoff_date32": cutoff,
        "q1_groups": {f"{k[0]}|{k[1]}": v for k, v in sorted(groups.items())},
        "q6_window": [d_lo, d_hi],
        "q6_count": q6_count,
        "q6_sum_scale4": q6_sum,
        "source": "ballista/scheduler/testdata/lineitem/partition0.tbl + "
                  "benchmarks/queries/q1.sql,q6.sql (pure-python cross-check)",
    }
    with open(os.path.join(HERE, "lineitem_expected.json"), "w") as f:
        json.dump(expected, f, indent=1)


def gen_hash_regression():
    """Self-generated hash vectors — REGRESSION ONLY (parity unpinned, see
    bg_ahash.h): freezes our ahash restatement so oracle/GPU drift is caught;
    does NOT pin against DataFusion."""
    import sys
    sys.path.insert(0, os.path.join(HERE, "..", ".."))
    import oracle
    vals = np.array([0, 1, -1, 42, 2**31, -2**40, 2**62, 123456789], dtype=np.int64)
    h = oracle.hash_columns([("i64", vals)], len(vals))
    out = {"i64_inputs": [int(v) for v in vals],
           "i64_hashes": [int(x) for x in h],
           "note": "regression vectors of the bg_ahash.h restatement; "
                   "NOT reference-pinned (SURVEY.md §8c: parity unpinned)"}
    with open(os.path.join(HERE, "hash_regression.json"), "w") as f:
        json.dump(out, f, indent=1)


if __name__ == "__main__":
    gen_alltypes()
    gen_lineitem()
    gen_hash_regression()
    print("golden fixtures written to", HERE)
