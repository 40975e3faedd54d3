#!/usr/bin/env python3
"""Stage-interpreter fuzz: random tables (dtypes x NULLs x cardinalities)
through random plan trees (filter -> optional hash join -> group-by
aggregate [-> sort/limit]), every run checked against pyarrow compute.
Covers the plan grammar end to end: expression eval, LIKE, join types,
partial/final over a real shuffle file round trip (every 5th round).

Usage: python scripts/fuzz_stage.py [rounds] [seed]"""
import decimal
import os
import sys
import tempfile
import time

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

import numpy as np  # noqa: E402
import pyarrow as pa  # noqa: E402
import pyarrow.compute as pc  # noqa: E402

from datafusion_ballista_amd import gpu, stage  # noqa: E402


def rand_table(rng, n, with_utf8=True):
    cols = {"k": pa.array(rng.integers(0, max(2, n // 50), size=n,
                                       dtype=np.int64))}
    vmask = rng.random(n) < rng.choice([0.0, 0.25])
    cols["v"] = pa.array(rng.integers(-10**8, 10**8, size=n,
                                      dtype=np.int64), mask=vmask)
    cols["d"] = pa.array(rng.integers(8000, 11000, size=n,
                                      dtype=np.int32), type=pa.date32())
    cols["x"] = pa.array(rng.standard_normal(n))
    if with_utf8:
        words = ["PROMO A", "alpha green", "BRASS x", "", "special req",
                 "zed"]
        smask = rng.random(n) < rng.choice([0.0, 0.1])
        cols["s"] = pa.array(
            [None if m else words[i % len(words)] + str(i % 13)
             for i, m in enumerate(smask)])
    return pa.table(cols)


def run_round(ctx, rng, trial, tmpdir):
    n = int(rng.integers(100, 30_000))
    t = rand_table(rng, n)
    name = f"fz{trial}"
    keep = stage.register_table(ctx, name, t)
    scan = {"op": "scan", "schema": stage.schema_json(t.schema),
            "source": {"kind": "device", "table": name}}
    doc = lambda plan, **kw: {  # noqa: E731
        "job_id": f"fz{trial}", "stage_id": 1, "task_id": 0,
        "work_dir": tmpdir, "plan": plan, **kw}

    # random filter
    preds = []
    lo = int(rng.integers(8000, 10500))
    hi = lo + int(rng.integers(50, 2000))
    which = rng.integers(0, 4)
    mask = np.ones(n, dtype=bool)
    d_np = t["d"].to_numpy().astype("datetime64[D]").astype(np.int64)
    if which == 0:
        preds = [{"col": "d", "cmp": "ge_lt", "lo": lo, "hi": hi}]
        mask = (d_np >= lo) & (d_np < hi)
    elif which == 1:
        preds = [{"col": "d", "cmp": "lt", "hi": hi}]
        mask = d_np < hi
    elif which == 2:
        import re
        pat = str(rng.choice(["PROMO%", "%green%", "%x1%", "%req%"]))
        preds = [{"col": "s", "like": pat}]
        body = ".*".join(re.escape(f) for f in pat.split("%") if f)
        if pat.startswith("%"):
            body = ".*" + body
        if pat.endswith("%"):
            body = body + ".*"
        rx = re.compile(body, re.S)
        sv = t["s"].to_pylist()
        mask = np.array([v is not None and bool(rx.fullmatch(v))
                         for v in sv])
    else:
        preds = []

    ft = t.filter(pa.array(mask)) if preds else t
    plan = {"op": "filter", "predicates": preds, "input": scan} if preds \
        else scan

    # group-by aggregate
    agg_plan = {"op": "hash_aggregate", "mode": "single",
                "group_by": ["k"],
                "aggs": [{"fn": "sum", "as": "sv", "expr": {"col": "v"}},
                         {"fn": "min", "as": "mn", "expr": {"col": "v"}},
                         {"fn": "max", "as": "mxx", "expr": {"col": "x"}},
                         {"fn": "count", "as": "c"},
                         {"fn": "avg", "as": "ax", "expr": {"col": "x"}}],
                "input": plan}
    res = stage.execute(doc({"op": "collect", "input": agg_plan}))
    if ft.num_rows:
        want = ft.group_by("k").aggregate(
            [("v", "sum"), ("v", "min"), ("x", "max"), ("k", "count"),
             ("x", "mean")]).sort_by("k")
        got = sorted(res["rows"], key=lambda r: r[0])
        assert len(got) == want.num_rows, (trial, len(got), want.num_rows)
        for i, r in enumerate(got):
            assert r[0] == want["k"][i].as_py()
            assert r[1] == want["v_sum"][i].as_py(), (trial, "sum", r)
            assert r[2] == want["v_min"][i].as_py(), (trial, "min", r)
            wx = want["x_max"][i].as_py()
            assert (r[3] is None) == (wx is None)
            if wx is not None:
                assert abs(r[3] - wx) < 1e-12
            assert r[4] == want["k_count"][i].as_py()
    else:
        assert res["rows"] == []

    # every 5th: two-phase partial -> shuffle file -> final
    if trial % 5 == 0 and ft.num_rows:
        part_schema = pa.schema([
            ("k", pa.int64()), ("sv", pa.int64()), ("sv$n", pa.int64()),
            ("c", pa.int64())])
        partial = {"op": "hash_aggregate", "mode": "partial",
                   "group_by": ["k"],
                   "aggs": [{"fn": "sum", "as": "sv", "expr": {"col": "v"}},
                            {"fn": "count", "as": "c"}],
                   "input": plan}
        k = int(rng.integers(2, 6))
        r1 = stage.execute(doc(
            {"op": "sort_shuffle_write", "k": k, "keys": [{"col": "k"}],
             "input": partial},
            schema_msg_hex=stage.schema_msg_hex(part_schema)))
        pth = r1["partitions"][0]["path"]
        final = {"op": "hash_aggregate", "mode": "final", "group_by": ["k"],
                 "aggs": [{"fn": "sum", "as": "sv"},
                          {"fn": "count", "as": "c"}],
                 "input": {"op": "scan",
                           "schema": stage.schema_json(part_schema),
                           "source": {"kind": "shuffle", "data": pth,
                                      "index": pth + ".index",
                                      "partitions": list(range(k))}}}
        res2 = stage.execute(doc({"op": "collect", "input": final}))
        want = ft.group_by("k").aggregate(
            [("v", "sum"), ("k", "count")]).sort_by("k")
        got = sorted(res2["rows"], key=lambda r: r[0])
        assert len(got) == want.num_rows
        for i, r in enumerate(got):
            assert r[0] == want["k"][i].as_py()
            assert r[1] == want["v_sum"][i].as_py(), (trial, "2p sum", r)
            assert r[2] == want["k_count"][i].as_py()

    # random join type against build side
    jt = str(rng.choice(["inner", "semi", "anti", "left"]))
    nb = int(rng.integers(10, 500))
    bkeys = rng.choice(np.arange(0, max(2, n // 50), dtype=np.int64),
                       size=nb, replace=True)
    bt = pa.table({"bk": pa.array(np.unique(bkeys)),
                   "bv": pa.array(np.arange(len(np.unique(bkeys)),
                                            dtype=np.int64))})
    keep2 = stage.register_table(ctx, name + "b", bt)
    jp = {"op": "hash_join",
          "build": {"op": "scan", "schema": stage.schema_json(bt.schema),
                    "source": {"kind": "device", "table": name + "b"}},
          "probe": scan, "build_keys": ["bk"], "probe_keys": ["k"],
          "join_type": jt,
          "output": [{"side": "probe", "col": "k"}]}
    res3 = stage.execute(doc({"op": "collect", "input": jp}))
    bset = set(np.unique(bkeys).tolist())
    kv = t["k"].to_numpy()
    if jt == "inner" or jt == "semi":
        want_n = sum(1 for x in kv if int(x) in bset)
    elif jt == "anti":
        want_n = sum(1 for x in kv if int(x) not in bset)
    else:
        want_n = n
    assert len(res3["rows"]) == want_n, (trial, jt, len(res3["rows"]),
                                         want_n)
    stage.unregister_table(name)
    stage.unregister_table(name + "b")
    del keep, keep2


def main():
    rounds = int(sys.argv[1]) if len(sys.argv) > 1 else 25
    seed = int(sys.argv[2]) if len(sys.argv) > 2 else int(time.time())
    print(f"STAGE FUZZ seed={seed} rounds={rounds}", flush=True)
    rng = np.random.default_rng(seed)
    ctx = gpu.GpuStageContext(0)
    t0 = time.time()
    with tempfile.TemporaryDirectory() as td:
        for trial in range(rounds):
            run_round(ctx, rng, trial, td)
            if (trial + 1) % 10 == 0:
                print(f"round {trial+1}: ok ({time.time()-t0:.1f}s)",
                      flush=True)
    print(f"STAGE FUZZ OK: {rounds} rounds in {time.time()-t0:.1f}s",
          flush=True)


if __name__ == "__main__":
    main()
