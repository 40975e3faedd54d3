"""GPU parity tests of the C++ stage interpreter (bg_execute_stage): whole
stages — scan -> filter/project/join/aggregate/sort -> shuffle write /
collect — driven by plan JSON, checked against pyarrow compute (the same
arrow implementation family the reference executes on) and the oracle.

The interpreter is the product path of SURVEY.md §8b seam 1: one C call
per task, mirroring QueryStageExecutor::execute_query_stage
(executor/src/execution_engine.rs:78-103)."""
import decimal
import json
import os

import numpy as np
import pyarrow as pa
import pyarrow.compute as pc
import pytest

import oracle
from datafusion_ballista_amd import gpu, shuffle, stage

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ctx():
    c = gpu.GpuStageContext(0)
    yield c
    c.close()


@pytest.fixture()
def reg(ctx):
    """Register tables for a test; unregister afterwards."""
    keep = []
    names = []

    def _reg(name, table):
        keep.append(stage.register_table(ctx, name, table))
        names.append(name)
        return table

    yield _reg
    for n in names:
        stage.unregister_table(n)


def _doc(plan, **kw):
    d = {"job_id": "job-s", "stage_id": 1, "task_id": 0,
         "work_dir": "/tmp/stage-tests", "plan": plan}
    d.update(kw)
    return d


def scan_of(table, name):
    return {"op": "scan", "schema": stage.schema_json(table.schema),
            "source": {"kind": "device", "table": name}}


def make_lineitem(n, seed=0):
    rng = np.random.default_rng(seed)
    qty = rng.integers(100, 5100, size=n)          # scale 2
    price = rng.integers(90000, 10500000, size=n)  # scale 2
    disc = rng.integers(0, 11, size=n) * 100       # scale 4-ish (use 2)
    ship = rng.integers(8400, 9800, size=n).astype(np.int32)
    okey = rng.integers(0, n // 4 + 1, size=n, dtype=np.int64)
    return pa.table({
        "l_orderkey": pa.array(okey),
        "l_quantity": pa.array([decimal.Decimal(int(v)) / 100 for v in qty],
                               type=pa.decimal128(15, 2)),
        "l_extendedprice": pa.array(
            [decimal.Decimal(int(v)) / 100 for v in price],
            type=pa.decimal128(15, 2)),
        "l_discount": pa.array([decimal.Decimal(int(v)) / 100 for v in disc],
                               type=pa.decimal128(15, 2)),
        "l_shipdate": pa.array(ship, type=pa.date32()),
    })


def test_collect_filter_project(ctx, reg):
    t = make_lineitem(50_000, seed=1)
    reg("li", t)
    plan = {"op": "collect", "input": {
        "op": "project", "exprs": [
            {"as": "okey", "expr": {"col": "l_orderkey"}},
            {"as": "rev", "expr": {"mul": [{"col": "l_extendedprice"},
                                           {"col": "l_discount"}]}},
        ],
        "input": {"op": "filter", "predicates": [
            {"col": "l_shipdate", "cmp": "ge_lt", "lo": 8500, "hi": 8600}],
            "input": scan_of(t, "li")}}}
    res = stage.execute(_doc(plan))
    mask = pc.and_(pc.greater_equal(t["l_shipdate"], pa.scalar(8500,
                                                               pa.date32())),
                   pc.less(t["l_shipdate"], pa.scalar(8600, pa.date32())))
    want = t.filter(mask)
    assert len(res["rows"]) == want.num_rows
    got_keys = [r[0] for r in res["rows"]]
    assert got_keys == want["l_orderkey"].to_pylist()
    # revenue = price * discount exactly (scale 4)
    wp = want["l_extendedprice"].to_pylist()
    wd = want["l_discount"].to_pylist()
    got_rev = [int(r[1]) for r in res["rows"]]
    want_rev = [int(p.scaleb(2)) * int(d.scaleb(2)) for p, d in zip(wp, wd)]
    assert got_rev == want_rev


def test_q6_stage_fused_matches_oracle(ctx, reg):
    from datafusion_ballista_amd import tpch_synth
    n = 200_000
    li = tpch_synth.lineitem_numpy(n, seed=7)
    d16 = tpch_synth.dec128_pairs_np(li["l_discount"]).view(np.uint8).reshape(-1)
    q16 = tpch_synth.dec128_pairs_np(li["l_quantity"]).view(np.uint8).reshape(-1)
    p16 = tpch_synth.dec128_pairs_np(li["l_extendedprice"]).view(np.uint8).reshape(-1)

    def dec_col(raw16):
        vals = [int.from_bytes(bytes(raw16[16*i:16*(i+1)]), "little",
                               signed=True) for i in range(n)]
        return pa.array([decimal.Decimal(v).scaleb(-2) for v in vals],
                        type=pa.decimal128(15, 2))

    t = pa.table({
        "l_shipdate": pa.array(li["l_shipdate"], type=pa.date32()),
        "l_discount": dec_col(d16),
        "l_quantity": dec_col(q16),
        "l_extendedprice": dec_col(p16),
    })
    reg("li6", t)
    plan = {"op": "collect", "input": {
        "op": "hash_aggregate", "mode": "single", "group_by": [],
        "aggs": [
            {"fn": "sum", "as": "revenue",
             "expr": {"mul": [{"col": "l_extendedprice"},
                              {"col": "l_discount"}]}},
            {"fn": "count", "as": "cnt"},
        ],
        "input": {"op": "filter", "predicates": [
            {"col": "l_shipdate", "cmp": "ge_lt",
             "lo": tpch_synth.Q6_DATE_LO, "hi": tpch_synth.Q6_DATE_HI},
            {"col": "l_discount", "cmp": "between",
             "lo": tpch_synth.Q6_DISC_LO, "hi": tpch_synth.Q6_DISC_HI},
            {"col": "l_quantity", "cmp": "lt", "hi": tpch_synth.Q6_QTY_LT}],
            "input": scan_of(t, "li6")}}}
    res = stage.execute(_doc(plan))
    want_cnt, want_sum = oracle.q6(li["l_shipdate"], d16, q16, p16,
                                   tpch_synth.Q6_DATE_LO,
                                   tpch_synth.Q6_DATE_HI,
                                   tpch_synth.Q6_DISC_LO,
                                   tpch_synth.Q6_DISC_HI,
                                   tpch_synth.Q6_QTY_LT)
    assert len(res["rows"]) == 1
    assert int(res["rows"][0][0]) == want_sum
    assert res["rows"][0][1] == want_cnt
    # the fused kernel must actually have run (kernel_ms recorded)
    assert res["metrics"]["gpu_kernel_ms"] > 0


def test_grouped_aggregate_with_nulls(ctx, reg):
    rng = np.random.default_rng(11)
    n = 80_000
    keys = rng.integers(0, 500, size=n, dtype=np.int64)
    vals = rng.integers(-10**6, 10**6, size=n, dtype=np.int64)
    vmask = rng.random(n) < 0.3  # 30% NULL values
    f64 = rng.standard_normal(n)
    t = pa.table({
        "k": pa.array(keys),
        "v": pa.array(vals, mask=vmask),
        "x": pa.array(f64),
    })
    reg("g", t)
    plan = {"op": "collect", "input": {
        "op": "hash_aggregate", "mode": "single", "group_by": ["k"],
        "aggs": [
            {"fn": "sum", "as": "sv", "expr": {"col": "v"}},
            {"fn": "min", "as": "mn", "expr": {"col": "v"}},
            {"fn": "max", "as": "mx", "expr": {"col": "v"}},
            {"fn": "count", "as": "c"},
            {"fn": "avg", "as": "ax", "expr": {"col": "x"}},
        ], "input": scan_of(t, "g")}}
    res = stage.execute(_doc(plan))
    want = t.group_by("k").aggregate([
        ("v", "sum"), ("v", "min"), ("v", "max"), ("k", "count"),
        ("x", "mean")]).sort_by("k")
    got = sorted(res["rows"], key=lambda r: r[0])
    assert len(got) == want.num_rows
    for i, r in enumerate(got):
        assert r[0] == want["k"][i].as_py()
        assert r[1] == want["v_sum"][i].as_py()
        assert r[2] == want["v_min"][i].as_py()
        assert r[3] == want["v_max"][i].as_py()
        assert r[4] == want["k_count"][i].as_py()
        assert abs(r[5] - want["x_mean"][i].as_py()) < 1e-9


def test_grouped_aggregate_low_cardinality_lds(ctx, reg):
    """~50 groups over 200k rows: hits the per-block LDS aggregate path
    (k_hashagg_lds; the q1-class shape whose global-atomic version was
    346x slower at SF100) — values incl. NULLs and min/max/avg must match
    pyarrow exactly."""
    rng = np.random.default_rng(53)
    n = 200_000
    keys = rng.integers(0, 50, size=n, dtype=np.int64)
    vals = rng.integers(-10**7, 10**7, size=n, dtype=np.int64)
    vmask = rng.random(n) < 0.25
    f64 = rng.standard_normal(n)
    t = pa.table({"k": pa.array(keys), "v": pa.array(vals, mask=vmask),
                  "x": pa.array(f64)})
    reg("lds", t)
    plan = {"op": "collect", "input": {
        "op": "hash_aggregate", "mode": "single", "group_by": ["k"],
        "aggs": [
            {"fn": "sum", "as": "sv", "expr": {"col": "v"}},
            {"fn": "min", "as": "mn", "expr": {"col": "v"}},
            {"fn": "max", "as": "mx", "expr": {"col": "v"}},
            {"fn": "min", "as": "mnx", "expr": {"col": "x"}},
            {"fn": "max", "as": "mxx", "expr": {"col": "x"}},
            {"fn": "count", "as": "c"},
        ], "input": scan_of(t, "lds")}}
    res = stage.execute(_doc(plan))
    want = t.group_by("k").aggregate([
        ("v", "sum"), ("v", "min"), ("v", "max"), ("x", "min"),
        ("x", "max"), ("k", "count")]).sort_by("k")
    got = sorted(res["rows"], key=lambda r: r[0])
    assert len(got) == want.num_rows == 50
    for i, r in enumerate(got):
        assert r[0] == want["k"][i].as_py()
        assert r[1] == want["v_sum"][i].as_py()
        assert r[2] == want["v_min"][i].as_py()
        assert r[3] == want["v_max"][i].as_py()
        assert r[4] == want["x_min"][i].as_py()
        assert r[5] == want["x_max"][i].as_py()
        assert r[6] == want["k_count"][i].as_py()


def test_join_stage(ctx, reg):
    rng = np.random.default_rng(13)
    nb, np_ = 3_000, 40_000
    ck = np.arange(nb, dtype=np.int64)
    seg = rng.integers(0, 5, size=nb).astype(np.uint8)
    ok = rng.integers(0, nb * 2, size=np_, dtype=np.int64)  # half miss
    od = rng.integers(9000, 9400, size=np_).astype(np.int32)
    cust = pa.table({"c_custkey": pa.array(ck), "c_seg": pa.array(seg)})
    orders = pa.table({"o_custkey": pa.array(ok),
                       "o_orderdate": pa.array(od, type=pa.date32()),
                       "o_orderkey": pa.array(
                           np.arange(np_, dtype=np.int64))})
    reg("cust", cust)
    reg("ord", orders)
    plan = {"op": "collect", "input": {
        "op": "hash_join",
        "build": {"op": "filter",
                  "predicates": [{"col": "c_seg", "cmp": "eq", "lo": 2}],
                  "input": scan_of(cust, "cust")},
        "probe": scan_of(orders, "ord"),
        "build_keys": ["c_custkey"], "probe_keys": ["o_custkey"],
        "join_type": "inner",
        "output": [{"side": "probe", "col": "o_orderkey"},
                   {"side": "probe", "col": "o_orderdate"},
                   {"side": "build", "col": "c_custkey"}]}}
    res = stage.execute(_doc(plan))
    want_keys = set(ck[seg == 2].tolist())
    want_pairs = [(int(k), int(d), int(c)) for k, d, c in
                  zip(np.arange(np_), od, ok) if c in want_keys]
    got_pairs = [(r[0], r[1], r[2]) for r in res["rows"]]
    assert sorted(got_pairs) == sorted(want_pairs)


def test_sort_topk_stage(ctx, reg):
    rng = np.random.default_rng(17)
    n = 30_000
    t = pa.table({"a": pa.array(rng.integers(0, 10**9, size=n,
                                             dtype=np.int64)),
                  "b": pa.array(rng.integers(0, 100, size=n,
                                             dtype=np.int64))})
    reg("s", t)
    plan = {"op": "collect", "input": {
        "op": "sort", "keys": [{"col": "a", "desc": True}], "limit": 25,
        "input": scan_of(t, "s")}}
    res = stage.execute(_doc(plan))
    want = t.sort_by([("a", "descending")]).slice(0, 25)
    assert [r[0] for r in res["rows"]] == want["a"].to_pylist()


def test_sort_shuffle_write_roundtrip(ctx, reg, tmp_path):
    """The consolidated file the C++ writer emits must be byte-compatible:
    pyarrow's own IPC reader (via shuffle.read_partition) reconstructs
    every partition exactly, and the index/partition assignment matches
    the oracle's compute_partition_indices restatement."""
    rng = np.random.default_rng(19)
    n, k = 57_000, 8
    keys = rng.integers(-2**62, 2**62, size=n, dtype=np.int64)
    vals = rng.integers(-10**9, 10**9, size=n, dtype=np.int64)
    vmask = rng.random(n) < 0.15
    strs = [f"s{int(v) % 997}" if v % 7 else "" for v in vals]
    t = pa.table({
        "k": pa.array(keys),
        "v": pa.array(vals, mask=vmask),
        "s": pa.array(strs, type=pa.string()),
    })
    reg("sw", t)
    doc = _doc({"op": "sort_shuffle_write", "k": k,
                "keys": [{"col": "k"}],
                "input": scan_of(t, "sw")},
               work_dir=str(tmp_path), stage_id=3, task_id=5,
               schema_msg_hex=stage.schema_msg_hex(t.schema),
               batch_size=4096)
    res = stage.execute(doc)
    parts = res["partitions"]
    assert len(parts) == k
    assert sum(p["num_rows"] for p in parts) == n
    data_path = parts[0]["path"]
    assert data_path.endswith(
        os.path.join(str(tmp_path), "job-s", "3", "5", "data.arrow"))

    # oracle partition assignment
    h = oracle.hash_columns([("i64", keys)], n)
    pids = oracle.partition_ids(h, k)
    idx, offs = oracle.partition_indices(pids, k)

    schema = shuffle.read_schema(data_path)
    assert schema.equals(t.schema)
    for p in range(k):
        batches = shuffle.read_partition(data_path, data_path + ".index", p)
        got = (pa.Table.from_batches(batches, schema=schema)
               if batches else t.schema.empty_table())
        rows = idx[offs[p]:offs[p + 1]]
        want = t.take(pa.array(rows, type=pa.uint32()))
        assert got.equals(want), f"partition {p}"
        assert got.num_rows == parts[p]["num_rows"]
        nb_expect = (got.num_rows + 4095) // 4096
        assert parts[p]["num_batches"] == nb_expect


def test_two_phase_aggregate_over_shuffle(ctx, reg, tmp_path):
    """Partial aggregate -> sort_shuffle_write -> (scan shuffle) -> final
    aggregate == single-shot aggregate: the whole two-stage exchange runs
    through the interpreter, files in between (the reference's execution
    shape for FinalPartitioned)."""
    rng = np.random.default_rng(23)
    n, k = 120_000, 4
    keys = rng.integers(0, 3_000, size=n, dtype=np.int64)
    vals = rng.integers(-10**8, 10**8, size=n, dtype=np.int64)
    vmask = rng.random(n) < 0.2
    t = pa.table({"k": pa.array(keys), "v": pa.array(vals, mask=vmask)})
    reg("tp", t)

    partial = {"op": "hash_aggregate", "mode": "partial", "group_by": ["k"],
               "aggs": [{"fn": "sum", "as": "sv", "expr": {"col": "v"}},
                        {"fn": "count", "as": "c"},
                        {"fn": "avg", "as": "av", "expr": {"col": "v"}}],
               "input": scan_of(t, "tp")}
    part_schema = pa.schema([
        ("k", pa.int64()), ("sv", pa.decimal128(38, 0)), ("sv$n", pa.int64()),
        ("c", pa.int64()), ("av$s", pa.decimal128(38, 0)),
        ("av$n", pa.int64())])
    # NOTE: SUM(Int64) partial is int64 in Arrow; here we declare the
    # partial sum columns as the interpreter emits them (int64)
    part_schema = pa.schema([
        ("k", pa.int64()), ("sv", pa.int64()), ("sv$n", pa.int64()),
        ("c", pa.int64()), ("av$s", pa.int64()), ("av$n", pa.int64())])
    doc1 = _doc({"op": "sort_shuffle_write", "k": k,
                 "keys": [{"col": "k"}], "input": partial},
                work_dir=str(tmp_path), stage_id=1, task_id=0,
                schema_msg_hex=stage.schema_msg_hex(part_schema))
    res1 = stage.execute(doc1)
    data_path = res1["partitions"][0]["path"]

    # stage 2: scan every partition of the shuffle file, final-aggregate
    sc2 = {"op": "scan",
           "schema": stage.schema_json(part_schema),
           "source": {"kind": "shuffle", "data": data_path,
                      "index": data_path + ".index",
                      "partitions": list(range(k))}}
    final = {"op": "hash_aggregate", "mode": "final", "group_by": ["k"],
             "aggs": [{"fn": "sum", "as": "sv"}, {"fn": "count", "as": "c"},
                      {"fn": "avg", "as": "av"}], "input": sc2}
    res2 = stage.execute(_doc({"op": "collect", "input": final}))

    df = t.group_by("k").aggregate(
        [("v", "sum"), ("k", "count"), ("v", "mean")]).sort_by("k")
    got = sorted(res2["rows"], key=lambda r: r[0])
    assert len(got) == df.num_rows
    for i, r in enumerate(got):
        assert r[0] == df["k"][i].as_py()
        assert r[1] == df["v_sum"][i].as_py()
        assert r[2] == df["k_count"][i].as_py()
        want_avg = df["v_mean"][i].as_py()
        # decimal avg at scale +4 vs float mean
        assert abs(float(r[3]) / 10**4 - want_avg) < 0.51 / 10**4


def test_passthrough_write_stage(ctx, reg, tmp_path):
    rng = np.random.default_rng(29)
    n = 23_000
    t = pa.table({"a": pa.array(rng.integers(0, 10**9, size=n,
                                             dtype=np.int64)),
                  "s": pa.array([f"v{i%311}" for i in range(n)])})
    reg("pt", t)
    doc = _doc({"op": "passthrough_write", "global_partition": 7,
                "input": scan_of(t, "pt")},
               work_dir=str(tmp_path), stage_id=2, task_id=9,
               schema_msg_hex=stage.schema_msg_hex(t.schema))
    res = stage.execute(doc)
    path = res["partitions"][0]["path"]
    assert path.endswith(os.path.join("2", "7", "data-9.arrow"))
    got = pa.Table.from_batches(shuffle.read_passthrough_partition(path),
                                schema=t.schema)
    assert got.equals(t)


def test_q3_stage_real_group_key(ctx, reg):
    """TPC-H q3 as ONE plan-driven stage with its REAL shape
    (approved/q3.txt): customer(mktsegment=BUILDING) ⨝ orders(date<cutoff)
    ⨝ lineitem(date>cutoff), GROUP BY (l_orderkey, o_orderdate,
    o_shippriority) — the 3-column key the reference groups on, not the
    single-key stand-in of round 1 — then ORDER BY revenue DESC,
    o_orderdate ASC LIMIT 10.  Per-group values checked exactly."""
    rng = np.random.default_rng(37)
    ncust, nord, nli = 20_000, 100_000, 400_000
    cutoff = 9204  # 1995-03-15

    c_custkey = np.arange(1, ncust + 1, dtype=np.int64)
    c_seg = rng.integers(0, 5, size=ncust).astype(np.uint8)
    o_orderkey = np.arange(1, nord + 1, dtype=np.int64)
    o_custkey = rng.integers(1, ncust + 1, size=nord, dtype=np.int64)
    o_orderdate = rng.integers(8900, 9500, size=nord, dtype=np.int32)
    o_shipprio = rng.integers(0, 3, size=nord, dtype=np.int32)
    l_orderkey = rng.integers(1, nord + 1, size=nli, dtype=np.int64)
    l_shipdate = rng.integers(8900, 9500, size=nli, dtype=np.int32)
    l_price = rng.integers(90000, 10495100, size=nli, dtype=np.int64)
    l_disc = rng.integers(0, 11, size=nli, dtype=np.int64)

    cust = pa.table({"c_custkey": pa.array(c_custkey),
                     "c_mktsegment": pa.array(c_seg)})
    orders = pa.table({"o_orderkey": pa.array(o_orderkey),
                       "o_custkey": pa.array(o_custkey),
                       "o_orderdate": pa.array(o_orderdate,
                                               type=pa.date32()),
                       "o_shippriority": pa.array(o_shipprio)})
    li = pa.table({
        "l_orderkey": pa.array(l_orderkey),
        "l_shipdate": pa.array(l_shipdate, type=pa.date32()),
        "l_extendedprice": pa.array(
            [decimal.Decimal(int(v)) / 100 for v in l_price],
            type=pa.decimal128(15, 2)),
        "l_discount": pa.array(
            [decimal.Decimal(int(v)) / 100 for v in l_disc],
            type=pa.decimal128(15, 2)),
    })
    reg("q3c", cust)
    reg("q3o", orders)
    reg("q3l", li)

    join1 = {"op": "hash_join",
             "build": {"op": "filter",
                       "predicates": [{"col": "c_mktsegment", "cmp": "eq",
                                       "lo": 1}],  # BUILDING
                       "input": scan_of(cust, "q3c")},
             "probe": {"op": "filter",
                       "predicates": [{"col": "o_orderdate", "cmp": "lt",
                                       "hi": cutoff}],
                       "input": scan_of(orders, "q3o")},
             "build_keys": ["c_custkey"], "probe_keys": ["o_custkey"],
             "join_type": "inner",
             "output": [{"side": "probe", "col": "o_orderkey"},
                        {"side": "probe", "col": "o_orderdate"},
                        {"side": "probe", "col": "o_shippriority"}]}
    join2 = {"op": "hash_join",
             "build": join1,
             "probe": {"op": "filter",
                       "predicates": [{"col": "l_shipdate", "cmp": "gt",
                                       "lo": cutoff}],
                       "input": scan_of(li, "q3l")},
             "build_keys": ["o_orderkey"], "probe_keys": ["l_orderkey"],
             "join_type": "inner",
             "output": [{"side": "probe", "col": "l_orderkey"},
                        {"side": "build", "col": "o_orderdate"},
                        {"side": "build", "col": "o_shippriority"},
                        {"side": "probe", "col": "l_extendedprice"},
                        {"side": "probe", "col": "l_discount"}]}
    agg = {"op": "hash_aggregate", "mode": "single",
           "group_by": ["l_orderkey", "o_orderdate", "o_shippriority"],
           "aggs": [{"fn": "sum", "as": "revenue",
                     "expr": {"mul": [{"col": "l_extendedprice"},
                                      {"sub": [{"lit": 100},
                                               {"col": "l_discount"}]}]}}],
           "input": join2}
    top = {"op": "sort", "keys": [{"col": "revenue", "desc": True},
                                  {"col": "o_orderdate", "desc": False}],
           "limit": 10, "input": agg}
    res = stage.execute(_doc({"op": "collect", "input": top}))

    # exact python restatement (per-group, not just global invariants)
    keep_cust = set(c_custkey[c_seg == 1].tolist())
    omult = {}
    for i in range(nord):
        if o_orderdate[i] < cutoff and int(o_custkey[i]) in keep_cust:
            key = int(o_orderkey[i])
            omult.setdefault(key, []).append(
                (int(o_orderdate[i]), int(o_shipprio[i])))
    want = {}
    for i in range(nli):
        if l_shipdate[i] <= cutoff:
            continue
        okey = int(l_orderkey[i])
        if okey not in omult:
            continue
        rev = int(l_price[i]) * (100 - int(l_disc[i]))
        for (odate, oprio) in omult[okey]:
            g = (okey, odate, oprio)
            want[g] = want.get(g, 0) + rev
    assert want, "degenerate test"
    top10 = sorted(want.items(), key=lambda kv: (-kv[1], kv[0][1]))[:10]
    got = [((r[0], r[1], r[2]), int(r[3])) for r in res["rows"]]
    # revenue DESC is a total order on values; ties broken by o_orderdate
    assert [g[1] for g in got] == [w[1] for w in top10]
    assert sorted(g[0] for g in got) == sorted(w[0] for w in top10) or \
        [g[0] for g in got] == [w[0] for w in top10]


def _join_plan(build_scan, probe_scan, bk, pk, jt, output):
    return {"op": "hash_join", "build": build_scan, "probe": probe_scan,
            "build_keys": bk, "probe_keys": pk, "join_type": jt,
            "output": output}


def test_join_types_semi_anti_left(ctx, reg):
    """Probe-side join types (q2/q9/q16-class shapes): semi, anti, and
    left (probe side preserved, build side NULL for unmatched), with NULL
    probe keys — null_equals_null=false: nulls never match, so they are
    excluded by semi/inner and EMITTED by anti/left."""
    rng = np.random.default_rng(41)
    nb, np_ = 500, 20_000
    bkeys = np.arange(0, 2 * nb, 2, dtype=np.int64)  # evens only
    bval = rng.integers(0, 100, size=nb, dtype=np.int64)
    pkeys = rng.integers(0, 2 * nb, size=np_, dtype=np.int64)
    pmask = rng.random(np_) < 0.1  # 10% NULL probe keys
    build = pa.table({"bk": pa.array(bkeys), "bv": pa.array(bval)})
    probe = pa.table({"pk": pa.array(pkeys, mask=pmask),
                      "pid": pa.array(np.arange(np_, dtype=np.int64))})
    reg("jb", build)
    reg("jp", probe)
    bs, ps = scan_of(build, "jb"), scan_of(probe, "jp")

    matched = {int(k) for k in bkeys}
    pk_list = [None if m else int(k) for k, m in zip(pkeys, pmask)]

    # SEMI: probe rows with a match, once
    res = stage.execute(_doc({"op": "collect", "input": _join_plan(
        bs, ps, ["bk"], ["pk"], "semi",
        [{"side": "probe", "col": "pid"}])}))
    want = [i for i, k in enumerate(pk_list)
            if k is not None and k in matched]
    assert sorted(r[0] for r in res["rows"]) == want

    # ANTI: probe rows with no match, including NULL-key rows
    res = stage.execute(_doc({"op": "collect", "input": _join_plan(
        bs, ps, ["bk"], ["pk"], "anti",
        [{"side": "probe", "col": "pid"}])}))
    want = [i for i, k in enumerate(pk_list) if k is None or k not in matched]
    assert sorted(r[0] for r in res["rows"]) == want

    # LEFT (probe preserved): every probe row; unmatched build side -> NULL
    res = stage.execute(_doc({"op": "collect", "input": _join_plan(
        bs, ps, ["bk"], ["pk"], "left",
        [{"side": "probe", "col": "pid"},
         {"side": "build", "col": "bv"}])}))
    assert len(res["rows"]) == np_
    bmap = {int(k): int(v) for k, v in zip(bkeys, bval)}
    for pid, bv in res["rows"]:
        k = pk_list[pid]
        if k is None or k not in bmap:
            assert bv is None
        else:
            assert bv == bmap[k]


def test_join_utf8_and_composite_keys(ctx, reg):
    """Utf8 and composite (Int64, Utf8) join keys through the generalized
    build/probe — the q2/q9-class key shapes round 1 lacked."""
    rng = np.random.default_rng(43)
    nb, np_ = 800, 30_000
    names = [f"part#{i:05d}" for i in range(nb)]
    bval = rng.integers(0, 10**6, size=nb, dtype=np.int64)
    build = pa.table({"name": pa.array(names), "bv": pa.array(bval)})
    p_name = [names[i] if i < nb else f"miss#{i}"
              for i in rng.integers(0, nb + 200, size=np_)]
    probe = pa.table({"pname": pa.array(p_name),
                      "pid": pa.array(np.arange(np_, dtype=np.int64))})
    reg("ub", build)
    reg("up", probe)
    res = stage.execute(_doc({"op": "collect", "input": _join_plan(
        scan_of(build, "ub"), scan_of(probe, "up"),
        ["name"], ["pname"], "inner",
        [{"side": "probe", "col": "pid"},
         {"side": "build", "col": "bv"}])}))
    bmap = dict(zip(names, (int(v) for v in bval)))
    want = sorted((i, bmap[nm]) for i, nm in enumerate(p_name)
                  if nm in bmap)
    got = sorted((r[0], r[1]) for r in res["rows"])
    assert got == want

    # composite (int64, utf8) key
    b2 = pa.table({"k1": pa.array(np.arange(nb, dtype=np.int64) % 50),
                   "k2": pa.array(names),
                   "bv": pa.array(bval)})
    p2 = pa.table({"q1": pa.array(
        rng.integers(0, 50, size=np_, dtype=np.int64)),
        "q2": pa.array(p_name),
        "pid": pa.array(np.arange(np_, dtype=np.int64))})
    reg("cb", b2)
    reg("cp", p2)
    res = stage.execute(_doc({"op": "collect", "input": _join_plan(
        scan_of(b2, "cb"), scan_of(p2, "cp"),
        ["k1", "k2"], ["q1", "q2"], "inner",
        [{"side": "probe", "col": "pid"},
         {"side": "build", "col": "bv"}])}))
    bmap2 = {(int(i % 50), names[i]): int(bval[i]) for i in range(nb)}
    want2 = sorted((i, bmap2[(int(q1), q2)])
                   for i, (q1, q2) in enumerate(zip(p2["q1"].to_numpy(),
                                                    p_name))
                   if (int(q1), q2) in bmap2)
    got2 = sorted((r[0], r[1]) for r in res["rows"])
    assert got2 == want2


def test_join_chunked_probe_matches_unchunked(ctx, reg):
    """probe_chunk_rows (the join-temporary spill bound, writer.rs:650-686
    analogue): slicing the probe side must reproduce the unchunked result
    exactly — incl. Utf8 payloads, NULL probe keys, and a left join whose
    unmatched rows carry NULL build columns across chunk boundaries."""
    rng = np.random.default_rng(47)
    nb, np_ = 700, 25_000
    bkeys = np.arange(0, 2 * nb, 2, dtype=np.int64)
    bstr = [f"b{int(k) % 313}" for k in bkeys]
    pkeys = rng.integers(0, 2 * nb, size=np_, dtype=np.int64)
    pmask = rng.random(np_) < 0.08
    pstr = [f"p{i % 511}" for i in range(np_)]
    build = pa.table({"bk": pa.array(bkeys), "bs": pa.array(bstr)})
    probe = pa.table({"pk": pa.array(pkeys, mask=pmask),
                      "ps": pa.array(pstr),
                      "pid": pa.array(np.arange(np_, dtype=np.int64))})
    reg("ckb", build)
    reg("ckp", probe)
    for jt in ("inner", "left"):
        plans = []
        for chunk in (0, 4096):
            jp = _join_plan(scan_of(build, "ckb"), scan_of(probe, "ckp"),
                            ["bk"], ["pk"], jt,
                            [{"side": "probe", "col": "pid"},
                             {"side": "probe", "col": "ps"},
                             {"side": "build", "col": "bs"}])
            if chunk:
                jp = dict(jp, probe_chunk_rows=chunk)
            plans.append(stage.execute(_doc({"op": "collect", "input": jp})))
        r0 = sorted(map(tuple, plans[0]["rows"]))
        r1 = sorted(map(tuple, plans[1]["rows"]))
        assert r0 == r1, jt


def test_q3_staged_pipeline_with_shuffle_files(ctx, reg, tmp_path):
    """q3 as the reference actually RUNS it: four tasks exchanging real
    consolidated shuffle files (the staged execution of
    execution_graph.rs:62-105 — each stage's output is a durable
    addressable artifact, shuffle.md:66-68), every stage one
    bg_execute_stage call:
      stage 1: customer filter -> sort_shuffle_write (hash c_custkey, k=4)
      stage 2: orders filter -> sort_shuffle_write (hash o_custkey, k=4)
      stage 3 (x4 tasks, one per partition p): scan both shuffles at p ->
               join -> partial agg -> sort_shuffle_write (hash group key)
      stage 4: scan stage-3 shuffles -> final agg -> top-10
    Cross-checked against the single-stage q3 plan on the same data."""
    rng = np.random.default_rng(59)
    ncust, nord, nli = 8_000, 60_000, 240_000
    cutoff = 9204
    cust = pa.table({
        "c_custkey": pa.array(np.arange(1, ncust + 1, dtype=np.int64)),
        "c_mktsegment": pa.array(rng.integers(0, 5, size=ncust)
                                 .astype(np.uint8))})
    orders = pa.table({
        "o_orderkey": pa.array(np.arange(1, nord + 1, dtype=np.int64)),
        "o_custkey": pa.array(rng.integers(1, ncust + 1, size=nord,
                                           dtype=np.int64)),
        "o_orderdate": pa.array(rng.integers(8900, 9500, size=nord,
                                             dtype=np.int32),
                                type=pa.date32()),
        "o_shippriority": pa.array(np.zeros(nord, dtype=np.int32))})
    li = pa.table({
        "l_orderkey": pa.array(rng.integers(1, nord + 1, size=nli,
                                            dtype=np.int64)),
        "l_shipdate": pa.array(rng.integers(8900, 9500, size=nli,
                                            dtype=np.int32),
                               type=pa.date32()),
        "l_extendedprice": pa.array(
            [decimal.Decimal(int(v)) / 100 for v in
             rng.integers(90000, 10495100, size=nli)],
            type=pa.decimal128(15, 2)),
        "l_discount": pa.array(
            [decimal.Decimal(int(v)) / 100 for v in
             rng.integers(0, 11, size=nli)],
            type=pa.decimal128(15, 2))})
    reg("sgc", cust)
    reg("sgo", orders)
    reg("sgl", li)
    K = 4
    wd = str(tmp_path)

    def doc(plan, stage_id, task_id=0, schema=None, **kw):
        d = {"job_id": "q3stg", "stage_id": stage_id, "task_id": task_id,
             "work_dir": wd, "plan": plan}
        if schema is not None:
            d["schema_msg_hex"] = stage.schema_msg_hex(schema)
        d.update(kw)
        return d

    # stage 1: filtered customer, hash-partitioned by c_custkey
    s1_schema = pa.schema([("c_custkey", pa.int64())])
    r1 = stage.execute(doc({
        "op": "sort_shuffle_write", "k": K, "keys": [{"col": "c_custkey"}],
        "input": {"op": "project",
                  "exprs": [{"as": "c_custkey", "expr": {"col": "c_custkey"}}],
                  "input": {"op": "filter",
                            "predicates": [{"col": "c_mktsegment",
                                            "cmp": "eq", "lo": 1}],
                            "input": scan_of(cust, "sgc")}}},
        1, schema=s1_schema))
    s1_path = r1["partitions"][0]["path"]

    # stage 2: filtered orders, hash-partitioned by o_custkey
    s2_schema = pa.schema([("o_orderkey", pa.int64()),
                           ("o_custkey", pa.int64()),
                           ("o_orderdate", pa.date32()),
                           ("o_shippriority", pa.int32())])
    r2 = stage.execute(doc({
        "op": "sort_shuffle_write", "k": K, "keys": [{"col": "o_custkey"}],
        "input": {"op": "filter",
                  "predicates": [{"col": "o_orderdate", "cmp": "lt",
                                  "hi": cutoff}],
                  "input": scan_of(orders, "sgo")}},
        2, schema=s2_schema))
    s2_path = r2["partitions"][0]["path"]

    # stage 3 (one task per partition): co-partitioned join + partial agg,
    # output hash-partitioned by l_orderkey for the final stage
    s3_schema = pa.schema([
        ("l_orderkey", pa.int64()), ("o_orderdate", pa.date32()),
        ("o_shippriority", pa.int32()),
        ("revenue", pa.decimal128(38, 4)), ("revenue$n", pa.int64())])
    s3_paths = []
    for p in range(K):
        join1 = {"op": "hash_join",
                 "build": {"op": "scan",
                           "schema": stage.schema_json(s1_schema),
                           "source": {"kind": "shuffle", "data": s1_path,
                                      "index": s1_path + ".index",
                                      "partitions": [p]}},
                 "probe": {"op": "scan",
                           "schema": stage.schema_json(s2_schema),
                           "source": {"kind": "shuffle", "data": s2_path,
                                      "index": s2_path + ".index",
                                      "partitions": [p]}},
                 "build_keys": ["c_custkey"], "probe_keys": ["o_custkey"],
                 "join_type": "inner",
                 "output": [{"side": "probe", "col": "o_orderkey"},
                            {"side": "probe", "col": "o_orderdate"},
                            {"side": "probe", "col": "o_shippriority"}]}
        join2 = {"op": "hash_join", "build": join1,
                 "probe": {"op": "filter",
                           "predicates": [{"col": "l_shipdate", "cmp": "gt",
                                           "lo": cutoff}],
                           "input": scan_of(li, "sgl")},
                 "build_keys": ["o_orderkey"], "probe_keys": ["l_orderkey"],
                 "join_type": "inner",
                 "output": [{"side": "probe", "col": "l_orderkey"},
                            {"side": "build", "col": "o_orderdate"},
                            {"side": "build", "col": "o_shippriority"},
                            {"side": "probe", "col": "l_extendedprice"},
                            {"side": "probe", "col": "l_discount"}]}
        partial = {"op": "hash_aggregate", "mode": "partial",
                   "group_by": ["l_orderkey", "o_orderdate",
                                "o_shippriority"],
                   "aggs": [{"fn": "sum", "as": "revenue",
                             "expr": {"mul": [
                                 {"col": "l_extendedprice"},
                                 {"sub": [{"lit": 100},
                                          {"col": "l_discount"}]}]}}],
                   "input": join2}
        rp = stage.execute(doc({
            "op": "sort_shuffle_write", "k": K,
            "keys": [{"col": "l_orderkey"}], "input": partial},
            3, task_id=p, schema=s3_schema))
        s3_paths.append(rp["partitions"][0]["path"])

    # stage 4: final aggregate over every stage-3 output, top-10
    final = {"op": "hash_aggregate", "mode": "final",
             "group_by": ["l_orderkey", "o_orderdate", "o_shippriority"],
             "aggs": [{"fn": "sum", "as": "revenue"}],
             "input": {"op": "scan",
                       "schema": stage.schema_json(s3_schema),
                       "source": {"kind": "shuffle", "locations": [
                           {"data": pth, "index": pth + ".index",
                            "partitions": list(range(K))}
                           for pth in s3_paths]}}}
    r4 = stage.execute(doc({
        "op": "collect", "limit": 10, "input": {
            "op": "sort", "keys": [{"col": "revenue", "desc": True},
                                   {"col": "o_orderdate", "desc": False}],
            "limit": 10, "input": final}}, 4))

    # reference: the single-stage q3 plan over the same registered tables
    single = stage.execute(_doc({"op": "collect", "limit": 10, "input": {
        "op": "sort", "keys": [{"col": "revenue", "desc": True},
                               {"col": "o_orderdate", "desc": False}],
        "limit": 10, "input": {
            "op": "hash_aggregate", "mode": "single",
            "group_by": ["l_orderkey", "o_orderdate", "o_shippriority"],
            "aggs": [{"fn": "sum", "as": "revenue",
                      "expr": {"mul": [{"col": "l_extendedprice"},
                                       {"sub": [{"lit": 100},
                                                {"col": "l_discount"}]}]}}],
            "input": {"op": "hash_join",
                      "build": {"op": "hash_join",
                                "build": {"op": "filter",
                                          "predicates": [
                                              {"col": "c_mktsegment",
                                               "cmp": "eq", "lo": 1}],
                                          "input": scan_of(cust, "sgc")},
                                "probe": {"op": "filter",
                                          "predicates": [
                                              {"col": "o_orderdate",
                                               "cmp": "lt", "hi": cutoff}],
                                          "input": scan_of(orders, "sgo")},
                                "build_keys": ["c_custkey"],
                                "probe_keys": ["o_custkey"],
                                "join_type": "inner",
                                "output": [
                                    {"side": "probe", "col": "o_orderkey"},
                                    {"side": "probe", "col": "o_orderdate"},
                                    {"side": "probe",
                                     "col": "o_shippriority"}]},
                      "probe": {"op": "filter",
                                "predicates": [{"col": "l_shipdate",
                                                "cmp": "gt", "lo": cutoff}],
                                "input": scan_of(li, "sgl")},
                      "build_keys": ["o_orderkey"],
                      "probe_keys": ["l_orderkey"],
                      "join_type": "inner",
                      "output": [{"side": "probe", "col": "l_orderkey"},
                                 {"side": "build", "col": "o_orderdate"},
                                 {"side": "build",
                                  "col": "o_shippriority"},
                                 {"side": "probe",
                                  "col": "l_extendedprice"},
                                 {"side": "probe",
                                  "col": "l_discount"}]}}}}))
    assert r4["rows"] == single["rows"]
    assert len(r4["rows"]) == 10


def test_parquet_scan_in_stage(ctx, tmp_path):
    """scan{kind:"parquet"}: the C++ page walk + device decode inside
    bg_execute_stage — q6 over a pyarrow-written snappy lineitem file
    (dictionary pages for low-cardinality columns, PLAIN fallback for
    price, FLBA decimals, OPTIONAL columns), vs the oracle exactly.
    Also: a whole-file collect must equal pyarrow's own reader."""
    import pyarrow.parquet as pq

    import oracle
    from datafusion_ballista_amd import tpch_synth
    n = 150_000
    li = tpch_synth.lineitem_numpy(n, seed=29)

    def dec_col(vals):
        return pa.array([decimal.Decimal(int(v)) / 100 for v in vals],
                        type=pa.decimal128(15, 2))

    t = pa.table({
        "l_shipdate": pa.array(li["l_shipdate"], type=pa.date32()),
        "l_quantity": dec_col(li["l_quantity"]),
        "l_extendedprice": dec_col(li["l_extendedprice"]),
        "l_discount": dec_col(li["l_discount"]),
    })
    path = str(tmp_path / "li_stage.parquet")
    pq.write_table(t, path, compression="snappy",
                   data_page_size=32 * 1024, write_statistics=False)

    scan_node = {"op": "scan", "schema": stage.schema_json(t.schema),
                 "source": stage.parquet_source(path)}
    plan = {"op": "collect", "input": {
        "op": "hash_aggregate", "mode": "single", "group_by": [],
        "aggs": [{"fn": "sum", "as": "revenue",
                  "expr": {"mul": [{"col": "l_extendedprice"},
                                   {"col": "l_discount"}]}},
                 {"fn": "count", "as": "cnt"}],
        "input": {"op": "filter", "predicates": [
            {"col": "l_shipdate", "cmp": "ge_lt",
             "lo": tpch_synth.Q6_DATE_LO, "hi": tpch_synth.Q6_DATE_HI},
            {"col": "l_discount", "cmp": "between",
             "lo": tpch_synth.Q6_DISC_LO, "hi": tpch_synth.Q6_DISC_HI},
            {"col": "l_quantity", "cmp": "lt",
             "hi": tpch_synth.Q6_QTY_LT}],
            "input": scan_node}}}
    res = stage.execute(_doc(plan))
    d16 = tpch_synth.dec128_pairs_np(li["l_discount"]).view(np.uint8).reshape(-1)
    q16 = tpch_synth.dec128_pairs_np(li["l_quantity"]).view(np.uint8).reshape(-1)
    p16 = tpch_synth.dec128_pairs_np(li["l_extendedprice"]).view(np.uint8).reshape(-1)
    want_cnt, want_sum = oracle.q6(
        li["l_shipdate"], d16, q16, p16, tpch_synth.Q6_DATE_LO,
        tpch_synth.Q6_DATE_HI, tpch_synth.Q6_DISC_LO, tpch_synth.Q6_DISC_HI,
        tpch_synth.Q6_QTY_LT)
    assert res["rows"][0][1] == want_cnt
    assert int(res["rows"][0][0]) == want_sum

    # whole-file scan collect vs pyarrow reader (first/last 100 rows)
    res2 = stage.execute(_doc({"op": "collect", "limit": 100,
                               "input": scan_node}))
    ref = pq.read_table(path)
    for i, row in enumerate(res2["rows"]):
        assert row[0] == ref["l_shipdate"][i].as_py().toordinal() - 719163
        assert int(row[2]) == int(ref["l_extendedprice"][i].as_py()
                                  .scaleb(2))


def test_like_predicates(ctx, reg):
    """LIKE / NOT LIKE filter predicates (the q9/q13/q14/q16 pattern
    shapes): prefix, suffix, contains, multi-fragment, and negation —
    each checked against a Python regex restatement, incl. NULLs
    (NULL LIKE ... => excluded; NULL NOT LIKE ... => excluded too)."""
    import re
    rng = np.random.default_rng(61)
    n = 40_000
    words = ["PROMO BRUSHED BRASS", "STANDARD green shiny", "promo brass",
             "special packages requests", "special requests",
             "ECONOMY green BRASS", "MEDIUM POLISHED TIN", "",
             "the specials make requests later", "PROMOTION green"]
    strs = [words[i % len(words)] + (f" #{i%97}" if i % 3 else "")
            for i in range(n)]
    mask = rng.random(n) < 0.07
    t = pa.table({"s": pa.array([None if m else v
                                 for v, m in zip(strs, mask)]),
                  "pid": pa.array(np.arange(n, dtype=np.int64))})
    reg("lk", t)

    cases = [("PROMO%", False, r"\APROMO.*\Z"),
             ("%BRASS", False, r"\A.*BRASS\Z"),
             ("%green%", False, r"\A.*green.*\Z"),
             ("%special%requests%", False, r"\A.*special.*requests.*\Z"),
             ("%green%", True, None)]
    for pat, neg, rx in cases:
        pred = {"col": "s", "like": pat}
        if neg:
            pred["negate"] = True
        res = stage.execute(_doc({"op": "collect", "input": {
            "op": "filter", "predicates": [pred],
            "input": scan_of(t, "lk")}}))
        if rx is None:
            rx_pos = re.compile(r"\A.*green.*\Z", re.S)
            want = [i for i, (v, m) in enumerate(zip(strs, mask))
                    if not m and not rx_pos.match(v)]
        else:
            creg = re.compile(rx, re.S)
            want = [i for i, (v, m) in enumerate(zip(strs, mask))
                    if not m and creg.match(v)]
        got = sorted(r[1] for r in res["rows"])
        assert got == want, (pat, neg, len(got), len(want))


def test_case_expression_q14_shape(ctx, reg):
    """q14's CASE WHEN p_type LIKE 'PROMO%' THEN revenue ELSE 0 shape:
    conditional projection feeding SUMs, exact vs a python restatement."""
    rng = np.random.default_rng(67)
    n = 50_000
    types = ["PROMO BRUSHED", "STANDARD TIN", "PROMO POLISHED", "ECONOMY",
             "MEDIUM PLATED"]
    tcol = [types[i % len(types)] for i in range(n)]
    price = rng.integers(90000, 10495100, size=n)
    disc = rng.integers(0, 11, size=n)
    t = pa.table({
        "p_type": pa.array(tcol),
        "l_extendedprice": pa.array(
            [decimal.Decimal(int(v)) / 100 for v in price],
            type=pa.decimal128(15, 2)),
        "l_discount": pa.array(
            [decimal.Decimal(int(v)) / 100 for v in disc],
            type=pa.decimal128(15, 2)),
    })
    reg("q14", t)
    rev = {"mul": [{"col": "l_extendedprice"},
                   {"sub": [{"lit": 100}, {"col": "l_discount"}]}]}
    plan = {"op": "collect", "input": {
        "op": "hash_aggregate", "mode": "single", "group_by": [],
        "aggs": [
            {"fn": "sum", "as": "promo",
             "expr": {"case": {"when": [{"col": "p_type",
                                         "like": "PROMO%"}],
                               "then": rev, "else": {"lit": 0}}}},
            {"fn": "sum", "as": "total", "expr": rev}],
        "input": {"op": "project", "exprs": [
            {"as": "p_type", "expr": {"col": "p_type"}},
            {"as": "l_extendedprice", "expr": {"col": "l_extendedprice"}},
            {"as": "l_discount", "expr": {"col": "l_discount"}}],
            "input": scan_of(t, "q14")}}}
    res = stage.execute(_doc(plan))
    want_promo = sum(int(p) * (100 - int(d))
                     for ty, p, d in zip(tcol, price, disc)
                     if ty.startswith("PROMO"))
    want_total = sum(int(p) * (100 - int(d))
                     for p, d in zip(price, disc))
    assert int(res["rows"][0][0]) == want_promo
    assert int(res["rows"][0][1]) == want_total


def test_in_list_and_or_groups(ctx, reg):
    """IN-list predicates (q12 l_shipmode IN (...)) and OR-of-AND groups
    (q19's (p=a AND qty in r1) OR (p=b AND qty in r2) shape), exact vs
    numpy."""
    rng = np.random.default_rng(71)
    n = 30_000
    mode = rng.integers(0, 7, size=n).astype(np.uint8)
    qty = rng.integers(1, 51, size=n, dtype=np.int64)
    d = rng.integers(8000, 11000, size=n, dtype=np.int32)
    t = pa.table({"mode": pa.array(mode), "qty": pa.array(qty),
                  "d": pa.array(d, type=pa.date32()),
                  "pid": pa.array(np.arange(n, dtype=np.int64))})
    reg("inor", t)

    # IN over dict8
    res = stage.execute(_doc({"op": "collect", "input": {
        "op": "filter", "predicates": [{"col": "mode", "in": [1, 4, 6]}],
        "input": scan_of(t, "inor")}}))
    want = sorted(np.nonzero(np.isin(mode, [1, 4, 6]))[0].tolist())
    assert sorted(r[3] for r in res["rows"]) == want

    # OR of AND-groups (q19 shape)
    res = stage.execute(_doc({"op": "collect", "input": {
        "op": "filter", "any": [
            [{"col": "mode", "in": [1]},
             {"col": "qty", "cmp": "between", "lo": 1, "hi": 11}],
            [{"col": "mode", "in": [2]},
             {"col": "qty", "cmp": "between", "lo": 10, "hi": 20}],
            [{"col": "mode", "in": [3]},
             {"col": "qty", "cmp": "between", "lo": 20, "hi": 30}]],
        "input": scan_of(t, "inor")}}))
    want_mask = (((mode == 1) & (qty >= 1) & (qty <= 11)) |
                 ((mode == 2) & (qty >= 10) & (qty <= 20)) |
                 ((mode == 3) & (qty >= 20) & (qty <= 30)))
    assert sorted(r[3] for r in res["rows"]) == \
        sorted(np.nonzero(want_mask)[0].tolist())

    # OR groups fused into an aggregate (mask path)
    res = stage.execute(_doc({"op": "collect", "input": {
        "op": "hash_aggregate", "mode": "single", "group_by": [],
        "aggs": [{"fn": "count", "as": "c"}],
        "input": {"op": "filter", "any": [
            [{"col": "d", "cmp": "lt", "hi": 9000}],
            [{"col": "qty", "cmp": "gt", "lo": 45}]],
            "input": scan_of(t, "inor")}}}))
    want_c = int(((d < 9000) | (qty > 45)).sum())
    assert res["rows"][0][0] == want_c


def test_sort_float64_keys(ctx, reg):
    """ORDER BY over Float64 (IEEE total order incl. negatives/infs), asc
    and desc, with NULLs defaulting to the SQL convention."""
    rng = np.random.default_rng(73)
    n = 20_000
    x = rng.standard_normal(n) * 1e6
    x[0] = float("inf")
    x[1] = float("-inf")
    mask = rng.random(n) < 0.05
    t = pa.table({"x": pa.array(x, mask=mask),
                  "pid": pa.array(np.arange(n, dtype=np.int64))})
    reg("f64s", t)
    for desc in (False, True):
        # pyarrow sort_by places nulls at_end for BOTH directions; pin the
        # plan's nulls_first accordingly (the kernel's default is the SQL
        # convention: DESC -> NULLS FIRST)
        res = stage.execute(_doc({"op": "collect", "input": {
            "op": "sort", "keys": [{"col": "x", "desc": desc,
                                    "nulls_first": False}],
            "limit": 200, "input": scan_of(t, "f64s")}}))
        order = "descending" if desc else "ascending"
        want = t.sort_by([("x", order)]).slice(0, 200)
        got = [r[0] for r in res["rows"]]
        assert got == want["x"].to_pylist()
    # nulls: ASC -> last; with limit < non-null count none appear
    assert all(g is not None for g in got)


def test_stage_errors_fail_loudly(ctx):
    with pytest.raises(RuntimeError, match="unregistered device table"):
        stage.execute(_doc({"op": "collect", "input": {
            "op": "scan", "schema": [{"name": "x", "dtype": "int64"}],
            "source": {"kind": "device", "table": "missing-table"}}}))
