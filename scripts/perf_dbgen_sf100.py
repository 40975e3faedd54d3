#!/usr/bin/env python3
"""q6 / q1 / q3 at SF100 on dbgen-shaped data: the
TPC-H spec's correlated distributions (per-order 1..7 lines, shipdate =
orderdate + 1..121, partkey-derived prices — tpch_dbgen.py) instead of
round 1's independent uniforms, all three queries driven as plan documents
through the C++ stage interpreter, with independent device-side
cross-checks (torch recomputation) of the aggregates.

Row counts pinned to fixtures.rs: lineitem 600,037,902 / orders
150,000,000 / customer 15,000,000."""
import ctypes
import json
import os
import sys
import time

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

import numpy as np  # noqa: E402
import torch  # noqa: E402

from datafusion_ballista_amd import gpu, stage, tpch_dbgen  # noqa: E402

SF = 100
NLI = 600_037_902
CUTOFF = tpch_dbgen.DATE_1995_03_15


def register(name, cols):
    L = gpu.load_library()
    arr = (gpu.BgColumn * len(cols))()
    names = (ctypes.c_char_p * len(cols))()
    n = cols[0][2].shape[0]
    for i, (cn, dt, t, p, s) in enumerate(cols):
        arr[i] = gpu.BgColumn(dt, p, s, 0, ctypes.c_void_p(t.data_ptr()),
                              None, None, n)
        names[i] = cn.encode()
    gpu._check(L.bg_stage_register_table(name.encode(), arr, names,
                                         len(cols), ctypes.c_int64(n)),
               "register")


def main():
    assert torch.cuda.is_available()
    dev = torch.device("cuda:0")
    gpu.GpuStageContext(0)
    out = {}

    print("generating dbgen-shaped SF100 in HBM...", flush=True)
    t0 = time.perf_counter()
    orders, li = tpch_dbgen.orders_lineitem(SF, dev, lines_cap=NLI)
    cust = tpch_dbgen.customer(SF, dev)
    torch.cuda.synchronize()
    n = li["l_orderkey"].shape[0]
    assert n == NLI, n
    print(f"  generated in {time.perf_counter()-t0:.1f}s "
          f"(lineitem {n} rows)", flush=True)

    register("lineitem", [
        ("l_orderkey", gpu.BG_DT_INT64, li["l_orderkey"], 0, 0),
        ("l_shipdate", gpu.BG_DT_DATE32, li["l_shipdate"], 0, 0),
        ("l_quantity", gpu.BG_DT_DECIMAL128, li["l_quantity"], 15, 2),
        ("l_extendedprice", gpu.BG_DT_DECIMAL128, li["l_extendedprice"],
         15, 2),
        ("l_discount", gpu.BG_DT_DECIMAL128, li["l_discount"], 15, 2),
        ("l_tax", gpu.BG_DT_DECIMAL128, li["l_tax"], 15, 2),
        ("l_returnflag", gpu.BG_DT_DICT8, li["l_returnflag"], 0, 0),
        ("l_linestatus", gpu.BG_DT_DICT8, li["l_linestatus"], 0, 0),
    ])
    register("orders", [
        ("o_orderkey", gpu.BG_DT_INT64, orders["o_orderkey"], 0, 0),
        ("o_custkey", gpu.BG_DT_INT64, orders["o_custkey"], 0, 0),
        ("o_orderdate", gpu.BG_DT_DATE32, orders["o_orderdate"], 0, 0),
        ("o_shippriority", gpu.BG_DT_INT32, orders["o_shippriority"], 0, 0),
    ])
    register("customer", [
        ("c_custkey", gpu.BG_DT_INT64, cust["c_custkey"], 0, 0),
        ("c_mktsegment", gpu.BG_DT_DICT8, cust["c_mktsegment"], 0, 0),
    ])

    li_schema = [
        {"name": "l_orderkey", "dtype": "int64"},
        {"name": "l_shipdate", "dtype": "date32"},
        {"name": "l_quantity", "dtype": "decimal128", "precision": 15,
         "scale": 2},
        {"name": "l_extendedprice", "dtype": "decimal128", "precision": 15,
         "scale": 2},
        {"name": "l_discount", "dtype": "decimal128", "precision": 15,
         "scale": 2},
        {"name": "l_tax", "dtype": "decimal128", "precision": 15,
         "scale": 2},
        {"name": "l_returnflag", "dtype": "dict8"},
        {"name": "l_linestatus", "dtype": "dict8"},
    ]

    def li_scan(proj):
        return {"op": "scan", "schema": li_schema,
                "source": {"kind": "device", "table": "lineitem"},
                "projection": proj}

    def run_doc(plan, stage_id):
        doc = {"job_id": "dbgen", "stage_id": stage_id, "task_id": 0,
               "work_dir": "/tmp/x", "plan": plan}
        t0 = time.perf_counter()
        r = stage.execute(doc)
        cold = time.perf_counter() - t0
        t0 = time.perf_counter()
        r2 = stage.execute(doc)
        warm = time.perf_counter() - t0
        assert r["rows"] == r2["rows"], "nondeterministic"
        return r, cold, warm

    # ---------------- q6 ----------------
    q6 = {"op": "collect", "input": {
        "op": "hash_aggregate", "mode": "single", "group_by": [],
        "aggs": [{"fn": "sum", "as": "revenue",
                  "expr": {"mul": [{"col": "l_extendedprice"},
                                   {"col": "l_discount"}]}},
                 {"fn": "count", "as": "cnt"}],
        "input": {"op": "filter", "predicates": [
            {"col": "l_shipdate", "cmp": "ge_lt", "lo": 8766, "hi": 9131},
            {"col": "l_discount", "cmp": "between", "lo": 5, "hi": 7},
            {"col": "l_quantity", "cmp": "lt", "hi": 2400}],
            "input": li_scan(["l_shipdate", "l_discount", "l_quantity",
                              "l_extendedprice"])}}}
    r, cold, warm = run_doc(q6, 6)
    # independent device cross-check with torch
    sd = li["l_shipdate"]
    disc = li["l_discount"][:, 0]
    qty = li["l_quantity"][:, 0]
    price = li["l_extendedprice"][:, 0]
    mask = ((sd >= 8766) & (sd < 9131) & (disc >= 5) & (disc <= 7) &
            (qty < 2400))
    want_cnt = int(mask.sum().item())
    want_sum = int((price[mask] * disc[mask]).sum().item())
    assert r["rows"][0][1] == want_cnt
    assert int(r["rows"][0][0]) == want_sum
    out["q6"] = {"rows": n, "wall_s_warm": warm, "wall_s_cold": cold,
                 "selectivity": want_cnt / n,
                 "rows_per_s": n / warm}
    print("q6:", json.dumps(out["q6"]), flush=True)

    # ---------------- q1 ----------------
    q1 = {"op": "collect", "input": {
            "op": "hash_aggregate", "mode": "single",
            "group_by": ["l_returnflag", "l_linestatus"],
            "aggs": [
                {"fn": "sum", "as": "sum_qty", "expr": {"col": "l_quantity"}},
                {"fn": "sum", "as": "sum_base",
                 "expr": {"col": "l_extendedprice"}},
                {"fn": "sum", "as": "sum_disc",
                 "expr": {"mul": [{"col": "l_extendedprice"},
                                  {"sub": [{"lit": 100},
                                           {"col": "l_discount"}]}]}},
                {"fn": "sum", "as": "sum_charge",
                 "expr": {"mul": [
                     {"mul": [{"col": "l_extendedprice"},
                              {"sub": [{"lit": 100},
                                       {"col": "l_discount"}]}]},
                     {"add": [{"col": "l_tax"}, {"lit": 100}]}]}},
                {"fn": "avg", "as": "avg_qty", "expr": {"col": "l_quantity"}},
                {"fn": "avg", "as": "avg_price",
                 "expr": {"col": "l_extendedprice"}},
                {"fn": "avg", "as": "avg_disc",
                 "expr": {"col": "l_discount"}},
                {"fn": "count", "as": "count_order"}],
            "input": {"op": "filter",
                      "predicates": [{"col": "l_shipdate", "cmp": "lt",
                                      "hi": 10471}],
                      "input": li_scan(["l_returnflag", "l_linestatus",
                                        "l_quantity", "l_extendedprice",
                                        "l_discount", "l_tax",
                                        "l_shipdate"])}}}
    r1_, cold, warm = run_doc(q1, 1)
    # torch cross-check: per-(rf,ls) count and sum_qty
    fmask = li["l_shipdate"] < 10471
    gid = (li["l_returnflag"].to(torch.int64) * 2 +
           li["l_linestatus"].to(torch.int64))[fmask]
    cnts = torch.bincount(gid, minlength=6)
    sq = torch.bincount(gid, weights=qty[fmask].to(torch.float64),
                        minlength=6)
    got = {(r[0], r[1]): r for r in r1_["rows"]}
    for rf in range(3):
        for ls in range(2):
            want_c = int(cnts[rf * 2 + ls].item())
            if want_c == 0:
                continue
            row = got[(rf, ls)]
            assert row[9] == want_c, (row, want_c)
            assert int(row[2]) == int(sq[rf * 2 + ls].item())
    out["q1"] = {"rows": n, "wall_s_warm": warm, "wall_s_cold": cold,
                 "groups": len(r1_["rows"]), "rows_per_s": n / warm}
    print("q1:", json.dumps(out["q1"]), flush=True)

    if "--stop-after-q1" in sys.argv:
        path = os.path.join(ROOT, "gpurun_out", "perf_dbgen_sf100.json")
        os.makedirs(os.path.dirname(path), exist_ok=True)
        with open(path, "w") as f:
            json.dump(out, f, indent=1)
        print("stopped after q1 (flag)", flush=True)
        return

    # ---------------- q3 (real group key, dbgen correlations) ----------
    join1 = {"op": "hash_join",
             "build": {"op": "filter",
                       "predicates": [{"col": "c_mktsegment", "cmp": "eq",
                                       "lo": 1}],
                       "input": {"op": "scan", "schema": [
                           {"name": "c_custkey", "dtype": "int64"},
                           {"name": "c_mktsegment", "dtype": "dict8"}],
                           "source": {"kind": "device",
                                      "table": "customer"}}},
             "probe": {"op": "filter",
                       "predicates": [{"col": "o_orderdate", "cmp": "lt",
                                       "hi": CUTOFF}],
                       "input": {"op": "scan", "schema": [
                           {"name": "o_orderkey", "dtype": "int64"},
                           {"name": "o_custkey", "dtype": "int64"},
                           {"name": "o_orderdate", "dtype": "date32"},
                           {"name": "o_shippriority", "dtype": "int32"}],
                           "source": {"kind": "device", "table": "orders"}}},
             "build_keys": ["c_custkey"], "probe_keys": ["o_custkey"],
             "join_type": "inner",
             "output": [{"side": "probe", "col": "o_orderkey"},
                        {"side": "probe", "col": "o_orderdate"},
                        {"side": "probe", "col": "o_shippriority"}]}
    join2 = {"op": "hash_join", "build": join1,
             "probe": {"op": "filter",
                       "predicates": [{"col": "l_shipdate", "cmp": "gt",
                                       "lo": CUTOFF}],
                       "input": li_scan(["l_orderkey", "l_shipdate",
                                         "l_extendedprice",
                                         "l_discount"])},
             "build_keys": ["o_orderkey"], "probe_keys": ["l_orderkey"],
             "join_type": "inner",
             "output": [{"side": "probe", "col": "l_orderkey"},
                        {"side": "build", "col": "o_orderdate"},
                        {"side": "build", "col": "o_shippriority"},
                        {"side": "probe", "col": "l_extendedprice"},
                        {"side": "probe", "col": "l_discount"}]}
    q3 = {"op": "collect", "limit": 10, "input": {
        "op": "sort", "keys": [{"col": "revenue", "desc": True},
                               {"col": "o_orderdate", "desc": False}],
        "limit": 10, "input": {
            "op": "hash_aggregate", "mode": "single",
            "group_by": ["l_orderkey", "o_orderdate", "o_shippriority"],
            "estimated_groups": 40_000_000,
            "aggs": [{"fn": "sum", "as": "revenue",
                      "expr": {"mul": [{"col": "l_extendedprice"},
                                       {"sub": [{"lit": 100},
                                                {"col": "l_discount"}]}]}}],
            "input": join2}}}
    r3, cold, warm = run_doc(q3, 3)
    # torch cross-check: revenue per group == per orderkey (okey unique)
    keep = torch.zeros(orders["o_orderkey"].shape[0] + 1, dtype=torch.bool,
                       device=dev)
    seg_ok = cust["c_mktsegment"] == 1
    cust_ok = torch.zeros(cust["c_custkey"].shape[0] + 1, dtype=torch.bool,
                          device=dev)
    cust_ok[cust["c_custkey"][seg_ok]] = True
    omask = (orders["o_orderdate"] < CUTOFF) & cust_ok[orders["o_custkey"]]
    keep[orders["o_orderkey"][omask]] = True
    lmask = (li["l_shipdate"] > CUTOFF) & keep[li["l_orderkey"]]
    rev = (price[lmask] * (100 - disc[lmask]))
    sums = torch.zeros(orders["o_orderkey"].shape[0] + 1, dtype=torch.int64,
                       device=dev)
    sums.scatter_add_(0, li["l_orderkey"][lmask], rev)
    topv, topi = torch.topk(sums, 10)
    want = sorted([int(v) for v in topv.tolist()], reverse=True)
    got = [int(r[3]) for r in r3["rows"]]
    assert got == want, (got, want)
    for row in r3["rows"]:
        okey = row[0]
        assert int(sums[okey].item()) == int(row[3])
        assert int(orders["o_orderdate"][okey - 1].item()) == row[1]
    out["q3"] = {"wall_s_warm": warm, "wall_s_cold": cold,
                 "top10": r3["rows"]}
    print("q3:", json.dumps(out["q3"]), flush=True)

    # ---------------- q14 (LIKE + CASE + part join at SF100) ----------
    # part: 20M rows, p_type from the spec's 150-combo vocabulary; the
    # first 25 combos are 'PROMO ...' so the cross-check knows membership
    # by type code.  The utf8 column is built ON DEVICE by gathering the
    # 150-string dictionary through the per-row codes.
    import ctypes as ct
    npart = SF * 200_000
    t1s = ["PROMO", "STANDARD", "SMALL", "MEDIUM", "LARGE", "ECONOMY"]
    t2s = ["ANODIZED", "BURNISHED", "PLATED", "POLISHED", "BRUSHED"]
    t3s = ["TIN", "NICKEL", "BRASS", "STEEL", "COPPER"]
    vocab = [f"{a} {b} {c}" for a in t1s for b in t2s for c in t3s]
    npromo = len(t2s) * len(t3s)  # 'PROMO *' codes are 0..24
    g2 = torch.Generator(device=dev)
    g2.manual_seed(77)
    pcodes = torch.randint(0, len(vocab), (npart,), generator=g2,
                           device=dev, dtype=torch.int32)
    dict_data = "".join(vocab).encode()
    dict_offs = np.zeros(len(vocab) + 1, dtype=np.int32)
    for i, w in enumerate(vocab):
        dict_offs[i + 1] = dict_offs[i] + len(w)
    ctx2 = gpu.GpuStageContext(0)
    ddata = ctx2.upload(np.frombuffer(dict_data, dtype=np.uint8))
    doffs = ctx2.upload(dict_offs)
    ptype_offs = ctx2.alloc(4 * (npart + 1))
    cap = npart * 32
    ptype_data = ctx2.alloc(cap)
    tot = ct.c_int64()
    gpu._check(gpu.load_library().bg_gather_varlen(
        ddata.ptr, doffs.ptr, ct.c_void_p(pcodes.data_ptr()),
        ct.c_int64(npart), ptype_offs.ptr, ptype_data.ptr,
        ct.c_int64(cap), ct.byref(tot)), "gather_varlen(p_type)")
    p_partkey = torch.arange(1, npart + 1, device=dev, dtype=torch.int64)
    L = gpu.load_library()
    arr = (gpu.BgColumn * 2)(
        gpu.BgColumn(gpu.BG_DT_INT64, 0, 0, 0,
                     ct.c_void_p(p_partkey.data_ptr()), None, None, npart),
        gpu.BgColumn(gpu.BG_DT_UTF8, 0, 0, 0, ptype_data.ptr, None,
                     ptype_offs.ptr, npart))
    names = (ct.c_char_p * 2)(b"p_partkey", b"p_type")
    gpu._check(L.bg_stage_register_table(b"part", arr, names, 2,
                                         ct.c_int64(npart)), "register")

    rev = {"mul": [{"col": "l_extendedprice"},
                   {"sub": [{"lit": 100}, {"col": "l_discount"}]}]}
    q14 = {"op": "collect", "input": {
        "op": "hash_aggregate", "mode": "single", "group_by": [],
        "aggs": [
            {"fn": "sum", "as": "promo",
             "expr": {"case": {"when": [{"col": "p_type",
                                         "like": "PROMO%"}],
                               "then": rev, "else": {"lit": 0}}}},
            {"fn": "sum", "as": "total", "expr": rev},
            {"fn": "count", "as": "cnt"}],
        "input": {"op": "hash_join",
                  "build": {"op": "scan", "schema": [
                      {"name": "p_partkey", "dtype": "int64"},
                      {"name": "p_type", "dtype": "utf8"}],
                      "source": {"kind": "device", "table": "part"}},
                  "probe": {"op": "filter",
                            "predicates": [{"col": "l_shipdate",
                                            "cmp": "ge_lt", "lo": 9374,
                                            "hi": 9404}],
                            "input": {"op": "scan", "schema": li_schema + [
                                {"name": "l_partkey", "dtype": "int64"}],
                                "source": {"kind": "device",
                                           "table": "lineitem2"},
                                "projection": ["l_partkey", "l_shipdate",
                                               "l_extendedprice",
                                               "l_discount"]}},
                  "build_keys": ["p_partkey"], "probe_keys": ["l_partkey"],
                  "join_type": "inner",
                  "output": [{"side": "build", "col": "p_type"},
                             {"side": "probe", "col": "l_extendedprice"},
                             {"side": "probe", "col": "l_discount"}]}}}
    register("lineitem2", [
        ("l_orderkey", gpu.BG_DT_INT64, li["l_orderkey"], 0, 0),
        ("l_shipdate", gpu.BG_DT_DATE32, li["l_shipdate"], 0, 0),
        ("l_quantity", gpu.BG_DT_DECIMAL128, li["l_quantity"], 15, 2),
        ("l_extendedprice", gpu.BG_DT_DECIMAL128, li["l_extendedprice"],
         15, 2),
        ("l_discount", gpu.BG_DT_DECIMAL128, li["l_discount"], 15, 2),
        ("l_tax", gpu.BG_DT_DECIMAL128, li["l_tax"], 15, 2),
        ("l_returnflag", gpu.BG_DT_DICT8, li["l_returnflag"], 0, 0),
        ("l_linestatus", gpu.BG_DT_DICT8, li["l_linestatus"], 0, 0),
        ("l_partkey", gpu.BG_DT_INT64, li["l_partkey"], 0, 0),
    ])
    r14, cold, warm = run_doc(q14, 14)
    # torch cross-check
    lm = (li["l_shipdate"] >= 9374) & (li["l_shipdate"] < 9404)
    pk = li["l_partkey"][lm]
    rev_t = (li["l_extendedprice"][:, 0][lm] *
             (100 - li["l_discount"][:, 0][lm]))
    promo_m = pcodes.to(torch.int64)[pk - 1] < npromo
    want_promo = int(rev_t[promo_m].sum().item())
    want_total = int(rev_t.sum().item())
    want_cnt = int(lm.sum().item())
    assert int(r14["rows"][0][0]) == want_promo
    assert int(r14["rows"][0][1]) == want_total
    assert r14["rows"][0][2] == want_cnt
    out["q14"] = {"wall_s_warm": warm, "wall_s_cold": cold,
                  "joined_rows": want_cnt,
                  "promo_frac": want_promo / max(want_total, 1)}
    print("q14:", json.dumps(out["q14"]), flush=True)

    os.makedirs(os.path.join(ROOT, "gpurun_out"), exist_ok=True)
    with open(os.path.join(ROOT, "gpurun_out", "perf_dbgen_sf100.json"),
              "w") as f:
        json.dump(out, f, indent=1)
    print(json.dumps(out), flush=True)


if __name__ == "__main__":
    main()
