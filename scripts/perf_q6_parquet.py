#!/usr/bin/env python3
"""Parquet-fed TPC-H q6 (SURVEY.md §8f row 1 feeding §8a row 1): lineitem
Parquet file resident in HBM -> device page decode (Snappy + dict/PLAIN +
FLBA decimals) -> fused q6 kernel, timed as one leg; exact cross-check
against the oracle over the host-decoded values.

The reference's q6 starts at DataSourceExec(Parquet); round 1's decode
peaked at 5.4 GB/s making a parquet-fed q6 decode-bound at ~0.1% of the
resident-column rate (VERDICT weak-2).  This measures the round-2 decode
path end to end.  File written with pyarrow defaults (dictionary encoding
where it fits — the shapes parquet-rs/arrow writers produce)."""
import json
import os
import sys
import time

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

import numpy as np  # noqa: E402
import pyarrow as pa  # noqa: E402
import pyarrow.parquet as pq  # noqa: E402
import decimal  # noqa: E402

import oracle  # noqa: E402  (test infrastructure: the cross-check only)
from datafusion_ballista_amd import gpu, tpch_synth  # noqa: E402
from datafusion_ballista_amd.parquet import GpuParquetColumnReader  # noqa: E402


def main():
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 60_000_000
    path = f"/tmp/lineitem_q6_{n}.parquet"
    li = tpch_synth.lineitem_numpy(n, seed=11)
    if not os.path.exists(path):
        print(f"writing {n}-row lineitem parquet...", flush=True)
        t0 = time.perf_counter()
        dec = pa.decimal128(15, 2)

        def dcol(v):
            return pa.array(v, type=dec)

        # decimal columns via int -> Decimal at scale 2
        qty = [decimal.Decimal(int(x)) / 100 for x in li["l_quantity"]]
        # price/discount likewise (price random-ish -> PLAIN fallback,
        # qty/discount/shipdate low-cardinality -> dictionary pages)
        price = [decimal.Decimal(int(x)) / 100
                 for x in li["l_extendedprice"]]
        disc = [decimal.Decimal(int(x)) / 100 for x in li["l_discount"]]
        table = pa.table({
            "l_shipdate": pa.array(li["l_shipdate"], type=pa.date32()),
            "l_quantity": dcol(qty),
            "l_extendedprice": dcol(price),
            "l_discount": dcol(disc),
        })
        pq.write_table(table, path, compression="snappy",
                       write_statistics=False)
        print(f"  wrote in {time.perf_counter()-t0:.1f}s "
              f"({os.path.getsize(path)/1e6:.0f} MB)", flush=True)

    ctx = gpu.GpuStageContext(0)
    t_up0 = time.perf_counter()
    rd = GpuParquetColumnReader(ctx, path)   # file bytes -> HBM (untimed leg)
    ctx.synchronize()
    t_upload = time.perf_counter() - t_up0

    def run():
        t0 = time.perf_counter()
        cols = {}
        decoded_bytes = 0
        percol = {}
        for ci, name in enumerate(["l_shipdate", "l_quantity",
                                   "l_extendedprice", "l_discount"]):
            tc = time.perf_counter()
            buf, nv, phys, valid = rd.read_column_all(ci)
            ctx.synchronize()
            percol[name] = round(time.perf_counter() - tc, 4)
            # pyarrow writes these as OPTIONAL; no actual nulls exist
            assert nv == n
            esz = 4 if name == "l_shipdate" else 16
            decoded_bytes += nv * esz
            cols[name] = buf
        ctx.synchronize()
        t_dec = time.perf_counter() - t0
        if os.environ.get("BG_PQ_DEBUG"):
            print("percol:", json.dumps(percol), flush=True)
        sd = ctx.column(gpu.BG_DT_DATE32, cols["l_shipdate"], n)
        cq = ctx.column(gpu.BG_DT_DECIMAL128, cols["l_quantity"], n)
        cp = ctx.column(gpu.BG_DT_DECIMAL128, cols["l_extendedprice"], n)
        cd = ctx.column(gpu.BG_DT_DECIMAL128, cols["l_discount"], n)
        cnt, total = ctx.q6_agg(sd, cd, cq, cp, tpch_synth.Q6_DATE_LO,
                                tpch_synth.Q6_DATE_HI, tpch_synth.Q6_DISC_LO,
                                tpch_synth.Q6_DISC_HI, tpch_synth.Q6_QTY_LT)
        ctx.synchronize()
        t_all = time.perf_counter() - t0
        return cnt, total, t_dec, t_all, decoded_bytes

    cnt0, tot0, *_ = run()  # warmup
    best = None
    for _ in range(5):
        cnt, total, t_dec, t_all, dbytes = run()
        assert (cnt, total) == (cnt0, tot0)
        if best is None or t_all < best[3]:
            best = (cnt, total, t_dec, t_all, dbytes)
    cnt, total, t_dec, t_all, dbytes = best

    # oracle cross-check (host, exact)
    d16 = tpch_synth.dec128_pairs_np(li["l_discount"]).view(np.uint8).reshape(-1)
    q16 = tpch_synth.dec128_pairs_np(li["l_quantity"]).view(np.uint8).reshape(-1)
    p16 = tpch_synth.dec128_pairs_np(li["l_extendedprice"]).view(np.uint8).reshape(-1)
    want_cnt, want_sum = oracle.q6(li["l_shipdate"], d16, q16, p16,
                                   tpch_synth.Q6_DATE_LO,
                                   tpch_synth.Q6_DATE_HI,
                                   tpch_synth.Q6_DISC_LO,
                                   tpch_synth.Q6_DISC_HI,
                                   tpch_synth.Q6_QTY_LT)
    assert cnt == want_cnt and total == want_sum, \
        (cnt, total, want_cnt, want_sum)
    # interpreter leg: the SAME q6 over scan{kind:"parquet"} — C++ page
    # walk + device decode + fused kernel in one bg_execute_stage call
    from datafusion_ballista_amd import stage as bgstage
    scan_node = {"op": "scan", "schema": [
        {"name": "l_shipdate", "dtype": "date32"},
        {"name": "l_quantity", "dtype": "decimal128", "precision": 15,
         "scale": 2},
        {"name": "l_extendedprice", "dtype": "decimal128", "precision": 15,
         "scale": 2},
        {"name": "l_discount", "dtype": "decimal128", "precision": 15,
         "scale": 2}],
        "source": bgstage.parquet_source(path)}
    q6_doc = {"job_id": "pq", "stage_id": 6, "task_id": 0,
              "work_dir": "/tmp/x", "plan": {"op": "collect", "input": {
                  "op": "hash_aggregate", "mode": "single", "group_by": [],
                  "aggs": [{"fn": "sum", "as": "revenue",
                            "expr": {"mul": [{"col": "l_extendedprice"},
                                             {"col": "l_discount"}]}},
                           {"fn": "count", "as": "cnt"}],
                  "input": {"op": "filter", "predicates": [
                      {"col": "l_shipdate", "cmp": "ge_lt",
                       "lo": tpch_synth.Q6_DATE_LO,
                       "hi": tpch_synth.Q6_DATE_HI},
                      {"col": "l_discount", "cmp": "between",
                       "lo": tpch_synth.Q6_DISC_LO,
                       "hi": tpch_synth.Q6_DISC_HI},
                      {"col": "l_quantity", "cmp": "lt",
                       "hi": tpch_synth.Q6_QTY_LT}],
                      "input": scan_node}}}}
    ri = bgstage.execute(q6_doc)  # warmup (file-read + pool warm)
    assert ri["rows"][0][1] == want_cnt
    assert int(ri["rows"][0][0]) == want_sum
    t_i = 1e30
    for _ in range(5):
        t0 = time.perf_counter()
        ri = bgstage.execute(q6_doc)
        t_i = min(t_i, time.perf_counter() - t0)
    # NOTE: the interpreter leg re-reads the FILE from page cache and
    # re-uploads every call (stateless stages) — PCIe-inclusive

    rec = {"rows": n, "file_mb": os.path.getsize(path) / 1e6,
           "upload_s": t_upload, "decode_s": t_dec,
           "decode_gbps": dbytes / t_dec / 1e9,
           "q6_from_parquet_s": t_all,
           "rows_per_s": n / t_all, "count": cnt,
           "interp_q6_s_incl_file_read": t_i,
           "interp_rows_per_s": n / t_i,
           "crosscheck": "exact vs oracle"}
    print(json.dumps(rec), flush=True)
    os.makedirs(os.path.join(ROOT, "gpurun_out"), exist_ok=True)
    with open(os.path.join(ROOT, "gpurun_out", "perf_q6_parquet.json"),
              "w") as f:
        json.dump(rec, f, indent=1)


if __name__ == "__main__":
    main()
