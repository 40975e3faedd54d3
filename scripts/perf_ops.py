#!/usr/bin/env python3
"""Per-op perf sweep at TPC-H SF100-like scale (1 GPU).

Times each hot-path op with wall clocks around bg_synchronize (ms-scale
ops; kernel-level splits come from the rocprofv3 --stats run committed under
profiles/).  Reports GB/s against the ALGORITHMIC byte model of each op
(DESIGN.md §3).  Shapes:
  - hash repartition: q3 stage-2-like — 150M orders rows, key Int64,
    payload (Int64 key + Date32 + Int32 + Int64), k=16
  - hash join: customer⨝orders — build 15M, probe 150M, Int64 keys
  - hash group-by: 150M rows -> ~12M groups, SUM(Decimal128)
  - q1 fused: 600M lineitem rows
"""
import json
import os
import sys
import time

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

import numpy as np  # noqa: E402
import torch  # noqa: E402

from datafusion_ballista_amd import gpu, tpch_synth  # noqa: E402


def timeit(ctx, fn, iters=3):
    fn()  # warmup
    ctx.synchronize()
    best = 1e18
    for _ in range(iters):
        t0 = time.perf_counter()
        fn()
        ctx.synchronize()
        best = min(best, time.perf_counter() - t0)
    return best


def main():
    assert torch.cuda.is_available()
    dev = torch.device("cuda:0")
    ctx = gpu.GpuStageContext(0)
    g = torch.Generator(device=dev)
    g.manual_seed(7)
    out = []

    def report(name, sec, algo_bytes, extra=None):
        rec = {"op": name, "ms": sec * 1e3, "algo_bytes": algo_bytes,
               "gbps": algo_bytes / sec / 1e9}
        if extra:
            rec.update(extra)
        out.append(rec)
        print(json.dumps(rec), flush=True)

    def col_of(t, dtype):
        return gpu.BgColumn(dtype, 15, 2, 0, t.data_ptr(), None, None, t.shape[0])

    # ---- hash repartition (150M rows, k=16) ----
    n = 150_000_000
    k = 16
    keys = torch.randint(1, n, (n,), generator=g, device=dev, dtype=torch.int64)
    dates = torch.randint(8000, 11000, (n,), generator=g, device=dev,
                          dtype=torch.int32)
    prio = torch.randint(0, 5, (n,), generator=g, device=dev, dtype=torch.int32)
    cust = torch.randint(1, 15_000_000, (n,), generator=g, device=dev,
                         dtype=torch.int64)
    kc = col_of(keys, gpu.BG_DT_INT64)
    payload = [kc, col_of(dates, gpu.BG_DT_DATE32),
               col_of(prio, gpu.BG_DT_INT32), col_of(cust, gpu.BG_DT_INT64)]

    hbuf = ctx.alloc(8 * n)
    pbuf = ctx.alloc(4 * n)
    ibuf = ctx.alloc(4 * n)
    obuf = ctx.alloc(8 * (k + 1))
    import ctypes
    L = ctx.L
    sec = timeit(ctx, lambda: gpu._check(
        L.bg_hash_columns((gpu.BgColumn * 1)(kc), 1, ctypes.c_int64(n),
                          hbuf.ptr), "hash"))
    report("hash_columns_i64", sec, 8 * n + 8 * n)  # read key, write hash
    sec = timeit(ctx, lambda: gpu._check(
        L.bg_partition_ids(hbuf.ptr, ctypes.c_int64(n), k, pbuf.ptr), "pids"))
    report("partition_ids", sec, 8 * n + 4 * n)
    sec = timeit(ctx, lambda: gpu._check(
        L.bg_partition_indices(pbuf.ptr, ctypes.c_int64(n), k, ibuf.ptr,
                               obuf.ptr), "split"))
    report("partition_indices_stable_split_k16", sec, 4 * n * 2 + 4 * n)
    # gather all four payload columns through the permutation
    outs = [ctx.alloc(8 * n), ctx.alloc(4 * n), ctx.alloc(4 * n),
            ctx.alloc(8 * n)]
    def do_gather():
        for c, o in zip(payload, outs):
            esz = gpu._DT_SIZE[c.dtype]
            gpu._check(L.bg_gather(ctypes.c_void_p(c.d_data),
                                   ctypes.c_int64(esz), ibuf.ptr,
                                   ctypes.c_int64(n), o.ptr), "gather")
    gather_bytes = sum((4 + 2 * gpu._DT_SIZE[c.dtype]) * n for c in payload)
    sec = timeit(ctx, do_gather)
    report("gather_4cols_24B_row", sec, gather_bytes,
           {"note": "4B idx + read+write per col elem; random reads"})
    # scatter-based materialisation (sequential reads, k write streams)
    rankbuf = ctx.alloc(4 * n)
    gpu._check(L.bg_partition_indices_ex(pbuf.ptr, ctypes.c_int64(n), k,
                                         ibuf.ptr, obuf.ptr, rankbuf.ptr),
               "split_ex")
    def do_scatter():
        for c, o in zip(payload, outs):
            esz = gpu._DT_SIZE[c.dtype]
            gpu._check(L.bg_scatter_rows(ctypes.c_void_p(c.d_data),
                                         ctypes.c_int64(esz), rankbuf.ptr,
                                         ctypes.c_int64(n), o.ptr), "scat")
    sec = timeit(ctx, do_scatter)
    report("scatter_4cols_24B_row", sec, gather_bytes,
           {"note": "sequential reads via inverse perm, k=16 write streams"})
    whole_bytes = (8 + 8) * n + (8 + 4) * n + (4 * 3) * n + gather_bytes
    sec = timeit(ctx, lambda: (
        gpu._check(L.bg_hash_columns((gpu.BgColumn * 1)(kc), 1,
                                     ctypes.c_int64(n), hbuf.ptr), "h"),
        gpu._check(L.bg_partition_ids(hbuf.ptr, ctypes.c_int64(n), k,
                                      pbuf.ptr), "p"),
        gpu._check(L.bg_partition_indices_ex(pbuf.ptr, ctypes.c_int64(n), k,
                                             ibuf.ptr, obuf.ptr,
                                             rankbuf.ptr), "s"),
        do_scatter()))
    report("repartition_pipeline_total", sec, whole_bytes,
           {"rows_per_s": n / sec})

    # ---- fused one-pass repartition materialiser ----
    fidx = ctx.alloc(4 * n)
    foffs = ctx.alloc(8 * (k + 1))
    frank = ctx.alloc(4 * n)
    fouts = [ctx.alloc(8 * n), ctx.alloc(4 * n), ctx.alloc(4 * n),
             ctx.alloc(8 * n)]
    foptrs = (ctypes.c_void_p * 4)(*[b.ptr.value for b in fouts])
    karr1 = (gpu.BgColumn * 1)(kc)
    parr4 = (gpu.BgColumn * 4)(*payload)
    def do_fused():
        gpu._check(L.bg_hash_repartition_fused(
            karr1, 1, parr4, 4, ctypes.c_int64(n), k, fidx.ptr, foffs.ptr,
            frank.ptr, foptrs), "fused")
    sec = timeit(ctx, do_fused)
    report("repartition_fused_total", sec, whole_bytes,
           {"rows_per_s": n / sec, "kernel_ms": L.bg_last_kernel_ms()})

    # ---- hash join: build 15M, probe 150M ----
    nb = 15_000_000
    bkeys = torch.randperm(nb, device=dev, dtype=torch.int64) + 1
    bcol = col_of(bkeys, gpu.BG_DT_INT64)
    pcol = col_of(cust, gpu.BG_DT_INT64)  # cust in [1, 15M): ~every row matches
    join = gpu.GpuHashJoin(ctx, bcol, nb)
    ctx.synchronize()
    t0 = time.perf_counter()
    join2 = gpu.GpuHashJoin(ctx, bcol, nb)
    ctx.synchronize()
    build_sec = time.perf_counter() - t0
    report("hashjoin_build_15M", build_sec, nb * (8 + 8 + 4 + 4),
           {"note": "read keys, write key copy + head CAS + next"})
    pb, bb, m = join2.probe(pcol, n)
    ctx.synchronize()
    pb.free(); bb.free()  # return pair buffers to the pool (steady state)
    t0 = time.perf_counter()
    pb2, bb2, m2 = join2.probe(pcol, n)
    ctx.synchronize()
    probe_sec = time.perf_counter() - t0
    # probe: read probe key (8) x2 phases + ~1 random line (head 4 + next 4
    # + build key 8) x2 + write pairs 8
    probe_bytes = n * (16 + 2 * 16 + 8)
    report("hashjoin_probe_150M_vs_15M", probe_sec, probe_bytes,
           {"matches": m2, "rows_per_s": n / probe_sec})
    join.free(); join2.free()

    # ---- hash group-by: 150M rows -> 12M groups ----
    gk = torch.randint(0, 12_000_000, (n,), generator=g, device=dev,
                       dtype=torch.int64)
    dec = torch.zeros((n, 2), dtype=torch.int64, device=dev)
    dec[:, 0] = torch.randint(0, 10**9, (n,), generator=g, device=dev)
    gkc = col_of(gk, gpu.BG_DT_INT64)
    dcc = col_of(dec, gpu.BG_DT_DECIMAL128)
    first = ctx.alloc(4 * 13_000_000)
    acc = ctx.alloc(16 * 13_000_000)
    cnts = ctx.alloc(8 * 13_000_000)
    ng = ctypes.c_int64()
    aarr = (gpu.BgColumn * 1)(dcc)
    oarr = (ctypes.c_int32 * 1)(gpu.BG_AGG_OP_SUM_DEC128)
    def do_agg():
        gpu._check(L.bg_hashagg((gpu.BgColumn * 1)(gkc), 1, aarr, oarr, 1,
                                None, ctypes.c_int64(n),
                                ctypes.c_int64(12_500_000), first.ptr,
                                acc.ptr, cnts.ptr, ctypes.byref(ng)), "agg")
    sec = timeit(ctx, do_agg, iters=2)
    # per row: key 8 + dec 16 + ~1 random slot line (CAS 4 + key cmp 8) +
    # acc atomics 16+8
    agg_bytes = n * (8 + 16 + 12 + 24)
    report("hashagg_150M_rows_12M_groups", sec, agg_bytes,
           {"ngroups": ng.value, "rows_per_s": n / sec,
            "kernel_ms": L.bg_last_kernel_ms()})

    # ---- stable split at k=512 (the >64-partition multi-split path) ----
    ob512 = ctx.alloc(8 * 513)
    gpu._check(L.bg_partition_ids(hbuf.ptr, ctypes.c_int64(n), 512,
                                  pbuf.ptr), "pids512")
    sec = timeit(ctx, lambda: gpu._check(
        L.bg_partition_indices(pbuf.ptr, ctypes.c_int64(n), 512, ibuf.ptr,
                               ob512.ptr), "split512"))
    report("partition_indices_stable_split_k512", sec, 4 * n * 2 + 4 * n,
           {"rows_per_s": n / sec})
    gpu._check(L.bg_partition_ids(hbuf.ptr, ctypes.c_int64(n), k,
                                  pbuf.ptr), "pids-restore")

    # ---- sort: 150M i64 keys (8 radix passes) ----
    permbuf = ctx.alloc(4 * n)
    kcarr = (gpu.BgColumn * 1)(kc)
    darr = (ctypes.c_int32 * 1)(0)
    def do_sort():
        gpu._check(L.bg_sort_rows(kcarr, darr, 1, ctypes.c_int64(n),
                                  permbuf.ptr), "sort")
    sec = timeit(ctx, do_sort, iters=2)
    report("sort_rows_i64_150M", sec, n * 8 * 16,
           {"note": "8 stable radix passes; bytes = 16B/row/pass model",
            "rows_per_s": n / sec})

    # free big tensors before q1
    del keys, dates, prio, cust, bkeys, gk, dec
    torch.cuda.empty_cache()

    # ---- q1 fused (600M rows) ----
    n1 = 600_037_902
    cols = tpch_synth.lineitem_torch(n1, dev, seed=5)
    rf = col_of(cols["l_returnflag"], gpu.BG_DT_DICT8)
    ls = col_of(cols["l_linestatus"], gpu.BG_DT_DICT8)
    cq = col_of(cols["l_quantity"], gpu.BG_DT_DECIMAL128)
    cp = col_of(cols["l_extendedprice"], gpu.BG_DT_DECIMAL128)
    cd = col_of(cols["l_discount"], gpu.BG_DT_DECIMAL128)
    ct = col_of(cols["l_tax"], gpu.BG_DT_DECIMAL128)
    sd = col_of(cols["l_shipdate"], gpu.BG_DT_DATE32)
    counts = np.zeros(256, dtype=np.int64)
    sums = np.zeros(256 * 5 * 16, dtype=np.uint8)
    def do_q1():
        gpu._check(L.bg_q1_agg(
            ctypes.byref(rf), ctypes.byref(ls), ctypes.byref(cq),
            ctypes.byref(cp), ctypes.byref(cd), ctypes.byref(ct),
            ctypes.byref(sd), tpch_synth.Q1_DATE_LE,
            counts.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)),
            sums.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8))), "q1")
    sec = timeit(ctx, do_q1, iters=3)
    q1_bytes = n1 * 70  # 2xu8 + 4xDec128 + Date32
    report("q1_fused_600M", sec, q1_bytes,
           {"kernel_ms": L.bg_last_kernel_ms(), "rows_per_s": n1 / sec})

    with open(os.path.join(ROOT, "gpurun_out", "perf_ops.json"), "w") as f:
        json.dump(out, f, indent=1)


if __name__ == "__main__":
    main()
