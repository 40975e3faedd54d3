#!/usr/bin/env python3
"""TPC-H q3 at SF100 scale on 1 MI355X (BASELINE.json configs[2]):
customer(filter) ⨝ orders(filter) ⨝ lineitem(filter) → revenue group-by →
top-10 — the full 3-way-join query chain on device, timed end to end, with
a full-scale numpy cross-check of the global aggregates.

Row counts per tpch_plan_stability/fixtures.rs: customer 15M, orders 150M,
lineitem 600,037,902.  Predicates use stand-ins with the reference
selectivities (mktsegment 1/5, o_orderdate<1995-03-15 ≈46%,
l_shipdate>1995-03-15 ≈54%)."""
import ctypes
import json
import os
import sys
import time

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

import numpy as np  # noqa: E402
import torch  # noqa: E402

from datafusion_ballista_amd import gpu  # noqa: E402

NCUST = 15_000_000
NORD = 150_000_000
NLI = 600_037_902
CUTOFF = 9204  # 1995-03-15


def col_of(t, dtype):
    return gpu.BgColumn(dtype, 15, 2, 0, t.data_ptr(), None, None, t.shape[0])


def main():
    assert torch.cuda.is_available()
    dev = torch.device("cuda:0")
    ctx = gpu.GpuStageContext(0)
    g = torch.Generator(device=dev)
    g.manual_seed(42)

    print("generating SF100-scale tables in HBM...", flush=True)
    c_custkey = torch.randperm(NCUST, device=dev, dtype=torch.int64) + 1
    o_custkey = torch.randint(1, NCUST + 1, (NORD,), generator=g, device=dev,
                              dtype=torch.int64)
    o_orderdate = torch.randint(8036, 10561, (NORD,), generator=g, device=dev,
                                dtype=torch.int32)
    l_orderkey = torch.randint(1, NORD + 1, (NLI,), generator=g, device=dev,
                               dtype=torch.int64)
    l_shipdate = torch.randint(8036, 10561, (NLI,), generator=g, device=dev,
                               dtype=torch.int32)
    l_price = torch.zeros((NLI, 2), dtype=torch.int64, device=dev)
    l_price[:, 0] = torch.randint(90000, 10495100, (NLI,), generator=g,
                                  device=dev)
    l_disc = torch.zeros((NLI, 2), dtype=torch.int64, device=dev)
    l_disc[:, 0] = torch.randint(0, 11, (NLI,), generator=g, device=dev)
    torch.cuda.synchronize()

    def run_query():
        marks = []
        def mk(name):
            ctx.synchronize()
            marks.append((name, time.perf_counter()))
        t0 = time.perf_counter()
        mk("start")
        # stage A: customer filter -> build
        cc = col_of(c_custkey, gpu.BG_DT_INT64)
        cmask = ctx.eval_predicates([cc], [(0, gpu.BG_PRED_LT, 0,
                                            NCUST // 5 + 1)], NCUST)
        cidx, nc = ctx.mask_to_indices(cmask, NCUST)
        ckeys = ctx.gather(_wrap(ctx, c_custkey), 8, cidx, nc)
        cjoin = gpu.GpuHashJoin(ctx, ctx.column(gpu.BG_DT_INT64, ckeys, nc),
                                nc)
        mk("stageA_cust_build")
        # stage B: orders filter -> probe customers -> build orderkeys
        od = col_of(o_orderdate, gpu.BG_DT_DATE32)
        omask = ctx.eval_predicates([od], [(0, gpu.BG_PRED_LT, 0, CUTOFF)],
                                    NORD)
        oidx, no = ctx.mask_to_indices(omask, NORD)
        ock = ctx.gather(_wrap(ctx, o_custkey), 8, oidx, no)
        ook_all = ctx.alloc(8 * no)  # orderkey = row index + 1 (generated)
        # o_orderkey is 1..NORD dense: orderkey of filtered row i = oidx[i]+1
        # compute via project: gather row indices then +1 on host model —
        # build a device orderkey column from oidx (u32 -> i64 + 1)
        oidx_i64 = _u32_to_i64_plus1(ctx, oidx, no)
        ppos, bpos, nmatch = cjoin.probe(
            ctx.column(gpu.BG_DT_INT64, ock, no), no)
        ok_matched = ctx.gather(oidx_i64, 8, ppos, nmatch)
        ojoin = gpu.GpuHashJoin(
            ctx, ctx.column(gpu.BG_DT_INT64, ok_matched, nmatch), nmatch)
        mk("stageB_orders")
        # stage C: lineitem filter -> probe orders -> revenue -> group-by
        ld = col_of(l_shipdate, gpu.BG_DT_DATE32)
        lmask = ctx.eval_predicates([ld], [(0, gpu.BG_PRED_GT, CUTOFF, 0)],
                                    NLI)
        mk("C.filter")
        lidx, nl = ctx.mask_to_indices(lmask, NLI)
        mk("C.indices")
        lk = ctx.gather(_wrap(ctx, l_orderkey), 8, lidx, nl)
        lp = ctx.gather(_wrap(ctx, l_price), 16, lidx, nl)
        ldc = ctx.gather(_wrap(ctx, l_disc), 16, lidx, nl)
        mk("C.gather3")
        lppos, lbpos, nlm = ojoin.probe(
            ctx.column(gpu.BG_DT_INT64, lk, nl), nl)
        mk("C.probe")
        jk = ctx.gather(lk, 8, lppos, nlm)
        jp = ctx.gather(lp, 16, lppos, nlm)
        jd = ctx.gather(ldc, 16, lppos, nlm)
        mk("C.gatherj")
        one_minus = ctx.project_dec128(
            gpu.BG_PROJ_RSUB_LIT,
            ctx.column(gpu.BG_DT_DECIMAL128, jd, nlm), None, 100, nlm)
        rev = ctx.project_dec128(
            gpu.BG_PROJ_MUL, ctx.column(gpu.BG_DT_DECIMAL128, jp, nlm),
            ctx.column(gpu.BG_DT_DECIMAL128, one_minus, nlm), 0, nlm)
        mk("C.proj")
        mk("stageC_lineitem_join_proj")
        first, acc, counts = ctx.hashagg(
            [ctx.column(gpu.BG_DT_INT64, jk, nlm)],
            [ctx.column(gpu.BG_DT_DECIMAL128, rev, nlm)],
            [gpu.BG_AGG_OP_SUM_DEC128], nlm, max_groups=32_000_000)
        mk("hashagg_and_downloads")
        ngroups = len(first)
        # top-10 by revenue desc: sort the per-group sums (dec128 keys)
        gsum = ctx.upload(acc.reshape(-1))  # [g,1,16] bytes back to device
        perm = ctx.sort_rows(
            [ctx.column(gpu.BG_DT_DECIMAL128, gsum, ngroups)], [True],
            ngroups)
        top = perm.download(np.uint32, min(10, ngroups))
        mk("upload_sort_topk")
        ctx.synchronize()
        wall = time.perf_counter() - t0
        prev = t0
        for nm, tt in marks[1:]:
            print(f"  [{nm}] {(tt - prev) * 1e3:.1f} ms", flush=True)
            prev = tt
        lo = acc[:, 0, :8].copy().view(np.uint64).reshape(-1)
        hi = acc[:, 0, 8:].copy().view(np.int64).reshape(-1)
        total_rev = (int(hi.astype(object).sum()) << 64) + \
            int(lo.astype(object).sum())
        # top-1 group identity: fetch the group KEY (l_orderkey) via the
        # representative row — the claiming row itself is scheduling-
        # dependent (any row of the group is valid), the key is not.
        top1_key = -1
        if ngroups:
            ridx = ctx.upload(np.asarray([first[top[0]]], dtype=np.uint32))
            kbuf = ctx.gather(jk, 8, ridx, 1)
            top1_key = int(kbuf.download(np.int64, 1)[0])
        return wall, {"qual_cust": nc, "qual_orders": no,
                      "matched_orders": nmatch, "qual_lineitem": nl,
                      "joined_rows": nlm, "groups": ngroups,
                      "total_revenue_scale4": total_rev,
                      "top1_orderkey": top1_key}

    def _wrap(ctx_, t):
        b = gpu.DeviceBuffer.__new__(gpu.DeviceBuffer)
        b._ctx = ctx_
        b.ptr = ctypes.c_void_p(t.data_ptr())
        b.nbytes = t.numel() * t.element_size()
        return b

    def _u32_to_i64_plus1(ctx_, idxbuf, m):
        # tiny helper via torch: wrap the u32 buffer, convert on device
        t = torch.empty(m, dtype=torch.int64, device=dev)
        src = torch.empty(m, dtype=torch.int32, device=dev)
        gpu._check(ctx_.L.bg_memcpy_dtod(
            ctypes.c_void_p(src.data_ptr()), idxbuf.ptr,
            ctypes.c_uint64(4 * m)), "dtod")
        t.copy_(src.view(torch.int32).to(torch.int64))
        t.add_(1)
        b = _wrap(ctx_, t)
        b._keep = t  # keep tensor alive
        return b

    globals()["_wrap"] = _wrap
    globals()["_u32_to_i64_plus1"] = _u32_to_i64_plus1

    # warmup + timed
    w0, stats0 = run_query()
    w1, stats1 = run_query()
    if stats0 != stats1:
        print("run0:", stats0)
        print("run1:", stats1)
        raise AssertionError("nondeterministic aggregates")

    # full-scale cross-check of the global invariants on host (numpy)
    print("host cross-check...", flush=True)
    ck = c_custkey.cpu().numpy()
    ok_mask = (o_orderdate.cpu().numpy() < CUTOFF)
    ock_h = o_custkey.cpu().numpy()
    qual_cust = ck[ck <= NCUST // 5]
    cust_set = np.zeros(NCUST + 1, dtype=bool)
    cust_set[qual_cust] = True
    qual_orders_mask = ok_mask & cust_set[ock_h]
    qual_orderkeys = np.nonzero(qual_orders_mask)[0] + 1
    order_set = np.zeros(NORD + 1, dtype=bool)
    order_set[qual_orderkeys] = True
    lk_h = l_orderkey.cpu().numpy()
    ls_h = l_shipdate.cpu().numpy()
    li_mask = (ls_h > CUTOFF) & order_set[lk_h]
    joined = int(li_mask.sum())
    price_h = l_price[:, 0].cpu().numpy()
    disc_h = l_disc[:, 0].cpu().numpy()
    rev = price_h[li_mask] * (100 - disc_h[li_mask])  # fits int64 per row
    rev_total = ((int(np.sum(rev >> 32, dtype=np.int64)) << 32) +
                 int(np.sum(rev & 0xFFFFFFFF, dtype=np.int64)))
    assert stats1["joined_rows"] == joined, (stats1["joined_rows"], joined)
    assert stats1["total_revenue_scale4"] == rev_total
    print("cross-check OK", flush=True)

    rec = {"query": "q3_sf100_like", "wall_s_warm": w1, "wall_s_cold": w0,
           **stats1}
    print(json.dumps(rec), flush=True)
    with open(os.path.join(ROOT, "gpurun_out", "perf_q3_sf100.json"),
              "w") as f:
        json.dump(rec, f, indent=1)


if __name__ == "__main__":
    main()
