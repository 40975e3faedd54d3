"""End-to-end q3-class multi-operator pipeline on the GPU (one task's worth
of work through the C ABI): filter -> join -> filter -> join -> project ->
grouped aggregate, checked exactly against a python/numpy restatement.

Mirrors the TPC-H q3 stage chain the reference plans
(scheduler/tests/tpch_plan_stability/approved/q3.txt shape, hash-join form
per the prefer_hash_join opt-in, context_checks.rs:1034-1063):
  customer(filter) ⨝ orders(filter) ⨝ lineitem(filter),
  group by l_orderkey, sum(l_extendedprice * (1 - l_discount)).
"""
import numpy as np
import pytest

from datafusion_ballista_amd import gpu, tpch_synth

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ctx():
    c = gpu.GpuStageContext(0)
    yield c
    c.close()


def dec_bytes(vals):
    return tpch_synth.dec128_pairs_np(np.asarray(vals, dtype=np.int64)) \
        .view(np.uint8).reshape(-1)


def test_q3_pipeline_exact(ctx):
    rng = np.random.default_rng(33)
    ncust, nord, nli = 30_000, 150_000, 600_000
    cutoff = 9204  # 1995-03-15

    c_custkey = np.arange(1, ncust + 1, dtype=np.int64)
    rng.shuffle(c_custkey)
    o_orderkey = np.arange(1, nord + 1, dtype=np.int64)
    o_custkey = rng.integers(1, ncust + 1, size=nord, dtype=np.int64)
    o_orderdate = rng.integers(8000, 10500, size=nord, dtype=np.int32)
    l_orderkey = rng.integers(1, nord + 1, size=nli, dtype=np.int64)
    l_shipdate = rng.integers(8000, 10500, size=nli, dtype=np.int32)
    l_price = rng.integers(90000, 10495100, size=nli, dtype=np.int64)
    l_disc = rng.integers(0, 11, size=nli, dtype=np.int64)

    # ---------------- GPU pipeline ----------------
    # stage A: customer filter (c_custkey <= ncust/5 stands in for the
    # mktsegment predicate) -> build side
    cc, _ = ctx.upload_column(c_custkey, gpu.BG_DT_INT64)
    seg_hi = ncust // 5
    mask = ctx.eval_predicates([cc], [(0, gpu.BG_PRED_LT, 0, seg_hi + 1)],
                               ncust)
    cidx, ncust_f = ctx.mask_to_indices(mask, ncust)
    cbuf = ctx.upload(c_custkey)
    ckeys_f = ctx.gather(cbuf, 8, cidx, ncust_f)
    ccol_f = ctx.column(gpu.BG_DT_INT64, ckeys_f, ncust_f)
    cjoin = gpu.GpuHashJoin(ctx, ccol_f, ncust_f)

    # stage B: orders filter (o_orderdate < cutoff), probe customers
    oc, _ = ctx.upload_column(o_custkey, gpu.BG_DT_INT64)
    od, _ = ctx.upload_column(o_orderdate, gpu.BG_DT_DATE32)
    ok, _ = ctx.upload_column(o_orderkey, gpu.BG_DT_INT64)
    omask = ctx.eval_predicates([od], [(0, gpu.BG_PRED_LT, 0, cutoff)], nord)
    oidx, nord_f = ctx.mask_to_indices(omask, nord)
    obuf_ck = ctx.gather(ctx.upload(o_custkey), 8, oidx, nord_f)
    obuf_ok = ctx.gather(ctx.upload(o_orderkey), 8, oidx, nord_f)
    ocol_ck = ctx.column(gpu.BG_DT_INT64, obuf_ck, nord_f)
    pidx, bidx, nmatch = cjoin.probe(ocol_ck, nord_f)
    # surviving orders' orderkeys (probe-side gather through pair indices)
    ok_matched = ctx.gather(obuf_ok, 8, pidx, nmatch)
    okcol = ctx.column(gpu.BG_DT_INT64, ok_matched, nmatch)
    ojoin = gpu.GpuHashJoin(ctx, okcol, nmatch)

    # stage C: lineitem filter (l_shipdate > cutoff), probe orders,
    # project revenue, group by l_orderkey
    lk, _ = ctx.upload_column(l_orderkey, gpu.BG_DT_INT64)
    ld, _ = ctx.upload_column(l_shipdate, gpu.BG_DT_DATE32)
    lmask = ctx.eval_predicates([ld], [(0, gpu.BG_PRED_GT, cutoff, 0)], nli)
    lidx, nli_f = ctx.mask_to_indices(lmask, nli)
    lbuf_k = ctx.gather(ctx.upload(l_orderkey), 8, lidx, nli_f)
    lbuf_p = ctx.gather(ctx.upload(dec_bytes(l_price)), 16, lidx, nli_f)
    lbuf_d = ctx.gather(ctx.upload(dec_bytes(l_disc)), 16, lidx, nli_f)
    lkcol = ctx.column(gpu.BG_DT_INT64, lbuf_k, nli_f)
    lpidx, lbidx, nlm = ojoin.probe(lkcol, nli_f)
    # gather the joined lineitem rows
    jbuf_k = ctx.gather(lbuf_k, 8, lpidx, nlm)
    jbuf_p = ctx.gather(lbuf_p, 16, lpidx, nlm)
    jbuf_d = ctx.gather(lbuf_d, 16, lpidx, nlm)
    jp = ctx.column(gpu.BG_DT_DECIMAL128, jbuf_p, nlm)
    jd = ctx.column(gpu.BG_DT_DECIMAL128, jbuf_d, nlm)
    # revenue = price * (100 - disc)   (scaled ints; scale 2+2=4)
    one_minus = ctx.project_dec128(gpu.BG_PROJ_RSUB_LIT, jd, None, 100, nlm)
    omcol = ctx.column(gpu.BG_DT_DECIMAL128, one_minus, nlm)
    rev = ctx.project_dec128(gpu.BG_PROJ_MUL, jp, omcol, 0, nlm)
    revcol = ctx.column(gpu.BG_DT_DECIMAL128, rev, nlm)
    jkcol = ctx.column(gpu.BG_DT_INT64, jbuf_k, nlm)
    first, acc, counts = ctx.hashagg([jkcol], [revcol],
                                     [gpu.BG_AGG_OP_SUM_DEC128], nlm,
                                     max_groups=max(2 * nord, 64))
    jkeys = jbuf_k.download(np.int64, nlm)
    got = {}
    for g in range(len(first)):
        key = int(jkeys[first[g]])
        got[key] = (int(counts[g]),
                    int.from_bytes(bytes(acc[g, 0]), "little", signed=True))

    # ---------------- reference restatement (numpy/python) --------------
    cust_keep = set(c_custkey[c_custkey <= seg_hi].tolist())
    ord_keep = {}
    for i in range(nord):
        if o_orderdate[i] < cutoff and int(o_custkey[i]) in cust_keep:
            ord_keep[int(o_orderkey[i])] = ord_keep.get(int(o_orderkey[i]), 0) + 1
    want = {}
    for i in range(nli):
        if l_shipdate[i] <= cutoff:
            continue
        okey = int(l_orderkey[i])
        mult = ord_keep.get(okey, 0)
        if not mult:
            continue
        rev_i = int(l_price[i]) * (100 - int(l_disc[i]))
        cnt, s = want.get(okey, (0, 0))
        want[okey] = (cnt + mult, s + rev_i * mult)
    assert got == want
    assert len(got) > 1000  # non-degenerate

    cjoin.free()
    ojoin.free()
