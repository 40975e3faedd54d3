"""GPU parity tests: HIP kernels vs the CPU oracle on the same seeded
inputs, plus the committed golden vectors.  All tests here require an
MI355X (`-m gpu`); they call the product path through the C ABI
(libballista_gpu.so) and compare against oracle/ (test infrastructure).

Bar (BASELINE.json north_star): bit-exact for integer/byte/index work —
partition IDs, filter masks, compacted indices, gathered rows, Decimal128
sums (exact i128).
"""
import decimal
import os

import numpy as np
import pyarrow as pa
import pytest

import oracle
from datafusion_ballista_amd import gpu, tpch_synth

pytestmark = pytest.mark.gpu

HERE = os.path.dirname(os.path.abspath(__file__))
GOLD = os.path.join(HERE, "golden")


@pytest.fixture(scope="module")
def ctx():
    c = gpu.GpuStageContext(0)
    yield c
    c.close()


def dec_bytes(vals):
    return tpch_synth.dec128_pairs_np(np.asarray(vals, dtype=np.int64)) \
        .view(np.uint8).reshape(-1)


# ---------------------------------------------------------------------------
# hashing / partitioning
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("n", [1, 63, 64, 65, 100_000, 1_000_003])
def test_hash_i64_parity(ctx, n):
    rng = np.random.default_rng(n)
    keys = rng.integers(-2**62, 2**62, size=n, dtype=np.int64)
    col, _ = ctx.upload_column(keys, gpu.BG_DT_INT64)
    h = ctx.hash_columns([col], n).download(np.uint64, n)
    want = oracle.hash_columns([("i64", keys)], n)
    assert np.array_equal(h, want)


def test_hash_multi_column_and_dtypes_parity(ctx):
    n = 50_000
    rng = np.random.default_rng(9)
    k64 = rng.integers(-2**40, 2**40, size=n, dtype=np.int64)
    k32 = rng.integers(-2**30, 2**30, size=n, dtype=np.int32)
    dec = rng.integers(-10**12, 10**12, size=n, dtype=np.int64)
    dec16 = dec_bytes(dec)
    c64, _ = ctx.upload_column(k64, gpu.BG_DT_INT64)
    c32, _ = ctx.upload_column(k32, gpu.BG_DT_INT32)
    cd = ctx.column(gpu.BG_DT_DECIMAL128, ctx.upload(dec16), n)
    h = ctx.hash_columns([c64, c32, cd], n).download(np.uint64, n)
    want = oracle.hash_columns(
        [("i64", k64), ("i32", k32), ("dec128", dec16)], n)
    assert np.array_equal(h, want)


def test_hash_nulls_parity(ctx):
    n = 1000
    rng = np.random.default_rng(5)
    a = rng.integers(0, 100, size=n, dtype=np.int64)
    b = rng.integers(0, 100, size=n, dtype=np.int64)
    valid = rng.integers(0, 256, size=(n + 7) // 8, dtype=np.uint8) \
        .astype(np.uint8)
    ca, _ = ctx.upload_column(a, gpu.BG_DT_INT64)
    cb, _ = ctx.upload_column(b, gpu.BG_DT_INT64, validity=valid)
    h = ctx.hash_columns([ca, cb], n).download(np.uint64, n)
    want = oracle.hash_columns([("i64", a), ("i64", b, valid)], n)
    assert np.array_equal(h, want)


@pytest.mark.parametrize("n,k", [(1, 4), (65, 16), (16384, 16),
                                 (16385, 16), (1_000_000, 16),
                                 (300_000, 128), (10_000, 1)])
def test_partition_split_parity(ctx, n, k):
    """bit-exact vs compute_partition_indices restatement: same pids, same
    partition-major row order, same offsets (writer.rs:1259-1279)."""
    rng = np.random.default_rng(n + k)
    keys = rng.integers(-2**62, 2**62, size=n, dtype=np.int64)
    col, _ = ctx.upload_column(keys, gpu.BG_DT_INT64)
    hbuf = ctx.hash_columns([col], n)
    pbuf = ctx.partition_ids(hbuf, n, k)
    ibuf, obuf = ctx.partition_indices(pbuf, n, k)
    ctx.synchronize()
    pids = pbuf.download(np.uint32, n)
    idx = ibuf.download(np.uint32, n)
    offs = obuf.download(np.int64, k + 1)

    want_h = oracle.hash_columns([("i64", keys)], n)
    want_pids = oracle.partition_ids(want_h, k)
    want_idx, want_offs = oracle.partition_indices(want_pids, k)
    assert np.array_equal(pids, want_pids)
    assert np.array_equal(offs, want_offs)
    assert np.array_equal(idx, want_idx)


# ---------------------------------------------------------------------------
# filter / compaction / gather
# ---------------------------------------------------------------------------
def test_eval_predicates_q6_parity(ctx):
    n = 500_000
    li = tpch_synth.lineitem_numpy(n, seed=1)
    disc16 = dec_bytes(li["l_discount"])
    qty16 = dec_bytes(li["l_quantity"])
    sd, _ = ctx.upload_column(li["l_shipdate"], gpu.BG_DT_DATE32)
    cd = ctx.column(gpu.BG_DT_DECIMAL128, ctx.upload(disc16), n)
    cq = ctx.column(gpu.BG_DT_DECIMAL128, ctx.upload(qty16), n)
    preds = [(0, gpu.BG_PRED_GE_LT, tpch_synth.Q6_DATE_LO, tpch_synth.Q6_DATE_HI),
             (1, gpu.BG_PRED_BETWEEN, tpch_synth.Q6_DISC_LO, tpch_synth.Q6_DISC_HI),
             (2, gpu.BG_PRED_LT, 0, tpch_synth.Q6_QTY_LT)]
    mask = ctx.eval_predicates([sd, cd, cq], preds, n)
    ctx.synchronize()
    nbytes = ((n + 63) // 64) * 8
    got = mask.download(np.uint8, nbytes)

    want = oracle.filter_mask([
        ("i32", li["l_shipdate"], None, 0, tpch_synth.Q6_DATE_LO, tpch_synth.Q6_DATE_HI),
        ("dec128", disc16, None, 1, tpch_synth.Q6_DISC_LO, tpch_synth.Q6_DISC_HI),
        ("dec128", qty16, None, 2, 0, tpch_synth.Q6_QTY_LT),
    ], n)
    # oracle mask is byte-granular; compare bit-by-bit over n rows
    got_bits = np.unpackbits(got[: (n + 7) // 8], bitorder="little")[:n]
    want_bits = np.unpackbits(want, bitorder="little")[:n]
    assert np.array_equal(got_bits, want_bits)

    idxbuf, m = ctx.mask_to_indices(mask, n)
    want_idx = oracle.mask_to_indices(want, n)
    assert m == len(want_idx)
    assert np.array_equal(idxbuf.download(np.uint32, m), want_idx)


def test_filter_alltypes_golden(ctx):
    """The reference's own literal filter assertion, on the GPU
    (context_checks.rs:63-77 via tests/golden/)."""
    import json
    az = np.load(os.path.join(GOLD, "alltypes_plain.npz"))
    e = json.load(open(os.path.join(GOLD, "alltypes_expected.json")))
    ids = az["id"].astype(np.int32)
    n = len(ids)
    col, _ = ctx.upload_column(ids, gpu.BG_DT_INT32)
    mask = ctx.eval_predicates([col], [(0, gpu.BG_PRED_GT, 4, 0)], n)
    idxbuf, m = ctx.mask_to_indices(mask, n)
    idx = idxbuf.download(np.uint32, m)
    got_str = [str(az["string_col"][i]) for i in idx]
    assert got_str == e["filter_id_gt4"]["string_col"]


@pytest.mark.parametrize("n", [64, 1000, 100_001])
def test_mask_edges_all_and_none(ctx, n):
    v = np.arange(n, dtype=np.int64)
    col, _ = ctx.upload_column(v, gpu.BG_DT_INT64)
    m_all = ctx.eval_predicates([col], [(0, gpu.BG_PRED_GT, -1, 0)], n)
    _, cnt_all = ctx.mask_to_indices(m_all, n)
    assert cnt_all == n
    m_none = ctx.eval_predicates([col], [(0, gpu.BG_PRED_LT, 0, -10)], n)
    _, cnt_none = ctx.mask_to_indices(m_none, n)
    assert cnt_none == 0


def test_gather_parity(ctx):
    n, m = 200_000, 50_000
    rng = np.random.default_rng(2)
    src = rng.integers(-2**62, 2**62, size=n, dtype=np.int64)
    dec16 = dec_bytes(src)
    idx = rng.integers(0, n, size=m, dtype=np.uint32)
    sbuf = ctx.upload(src)
    dbuf = ctx.upload(dec16)
    ibuf = ctx.upload(idx)
    got8 = ctx.gather(sbuf, 8, ibuf, m).download(np.int64, m)
    assert np.array_equal(got8, src[idx])
    got16 = ctx.gather(dbuf, 16, ibuf, m).download(np.uint8, 16 * m)
    want16 = oracle.gather(dec16, 16, idx)
    assert np.array_equal(got16, want16)


# ---------------------------------------------------------------------------
# fused aggregates
# ---------------------------------------------------------------------------
def q6_cols(ctx, li, n):
    disc16 = dec_bytes(li["l_discount"])
    qty16 = dec_bytes(li["l_quantity"])
    price16 = dec_bytes(li["l_extendedprice"])
    sd, _ = ctx.upload_column(np.asarray(li["l_shipdate"], dtype=np.int32),
                              gpu.BG_DT_DATE32)
    cd = ctx.column(gpu.BG_DT_DECIMAL128, ctx.upload(disc16), n)
    cq = ctx.column(gpu.BG_DT_DECIMAL128, ctx.upload(qty16), n)
    cp = ctx.column(gpu.BG_DT_DECIMAL128, ctx.upload(price16), n)
    return sd, cd, cq, cp, disc16, qty16, price16


def test_q6_parity_synthetic(ctx):
    n = 2_000_000
    li = tpch_synth.lineitem_numpy(n, seed=7)
    sd, cd, cq, cp, disc16, qty16, price16 = q6_cols(ctx, li, n)
    cnt, total = ctx.q6_agg(sd, cd, cq, cp, tpch_synth.Q6_DATE_LO,
                            tpch_synth.Q6_DATE_HI, tpch_synth.Q6_DISC_LO,
                            tpch_synth.Q6_DISC_HI, tpch_synth.Q6_QTY_LT)
    want_cnt, want_sum = oracle.q6(li["l_shipdate"], disc16, qty16, price16,
                                   tpch_synth.Q6_DATE_LO, tpch_synth.Q6_DATE_HI,
                                   tpch_synth.Q6_DISC_LO, tpch_synth.Q6_DISC_HI,
                                   tpch_synth.Q6_QTY_LT)
    assert cnt == want_cnt and cnt > 0
    assert total == want_sum


def test_q6_golden_lineitem_slice(ctx):
    import json
    li = np.load(os.path.join(GOLD, "lineitem_slice.npz"))
    e = json.load(open(os.path.join(GOLD, "lineitem_expected.json")))
    n = len(li["l_shipdate"])
    cols = {"l_shipdate": li["l_shipdate"],
            "l_discount": li["l_discount"],
            "l_quantity": li["l_quantity"],
            "l_extendedprice": li["l_extendedprice"]}
    sd, cd, cq, cp, *_ = q6_cols(ctx, cols, n)
    cnt, total = ctx.q6_agg(sd, cd, cq, cp, e["q6_1996_window"][0],
                            e["q6_1996_window"][1], e["q6_1996_disc"][0],
                            e["q6_1996_disc"][1], e["q6_1996_qty_lt"])
    assert cnt == e["q6_1996_count"]
    assert total == e["q6_1996_sum_scale4"]


def test_q1_parity_synthetic(ctx):
    n = 1_000_000
    li = tpch_synth.lineitem_numpy(n, seed=11)
    qty16 = dec_bytes(li["l_quantity"])
    price16 = dec_bytes(li["l_extendedprice"])
    disc16 = dec_bytes(li["l_discount"])
    tax16 = dec_bytes(li["l_tax"])
    rf, _ = ctx.upload_column(li["l_returnflag"], gpu.BG_DT_DICT8)
    ls, _ = ctx.upload_column(li["l_linestatus"], gpu.BG_DT_DICT8)
    cq = ctx.column(gpu.BG_DT_DECIMAL128, ctx.upload(qty16), n)
    cp = ctx.column(gpu.BG_DT_DECIMAL128, ctx.upload(price16), n)
    cd = ctx.column(gpu.BG_DT_DECIMAL128, ctx.upload(disc16), n)
    ct = ctx.column(gpu.BG_DT_DECIMAL128, ctx.upload(tax16), n)
    sd, _ = ctx.upload_column(li["l_shipdate"], gpu.BG_DT_DATE32)
    got = ctx.q1_agg(rf, ls, cq, cp, cd, ct, sd, tpch_synth.Q1_DATE_LE)
    want = oracle.q1(li["l_returnflag"], li["l_linestatus"], qty16, price16,
                     disc16, tax16, li["l_shipdate"], tpch_synth.Q1_DATE_LE)
    assert got == want
    assert len(got) == 6  # 3 returnflags x 2 linestatus


def test_q1_golden_lineitem_slice(ctx):
    import json
    li = np.load(os.path.join(GOLD, "lineitem_slice.npz"))
    e = json.load(open(os.path.join(GOLD, "lineitem_expected.json")))
    n = len(li["l_shipdate"])
    rf_vals = sorted(set(li["l_returnflag"]))
    ls_vals = sorted(set(li["l_linestatus"]))
    rf_codes = np.array([rf_vals.index(v) for v in li["l_returnflag"]],
                        dtype=np.uint8)
    ls_codes = np.array([ls_vals.index(v) for v in li["l_linestatus"]],
                        dtype=np.uint8)
    rf, _ = ctx.upload_column(rf_codes, gpu.BG_DT_DICT8)
    ls, _ = ctx.upload_column(ls_codes, gpu.BG_DT_DICT8)
    cq = ctx.column(gpu.BG_DT_DECIMAL128,
                    ctx.upload(dec_bytes(li["l_quantity"])), n)
    cp = ctx.column(gpu.BG_DT_DECIMAL128,
                    ctx.upload(dec_bytes(li["l_extendedprice"])), n)
    cd = ctx.column(gpu.BG_DT_DECIMAL128,
                    ctx.upload(dec_bytes(li["l_discount"])), n)
    ct = ctx.column(gpu.BG_DT_DECIMAL128,
                    ctx.upload(dec_bytes(li["l_tax"])), n)
    sd, _ = ctx.upload_column(li["l_shipdate"].astype(np.int32),
                              gpu.BG_DT_DATE32)
    got = ctx.q1_agg(rf, ls, cq, cp, cd, ct, sd, e["q1_cutoff_date32"])
    want_groups = e["q1_groups"]
    got_named = {}
    for g, (count, sums) in got.items():
        key = f"{rf_vals[g >> 4]}|{ls_vals[g & 15]}"
        got_named[key] = {"count": count, "sum_qty": sums[0],
                          "sum_price": sums[1], "sum_disc_price": sums[2],
                          "sum_charge": sums[3], "sum_disc": sums[4]}
    assert got_named == want_groups


# ---------------------------------------------------------------------------
# full sort-shuffle stage (engine + file format)
# ---------------------------------------------------------------------------
def test_sort_shuffle_stage_e2e(ctx, tmp_path):
    from datafusion_ballista_amd import engine, shuffle
    n, k = 100_000, 16
    rng = np.random.default_rng(21)
    table = pa.table({
        "k": pa.array(rng.integers(0, 10_000, size=n, dtype=np.int64)),
        "v": pa.array(rng.integers(-10**9, 10**9, size=n, dtype=np.int64)),
        "d": pa.array(rng.integers(8000, 11000, size=n, dtype=np.int32),
                      type=pa.int32()),
    })
    ex = engine.GpuQueryStageExecutor(ctx, "job-gpu", 3, str(tmp_path),
                                      key_columns=[0], num_partitions=k)
    summaries = ex.execute_query_stage(0, table)
    assert len(summaries) == k
    assert sum(s.num_rows for s in summaries) == n
    # MetricsSet name parity (sort_shuffle/writer.rs:328-440)
    m = ex.collect_plan_metrics()[0]
    for name in ("repart_time_ns", "write_time_ns", "spill_time_ns",
                 "spill_count", "spilled_bytes", "output_rows",
                 "gpu_kernel_time_ns"):
        assert name in m
    assert m["output_rows"] == n

    # oracle-side expectation
    keys = table.column("k").to_numpy()
    h = oracle.hash_columns([("i64", keys)], n)
    pids = oracle.partition_ids(h, k)
    idx, offs = oracle.partition_indices(pids, k)

    data_path = summaries[0].path
    index_path = data_path + ".index"
    for p in range(k):
        batches = shuffle.read_partition(data_path, index_path, p)
        got = pa.Table.from_batches(batches, schema=table.schema) if batches \
            else table.schema.empty_table()
        rows = idx[offs[p]:offs[p + 1]]
        want = table.take(pa.array(rows, type=pa.uint32()))
        assert got.equals(want), f"partition {p} mismatch"


def test_native_library_is_in_tree(ctx):
    """The loaded extension must be the in-tree .so (round-end check:
    'native code not loaded' guard)."""
    p = gpu.lib_path()
    assert os.path.exists(p)
    assert "datafusion_ballista_amd" in p


# ---------------------------------------------------------------------------
# hash join (build/probe)
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("nb,np_,kspace", [(1000, 5000, 500),
                                           (100_000, 300_000, 50_000),
                                           (8, 8, 4)])
def test_hashjoin_parity(ctx, nb, np_, kspace):
    """Inner-join pair multiset == oracle (sorted-pair comparison; SQL pins
    the result set, not intra-row order — the reference's own join goldens
    use ORDER BY, context_checks.rs:986-1003)."""
    rng = np.random.default_rng(nb + np_)
    build = rng.integers(0, kspace, size=nb, dtype=np.int64)
    probe = rng.integers(0, kspace, size=np_, dtype=np.int64)
    bcol, _ = ctx.upload_column(build, gpu.BG_DT_INT64)
    pcol, _ = ctx.upload_column(probe, gpu.BG_DT_INT64)
    join = gpu.GpuHashJoin(ctx, bcol, nb)
    pbuf, bbuf, m = join.probe(pcol, np_)
    got_p = pbuf.download(np.uint32, m)
    got_b = bbuf.download(np.uint32, m)
    join.free()

    want_p, want_b = oracle.hashjoin_pairs(build, probe)
    assert m == len(want_p)
    got = np.sort(got_p.astype(np.uint64) << np.uint64(32) | got_b)
    want = np.sort(want_p.astype(np.uint64) << np.uint64(32) | want_b)
    assert np.array_equal(got, want)
    # probe-major ordering contract: probe indices non-decreasing
    assert np.all(np.diff(got_p.astype(np.int64)) >= 0)


def test_hashjoin_golden_alltypes(ctx):
    """The reference's own join golden: self equi-join of alltypes_plain on
    id, order by id desc limit 5 -> [7,6,5,4,3]
    (context_checks.rs:986-1003 SMJ form and :1034-1063 hash-join opt-in
    form assert the identical table)."""
    import json
    az = np.load(os.path.join(GOLD, "alltypes_plain.npz"))
    e = json.load(open(os.path.join(GOLD, "alltypes_expected.json")))
    ids = az["id"].astype(np.int64)
    n = len(ids)
    col, _ = ctx.upload_column(ids, gpu.BG_DT_INT64)
    join = gpu.GpuHashJoin(ctx, col, n)
    pbuf, bbuf, m = join.probe(col, n)
    got_p = pbuf.download(np.uint32, m)
    join.free()
    # every id unique -> m == n; project t0.id, order desc, limit 5
    joined_ids = sorted((int(ids[i]) for i in got_p), reverse=True)[:5]
    assert joined_ids == e["join_ids_desc5"]


# ---------------------------------------------------------------------------
# general hash group-by
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("n,ngroups_hint", [(10_000, 50), (500_000, 100_000),
                                            (64, 64)])
def test_hashagg_parity(ctx, n, ngroups_hint):
    """q3-class group-by: composite (i64, date32) keys, SUM(Decimal128) +
    SUM(Int64) + COUNT — exact vs the oracle as a key->values map."""
    rng = np.random.default_rng(n)
    k1 = rng.integers(0, max(ngroups_hint // 2, 2), size=n, dtype=np.int64)
    k2 = rng.integers(0, 4, size=n, dtype=np.int32)
    dec = rng.integers(-10**10, 10**10, size=n, dtype=np.int64)
    dec16 = dec_bytes(dec)
    v64 = rng.integers(-10**6, 10**6, size=n, dtype=np.int64)

    c1, _ = ctx.upload_column(k1, gpu.BG_DT_INT64)
    c2, _ = ctx.upload_column(k2, gpu.BG_DT_INT32)
    ca = ctx.column(gpu.BG_DT_DECIMAL128, ctx.upload(dec16), n)
    cb, _ = ctx.upload_column(v64, gpu.BG_DT_INT64)
    first, acc, counts = ctx.hashagg(
        [c1, c2], [ca, cb],
        [gpu.BG_AGG_OP_SUM_DEC128, gpu.BG_AGG_OP_SUM_I64], n,
        max_groups=max(ngroups_hint * 4, 64))

    got = {}
    for g in range(len(first)):
        r = int(first[g])
        key = (int(k1[r]), int(k2[r]))
        sums = [int.from_bytes(bytes(acc[g, a]), "little", signed=True)
                for a in range(2)]
        assert key not in got, "duplicate group emitted"
        got[key] = (int(counts[g]), sums)

    want = oracle.hashagg([k1, k2], [("sum", dec), ("sum", v64)], n)
    assert got == want


def test_hashagg_with_mask_fused_filter(ctx):
    """Fused Filter+Aggregate: the mask from bg_eval_predicates feeds
    bg_hashagg (the q1/q3 stage-1 shape)."""
    n = 200_000
    li = tpch_synth.lineitem_numpy(n, seed=31)
    sd, _ = ctx.upload_column(li["l_shipdate"], gpu.BG_DT_DATE32)
    mask = ctx.eval_predicates(
        [sd], [(0, gpu.BG_PRED_LT, 0, tpch_synth.Q1_DATE_LE + 1)], n)
    rf, _ = ctx.upload_column(li["l_returnflag"], gpu.BG_DT_DICT8)
    qty16 = dec_bytes(li["l_quantity"])
    cq = ctx.column(gpu.BG_DT_DECIMAL128, ctx.upload(qty16), n)
    first, acc, counts = ctx.hashagg([rf], [cq], [gpu.BG_AGG_OP_SUM_DEC128],
                                     n, max_groups=64, mask=mask)
    got = {}
    for g in range(len(first)):
        r = int(first[g])
        got[int(li["l_returnflag"][r])] = (
            int(counts[g]),
            [int.from_bytes(bytes(acc[g, 0]), "little", signed=True)])
    np_mask = li["l_shipdate"] <= tpch_synth.Q1_DATE_LE
    want = oracle.hashagg([li["l_returnflag"]], [("sum", li["l_quantity"])],
                          n, mask=np_mask)
    want = {k[0]: v for k, v in want.items()}
    assert got == want


# ---------------------------------------------------------------------------
# SortExec / Top-K
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("n", [1, 64, 1000, 1_000_000])
def test_sort_rows_i64_parity(ctx, n):
    """Stable sort permutation == numpy stable argsort (both stable =>
    unique answer), asc and desc."""
    rng = np.random.default_rng(n)
    keys = rng.integers(-2**62, 2**62, size=n, dtype=np.int64)
    # inject duplicates to exercise stability
    if n >= 1000:
        keys[::7] = 42
    col, _ = ctx.upload_column(keys, gpu.BG_DT_INT64)
    # order-preserving u64 transform; stable desc = stable argsort of ~u
    u = keys.view(np.uint64) ^ np.uint64(1 << 63)
    for desc in (False, True):
        perm = ctx.sort_rows([col], [desc], n).download(np.uint32, n)
        want = np.argsort(~u if desc else u, kind="stable")
        assert np.array_equal(perm.astype(np.int64), want)


def test_sort_multi_column_topk(ctx):
    """q3 final-stage shape: ORDER BY revenue DESC, o_orderdate ASC,
    LIMIT 10 (SortExec: TopK(fetch=10), approved/q3.txt)."""
    n = 100_000
    rng = np.random.default_rng(8)
    revenue = rng.integers(0, 5000, size=n, dtype=np.int64)  # many ties
    odate = rng.integers(8000, 11000, size=n, dtype=np.int32)
    rc, _ = ctx.upload_column(revenue, gpu.BG_DT_INT64)
    dc, _ = ctx.upload_column(odate, gpu.BG_DT_DATE32)
    perm = ctx.sort_rows([rc, dc], [True, False], n).download(np.uint32, n)
    order = np.lexsort((np.arange(n), odate, -revenue))
    assert np.array_equal(perm[:10].astype(np.int64), order[:10])
    assert np.array_equal(perm.astype(np.int64), order)


def test_hash_utf8_parity(ctx):
    """Utf8 key hashing (q1's repartition keys are 1-char Utf8) — bit-exact
    vs the oracle's bg_hash_str restatement, incl. multi-column combine."""
    strings = [b"A", b"N", b"R", b"", b"hello world", b"x" * 40,
               b"A", b"NFNF", b"\x00\x01\x02"] * 300
    n = len(strings)
    col = ctx.upload_utf8_column(strings)
    h = ctx.hash_columns([col], n).download(np.uint64, n)
    data = np.frombuffer(b"".join(strings), dtype=np.uint8)
    offs = np.zeros(n + 1, dtype=np.int32)
    for i, b in enumerate(strings):
        offs[i + 1] = offs[i] + len(b)
    want = oracle.hash_columns([("utf8", data, offs)], n)
    assert np.array_equal(h, want)
    # multi-column: (utf8, i64) combine
    ids = np.arange(n, dtype=np.int64) % 7
    icol, _ = ctx.upload_column(ids, gpu.BG_DT_INT64)
    h2 = ctx.hash_columns([col, icol], n).download(np.uint64, n)
    want2 = oracle.hash_columns([("utf8", data, offs), ("i64", ids)], n)
    assert np.array_equal(h2, want2)


def test_hashagg_utf8_keys(ctx):
    """Group-by on a Utf8 key column (q1's shape pre-dictionary)."""
    strings = ([b"A|F", b"N|F", b"N|O", b"R|F"] * 2500)
    n = len(strings)
    col = ctx.upload_utf8_column(strings)
    vals = np.arange(n, dtype=np.int64)
    vcol, _ = ctx.upload_column(vals, gpu.BG_DT_INT64)
    first, acc, counts = ctx.hashagg([col], [vcol], [gpu.BG_AGG_OP_SUM_I64],
                                     n, max_groups=64)
    got = {}
    for g in range(len(first)):
        key = strings[first[g]]
        got[key] = (int(counts[g]),
                    int.from_bytes(bytes(acc[g, 0]), "little", signed=True))
    want = {}
    for i, s_ in enumerate(strings):
        c, t = want.get(s_, (0, 0))
        want[s_] = (c + 1, t + i)
    assert got == want


def test_hashagg_utf8_high_cardinality(ctx):
    """Regression: hash_keys_row must actually hash
    Utf8 keys — with the missing case every row collided into one slot and
    a high-cardinality Utf8 group-by degenerated to one linear chain.
    20k distinct keys over 100k rows both pins correctness and would hang
    (O(n*groups)) under the old degenerate path."""
    rng = np.random.default_rng(23)
    n = 100_000
    gids = rng.integers(0, 20_000, size=n)
    strings = [b"cust#%08d" % g for g in gids]
    col = ctx.upload_utf8_column(strings)
    vals = rng.integers(-10**9, 10**9, size=n, dtype=np.int64)
    vcol, _ = ctx.upload_column(vals, gpu.BG_DT_INT64)
    first, acc, counts = ctx.hashagg([col], [vcol], [gpu.BG_AGG_OP_SUM_I64],
                                     n, max_groups=40_000)
    got = {}
    for g in range(len(first)):
        key = strings[first[g]]
        got[key] = (int(counts[g]),
                    int.from_bytes(bytes(acc[g, 0]), "little", signed=True))
    want = {}
    for s_, v in zip(strings, vals):
        c, t = want.get(s_, (0, 0))
        want[s_] = (c + 1, t + int(v))
    assert got == want


def test_sort_dec128_keys(ctx):
    """Decimal128 ORDER BY (q3's revenue DESC is Decimal(38,4)): 16-pass
    radix over the (hi, lo) order-preserving pair."""
    n = 50_000
    rng = np.random.default_rng(12)
    lo_part = rng.integers(-10**12, 10**12, size=n, dtype=np.int64)
    vals = [int(v) * (10**6) + int(w) for v, w in
            zip(lo_part, rng.integers(0, 10**6, size=n))]
    dec16 = np.zeros(16 * n, dtype=np.uint8)
    for i, v in enumerate(vals):
        dec16[16*i:16*(i+1)] = np.frombuffer(
            int(v).to_bytes(16, "little", signed=True), dtype=np.uint8)
    col = ctx.column(gpu.BG_DT_DECIMAL128, ctx.upload(dec16), n)
    for desc in (False, True):
        perm = ctx.sort_rows([col], [desc], n).download(np.uint32, n)
        order = sorted(range(n), key=lambda i: (-vals[i] if desc else vals[i], i))
        assert perm.astype(np.int64).tolist() == order


def test_sort_shuffle_stage_multi_input(ctx, tmp_path):
    """Multi-input task (M local input partitions): per-input encoded
    streams concatenated per output partition in the consolidated file
    (execute_shuffle_write writer.rs:564-753, 861-884); readers cross the
    sub-stream boundaries transparently."""
    from datafusion_ballista_amd import engine, shuffle
    k = 8
    rng = np.random.default_rng(55)
    tables = []
    for i in range(3):
        n = 20_000 + i * 5_000
        tables.append(pa.table({
            "k": pa.array(rng.integers(0, 3_000, size=n, dtype=np.int64)),
            "v": pa.array(rng.integers(-10**9, 10**9, size=n, dtype=np.int64)),
        }))
    ex = engine.GpuQueryStageExecutor(ctx, "job-mi", 5, str(tmp_path),
                                      key_columns=[0], num_partitions=k)
    summaries = ex.execute_query_stage(2, tables)
    total = sum(t.num_rows for t in tables)
    assert sum(s.num_rows for s in summaries) == total

    data_path = summaries[0].path
    index_path = data_path + ".index"
    for p in range(k):
        batches = shuffle.read_partition(data_path, index_path, p)
        got = pa.Table.from_batches(batches, schema=tables[0].schema) \
            if batches else tables[0].schema.empty_table()
        # expected: concat of each input's partition-p rows, in input order
        chunks = []
        for t in tables:
            n = t.num_rows
            keys = t.column("k").to_numpy()
            h = oracle.hash_columns([("i64", keys)], n)
            pids = oracle.partition_ids(h, k)
            idx, offs = oracle.partition_indices(pids, k)
            rows = idx[offs[p]:offs[p + 1]]
            chunks.append(t.take(pa.array(rows, type=pa.uint32())))
        want = pa.concat_tables(chunks)
        assert got.equals(want.combine_chunks()), f"partition {p}"


def test_q1_partial_final_two_stage(ctx, tmp_path):
    """Partial/Final aggregate decomposition (AggregateExec Partial on each
    task shard -> shuffle -> Final merge; approved/q1.txt stages 1-2):
    partials from two GPU tasks merge to exactly the single-pass oracle
    answer; AVG at the final stage checked at 1e-6 relative
    (benchmarks/src/lib.rs:35 float tolerance)."""
    n = 400_000
    li = tpch_synth.lineitem_numpy(n, seed=77)
    half = n // 2

    def run_partial(sl):
        m = sl.stop - sl.start
        rf, _ = ctx.upload_column(li["l_returnflag"][sl], gpu.BG_DT_DICT8)
        ls, _ = ctx.upload_column(li["l_linestatus"][sl], gpu.BG_DT_DICT8)
        cq = ctx.column(gpu.BG_DT_DECIMAL128,
                        ctx.upload(dec_bytes(li["l_quantity"][sl])), m)
        cp = ctx.column(gpu.BG_DT_DECIMAL128,
                        ctx.upload(dec_bytes(li["l_extendedprice"][sl])), m)
        cd = ctx.column(gpu.BG_DT_DECIMAL128,
                        ctx.upload(dec_bytes(li["l_discount"][sl])), m)
        ct = ctx.column(gpu.BG_DT_DECIMAL128,
                        ctx.upload(dec_bytes(li["l_tax"][sl])), m)
        sd, _ = ctx.upload_column(li["l_shipdate"][sl], gpu.BG_DT_DATE32)
        return ctx.q1_agg(rf, ls, cq, cp, cd, ct, sd, tpch_synth.Q1_DATE_LE)

    p1 = run_partial(slice(0, half))
    p2 = run_partial(slice(half, n))

    # FINAL aggregate: merge partial states (exact integer adds)
    merged = {}
    for part in (p1, p2):
        for g, (cnt, sums) in part.items():
            c0, s0 = merged.get(g, (0, [0] * 5))
            merged[g] = (c0 + cnt, [a + b for a, b in zip(s0, sums)])

    want = oracle.q1(li["l_returnflag"], li["l_linestatus"],
                     dec_bytes(li["l_quantity"]),
                     dec_bytes(li["l_extendedprice"]),
                     dec_bytes(li["l_discount"]), dec_bytes(li["l_tax"]),
                     li["l_shipdate"], tpch_synth.Q1_DATE_LE)
    assert merged == want  # partial+final == single-pass, bit-exact

    # final-stage AVG columns (avg_qty, avg_price, avg_disc) at 1e-6 rel
    for g, (cnt, sums) in merged.items():
        wc, ws = want[g]
        for idx_a, scale in ((0, 100.0), (1, 100.0), (4, 100.0)):
            got_avg = sums[idx_a] / cnt / scale
            want_avg = ws[idx_a] / wc / scale
            assert abs(got_avg - want_avg) <= 1e-6 * max(abs(want_avg), 1e-30)


# ---------------------------------------------------------------------------
# sort-merge join (reference-default partitioned join shape)
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("nb,np_,kspace", [(1000, 5000, 500),
                                           (100_000, 300_000, 50_000)])
def test_merge_join_parity(ctx, nb, np_, kspace):
    """SMJ(sorted build, sorted probe) emits exactly the hash-join pair
    multiset, mapped through the sort permutations; output fully ordered
    (probe-major, build ascending)."""
    rng = np.random.default_rng(nb * 3 + np_)
    build = rng.integers(0, kspace, size=nb, dtype=np.int64)
    probe = rng.integers(0, kspace, size=np_, dtype=np.int64)

    bcol, bbuf_raw = ctx.upload_column(build, gpu.BG_DT_INT64)
    pcol, pbuf_raw = ctx.upload_column(probe, gpu.BG_DT_INT64)
    bperm = ctx.sort_rows([bcol], [False], nb)
    pperm = ctx.sort_rows([pcol], [False], np_)
    bsorted = ctx.gather(bbuf_raw, 8, bperm, nb)
    psorted = ctx.gather(pbuf_raw, 8, pperm, np_)
    ppos_buf, bpos_buf, m = ctx.merge_join(bsorted, nb, psorted, np_)
    ppos = ppos_buf.download(np.uint32, m)
    bpos = bpos_buf.download(np.uint32, m)
    # map sorted positions back to original row ids
    bperm_h = bperm.download(np.uint32, nb)
    pperm_h = pperm.download(np.uint32, np_)
    got_pairs = np.sort(pperm_h[ppos].astype(np.uint64) << np.uint64(32)
                        | bperm_h[bpos])

    want_p, want_b = oracle.hashjoin_pairs(build, probe)
    want_pairs = np.sort(want_p.astype(np.uint64) << np.uint64(32) | want_b)
    assert m == len(want_p)
    assert np.array_equal(got_pairs, want_pairs)
    # ordered-output contract: probe positions non-decreasing, build
    # positions ascending within a probe row
    assert np.all(np.diff(ppos.astype(np.int64)) >= 0)
    same = np.diff(ppos.astype(np.int64)) == 0
    assert np.all(np.diff(bpos.astype(np.int64))[same] > 0)


# ---------------------------------------------------------------------------
# Parquet decode stage 1: Snappy pages
# ---------------------------------------------------------------------------
def test_snappy_decompress_parity(ctx):
    """Device Snappy vs pyarrow's reference codec on structured, random and
    highly-compressible pages (incl. overlapping copies / long literals)."""
    rng = np.random.default_rng(4)
    payloads = [
        b"hello world " * 1000,                       # short-offset copies
        bytes(rng.integers(0, 256, 100_000, dtype=np.uint8)),  # incompressible
        np.arange(50_000, dtype=np.int64).tobytes(),  # structured
        b"\x00" * 65_000,                             # RLE (offset-1 copies)
        (b"abcdefgh" * 9000)[:70_001],                # odd length
        b"x",                                         # tiny
    ]
    comp = [pa.compress(p, codec="snappy", asbytes=True) for p in payloads]
    pages = []
    dsts = []
    for c, p in zip(comp, payloads):
        src = ctx.upload(np.frombuffer(c, dtype=np.uint8))
        dst = ctx.alloc(max(len(p), 1))
        pages.append((src, len(c), dst, len(p)))
        dsts.append(dst)
    lens = ctx.snappy_decompress(pages)
    for i, p in enumerate(payloads):
        assert lens[i] == len(p), f"page {i} length mismatch"
        got = dsts[i].download(np.uint8, len(p)).tobytes()
        assert got == p, f"page {i} content mismatch"


def test_snappy_big_page_list_ranking_parity(ctx):
    """Pages with compressed size above the big-page threshold (98304)
    replay matches by list ranking (parent pointers + jump-4 doubling +
    root gather) instead of the serial in-order wave replay.  Exact-byte
    parity on the page shapes that route there: FLBA(7) decimal pages
    (the q6-feed bound: ~150k matches, chain depth ~450), low-cardinality
    int64 pages (depth ~200), and a just-above-threshold page."""
    rng = np.random.default_rng(11)
    payloads = []
    vals = rng.integers(90000, 10495100, size=1_048_576 // 7)
    payloads.append(b"".join(int(v).to_bytes(7, "big") for v in vals))
    payloads.append(rng.integers(0, 100, size=1_048_576 // 8,
                                 dtype=np.int64).tobytes())
    # heavy-RLE page (class A: always list-ranked regardless of count):
    # a repeated 64-B pattern compresses ~20x with deep offset-64 chains
    payloads.append(rng.integers(0, 256, size=64,
                                 dtype=np.uint8).tobytes() * (1_048_576
                                                              // 64))
    # and a pure zero page (offset-1 chains, the deepest possible)
    payloads.append(b"\x00" * 1_048_576)
    # binary-search a raw size whose flba7 compressed size lands just
    # above the threshold
    for raw_bytes in (131_072, 160_000, 200_000):
        vv = rng.integers(90000, 10495100, size=raw_bytes // 7)
        p = b"".join(int(v).to_bytes(7, "big") for v in vv)
        if len(pa.compress(p, codec="snappy", asbytes=True)) > 98304:
            payloads.append(p)
            break
    comp = [pa.compress(p, codec="snappy", asbytes=True) for p in payloads]
    assert any(len(c) > 98304 for c in comp)
    pages = []
    dsts = []
    for c, p in zip(comp, payloads):
        src = ctx.upload(np.frombuffer(c, dtype=np.uint8))
        dst = ctx.alloc(len(p))
        pages.append((src, len(c), dst, len(p)))
        dsts.append(dst)
    lens = ctx.snappy_decompress(pages)
    for i, p in enumerate(payloads):
        assert lens[i] == len(p), f"big page {i} length mismatch"
        got = dsts[i].download(np.uint8, len(p)).tobytes()
        assert got == p, f"big page {i} content mismatch"


def test_snappy_malformed_is_rejected(ctx):
    """Truncated/garbage pages must fail loudly (-1), not write junk."""
    good = pa.compress(b"A" * 5000, codec="snappy", asbytes=True)
    bad1 = good[: len(good) // 2]           # truncated
    bad2 = b"\xff\xff\xff\xff\xff\xff"      # absurd varint
    pages = []
    for c in (bad1, bad2):
        src = ctx.upload(np.frombuffer(c, dtype=np.uint8))
        dst = ctx.alloc(5000)
        pages.append((src, len(c), dst, 5000))
    lens = ctx.snappy_decompress(pages)
    assert lens == [-1, -1]


def test_gather_varlen_parity(ctx):
    """Utf8 take (string shuffle payload): offsets + bytes vs pyarrow."""
    rng = np.random.default_rng(17)
    n, m = 50_000, 20_000
    strings = [bytes(rng.integers(65, 91, size=int(L), dtype=np.uint8))
               for L in rng.integers(0, 40, size=n)]
    data = np.frombuffer(b"".join(strings), dtype=np.uint8)
    offs = np.zeros(n + 1, dtype=np.int32)
    for i, s_ in enumerate(strings):
        offs[i + 1] = offs[i] + len(s_)
    idx = rng.integers(0, n, size=m, dtype=np.uint32)

    dbuf = ctx.upload(data if len(data) else np.zeros(1, dtype=np.uint8))
    obuf = ctx.upload(offs)
    ibuf = ctx.upload(idx)
    out_offs, out_data, total = ctx.gather_varlen(dbuf, obuf, ibuf, m,
                                                  max_bytes=int(data.nbytes) + 1)
    got_offs = out_offs.download(np.int32, m + 1)
    got_data = out_data.download(np.uint8, max(total, 1))[:total].tobytes()

    arr = pa.array([s_.decode() for s_ in strings], type=pa.utf8())
    want = arr.take(pa.array(idx, type=pa.uint32()))
    want_strings = [v.as_py().encode() for v in want]
    assert got_offs[0] == 0 and got_offs[-1] == total
    got_strings = [got_data[got_offs[i]:got_offs[i + 1]] for i in range(m)]
    assert got_strings == want_strings


def test_sort_shuffle_stage_with_utf8_payload(ctx, tmp_path):
    """Shuffle stage carrying a Utf8 payload column AND a Utf8 key column
    (q1's repartition keys are 1-char Utf8 — SURVEY.md §8a row 4)."""
    from datafusion_ballista_amd import engine, shuffle
    n, k = 30_000, 8
    rng = np.random.default_rng(71)
    flags = ["A", "N", "R"]
    strs = [flags[i] for i in rng.integers(0, 3, size=n)]
    comments = ["".join(chr(65 + c) for c in rng.integers(0, 26, size=int(L)))
                for L in rng.integers(0, 30, size=n)]
    table = pa.table({
        "flag": pa.array(strs, type=pa.utf8()),
        "v": pa.array(rng.integers(-10**9, 10**9, size=n, dtype=np.int64)),
        "comment": pa.array(comments, type=pa.utf8()),
    })
    ex = engine.GpuQueryStageExecutor(ctx, "job-utf8", 6, str(tmp_path),
                                      key_columns=[0], num_partitions=k)
    summaries = ex.execute_query_stage(0, table)
    assert sum(s.num_rows for s in summaries) == n

    # oracle expectation: hash the utf8 key column
    data = np.frombuffer("".join(strs).encode(), dtype=np.uint8)
    offs = np.zeros(n + 1, dtype=np.int32)
    for i, s_ in enumerate(strs):
        offs[i + 1] = offs[i] + len(s_)
    h = oracle.hash_columns([("utf8", data, offs)], n)
    pids = oracle.partition_ids(h, k)
    idx, poffs = oracle.partition_indices(pids, k)

    data_path = summaries[0].path
    for p in range(k):
        batches = shuffle.read_partition(data_path, data_path + ".index", p)
        got = pa.Table.from_batches(batches, schema=table.schema) if batches \
            else table.schema.empty_table()
        rows = idx[poffs[p]:poffs[p + 1]]
        want = table.take(pa.array(rows, type=pa.uint32()))
        assert got.equals(want), f"partition {p}"


def test_hashagg_min_max(ctx):
    """MIN/MAX aggregate ops (AggregateExec breadth) vs a numpy oracle."""
    n = 300_000
    rng = np.random.default_rng(23)
    keys = rng.integers(0, 1000, size=n, dtype=np.int64)
    vals = rng.integers(-2**62, 2**62, size=n, dtype=np.int64)
    kc, _ = ctx.upload_column(keys, gpu.BG_DT_INT64)
    vc, _ = ctx.upload_column(vals, gpu.BG_DT_INT64)
    first, acc, counts = ctx.hashagg(
        [kc], [vc, vc, vc],
        [gpu.BG_AGG_OP_MIN_I64, gpu.BG_AGG_OP_MAX_I64, gpu.BG_AGG_OP_SUM_I64],
        n, max_groups=4096)
    got = {}
    for g in range(len(first)):
        key = int(keys[first[g]])
        got[key] = (gpu.decode_agg_value(gpu.BG_AGG_OP_MIN_I64, bytes(acc[g, 0])),
                    gpu.decode_agg_value(gpu.BG_AGG_OP_MAX_I64, bytes(acc[g, 1])),
                    gpu.decode_agg_value(gpu.BG_AGG_OP_SUM_I64, bytes(acc[g, 2])),
                    int(counts[g]))
    want = {}
    for k_ in np.unique(keys):
        sel = vals[keys == k_]
        want[int(k_)] = (int(sel.min()), int(sel.max()),
                         int(sel.astype(object).sum()), int(len(sel)))
    assert got == want


def test_resident_shuffle_matches_file_path(ctx, tmp_path):
    """ballista.gpu.resident_shuffle: the HBM-resident partition buffers
    hold exactly the rows the file path writes (the RCCL exchange then
    moves these buffers instead of Flight moving file bytes)."""
    from datafusion_ballista_amd import engine, shuffle
    n, k = 60_000, 8
    rng = np.random.default_rng(99)
    table = pa.table({
        "k": pa.array(rng.integers(0, 5_000, size=n, dtype=np.int64)),
        "v": pa.array(rng.integers(-10**9, 10**9, size=n, dtype=np.int64)),
    })
    ex = engine.GpuResidentShuffleStage(ctx, "job-res", 9, str(tmp_path),
                                        key_columns=[0], num_partitions=k)
    offsets, bufs, schema = ex.execute_query_stage_resident(0, table)
    assert offsets[0] == 0 and offsets[-1] == n
    keys_pm = bufs[0].download(np.int64, n)
    vals_pm = bufs[1].download(np.int64, n)

    # file path on the same input
    summaries = ex.execute_query_stage(1, table)
    data_path = summaries[0].path
    for p in range(k):
        lo, hi = int(offsets[p]), int(offsets[p + 1])
        batches = shuffle.read_partition(data_path, data_path + ".index", p)
        want = pa.Table.from_batches(batches, schema=schema) if batches \
            else schema.empty_table()
        assert np.array_equal(keys_pm[lo:hi], want.column("k").to_numpy())
        assert np.array_equal(vals_pm[lo:hi], want.column("v").to_numpy())


def test_gather_bits_parity(ctx):
    """Validity-bitmap take vs a numpy bit oracle (null columns through the
    shuffle/filter materialisation)."""
    rng = np.random.default_rng(41)
    n, m = 100_000, 40_000
    valid = rng.integers(0, 256, size=(n + 7) // 8, dtype=np.uint8)
    idx = rng.integers(0, n, size=m, dtype=np.uint32)
    vbuf = ctx.upload(valid)
    ibuf = ctx.upload(idx)
    nwords = (m + 63) // 64
    out = ctx.alloc(nwords * 8)
    import ctypes
    gpu._check(ctx.L.bg_gather_bits(vbuf.ptr, ibuf.ptr, ctypes.c_int64(m),
                                    out.ptr), "bg_gather_bits")
    got_bits = np.unpackbits(out.download(np.uint8, nwords * 8),
                             bitorder="little")[:m]
    src_bits = np.unpackbits(valid, bitorder="little")[:n]
    want = src_bits[idx]
    assert np.array_equal(got_bits, want)


# ---------------------------------------------------------------------------
# device LZ4 (shuffle codec compress half)
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("case", ["zeros", "random", "text", "ints",
                                  "multiblock", "tiny"])
def test_lz4_compress_roundtrip(ctx, case):
    """Device-compressed LZ4 frames must decode bit-exact with Arrow's own
    (reference) LZ4_FRAME codec — the same decoder ShuffleReaderExec uses
    for shuffle batches (codec default lz4, config.rs:413-415)."""
    rng = np.random.default_rng(len(case))
    payloads = {
        "zeros": b"\x00" * 200_000,
        "random": bytes(rng.integers(0, 256, 150_000, dtype=np.uint8)),
        "text": (b"the quick brown fox jumps over the lazy dog. " * 5000),
        "ints": np.arange(50_000, dtype=np.int64).tobytes(),
        "multiblock": bytes(rng.integers(0, 64, 300_000, dtype=np.uint8)),
        "tiny": b"abc",
    }
    data = payloads[case]
    src = ctx.upload(np.frombuffer(data, dtype=np.uint8))
    sizes, slots = ctx.lz4_compress(src, len(data))
    slot_bytes = slots.download(np.uint8, max(len(sizes) * 65544, 1)).tobytes()
    frame = gpu.lz4_frame_assemble(sizes, slot_bytes, len(data))
    got = pa.decompress(frame, len(data), codec="lz4", asbytes=True)
    assert got == data
    if case in ("zeros", "text"):
        assert len(frame) < len(data) // 2, "compressible data must shrink"
    if case == "ints":
        assert len(frame) < len(data), "structured data must shrink some"


def test_sort_shuffle_stage_gpu_codec(ctx, tmp_path):
    """gpu_codec mode: device-LZ4 batch bodies + handwritten IPC metadata
    must read back (pyarrow reader == the arrow decoder family
    ShuffleReaderExec uses) identical to the host-codec path."""
    from datafusion_ballista_amd import engine, shuffle
    n, k = 120_000, 16
    rng = np.random.default_rng(63)
    table = pa.table({
        "k": pa.array(rng.integers(0, 9_000, size=n, dtype=np.int64)),
        "d": pa.array(rng.integers(8000, 11000, size=n, dtype=np.int32),
                      type=pa.int32()),
        "v": pa.array(rng.integers(-10**9, 10**9, size=n, dtype=np.int64)),
    })
    ex_cpu = engine.GpuQueryStageExecutor(ctx, "job-cc", 1, str(tmp_path),
                                          key_columns=[0], num_partitions=k)
    ex_gpu = engine.GpuQueryStageExecutor(ctx, "job-gc", 1, str(tmp_path),
                                          key_columns=[0], num_partitions=k,
                                          gpu_codec=True)
    s_cpu = ex_cpu.execute_query_stage(0, table)
    s_gpu = ex_gpu.execute_query_stage(0, table)
    assert sum(x.num_rows for x in s_gpu) == n
    p_cpu = s_cpu[0].path
    p_gpu = s_gpu[0].path
    for p in range(k):
        want = shuffle.read_partition(p_cpu, p_cpu + ".index", p)
        got = shuffle.read_partition(p_gpu, p_gpu + ".index", p)
        wt = pa.Table.from_batches(want, schema=table.schema) if want else \
            table.schema.empty_table()
        gt = pa.Table.from_batches(got, schema=table.schema) if got else \
            table.schema.empty_table()
        assert gt.equals(wt), f"partition {p} mismatch"


@pytest.mark.parametrize("n,k", [(1000, 4), (16384, 16), (16500, 16),
                                 (500_000, 16), (300_000, 64), (5_000, 1)])
def test_hash_repartition_fused_parity(ctx, n, k):
    """The fused LDS-write-combining materialiser must produce bit-exact
    the unfused path's outputs: indices, offsets, and every partition-major
    payload buffer (stable order included)."""
    rng = np.random.default_rng(n * 7 + k)
    keys = rng.integers(-2**62, 2**62, size=n, dtype=np.int64)
    vals8 = rng.integers(-10**9, 10**9, size=n, dtype=np.int64)
    vals4 = rng.integers(-10**6, 10**6, size=n, dtype=np.int32)
    dec16 = dec_bytes(rng.integers(-10**10, 10**10, size=n, dtype=np.int64))
    kc, _ = ctx.upload_column(keys, gpu.BG_DT_INT64)
    c8, _ = ctx.upload_column(vals8, gpu.BG_DT_INT64)
    c4, _ = ctx.upload_column(vals4, gpu.BG_DT_INT32)
    cd = ctx.column(gpu.BG_DT_DECIMAL128, ctx.upload(dec16), n)
    if k <= 16:
        payload, esizes = [kc, c8, c4, cd], (8, 8, 4, 16)
    else:  # large k: smaller payload so the per-wave LDS tiles fit
        payload, esizes = [kc, c4], (8, 4)

    i1, o1, b1 = ctx.hash_repartition([kc], payload, n, k)
    i2, o2, b2 = ctx.hash_repartition_fused([kc], payload, n, k)
    ctx.synchronize()
    assert np.array_equal(o1.download(np.int64, k + 1),
                          o2.download(np.int64, k + 1))
    assert np.array_equal(i1.download(np.uint32, n), i2.download(np.uint32, n))
    for a, b, esz in zip(b1, b2, esizes):
        assert np.array_equal(a.download(np.uint8, n * esz),
                              b.download(np.uint8, n * esz))


@pytest.mark.parametrize("case", ["pyarrow", "own_codec", "stored", "mixed"])
def test_lz4_decompress_roundtrip(ctx, case):
    """Device LZ4-frame decode vs frames produced by (a) Arrow's reference
    encoder, (b) our own device compressor, (c) stored-block frames —
    the shuffle-read ingest path."""
    rng = np.random.default_rng(len(case) * 3)
    data = (b"shuffle bytes " * 20_000 +
            bytes(rng.integers(0, 256, 50_000, dtype=np.uint8)))
    if case == "pyarrow":
        frame = pa.compress(data, codec="lz4", asbytes=True)
    elif case == "own_codec":
        src = ctx.upload(np.frombuffer(data, dtype=np.uint8))
        sizes, slots = ctx.lz4_compress(src, len(data))
        slot_bytes = slots.download(np.uint8,
                                    max(len(sizes) * 65544, 1)).tobytes()
        frame = gpu.lz4_frame_assemble(sizes, slot_bytes, len(data))
    elif case == "stored":
        import struct
        frame = (gpu.LZ4_FRAME_HEADER +
                 struct.pack("<I", len(data) | 0x80000000) + data +
                 b"\x00\x00\x00\x00")
    else:  # mixed: compressible + stored blocks from our codec
        data = b"\x00" * 70_000 + bytes(rng.integers(0, 256, 70_000,
                                                     dtype=np.uint8))
        src = ctx.upload(np.frombuffer(data, dtype=np.uint8))
        sizes, slots = ctx.lz4_compress(src, len(data))
        slot_bytes = slots.download(np.uint8,
                                    max(len(sizes) * 65544, 1)).tobytes()
        frame = gpu.lz4_frame_assemble(sizes, slot_bytes, len(data))
    fsrc = ctx.upload(np.frombuffer(frame, dtype=np.uint8))
    fdst = ctx.alloc(len(data))
    lens = ctx.lz4_decompress([(fsrc, len(frame), fdst, len(data))])
    assert lens == [len(data)]
    assert fdst.download(np.uint8, len(data)).tobytes() == data


def test_device_shuffle_read_roundtrip(ctx, tmp_path):
    """Stage N writes with the GPU codec; stage N+1 ingests the partition
    ON DEVICE (read_partition_gpu) — columns must equal the host reader's
    view (ShuffleReaderExec local-read parity, shuffle_reader.rs:1120-1168)."""
    from datafusion_ballista_amd import engine, shuffle
    n, k = 80_000, 8
    rng = np.random.default_rng(311)
    table = pa.table({
        "k": pa.array(rng.integers(0, 7_000, size=n, dtype=np.int64)),
        "d": pa.array(rng.integers(8000, 11000, size=n, dtype=np.int32),
                      type=pa.int32()),
    })
    ex = engine.GpuQueryStageExecutor(ctx, "job-dr", 2, str(tmp_path),
                                      key_columns=[0], num_partitions=k,
                                      gpu_codec=True)
    summaries = ex.execute_query_stage(0, table)
    data_path = summaries[0].path
    index_path = data_path + ".index"
    for p in range(k):
        want_batches = shuffle.read_partition(data_path, index_path, p)
        want = pa.Table.from_batches(want_batches, schema=table.schema) \
            if want_batches else table.schema.empty_table()
        m, cols = shuffle.read_partition_gpu(ctx, data_path, index_path, p,
                                             table.schema)
        assert m == want.num_rows
        if m:
            got_k = cols[0].download(np.int64, m)
            got_d = cols[1].download(np.int32, m)
            assert np.array_equal(got_k, want.column("k").to_numpy())
            assert np.array_equal(got_d, want.column("d").to_numpy())


# ---------------------------------------------------------------------------
# edge hardening
# ---------------------------------------------------------------------------
def test_empty_inputs(ctx):
    """n=0 through filter/compact/hash/split/join/sort — no crashes, empty
    results (the reference's empty-partition paths, writer.rs tests)."""
    empty64 = np.zeros(0, dtype=np.int64)
    col, _ = ctx.upload_column(empty64, gpu.BG_DT_INT64)
    mask = ctx.eval_predicates([col], [(0, gpu.BG_PRED_GT, 0, 0)], 0)
    _, cnt = ctx.mask_to_indices(mask, 0)
    assert cnt == 0
    h = ctx.hash_columns([col], 0)
    p = ctx.partition_ids(h, 0, 4)
    idx, offs = ctx.partition_indices(p, 0, 4)
    assert np.array_equal(offs.download(np.int64, 5), np.zeros(5, np.int64))
    join = gpu.GpuHashJoin(ctx, col, 0)
    _, _, m = join.probe(col, 0)
    assert m == 0
    join.free()
    perm = ctx.sort_rows([col], [False], 0)
    assert perm.download(np.uint32, 0).size == 0


def test_join_empty_build_side(ctx):
    """Probing an empty build side yields zero matches (inner join)."""
    build = np.zeros(0, dtype=np.int64)
    probe = np.arange(1000, dtype=np.int64)
    bcol, _ = ctx.upload_column(build, gpu.BG_DT_INT64)
    pcol, _ = ctx.upload_column(probe, gpu.BG_DT_INT64)
    join = gpu.GpuHashJoin(ctx, bcol, 0)
    _, _, m = join.probe(pcol, 1000)
    assert m == 0
    join.free()


def test_hashagg_table_full_fails_loudly(ctx):
    """max_groups too small must error (BG_ERR_INVALID) at the RAW ABI, not
    corrupt (the python wrapper adds auto-grow on top, tested separately)."""
    import ctypes
    n = 10_000
    keys = np.arange(n, dtype=np.int64)  # n distinct groups
    vals = np.ones(n, dtype=np.int64)
    kc, _ = ctx.upload_column(keys, gpu.BG_DT_INT64)
    vc, _ = ctx.upload_column(vals, gpu.BG_DT_INT64)
    first = ctx.alloc(4 * 16)
    acc = ctx.alloc(16 * 16)
    counts = ctx.alloc(8 * 16)
    ng = ctypes.c_int64()
    rc = ctx.L.bg_hashagg((gpu.BgColumn * 1)(kc), 1, (gpu.BgColumn * 1)(vc),
                          (ctypes.c_int32 * 1)(gpu.BG_AGG_OP_SUM_I64), 1,
                          None, ctypes.c_int64(n), ctypes.c_int64(16),
                          first.ptr, acc.ptr, counts.ptr, ctypes.byref(ng))
    assert rc == -3  # BG_ERR_INVALID
    assert b"table full" in ctx.L.bg_last_error()


def test_partition_split_k4096(ctx):
    """Largest supported k (leader-loop path, 4096 partitions)."""
    n, k = 300_000, 4096
    rng = np.random.default_rng(4096)
    keys = rng.integers(-2**62, 2**62, size=n, dtype=np.int64)
    col, _ = ctx.upload_column(keys, gpu.BG_DT_INT64)
    hbuf = ctx.hash_columns([col], n)
    pbuf = ctx.partition_ids(hbuf, n, k)
    ibuf, obuf = ctx.partition_indices(pbuf, n, k)
    ctx.synchronize()
    idx = ibuf.download(np.uint32, n)
    offs = obuf.download(np.int64, k + 1)
    want_h = oracle.hash_columns([("i64", keys)], n)
    want_p = oracle.partition_ids(want_h, k)
    want_idx, want_offs = oracle.partition_indices(want_p, k)
    assert np.array_equal(offs, want_offs)
    assert np.array_equal(idx, want_idx)


def test_two_stage_pipeline_device_resident(ctx, tmp_path):
    """Full two-stage flow, all compute on device: stage 1 hash-repartitions
    orders by custkey and writes GPU-codec shuffle files; stage 2 device-
    reads each partition, joins against a filtered customer build side and
    group-by-sums — exact vs a python restatement (the q3-class two-stage
    execution model end to end)."""
    from datafusion_ballista_amd import engine, shuffle
    rng = np.random.default_rng(555)
    ncust, nord, k = 5_000, 120_000, 8
    c_custkey = np.arange(1, ncust + 1, dtype=np.int64)
    o_custkey = rng.integers(1, ncust + 1, size=nord, dtype=np.int64)
    o_total = rng.integers(1, 10**6, size=nord, dtype=np.int64)
    table = pa.table({"o_custkey": pa.array(o_custkey),
                      "o_total": pa.array(o_total)})
    ex = engine.GpuQueryStageExecutor(ctx, "job-2s", 1, str(tmp_path),
                                      key_columns=[0], num_partitions=k,
                                      gpu_codec=True)
    summaries = ex.execute_query_stage(0, table)
    data_path = summaries[0].path

    # stage 2: per partition — device read, semi-join filter on custkey
    # (custkey <= ncust//3 stands in for the dimension filter), group-by sum
    keep_hi = ncust // 3
    got = {}
    for p in range(k):
        m, cols = shuffle.read_partition_gpu(ctx, data_path,
                                             data_path + ".index", p,
                                             table.schema)
        if m == 0:
            continue
        ck = ctx.column(gpu.BG_DT_INT64, cols[0], m)
        mask = ctx.eval_predicates([ck], [(0, gpu.BG_PRED_LT, 0, keep_hi + 1)],
                                   m)
        tot = ctx.column(gpu.BG_DT_INT64, cols[1], m)
        first, acc, counts = ctx.hashagg([ck], [tot],
                                         [gpu.BG_AGG_OP_SUM_I64], m,
                                         max_groups=max(2 * keep_hi, 64),
                                         mask=mask)
        keys_h = cols[0].download(np.int64, m)
        for g in range(len(first)):
            key = int(keys_h[first[g]])
            cnt, s_ = got.get(key, (0, 0))
            got[key] = (cnt + int(counts[g]),
                        s_ + gpu.decode_agg_value(gpu.BG_AGG_OP_SUM_I64,
                                                  bytes(acc[g, 0])))
    want = {}
    for i in range(nord):
        key = int(o_custkey[i])
        if key > keep_hi:
            continue
        cnt, s_ = want.get(key, (0, 0))
        want[key] = (cnt + 1, s_ + int(o_total[i]))
    assert got == want


def test_f64_filter_and_sum(ctx):
    """Float64 predicates (exact compares) + grouped SUM(f64): sums within
    1e-6 relative of numpy (unordered atomic accumulation; the reference
    comparator's float tolerance, benchmarks/src/lib.rs:35)."""
    n = 500_000
    rng = np.random.default_rng(77)
    vals = rng.standard_normal(n) * 1000.0
    keys = rng.integers(0, 100, size=n, dtype=np.int64)
    vcol, _ = ctx.upload_column(vals, gpu.BG_DT_FLOAT64)
    kcol, _ = ctx.upload_column(keys, gpu.BG_DT_INT64)

    mask = ctx.eval_predicates([vcol], [(0, gpu.BG_PRED_GT, 12.5, 0.0)], n)
    _, m = ctx.mask_to_indices(mask, n)
    assert m == int((vals > 12.5).sum())

    first, acc, counts = ctx.hashagg([kcol], [vcol],
                                     [gpu.BG_AGG_OP_SUM_F64], n,
                                     max_groups=512, mask=mask)
    got = {}
    import struct as st
    for g in range(len(first)):
        key = int(keys[first[g]])
        got[key] = (int(counts[g]),
                    st.unpack("<d", bytes(acc[g, 0][:8]))[0])
    sel = vals > 12.5
    for key in np.unique(keys[sel]):
        cnt = int((sel & (keys == key)).sum())
        s_ = float(vals[sel & (keys == key)].sum())
        gc, gs = got[int(key)]
        assert gc == cnt
        assert abs(gs - s_) <= 1e-6 * max(abs(s_), 1e-30)


def test_hashagg_auto_grow_and_dec128_keys(ctx):
    """Table-full auto-retry (scheduler underestimates) + Decimal128 group
    keys."""
    n = 50_000
    rng = np.random.default_rng(19)
    kdec = rng.integers(0, 20_000, size=n, dtype=np.int64)  # ~18k groups
    dec16 = dec_bytes(kdec)
    vals = rng.integers(0, 1000, size=n, dtype=np.int64)
    kc = ctx.column(gpu.BG_DT_DECIMAL128, ctx.upload(dec16), n)
    vc, _ = ctx.upload_column(vals, gpu.BG_DT_INT64)
    # deliberately low estimate -> auto-grow path
    first, acc, counts = ctx.hashagg([kc], [vc], [gpu.BG_AGG_OP_SUM_I64], n,
                                     max_groups=32)
    got = {}
    for g in range(len(first)):
        key = int(kdec[first[g]])
        got[key] = (int(counts[g]),
                    [gpu.decode_agg_value(gpu.BG_AGG_OP_SUM_I64,
                                          bytes(acc[g, 0]))])
    want = oracle.hashagg([kdec], [("sum", vals)], n)
    want = {k_[0]: v for k_, v in want.items()}
    assert got == want


def test_filter_with_validity(ctx):
    """Predicate over a null-carrying column: null rows are excluded
    (SQL WHERE three-valued logic -> false; FilterExec semantics)."""
    n = 10_000
    rng = np.random.default_rng(8)
    vals = rng.integers(0, 100, size=n, dtype=np.int64)
    valid = rng.integers(0, 256, size=(n + 7) // 8, dtype=np.uint8)
    col, _ = ctx.upload_column(vals, gpu.BG_DT_INT64, validity=valid)
    mask = ctx.eval_predicates([col], [(0, gpu.BG_PRED_GT, 10, 0)], n)
    _, m = ctx.mask_to_indices(mask, n)
    want = oracle.filter_mask([("i64", vals, valid, 4, 10, 0)], n)
    assert m == len(oracle.mask_to_indices(want, n))


# ---------------------------------------------------------------------------
# NULL semantics through the aggregate and join paths
# ---------------------------------------------------------------------------
def _bits(bm, i):
    return (int(bm[i >> 3]) >> (i & 7)) & 1


def test_hashagg_nulls(ctx):
    """GROUP BY a null-carrying key with SUM/MIN/MAX over null-carrying
    values: NULL keys group together, NULL inputs contribute nothing, and
    an all-NULL group's aggregate is NULL (nncnt 0) — the reference's
    group_values + functions-aggregate null semantics, vs the oracle."""
    n = 100_000
    rng = np.random.default_rng(77)
    k1 = rng.integers(0, 50, size=n, dtype=np.int64)
    kv = rng.integers(0, 256, size=(n + 7) // 8, dtype=np.uint8)
    dec = rng.integers(-10**10, 10**10, size=n, dtype=np.int64)
    dec16 = dec_bytes(dec)
    decv = rng.integers(0, 256, size=(n + 7) // 8, dtype=np.uint8)
    v64 = rng.integers(-10**6, 10**6, size=n, dtype=np.int64)
    v64v = rng.integers(0, 256, size=(n + 7) // 8, dtype=np.uint8)

    c1, _ = ctx.upload_column(k1, gpu.BG_DT_INT64, validity=kv)
    ca = ctx.column(gpu.BG_DT_DECIMAL128, ctx.upload(dec16), n,
                    validity=ctx.upload(decv))
    cb, _ = ctx.upload_column(v64, gpu.BG_DT_INT64, validity=v64v)
    cc, _ = ctx.upload_column(v64, gpu.BG_DT_INT64, validity=v64v)
    first, acc, counts, nn = ctx.hashagg(
        [c1], [ca, cb, cc],
        [gpu.BG_AGG_OP_SUM_DEC128, gpu.BG_AGG_OP_MIN_I64,
         gpu.BG_AGG_OP_MAX_I64], n, max_groups=256, want_nncnt=True)

    got = {}
    for g in range(len(first)):
        r = int(first[g])
        key = int(k1[r]) if _bits(kv, r) else None
        accs = [None if nn[g, a] == 0 else
                gpu.decode_agg_value(op, bytes(acc[g, a]))
                for a, op in enumerate([gpu.BG_AGG_OP_SUM_DEC128,
                                        gpu.BG_AGG_OP_MIN_I64,
                                        gpu.BG_AGG_OP_MAX_I64])]
        assert key not in got, "NULL keys must form one group"
        got[key] = (int(counts[g]), accs, [int(x) for x in nn[g]])

    kvb = np.array([_bits(kv, i) for i in range(n)], dtype=bool)
    dvb = np.array([_bits(decv, i) for i in range(n)], dtype=bool)
    vvb = np.array([_bits(v64v, i) for i in range(n)], dtype=bool)
    want = oracle.hashagg_nulls(
        [(k1, kvb)], [("sum", dec, dvb), ("min", v64, vvb),
                      ("max", v64, vvb)], n)
    want = {k[0]: (c, a, nnw) for k, (c, a, nnw) in want.items()}
    assert got == want


def test_hashjoin_nulls(ctx):
    """Inner join with NULL keys on both sides: null_equals_null=false —
    a NULL key matches nothing (the reference HashJoinExec default)."""
    nb, np_ = 5_000, 20_000
    rng = np.random.default_rng(88)
    bk = rng.integers(0, 2_000, size=nb, dtype=np.int64)
    bv = rng.integers(0, 256, size=(nb + 7) // 8, dtype=np.uint8)
    pk = rng.integers(0, 2_000, size=np_, dtype=np.int64)
    pv = rng.integers(0, 256, size=(np_ + 7) // 8, dtype=np.uint8)

    cb, _ = ctx.upload_column(bk, gpu.BG_DT_INT64, validity=bv)
    cp, _ = ctx.upload_column(pk, gpu.BG_DT_INT64, validity=pv)
    j = gpu.GpuHashJoin(ctx, cb, nb)
    ppos, bpos, nm = j.probe(cp, np_)
    gp = ppos.download(np.uint32, nm)
    gb = bpos.download(np.uint32, nm)
    got = set(zip(gp.tolist(), gb.tolist()))

    bvb = np.array([_bits(bv, i) for i in range(nb)], dtype=bool)
    pvb = np.array([_bits(pv, i) for i in range(np_)], dtype=bool)
    want = set()
    from collections import defaultdict
    idx = defaultdict(list)
    for i in range(nb):
        if bvb[i]:
            idx[int(bk[i])].append(i)
    for i in range(np_):
        if pvb[i]:
            for b in idx.get(int(pk[i]), ()):
                want.add((i, b))
    assert got == want


def test_sort_shuffle_stage_with_nulls_gpu_codec(ctx, tmp_path):
    """gpu_codec stage over null-carrying columns INCLUDING strings: data
    and rebased utf8 offsets are device-LZ4'd, validity rides as
    per-partition LZ4-framed bitmap parts; partitions must equal
    table.take(oracle order)."""
    from datafusion_ballista_amd import engine, shuffle
    n, k = 50_000, 8
    rng = np.random.default_rng(41)
    kmask = rng.random(n) < 0.15
    vmask = rng.random(n) < 0.4
    smask = rng.random(n) < 0.25
    kvals = rng.integers(0, 4_000, size=n, dtype=np.int64)
    table = pa.table({
        "k": pa.array(kvals, mask=kmask),
        "v": pa.array(rng.integers(-10**9, 10**9, size=n, dtype=np.int64),
                      mask=vmask),
        "d": pa.array(rng.integers(0, 10**6, size=n, dtype=np.int32)),
        "s": pa.array([None if m else f"str{i % 1009}-{i % 7}"
                       for i, m in enumerate(smask)], type=pa.string()),
    })
    ex = engine.GpuQueryStageExecutor(ctx, "job-ngc", 8, str(tmp_path),
                                      key_columns=[0], num_partitions=k,
                                      gpu_codec=True)
    summaries = ex.execute_query_stage(0, table)
    assert sum(s.num_rows for s in summaries) == n
    kvalid = np.packbits(~kmask, bitorder="little")
    h = oracle.hash_columns([("i64", kvals, kvalid)], n)
    pids = oracle.partition_ids(h, k)
    idx, offs = oracle.partition_indices(pids, k)
    data_path = summaries[0].path
    for p in range(k):
        batches = shuffle.read_partition(data_path, data_path + ".index", p)
        got = pa.Table.from_batches(batches, schema=table.schema) if batches             else table.schema.empty_table()
        rows = idx[offs[p]:offs[p + 1]]
        want = table.take(pa.array(rows, type=pa.uint32()))
        assert got.equals(want), f"partition {p} mismatch"


def test_sort_shuffle_stage_with_nulls(ctx, tmp_path):
    """Full stage over null-carrying key, payload and string columns:
    NULL keys hash as no-contribution (hash_utils create_hashes), validity
    travels through the device split (bg_gather_bits) and lands in the
    partition streams; the reassembled partitions must equal
    table.take(oracle split order) including every null."""
    from datafusion_ballista_amd import engine, shuffle
    n, k = 60_000, 8
    rng = np.random.default_rng(31)
    kmask = rng.random(n) < 0.1
    vmask = rng.random(n) < 0.3
    smask = rng.random(n) < 0.2
    kvals = rng.integers(0, 5_000, size=n, dtype=np.int64)
    words = ["aa", "bbb", "", "dddd", "eee-ee"]
    table = pa.table({
        "k": pa.array(kvals, mask=kmask),
        "v": pa.array(rng.integers(-10**9, 10**9, size=n, dtype=np.int64),
                      mask=vmask),
        "s": pa.array([None if m else words[i % 5] + str(i % 97)
                       for i, m in enumerate(smask)], type=pa.string()),
    })
    ex = engine.GpuQueryStageExecutor(ctx, "job-null", 7, str(tmp_path),
                                      key_columns=[0], num_partitions=k)
    summaries = ex.execute_query_stage(0, table)
    assert sum(s.num_rows for s in summaries) == n

    kvalid = np.packbits(~kmask, bitorder="little")
    h = oracle.hash_columns([("i64", kvals, kvalid)], n)
    pids = oracle.partition_ids(h, k)
    idx, offs = oracle.partition_indices(pids, k)

    data_path = summaries[0].path
    index_path = data_path + ".index"
    for p in range(k):
        batches = shuffle.read_partition(data_path, index_path, p)
        got = pa.Table.from_batches(batches, schema=table.schema) if batches \
            else table.schema.empty_table()
        rows = idx[offs[p]:offs[p + 1]]
        want = table.take(pa.array(rows, type=pa.uint32()))
        assert got.equals(want), f"partition {p} mismatch"


def test_hashagg_f64_min_max_sum(ctx):
    """Float64 aggregates: MIN/MAX via the totally-ordered u64 transform
    (exact), SUM within the reference comparator's float tolerance
    (benchmarks/src/lib.rs:35 — atomics reorder the additions)."""
    n = 200_000
    rng = np.random.default_rng(41)
    keys = rng.integers(0, 300, size=n, dtype=np.int64)
    vals = rng.standard_normal(n) * 1e6
    vals[::97] = -vals[::97]  # plenty of negatives
    kc, _ = ctx.upload_column(keys, gpu.BG_DT_INT64)
    vc, _ = ctx.upload_column(vals, gpu.BG_DT_FLOAT64)
    first, acc, counts = ctx.hashagg(
        [kc], [vc, vc, vc],
        [gpu.BG_AGG_OP_MIN_F64, gpu.BG_AGG_OP_MAX_F64,
         gpu.BG_AGG_OP_SUM_F64], n, max_groups=1024)
    for g in range(len(first)):
        k = int(keys[first[g]])
        sel = vals[keys == k]
        gmin = gpu.decode_agg_value(gpu.BG_AGG_OP_MIN_F64, bytes(acc[g, 0]))
        gmax = gpu.decode_agg_value(gpu.BG_AGG_OP_MAX_F64, bytes(acc[g, 1]))
        gsum = gpu.decode_agg_value(gpu.BG_AGG_OP_SUM_F64, bytes(acc[g, 2]))
        assert gmin == sel.min() and gmax == sel.max(), k
        assert abs(gsum - sel.sum()) <= 1e-6 * max(1.0, abs(sel.sum())), k
    assert len(first) == len(np.unique(keys))


def test_two_phase_aggregate_via_shuffle(ctx, tmp_path):
    """The reference's AggregateExec Partial -> hash shuffle -> Final
    composition: per-split partial SUM(dec)/COUNT, written
    through the GPU sort-shuffle stage partitioned by group key, then a
    final merge aggregate per partition; the union of final groups must
    equal the oracle's one-shot global aggregate exactly (i128 sums)."""
    from datafusion_ballista_amd import engine, shuffle
    rng = np.random.default_rng(55)
    nsplits, k = 3, 4
    all_keys, all_vals = [], []
    partial_tables = []
    for s_i in range(nsplits):
        n = 40_000 + 1000 * s_i
        keys = rng.integers(0, 3_000, size=n, dtype=np.int64)
        vals = rng.integers(-10**9, 10**9, size=n, dtype=np.int64)
        all_keys.append(keys)
        all_vals.append(vals)
        kc, _ = ctx.upload_column(keys, gpu.BG_DT_INT64)
        dec16 = dec_bytes(vals)
        vc = ctx.column(gpu.BG_DT_DECIMAL128, ctx.upload(dec16), n)
        first, acc, counts = ctx.hashagg(
            [kc], [vc], [gpu.BG_AGG_OP_SUM_DEC128], n, max_groups=8192)
        g = len(first)
        pk = keys[first]
        psum = np.array([int.from_bytes(bytes(acc[i, 0]), "little",
                                        signed=True) for i in range(g)],
                        dtype=np.int64)  # fits: |vals| small enough
        partial_tables.append(pa.table({
            "k": pa.array(pk.astype(np.int64)),
            "s": pa.array([decimal.Decimal(int(v)) for v in psum],
                          type=pa.decimal128(38, 0)),
            "c": pa.array(counts.astype(np.int64)),
        }))
    ex = engine.GpuQueryStageExecutor(ctx, "job-2ph", 9, str(tmp_path),
                                      key_columns=[0], num_partitions=k)
    summaries = ex.execute_query_stage(0, partial_tables)
    data_path = summaries[0].path
    final = {}
    for p in range(k):
        batches = shuffle.read_partition(data_path, data_path + ".index", p)
        if not batches:
            continue
        t = pa.Table.from_batches(batches)
        pk = t.column("k").to_numpy()
        ps = t.column("s").combine_chunks()
        raw = np.frombuffer(ps.buffers()[1], dtype=np.uint8,
                            count=16 * len(t)).reshape(len(t), 16)
        pc = t.column("c").to_numpy()
        n2 = len(t)
        kc2, _ = ctx.upload_column(pk.astype(np.int64), gpu.BG_DT_INT64)
        sc2 = ctx.column(gpu.BG_DT_DECIMAL128,
                         ctx.upload(raw.reshape(-1).copy()), n2)
        cc2, _ = ctx.upload_column(pc.astype(np.int64), gpu.BG_DT_INT64)
        f2, a2, c2 = ctx.hashagg(
            [kc2], [sc2, cc2],
            [gpu.BG_AGG_OP_SUM_DEC128, gpu.BG_AGG_OP_SUM_I64], n2,
            max_groups=8192)
        for g in range(len(f2)):
            key = int(pk[f2[g]])
            assert key not in final, "group split across partitions"
            final[key] = (
                int.from_bytes(bytes(a2[g, 0]), "little", signed=True),
                gpu.decode_agg_value(gpu.BG_AGG_OP_SUM_I64,
                                     bytes(a2[g, 1])))
    keys_all = np.concatenate(all_keys)
    vals_all = np.concatenate(all_vals)
    want = oracle.hashagg([keys_all], [("sum", vals_all)], len(keys_all))
    want = {kk[0]: (v[1][0], v[0]) for kk, v in want.items()}
    assert final == want


@pytest.mark.parametrize("desc", [False, True])
def test_sort_utf8_keys(ctx, desc):
    """ORDER BY a Utf8 column: exact memcmp-then-length order (the
    reference's Utf8 comparator), stable, incl. embedded NULs, empty
    strings, shared prefixes and length ties vs python's stable sort."""
    n = 40_000
    rng = np.random.default_rng(91)
    pool = [b"", b"a", b"a\x00", b"a\x00b", b"ab", b"abc", b"abcdefgh",
            b"abcdefghi", b"abcdefgh\x00", b"zz", b"z" * 23]
    strs = [pool[i % len(pool)] + (str(rng.integers(0, 50)).encode()
                                   if i % 3 else b"") for i in range(n)]
    col, keep = None, []
    data = b"".join(strs)
    offs = np.zeros(n + 1, dtype=np.int32)
    for i, b in enumerate(strs):
        offs[i + 1] = offs[i] + len(b)
    dbuf = ctx.upload(np.frombuffer(data, dtype=np.uint8)
                      if data else np.zeros(1, np.uint8))
    obuf = ctx.upload(offs)
    col = ctx.column(gpu.BG_DT_UTF8, dbuf, n, offsets=obuf)
    perm = ctx.sort_rows([col], [desc], n).download(np.uint32, n)
    if desc:
        # stable descending: invert every byte and use inverted length as
        # the tiebreak (matches the engine's ~key transform exactly)
        want = sorted(range(n),
                      key=lambda i: (bytes(255 - b for b in strs[i]) +
                                     b"\xff" * (64 - len(strs[i])),
                                     -len(strs[i])))
    else:
        want = sorted(range(n), key=lambda i: strs[i])
    assert list(perm) == want


def test_sort_null_ordering(ctx):
    """ORDER BY with NULL keys: SQL default (ASC -> NULLS LAST, DESC ->
    NULLS FIRST) via bg_sort_rows, explicit override via bg_sort_rows2;
    nulls form one stable group (original order preserved)."""
    import ctypes
    n = 30_000
    rng = np.random.default_rng(93)
    vals = rng.integers(0, 1000, size=n, dtype=np.int64)
    mask = rng.random(n) < 0.2
    vbits = np.packbits(~mask, bitorder="little")
    col, _ = ctx.upload_column(vals, gpu.BG_DT_INT64, validity=vbits)
    null_rows = [i for i in range(n) if mask[i]]
    valid_sorted = sorted((i for i in range(n) if not mask[i]),
                          key=lambda i: (vals[i], i))

    perm = ctx.sort_rows([col], [False], n).download(np.uint32, n)
    assert list(perm) == valid_sorted + null_rows  # ASC: NULLS LAST

    perm = ctx.sort_rows([col], [True], n).download(np.uint32, n)
    desc_sorted = sorted((i for i in range(n) if not mask[i]),
                         key=lambda i: (-vals[i], i))
    assert list(perm) == null_rows + desc_sorted   # DESC: NULLS FIRST

    # explicit override: ASC + NULLS FIRST
    pbuf = ctx.alloc(4 * n)
    karr = (gpu.BgColumn * 1)(col)
    darr = (ctypes.c_int32 * 1)(0)
    nfarr = (ctypes.c_int32 * 1)(1)
    gpu._check(ctx.L.bg_sort_rows2(karr, darr, nfarr, 1,
                                   ctypes.c_int64(n), pbuf.ptr), "sort2")
    ctx.synchronize()
    perm = pbuf.download(np.uint32, n)
    assert list(perm) == null_rows + valid_sorted
