#!/usr/bin/env python3
"""Randomised parity fuzz: random shapes/dtypes/selectivities through every
hot-path op, each checked bit-exactly against the oracle.  Run on the GPU
box; any mismatch prints the failing seed and aborts."""
import os
import sys
import time

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

import numpy as np  # noqa: E402

import oracle  # noqa: E402
from datafusion_ballista_amd import gpu  # noqa: E402


def fuzz_round(ctx, seed):
    rng = np.random.default_rng(seed)
    n = int(rng.integers(1, 2_000_00))
    k = int(rng.choice([1, 2, 3, 5, 16, 64, 100, 333, 1024]))

    keys = rng.integers(-2**62, 2**62, size=n, dtype=np.int64)
    col, kbuf = ctx.upload_column(keys, gpu.BG_DT_INT64)

    # hash + split
    hbuf = ctx.hash_columns([col], n)
    pbuf = ctx.partition_ids(hbuf, n, k)
    ibuf, obuf = ctx.partition_indices(pbuf, n, k)
    ctx.synchronize()
    h = hbuf.download(np.uint64, n)
    want_h = oracle.hash_columns([("i64", keys)], n)
    assert np.array_equal(h, want_h), f"hash mismatch seed={seed}"
    idx = ibuf.download(np.uint32, n)
    offs = obuf.download(np.int64, k + 1)
    want_p = oracle.partition_ids(want_h, k)
    want_idx, want_offs = oracle.partition_indices(want_p, k)
    assert np.array_equal(offs, want_offs), f"offs mismatch seed={seed}"
    assert np.array_equal(idx, want_idx), f"split mismatch seed={seed}"

    # filter + compact at a random selectivity
    lo = int(rng.integers(-2**62, 2**62))
    hi = int(rng.integers(lo, 2**62))
    mask = ctx.eval_predicates([col], [(0, gpu.BG_PRED_GE_LT, lo, hi)], n)
    fidx, m = ctx.mask_to_indices(mask, n)
    want_mask = oracle.filter_mask([("i64", keys, None, 0, lo, hi)], n)
    want_fidx = oracle.mask_to_indices(want_mask, n)
    assert m == len(want_fidx), f"filter count mismatch seed={seed}"
    if m:
        assert np.array_equal(fidx.download(np.uint32, m), want_fidx), \
            f"filter idx mismatch seed={seed}"

    # join on a narrowed key space
    kb = int(rng.integers(1, 50_000))
    nb = int(rng.integers(1, 50_000))
    build = rng.integers(0, kb, size=nb, dtype=np.int64)
    probe = keys % kb
    bcol, _ = ctx.upload_column(build, gpu.BG_DT_INT64)
    pcol, _ = ctx.upload_column(probe, gpu.BG_DT_INT64)
    join = gpu.GpuHashJoin(ctx, bcol, nb)
    jp, jb, jm = join.probe(pcol, n)
    wp, wb = oracle.hashjoin_pairs(build, probe)
    assert jm == len(wp), f"join count mismatch seed={seed}"
    if jm:
        got = np.sort(jp.download(np.uint32, jm).astype(np.uint64) << np.uint64(32)
                      | jb.download(np.uint32, jm))
        want = np.sort(wp.astype(np.uint64) << np.uint64(32) | wb)
        assert np.array_equal(got, want), f"join pairs mismatch seed={seed}"
    join.free()

    # group-by SUM on the narrowed keys
    vals = rng.integers(-10**9, 10**9, size=n, dtype=np.int64)
    vcol, _ = ctx.upload_column(vals, gpu.BG_DT_INT64)
    gcol, _ = ctx.upload_column(probe, gpu.BG_DT_INT64)
    first, acc, counts = ctx.hashagg([gcol], [vcol], [gpu.BG_AGG_OP_SUM_I64],
                                     n, max_groups=max(2 * kb, 64))
    got_groups = {}
    for g in range(len(first)):
        got_groups[int(probe[first[g]])] = (
            int(counts[g]),
            [gpu.decode_agg_value(gpu.BG_AGG_OP_SUM_I64, bytes(acc[g, 0]))])
    want_groups = oracle.hashagg([probe], [("sum", vals)], n)
    want_groups = {kk[0]: v for kk, v in want_groups.items()}
    assert got_groups == want_groups, f"hashagg mismatch seed={seed}"

    # stable sort (asc/desc) vs numpy stable argsort — both full-range
    # keys and a narrow range that exercises constant-digit pass skipping
    desc = bool(rng.integers(0, 2))
    perm = ctx.sort_rows([col], [desc], n).download(np.uint32, n)
    u = keys.view(np.uint64) ^ np.uint64(1 << 63)
    want_perm = np.argsort(~u if desc else u, kind="stable")
    assert np.array_equal(perm.astype(np.int64), want_perm), \
        f"sort mismatch seed={seed} desc={desc}"
    ncol, _ = ctx.upload_column(probe, gpu.BG_DT_INT64)  # narrow range
    perm2 = ctx.sort_rows([ncol], [desc], n).download(np.uint32, n)
    u2 = probe.view(np.uint64) ^ np.uint64(1 << 63)
    want2 = np.argsort(~u2 if desc else u2, kind="stable")
    assert np.array_equal(perm2.astype(np.int64), want2), \
        f"narrow sort mismatch seed={seed} desc={desc}"

    # NULL semantics: hash with validity; join with null keys both sides
    vbits = rng.integers(0, 256, size=(n + 7) // 8, dtype=np.uint8)
    colv, _ = ctx.upload_column(keys, gpu.BG_DT_INT64, validity=vbits)
    hv = ctx.hash_columns([colv], n)
    ctx.synchronize()
    want_hv = oracle.hash_columns([("i64", keys, vbits)], n)
    assert np.array_equal(hv.download(np.uint64, n), want_hv), \
        f"null hash mismatch seed={seed}"
    bbits = rng.integers(0, 256, size=(nb + 7) // 8, dtype=np.uint8)
    bcolv, _ = ctx.upload_column(build, gpu.BG_DT_INT64, validity=bbits)
    pcolv, _ = ctx.upload_column(probe, gpu.BG_DT_INT64, validity=vbits)
    jn = gpu.GpuHashJoin(ctx, bcolv, nb)
    jp2, jb2, jm2 = jn.probe(pcolv, n)
    bmask = np.unpackbits(bbits, bitorder="little")[:nb].astype(bool)
    pmask = np.unpackbits(vbits, bitorder="little")[:n].astype(bool)
    bi = np.where(bmask)[0]
    pi = np.where(pmask)[0]
    cwp, cwb = oracle.hashjoin_pairs(build[bi], probe[pi])
    wp2 = pi[cwp.astype(np.int64)].astype(np.uint64)
    wb2 = bi[cwb.astype(np.int64)].astype(np.uint64)
    assert jm2 == len(wp2), f"null join count mismatch seed={seed}"
    if jm2:
        got2 = np.sort(jp2.download(np.uint32, jm2).astype(np.uint64)
                       << np.uint64(32) | jb2.download(np.uint32, jm2))
        want2j = np.sort(wp2 << np.uint64(32) | wb2)
        assert np.array_equal(got2, want2j), \
            f"null join pairs mismatch seed={seed}"
    jn.free()

    # sort with NULL keys: SQL default null ordering + stable null group
    perm3 = ctx.sort_rows([colv], [desc], n).download(np.uint32, n)
    vb = np.unpackbits(vbits, bitorder="little")[:n].astype(bool)
    nulls = [i for i in range(n) if not vb[i]]
    u3 = keys.view(np.uint64) ^ np.uint64(1 << 63)
    if desc:
        order = sorted((i for i in range(n) if vb[i]),
                       key=lambda i: (~u3[i] & 0xFFFFFFFFFFFFFFFF, i))
        want3 = nulls + order
    else:
        order = sorted((i for i in range(n) if vb[i]),
                       key=lambda i: (u3[i], i))
        want3 = order + nulls
    assert np.array_equal(perm3.astype(np.int64), np.array(want3)), \
        f"null sort mismatch seed={seed} desc={desc}"

    # utf8 sort on derived strings (shared prefixes + length ties)
    if n <= 60_000:
        strs = [b"p%d/%d" % (int(k2) % 97, int(k2) % 13) for k2 in probe]
        data = b"".join(strs)
        offs = np.zeros(n + 1, dtype=np.int32)
        for i2, b2 in enumerate(strs):
            offs[i2 + 1] = offs[i2] + len(b2)
        db = ctx.upload(np.frombuffer(data, dtype=np.uint8)
                        if data else np.zeros(1, np.uint8))
        ob = ctx.upload(offs)
        ucol = ctx.column(gpu.BG_DT_UTF8, db, n, offsets=ob)
        permu = ctx.sort_rows([ucol], [False], n).download(np.uint32, n)
        wantu = sorted(range(n), key=lambda i2: strs[i2])
        assert list(permu) == wantu, f"utf8 sort mismatch seed={seed}"

    # null-aware group-by with non-null counts (bg_hashagg2) on small n
    if n <= 60_000:
        gbits = rng.integers(0, 256, size=(n + 7) // 8, dtype=np.uint8)
        gv, _ = ctx.upload_column(probe, gpu.BG_DT_INT64, validity=gbits)
        vv, _ = ctx.upload_column(vals, gpu.BG_DT_INT64, validity=vbits)
        f2, a2, c2, nn2 = ctx.hashagg(
            [gv], [vv], [gpu.BG_AGG_OP_SUM_I64], n,
            max_groups=max(2 * kb, 64), want_nncnt=True)
        gmask = np.unpackbits(gbits, bitorder="little")[:n].astype(bool)
        got_n = {}
        for g in range(len(f2)):
            r = int(f2[g])
            kk = int(probe[r]) if gmask[r] else None
            acc_v = None if nn2[g, 0] == 0 else gpu.decode_agg_value(
                gpu.BG_AGG_OP_SUM_I64, bytes(a2[g, 0]))
            got_n[kk] = (int(c2[g]), [acc_v], [int(nn2[g, 0])])
        want_n = oracle.hashagg_nulls([(probe, gmask)],
                                      [("sum", vals, pmask)], n)
        want_n = {kk[0]: v for kk, v in want_n.items()}
        assert got_n == want_n, f"null hashagg mismatch seed={seed}"
    return n, k


def main():
    import torch
    assert torch.cuda.is_available()
    ctx = gpu.GpuStageContext(0)
    t0 = time.perf_counter()
    rounds = int(sys.argv[1]) if len(sys.argv) > 1 else 50
    base = int(sys.argv[2]) if len(sys.argv) > 2 else 10_000
    for i in range(rounds):
        n, k = fuzz_round(ctx, base + i)
        if i % 10 == 0:
            print(f"round {i}: ok (n={n}, k={k}, "
                  f"{time.perf_counter()-t0:.1f}s)", flush=True)
        ctx.close()
        ctx = gpu.GpuStageContext(0)
    print(f"FUZZ OK: {rounds} rounds in {time.perf_counter()-t0:.1f}s")


if __name__ == "__main__":
    main()
