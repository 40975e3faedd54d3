"""Multi-process CPU coverage of the N>1 path (gloo, world_size=2).

bench.py's distributed pieces — the exact Decimal128 limb all-reduce that
merges per-rank q6 partial aggregates, and the MAX-over-ranks elapsed-time
reduction — must be correct by construction before the driver ever runs the
8-GPU scaling bench (RCCL path, same torch.distributed calls)."""

import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

from bench import i128_to_limbs, limbs_to_i128


def test_limb_codec_roundtrip():
    rng = np.random.default_rng(0)
    vals = [0, 1, -1, 2**64, -(2**64), 2**126, -(2**126) + 5]
    vals += [int(rng.integers(-2**62, 2**62)) * int(rng.integers(1, 2**40))
             for _ in range(50)]
    for v in vals:
        assert limbs_to_i128(i128_to_limbs(v)) == v


def _worker(rank, world, port, results):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.distributed.init_process_group("gloo", rank=rank, world_size=world)
    try:
        # each rank holds an exact i128 partial (as a q6 shard would)
        partials = [123456789012345678901234567890 * 3,
                    -987654321098765432109876543210]
        cnt = [17, 25]
        t = torch.tensor(i128_to_limbs(partials[rank]) + [cnt[rank]],
                         dtype=torch.int64)
        torch.distributed.all_reduce(t)
        merged = t.tolist()
        total = limbs_to_i128(merged[:4])
        count = merged[4]

        # MAX-over-ranks elapsed
        elapsed = [1.5, 2.5][rank]
        e = torch.tensor([elapsed], dtype=torch.float64)
        torch.distributed.all_reduce(e, op=torch.distributed.ReduceOp.MAX)

        results[rank] = (total, count, float(e.item()))
    finally:
        torch.distributed.destroy_process_group()


def test_exact_i128_allreduce_world2():
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        port = 29631
        ps = [ctx.Process(target=_worker, args=(r, 2, port, results))
              for r in range(2)]
        for p in ps:
            p.start()
        for p in ps:
            p.join(120)
            assert p.exitcode == 0
        want_total = (123456789012345678901234567890 * 3
                      - 987654321098765432109876543210)
        for r in range(2):
            total, count, emax = results[r]
            assert total == want_total
            assert count == 42
            assert emax == pytest.approx(2.5)


def _exchange_worker(rank, world, port, results):
    import numpy as np
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.distributed.init_process_group("gloo", rank=rank, world_size=world)
    try:
        import oracle
        from datafusion_ballista_amd import exchange
        k_local = 4
        k = k_local * world
        n = 10_000 + rank * 777
        rng = np.random.default_rng(100 + rank)
        keys = rng.integers(0, 5000, size=n, dtype=np.int64)
        h = oracle.hash_columns([("i64", keys)], n)
        pids = oracle.partition_ids(h, k)
        idx, offs = oracle.partition_indices(pids, k)
        part_major = keys[idx]  # partition-major materialisation (oracle)

        t = torch.from_numpy(part_major.copy())
        out, out_splits = exchange.all_to_all_rows(t, offs, world)
        got = out.numpy()
        # every received key must hash to a partition owned by this rank
        gh = oracle.hash_columns([("i64", got)], len(got))
        gp = oracle.partition_ids(gh, k)
        owners = gp // k_local
        assert (owners == rank).all(), "received a row owned by another rank"
        results[rank] = (int(len(got)), int(got.astype(np.int64).sum()))
    finally:
        torch.distributed.destroy_process_group()


def test_all_to_all_exchange_world2():
    """RCCL-exchange logic on gloo (world 2): row conservation and correct
    ownership after the all-to-all (exchange.py)."""
    import numpy as np
    import oracle
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        port = 29637
        ps = [ctx.Process(target=_exchange_worker, args=(r, 2, port, results))
              for r in range(2)]
        for p in ps:
            p.start()
        for p in ps:
            p.join(180)
            assert p.exitcode == 0
        # global row + checksum conservation
        total_rows = sum(results[r][0] for r in range(2))
        total_sum = sum(results[r][1] for r in range(2))
        want_rows, want_sum = 0, 0
        for rank in range(2):
            n = 10_000 + rank * 777
            rng = np.random.default_rng(100 + rank)
            keys = rng.integers(0, 5000, size=n, dtype=np.int64)
            want_rows += n
            want_sum += int(keys.sum())
        assert total_rows == want_rows
        assert total_sum == want_sum


def _broadcast_worker(rank, world, port, results):
    import numpy as np
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.distributed.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from datafusion_ballista_amd import exchange
        if rank == 0:
            build = torch.arange(1000, dtype=torch.int64) * 7
        else:
            build = torch.empty(0, dtype=torch.int64)
        got = exchange.broadcast_build_side(build, src_rank=0)
        results[rank] = (int(got.shape[0]), int(got.sum().item()))
    finally:
        torch.distributed.destroy_process_group()


def test_broadcast_build_side_world2():
    """Broadcast-join build replication (RCCL broadcast stand-in on gloo)."""
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        ps = [ctx.Process(target=_broadcast_worker, args=(r, 2, 29641, results))
              for r in range(2)]
        for p in ps:
            p.start()
        for p in ps:
            p.join(120)
            assert p.exitcode == 0
        want = (1000, sum(i * 7 for i in range(1000)))
        assert results[0] == want and results[1] == want


def _twophase_worker(rank, world, port, results):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")
    torch.distributed.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from datafusion_ballista_amd import exchange
        import oracle

        # each rank holds a shard; partial-aggregate it (oracle restatement
        # stands in for the GPU hashagg — the exchange LOGIC is under test)
        rng = np.random.default_rng(100 + rank)
        n = 5_000
        keys = rng.integers(0, 200, size=n, dtype=np.int64)
        vals = rng.integers(-10**6, 10**6, size=n, dtype=np.int64)
        partial = oracle.hashagg([keys], [("sum", vals)], n)
        pk = np.array([k[0] for k in partial.keys()], dtype=np.int64)
        pc = np.array([v[0] for v in partial.values()], dtype=np.int64)
        ps = np.array([v[1][0] for v in partial.values()], dtype=np.int64)

        # route each partial row to the rank owning hash(key) % world
        # (the reference's Partial -> RepartitionExec(Hash) -> Final shape)
        h = oracle.hash_columns([("i64", pk)], len(pk))
        owner = (h % world).astype(np.int64)
        order = np.argsort(owner, kind="stable")
        t = torch.tensor(
            np.stack([pk[order], pc[order], ps[order]],
                     axis=1)).reshape(-1)  # 3 units per row (flat exchange)
        offs = np.zeros(world + 1, dtype=np.int64)
        for o in owner:
            offs[o + 1] += 1
        offs = np.cumsum(offs) * 3
        out, _ = exchange.all_to_all_rows(t, offs, world)

        # final merge on the owning rank
        rows = out.numpy().reshape(-1, 3)
        final = {}
        for k, c, s_ in rows:
            e = final.setdefault(int(k), [0, 0])
            e[0] += int(c)
            e[1] += int(s_)
        results[rank] = final
    finally:
        torch.distributed.destroy_process_group()


def test_two_phase_aggregate_exchange_world2():
    """Distributed Partial->exchange->Final aggregate (2 ranks, gloo): the
    union of per-rank final groups must equal the oracle's global
    aggregate over both shards, with every key on exactly one rank."""
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        port = 29637
        ps = [ctx.Process(target=_twophase_worker, args=(r, 2, port, results))
              for r in range(2)]
        for p in ps:
            p.start()
        for p in ps:
            p.join(120)
            assert p.exitcode == 0
        f0, f1 = results[0], results[1]
    assert not (set(f0) & set(f1)), "a group landed on both ranks"
    merged = {**f0, **f1}

    import oracle
    all_k, all_v = [], []
    for r in range(2):
        rng = np.random.default_rng(100 + r)
        all_k.append(rng.integers(0, 200, size=5_000, dtype=np.int64))
        all_v.append(rng.integers(-10**6, 10**6, size=5_000, dtype=np.int64))
    gk = np.concatenate(all_k)
    gv = np.concatenate(all_v)
    want = oracle.hashagg([gk], [("sum", gv)], len(gk))
    want = {k[0]: [v[0], v[1][0]] for k, v in want.items()}
    assert merged == want
