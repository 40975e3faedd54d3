#!/usr/bin/env python3
"""Randomised GPU parquet-decode fuzz: random schemas, encodings, page
sizes, codecs, versions and null densities, every column checked exactly
against pyarrow's own reader.  Usage: fuzz_parquet.py [rounds] [seed]."""
import decimal
import os
import sys
import tempfile
import time

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

import numpy as np  # noqa: E402
import pyarrow as pa  # noqa: E402
import pyarrow.parquet as pq  # noqa: E402

from datafusion_ballista_amd import gpu  # noqa: E402
from datafusion_ballista_amd.parquet import GpuParquetColumnReader  # noqa: E402


def one_round(ctx, seed, tmpdir):
    rng = np.random.default_rng(seed)
    n = int(rng.integers(1, 120_000))
    version = rng.choice(["1.0", "2.0"])
    compression = rng.choice(["snappy", "none"])
    page = int(rng.choice([4096, 16384, 65536]))
    cols, enc, checkers = {}, {}, []
    ncols = int(rng.integers(1, 5))
    for c in range(ncols):
        name = f"c{c}"
        kinds = ["i64", "i32", "f64", "dec", "str"]
        if version == "1.0" and compression in ("snappy", "none"):
            kinds.append("list")  # LIST decode is V1-only
        kind = rng.choice(kinds)
        nullp = float(rng.choice([0.0, 0.05, 0.4]))
        mask = rng.random(n) < nullp
        if kind == "list":
            rows = []
            for i in range(n):
                if mask[i]:
                    rows.append(None)
                elif rng.random() < 0.12:
                    rows.append([])
                else:
                    rows.append([None if rng.random() < 0.2
                                 else int(v) for v in
                                 rng.integers(-2**50, 2**50,
                                              size=int(rng.integers(1, 7)))])
            cols[name] = pa.array(rows, type=pa.list_(pa.int64()))
            checkers.append((name, kind, mask))
            continue
        if kind == "i64":
            v = rng.integers(-2**62, 2**62, n, dtype=np.int64)
            arr = pa.array(v, mask=mask)
            e = rng.choice([None, "PLAIN", "DELTA_BINARY_PACKED"])
        elif kind == "i32":
            v = rng.integers(-2**30, 2**30, n, dtype=np.int32)
            arr = pa.array(v, mask=mask)
            e = rng.choice([None, "PLAIN", "DELTA_BINARY_PACKED"])
        elif kind == "f64":
            v = rng.standard_normal(n) * 10**rng.integers(0, 9)
            arr = pa.array(v, mask=mask)
            e = rng.choice([None, "PLAIN", "BYTE_STREAM_SPLIT"])
        elif kind == "dec":
            v = rng.integers(-10**12, 10**12, n)
            arr = pa.array([None if m else decimal.Decimal(int(x)) / 100
                            for x, m in zip(v, mask)],
                           type=pa.decimal128(15, 2))
            e = rng.choice([None, "PLAIN"])
        else:
            base = [f"p{i % 13}/q{i % 101}/v{i}" if i % 3 else f"s{i % 7}"
                    for i in range(n)]
            arr = pa.array([None if m else b for b, m in zip(base, mask)],
                           type=pa.string())
            e = rng.choice([None, "PLAIN", "DELTA_LENGTH_BYTE_ARRAY",
                            "DELTA_BYTE_ARRAY"])
        cols[name] = arr
        if e is not None:
            enc[name] = e
        checkers.append((name, kind, mask))
    table = pa.table(cols)
    path = os.path.join(tmpdir, f"f{seed}.parquet")
    kw = dict(compression=compression, data_page_version=version,
              data_page_size=page, write_statistics=False)
    if enc:
        kw["use_dictionary"] = False
        kw["column_encoding"] = enc
    else:
        kw["use_dictionary"] = bool(rng.integers(0, 2))
    pq.write_table(table, path, **kw)
    rd = GpuParquetColumnReader(ctx, path)
    ref = pq.read_table(path)
    for ci, (name, kind, mask) in enumerate(checkers):
        if kind == "list":
            res = rd.read_list_column_all(ci)
            col = ref.column(name).combine_chunks()
            assert res["n_rows"] == n, (seed, name)
            w_offs = col.offsets.to_numpy(zero_copy_only=False)
            assert np.array_equal(res["offsets"],
                                  w_offs.astype(np.int32)), (seed, name)
            lv = res["list_valid"]
            if lv is not None:
                w = lv.download(np.uint32, (n + 31) // 32)
                gv = np.unpackbits(w.view(np.uint8),
                                   bitorder="little")[:n].astype(bool)
                assert np.array_equal(gv, ~np.array(col.is_null())), \
                    (seed, name, "list validity")
            ne = res["n_entries"]
            w_elems = col.values
            assert ne == len(w_elems), (seed, name)
            ew = res["elem_valid"].download(np.uint32, (ne + 31) // 32)
            gev = np.unpackbits(ew.view(np.uint8),
                                bitorder="little")[:ne].astype(bool)
            w_ev = ~np.array(w_elems.is_null())
            assert np.array_equal(gev, w_ev), (seed, name, "elem validity")
            got = res["values"].download(np.int64, max(ne, 1))[:ne]
            want = np.asarray(
                w_elems.to_numpy(zero_copy_only=False))
            assert np.array_equal(got[w_ev], want[w_ev].astype(np.int64)), \
                (seed, name, "values")
            continue
        out, nv, phys, valid = rd.read_column_all(ci)
        assert nv == n, (seed, name)
        col = ref.column(name).combine_chunks()
        if mask.any():
            vw = valid.download(np.uint32, (n + 31) // 32)
            gv = np.unpackbits(vw.view(np.uint8),
                               bitorder="little")[:n].astype(bool)
            assert np.array_equal(gv, ~mask), (seed, name, "validity")
        else:
            gv = np.ones(n, bool)
        if kind == "str":
            offs_buf, data_buf, total = out
            offs = offs_buf.download(np.int32, n + 1)
            data = data_buf.download(np.uint8, max(total, 1))
            want = col.drop_null()
            got = [bytes(data[offs[i]:offs[i + 1]]) for i in range(n)
                   if gv[i]]
            assert got == [w.as_py().encode() for w in want], (seed, name)
        elif kind == "dec":
            raw = out.download(np.uint8, 16 * n).reshape(n, 16)
            want = [int(w.as_py() * 100) for w in col.drop_null()]
            got = [int.from_bytes(bytes(raw[i]), "little", signed=True)
                   for i in range(n) if gv[i]]
            assert got == want, (seed, name)
        else:
            npdt = {"i64": np.int64, "i32": np.int32,
                    "f64": np.float64}[kind]
            got = out.download(npdt, n)
            want = col.drop_null().to_numpy(zero_copy_only=False)
            if kind == "f64":
                assert np.array_equal(got[gv].view(np.uint64),
                                      want.view(np.uint64)), (seed, name)
            else:
                assert np.array_equal(got[gv], want.astype(npdt)), \
                    (seed, name)
    os.unlink(path)
    return n, ncols


def main():
    rounds = int(sys.argv[1]) if len(sys.argv) > 1 else 30
    seed0 = int(sys.argv[2]) if len(sys.argv) > 2 else 0
    ctx = gpu.GpuStageContext(0)
    t0 = time.time()
    with tempfile.TemporaryDirectory() as td:
        for r in range(rounds):
            n, k = one_round(ctx, seed0 + r, td)
            if r % 10 == 9:
                print(f"round {r + 1}: ok (n={n} ncols={k}, "
                      f"{time.time() - t0:.1f}s)", flush=True)
    print(f"PARQUET FUZZ OK: {rounds} rounds in {time.time() - t0:.1f}s",
          flush=True)


if __name__ == "__main__":
    main()
