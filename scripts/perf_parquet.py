#!/usr/bin/env python3
"""Parquet GPU-decode throughput: lineitem-shaped file -> device columns.
Reports decode GB/s (decoded output bytes / wall) per codec/encoding."""
import json
import os
import sys
import time
import decimal

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

import numpy as np  # noqa: E402
import pyarrow as pa  # noqa: E402
import pyarrow.parquet as pq  # noqa: E402

from datafusion_ballista_amd import gpu  # noqa: E402
from datafusion_ballista_amd.parquet import GpuParquetColumnReader  # noqa: E402


def main():
    n = 20_000_000
    rng = np.random.default_rng(5)
    print(f"writing {n}-row file...", flush=True)
    dec_pool = np.array([decimal.Decimal(int(v)) / 100 for v in
                         rng.integers(90000, 10495100, size=100_000)])
    table = pa.table({
        "l_orderkey": pa.array(rng.integers(1, n, size=n, dtype=np.int64)),
        "l_shipdate": pa.array(rng.integers(8000, 11000, size=n,
                                            dtype=np.int32)),
        "l_extendedprice": pa.array(dec_pool[rng.integers(
            0, len(dec_pool), size=n)], type=pa.decimal128(15, 2)),
    })
    results = []
    for codec, use_dict, tag in [("snappy", True, "snappy+dict"),
                                 ("snappy", False, "snappy+plain"),
                                 ("none", False, "plain")]:
        path = f"/tmp/li_{tag}.parquet"
        pq.write_table(table, path, compression=codec,
                       use_dictionary=use_dict, write_statistics=False)
        fsize = os.path.getsize(path)
        ctx = gpu.GpuStageContext(0)
        rd = GpuParquetColumnReader(ctx, path)
        t0 = time.perf_counter()
        out_bytes = 0
        for col, esz in [(0, 8), (1, 4), (2, 16)]:
            buf, nv, _, _ = rd.read_column_all(col)
            out_bytes += nv * esz
        dt = time.perf_counter() - t0
        rec = {"case": tag, "file_mb": fsize / 1e6, "rows": n,
               "decoded_gb": out_bytes / 1e9, "s": dt,
               "decoded_gbps": out_bytes / dt / 1e9,
               "rows_per_s": n / dt}
        results.append(rec)
        print(json.dumps(rec), flush=True)
        ctx.close()
    with open(os.path.join(ROOT, "gpurun_out", "perf_parquet.json"), "w") as f:
        json.dump(results, f, indent=1)


if __name__ == "__main__":
    main()
