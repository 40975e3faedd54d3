#!/usr/bin/env python3
"""End-to-end sort-shuffle STAGE timing (device compute vs host IPC+file
write) — quantifies the host-encode share that a GPU LZ4/IPC codec
 would remove."""
import json
import os
import sys
import time

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

import numpy as np  # noqa: E402
import pyarrow as pa  # noqa: E402

from datafusion_ballista_amd import engine, gpu  # noqa: E402


def main():
    n, k = 20_000_000, 16
    rng = np.random.default_rng(3)
    table = pa.table({
        "k": pa.array(rng.integers(0, 10_000_000, size=n, dtype=np.int64)),
        "d": pa.array(rng.integers(8000, 11000, size=n, dtype=np.int32),
                      type=pa.int32()),
        "v": pa.array(rng.integers(-10**9, 10**9, size=n, dtype=np.int64)),
    })
    ctx = gpu.GpuStageContext(0)
    recs = []
    for codec in (False, True):
        work = f"/tmp/stage_perf_{codec}"
        ex = engine.GpuQueryStageExecutor(ctx, "job-perf", 1, work,
                                          key_columns=[0], num_partitions=k,
                                          gpu_codec=codec)
        ex.execute_query_stage(0, table)  # warmup (pool, imports)
        t0 = time.perf_counter()
        summaries = ex.execute_query_stage(1, table)
        wall = time.perf_counter() - t0
        m = ex.collect_plan_metrics()[0]
        rec = {"gpu_codec": codec, "rows": n, "k": k, "wall_s": wall,
               "repart_device_ms": m["repart_time_ns"] / 1e6,
               "write_host_ms": m["write_time_ns"] / 1e6,
               "rows_per_s": n / wall,
               "file_mb": os.path.getsize(summaries[0].path) / 1e6}
        recs.append(rec)
        print(json.dumps(rec), flush=True)
    with open(os.path.join(ROOT, "gpurun_out", "perf_stage.json"), "w") as f:
        json.dump(recs, f, indent=1)


if __name__ == "__main__":
    main()
