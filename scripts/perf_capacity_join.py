#!/usr/bin/env python3
"""SF300-class capacity: a 1.8B-row q3-class join + grouped aggregate
through the stage interpreter with probe_chunk_rows bounding the join
temporaries ("1.8B-row q3-class join completing with a
capped HBM budget").  The probe side is processed in 256M-row slices, so
pair buffers + gathered outputs stay ~10 GB regardless of probe size —
the join-temporary analogue of the reference's spill path
(sort_shuffle/writer.rs:650-686).  Invariants cross-checked with torch."""
import ctypes
import json
import os
import sys
import time

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

import torch  # noqa: E402

from datafusion_ballista_amd import gpu, stage  # noqa: E402

NPROBE = 1_800_000_000
NBUILD = 50_000_000
CHUNK = 256 * 1024 * 1024


def main():
    assert torch.cuda.is_available()
    dev = torch.device("cuda:0")
    gpu.GpuStageContext(0)
    g = torch.Generator(device=dev)
    g.manual_seed(7)

    print(f"generating probe {NPROBE} rows + build {NBUILD} in HBM...",
          flush=True)
    keys = torch.randint(0, 2 * NBUILD, (NPROBE,), generator=g, device=dev,
                         dtype=torch.int64)
    val = torch.zeros((NPROBE, 2), dtype=torch.int64, device=dev)
    val[:, 0] = torch.randint(0, 10**6, (NPROBE,), generator=g, device=dev)
    bkey = torch.arange(0, 2 * NBUILD, 2, device=dev, dtype=torch.int64)
    bval = torch.randint(0, 100, (NBUILD,), generator=g, device=dev,
                         dtype=torch.int64)
    torch.cuda.synchronize()
    free0, total0 = torch.cuda.mem_get_info()
    print(f"  resident; HBM free {free0/2**30:.0f} GiB of "
          f"{total0/2**30:.0f}", flush=True)

    def reg(name, cols):
        L = gpu.load_library()
        arr = (gpu.BgColumn * len(cols))()
        names = (ctypes.c_char_p * len(cols))()
        n = cols[0][2].shape[0]
        for i, (cn, dt, t, p, s) in enumerate(cols):
            arr[i] = gpu.BgColumn(dt, p, s, 0,
                                  ctypes.c_void_p(t.data_ptr()), None, None,
                                  n)
            names[i] = cn.encode()
        gpu._check(L.bg_stage_register_table(name.encode(), arr, names,
                                             len(cols), ctypes.c_int64(n)),
                   "register")

    reg("probe", [("k", gpu.BG_DT_INT64, keys, 0, 0),
                  ("v", gpu.BG_DT_DECIMAL128, val, 15, 2)])
    reg("build", [("bk", gpu.BG_DT_INT64, bkey, 0, 0),
                  ("bv", gpu.BG_DT_INT64, bval, 0, 0)])

    plan = {"op": "collect", "input": {
        "op": "hash_aggregate", "mode": "single", "group_by": [],
        "aggs": [{"fn": "sum", "as": "sv", "expr": {"col": "v"}},
                 {"fn": "count", "as": "c"}],
        "input": {"op": "hash_join",
                  "probe_chunk_rows": CHUNK,
                  "build": {"op": "scan", "schema": [
                      {"name": "bk", "dtype": "int64"},
                      {"name": "bv", "dtype": "int64"}],
                      "source": {"kind": "device", "table": "build"}},
                  "probe": {"op": "scan", "schema": [
                      {"name": "k", "dtype": "int64"},
                      {"name": "v", "dtype": "decimal128", "precision": 15,
                       "scale": 2}],
                      "source": {"kind": "device", "table": "probe"}},
                  "build_keys": ["bk"], "probe_keys": ["k"],
                  "join_type": "inner",
                  "output": [{"side": "probe", "col": "v"},
                             {"side": "build", "col": "bv"}]}}}
    doc = {"job_id": "cap", "stage_id": 1, "task_id": 0,
           "work_dir": "/tmp/x", "plan": plan}
    t0 = time.perf_counter()
    r = stage.execute(doc)
    cold = time.perf_counter() - t0
    t0 = time.perf_counter()
    r2 = stage.execute(doc)
    warm = time.perf_counter() - t0
    assert r["rows"] == r2["rows"]
    free1, _ = torch.cuda.mem_get_info()

    # torch cross-check: even keys match
    match_mask = (keys % 2 == 0) & (keys < 2 * NBUILD)
    want_cnt = int(match_mask.sum().item())
    want_sum = int(val[:, 0][match_mask].sum().item())
    assert r["rows"][0][1] == want_cnt, (r["rows"], want_cnt)
    assert int(r["rows"][0][0]) == want_sum
    rec = {"probe_rows": NPROBE, "build_rows": NBUILD,
           "chunk_rows": CHUNK, "matches": want_cnt,
           "wall_s_cold": cold, "wall_s_warm": warm,
           "probe_rows_per_s": NPROBE / warm,
           "hbm_free_before_gib": free0 / 2**30,
           "hbm_free_after_gib": free1 / 2**30,
           "crosscheck": "exact (torch)"}
    print(json.dumps(rec), flush=True)
    os.makedirs(os.path.join(ROOT, "gpurun_out"), exist_ok=True)
    with open(os.path.join(ROOT, "gpurun_out", "perf_capacity_join.json"),
              "w") as f:
        json.dump(rec, f, indent=1)


if __name__ == "__main__":
    main()
