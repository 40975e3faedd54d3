#!/usr/bin/env python3
"""Device Snappy decode micro-benchmark: times bg_snappy_decompress alone
(warmed, pages resident) for incompressible (PLAIN int data), mixed, and
highly compressible (RLE-ish) page sets at two page-population sizes.
Run under rocprofv3 --kernel-trace --stats to split the three passes."""
import ctypes
import json
import os
import sys
import time

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

import numpy as np  # noqa: E402
import pyarrow as pa  # noqa: E402

from datafusion_ballista_amd import gpu  # noqa: E402

CODEC = pa.Codec("snappy")


def make_pages(kind, page_bytes, npages, rng):
    pages = []
    for i in range(npages):
        if kind == "random":
            raw = rng.integers(0, 2**63, size=page_bytes // 8,
                               dtype=np.int64).tobytes()
        elif kind == "lowcard":
            raw = rng.integers(0, 100, size=page_bytes // 8,
                               dtype=np.int64).tobytes()
        elif kind == "flba7":  # parquet FLBA(7) decimals (price column)
            vals = rng.integers(90000, 10495100, size=page_bytes // 7)
            raw = b"".join(int(v).to_bytes(7, "big") for v in vals)
        else:  # runs
            raw = (rng.integers(0, 256, size=64, dtype=np.uint8).tobytes()
                   * (page_bytes // 64))
        pages.append((CODEC.compress(raw).to_pybytes(), raw))
    return pages


def bench(ctx, pages, iters=5):
    L = gpu.load_library()
    blobs = b"".join(p[0] for p in pages)
    src = ctx.upload(np.frombuffer(blobs, dtype=np.uint8))
    total_u = sum(len(p[1]) for p in pages)
    dst = ctx.alloc(total_u)
    arr = (gpu.BgSnappyPage * len(pages))()
    soff = doff = 0
    for i, (comp, raw) in enumerate(pages):
        arr[i] = gpu.BgSnappyPage(
            ctypes.c_void_p(src.ptr.value + soff),
            ctypes.c_void_p(dst.ptr.value + doff), len(comp), len(raw))
        soff += len(comp)
        doff += len(raw)
    lens = np.zeros(len(pages), dtype=np.int64)
    lp = lens.ctypes.data_as(ctypes.POINTER(ctypes.c_int64))
    # warmup + verify
    gpu._check(L.bg_snappy_decompress(arr, ctypes.c_int64(len(pages)), lp),
               "snappy")
    if not os.environ.get("BG_SNAP_ABLATE"):
        assert all(lens[i] == len(pages[i][1]) for i in range(len(pages)))
        got = dst.download(np.uint8, min(total_u, 1 << 20))
        want = b"".join(p[1] for p in pages)[:1 << 20]
        assert got.tobytes() == want, "decode mismatch"
    ctx.synchronize()
    best = 1e30
    for _ in range(iters):
        t0 = time.perf_counter()
        gpu._check(L.bg_snappy_decompress(arr, ctypes.c_int64(len(pages)),
                                          lp), "snappy")
        dt = time.perf_counter() - t0
        best = min(best, dt)
    return total_u, best


def main():
    ctx = gpu.GpuStageContext(0)
    rng = np.random.default_rng(3)
    out = []
    only = sys.argv[1] if len(sys.argv) > 1 else None
    cases = [("random", 1024, 160), ("random", 1024, 2000),
             ("random", 64, 4000), ("lowcard", 1024, 160),
             ("lowcard", 1024, 2000), ("lowcard", 64, 4000),
             ("runs", 1024, 160), ("runs", 1024, 2000), ("runs", 64, 4000)]
    if only:
        kind, page_kb, npages = only.split(",")
        cases = [(kind, int(page_kb), int(npages))]
    for kind, page_kb, npages in cases:
            pages = make_pages(kind, page_kb * 1024, npages, rng)
            comp = sum(len(p[0]) for p in pages)
            total_u, dt = bench(ctx, pages)
            rec = {"kind": kind, "page_kb": page_kb, "npages": npages,
                   "comp_mb": comp / 1e6, "decoded_mb": total_u / 1e6,
                   "s": dt, "gbps": total_u / dt / 1e9}
            out.append(rec)
            print(json.dumps(rec), flush=True)
    os.makedirs(os.path.join(ROOT, "gpurun_out"), exist_ok=True)
    with open(os.path.join(ROOT, "gpurun_out", "perf_snappy.json"), "w") as f:
        json.dump(out, f, indent=1)


if __name__ == "__main__":
    main()
