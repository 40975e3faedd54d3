#!/usr/bin/env python3
"""HBM-traffic calibration for k_q6_agg (run under rocprofv3 --pmc).

Launches the fused q6 kernel at three selectivities with analytically known
algorithmic byte counts, so the PMC FETCH_SIZE per dispatch can be
calibrated on this exact access pattern (MI355X_MICROARCH.md §HBM: gfx950
FETCH_SIZE reports 1/2 of wide coalesced reads — calibrate, don't assume):
  case all-fail-date: date window empty  -> reads date+disc+qty = 36 B/row
     (disc/qty feed the predicate so their loads cannot be elided; price is
      never needed)
  case all-pass:      full ranges        -> reads all four columns = 52 B/row
  case real-q6:       1994 window        -> 36 B/row + touched price lines
"""
import json
import os
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

import torch  # noqa: E402
from datafusion_ballista_amd import gpu, tpch_synth  # noqa: E402

N = 200_000_000


def main():
    assert torch.cuda.is_available()
    ctx = gpu.GpuStageContext(0)
    cols = tpch_synth.lineitem_torch(N, torch.device("cuda:0"), seed=77)
    torch.cuda.synchronize()

    def col_of(t, dtype):
        return gpu.BgColumn(dtype, 15, 2, 0, t.data_ptr(), None, None, t.shape[0])

    sd = col_of(cols["l_shipdate"], gpu.BG_DT_DATE32)
    cd = col_of(cols["l_discount"], gpu.BG_DT_DECIMAL128)
    cq = col_of(cols["l_quantity"], gpu.BG_DT_DECIMAL128)
    cp = col_of(cols["l_extendedprice"], gpu.BG_DT_DECIMAL128)

    cases = [
        ("all_fail_date", (0, 0, 0, 10, 10**9), 36 * N),
        ("all_pass", (0, 20000, 0, 10, 10**9), 52 * N),
        ("real_q6", (tpch_synth.Q6_DATE_LO, tpch_synth.Q6_DATE_HI,
                     tpch_synth.Q6_DISC_LO, tpch_synth.Q6_DISC_HI,
                     tpch_synth.Q6_QTY_LT), None),
    ]
    results = []
    for name, (dlo, dhi, plo, phi, qlt), algo in cases:
        cnt, total = ctx.q6_agg(sd, cd, cq, cp, dlo, dhi, plo, phi, qlt)
        ms = ctx.L.bg_last_kernel_ms()
        results.append({"case": name, "rows": N, "count": cnt,
                        "kernel_ms": ms, "algo_bytes": algo,
                        "gbps_vs_algo": (algo / ms / 1e6) if algo else None})
        print(json.dumps(results[-1]), flush=True)
    with open(os.path.join(ROOT, "gpurun_out", "pmc_cases.json"), "w") as f:
        json.dump(results, f, indent=1)


if __name__ == "__main__":
    main()
