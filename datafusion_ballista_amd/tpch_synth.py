"""Synthetic TPC-H-shaped lineitem columns (no network => no real dbgen data;
BASELINE.md sanctions synthetic inputs of the pinned shape).

Shapes follow scheduler/tests/tpch_plan_stability/fixtures.rs:56-130:
  l_shipdate Date32, l_discount/l_quantity/l_extendedprice/l_tax
  Decimal128(15,2), l_returnflag/l_linestatus 1-char Utf8 (dict-encoded u8
  here), l_orderkey Int64.  Value ranges follow the TPC-H spec so q1/q6
  selectivities are realistic: shipdate 1992-01-02..1998-12-01, discount
  0.00..0.10, quantity 1..50, extendedprice ~ 901..104950, tax 0.00..0.08.

Two generators: numpy (host, parity tests) and torch (device-resident, used
by bench.py so SF100 never exists in host RAM).
"""

import numpy as np

DATE_LO = 8036   # 1992-01-02
DATE_HI = 10561  # 1998-12-01 (exclusive-ish upper bound of shipdate)

# q6 parameters (1994 window, disc 0.05..0.07, qty < 24)
Q6_DATE_LO = 8766   # 1994-01-01
Q6_DATE_HI = 9131   # 1995-01-01
Q6_DISC_LO = 5
Q6_DISC_HI = 7
Q6_QTY_LT = 2400

# q1 cutoff: 1998-09-02
Q1_DATE_LE = 10471


def dec128_pairs_np(vals_i64: np.ndarray) -> np.ndarray:
    """int64 scaled values -> [n,2] int64 (lo, hi) = Arrow Decimal128 LE."""
    out = np.empty((len(vals_i64), 2), dtype=np.int64)
    out[:, 0] = vals_i64
    out[:, 1] = np.where(vals_i64 < 0, -1, 0)
    return out


def lineitem_numpy(n: int, seed: int = 42):
    rng = np.random.default_rng(seed)
    cols = {
        "l_orderkey": rng.integers(1, max(n, 2), size=n, dtype=np.int64),
        "l_shipdate": rng.integers(DATE_LO, DATE_HI, size=n, dtype=np.int32),
        "l_quantity": rng.integers(1, 51, size=n, dtype=np.int64) * 100,
        "l_extendedprice": rng.integers(90000, 10495100, size=n, dtype=np.int64),
        "l_discount": rng.integers(0, 11, size=n, dtype=np.int64),
        "l_tax": rng.integers(0, 9, size=n, dtype=np.int64),
        "l_returnflag": rng.integers(0, 3, size=n).astype(np.uint8),  # A/N/R
        "l_linestatus": rng.integers(0, 2, size=n).astype(np.uint8),  # F/O
    }
    return cols


def lineitem_torch(n: int, device, seed: int = 42):
    """Device-resident generation (bench path): columns live in HBM from
    birth; Decimal128 as [n,2] int64 (lo, hi)."""
    import torch
    g = torch.Generator(device=device)
    g.manual_seed(seed)

    def ri(lo, hi, dtype=torch.int64):
        return torch.randint(lo, hi, (n,), generator=g, device=device,
                             dtype=dtype)

    def dec(v):
        out = torch.zeros((n, 2), dtype=torch.int64, device=device)
        out[:, 0] = v
        return out.contiguous()

    cols = {}
    cols["l_orderkey"] = ri(1, max(n, 2))
    cols["l_shipdate"] = ri(DATE_LO, DATE_HI, torch.int32)
    cols["l_quantity"] = dec(ri(1, 51) * 100)
    cols["l_extendedprice"] = dec(ri(90000, 10495100))
    cols["l_discount"] = dec(ri(0, 11))
    cols["l_tax"] = dec(ri(0, 9))
    cols["l_returnflag"] = ri(0, 3, torch.uint8)
    cols["l_linestatus"] = ri(0, 2, torch.uint8)
    return cols
