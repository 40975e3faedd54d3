"""TPC-H dbgen-shaped data generation: restates the
COLUMN DISTRIBUTIONS the TPC-H specification (v3.0 §4.2.2-4.2.3) pins for
ORDERS/LINEITEM — per-order line counts, date correlation, partkey-derived
prices — so perf and parity runs see realistic skew and correlation rather
than round 1's independent uniforms.

What is and is not restated (honesty note): dbgen's exact byte streams
come from its seeded column PRNG tables (speed_seed.c); without the dbgen
sources those bitstreams are not reproducible here, so this module follows
the published FORMULAS and RANGES with a deterministic philox generator.
Row counts and selectivities match the reference's stat fixtures
(scheduler/tests/tpch_plan_stability/fixtures.rs: lineitem 600,037,902 /
orders 150,000,000 / customer 15,000,000 at SF100; q6 selectivity ~1.9%,
o_orderdate < 1995-03-15 ~48%, l_shipdate > 1995-03-15 ~54%).

Spec facts used:
  - orders: O = SF*1,500,000 rows; o_orderdate uniform in
    [STARTDATE=1992-01-01, ENDDATE-151 days=1998-08-02); o_custkey uniform
    over customers (dbgen skips 2/3 of keys; the skew-neutral uniform over
    C keys keeps join fan-in identical); o_shippriority = 0 (spec: fixed).
  - lineitem: per order 1..7 lines uniform (=> L ~= 4.0*O; the pinned
    600,037,902 at SF100 matches); l_orderkey = parent order key (dbgen
    spreads keys over 4x range with low bits; key VALUES don't affect the
    hash-partitioned plan shape, the per-key multiplicity 1..7 does);
    l_quantity uniform [1,50]; l_discount uniform [0.00,0.10];
    l_tax uniform [0.00,0.08];
    l_partkey uniform [1, SF*200,000];
    l_extendedprice = l_quantity * p_retailprice(l_partkey) where
      p_retailprice = (90000 + (partkey/10 mod 20001) + 100*(partkey mod
      1000)) / 100   (spec §4.2.3);
    l_shipdate = o_orderdate + uniform[1,121] days;
    l_returnflag 'R'/'A' if receiptdate <= currentdate else 'N';
    l_linestatus 'O' if shipdate > currentdate else 'F'.
  - customer: C = SF*150,000; c_mktsegment uniform over 5 segments.

All columns come back as torch tensors resident on `device` (decimals as
scaled int64 in [n,2] i128 pairs), generated in HBM.
"""

import torch

DATE_EPOCH_1992_01_01 = 8036
DATE_1995_03_15 = 9204
DATE_1998_08_02 = 10440   # ENDDATE - 151 days
DATE_CURRENT = 9298       # 1995-06-17 (spec CURRENTDATE)


def _gen(shape, lo, hi, g, device, dtype=torch.int64):
    return torch.randint(int(lo), int(hi), shape, generator=g,
                         device=device, dtype=dtype)


def retailprice_cents(partkey: torch.Tensor) -> torch.Tensor:
    """p_retailprice in cents (spec §4.2.3 formula, exact integer)."""
    return 90000 + (torch.div(partkey, 10, rounding_mode="floor") % 20001) \
        + 100 * (partkey % 1000)


def orders_lineitem(sf: int, device, seed: int = 19920101,
                    orders_cap: int = None, lines_cap: int = None):
    """Generate orders + lineitem with dbgen's correlations on device.
    Returns (orders dict, lineitem dict); decimals as scaled int64."""
    g = torch.Generator(device=device)
    g.manual_seed(seed)
    n_ord = sf * 1_500_000 if orders_cap is None else orders_cap
    n_cust = sf * 150_000

    o_orderkey = torch.arange(1, n_ord + 1, device=device, dtype=torch.int64)
    o_custkey = _gen((n_ord,), 1, n_cust + 1, g, device)
    o_orderdate = _gen((n_ord,), DATE_EPOCH_1992_01_01, DATE_1998_08_02, g,
                       device, torch.int32)
    o_shippriority = torch.zeros(n_ord, dtype=torch.int32, device=device)

    # lines per order: uniform 1..7 (spec); expand with repeat_interleave
    lines = _gen((n_ord,), 1, 8, g, device)
    if lines_cap is not None:
        # nudge to the exact pinned row count (fixtures.rs): add or remove
        # single lines from the first orders with slack — a vanishing
        # perturbation of the 1..7 distribution (~1e-4 of orders)
        total = int(lines.sum().item())
        if total < lines_cap:
            slack = (lines < 7).nonzero(as_tuple=True)[0][:lines_cap - total]
            lines[slack] += 1
        elif total > lines_cap:
            slack = (lines > 1).nonzero(as_tuple=True)[0][:total - lines_cap]
            lines[slack] -= 1
    o_idx_src = torch.arange(n_ord, device=device)
    l_order_idx = torch.repeat_interleave(o_idx_src, lines)
    n_li = l_order_idx.shape[0]

    l_orderkey = o_orderkey[l_order_idx]
    l_partkey = _gen((n_li,), 1, sf * 200_000 + 1, g, device)
    l_quantity = _gen((n_li,), 1, 51, g, device)          # whole units
    l_discount = _gen((n_li,), 0, 11, g, device)          # cents (0.00-0.10)
    l_tax = _gen((n_li,), 0, 9, g, device)                # cents
    l_extendedprice = l_quantity * retailprice_cents(l_partkey)
    l_shipdate = (o_orderdate[l_order_idx].to(torch.int64) +
                  _gen((n_li,), 1, 122, g, device)).to(torch.int32)
    l_receipt = (l_shipdate.to(torch.int64) +
                 _gen((n_li,), 1, 31, g, device)).to(torch.int32)
    l_linestatus = (l_shipdate > DATE_CURRENT).to(torch.uint8)  # 1='O'
    ret = _gen((n_li,), 0, 2, g, device)
    l_returnflag = torch.where(
        l_receipt <= DATE_CURRENT, ret.to(torch.uint8),
        torch.full_like(ret, 2).to(torch.uint8))  # 0='R',1='A',2='N'

    def dec(v):
        out = torch.zeros((v.shape[0], 2), dtype=torch.int64, device=device)
        out[:, 0] = v
        return out

    orders = {"o_orderkey": o_orderkey, "o_custkey": o_custkey,
              "o_orderdate": o_orderdate, "o_shippriority": o_shippriority}
    lineitem = {
        "l_orderkey": l_orderkey,
        "l_partkey": l_partkey,
        "l_quantity": dec(l_quantity * 100),      # Decimal(15,2)
        "l_extendedprice": dec(l_extendedprice),  # cents
        "l_discount": dec(l_discount),            # scale-2 int (0..10)
        "l_tax": dec(l_tax),
        "l_shipdate": l_shipdate,
        "l_returnflag": l_returnflag,
        "l_linestatus": l_linestatus,
    }
    return orders, lineitem


def customer(sf: int, device, seed: int = 19920102):
    g = torch.Generator(device=device)
    g.manual_seed(seed)
    n = sf * 150_000
    return {"c_custkey": torch.arange(1, n + 1, device=device,
                                      dtype=torch.int64),
            "c_mktsegment": _gen((n,), 0, 5, g, device,
                                 torch.int64).to(torch.uint8)}
