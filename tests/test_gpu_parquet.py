"""GPU Parquet column decode vs pyarrow's own reader.

The files are written BY pyarrow in the test (PLAIN encoding, Snappy or
uncompressed, multiple pages per chunk) — pyarrow's read path is the pinned
reference for the decoded bytes."""
import decimal

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest

from datafusion_ballista_amd import gpu
from datafusion_ballista_amd.parquet import GpuParquetColumnReader

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ctx():
    c = gpu.GpuStageContext(0)
    yield c
    c.close()


def make_file(tmp_path, compression, n=200_000, use_dictionary=False):
    rng = np.random.default_rng(3)
    dec_vals = [decimal.Decimal(int(v)) / 100 for v in
                rng.integers(-10**9, 10**9, size=n)]
    table = pa.table({
        "k64": pa.array(rng.integers(-2**60, 2**60, size=n, dtype=np.int64)),
        "d32": pa.array(rng.integers(-10**9, 10**9, size=n, dtype=np.int32)),
        "f64": pa.array(rng.standard_normal(n)),
        "dec": pa.array(dec_vals, type=pa.decimal128(15, 2)),
        "lowcard": pa.array(rng.integers(0, 50, size=n, dtype=np.int64)),
    })
    path = str(tmp_path / f"t_{compression}_{use_dictionary}.parquet")
    pq.write_table(table, path, compression=compression,
                   use_dictionary=use_dictionary,
                   data_page_size=64 * 1024, write_statistics=False)
    return path, table


@pytest.mark.parametrize("compression", ["snappy", "none", "zstd", "gzip"])
@pytest.mark.parametrize("use_dictionary", [False, True])
def test_parquet_column_decode(ctx, tmp_path, compression, use_dictionary):
    """PLAIN and dictionary-encoded chunks (pyarrow's default is
    dictionary; low-cardinality columns stay dict-coded, high-cardinality
    ones fall back to PLAIN mid-chunk — both paths on device)."""
    path, table = make_file(tmp_path, compression,
                            use_dictionary=use_dictionary)
    rd = GpuParquetColumnReader(ctx, path)
    n = table.num_rows

    buf, nv, phys, _ = rd.read_column(0, 0)   # k64
    assert nv == n and phys == "INT64"
    assert np.array_equal(buf.download(np.int64, n),
                          table.column("k64").to_numpy())

    buf, nv, phys, _ = rd.read_column(0, 1)   # d32
    assert np.array_equal(buf.download(np.int32, n),
                          table.column("d32").to_numpy())

    buf, nv, phys, _ = rd.read_column(0, 2)   # f64
    assert np.array_equal(buf.download(np.float64, n),
                          table.column("f64").to_numpy())

    buf, nv, phys, _ = rd.read_column(0, 3)   # decimal128 -> Arrow LE bytes
    assert phys == "DECIMAL128"
    got = buf.download(np.uint8, 16 * n)
    want = table.column("dec").combine_chunks().buffers()[1]
    want_np = np.frombuffer(want, dtype=np.uint8, count=16 * n)
    assert np.array_equal(got, want_np)

    buf, nv, phys, _ = rd.read_column(0, 4)   # lowcard: dict-coded when enabled
    assert np.array_equal(buf.download(np.int64, n),
                          table.column("lowcard").to_numpy())


def test_parquet_decode_feeds_q6_kernel(ctx, tmp_path):
    """Decoded-parquet columns drive the fused q6 kernel directly (the
    DataSourceExec -> FilterExec -> AggregateExec chain, file-fed)."""
    import oracle
    rng = np.random.default_rng(9)
    n = 100_000
    sd = rng.integers(8000, 11000, size=n, dtype=np.int32)
    disc = rng.integers(0, 11, size=n, dtype=np.int64)
    qty = rng.integers(100, 5100, size=n, dtype=np.int64)
    price = rng.integers(90000, 10495100, size=n, dtype=np.int64)
    table = pa.table({
        "l_shipdate": pa.array(sd, type=pa.int32()),
        "l_discount": pa.array([decimal.Decimal(int(v)) / 100 for v in disc],
                               type=pa.decimal128(15, 2)),
        "l_quantity": pa.array([decimal.Decimal(int(v)) / 100 for v in qty],
                               type=pa.decimal128(15, 2)),
        "l_extendedprice": pa.array(
            [decimal.Decimal(int(v)) / 100 for v in price],
            type=pa.decimal128(15, 2)),
    })
    path = str(tmp_path / "lineitem.parquet")
    pq.write_table(table, path, compression="snappy", use_dictionary=False,
                   data_page_size=128 * 1024, write_statistics=False)
    rd = GpuParquetColumnReader(ctx, path)
    sd_buf, _, _, _ = rd.read_column(0, 0)
    disc_buf, _, _, _ = rd.read_column(0, 1)
    qty_buf, _, _, _ = rd.read_column(0, 2)
    price_buf, _, _, _ = rd.read_column(0, 3)
    from datafusion_ballista_amd import tpch_synth
    c_sd = ctx.column(gpu.BG_DT_DATE32, sd_buf, n)
    c_d = ctx.column(gpu.BG_DT_DECIMAL128, disc_buf, n)
    c_q = ctx.column(gpu.BG_DT_DECIMAL128, qty_buf, n)
    c_p = ctx.column(gpu.BG_DT_DECIMAL128, price_buf, n)
    cnt, total = ctx.q6_agg(c_sd, c_d, c_q, c_p, 8766, 9131, 5, 7, 2400)

    d16 = np.zeros(16 * n, dtype=np.uint8)
    q16 = np.zeros(16 * n, dtype=np.uint8)
    p16 = np.zeros(16 * n, dtype=np.uint8)
    for i in range(n):
        d16[16*i:16*i+8] = np.frombuffer(
            int(disc[i]).to_bytes(8, "little", signed=True), dtype=np.uint8)
        q16[16*i:16*i+8] = np.frombuffer(
            int(qty[i]).to_bytes(8, "little", signed=True), dtype=np.uint8)
        p16[16*i:16*i+8] = np.frombuffer(
            int(price[i]).to_bytes(8, "little", signed=True), dtype=np.uint8)
    want_cnt, want_sum = oracle.q6(sd, d16, q16, p16, 8766, 9131, 5, 7, 2400)
    assert cnt == want_cnt and total == want_sum and cnt > 0


@pytest.mark.parametrize("compression", ["snappy", "none"])
@pytest.mark.parametrize("use_dictionary", [False, True])
@pytest.mark.parametrize("version", ["1.0", "2.0"])
def test_parquet_nullable_decode(ctx, tmp_path, compression,
                                 use_dictionary, version):
    """Columns with real NULLs: definition levels decode on device to the
    Arrow validity bitmap + value scatter; values at valid slots and the
    bitmap itself must match pyarrow's own reader bit-for-bit."""
    n = 150_000
    rng = np.random.default_rng(17)
    k64 = rng.integers(-2**60, 2**60, size=n, dtype=np.int64)
    f64 = rng.standard_normal(n)
    low = rng.integers(0, 40, size=n, dtype=np.int64)
    dec_vals = [decimal.Decimal(int(v)) / 100 for v in
                rng.integers(-10**9, 10**9, size=n)]
    masks = {name: rng.random(n) < p for name, p in
             [("k64", 0.2), ("f64", 0.5), ("dec", 0.03), ("low", 0.25)]}
    table = pa.table({
        "k64": pa.array(k64, mask=masks["k64"]),
        "f64": pa.array(f64, mask=masks["f64"]),
        "dec": pa.array([None if m else v
                         for v, m in zip(dec_vals, masks["dec"])],
                        type=pa.decimal128(15, 2)),
        "low": pa.array(low, mask=masks["low"]),
    })
    path = str(tmp_path /
               f"null_{compression}_{use_dictionary}_{version}.parquet")
    pq.write_table(table, path, compression=compression,
                   use_dictionary=use_dictionary, data_page_version=version,
                   data_page_size=32 * 1024, write_statistics=False)
    rd = GpuParquetColumnReader(ctx, path)
    ref = pq.read_table(path)

    for ci, (name, npdt, esz) in enumerate(
            [("k64", np.int64, 8), ("f64", np.float64, 8),
             ("dec", None, 16), ("low", np.int64, 8)]):
        buf, nv, phys, valid = rd.read_column_all(ci)
        assert nv == n and valid is not None
        vw = valid.download(np.uint32, (n + 31) // 32)
        got_valid = np.unpackbits(vw.view(np.uint8),
                                  bitorder="little")[:n].astype(bool)
        want_valid = ~np.asarray(masks[name])
        assert np.array_equal(got_valid, want_valid), name
        col = ref.column(name).combine_chunks()
        if name == "dec":
            raw = buf.download(np.uint8, 16 * n).reshape(n, 16)
            want = [v for v, m in zip(dec_vals, masks["dec"]) if not m]
            got = [int.from_bytes(bytes(raw[i]), "little", signed=True)
                   for i in range(n) if got_valid[i]]
            assert got == [int(w * 100) for w in want]
        else:
            vals = buf.download(npdt, n)
            want = col.drop_null().to_numpy(zero_copy_only=False)
            assert np.array_equal(vals[got_valid], want.astype(npdt))


@pytest.mark.parametrize("compression", ["none", "snappy"])
def test_parquet_v2_all_null_pages(ctx, tmp_path, compression):
    """Regression: an all-null DataPageV2 page has an
    EMPTY values section; the old sign-encoded fixup misread the 0-length
    raw copy as a def-len prefix entry and desynchronized every later V2
    page.  Small pages force several all-null pages before valued ones."""
    n = 60_000
    rng = np.random.default_rng(31)
    vals = rng.integers(-2**60, 2**60, size=n, dtype=np.int64)
    # first half all-null, second half mixed -> the column chunk starts
    # with multiple all-null pages followed by valued pages
    mask = np.concatenate([np.ones(n // 2, dtype=bool),
                           rng.random(n - n // 2) < 0.3])
    table = pa.table({"v": pa.array(vals, mask=mask)})
    path = str(tmp_path / f"v2null_{compression}.parquet")
    pq.write_table(table, path, compression=compression,
                   use_dictionary=False, data_page_version="2.0",
                   data_page_size=8 * 1024, write_statistics=False)
    rd = GpuParquetColumnReader(ctx, path)
    buf, nv, phys, valid = rd.read_column_all(0)
    assert nv == n and valid is not None
    vw = valid.download(np.uint32, (n + 31) // 32)
    got_valid = np.unpackbits(vw.view(np.uint8),
                              bitorder="little")[:n].astype(bool)
    assert np.array_equal(got_valid, ~mask)
    got = buf.download(np.int64, n)
    assert np.array_equal(got[got_valid], vals[~mask])


@pytest.mark.parametrize("compression", ["snappy", "none"])
@pytest.mark.parametrize("use_dictionary", [False, True])
@pytest.mark.parametrize("with_nulls", [False, True])
@pytest.mark.parametrize("version", ["1.0", "2.0"])
def test_parquet_byte_array_decode(ctx, tmp_path, compression,
                                   use_dictionary, with_nulls, version):
    """BYTE_ARRAY (Utf8) columns: PLAIN [u32 len][bytes] pages and dict-
    coded pages decode on device to Arrow offsets+data; strings at valid
    slots must match pyarrow's reader exactly."""
    n = 80_000
    rng = np.random.default_rng(23)
    words = ["alpha", "bravo", "charlie", "delta", "echo", "foxtrot",
             "", "x" * 120, "golf-hotel-india", "juliett"]
    strs = [words[i % len(words)] + str(rng.integers(0, 1000))
            if i % 7 else words[i % len(words)] for i in range(n)]
    mask = (rng.random(n) < 0.15) if with_nulls else np.zeros(n, bool)
    arr = pa.array([None if m else v for v, m in zip(strs, mask)],
                   type=pa.string())
    table = pa.table({"s": arr})
    path = str(tmp_path / f"ba_{compression}_{use_dictionary}_{with_nulls}"
               f"_{version}.parquet")
    pq.write_table(table, path, compression=compression,
                   use_dictionary=use_dictionary, data_page_version=version,
                   data_page_size=16 * 1024, write_statistics=False)
    rd = GpuParquetColumnReader(ctx, path)
    (offs_buf, data_buf, total), nv, phys, valid = rd.read_column_all(0)
    assert nv == n and phys == "BYTE_ARRAY"
    offs = offs_buf.download(np.int32, n + 1)
    data = data_buf.download(np.uint8, max(total, 1))
    if with_nulls:
        assert valid is not None
        vw = valid.download(np.uint32, (n + 31) // 32)
        got_valid = np.unpackbits(vw.view(np.uint8),
                                  bitorder="little")[:n].astype(bool)
        assert np.array_equal(got_valid, ~mask)
    assert offs[0] == 0 and np.all(np.diff(offs) >= 0)
    for i in rng.choice(n, 2000, replace=False):
        if mask[i]:
            continue
        got = bytes(data[offs[i]:offs[i + 1]]).decode()
        assert got == strs[i], (i, got, strs[i])
    # full equality on the packed bytes of valid slots
    want_all = "".join(v for v, m in zip(strs, mask) if not m).encode()
    got_all = b"".join(bytes(data[offs[i]:offs[i + 1]])
                       for i in range(n) if not mask[i])
    assert got_all == want_all


@pytest.mark.parametrize("compression", ["snappy", "none"])
@pytest.mark.parametrize("with_nulls", [False, True])
@pytest.mark.parametrize("version", ["1.0", "2.0"])
def test_parquet_delta_binary_packed(ctx, tmp_path, compression,
                                     with_nulls, version):
    """DELTA_BINARY_PACKED integer pages (encoding 5 — the reference's
    parquet-rs V2 default for ints): prefix-chain decode on device must
    match pyarrow's reader exactly, incl. nullable slot scatter."""
    n = 120_000
    rng = np.random.default_rng(67)
    a64 = rng.integers(-10**14, 10**14, size=n, dtype=np.int64)
    a64[::11] = a64[::11] // 10**9  # mixed magnitudes -> varied bit widths
    a32 = rng.integers(-2**30, 2**30, size=n, dtype=np.int32)
    mask = (rng.random(n) < 0.2) if with_nulls else np.zeros(n, bool)
    table = pa.table({
        "a": pa.array(a64, mask=mask),
        "b": pa.array(a32, mask=mask),
    })
    path = str(tmp_path / f"delta_{compression}_{with_nulls}_{version}"
               ".parquet")
    pq.write_table(table, path, compression=compression,
                   use_dictionary=False, data_page_version=version,
                   column_encoding={"a": "DELTA_BINARY_PACKED",
                                    "b": "DELTA_BINARY_PACKED"},
                   data_page_size=16 * 1024, write_statistics=False)
    rd = GpuParquetColumnReader(ctx, path)
    for ci, (vals, npdt) in enumerate([(a64, np.int64), (a32, np.int32)]):
        buf, nv, phys, valid = rd.read_column_all(ci)
        assert nv == n
        got = buf.download(npdt, n)
        if with_nulls:
            vw = valid.download(np.uint32, (n + 31) // 32)
            gv = np.unpackbits(vw.view(np.uint8),
                               bitorder="little")[:n].astype(bool)
            assert np.array_equal(gv, ~mask)
            assert np.array_equal(got[gv], vals[~mask])
        else:
            assert np.array_equal(got, vals)


@pytest.mark.parametrize("encname", ["DELTA_LENGTH_BYTE_ARRAY",
                                     "DELTA_BYTE_ARRAY"])
@pytest.mark.parametrize("compression", ["snappy", "none"])
@pytest.mark.parametrize("with_nulls", [False, True])
def test_parquet_delta_strings(ctx, tmp_path, encname, compression,
                               with_nulls):
    """DELTA_LENGTH_BYTE_ARRAY (6) and DELTA_BYTE_ARRAY (7) string pages —
    the reference's parquet-rs V2 defaults for BYTE_ARRAY: device decode
    must reproduce pyarrow's strings exactly (shared-prefix chains for 7)."""
    n = 60_000
    rng = np.random.default_rng(71)
    # heavy shared prefixes make enc 7 chains meaningful
    strs = [f"common/prefix/dir{i % 37:03d}/file_{i % 911:04d}.part"
            if i % 5 else f"x{i}" for i in range(n)]
    mask = (rng.random(n) < 0.15) if with_nulls else np.zeros(n, bool)
    table = pa.table({"s": pa.array(
        [None if m else v for v, m in zip(strs, mask)], type=pa.string())})
    path = str(tmp_path / f"dba_{encname}_{compression}_{with_nulls}"
               ".parquet")
    pq.write_table(table, path, compression=compression,
                   use_dictionary=False, column_encoding={"s": encname},
                   data_page_size=16 * 1024, write_statistics=False)
    rd = GpuParquetColumnReader(ctx, path)
    (offs_buf, data_buf, total), nv, phys, valid = rd.read_column_all(0)
    assert nv == n and phys == "BYTE_ARRAY"
    offs = offs_buf.download(np.int32, n + 1)
    data = data_buf.download(np.uint8, max(total, 1))
    if with_nulls:
        vw = valid.download(np.uint32, (n + 31) // 32)
        gv = np.unpackbits(vw.view(np.uint8),
                           bitorder="little")[:n].astype(bool)
        assert np.array_equal(gv, ~mask)
    got_all = b"".join(bytes(data[offs[i]:offs[i + 1]])
                       for i in range(n) if not mask[i])
    want_all = "".join(v for v, m in zip(strs, mask) if not m).encode()
    assert got_all == want_all
    for i in rng.choice(n, 500, replace=False):
        if not mask[i]:
            assert bytes(data[offs[i]:offs[i + 1]]).decode() == strs[i]


@pytest.mark.parametrize("compression", ["snappy", "none"])
@pytest.mark.parametrize("with_nulls", [False, True])
def test_parquet_byte_stream_split(ctx, tmp_path, compression, with_nulls):
    """BYTE_STREAM_SPLIT float pages (encoding 9): the per-byte stream
    transpose must reproduce pyarrow's doubles bit-exactly."""
    n = 90_000
    rng = np.random.default_rng(83)
    f64 = rng.standard_normal(n) * 1e9
    mask = (rng.random(n) < 0.2) if with_nulls else np.zeros(n, bool)
    table = pa.table({"f": pa.array(f64, mask=mask)})
    path = str(tmp_path / f"bss_{compression}_{with_nulls}.parquet")
    pq.write_table(table, path, compression=compression,
                   use_dictionary=False,
                   column_encoding={"f": "BYTE_STREAM_SPLIT"},
                   data_page_size=16 * 1024, write_statistics=False)
    rd = GpuParquetColumnReader(ctx, path)
    buf, nv, phys, valid = rd.read_column_all(0)
    assert nv == n
    got = buf.download(np.float64, n)
    if with_nulls:
        vw = valid.download(np.uint32, (n + 31) // 32)
        gv = np.unpackbits(vw.view(np.uint8),
                           bitorder="little")[:n].astype(bool)
        assert np.array_equal(gv, ~mask)
        assert np.array_equal(got[gv].view(np.uint64),
                              f64[~mask].view(np.uint64))
    else:
        assert np.array_equal(got.view(np.uint64), f64.view(np.uint64))


def _bitmap_bools(words, n):
    bits = np.unpackbits(words.view(np.uint8), bitorder="little")
    return bits[:n].astype(bool)


@pytest.mark.parametrize("compression", ["snappy", "none"])
@pytest.mark.parametrize("use_dictionary", [False, True])
def test_parquet_list_column_decode(ctx, tmp_path, compression,
                                    use_dictionary):
    """LIST<int64> / LIST<double> with null lists, empty lists and null
    elements (repetition + definition levels walked on device by
    bg_list_levels_batch) vs pyarrow's own reader."""
    rng = np.random.default_rng(21)
    n = 40_000

    def make_lists(values_f, null_lists=True):
        out = []
        for i in range(n):
            r = rng.random()
            if null_lists and r < 0.10:
                out.append(None)
            elif r < 0.20:
                out.append([])
            else:
                ln = int(rng.integers(1, 9))
                out.append([None if rng.random() < 0.15 else values_f()
                            for _ in range(ln)])
        return out

    li = make_lists(lambda: int(rng.integers(-2**50, 2**50)))
    ld = make_lists(lambda: float(rng.standard_normal()))
    table = pa.table({
        "li": pa.array(li, type=pa.list_(pa.int64())),
        "ld": pa.array(ld, type=pa.list_(pa.float64())),
    })
    path = str(tmp_path / f"lists_{compression}_{use_dictionary}.parquet")
    pq.write_table(table, path, compression=compression,
                   use_dictionary=use_dictionary,
                   data_page_size=64 * 1024, row_group_size=n // 2,
                   write_statistics=False)
    rd = GpuParquetColumnReader(ctx, path)
    want_tbl = pq.read_table(path)
    for col, name, npdt in ((0, "li", np.int64), (1, "ld", np.float64)):
        res = rd.read_list_column_all(col)
        want = want_tbl.column(name).combine_chunks()
        assert res["n_rows"] == len(want)
        w_offs = want.offsets.to_numpy(zero_copy_only=False)
        assert np.array_equal(res["offsets"], w_offs.astype(np.int32))
        # list validity
        got_lv = _bitmap_bools(
            res["list_valid"].download(np.uint32,
                                       (res["n_rows"] + 31) // 32),
            res["n_rows"])
        assert np.array_equal(got_lv, ~np.array(want.is_null()))
        # element validity + values
        w_elems = want.values
        assert res["n_entries"] == len(w_elems)
        got_ev = _bitmap_bools(
            res["elem_valid"].download(np.uint32,
                                       (res["n_entries"] + 31) // 32),
            res["n_entries"])
        w_ev = ~np.array(w_elems.is_null())
        assert np.array_equal(got_ev, w_ev)
        got_vals = res["values"].download(npdt, max(res["n_entries"], 1))
        w_vals = w_elems.to_numpy(zero_copy_only=False)
        assert np.array_equal(got_vals[:res["n_entries"]][w_ev],
                              np.asarray(w_vals)[w_ev].astype(npdt))


def test_parquet_list_no_null_elements(ctx, tmp_path):
    """LIST<int64> with non-null elements (max_def=2 shape) and the flat
    path's loud rejection of nested columns."""
    rng = np.random.default_rng(5)
    rows = []
    for _ in range(10_000):
        r = rng.random()
        rows.append(None if r < 0.1 else
                    [int(v) for v in
                     rng.integers(0, 100, size=int(rng.integers(0, 6)))])
    table = pa.table({"x": pa.array(
        rows, type=pa.list_(pa.field("element", pa.int64(),
                                     nullable=False)))})
    path = str(tmp_path / "lists_nonnull.parquet")
    pq.write_table(table, path, compression="snappy",
                   use_dictionary=False, write_statistics=False)
    rd = GpuParquetColumnReader(ctx, path)
    res = rd.read_list_column_all(0)
    want = pq.read_table(path).column("x").combine_chunks()
    w_offs = want.offsets.to_numpy(zero_copy_only=False)
    assert np.array_equal(res["offsets"], w_offs.astype(np.int32))
    got = res["values"].download(np.int64, max(res["n_entries"], 1))
    assert np.array_equal(got[:res["n_entries"]],
                          want.values.to_numpy(zero_copy_only=False))
    with pytest.raises(RuntimeError, match="nested"):
        rd.read_column_all(0)


@pytest.mark.parametrize("use_dictionary", [False, True])
def test_parquet_list_strings_and_decimals(ctx, tmp_path, use_dictionary):
    """LIST<string> (BYTE_ARRAY elements via the lens/addr staging +
    materialize path) and LIST<decimal(15,2)> (FLBA big-endian
    sign-extension) vs pyarrow."""
    rng = np.random.default_rng(33)
    n = 20_000
    words = ["green", "lemon", "", "navajo peru", "x" * 40, "midnight"]

    def mk(valf):
        out = []
        for _ in range(n):
            r = rng.random()
            if r < 0.10:
                out.append(None)
            elif r < 0.20:
                out.append([])
            else:
                out.append([None if rng.random() < 0.15 else valf()
                            for _ in range(int(rng.integers(1, 6)))])
        return out

    ls = mk(lambda: words[int(rng.integers(0, len(words)))])
    dv = mk(lambda: decimal.Decimal(int(rng.integers(-10**9, 10**9))) / 100)
    table = pa.table({
        "ls": pa.array(ls, type=pa.list_(pa.string())),
        "ldec": pa.array(dv, type=pa.list_(pa.decimal128(15, 2))),
    })
    path = str(tmp_path / f"lists_sd_{use_dictionary}.parquet")
    pq.write_table(table, path, compression="snappy",
                   use_dictionary=use_dictionary,
                   data_page_size=32 * 1024, row_group_size=n // 2,
                   write_statistics=False)
    rd = GpuParquetColumnReader(ctx, path)
    want_tbl = pq.read_table(path)

    res = rd.read_list_column_all(0)
    want = want_tbl.column("ls").combine_chunks()
    w_offs = want.offsets.to_numpy(zero_copy_only=False)
    assert np.array_equal(res["offsets"], w_offs.astype(np.int32))
    offs32, data, tot = res["values"]
    got_offs = offs32.download(np.int32, res["n_entries"] + 1)
    got_data = data.download(np.uint8, max(tot, 1)).tobytes()
    w_elems = want.values
    w_ev = ~np.array(w_elems.is_null())
    got_ev = _bitmap_bools(
        res["elem_valid"].download(np.uint32,
                                   (res["n_entries"] + 31) // 32),
        res["n_entries"])
    assert np.array_equal(got_ev, w_ev)
    for i in range(res["n_entries"]):
        if w_ev[i]:
            s = got_data[got_offs[i]:got_offs[i + 1]].decode()
            assert s == w_elems[i].as_py(), i

    res = rd.read_list_column_all(1)
    want = want_tbl.column("ldec").combine_chunks()
    w_offs = want.offsets.to_numpy(zero_copy_only=False)
    assert np.array_equal(res["offsets"], w_offs.astype(np.int32))
    got = res["values"].download(np.int64,
                                 2 * max(res["n_entries"], 1))
    w_elems = want.values
    w_ev = ~np.array(w_elems.is_null())
    for i in range(res["n_entries"]):
        if w_ev[i]:
            lo, hi = int(got[2 * i]), int(got[2 * i + 1])
            v = (hi << 64) | (lo & (2**64 - 1))
            assert v == int(w_elems[i].as_py() * 100), i
