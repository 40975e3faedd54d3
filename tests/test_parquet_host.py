"""Host-side parquet machinery (no GPU): the thrift compact-protocol page
-header parser and page classification against pyarrow's own metadata —
the same walk GpuParquetColumnReader drives on device buffers."""
import decimal

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest

from datafusion_ballista_amd.parquet import parse_page_header


def _walk(path, rg, col):
    pf = pq.ParquetFile(path)
    raw = open(path, "rb").read()
    m = pf.metadata.row_group(rg).column(col)
    start = m.data_page_offset
    if m.has_dictionary_page and m.dictionary_page_offset is not None and \
            m.dictionary_page_offset < start:
        start = m.dictionary_page_offset
    end = start + m.total_compressed_size
    pos, pages = start, []
    while pos < end:
        h, data_pos = parse_page_header(raw, pos)
        pages.append(h)
        pos = data_pos + h[3]
    return pf, pages


@pytest.mark.parametrize("version", ["1.0", "2.0"])
@pytest.mark.parametrize("compression", ["snappy", "none"])
def test_page_walk_matches_metadata(tmp_path, version, compression):
    n = 50_000
    rng = np.random.default_rng(5)
    table = pa.table({
        "a": pa.array(rng.integers(0, 10**9, n, dtype=np.int64),
                      mask=rng.random(n) < 0.1),
        "s": pa.array([f"w{i % 997}" for i in range(n)]),
    })
    path = str(tmp_path / f"t_{version}_{compression}.parquet")
    pq.write_table(table, path, compression=compression,
                   data_page_version=version, data_page_size=16 * 1024,
                   write_statistics=False)
    for col in (0, 1):
        pf, pages = _walk(path, 0, col)
        want_vals = pf.metadata.row_group(0).column(col).num_values
        got_vals = 0
        for h in pages:
            ptype = h.get(1, 0)
            if ptype == 0:
                got_vals += h.get(5, {}).get(1, 0)
            elif ptype == 3:
                dph = h.get(8, {})
                got_vals += dph.get(1, 0)
                assert dph.get(6, 0) == 0  # flat: no repetition levels
                assert 2 + dph.get(5, 0) <= h[2]  # def levels fit the page
            elif ptype == 2:
                assert h.get(7, {}).get(1, 0) > 0  # dictionary entries
        assert got_vals == want_vals
        # sizes: compressed sections must tile the chunk exactly (the walk
        # landing exactly on `end` proves every header parse consumed the
        # right byte count)


def test_page_header_rejects_garbage():
    with pytest.raises(Exception):
        parse_page_header(b"\xff\xfe\xfd\xfc" * 4, 0)


def _lvl_walk(buf, w, count):
    """CPU restatement of the kernel's LvlRd hybrid RLE/bit-packed level
    reader (kernels.hip k_list_levels_body): groups of 8 w-bit values per
    bit-packed run (refill bounded to the run's own bytes), 1-byte RLE
    values."""
    i, out = 0, []
    while len(out) < count:
        assert i < len(buf), "level stream exhausted"
        h, sh = 0, 0
        while True:
            b = buf[i]
            i += 1
            h |= (b & 0x7F) << sh
            if not (b & 0x80):
                break
            sh += 7
        if h & 1:
            nbytes = (h >> 1) * w
            bits = int.from_bytes(buf[i:i + nbytes], "little")
            i += nbytes
            for k in range((h >> 1) * 8):
                if len(out) >= count:
                    break
                out.append((bits >> (k * w)) & ((1 << w) - 1))
        else:
            run, v = h >> 1, buf[i]
            i += 1
            out.extend([v] * min(run, count - len(out)))
    return out


def test_list_level_walk_matches_pyarrow(tmp_path):
    """Walk a LIST column's rep+def level streams with the kernel's
    algorithm (CPU restatement) and check rows/entries/present/offsets
    against pyarrow's own reader — pins bg_list_levels_batch's level
    semantics without a GPU."""
    rng = np.random.default_rng(9)
    n = 30_000
    rows = []
    for _ in range(n):
        r = rng.random()
        if r < 0.1:
            rows.append(None)
        elif r < 0.2:
            rows.append([])
        else:
            rows.append([None if rng.random() < 0.2 else int(v)
                         for v in rng.integers(0, 10**6,
                                               size=int(rng.integers(1, 8)))])
    t = pa.table({"x": pa.array(rows, type=pa.list_(pa.int64()))})
    path = str(tmp_path / "l.parquet")
    pq.write_table(t, path, compression="none", use_dictionary=False,
                   data_page_size=16 * 1024, write_statistics=False)
    pf, pages = _walk(path, 0, 0)
    raw = open(path, "rb").read()
    assert pf.schema.column(0).max_repetition_level == 1
    max_def = pf.schema.column(0).max_definition_level
    assert max_def == 3
    want = pq.read_table(path).column("x").combine_chunks()
    w_offs = want.offsets.to_numpy(zero_copy_only=False)

    m = pf.metadata.row_group(0).column(0)
    pos = m.data_page_offset
    end = pos + m.total_compressed_size
    rows_seen = entries = present = 0
    sizes = []
    while pos < end:
        h, data_pos = parse_page_header(raw, pos)
        page = raw[data_pos:data_pos + h[3]]
        nslots = h[5][1]
        rlen = int.from_bytes(page[0:4], "little")
        dpos = 4 + rlen
        dlen = int.from_bytes(page[dpos:dpos + 4], "little")
        reps = _lvl_walk(page[4:4 + rlen], 1, nslots)
        defs = _lvl_walk(page[dpos + 4:dpos + 4 + dlen], 2, nslots)
        assert reps[0] == 0  # rows never span pages
        for rep, d in zip(reps, defs):
            if rep == 0:
                rows_seen += 1
                sizes.append(0)
            if d >= 2:
                sizes[-1] += 1
                entries += 1
            if d == max_def:
                present += 1
        pos = data_pos + h[3]
    assert rows_seen == n
    offsets = np.zeros(n + 1, dtype=np.int64)
    np.cumsum(sizes, out=offsets[1:])
    assert np.array_equal(offsets, w_offs.astype(np.int64))
    assert entries == len(want.values)
    assert present == len(want.values.drop_null())
