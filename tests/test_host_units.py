"""CPU unit tests for host-side logic (exchange splits, parquet metadata
parsing, LZ4 frame glue, aggregate decode) — GPU-free."""
import struct

import numpy as np
import pyarrow as pa
import pytest


def test_exchange_send_splits_and_owner():
    from datafusion_ballista_amd import exchange
    # K = 8 partitions over world 4 -> 2 local partitions per rank
    offsets = np.array([0, 5, 10, 10, 12, 20, 23, 30, 31], dtype=np.int64)
    splits = exchange.send_splits(offsets, world=4)
    assert splits == [10, 2, 11, 8]
    assert sum(splits) == 31
    assert [exchange.owner_of(p, 2) for p in range(8)] == \
        [0, 0, 1, 1, 2, 2, 3, 3]
    # element-size scaling (byte buffers)
    assert exchange.send_splits(offsets, world=4, elem_size=8) == \
        [80, 16, 88, 64]


@pytest.mark.parametrize("compression,use_dict", [("none", False),
                                                  ("snappy", True)])
def test_parquet_page_walk_geometry(tmp_path, compression, use_dict):
    """The thrift PageHeader parser accounts for every value in every page
    of every row group (pinned against pyarrow's own metadata)."""
    import pyarrow.parquet as pq
    from datafusion_ballista_amd.parquet import parse_page_header
    rng = np.random.default_rng(1)
    n = 150_000
    table = pa.table({
        "a": pa.array(rng.integers(0, 500, size=n, dtype=np.int64)),
        "b": pa.array(rng.standard_normal(n)),
    })
    path = str(tmp_path / "t.parquet")
    pq.write_table(table, path, compression=compression,
                   use_dictionary=use_dict, data_page_size=32 * 1024,
                   row_group_size=60_000, write_statistics=False)
    pf = pq.ParquetFile(path)
    raw = open(path, "rb").read()
    for rg in range(pf.metadata.num_row_groups):
        for col in range(2):
            meta = pf.metadata.row_group(rg).column(col)
            start = meta.data_page_offset
            if meta.has_dictionary_page and \
                    meta.dictionary_page_offset is not None and \
                    meta.dictionary_page_offset < start:
                start = meta.dictionary_page_offset
            end = start + meta.total_compressed_size
            pos = start
            nvals = 0
            while pos < end:
                h, data_pos = parse_page_header(raw, pos)
                if h.get(1, 0) == 0:
                    nvals += h.get(5, {}).get(1, 0)
                pos = data_pos + h[3]
            assert nvals == meta.num_values


def test_lz4_stored_frame_assembly_cpu():
    """Stored-block frame glue round-trips through Arrow's LZ4 codec with
    no GPU (the device path emits the same bytes)."""
    from datafusion_ballista_amd import gpu
    data = bytes(np.random.default_rng(2).integers(0, 256, 200_000,
                                                   dtype=np.uint8))
    nblocks = (len(data) + 65535) // 65536
    sizes = np.full(nblocks, -1, dtype=np.int64)  # all stored
    slot_bytes = bytearray()
    for i in range(nblocks):
        blk = data[i * 65536:(i + 1) * 65536]
        slot_bytes += blk + b"\x00" * (65544 - len(blk))
    frame = gpu.lz4_frame_assemble(sizes, bytes(slot_bytes), len(data))
    got = pa.decompress(frame, len(data), codec="lz4", asbytes=True)
    assert got == data


def test_agg_value_codec():
    from datafusion_ballista_amd import gpu
    for v in (0, 1, -1, 2**62, -(2**62), 123456789):
        enc_max = (v ^ (1 << 63)) & 0xFFFFFFFFFFFFFFFF if v >= 0 else \
            (v + (1 << 64)) ^ (1 << 63)
        raw = struct.pack("<Q", enc_max) + b"\x00" * 8
        assert gpu.decode_agg_value(gpu.BG_AGG_OP_MAX_I64, raw) == v
        enc_min = (~enc_max) & 0xFFFFFFFFFFFFFFFF
        raw = struct.pack("<Q", enc_min) + b"\x00" * 8
        assert gpu.decode_agg_value(gpu.BG_AGG_OP_MIN_I64, raw) == v
        raw16 = int(v).to_bytes(16, "little", signed=True)
        assert gpu.decode_agg_value(gpu.BG_AGG_OP_SUM_DEC128, raw16) == v
    # Float64 MIN/MAX: the IEEE totally-ordered u64 transform round-trips
    for f in (0.0, -0.0, 1.5, -1.5, 1e300, -1e300, 3.14159, -2.5e-308,
              float("inf"), float("-inf")):
        bits = struct.unpack("<Q", struct.pack("<d", f))[0]
        enc = (~bits) & 0xFFFFFFFFFFFFFFFF if bits >> 63 else             bits ^ (1 << 63)
        raw = struct.pack("<Q", enc) + b"\x00" * 8
        got = gpu.decode_agg_value(gpu.BG_AGG_OP_MAX_F64, raw)
        assert struct.pack("<d", got) == struct.pack("<d", f), f
        raw = struct.pack("<Q", (~enc) & 0xFFFFFFFFFFFFFFFF) + b"\x00" * 8
        got = gpu.decode_agg_value(gpu.BG_AGG_OP_MIN_F64, raw)
        assert struct.pack("<d", got) == struct.pack("<d", f), f
    # the transform is monotone: ordering of encodings == ordering of floats
    import random
    rnd = random.Random(5)
    fs = sorted(rnd.uniform(-1e12, 1e12) for _ in range(200))
    encs = []
    for f in fs:
        bits = struct.unpack("<Q", struct.pack("<d", f))[0]
        encs.append((~bits) & 0xFFFFFFFFFFFFFFFF if bits >> 63 else
                    bits ^ (1 << 63))
    assert encs == sorted(encs)


def test_ipc_stream_empty_batchless():
    """A schema-only stream from the handwritten writer framing (schema
    message verbatim + EOS) reads back as an empty table."""
    from datafusion_ballista_amd import ipc as bgipc
    schema = pa.schema([("x", pa.int64())])
    stream = bgipc.schema_message_bytes(schema) + bgipc.EOS
    got = pa.ipc.open_stream(pa.BufferReader(stream)).read_all()
    assert got.num_rows == 0 and got.schema.equals(schema)


def test_snappy_list_ranking_replay_restatement():
    """CPU restatement of the giant-page match-replay algorithm
    (kernels.hip k_snap_par_*): every output byte of a match has exactly
    ONE parent byte, (dst-off) + ((b-dst) mod off) — periodic/RLE matches
    included — literal bytes are roots, jump-4 pointer doubling converges
    in ceil(log4 depth) rounds, and the final gather from literal roots
    reproduces the stream byte-exactly.  Pinned against pyarrow's own
    snappy codec on the page shapes that route to this path."""
    import numpy as np
    import pyarrow as pa

    rng = np.random.default_rng(13)
    payloads = [
        # structured FLBA-like (deep-ish chains)
        b"".join(int(v).to_bytes(7, "big")
                 for v in rng.integers(90000, 10495100, size=40_000)),
        # heavy RLE: repeated 64-B pattern and pure zeros (deepest chains)
        rng.integers(0, 256, size=64, dtype=np.uint8).tobytes() * 2048,
        b"\x00" * 100_000,
        # low-cardinality ints
        rng.integers(0, 50, size=20_000, dtype=np.int64).tobytes(),
    ]
    for raw in payloads:
        comp = pa.compress(raw, codec="snappy", asbytes=True)
        # parse the element stream (tags per snappy format)
        s, i = comp, 0
        ulen, sh = 0, 0
        while True:
            b = s[i]
            i += 1
            ulen |= (b & 0x7F) << sh
            if not (b & 0x80):
                break
            sh += 7
        assert ulen == len(raw)
        out = bytearray(ulen)
        parents = np.arange(ulen, dtype=np.int64)  # identity = roots
        dst = 0
        while i < len(s):
            tag = s[i]
            t = tag & 3
            if t == 0:
                ln = (tag >> 2) + 1
                nb = 0
                if ln > 60:
                    nb = ln - 60
                    ln = int.from_bytes(s[i + 1:i + 1 + nb], "little") + 1
                out[dst:dst + ln] = s[i + 1 + nb:i + 1 + nb + ln]  # literal
                dst += ln
                i += 1 + nb + ln
                continue
            if t == 1:
                ln = ((tag >> 2) & 7) + 4
                off = ((tag >> 5) << 8) | s[i + 1]
                i += 2
            elif t == 2:
                ln = (tag >> 2) + 1
                off = s[i + 1] | (s[i + 2] << 8)
                i += 3
            else:
                ln = (tag >> 2) + 1
                off = int.from_bytes(s[i + 1:i + 5], "little")
                i += 5
            assert 0 < off <= dst
            base = dst - off
            idx = np.arange(ln, dtype=np.int64)
            parents[dst:dst + ln] = base + (idx % off)
            dst += ln
        assert dst == ulen
        # jump-4 doubling until fixpoint; rounds must stay logarithmic
        rounds = 0
        while True:
            nxt = parents[parents[parents[parents[parents]]]]
            rounds += 1
            if np.array_equal(nxt, parents):
                break
            parents = nxt
            assert rounds < 20
        # all roots are literal bytes (already written); one gather
        ob = np.frombuffer(bytes(out), dtype=np.uint8).copy()
        ob = ob[parents]
        assert ob.tobytes() == raw
