"""Sort-shuffle file format round-trip (CPU; oracle feeds the partitioner).

Pins the byte layout restated from the reference:
  - data file: leading schema-only IPC stream, then K partition-major byte
    ranges of concatenated complete IPC streams, LZ4_FRAME batches
    (write_task_consolidated writer.rs:794-895; codec config.rs:413-415)
  - index: (K+1) LE i64 absolute offsets, last = total length
    (index.rs:21-33)
  - readers cross sub-stream boundaries (multi_stream_reader.rs:17-34)
"""
import os
import struct

import numpy as np
import pyarrow as pa
import pytest

import oracle
from datafusion_ballista_amd import shuffle


@pytest.fixture()
def tmp_work(tmp_path):
    return str(tmp_path)


def make_table(n, seed=0):
    rng = np.random.default_rng(seed)
    keys = rng.integers(0, 1000, size=n, dtype=np.int64)
    vals = rng.integers(-10**9, 10**9, size=n, dtype=np.int64)
    dates = rng.integers(8000, 11000, size=n, dtype=np.int32)
    return pa.table({
        "k": pa.array(keys, type=pa.int64()),
        "v": pa.array(vals, type=pa.int64()),
        "d": pa.array(dates, type=pa.int32()),
    })


def cpu_partition_streams(table, k):
    """Partition with the oracle (test infrastructure) and encode."""
    n = table.num_rows
    keys = table.column("k").to_numpy()
    h = oracle.hash_columns([("i64", keys)], n)
    pids = oracle.partition_ids(h, k)
    idx, offs = oracle.partition_indices(pids, k)
    streams = []
    for p in range(k):
        rows = idx[offs[p]:offs[p + 1]]
        part = table.take(pa.array(rows, type=pa.uint32()))
        batches = shuffle.rechunk(part, 100)  # small batches => many per stream
        streams.append([shuffle.encode_partition_stream(batches, table.schema)])
    return streams, pids, idx, offs


def test_roundtrip_and_layout(tmp_work):
    k, n = 8, 5000
    table = make_table(n)
    streams, pids, idx, offs = cpu_partition_streams(table, k)
    data_path, index_path, stats = shuffle.write_task_consolidated(
        tmp_work, "job-1", 2, 0, table.schema, streams)

    # path layout: {work_dir}/{job}/{stage}/{task_id}/data.arrow (writer.rs:821-828)
    assert data_path == os.path.join(tmp_work, "job-1", "2", "0", "data.arrow")
    assert index_path == data_path + ".index"

    # index structure (index.rs:21-33): K+1 LE i64, monotonic, last == filesize
    raw = open(index_path, "rb").read()
    assert len(raw) == 8 * (k + 1)
    offsets = struct.unpack(f"<{k + 1}q", raw)
    assert all(offsets[i] <= offsets[i + 1] for i in range(k))
    assert offsets[-1] == os.path.getsize(data_path)
    assert offsets[0] > 0  # leading schema-only header stream precedes p0

    # the leading bytes decode as a schema-only IPC stream
    schema = shuffle.read_schema(data_path)
    assert schema.equals(table.schema)

    # per-partition ranged reads reconstruct exactly the oracle's partitions
    total = 0
    for p in range(k):
        batches = shuffle.read_partition(data_path, index_path, p)
        got = pa.Table.from_batches(batches, schema=schema) if batches else \
            table.schema.empty_table()
        rows = idx[offs[p]:offs[p + 1]]
        want = table.take(pa.array(rows, type=pa.uint32()))
        assert got.equals(want), f"partition {p} mismatch"
        total += got.num_rows
    assert total == n

    # stats row conservation (ShuffleWritePartition counts, proto :779-791)
    assert sum(s[2] for s in stats) == n


def test_multi_input_concatenated_streams(tmp_work):
    """A partition range holding SEVERAL complete IPC streams (one per input
    partition) must read back transparently (multi_stream_reader.rs:17-34,
    writer.rs:861-884 concatenates inputs verbatim)."""
    k = 2
    t1 = make_table(300, seed=1)
    t2 = make_table(400, seed=2)
    streams = []
    for p in range(k):
        per_input = []
        for t in (t1, t2):
            n = t.num_rows
            h = oracle.hash_columns([("i64", t.column("k").to_numpy())], n)
            pids = oracle.partition_ids(h, k)
            rows = np.nonzero(pids == p)[0].astype(np.uint32)
            part = t.take(pa.array(rows, type=pa.uint32()))
            per_input.append(shuffle.encode_partition_stream(
                shuffle.rechunk(part, 128), t.schema))
        streams.append(per_input)
    data_path, index_path, stats = shuffle.write_task_consolidated(
        tmp_work, "job-2", 1, 7, t1.schema, streams)
    for p in range(k):
        batches = shuffle.read_partition(data_path, index_path, p)
        got = pa.Table.from_batches(batches, schema=t1.schema)
        want_rows = 0
        for t in (t1, t2):
            n = t.num_rows
            h = oracle.hash_columns([("i64", t.column("k").to_numpy())], n)
            pids = oracle.partition_ids(h, k)
            want_rows += int((pids == p).sum())
        assert got.num_rows == want_rows


def test_walk_partition_batches_multi_input():
    """Regression: the device-read metadata walk must
    cross the EOS marker between concatenated sub-streams — a partition
    range holding one complete IPC stream PER INPUT (writer.rs:861-884)
    must surface every batch, not just the first stream's."""
    schema = pa.schema([("k", pa.int64()), ("v", pa.int64()),
                       ("d", pa.int32())])
    # three inputs -> three complete IPC streams concatenated verbatim
    parts = []
    want_rows = []
    for seed, n in ((1, 300), (2, 257), (3, 129)):
        t = make_table(n, seed=seed)
        parts.append(shuffle.encode_partition_stream(
            shuffle.rechunk(t, 128), t.schema))
        want_rows += [128] * (n // 128) + ([n % 128] if n % 128 else [])
    raw = b"".join(parts)
    batches = shuffle.walk_partition_batches(raw, schema)
    assert [b[0] for b in batches] == want_rows
    assert sum(b[0] for b in batches) == 300 + 257 + 129
    # every parsed frame must carry a plausible (off, len, usize) triple
    for n_rows, cols in batches:
        for ci, c in enumerate(cols):
            assert c is not None
            foff, flen, usize = c
            esz = 4 if ci == 2 else 8
            assert usize == n_rows * esz
            assert 0 < foff < len(raw) and 0 < flen <= len(raw) - foff


def test_empty_partitions(tmp_work):
    """Empty partitions encode as zero bytes; the reader returns no batches
    and the schema survives via the header stream (writer.rs:838-846)."""
    table = make_table(10)
    streams = [[shuffle.encode_partition_stream(
        shuffle.rechunk(table, 8192), table.schema)]] + [[b""]] * 3
    data_path, index_path, _ = shuffle.write_task_consolidated(
        tmp_work, "job-3", 0, 0, table.schema, streams)
    offsets = shuffle.read_index(index_path)
    assert offsets[1] == offsets[2] == offsets[3]
    assert shuffle.read_partition(data_path, index_path, 2) == []
    assert shuffle.read_schema(data_path).equals(table.schema)


def test_lz4_frame_batches(tmp_work):
    """Batches are LZ4_FRAME-compressed (Ballista's default shuffle codec,
    config.rs:413-415): the IPC stream must carry the LZ4_FRAME body
    compression codec marker."""
    table = make_table(1000)
    enc = shuffle.encode_partition_stream(shuffle.rechunk(table, 8192),
                                          table.schema)
    # decode must succeed and match
    got = pa.ipc.open_stream(pa.BufferReader(enc)).read_all()
    assert got.equals(table)
    # a compressed stream of low-entropy data is smaller than raw
    raw_sink_len = sum(
        b.get_total_buffer_size() for b in shuffle.rechunk(table, 8192))
    assert len(enc) < raw_sink_len + 1000


def test_passthrough_writer_roundtrip(tmp_work):
    """Passthrough ShuffleWriterExec format: one LZ4 IPC file per global
    partition at {work}/{job}/{stage}/{global}/data-{task}.arrow
    (shuffle_writer.rs:528-622)."""
    table = make_table(3000, seed=9)
    batches = shuffle.rechunk(table, 512)
    path, nb, nr, nbytes = shuffle.write_passthrough_partition(
        tmp_work, "job-p", 4, 7, 3, table.schema, batches)
    assert path == os.path.join(tmp_work, "job-p", "4", "7", "data-3.arrow")
    assert nr == 3000 and nb == len(batches)
    got = pa.Table.from_batches(shuffle.read_passthrough_partition(path),
                                schema=table.schema)
    assert got.equals(table)


def test_global_partition_map():
    """Local->global mapping rules (shuffle_writer.rs:76-108):
    Collapsed -> 0; KSpace -> identity; PassThrough -> slice (identity
    fallback past the slice)."""
    m = shuffle.GlobalPartitionMap(shuffle.GlobalPartitionMap.COLLAPSED)
    assert [m.resolve(i) for i in range(3)] == [0, 0, 0]
    m = shuffle.GlobalPartitionMap(shuffle.GlobalPartitionMap.KSPACE)
    assert [m.resolve(i) for i in range(3)] == [0, 1, 2]
    m = shuffle.GlobalPartitionMap(shuffle.GlobalPartitionMap.PASSTHROUGH,
                                   [5, 9])
    assert [m.resolve(i) for i in range(3)] == [5, 9, 2]


def test_handwritten_ipc_writer_roundtrip():
    """The minimal flatbuffers IPC writer (datafusion_ballista_amd.ipc) must
    be readable by Arrow's own reader with CPU-compressed bodies (the GPU
    path swaps in device-compressed frames byte-for-byte)."""
    import struct
    from datafusion_ballista_amd import ipc as bgipc

    def cpu_part(data: bytes) -> bytes:
        comp = pa.compress(data, codec="lz4", asbytes=True)
        return struct.pack("<q", len(data)) + comp

    rng = np.random.default_rng(13)
    n = 10_000
    a = rng.integers(-2**60, 2**60, size=n, dtype=np.int64)
    b = rng.integers(-10**9, 10**9, size=n, dtype=np.int32)
    schema = pa.schema([("a", pa.int64()), ("b", pa.int32())])
    stream = bgipc.stream_from_compressed_batches(schema, [
        (n, [(n, 0), (n, 0)],
         [None, cpu_part(a.tobytes()), None, cpu_part(b.tobytes())]),
        (n, [(n, 0), (n, 0)],
         [None, cpu_part(a.tobytes()), None, cpu_part(b.tobytes())]),
    ])
    got = pa.ipc.open_stream(pa.BufferReader(stream)).read_all()
    one = pa.table({"a": a, "b": b})
    want = pa.concat_tables([one, one])
    assert got.equals(want)


def test_handwritten_ipc_writer_fuzz():
    """Host fuzz of the handwritten flatbuffers IPC writer (ipc.py):
    random schemas/batch shapes, buffer parts LZ4-framed with pyarrow's
    own codec, the stream decoded by pyarrow's reader — every value and
    null must round-trip (validates the metadata builder byte-for-byte
    against the arrow implementation the reference's readers use)."""
    import numpy as np
    import pyarrow as pa

    from datafusion_ballista_amd import ipc as bgipc

    lz4 = pa.Codec("lz4")
    rng = np.random.default_rng(17)

    def part(raw: bytes):
        import struct
        return struct.pack("<q", len(raw)) + lz4.compress(raw).to_pybytes()

    for trial in range(25):
        n = int(rng.integers(1, 5_000))
        fields, arrays = [], []
        nodes, parts = [], []
        ncols = int(rng.integers(1, 5))
        for c in range(ncols):
            kind = rng.choice(["i64", "i32", "f64", "str"])
            mask = rng.random(n) < rng.choice([0.0, 0.3])
            nulls = int(mask.sum())
            vb = None
            if nulls:
                vb = part(np.packbits(~mask, bitorder="little").tobytes())
            if kind == "str":
                strs = [b"" if m else b"s%d" % (i % 977)
                        for i, m in enumerate(mask)]
                offs = np.zeros(n + 1, dtype=np.int32)
                for i, b in enumerate(strs):
                    offs[i + 1] = offs[i] + len(b)
                fields.append(pa.field(f"c{c}", pa.string(),
                                       nullable=True))
                arrays.append(pa.array(
                    [None if m else s.decode()
                     for s, m in zip(strs, mask)], type=pa.string()))
                nodes.append((n, nulls))
                parts += [vb, part(offs.tobytes()),
                          part(b"".join(strs))]
            else:
                npdt = {"i64": np.int64, "i32": np.int32,
                        "f64": np.float64}[kind]
                vals = (rng.integers(-10**9, 10**9, n).astype(npdt)
                        if kind != "f64" else rng.standard_normal(n))
                pat = {"i64": pa.int64(), "i32": pa.int32(),
                       "f64": pa.float64()}[kind]
                fields.append(pa.field(f"c{c}", pat, nullable=True))
                arrays.append(pa.array(
                    [None if m else v.item()
                     for v, m in zip(vals, mask)], type=pat))
                nodes.append((n, nulls))
                parts += [vb, part(vals.tobytes())]
        schema = pa.schema(fields)
        stream = bgipc.stream_from_compressed_batches(
            schema, [(n, nodes, parts)])
        got = pa.ipc.open_stream(stream).read_all()
        want = pa.table(dict(zip([f.name for f in fields], arrays)),
                        schema=schema)
        assert got.equals(want), trial
