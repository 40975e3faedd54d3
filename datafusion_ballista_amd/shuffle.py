"""Sort-shuffle file format, byte-compatible with the reference.

Restates (reference citations, /root/reference):
  - consolidated data file layout — write_task_consolidated,
    ballista/core/src/execution_plans/sort_shuffle/writer.rs:794-895:
    `{work_dir}/{job_id}/{stage_id}/{file_id}/data.arrow` =
    leading schema-only Arrow IPC stream (schema message + EOS, no batches),
    then K partition-major byte ranges, each a concatenation of zero or more
    COMPLETE Arrow IPC streams (batches LZ4_FRAME-compressed; codec default
    `lz4`, core/src/config.rs:413-415).
  - index file — sort_shuffle/index.rs:21-33:
    `data.arrow.index` = (K+1) little-endian i64 absolute offsets; offset_i =
    absolute byte where partition i starts; last entry = total file length.
  - readers cross sub-stream boundaries transparently —
    sort_shuffle/multi_stream_reader.rs:17-34; a reader fetches partition p
    with one ranged read [offset_p, offset_{p+1}) (shuffle_reader.rs:1120-1168).
  - batches re-chunked to `batch_size` (default 8192,
    sort_shuffle/config.rs:35-43).

File IO + IPC framing stay host-side by design ("file
write stays CPU/IO"); the GPU produces the partition-major column buffers.
"""

import io
import os
import struct

import pyarrow as pa

DEFAULT_BATCH_SIZE = 8192  # sort_shuffle/config.rs:35-43


def ipc_write_options():
    # LZ4_FRAME body-buffer compression = Ballista's default shuffle codec
    # (config.rs:413-415, codec map :729-742).
    return pa.ipc.IpcWriteOptions(compression="lz4")


def encode_partition_stream(batches, schema) -> bytes:
    """One complete IPC stream (schema .. batches .. EOS) for one partition's
    in-memory rows — encode_buffered_partitions (writer.rs:121-164). Empty
    partitions encode to b'' (the reference skips the writer entirely)."""
    if not batches or all(b.num_rows == 0 for b in batches):
        return b""
    sink = io.BytesIO()
    with pa.ipc.new_stream(sink, schema, options=ipc_write_options()) as w:
        for b in batches:
            if b.num_rows:
                w.write_batch(b)
    return sink.getvalue()


def rechunk(table_like, batch_size=DEFAULT_BATCH_SIZE):
    """Slice a pyarrow Table/RecordBatch into <=batch_size record batches
    (PartitionedBatchIterator materialises interleaved rows in batch_size
    chunks — partitioned_batch_iterator.rs)."""
    if isinstance(table_like, pa.RecordBatch):
        table_like = pa.Table.from_batches([table_like])
    out = []
    n = table_like.num_rows
    for s in range(0, n, batch_size):
        chunk = table_like.slice(s, min(batch_size, n - s))
        out.extend(chunk.combine_chunks().to_batches())
    return out


def task_output_dir(work_dir, job_id, stage_id, file_id) -> str:
    # write_task_consolidated path layout (writer.rs:821-828)
    return os.path.join(work_dir, str(job_id), str(stage_id), str(file_id))


def write_task_consolidated(work_dir, job_id, stage_id, file_id, schema,
                            partition_streams):
    """partition_streams: list (len K) of lists of encoded IPC stream bytes
    (one per input partition, concatenated verbatim — writer.rs:861-884).

    Returns (data_path, index_path, [(partition_id, num_batches, num_rows,
    num_bytes)]) with stats counted like EncodedStats (writer.rs:778-844);
    num_rows/num_batches must be supplied via the streams' metadata, so this
    function re-parses each stream cheaply for counts (host-side, test/IO
    path only)."""
    out_dir = task_output_dir(work_dir, job_id, stage_id, file_id)
    os.makedirs(out_dir, exist_ok=True)
    data_path = os.path.join(out_dir, "data.arrow")
    index_path = os.path.join(out_dir, "data.arrow.index")

    k = len(partition_streams)
    offsets = [0] * (k + 1)
    stats = []
    with open(data_path, "wb") as f:
        # leading schema-only stream so a reader of an empty partition can
        # still recover the schema (writer.rs:838-846)
        sink = io.BytesIO()
        with pa.ipc.new_stream(sink, schema, options=ipc_write_options()):
            pass
        f.write(sink.getvalue())
        for p in range(k):
            offsets[p] = f.tell()
            num_batches = num_rows = num_bytes = 0
            for stream in partition_streams[p]:
                if not stream:
                    continue
                f.write(stream)
                rd = pa.ipc.open_stream(pa.BufferReader(stream))
                for b in rd:
                    num_batches += 1
                    num_rows += b.num_rows
                    num_bytes += b.get_total_buffer_size()
            stats.append((p, num_batches, num_rows, num_bytes))
        offsets[k] = f.tell()

    write_index(index_path, offsets)
    return data_path, index_path, stats


def write_index(index_path, offsets):
    # (K+1) little-endian i64 (index.rs:21-33)
    with open(index_path, "wb") as f:
        f.write(struct.pack(f"<{len(offsets)}q", *offsets))


def read_index(index_path):
    raw = open(index_path, "rb").read()
    n = len(raw) // 8
    return list(struct.unpack(f"<{n}q", raw))


def read_partition(data_path, index_path, partition_id):
    """Ranged read of one partition; crosses concatenated IPC sub-stream
    boundaries transparently (multi_stream_reader.rs:17-34).
    Returns a list of RecordBatches (possibly empty)."""
    offsets = read_index(index_path)
    lo = offsets[partition_id]
    hi = offsets[partition_id + 1]
    with open(data_path, "rb") as f:
        f.seek(lo)
        raw = f.read(hi - lo)
    batches = []
    buf = pa.BufferReader(raw)
    while buf.tell() < len(raw):
        sub = pa.BufferReader(raw[buf.tell():])
        rd = pa.ipc.open_stream(sub)
        for b in rd:
            batches.append(b)
        buf.seek(buf.tell() + sub.tell())
    return batches


def read_schema(data_path):
    """Recover the schema from the leading schema-only stream."""
    with open(data_path, "rb") as f:
        raw = f.read(1 << 16)
    rd = pa.ipc.open_stream(pa.BufferReader(raw))
    return rd.schema


# ---------------------------------------------------------------------------
# Passthrough shuffle writer (ShuffleWriterExec, no repartition):
# K child partitions -> K files `{work_dir}/{job}/{stage}/{global_partition}/
# data-{task_id}.arrow`, each one complete LZ4_FRAME IPC stream
# (shuffle_writer.rs:528-622; path layout core/src/execution_plans/
# mod.rs:95-127), with the local->global partition-id mapping of
# walk_child_partition_mapping (shuffle_writer.rs:76-165).
# ---------------------------------------------------------------------------

class GlobalPartitionMap:
    """Local output index -> global partition id (shuffle_writer.rs:76-108):
    COLLAPSED: every local -> 0 (SortPreservingMerge fan-in)
    KSPACE: local == global (fresh K-space: hash/round-robin/range repart)
    PASSTHROUGH: local i -> slice[i] (scheduler-stamped ids; identity when
    the slice is empty)."""
    COLLAPSED = "collapsed"
    KSPACE = "kspace"
    PASSTHROUGH = "passthrough"

    def __init__(self, kind, slice_ids=None):
        self.kind = kind
        self.slice_ids = slice_ids or []

    def resolve(self, local: int) -> int:
        if self.kind == self.COLLAPSED:
            return 0
        if self.kind == self.KSPACE:
            return local
        if local < len(self.slice_ids):
            return int(self.slice_ids[local])
        return local


def passthrough_partition_path(work_dir, job_id, stage_id, global_partition,
                               task_id) -> str:
    return os.path.join(work_dir, str(job_id), str(stage_id),
                        str(global_partition), f"data-{task_id}.arrow")


def write_passthrough_partition(work_dir, job_id, stage_id, global_partition,
                                task_id, schema, batches):
    """One local output partition -> one LZ4_FRAME IPC file.  Returns
    (path, num_batches, num_rows, num_bytes) — the ShuffleWritePartition
    fields (proto :779-791)."""
    path = passthrough_partition_path(work_dir, job_id, stage_id,
                                      global_partition, task_id)
    os.makedirs(os.path.dirname(path), exist_ok=True)
    num_batches = num_rows = num_bytes = 0
    with open(path, "wb") as f:
        with pa.ipc.new_stream(f, schema, options=ipc_write_options()) as w:
            for b in batches:
                if b.num_rows == 0:
                    continue
                w.write_batch(b)
                num_batches += 1
                num_rows += b.num_rows
                num_bytes += b.get_total_buffer_size()
    return path, num_batches, num_rows, num_bytes


def read_passthrough_partition(path):
    """ShuffleReaderExec local read of a passthrough file
    (shuffle_reader.rs:1120-1168, plain single-stream case)."""
    with open(path, "rb") as f:
        raw = f.read()
    return list(pa.ipc.open_stream(pa.BufferReader(raw)))


def walk_partition_batches(raw, schema):
    """Parse one partition's byte range — a concatenation of COMPLETE IPC
    sub-streams (one per input, writer.rs:861-884; crossed transparently
    like multi_stream_reader.rs:17-34) — into
    [(n_rows, per-col (frame_off_in_raw, frame_len, usize) | None)].
    Pure host logic (no GPU), so the sub-stream crossing is unit-testable."""
    from . import ipc as bgipc

    batches = []
    pos = 0
    while pos < len(raw):
        consumed = pos
        for msg, body_off, body_len in bgipc.walk_stream(raw[pos:]):
            if msg is None:
                consumed = pos + body_off
                continue
            n_rows, bufs, compressed = msg
            if not compressed:
                raise RuntimeError("uncompressed batches: host path")
            cols = []
            for ci in range(len(schema.types)):
                voff_, vlen_ = bufs[2 * ci]
                if vlen_:
                    # null-bearing batches stay on the host read path
                    # (read_partition), which reassembles validity exactly;
                    # silent null-dropping here would be a parity hole
                    raise RuntimeError(
                        "read_partition_gpu: null-carrying column "
                        f"{ci}: use read_partition (host) for this stream")
                boff, blen = bufs[2 * ci + 1]  # data buffer (validity at 2ci)
                if blen == 0:
                    cols.append(None)
                    continue
                abs_off = pos + body_off + boff
                usize = int.from_bytes(raw[abs_off:abs_off + 8], "little")
                cols.append((abs_off + 8, blen - 8, usize))
            batches.append((n_rows, cols))
            consumed = pos + body_off + body_len
        if consumed == pos:
            break
        # walk_stream ends at a sub-stream's EOS marker without consuming
        # it; skip it so the NEXT concatenated sub-stream is parsed too
        # (multi-input consolidated partitions, writer.rs:861-884)
        if raw[consumed:consumed + 8] == bgipc.EOS:
            consumed += 8
        pos = consumed
    return batches


def read_partition_gpu(ctx, data_path, index_path, partition_id, schema):
    """Device shuffle-read: decode a partition's LZ4-compressed batches
    straight into HBM column buffers (ShuffleReaderExec local-read
    equivalent for a GPU-resident next stage).  Fixed-width columns, null-
    free batches.  Returns (n_rows, {col_idx: DeviceBuffer}) with columns
    concatenated across the partition's batches."""
    import ctypes

    import numpy as np

    from . import gpu as g

    offsets = read_index(index_path)
    lo, hi = offsets[partition_id], offsets[partition_id + 1]
    with open(data_path, "rb") as f:
        f.seek(lo)
        raw = f.read(hi - lo)
    if not raw:
        return 0, {}

    esz = [16 if pa.types.is_decimal128(t) else t.bit_width // 8
           for t in schema.types]
    batches = walk_partition_batches(raw, schema)

    total = sum(b[0] for b in batches)
    raw_buf = ctx.upload(np.frombuffer(raw, dtype=np.uint8))
    out = {ci: ctx.alloc(max(total * esz[ci], esz[ci]))
           for ci in range(len(schema.types))}
    frames = []
    expected = []
    row_cursor = 0
    for n_rows, cols in batches:
        for ci, c in enumerate(cols):
            if c is None:
                continue
            foff, flen, usize = c
            dst = ctypes.c_void_p(out[ci].ptr.value + row_cursor * esz[ci])
            frames.append((ctypes.c_void_p(raw_buf.ptr.value + foff), flen,
                           dst, usize))
            expected.append(usize)
        row_cursor += n_rows
    if frames:
        lens = ctx.lz4_decompress(frames)
        for i, ln in enumerate(lens):
            if ln != expected[i]:
                raise RuntimeError(f"frame {i} decode failed ({ln})")
    ctx.synchronize()
    return total, out
