"""GPU query-stage executor — host mirror of Ballista's QueryStageExecutor
for the hash-repartition (sort-shuffle) stage.

Mirrors (reference citations, /root/reference):
  - ExecutionEngine::create_query_stage_exec /
    QueryStageExecutor::execute_query_stage —
    ballista/executor/src/execution_engine.rs:53-103 (the plugin seam the
    Rust binding in INTEGRATION.md drops this engine into).
  - SortShuffleWriterExec::execute_shuffle_write (writer.rs:564-753):
    per-input bucket rows by hash partition -> materialise per-partition
    batches -> encode IPC+LZ4 -> consolidate partition-major, one file per
    task + index; summaries returned as ShuffleWritePartition
    (proto ballista.proto:779-791: partition_id, path, num_batches,
    num_rows, num_bytes).

Device work (hash, partition split, gather) runs through libballista_gpu.so;
IPC encode + file write stay host-side.
"""

import ctypes as _ct
import time
from dataclasses import dataclass

import numpy as np
import pyarrow as pa

from . import gpu
from . import ipc as bgipc
from . import shuffle


@dataclass
class ShuffleWritePartition:
    """ShuffleWritePartition proto restatement (ballista.proto:779-791)."""
    partition_id: int
    path: str
    num_batches: int
    num_rows: int
    num_bytes: int


_PA_TO_BG = {
    pa.int32(): gpu.BG_DT_INT32,
    pa.int64(): gpu.BG_DT_INT64,
    pa.date32(): gpu.BG_DT_DATE32,
    pa.uint8(): gpu.BG_DT_DICT8,
    pa.float64(): gpu.BG_DT_FLOAT64,
}


def _bg_dtype(t: pa.DataType) -> int:
    if pa.types.is_decimal128(t):
        return gpu.BG_DT_DECIMAL128
    if pa.types.is_string(t):
        return gpu.BG_DT_UTF8
    try:
        return _PA_TO_BG[t]
    except KeyError:
        raise RuntimeError(f"dtype {t} not yet accelerated (BG_ERR_UNSUPPORTED)")


def _np_for(t: pa.DataType):
    if pa.types.is_decimal128(t):
        return np.uint8  # 16 B/elem raw
    return {pa.int32(): np.int32, pa.int64(): np.int64,
            pa.date32(): np.int32, pa.uint8(): np.uint8,
            pa.float64(): np.float64}[t]


def _col_raw(arr: pa.Array) -> np.ndarray:
    """Arrow array -> contiguous primitive numpy view (values buffer only;
    validity rides separately — see _repartition_one_input)."""
    t = arr.type
    if pa.types.is_decimal128(t):
        buf = arr.buffers()[1]
        off = arr.offset * 16
        return np.frombuffer(buf, dtype=np.uint8,
                             count=len(arr) * 16, offset=off)
    np_t = _np_for(t)
    buf = arr.buffers()[1]
    return np.frombuffer(buf, dtype=np_t, count=len(arr), offset=arr.offset *
                         np.dtype(np_t).itemsize)


def _array_from_raw(t: pa.DataType, raw: np.ndarray, n: int,
                    valid_bits: bytes = None) -> pa.Array:
    vb = pa.py_buffer(valid_bits) if valid_bits is not None else None
    return pa.Array.from_buffers(t, n, [vb, pa.py_buffer(raw.tobytes())])


class GpuQueryStageExecutor:
    """Executes one sort-shuffle task: hash-repartition `batch`'s rows into
    k partitions on the GPU, write the consolidated shuffle file, return
    ShuffleWritePartition summaries — the GPU implementation of
    execute_query_stage for a SortShuffleWriterExec-rooted stage."""

    def __init__(self, ctx: "gpu.GpuStageContext", job_id: str, stage_id: int,
                 work_dir: str, key_columns, num_partitions: int,
                 batch_size: int = shuffle.DEFAULT_BATCH_SIZE,
                 gpu_codec: bool = False):
        self.ctx = ctx
        self.job_id = job_id
        self.stage_id = stage_id
        self.work_dir = work_dir
        self.key_columns = key_columns  # column indices (already-evaluated keys)
        self.k = num_partitions
        self.batch_size = batch_size
        # gpu_codec: LZ4-compress the shuffle batch bodies ON DEVICE
        # (bg_lz4_compress) and emit IPC metadata with the handwritten
        # writer (ipc.py) — removes the host LZ4+encode cost (SURVEY.md
        # §8f row 3).  Fixed-width columns; one batch per partition.
        self.gpu_codec = gpu_codec
        self._metrics = {}

    def collect_plan_metrics(self):
        """Metric names mirror the reference writer's MetricsSet
        (sort_shuffle/writer.rs:328-440: write_time/repart_time/...) plus
        the GPU counters SURVEY.md §5 calls for, so EXPLAIN ANALYZE keeps
        rendering per-operator numbers."""
        return [dict(self._metrics)]

    def execute_query_stage(self, task_id: int, table: pa.Table):
        """Single- or multi-input task: a list of tables = the task's M
        local input partitions; each is bucketed on the GPU independently
        and its encoded streams are concatenated per output partition in
        the consolidated file, exactly like the reference's per-input
        tasks + coordinator concat (writer.rs:564-753, 861-884)."""
        tables = table if isinstance(table, (list, tuple)) else [table]
        schema = tables[0].schema
        per_input_streams = []  # [input][partition] -> bytes
        repart_time_s = 0.0
        total_rows = 0
        for t_in in tables:
            streams, dt = self._repartition_one_input(t_in)
            per_input_streams.append(streams)
            repart_time_s += dt
            total_rows += t_in.num_rows

        # transpose to [partition][input] for the consolidated writer
        partition_streams = [
            [per_input_streams[i][p] for i in range(len(tables))]
            for p in range(self.k)]

        t_write = time.perf_counter()
        data_path, index_path, stats = shuffle.write_task_consolidated(
            self.work_dir, self.job_id, self.stage_id, task_id, schema,
            partition_streams)
        write_time_s = time.perf_counter() - t_write
        self._metrics = {
            # reference writer metric names (sort_shuffle/writer.rs:328-440)
            "repart_time_ns": int(repart_time_s * 1e9),
            "write_time_ns": int(write_time_s * 1e9),
            "spill_time_ns": 0,
            "spill_count": 0,
            "spilled_bytes": 0,
            "output_rows": int(sum(s_[2] for s_ in stats)),
            # GPU additions
            "gpu_kernel_time_ns": int(repart_time_s * 1e9),
            "achieved_hbm_bytes": 0,
        }
        return [ShuffleWritePartition(p, data_path, nb, nr, nbytes)
                for (p, nb, nr, nbytes) in stats]

    def _repartition_one_input(self, table: pa.Table):
        """GPU-bucket one input partition -> (per-output-partition encoded
        IPC streams, device seconds)."""
        ctx = self.ctx
        table = table.combine_chunks()
        n = table.num_rows
        schema = table.schema

        cols = []           # BgColumn per table column (fixed OR utf8)
        utf8_src = {}       # col index -> (data buf, offsets buf, total)
        null_ids = {}       # col index -> host bool validity (null cols)
        for i in range(table.num_columns):
            arr = table.column(i).combine_chunks()
            if isinstance(arr, pa.ChunkedArray):
                arr = arr.chunk(0) if arr.num_chunks else pa.array(
                    [], type=arr.type)
            t = arr.type
            vbuf = None
            if arr.null_count:
                # Arrow LSB validity (rebased to offset 0); rows with NULL
                # keys hash as no-contribution (hash_utils create_hashes)
                vbool = np.asarray(arr.is_valid())
                vbuf = ctx.upload(np.packbits(vbool, bitorder="little"))
                null_ids[i] = vbool
            if pa.types.is_string(t):
                offs = np.frombuffer(arr.buffers()[1], dtype=np.int32,
                                     count=n + 1 + arr.offset)[arr.offset:]
                data_buf = arr.buffers()[2]
                nbytes = int(offs[-1]) - int(offs[0])
                base = int(offs[0])
                data_np = np.frombuffer(data_buf, dtype=np.uint8,
                                        count=nbytes, offset=base)                     if nbytes else np.zeros(1, dtype=np.uint8)
                offs_rb = (offs.astype(np.int32) - base).astype(np.int32)
                dbuf = ctx.upload(data_np)
                obuf = ctx.upload(offs_rb)
                utf8_src[i] = (dbuf, obuf, nbytes)
                cols.append(ctx.column(gpu.BG_DT_UTF8, dbuf, n,
                                       offsets=obuf, validity=vbuf))
            else:
                raw = _col_raw(arr)
                bgdt = _bg_dtype(t)
                buf = ctx.upload(raw)
                cols.append(ctx.column(bgdt, buf, n, validity=vbuf))

        key_cols = [cols[i] for i in self.key_columns]
        fixed_ids = [i for i in range(table.num_columns) if i not in utf8_src]
        fixed_cols = [cols[i] for i in fixed_ids]
        t_repart = time.perf_counter()
        idx_buf, offs_buf, out_bufs = ctx.hash_repartition(
            key_cols, fixed_cols, n, self.k)
        # variable-length payload: gather through the same permutation
        utf8_out = {}
        for i, (dbuf, obuf, nbytes) in utf8_src.items():
            oo, od, tot = ctx.gather_varlen(dbuf, obuf, idx_buf, n,
                                            max_bytes=max(nbytes, 1))
            utf8_out[i] = (oo, od, tot)
        valid_out = {}
        for i in null_ids:
            ob = ctx.alloc(max(((n + 63) // 64) * 8, 8))
            gpu._check(ctx.L.bg_gather_bits(
                _ct.c_void_p(cols[i].d_validity), idx_buf.ptr,
                _ct.c_int64(n), ob.ptr), "bg_gather_bits")
            valid_out[i] = ob
        ctx.synchronize()
        dt_device = time.perf_counter() - t_repart
        offsets = offs_buf.download(np.int64, self.k + 1)

        # download partition-major buffers and slice per partition
        col_raws = {}
        for ci, b in zip(fixed_ids, out_bufs):
            esz = gpu._DT_SIZE[cols[ci].dtype]
            col_raws[ci] = b.download(np.uint8, max(n, 1) * esz)
        utf8_raws = {}
        for i, (oo, od, tot) in utf8_out.items():
            utf8_raws[i] = (oo.download(np.int32, n + 1),
                            od.download(np.uint8, max(tot, 1))[:tot])
        valid_perm = {}
        for i, ob in valid_out.items():
            words = ob.download(np.uint8, ((n + 63) // 64) * 8)
            valid_perm[i] = np.unpackbits(
                words, bitorder="little")[:n].astype(bool)

        if self.gpu_codec:
            return self._encode_partitions_gpu(
                schema, offsets, cols, out_bufs, valid_perm, utf8_out,
                utf8_raws), dt_device

        streams = []
        for p in range(self.k):
            lo, hi = int(offsets[p]), int(offsets[p + 1])
            m = hi - lo
            if m == 0:
                streams.append(b"")
                continue
            arrays = []
            for ci in range(table.num_columns):
                t = schema.types[ci]
                vb = None
                if ci in valid_perm:
                    part_bool = valid_perm[ci][lo:hi]
                    if not part_bool.all():
                        vb = np.packbits(part_bool,
                                         bitorder="little").tobytes()
                if ci in utf8_raws:
                    offs_all, data_all = utf8_raws[ci]
                    o_lo, o_hi = int(offs_all[lo]), int(offs_all[hi])
                    sub_offs = (offs_all[lo:hi + 1] - o_lo).astype(np.int32)
                    sub_data = data_all[o_lo:o_hi]
                    arrays.append(pa.Array.from_buffers(
                        pa.utf8(), m,
                        [pa.py_buffer(vb) if vb is not None else None,
                         pa.py_buffer(sub_offs.tobytes()),
                         pa.py_buffer(sub_data.tobytes())]))
                    continue
                esz = 16 if pa.types.is_decimal128(t) else \
                    np.dtype(_np_for(t)).itemsize
                raw = col_raws[ci][lo * esz: hi * esz]
                arrays.append(_array_from_raw(t, raw, m, vb))
            part_table = pa.Table.from_arrays(arrays, schema=schema)
            batches = shuffle.rechunk(part_table, self.batch_size)
            streams.append(shuffle.encode_partition_stream(batches, schema))
        return streams, dt_device


class GpuResidentShuffleStage(GpuQueryStageExecutor):
    """`ballista.gpu.resident_shuffle` mode: the task's
    partition-major column buffers STAY in HBM for an RCCL all-to-all to
    co-located executors (exchange.all_to_all_rows) instead of being
    IPC-encoded to disk; the file path remains the durability/parity
    fallback (documented trade-off of shuffle.md:66-68 — resident shuffle
    gives up recompute-from-files recovery for the exchange's speed)."""

    def execute_query_stage_resident(self, task_id: int, table: pa.Table):
        """-> (offsets np.int64[k+1], {col_idx: DeviceBuffer partition-major},
        schema).  Fixed-width columns only in resident mode (strings fall
        back to the file path)."""
        ctx = self.ctx
        table = table.combine_chunks()
        n = table.num_rows
        cols = []
        for i in range(table.num_columns):
            arr = table.column(i).combine_chunks()
            if isinstance(arr, pa.ChunkedArray):
                arr = arr.chunk(0)
            if pa.types.is_string(arr.type):
                raise RuntimeError("resident shuffle: fixed-width only (r1)")
            raw = _col_raw(arr)
            cols.append(ctx.column(_bg_dtype(arr.type), ctx.upload(raw), n))
        key_cols = [cols[i] for i in self.key_columns]
        idx_buf, offs_buf, out_bufs = ctx.hash_repartition(
            key_cols, cols, n, self.k)
        ctx.synchronize()
        offsets = offs_buf.download(np.int64, self.k + 1)
        return offsets, {i: b for i, b in enumerate(out_bufs)}, table.schema


def _encode_partitions_gpu(self, schema, offsets, cols, out_bufs,
                           valid_perm=None, utf8_out=None, utf8_raws=None):
    """Device-LZ4 every partition's buffer slices in ONE flat launch, pack
    the frame bodies on device in ONE launch, then download only the
    compressed bytes; the handwritten IPC writer supplies the metadata.
    Fixed-width columns contribute one data buffer; Utf8 columns
    contribute rebased i32 offsets (bg_sub_i32 on device) + their byte
    range; null columns ride as host-LZ4'd validity parts (1/64 of the
    data bytes, already host-resident for metrics)."""
    import ctypes
    import struct
    ctx = self.ctx
    valid_perm = valid_perm or {}
    utf8_out = utf8_out or {}
    utf8_raws = utf8_raws or {}
    lz4 = pa.Codec("lz4") if valid_perm else None
    fixed_ids = [i for i in range(len(cols)) if i not in utf8_out]
    fixed_buf = {ci: b for ci, b in zip(fixed_ids, out_bufs)}

    # device-rebase each partition's offset slice for utf8 columns
    reb = {}
    for ci, (oo, od, tot) in utf8_out.items():
        r = ctx.alloc(max(4 * (int(offsets[-1]) + self.k), 4))
        cur = 0
        regions = {}
        offs_all = utf8_raws[ci][0]
        for p in range(self.k):
            lo, hi = int(offsets[p]), int(offsets[p + 1])
            m = hi - lo
            if m == 0:
                continue
            gpu._check(ctx.L.bg_sub_i32(
                ctypes.c_void_p(oo.ptr.value + 4 * lo),
                ctypes.c_int64(m + 1), ctypes.c_int32(int(offs_all[lo])),
                ctypes.c_void_p(r.ptr.value + 4 * cur)), "bg_sub_i32")
            regions[p] = cur
            cur += m + 1
        reb[ci] = (r, regions)

    # phase 1: enumerate every (partition, column, unit, 64KiB block)
    buf_meta = []   # (p, ci, kind, length, nblocks, first_job)
    jobs = []
    slot_cursor = 0
    for p in range(self.k):
        lo, hi = int(offsets[p]), int(offsets[p + 1])
        m = hi - lo
        if m == 0:
            continue
        for ci in range(len(cols)):
            if ci in utf8_out:
                oo, od, tot = utf8_out[ci]
                offs_all = utf8_raws[ci][0]
                r, regions = reb[ci]
                o_lo, o_hi = int(offs_all[lo]), int(offs_all[hi])
                units = [("offs", r.ptr.value + 4 * regions[p],
                          4 * (m + 1)),
                         ("data", od.ptr.value + o_lo, o_hi - o_lo)]
            else:
                esz = gpu._DT_SIZE[cols[ci].dtype]
                units = [("data", fixed_buf[ci].ptr.value + lo * esz,
                          m * esz)]
            for kind, src_ptr, length in units:
                nblocks = (length + 65536 - 1) // 65536
                buf_meta.append((p, ci, kind, length, nblocks, len(jobs)))
                for i in range(nblocks):
                    blen = min(65536, length - i * 65536)
                    jobs.append((src_ptr + i * 65536, slot_cursor, blen))
                    slot_cursor += 65544
    if not jobs:
        return [b""] * self.k
    slots = ctx.alloc(max(slot_cursor, 8))
    jarr = (gpu.BgLz4BlockJob * len(jobs))()
    for i, (src, soff, blen) in enumerate(jobs):
        jarr[i] = gpu.BgLz4BlockJob(src, slots.ptr.value + soff, blen, 0)
    sizes = np.zeros(len(jobs), dtype=np.int64)
    gpu._check(ctx.L.bg_lz4_compress_flat(
        jarr, ctypes.c_int64(len(jobs)),
        sizes.ctypes.data_as(ctypes.POINTER(ctypes.c_int64))),
        "bg_lz4_compress_flat")

    # phase 2: pack each buffer's [u32 size][block] sequence contiguously
    pack_jobs = []
    packed_layout = {}  # (p, ci, kind) -> (packed_off, body_len, length)
    packed_cursor = 0
    for (p, ci, kind, length, nblocks, job0) in buf_meta:
        body = 0
        for i in range(nblocks):
            sz = int(sizes[job0 + i])
            payload = -sz if sz < 0 else sz
            body += 4 + payload
        packed_layout[(p, ci, kind)] = (packed_cursor, body, length)
        off = packed_cursor
        for i in range(nblocks):
            sz = int(sizes[job0 + i])
            blen = min(65536, length - i * 65536)
            payload = blen if sz < 0 else sz
            word = (blen | 0x80000000) if sz < 0 else sz
            pack_jobs.append((jobs[job0 + i][1], off, payload, word))
            off += 4 + payload
        packed_cursor += body
    packed = ctx.alloc(max(packed_cursor, 8))
    parr = (gpu.BgPackJob * len(pack_jobs))()
    for i, (soff, doff, payload, word) in enumerate(pack_jobs):
        parr[i] = gpu.BgPackJob(slots.ptr.value + soff,
                                packed.ptr.value + doff, payload, word, 0)
    gpu._check(ctx.L.bg_pack_blocks(parr, ctypes.c_int64(len(pack_jobs))),
               "bg_pack_blocks")

    # phase 3: one D2H of the whole packed region, then assemble streams
    packed_host = packed.download(np.uint8, packed_cursor).tobytes()

    def part_of(p, ci, kind):
        poff, body_len, length = packed_layout[(p, ci, kind)]
        frame = (gpu.LZ4_FRAME_HEADER + packed_host[poff:poff + body_len] +
                 b"\x00\x00\x00\x00")
        return struct.pack("<q", length) + frame

    streams = []
    for p in range(self.k):
        lo, hi = int(offsets[p]), int(offsets[p + 1])
        m = hi - lo
        if m == 0:
            streams.append(b"")
            continue
        nodes = []
        buffer_parts = []
        for ci in range(len(cols)):
            if ci in valid_perm:
                part_bool = valid_perm[ci][lo:hi]
                nulls = int(m - part_bool.sum())
                nodes.append((m, nulls))
                if nulls:
                    vb = np.packbits(part_bool, bitorder="little").tobytes()
                    buffer_parts.append(
                        struct.pack("<q", len(vb)) +
                        lz4.compress(vb).to_pybytes())
                else:
                    buffer_parts.append(None)
            else:
                nodes.append((m, 0))
                buffer_parts.append(None)
            if ci in utf8_out:
                buffer_parts.append(part_of(p, ci, "offs"))
                buffer_parts.append(part_of(p, ci, "data"))
            else:
                buffer_parts.append(part_of(p, ci, "data"))
        streams.append(bgipc.stream_from_compressed_batches(
            schema, [(m, nodes, buffer_parts)]))
    slots.free()
    packed.free()
    return streams


GpuQueryStageExecutor._encode_partitions_gpu = _encode_partitions_gpu
