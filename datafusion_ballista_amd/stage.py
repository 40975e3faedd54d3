"""Python binding for the C++ stage interpreter (bg_execute_stage).

The interpreter is the product path: one call per task, the whole stage
(scan -> filter/join/aggregate/sort -> shuffle write) sequenced inside
libballista_gpu.so (csrc/stage.cpp), mirroring
QueryStageExecutor::execute_query_stage
(ballista/executor/src/execution_engine.rs:78-103).  This module only
builds plan JSON, supplies the Arrow schema-message bytes (as the Rust
host would with arrow-ipc), and parses the result JSON — no per-batch
Python in the execution path.
"""

import ctypes
import io
import json

import pyarrow as pa

from . import gpu


def schema_msg_hex(schema: pa.Schema) -> str:
    """The Arrow IPC schema message bytes (continuation+len+flatbuffer),
    hex-encoded for the plan document.  The host supplies these exactly as
    arrow-rs / pyarrow serialise them, so field metadata in the shuffle
    files stays byte-identical to the reference writers."""
    sink = io.BytesIO()
    with pa.ipc.new_stream(sink, schema):
        pass
    raw = sink.getvalue()
    assert raw.endswith(b"\xff\xff\xff\xff\x00\x00\x00\x00")
    return raw[:-8].hex()


def field_json(name: str, t: pa.DataType) -> dict:
    if pa.types.is_decimal128(t):
        return {"name": name, "dtype": "decimal128",
                "precision": t.precision, "scale": t.scale}
    m = {pa.int32(): "int32", pa.int64(): "int64", pa.date32(): "date32",
         pa.float64(): "float64", pa.uint8(): "dict8", pa.string(): "utf8"}
    return {"name": name, "dtype": m[t]}


def schema_json(schema: pa.Schema) -> list:
    return [field_json(n, t) for n, t in zip(schema.names, schema.types)]


def _lib():
    L = gpu.load_library()
    L.bg_execute_stage.argtypes = [ctypes.c_char_p,
                                   ctypes.POINTER(ctypes.c_char_p)]
    L.bg_stage_validate.argtypes = [ctypes.c_char_p,
                                    ctypes.POINTER(ctypes.c_char_p)]
    L.bg_stage_free.argtypes = [ctypes.c_char_p]
    L.bg_debug_rb_message.argtypes = [
        ctypes.c_int64, ctypes.POINTER(ctypes.c_int64), ctypes.c_int32,
        ctypes.POINTER(ctypes.c_int64), ctypes.c_int32, ctypes.c_int64,
        ctypes.c_int32, ctypes.POINTER(ctypes.c_char_p)]
    return L


def validate(plan: dict) -> dict:
    L = _lib()
    out = ctypes.c_char_p()
    rc = L.bg_stage_validate(json.dumps(plan).encode(), ctypes.byref(out))
    if rc != 0:
        raise RuntimeError(
            f"bg_stage_validate rc={rc}: {L.bg_last_error().decode()}")
    res = json.loads(out.value.decode())
    return res


def validate_error(plan: dict) -> str:
    """Returns the error message a bad plan produces (tests)."""
    L = _lib()
    out = ctypes.c_char_p()
    rc = L.bg_stage_validate(json.dumps(plan).encode(), ctypes.byref(out))
    if rc == 0:
        raise AssertionError("plan unexpectedly validated")
    return L.bg_last_error().decode()


def execute(plan: dict) -> dict:
    L = _lib()
    out = ctypes.c_char_p()
    rc = L.bg_execute_stage(json.dumps(plan).encode(), ctypes.byref(out))
    if rc != 0:
        raise RuntimeError(
            f"bg_execute_stage rc={rc}: {L.bg_last_error().decode()}")
    return json.loads(out.value.decode())


def register_table(ctx: "gpu.GpuStageContext", name: str, table: pa.Table):
    """Upload a pyarrow table and register it as a 'device' scan source.
    Returns the keep-alive handle (buffers must outlive the stages)."""
    import numpy as np
    from . import engine as eng

    table = table.combine_chunks()
    n = table.num_rows
    cols, names, keep = [], [], []
    for i in range(table.num_columns):
        arr = table.column(i).combine_chunks()
        if isinstance(arr, pa.ChunkedArray):
            arr = arr.chunk(0) if arr.num_chunks else pa.array(
                [], type=arr.type)
        vbuf = None
        if arr.null_count:
            vbool = np.asarray(arr.is_valid())
            vbuf = ctx.upload(np.packbits(vbool, bitorder="little"))
            keep.append(vbuf)
        if pa.types.is_string(arr.type):
            offs = np.frombuffer(arr.buffers()[1], dtype=np.int32,
                                 count=n + 1 + arr.offset)[arr.offset:]
            base = int(offs[0])
            nbytes = int(offs[-1]) - base
            data_np = (np.frombuffer(arr.buffers()[2], dtype=np.uint8,
                                     count=nbytes, offset=base)
                       if nbytes else np.zeros(1, dtype=np.uint8))
            dbuf = ctx.upload(data_np)
            obuf = ctx.upload((offs - base).astype(np.int32))
            keep += [dbuf, obuf]
            cols.append(ctx.column(gpu.BG_DT_UTF8, dbuf, n, offsets=obuf,
                                   validity=vbuf))
        else:
            raw = eng._col_raw(arr)
            buf = ctx.upload(raw)
            keep.append(buf)
            t = arr.type
            col = ctx.column(eng._bg_dtype(t), buf, n, validity=vbuf)
            if pa.types.is_decimal128(t):
                col.precision, col.scale = t.precision, t.scale
            cols.append(col)
        names.append(table.schema.names[i])

    L = gpu.load_library()
    carr = (gpu.BgColumn * len(cols))(*cols)
    narr = (ctypes.c_char_p * len(names))(*[s.encode() for s in names])
    gpu._check(L.bg_stage_register_table(name.encode(), carr, narr,
                                         len(cols), ctypes.c_int64(n)),
               "bg_stage_register_table")
    return keep


def unregister_table(name: str):
    gpu.load_library().bg_stage_unregister_table(name.encode())


def parquet_source(path: str) -> dict:
    """Build a scan{kind:"parquet"} source dict from the file's footer
    (pyarrow parses footers natively; the Rust host uses parquet-rs the
    same way).  The C++ scan walks page headers and decodes on device;
    columns outside its subset fail loudly at execute time."""
    import pyarrow.parquet as pq
    pf = pq.ParquetFile(path)
    md = pf.metadata
    cols = []
    for c in range(md.num_columns):
        sc = pf.schema.column(c)
        phys = md.row_group(0).column(c).physical_type
        spec = {"phys": "FLBA" if phys == "FIXED_LEN_BYTE_ARRAY" else phys,
                "max_def": sc.max_definition_level,
                "codec": md.row_group(0).column(c).compression,
                "chunks": []}
        if spec["phys"] == "FLBA":
            spec["flba_len"] = sc.length
        for rg in range(md.num_row_groups):
            m = md.row_group(rg).column(c)
            start = m.data_page_offset
            if m.has_dictionary_page and m.dictionary_page_offset is not None:
                start = min(start, m.dictionary_page_offset)
            spec["chunks"].append({"start": start,
                                   "size": m.total_compressed_size,
                                   "num_values": m.num_values})
        cols.append(spec)
    return {"kind": "parquet", "path": path, "columns": cols}


def debug_rb_message(n_rows, nodes, bufs, body_len, compressed) -> bytes:
    L = _lib()
    nd = (ctypes.c_int64 * (2 * len(nodes)))(
        *[x for p in nodes for x in p])
    bf = (ctypes.c_int64 * (2 * len(bufs)))(*[x for p in bufs for x in p])
    out = ctypes.c_char_p()
    rc = L.bg_debug_rb_message(n_rows, nd, len(nodes), bf, len(bufs),
                               body_len, 1 if compressed else 0,
                               ctypes.byref(out))
    if rc != 0:
        raise RuntimeError(L.bg_last_error().decode())
    return bytes.fromhex(out.value.decode())
