"""Minimal Arrow IPC stream writer for device-compressed record batches
(GPU shuffle codec).

Writes the encapsulated-message framing and the RecordBatch metadata
flatbuffer by hand (restating the published Arrow format/Message.fbs:
Message{version, header union, bodyLength}, RecordBatch{length, nodes,
buffers, compression}, BodyCompression{codec=LZ4_FRAME}), so the batch BODY
can be the device-compressed buffers verbatim:
  body = concat of per-buffer [i64 uncompressed_len][LZ4 frame], 8-aligned.
The schema message is taken verbatim from pyarrow (schema-only stream
prefix), so field metadata stays byte-identical to the reference writers.

Validity is pinned in tests by round-tripping through pyarrow's own IPC
reader (the same arrow decoder family ShuffleReaderExec uses).
"""

import io
import struct

import pyarrow as pa

# ---------------------------------------------------------------------------
# tiny back-to-front flatbuffers builder (only what Message.fbs needs)
# ---------------------------------------------------------------------------


class _FB:
    def __init__(self):
        self._parts = []   # list of bytes, later parts sit closer to the END
        self._size = 0

    def _prepend(self, b: bytes):
        self._parts.append(b)
        self._size += len(b)

    def _align(self, n: int, extra: int = 0):
        # pad so that (size + extra) % n == 0 after padding
        pad = (-(self._size + extra)) % n
        if pad:
            self._prepend(b"\x00" * pad)

    def end_rel(self) -> int:
        return self._size

    def struct_vector(self, elem_bytes: list) -> int:
        """Vector of structs (inline).  Returns end_rel offset of vector."""
        body = b"".join(elem_bytes)
        self._align(8)            # struct alignment (i64 members)
        self._prepend(body)
        self._align(4, extra=4)   # count must land 4-aligned
        self._prepend(struct.pack("<I", len(elem_bytes)))
        return self._size

    def table(self, fields) -> int:
        """fields: list of (field_id, kind, value) with kind in
        {'i16','i64','u8','ref'}; refs are end_rel offsets of targets.
        Returns end_rel offset of the table."""
        # lay out inline field area (sorted by size desc for alignment
        # simplicity: i64 first, then refs (u32), i16, u8)
        order = sorted(fields, key=lambda f: {"i64": 0, "ref": 1, "i16": 2,
                                              "u8": 3}[f[1]])
        # compute field positions relative to table start (after soffset i32)
        pos = 4
        placed = []
        for fid, kind, val in order:
            sz = {"i64": 8, "ref": 4, "i16": 2, "u8": 1}[kind]
            pad = (-pos) % sz
            pos += pad
            placed.append((fid, kind, val, pos, pad))
            pos += sz
        table_len = pos
        pad_tail = (-table_len) % 4
        table_len += pad_tail

        max_fid = max(f[0] for f in fields) if fields else -1
        vt_len = 4 + 2 * (max_fid + 1)
        vslots = [0] * (max_fid + 1)
        for fid, kind, val, p, pad in placed:
            vslots[fid] = p
        vtable = struct.pack("<HH", vt_len, table_len) + \
            b"".join(struct.pack("<H", s) for s in vslots)

        # write table: soffset + inline fields (+ tail pad)
        inline = bytearray()
        for fid, kind, val, p, pad in placed:
            inline += b"\x00" * pad
            if kind == "i64":
                inline += struct.pack("<q", val)
            elif kind == "i16":
                inline += struct.pack("<h", val)
            elif kind == "u8":
                inline += struct.pack("<B", val)
            elif kind == "ref":
                # patched below once positions are known
                inline += b"\x00\x00\x00\x00"
        inline += b"\x00" * pad_tail

        # Absolute alignment: the flatbuffers verifier requires every i64
        # scalar at an 8-aligned ABSOLUTE position.  finish() pads the total
        # to 8, so abs = -end_rel (mod 8): force table_rel (end_rel of the
        # soffset word) to 0 mod 8 with padding on the file-end side.
        pre_pad = (-(self._size + len(inline) + 4)) % 8
        if pre_pad:
            self._prepend(b"\x00" * pre_pad)
        self._prepend(bytes(inline))
        table_rel = self._size + 4        # account for soffset written next
        # resolve refs now that table position is known
        buf = bytearray(self._parts[-1])
        for fid, kind, val, p, pad in placed:
            if kind == "ref":
                field_rel = table_rel - p
                uoff = field_rel - val
                buf[p - 4:p] = struct.pack("<I", uoff)
        self._parts[-1] = bytes(buf)
        self._prepend(struct.pack("<i", 0))  # placeholder soffset
        table_rel = self._size
        self._prepend(vtable)
        vtable_rel = self._size
        # patch soffset = vtable_rel - table_rel? stored i32 = vtable offset
        # relative: soffset = table_abs_vt_field - vtable_abs...
        # flatbuffers: i32 at table start = (table_start_abs - vtable_abs)
        # NEGATED convention: value v means vtable at table_start - v.
        # abs = total - rel  =>  table_abs - vtable_abs = vtable_rel - table_rel
        soff = vtable_rel - table_rel
        buf = bytearray(self._parts[-2])  # the soffset part
        buf[0:4] = struct.pack("<i", soff)
        self._parts[-2] = bytes(buf)
        return table_rel

    def finish(self, root_rel: int) -> bytes:
        # total (root u32 + pad + content) must be 0 mod 8 so that
        # end-relative alignment == absolute alignment
        self._align(8, extra=4)
        root_pos = self._size + 4
        self._prepend(struct.pack("<I", root_pos - root_rel))
        return b"".join(reversed(self._parts))


# MessageHeader union types (format/Message.fbs)
_HDR_RECORD_BATCH = 3
_V5 = 4


def record_batch_message(n_rows: int, field_nodes, buffers, body_len: int,
                         compressed: bool) -> bytes:
    """Metadata flatbuffer for one RecordBatch.
    field_nodes: [(length, null_count)]; buffers: [(offset, length)]."""
    fb = _FB()
    comp_rel = None
    if compressed:
        # BodyCompression{codec=LZ4_FRAME(0)=default, method=BUFFER(0)=default}
        # -> empty table (defaults omitted), but it must be present
        comp_rel = fb.table([(0, "u8", 0)])  # write codec=0 explicitly
    bufs_rel = fb.struct_vector(
        [struct.pack("<qq", o, l) for (o, l) in buffers])
    nodes_rel = fb.struct_vector(
        [struct.pack("<qq", ln, nc) for (ln, nc) in field_nodes])
    rb_fields = [(0, "i64", n_rows), (1, "ref", nodes_rel),
                 (2, "ref", bufs_rel)]
    if compressed:
        rb_fields.append((3, "ref", comp_rel))
    rb_rel = fb.table(rb_fields)
    msg_rel = fb.table([
        (0, "i16", _V5),
        (1, "u8", _HDR_RECORD_BATCH),
        (2, "ref", rb_rel),
        (3, "i64", body_len),
    ])
    meta = fb.finish(msg_rel)
    # encapsulated framing: continuation + i32 len, meta padded to 8
    pad = (-(len(meta) + 8)) % 8
    meta = meta + b"\x00" * pad
    return b"\xff\xff\xff\xff" + struct.pack("<i", len(meta)) + meta


def schema_message_bytes(schema: pa.Schema) -> bytes:
    """The schema message verbatim from pyarrow (schema-only stream minus
    the trailing EOS marker)."""
    sink = io.BytesIO()
    with pa.ipc.new_stream(sink, schema):
        pass
    raw = sink.getvalue()
    assert raw.endswith(b"\xff\xff\xff\xff\x00\x00\x00\x00")
    return raw[:-8]


EOS = b"\xff\xff\xff\xff\x00\x00\x00\x00"


def stream_from_compressed_batches(schema: pa.Schema, batches) -> bytes:
    """batches: [(n_rows, field_nodes, buffer_parts)] where buffer_parts is
    a list of per-buffer COMPRESSED payloads (each already
    [i64 uncompressed_len][LZ4 frame]); None = empty buffer (validity of a
    null-free column).  Returns a complete IPC stream."""
    out = [schema_message_bytes(schema)]
    for n_rows, field_nodes, buffer_parts in batches:
        body = bytearray()
        bufspecs = []
        for part in buffer_parts:
            if part is None:
                bufspecs.append((len(body), 0))
                continue
            start = len(body)
            body += part
            body += b"\x00" * ((-len(body)) % 8)
            bufspecs.append((start, len(part)))
        meta = record_batch_message(n_rows, field_nodes, bufspecs,
                                    len(body), compressed=True)
        out.append(meta)
        out.append(bytes(body))
    out.append(EOS)
    return b"".join(out)


# ---------------------------------------------------------------------------
# minimal reader side: walk an IPC stream's encapsulated messages and parse
# RecordBatch metadata (the decode mirror of the writer above) — used by the
# device shuffle-read path to locate compressed buffer extents
# ---------------------------------------------------------------------------

def _parse_table(meta, tbl):
    soff = struct.unpack_from("<i", meta, tbl)[0]
    vt = tbl - soff
    vt_len, _ = struct.unpack_from("<HH", meta, vt)
    n = (vt_len - 4) // 2
    slots = struct.unpack_from(f"<{n}H", meta, vt + 4)
    return slots


def parse_record_batch_meta(meta: bytes):
    """-> (n_rows, [(buf_off, buf_len)], compressed, body_len) or None if
    the message is not a RecordBatch."""
    root = struct.unpack_from("<I", meta, 0)[0]
    slots = _parse_table(meta, root)

    def fld(i):
        return slots[i] if i < len(slots) and slots[i] else None

    if not fld(1) or meta[root + slots[1]] != _HDR_RECORD_BATCH:
        return None
    body_len = struct.unpack_from("<q", meta, root + slots[3])[0] \
        if fld(3) else 0
    hpos = root + slots[2]
    hoff = struct.unpack_from("<I", meta, hpos)[0]
    htbl = hpos + hoff
    hslots = _parse_table(meta, htbl)
    n_rows = struct.unpack_from("<q", meta, htbl + hslots[0])[0] \
        if hslots[0] else 0
    bpos = htbl + hslots[2]
    boff = struct.unpack_from("<I", meta, bpos)[0]
    bvec = bpos + boff
    cnt = struct.unpack_from("<I", meta, bvec)[0]
    bufs = [struct.unpack_from("<qq", meta, bvec + 4 + 16 * i)
            for i in range(cnt)]
    compressed = len(hslots) > 3 and hslots[3] != 0
    return n_rows, bufs, compressed, body_len


def walk_stream(raw: bytes):
    """Yield (meta_bytes, body_off, body_len) per message in an IPC stream
    (concatenated streams supported by the caller re-invoking)."""
    pos = 0
    while pos + 8 <= len(raw):
        if raw[pos:pos + 4] != b"\xff\xff\xff\xff":
            # legacy (no continuation marker) not supported
            return
        mlen = struct.unpack_from("<i", raw, pos + 4)[0]
        if mlen == 0:
            pos += 8
            return
        meta = raw[pos + 8:pos + 8 + mlen]
        parsed = parse_record_batch_meta(meta)
        body_off = pos + 8 + mlen
        if parsed is None:
            yield None, body_off, 0
            pos = body_off
            continue
        n_rows, bufs, compressed, body_len = parsed
        yield (n_rows, bufs, compressed), body_off, body_len
        pos = body_off + body_len
