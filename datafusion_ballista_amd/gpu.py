"""ctypes host bindings for libballista_gpu.so (include/ballista_gpu.h).

The GpuStageContext owns device memory for one task's columns and exposes
the hot-path ops.  It fails loudly (RuntimeError) when the HIP extension or
the GPU is missing — no CPU fallback exists on the product path.
"""

import ctypes
import os
import struct as _struct

import numpy as np


def struct_unpack_bits(v: float) -> int:
    """f64 -> its bit pattern as a SIGNED i64 (BgPred bound fields)."""
    return _struct.unpack("<q", _struct.pack("<d", v))[0]

_DIR = os.path.dirname(os.path.abspath(__file__))

BG_DT_INT32 = 1
BG_DT_INT64 = 2
BG_DT_DATE32 = 3
BG_DT_DECIMAL128 = 4
BG_DT_DICT8 = 5
BG_DT_UTF8 = 6
BG_DT_FLOAT64 = 7

BG_PRED_GE_LT = 0
BG_PRED_BETWEEN = 1
BG_PRED_LT = 2
BG_PRED_EQ = 3
BG_PRED_GT = 4

_DT_SIZE = {BG_DT_INT32: 4, BG_DT_DATE32: 4, BG_DT_INT64: 8,
            BG_DT_DECIMAL128: 16, BG_DT_DICT8: 1, BG_DT_FLOAT64: 8}

_NP_TO_DT = {
    np.dtype(np.int32): BG_DT_INT32,
    np.dtype(np.int64): BG_DT_INT64,
    np.dtype(np.uint8): BG_DT_DICT8,
    np.dtype(np.float64): BG_DT_FLOAT64,
}


class BgColumn(ctypes.Structure):
    _fields_ = [
        ("dtype", ctypes.c_int32),
        ("precision", ctypes.c_int32),
        ("scale", ctypes.c_int32),
        ("_pad", ctypes.c_int32),
        ("d_data", ctypes.c_void_p),
        ("d_validity", ctypes.c_void_p),
        ("d_offsets", ctypes.c_void_p),
        ("len", ctypes.c_int64),
    ]


class BgPred(ctypes.Structure):
    _fields_ = [
        ("column", ctypes.c_int32),
        ("op", ctypes.c_int32),
        ("lo_lo", ctypes.c_int64),
        ("lo_hi", ctypes.c_int64),
        ("hi_lo", ctypes.c_int64),
        ("hi_hi", ctypes.c_int64),
    ]


def lib_path() -> str:
    return os.path.join(_DIR, "libballista_gpu.so")


_lib = None


def load_library() -> ctypes.CDLL:
    """dlopen the HIP extension; raises if it is absent or unloadable."""
    global _lib
    if _lib is None:
        p = lib_path()
        if not os.path.exists(p):
            raise RuntimeError(
                f"libballista_gpu.so not found at {p}: the MI355X stage "
                "executor requires its HIP extension (run "
                "__graft_entry__.build()); there is no CPU fallback")
        _lib = ctypes.CDLL(p)
        _lib.bg_last_error.restype = ctypes.c_char_p
        _lib.bg_last_kernel_ms.restype = ctypes.c_double
    return _lib


def _check(rc: int, what: str):
    if rc != 0:
        err = load_library().bg_last_error().decode()
        raise RuntimeError(f"{what} failed (rc={rc}): {err}")


def _split_i128(v: int):
    v = int(v)
    return (ctypes.c_int64(v & 0xFFFFFFFFFFFFFFFF).value,
            ctypes.c_int64((v >> 64) & 0xFFFFFFFFFFFFFFFF).value)


class DeviceBuffer:
    """Caller-owned device allocation."""

    def __init__(self, ctx, nbytes: int):
        self._ctx = ctx
        self._owns = True
        self.nbytes = nbytes
        ptr = ctypes.c_void_p()
        _check(ctx.L.bg_malloc(ctypes.c_uint64(nbytes), ctypes.byref(ptr)),
               "bg_malloc")
        self.ptr = ptr

    def free(self):
        if self.ptr and self.ptr.value:
            self._ctx.L.bg_free(self.ptr)
            self.ptr = ctypes.c_void_p()

    def __del__(self):
        # return the buffer to the allocator pool when the last reference
        # dies — without this every stage leaks its temporaries and later
        # allocations pay fresh multi-GB hipMallocs (~100 ms/GB) instead
        # of a free-list hit.  Only buffers WE allocated: wrapper objects
        # built with __new__ around borrowed pointers (torch tensors) have
        # no _owns flag and must never be freed here.
        if not getattr(self, "_owns", False):
            return
        try:
            self.free()
        except Exception:
            pass  # interpreter shutdown: the library may already be gone

    def upload(self, arr: np.ndarray):
        a = np.ascontiguousarray(arr)
        assert a.nbytes <= self.nbytes
        _check(self._ctx.L.bg_memcpy_h2d(
            self.ptr, a.ctypes.data_as(ctypes.c_void_p),
            ctypes.c_uint64(a.nbytes)), "bg_memcpy_h2d")
        return self

    def download(self, dtype, count) -> np.ndarray:
        out = np.empty(count, dtype=dtype)
        _check(self._ctx.L.bg_memcpy_d2h(
            out.ctypes.data_as(ctypes.c_void_p), self.ptr,
            ctypes.c_uint64(out.nbytes)), "bg_memcpy_d2h")
        return out


class GpuStageContext:
    """One GPU-backed task context (one process per GPU; the device ordinal
    is normally fixed by HIP_VISIBLE_DEVICES, SURVEY.md §2 executor row)."""

    def __init__(self, device: int = 0):
        self.L = load_library()
        _check(self.L.bg_init(device), "bg_init")
        self._bufs = []

    # ---- memory ----
    def alloc(self, nbytes: int) -> DeviceBuffer:
        # ownership lives with the returned DeviceBuffer (__del__ returns
        # it to the pool when the last reference dies) — pinning every
        # allocation on the context kept multi-GB stage temporaries alive
        # for the context's whole life, so long pipelines paid fresh
        # hipMallocs instead of pool hits
        return DeviceBuffer(self, nbytes)

    def upload(self, arr: np.ndarray) -> DeviceBuffer:
        return self.alloc(arr.nbytes).upload(arr)

    def close(self):
        self._bufs = []  # legacy; buffers free themselves via __del__

    def synchronize(self):
        _check(self.L.bg_synchronize(), "bg_synchronize")

    # ---- columns ----
    def column(self, dtype: int, buf: DeviceBuffer, n: int,
               validity: DeviceBuffer = None, precision=0, scale=0,
               offsets: DeviceBuffer = None) -> BgColumn:
        col = BgColumn(dtype, precision, scale, 0, buf.ptr,
                       validity.ptr if validity else None,
                       offsets.ptr if offsets else None, n)
        # the struct carries raw device pointers: keep the backing
        # DeviceBuffers alive as long as the column handle is
        col._keep = (buf, validity, offsets)
        return col

    def upload_utf8_column(self, strings):
        """list[bytes] -> BG_DT_UTF8 column (Arrow i32 offsets + data)."""
        data = b"".join(strings)
        offs = np.zeros(len(strings) + 1, dtype=np.int32)
        for i, b in enumerate(strings):
            offs[i + 1] = offs[i] + len(b)
        dbuf = self.upload(np.frombuffer(data, dtype=np.uint8)
                           if data else np.zeros(1, dtype=np.uint8))
        obuf = self.upload(offs)
        return self.column(BG_DT_UTF8, dbuf, len(strings), offsets=obuf)

    def upload_column(self, arr: np.ndarray, dtype: int = None,
                      validity: np.ndarray = None):
        """Fixed-width column upload; Decimal128 arrives as raw 16-B rows."""
        if dtype is None:
            dtype = _NP_TO_DT[arr.dtype]
        buf = self.upload(arr)
        vbuf = self.upload(validity) if validity is not None else None
        n = arr.nbytes // _DT_SIZE[dtype]
        return self.column(dtype, buf, n, vbuf), buf

    # ---- ops ----
    def eval_predicates(self, cols, preds, n: int) -> DeviceBuffer:
        """cols: list[BgColumn]; preds: list[(col, op, lo, hi)] with python
        int bounds. Returns the device mask (ceil(n/64)*8 bytes)."""
        nwords = (n + 63) // 64
        mask = self.alloc(max(nwords * 8, 8))
        carr = (BgColumn * len(cols))(*cols)
        parr = (BgPred * len(preds))()
        for i, (c, op, lo, hi) in enumerate(preds):
            if isinstance(lo, float) or isinstance(hi, float):
                lo_lo = struct_unpack_bits(float(lo))
                hi_lo = struct_unpack_bits(float(hi))
                parr[i] = BgPred(c, op, lo_lo, 0, hi_lo, 0)
                continue
            lo_lo, lo_hi = _split_i128(lo)
            hi_lo, hi_hi = _split_i128(hi)
            parr[i] = BgPred(c, op, lo_lo, lo_hi, hi_lo, hi_hi)
        _check(self.L.bg_eval_predicates(carr, len(cols), parr, len(preds),
                                         ctypes.c_int64(n), mask.ptr),
               "bg_eval_predicates")
        return mask

    def mask_to_indices(self, mask: DeviceBuffer, n: int):
        idx = self.alloc(max(4 * n, 4))
        count = ctypes.c_int64()
        _check(self.L.bg_mask_to_indices(mask.ptr, ctypes.c_int64(n), idx.ptr,
                                         ctypes.byref(count)),
               "bg_mask_to_indices")
        return idx, count.value

    def gather(self, src: DeviceBuffer, elem_size: int, idx: DeviceBuffer,
               m: int) -> DeviceBuffer:
        dst = self.alloc(max(elem_size * m, elem_size))
        _check(self.L.bg_gather(src.ptr, ctypes.c_int64(elem_size), idx.ptr,
                                ctypes.c_int64(m), dst.ptr), "bg_gather")
        return dst

    def hash_columns(self, key_cols, n: int) -> DeviceBuffer:
        hashes = self.alloc(max(8 * n, 8))
        karr = (BgColumn * len(key_cols))(*key_cols)
        _check(self.L.bg_hash_columns(karr, len(key_cols), ctypes.c_int64(n),
                                      hashes.ptr), "bg_hash_columns")
        return hashes

    def partition_ids(self, hashes: DeviceBuffer, n: int, k: int) -> DeviceBuffer:
        pids = self.alloc(max(4 * n, 4))
        _check(self.L.bg_partition_ids(hashes.ptr, ctypes.c_int64(n), k,
                                       pids.ptr), "bg_partition_ids")
        return pids

    def partition_indices(self, pids: DeviceBuffer, n: int, k: int):
        idx = self.alloc(max(4 * n, 4))
        offs = self.alloc(8 * (k + 1))
        _check(self.L.bg_partition_indices(pids.ptr, ctypes.c_int64(n), k,
                                           idx.ptr, offs.ptr),
               "bg_partition_indices")
        return idx, offs

    def hash_repartition(self, key_cols, payload_cols, n: int, k: int):
        """-> (indices buf, offsets buf, [out bufs partition-major])."""
        idx = self.alloc(max(4 * n, 4))
        offs = self.alloc(8 * (k + 1))
        outs = []
        optrs = (ctypes.c_void_p * len(payload_cols))()
        for i, c in enumerate(payload_cols):
            esz = _DT_SIZE[c.dtype]
            b = self.alloc(max(esz * n, esz))
            outs.append(b)
            optrs[i] = b.ptr.value
        karr = (BgColumn * len(key_cols))(*key_cols)
        parr = (BgColumn * len(payload_cols))(*payload_cols)
        _check(self.L.bg_hash_repartition(karr, len(key_cols), parr,
                                          len(payload_cols),
                                          ctypes.c_int64(n), k, idx.ptr,
                                          offs.ptr, optrs),
               "bg_hash_repartition")
        return idx, offs, outs

    def q6_agg(self, shipdate: BgColumn, discount: BgColumn,
               quantity: BgColumn, price: BgColumn, date_lo: int, date_hi: int,
               disc_lo: int, disc_hi: int, qty_lt: int):
        """-> (count, exact i128 sum as python int)."""
        s_lo = ctypes.c_uint64()
        s_hi = ctypes.c_int64()
        cnt = ctypes.c_int64()
        _check(self.L.bg_q6_agg(ctypes.byref(shipdate), ctypes.byref(discount),
                                ctypes.byref(quantity), ctypes.byref(price),
                                date_lo, date_hi,
                                ctypes.c_int64(disc_lo), ctypes.c_int64(disc_hi),
                                ctypes.c_int64(qty_lt), ctypes.byref(s_lo),
                                ctypes.byref(s_hi), ctypes.byref(cnt)),
               "bg_q6_agg")
        return cnt.value, (s_hi.value << 64) + s_lo.value

    def q1_agg(self, rf, ls, qty, price, disc, tax, shipdate, date_le: int):
        """-> dict group -> (count, [5 exact i128 sums])."""
        counts = np.zeros(256, dtype=np.int64)
        sums = np.zeros(256 * 5 * 16, dtype=np.uint8)
        _check(self.L.bg_q1_agg(
            ctypes.byref(rf), ctypes.byref(ls), ctypes.byref(qty),
            ctypes.byref(price), ctypes.byref(disc), ctypes.byref(tax),
            ctypes.byref(shipdate), date_le,
            counts.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)),
            sums.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8))), "bg_q1_agg")
        out = {}
        raw = sums.reshape(256, 5, 16)
        for g in range(256):
            if counts[g] == 0:
                continue
            vals = [int.from_bytes(bytes(raw[g, a]), "little", signed=True)
                    for a in range(5)]
            out[g] = (int(counts[g]), vals)
        return out


class GpuHashJoin:
    """INNER equi-join (Int64 keys) — bg_hashjoin_* wrapper.
    Build once, probe many (HashJoinExec build/probe semantics)."""

    def __init__(self, ctx: GpuStageContext, build_keys: BgColumn, n_build: int):
        self.ctx = ctx
        self._handle = ctypes.c_void_p()
        _check(ctx.L.bg_hashjoin_build(ctypes.byref(build_keys),
                                       ctypes.c_int64(n_build),
                                       ctypes.byref(self._handle)),
               "bg_hashjoin_build")

    def probe(self, probe_keys: BgColumn, n_probe: int):
        """-> (probe_idx DeviceBuffer, build_idx DeviceBuffer, n_matches)."""
        ctx = self.ctx
        matches = ctypes.c_int64()
        _check(ctx.L.bg_hashjoin_probe_count(self._handle,
                                             ctypes.byref(probe_keys),
                                             ctypes.c_int64(n_probe),
                                             ctypes.byref(matches)),
               "bg_hashjoin_probe_count")
        m = matches.value
        pbuf = ctx.alloc(max(4 * m, 4))
        bbuf = ctx.alloc(max(4 * m, 4))
        _check(ctx.L.bg_hashjoin_probe_fill(self._handle,
                                            ctypes.byref(probe_keys),
                                            ctypes.c_int64(n_probe),
                                            pbuf.ptr, bbuf.ptr),
               "bg_hashjoin_probe_fill")
        return pbuf, bbuf, m

    def free(self):
        if self._handle and self._handle.value:
            self.ctx.L.bg_hashjoin_free(self._handle)
            self._handle = ctypes.c_void_p()


BG_AGG_OP_SUM_DEC128 = 0
BG_AGG_OP_SUM_I64 = 1
BG_AGG_OP_MIN_I64 = 2
BG_AGG_OP_MAX_I64 = 3
BG_AGG_OP_SUM_F64 = 4
BG_AGG_OP_MIN_F64 = 5
BG_AGG_OP_MAX_F64 = 6


def decode_agg_value(op, raw16: bytes):
    """Accumulator bytes -> python int/float per aggregate op."""
    import struct as _st
    if op in (BG_AGG_OP_SUM_DEC128, BG_AGG_OP_SUM_I64):
        return int.from_bytes(raw16, "little", signed=True)
    if op == BG_AGG_OP_SUM_F64:
        return _st.unpack("<d", raw16[:8])[0]
    enc = int.from_bytes(raw16[:8], "little")
    if op in (BG_AGG_OP_MIN_F64, BG_AGG_OP_MAX_F64):
        if op == BG_AGG_OP_MIN_F64:
            enc = ~enc & 0xFFFFFFFFFFFFFFFF
        bits = (~enc & 0xFFFFFFFFFFFFFFFF) if not (enc >> 63)             else enc ^ (1 << 63)
        return _st.unpack("<d", _st.pack("<Q", bits))[0]
    if op == BG_AGG_OP_MAX_I64:
        v = enc ^ (1 << 63)
    else:  # MIN_I64
        v = (~enc & 0xFFFFFFFFFFFFFFFF) ^ (1 << 63)
    return v - (1 << 64) if v >= 1 << 63 else v


def _ctx_hashagg(self, key_cols, agg_cols, agg_ops, n, max_groups,
                 mask: "DeviceBuffer" = None, want_nncnt: bool = False):
    """General hash group-by. -> (first_row u32[g], acc i128 bytes
    [g, naggs], counts i64[g]) as numpy arrays (downloaded); with
    want_nncnt also the per-(group, agg) NON-NULL input counts i64[g, naggs]
    (SQL: SUM/MIN/MAX of an all-NULL group is NULL — nncnt 0 marks it)."""
    naggs = len(agg_cols)
    first = self.alloc(max(4 * max_groups, 4))
    acc = self.alloc(max(16 * max_groups * max(naggs, 1), 16))
    counts = self.alloc(max(8 * max_groups, 8))
    nncnt = self.alloc(max(8 * max_groups * max(naggs, 1), 8))         if want_nncnt else None
    karr = (BgColumn * len(key_cols))(*key_cols)
    aarr = (BgColumn * max(naggs, 1))(*(agg_cols or [BgColumn()]))
    oarr = (ctypes.c_int32 * max(naggs, 1))(*(agg_ops or [0]))
    ng = ctypes.c_int64()
    # auto-grow on table-full (the scheduler's row estimates can be low,
    # like the reference's AQE-fed group estimates)
    attempt_groups = max_groups
    for _ in range(8):
        if want_nncnt:
            rc = self.L.bg_hashagg2(karr, len(key_cols), aarr, oarr, naggs,
                                    mask.ptr if mask else None,
                                    ctypes.c_int64(n),
                                    ctypes.c_int64(attempt_groups), first.ptr,
                                    acc.ptr, counts.ptr, nncnt.ptr,
                                    ctypes.byref(ng))
        else:
            rc = self.L.bg_hashagg(karr, len(key_cols), aarr, oarr, naggs,
                                   mask.ptr if mask else None,
                                   ctypes.c_int64(n),
                                   ctypes.c_int64(attempt_groups), first.ptr,
                                   acc.ptr, counts.ptr, ctypes.byref(ng))
        if rc == 0:
            break
        err = load_library().bg_last_error().decode()
        if "table full" not in err or attempt_groups >= n:
            _check(rc, "bg_hashagg")
        attempt_groups = min(max(attempt_groups * 8, 64), max(n, 64))
        # output buffers must grow with the group capacity
        first = self.alloc(max(4 * attempt_groups, 4))
        acc = self.alloc(max(16 * attempt_groups * max(naggs, 1), 16))
        counts = self.alloc(max(8 * attempt_groups, 8))
        if want_nncnt:
            nncnt = self.alloc(max(8 * attempt_groups * max(naggs, 1), 8))
    else:
        _check(rc, "bg_hashagg")
    g = ng.value
    ret = (first.download(np.uint32, g),
           acc.download(np.uint8, 16 * g * naggs).reshape(g, naggs, 16)
           if naggs else np.zeros((g, 0, 16), dtype=np.uint8),
           counts.download(np.int64, g))
    if want_nncnt:
        nn = nncnt.download(np.int64, g * naggs).reshape(g, max(naggs, 1))             if naggs else np.zeros((g, 0), dtype=np.int64)
        return ret + (nn,)
    return ret


GpuStageContext.hashagg = _ctx_hashagg


BG_PROJ_MUL = 0
BG_PROJ_ADD = 1
BG_PROJ_SUB = 2
BG_PROJ_RSUB_LIT = 3
BG_PROJ_MUL_LIT = 4
BG_PROJ_ADD_LIT = 5


def _ctx_project_dec128(self, op, a: BgColumn, b: BgColumn, lit, n):
    out = self.alloc(max(16 * n, 16))
    lo, hi = _split_i128(lit)
    _check(self.L.bg_project_dec128(op, ctypes.byref(a),
                                    ctypes.byref(b) if b else None,
                                    ctypes.c_int64(lo), ctypes.c_int64(hi),
                                    ctypes.c_int64(n), out.ptr),
           "bg_project_dec128")
    return out


GpuStageContext.project_dec128 = _ctx_project_dec128


def _ctx_sort_rows(self, key_cols, descending, n):
    """Stable multi-column ORDER BY -> row permutation DeviceBuffer."""
    perm = self.alloc(max(4 * n, 4))
    karr = (BgColumn * len(key_cols))(*key_cols)
    darr = (ctypes.c_int32 * len(key_cols))(*[1 if d else 0
                                              for d in descending])
    _check(self.L.bg_sort_rows(karr, darr, len(key_cols),
                               ctypes.c_int64(n), perm.ptr), "bg_sort_rows")
    return perm


GpuStageContext.sort_rows = _ctx_sort_rows


def _ctx_merge_join(self, build_sorted: "DeviceBuffer", nb: int,
                    probe_sorted: "DeviceBuffer", np_: int):
    """Inner merge join of key-sorted i64 buffers -> (probe_pos, build_pos
    DeviceBuffers into the sorted orders, n_matches)."""
    matches = ctypes.c_int64()
    _check(self.L.bg_merge_join(build_sorted.ptr, ctypes.c_int64(nb),
                                probe_sorted.ptr, ctypes.c_int64(np_),
                                ctypes.byref(matches), None, None),
           "bg_merge_join(count)")
    m = matches.value
    pbuf = self.alloc(max(4 * m, 4))
    bbuf = self.alloc(max(4 * m, 4))
    _check(self.L.bg_merge_join(build_sorted.ptr, ctypes.c_int64(nb),
                                probe_sorted.ptr, ctypes.c_int64(np_),
                                ctypes.byref(matches), pbuf.ptr, bbuf.ptr),
           "bg_merge_join(fill)")
    return pbuf, bbuf, m


GpuStageContext.merge_join = _ctx_merge_join


class BgSnappyPage(ctypes.Structure):
    _fields_ = [("d_src", ctypes.c_void_p), ("d_dst", ctypes.c_void_p),
                ("src_len", ctypes.c_int64), ("dst_cap", ctypes.c_int64)]


def _ctx_snappy_decompress(self, pages):
    """pages: list of (src DeviceBuffer, src_len, dst DeviceBuffer,
    dst_cap). -> list of decompressed lengths (-1 = malformed)."""
    n = len(pages)
    arr = (BgSnappyPage * n)()
    for i, (src, slen, dst, dcap) in enumerate(pages):
        arr[i] = BgSnappyPage(src.ptr, dst.ptr, slen, dcap)
    lens = np.zeros(n, dtype=np.int64)
    _check(self.L.bg_snappy_decompress(
        arr, ctypes.c_int64(n),
        lens.ctypes.data_as(ctypes.POINTER(ctypes.c_int64))),
        "bg_snappy_decompress")
    return lens.tolist()


GpuStageContext.snappy_decompress = _ctx_snappy_decompress


class BgPageExtractJob(ctypes.Structure):
    _fields_ = [("d_page", ctypes.c_void_p), ("d_out", ctypes.c_void_p),
                ("page_len", ctypes.c_int64), ("nvals", ctypes.c_int64),
                ("src_esz", ctypes.c_int64), ("has_def", ctypes.c_int32),
                ("flba_reverse", ctypes.c_int32),
                ("d_vidx", ctypes.c_void_p),
                ("d_n_present", ctypes.c_void_p)]


class BgDictIndicesJob(ctypes.Structure):
    _fields_ = [("d_page", ctypes.c_void_p), ("d_out_idx", ctypes.c_void_p),
                ("page_len", ctypes.c_int64), ("nvals", ctypes.c_int64),
                ("has_def", ctypes.c_int32), ("_pad", ctypes.c_int32),
                ("d_vidx", ctypes.c_void_p),
                ("d_n_present", ctypes.c_void_p),
                ("d_dense", ctypes.c_void_p)]


class BgBaPageJob(ctypes.Structure):
    _fields_ = [("d_page", ctypes.c_void_p),
                ("d_lens_out", ctypes.c_void_p),
                ("d_srcaddr_out", ctypes.c_void_p),
                ("page_len", ctypes.c_int64), ("nvals", ctypes.c_int64),
                ("has_def", ctypes.c_int32), ("_pad", ctypes.c_int32),
                ("d_vidx", ctypes.c_void_p),
                ("d_n_present", ctypes.c_void_p)]


class BgDeltaBpJob(ctypes.Structure):
    _fields_ = [("d_page", ctypes.c_void_p), ("d_out", ctypes.c_void_p),
                ("page_len", ctypes.c_int64), ("nvals", ctypes.c_int64),
                ("esz", ctypes.c_int64),
                ("has_def", ctypes.c_int32), ("_pad", ctypes.c_int32),
                ("d_vidx", ctypes.c_void_p),
                ("d_n_present", ctypes.c_void_p)]


class BgDeltaBaJob(ctypes.Structure):
    _fields_ = [("d_page", ctypes.c_void_p),
                ("d_lens_out", ctypes.c_void_p),
                ("d_srcaddr_out", ctypes.c_void_p),
                ("d_offs32", ctypes.c_void_p),
                ("d_data_out", ctypes.c_void_p),
                ("page_len", ctypes.c_int64), ("nvals", ctypes.c_int64),
                ("has_def", ctypes.c_int32), ("enc", ctypes.c_int32),
                ("d_vidx", ctypes.c_void_p),
                ("d_n_present", ctypes.c_void_p)]


class BgDefLevelsJob(ctypes.Structure):
    _fields_ = [("d_page", ctypes.c_void_p), ("d_vidx", ctypes.c_void_p),
                ("d_valid_out", ctypes.c_void_p),
                ("page_len", ctypes.c_int64), ("nvals", ctypes.c_int64),
                ("bit_off", ctypes.c_int64),
                ("d_n_present", ctypes.c_void_p)]


class BgListLevelsJob(ctypes.Structure):
    _fields_ = [("d_page", ctypes.c_void_p), ("page_len", ctypes.c_int64),
                ("nslots", ctypes.c_int64), ("max_def", ctypes.c_int32),
                ("def_entry", ctypes.c_int32),
                ("def_valid", ctypes.c_int32), ("_pad", ctypes.c_int32),
                ("row_base", ctypes.c_int64),
                ("entry_base", ctypes.c_int64),
                ("d_counts", ctypes.c_void_p),
                ("d_row_sizes", ctypes.c_void_p),
                ("d_list_valid", ctypes.c_void_p),
                ("d_elem_valid", ctypes.c_void_p),
                ("d_vidx", ctypes.c_void_p)]


def _ctx_gather_varlen(self, src_data: "DeviceBuffer", src_offsets: "DeviceBuffer",
                       idx: "DeviceBuffer", m: int, max_bytes: int):
    """Variable-length take -> (offsets buf i32[m+1], data buf, total)."""
    out_offs = self.alloc(max(4 * (m + 1), 8))
    out_data = self.alloc(max(max_bytes, 1))
    total = ctypes.c_int64()
    _check(self.L.bg_gather_varlen(src_data.ptr, src_offsets.ptr, idx.ptr,
                                   ctypes.c_int64(m), out_offs.ptr,
                                   out_data.ptr, ctypes.c_int64(max_bytes),
                                   ctypes.byref(total)), "bg_gather_varlen")
    return out_offs, out_data, total.value


GpuStageContext.gather_varlen = _ctx_gather_varlen


LZ4_FRAME_HEADER = bytes.fromhex("04224d184040c0")  # magic+FLG+BD+HC


def _ctx_lz4_compress(self, src: "DeviceBuffer", length: int):
    """Device LZ4 -> (list of (size, is_stored), slots DeviceBuffer)."""
    nblocks = (length + 65536 - 1) // 65536 if length else 0
    slots = self.alloc(max(nblocks * 65544, 8))
    sizes = np.zeros(max(nblocks, 1), dtype=np.int64)
    nb = ctypes.c_int64()
    _check(self.L.bg_lz4_compress(src.ptr, ctypes.c_int64(length), slots.ptr,
                                  sizes.ctypes.data_as(
                                      ctypes.POINTER(ctypes.c_int64)),
                                  ctypes.byref(nb)), "bg_lz4_compress")
    return sizes[:nblocks], slots


def lz4_frame_assemble(sizes, slot_bytes: bytes, length: int) -> bytes:
    """Host glue: device block slots -> one LZ4 frame (constant header,
    [u32 size] blocks — high bit = stored — end mark)."""
    import struct
    out = [LZ4_FRAME_HEADER]
    for i, sz in enumerate(sizes):
        base = i * 65544
        blen = min(65536, length - i * 65536)
        if sz < 0:  # stored raw
            out.append(struct.pack("<I", blen | 0x80000000))
            out.append(slot_bytes[base:base + blen])
        else:
            out.append(struct.pack("<I", int(sz)))
            out.append(slot_bytes[base:base + int(sz)])
    out.append(b"\x00\x00\x00\x00")
    return b"".join(out)


GpuStageContext.lz4_compress = _ctx_lz4_compress


class BgPackJob(ctypes.Structure):
    _fields_ = [("d_src", ctypes.c_void_p), ("d_dst", ctypes.c_void_p),
                ("nbytes", ctypes.c_int64), ("size_word", ctypes.c_uint32),
                ("_pad", ctypes.c_uint32)]


class BgLz4BlockJob(ctypes.Structure):
    _fields_ = [("d_src", ctypes.c_void_p), ("d_dst_slot", ctypes.c_void_p),
                ("blen", ctypes.c_int32), ("_pad", ctypes.c_int32)]


def _ctx_hash_repartition_fused(self, key_cols, payload_cols, n: int, k: int):
    """Fused one-pass materialiser (k<=64, <=4 fixed-width payload cols):
    same outputs/stable order as hash_repartition."""
    idx = self.alloc(max(4 * n, 4))
    offs = self.alloc(8 * (k + 1))
    rank = self.alloc(max(4 * n, 4))
    outs = []
    optrs = (ctypes.c_void_p * len(payload_cols))()
    for i, c in enumerate(payload_cols):
        esz = _DT_SIZE[c.dtype]
        b = self.alloc(max(esz * n, esz))
        outs.append(b)
        optrs[i] = b.ptr.value
    karr = (BgColumn * len(key_cols))(*key_cols)
    parr = (BgColumn * len(payload_cols))(*payload_cols)
    _check(self.L.bg_hash_repartition_fused(
        karr, len(key_cols), parr, len(payload_cols), ctypes.c_int64(n), k,
        idx.ptr, offs.ptr, rank.ptr, optrs), "bg_hash_repartition_fused")
    return idx, offs, outs


GpuStageContext.hash_repartition_fused = _ctx_hash_repartition_fused


class BgLz4Frame(ctypes.Structure):
    _fields_ = [("d_src", ctypes.c_void_p), ("d_dst", ctypes.c_void_p),
                ("src_len", ctypes.c_int64), ("dst_cap", ctypes.c_int64)]


def _ctx_lz4_decompress(self, frames):
    """frames: list of (src DeviceBuffer, src_len, dst DeviceBuffer,
    dst_cap) pointing at LZ4 frame magic. -> decompressed lengths."""
    n = len(frames)
    arr = (BgLz4Frame * n)()
    for i, (src, slen, dst, dcap) in enumerate(frames):
        arr[i] = BgLz4Frame(src.ptr if hasattr(src, "ptr") else src,
                            dst.ptr if hasattr(dst, "ptr") else dst,
                            slen, dcap)
    lens = np.zeros(n, dtype=np.int64)
    _check(self.L.bg_lz4_decompress(
        arr, ctypes.c_int64(n),
        lens.ctypes.data_as(ctypes.POINTER(ctypes.c_int64))),
        "bg_lz4_decompress")
    return lens.tolist()


GpuStageContext.lz4_decompress = _ctx_lz4_decompress
