"""CPU oracle bindings — TEST INFRASTRUCTURE ONLY.

Only ``tests/``, ``__graft_entry__.smoke()`` and ``bench.py``'s
``cpu_baseline`` leg may import this package (enforced by convention and by
the header of oracle/oracle.c).  The product path never imports it.

Restates (see oracle.c for full citations):
  - compute_partition_indices
    (reference: ballista/core/src/execution_plans/sort_shuffle/writer.rs:1259-1279)
  - FilterExec / AggregateExec semantics for the TPC-H hot-path shapes
  - the sort-shuffle index format (sort_shuffle/index.rs:21-33)
"""

import ctypes
import os
import subprocess

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_LIB_PATH = os.path.join(_DIR, "liboracle.so")


def build(force: bool = False) -> str:
    """Compile liboracle.so with gcc (idempotent)."""
    src = os.path.join(_DIR, "oracle.c")
    if force or (not os.path.exists(_LIB_PATH)) or (
        os.path.getmtime(_LIB_PATH) < os.path.getmtime(src)
    ):
        subprocess.run(["make", "-C", _DIR, "liboracle.so"], check=True,
                       capture_output=True)
    return _LIB_PATH


_lib = None


def lib() -> ctypes.CDLL:
    global _lib
    if _lib is None:
        build()
        _lib = ctypes.CDLL(_LIB_PATH)
        _decorate(_lib)
    return _lib


def _decorate(L):
    u8p = ctypes.POINTER(ctypes.c_uint8)
    i32p = ctypes.POINTER(ctypes.c_int32)
    i64p = ctypes.POINTER(ctypes.c_int64)
    u32p = ctypes.POINTER(ctypes.c_uint32)
    u64p = ctypes.POINTER(ctypes.c_uint64)
    L.oracle_hash_col_i64.argtypes = [i64p, u8p, ctypes.c_int64, ctypes.c_int, u64p]
    L.oracle_hash_col_i32.argtypes = [i32p, u8p, ctypes.c_int64, ctypes.c_int, u64p]
    L.oracle_hash_col_dec128.argtypes = [u8p, u8p, ctypes.c_int64, ctypes.c_int, u64p]
    L.oracle_hash_col_utf8.argtypes = [u8p, i32p, u8p, ctypes.c_int64, ctypes.c_int, u64p]
    L.oracle_partition_ids.argtypes = [u64p, ctypes.c_int64, ctypes.c_uint32, u32p]
    L.oracle_partition_indices.argtypes = [u32p, ctypes.c_int64, ctypes.c_uint32, u32p, i64p]
    L.oracle_filter_i32.argtypes = [i32p, u8p, ctypes.c_int64, ctypes.c_int,
                                    ctypes.c_int32, ctypes.c_int32, u8p, ctypes.c_int]
    L.oracle_filter_i64.argtypes = [i64p, u8p, ctypes.c_int64, ctypes.c_int,
                                    ctypes.c_int64, ctypes.c_int64, u8p, ctypes.c_int]
    L.oracle_filter_dec128.argtypes = [u8p, u8p, ctypes.c_int64, ctypes.c_int,
                                       ctypes.c_int64, ctypes.c_int64,
                                       ctypes.c_int64, ctypes.c_int64, u8p, ctypes.c_int]
    L.oracle_mask_to_indices.argtypes = [u8p, ctypes.c_int64, u32p]
    L.oracle_mask_to_indices.restype = ctypes.c_int64
    L.oracle_gather.argtypes = [u8p, ctypes.c_int64, u32p, ctypes.c_int64, u8p]
    L.oracle_q6.argtypes = [i32p, u8p, u8p, u8p, ctypes.c_int64,
                            ctypes.c_int32, ctypes.c_int32,
                            ctypes.c_int64, ctypes.c_int64, ctypes.c_int64,
                            u64p, i64p]
    L.oracle_q6.restype = ctypes.c_int64
    L.oracle_q1.argtypes = [u8p, u8p, u8p, u8p, u8p, u8p, i32p,
                            ctypes.c_int64, ctypes.c_int32, i64p, u8p]


def _p(arr, ty):
    return arr.ctypes.data_as(ctypes.POINTER(ty))


def _valid_p(valid):
    if valid is None:
        return ctypes.cast(None, ctypes.POINTER(ctypes.c_uint8))
    return _p(valid, ctypes.c_uint8)


def hash_columns(cols, n: int) -> np.ndarray:
    """create_hashes restatement over a list of (kind, arrays...) columns.

    cols: list of tuples —
      ("i64", values[, valid_bitmap])
      ("i32", values[, valid])
      ("dec128", bytes16xN uint8 array[, valid])
      ("utf8", data_u8, offsets_i32[, valid])
    """
    L = lib()
    hashes = np.zeros(n, dtype=np.uint64)
    hp = _p(hashes, ctypes.c_uint64)
    for ci, col in enumerate(cols):
        kind = col[0]
        first = 1 if ci == 0 else 0
        if kind == "i64":
            v = np.ascontiguousarray(col[1], dtype=np.int64)
            valid = col[2] if len(col) > 2 else None
            L.oracle_hash_col_i64(_p(v, ctypes.c_int64), _valid_p(valid), n, first, hp)
        elif kind == "i32":
            v = np.ascontiguousarray(col[1], dtype=np.int32)
            valid = col[2] if len(col) > 2 else None
            L.oracle_hash_col_i32(_p(v, ctypes.c_int32), _valid_p(valid), n, first, hp)
        elif kind == "dec128":
            v = np.ascontiguousarray(col[1], dtype=np.uint8)
            assert v.size == 16 * n
            valid = col[2] if len(col) > 2 else None
            L.oracle_hash_col_dec128(_p(v, ctypes.c_uint8), _valid_p(valid), n, first, hp)
        elif kind == "utf8":
            data = np.ascontiguousarray(col[1], dtype=np.uint8)
            offs = np.ascontiguousarray(col[2], dtype=np.int32)
            valid = col[3] if len(col) > 3 else None
            L.oracle_hash_col_utf8(_p(data, ctypes.c_uint8), _p(offs, ctypes.c_int32),
                                   _valid_p(valid), n, first, hp)
        else:
            raise ValueError(kind)
    return hashes


def partition_ids(hashes: np.ndarray, k: int) -> np.ndarray:
    L = lib()
    n = len(hashes)
    pids = np.empty(n, dtype=np.uint32)
    L.oracle_partition_ids(_p(hashes, ctypes.c_uint64), n, k, _p(pids, ctypes.c_uint32))
    return pids


def partition_indices(pids: np.ndarray, k: int):
    """-> (indices u32[n] partition-major, offsets i64[k+1])."""
    L = lib()
    n = len(pids)
    idx = np.empty(n, dtype=np.uint32)
    offs = np.zeros(k + 1, dtype=np.int64)
    L.oracle_partition_indices(_p(pids, ctypes.c_uint32), n, k,
                               _p(idx, ctypes.c_uint32), _p(offs, ctypes.c_int64))
    return idx, offs


def filter_mask(preds, n: int) -> np.ndarray:
    """AND-fold predicates into an Arrow LSB bitmask.

    preds: list of tuples:
      ("i32", values, valid_or_None, op, lo, hi)
      ("i64", values, valid_or_None, op, lo, hi)
      ("dec128", bytes16, valid_or_None, op, lo_int, hi_int)  # python ints
    """
    L = lib()
    mask = np.zeros((n + 7) // 8, dtype=np.uint8)
    for pi, pr in enumerate(preds):
        kind, vals, valid, op, lo, hi = pr
        first = 1 if pi == 0 else 0
        if kind == "i32":
            v = np.ascontiguousarray(vals, dtype=np.int32)
            L.oracle_filter_i32(_p(v, ctypes.c_int32), _valid_p(valid), n, op,
                                int(lo), int(hi), _p(mask, ctypes.c_uint8), first)
        elif kind == "i64":
            v = np.ascontiguousarray(vals, dtype=np.int64)
            L.oracle_filter_i64(_p(v, ctypes.c_int64), _valid_p(valid), n, op,
                                int(lo), int(hi), _p(mask, ctypes.c_uint8), first)
        elif kind == "dec128":
            v = np.ascontiguousarray(vals, dtype=np.uint8)
            lo, hi = int(lo), int(hi)
            L.oracle_filter_dec128(
                _p(v, ctypes.c_uint8), _valid_p(valid), n, op,
                lo & 0xFFFFFFFFFFFFFFFF, (lo >> 64) & 0xFFFFFFFFFFFFFFFF,
                hi & 0xFFFFFFFFFFFFFFFF, (hi >> 64) & 0xFFFFFFFFFFFFFFFF,
                _p(mask, ctypes.c_uint8), first)
        else:
            raise ValueError(kind)
    return mask


def mask_to_indices(mask: np.ndarray, n: int) -> np.ndarray:
    L = lib()
    out = np.empty(n, dtype=np.uint32)
    m = L.oracle_mask_to_indices(_p(mask, ctypes.c_uint8), n, _p(out, ctypes.c_uint32))
    return out[:m].copy()


def gather(src: np.ndarray, elem_size: int, idx: np.ndarray) -> np.ndarray:
    L = lib()
    src8 = np.ascontiguousarray(src).view(np.uint8).reshape(-1)
    idx = np.ascontiguousarray(idx, dtype=np.uint32)
    out = np.empty(len(idx) * elem_size, dtype=np.uint8)
    L.oracle_gather(_p(src8, ctypes.c_uint8), elem_size, _p(idx, ctypes.c_uint32),
                    len(idx), _p(out, ctypes.c_uint8))
    return out


def q6(shipdate, discount16, quantity16, price16, date_lo, date_hi,
       disc_lo, disc_hi, qty_lt):
    """-> (count, exact i128 sum as python int)."""
    L = lib()
    sd = np.ascontiguousarray(shipdate, dtype=np.int32)
    n = len(sd)
    s_lo = ctypes.c_uint64(0)
    s_hi = ctypes.c_int64(0)
    cnt = L.oracle_q6(_p(sd, ctypes.c_int32),
                      _p(np.ascontiguousarray(discount16, dtype=np.uint8), ctypes.c_uint8),
                      _p(np.ascontiguousarray(quantity16, dtype=np.uint8), ctypes.c_uint8),
                      _p(np.ascontiguousarray(price16, dtype=np.uint8), ctypes.c_uint8),
                      n, int(date_lo), int(date_hi), int(disc_lo), int(disc_hi),
                      int(qty_lt), ctypes.byref(s_lo), ctypes.byref(s_hi))
    total = (s_hi.value << 64) + s_lo.value
    return cnt, total


def q1(rf_code, ls_code, qty16, price16, disc16, tax16, shipdate, date_le):
    """-> dict group -> (count, [5 exact i128 sums])."""
    L = lib()
    sd = np.ascontiguousarray(shipdate, dtype=np.int32)
    n = len(sd)
    counts = np.zeros(256, dtype=np.int64)
    sums = np.zeros(256 * 5 * 16, dtype=np.uint8)
    L.oracle_q1(_p(np.ascontiguousarray(rf_code, dtype=np.uint8), ctypes.c_uint8),
                _p(np.ascontiguousarray(ls_code, dtype=np.uint8), ctypes.c_uint8),
                _p(np.ascontiguousarray(qty16, dtype=np.uint8), ctypes.c_uint8),
                _p(np.ascontiguousarray(price16, dtype=np.uint8), ctypes.c_uint8),
                _p(np.ascontiguousarray(disc16, dtype=np.uint8), ctypes.c_uint8),
                _p(np.ascontiguousarray(tax16, dtype=np.uint8), ctypes.c_uint8),
                _p(sd, ctypes.c_int32), n, int(date_le),
                _p(counts, ctypes.c_int64), _p(sums, ctypes.c_uint8))
    out = {}
    raw = sums.reshape(256, 5, 16)
    for g in range(256):
        if counts[g] == 0:
            continue
        vals = []
        for a in range(5):
            b = bytes(raw[g, a])
            vals.append(int.from_bytes(b, "little", signed=True))
        out[g] = (int(counts[g]), vals)
    return out


_omp_lib = None


def omp_lib():
    """Multithreaded oracle build — bench.py cpu_baseline leg only."""
    global _omp_lib
    if _omp_lib is None:
        subprocess.run(["make", "-C", _DIR, "liboracle_omp.so"], check=True,
                       capture_output=True)
        _omp_lib = ctypes.CDLL(os.path.join(_DIR, "liboracle_omp.so"))
        _decorate(_omp_lib)
    return _omp_lib


def q6_omp(shipdate, discount16, quantity16, price16, date_lo, date_hi,
           disc_lo, disc_hi, qty_lt):
    """All-core q6 (exact; i128 wrap-add is associative)."""
    L = omp_lib()
    sd = np.ascontiguousarray(shipdate, dtype=np.int32)
    n = len(sd)
    s_lo = ctypes.c_uint64(0)
    s_hi = ctypes.c_int64(0)
    cnt = L.oracle_q6(_p(sd, ctypes.c_int32),
                      _p(np.ascontiguousarray(discount16, dtype=np.uint8), ctypes.c_uint8),
                      _p(np.ascontiguousarray(quantity16, dtype=np.uint8), ctypes.c_uint8),
                      _p(np.ascontiguousarray(price16, dtype=np.uint8), ctypes.c_uint8),
                      n, int(date_lo), int(date_hi), int(disc_lo), int(disc_hi),
                      int(qty_lt), ctypes.byref(s_lo), ctypes.byref(s_hi))
    return cnt, (s_hi.value << 64) + s_lo.value


def dec128_from_ints(vals) -> np.ndarray:
    """Scaled python ints -> Arrow Decimal128 byte layout (16B LE each)."""
    out = np.zeros(len(vals) * 16, dtype=np.uint8)
    for i, v in enumerate(vals):
        out[i * 16:(i + 1) * 16] = np.frombuffer(
            int(v).to_bytes(16, "little", signed=True), dtype=np.uint8)
    return out


def hashjoin_pairs(build: np.ndarray, probe: np.ndarray):
    """-> (probe_idx u32[], build_idx u32[]) inner-join pair set (O(n*m))."""
    L = lib()
    L.oracle_hashjoin_count.restype = ctypes.c_int64
    L.oracle_hashjoin_count.argtypes = [ctypes.POINTER(ctypes.c_int64),
                                        ctypes.c_int64,
                                        ctypes.POINTER(ctypes.c_int64),
                                        ctypes.c_int64]
    L.oracle_hashjoin_pairs.argtypes = [ctypes.POINTER(ctypes.c_int64),
                                        ctypes.c_int64,
                                        ctypes.POINTER(ctypes.c_int64),
                                        ctypes.c_int64,
                                        ctypes.POINTER(ctypes.c_uint32),
                                        ctypes.POINTER(ctypes.c_uint32)]
    b = np.ascontiguousarray(build, dtype=np.int64)
    p = np.ascontiguousarray(probe, dtype=np.int64)
    total = L.oracle_hashjoin_count(_p(b, ctypes.c_int64), len(b),
                                    _p(p, ctypes.c_int64), len(p))
    op = np.empty(total, dtype=np.uint32)
    ob = np.empty(total, dtype=np.uint32)
    L.oracle_hashjoin_pairs(_p(b, ctypes.c_int64), len(b),
                            _p(p, ctypes.c_int64), len(p),
                            _p(op, ctypes.c_uint32), _p(ob, ctypes.c_uint32))
    return op, ob


def hashagg_nulls(keys, aggs, n, mask=None):
    """Null-aware serial group-by restatement (SQL semantics, exact python
    ints): keys = [(array, valid_bool_or_None)], aggs = [(op, vals,
    valid_bool_or_None)] with op in {"sum", "min", "max"}.  NULL group keys
    group together (None in the key tuple); NULL aggregate inputs contribute
    nothing.  -> dict key_tuple -> (count, [acc-or-None], [nncnt]).
    Restates the reference's accumulators (datafusion/functions-aggregate
    sum.rs/min_max.rs: null input -> no update) and group_values null keys."""
    out = {}
    for i in range(n):
        if mask is not None and not mask[i]:
            continue
        kt = tuple(
            None if (kv is not None and not kv[i]) else
            (int(k[i]) if hasattr(k[i], "item") else k[i])
            for k, kv in keys)
        e = out.setdefault(kt, [0, [None] * len(aggs), [0] * len(aggs)])
        e[0] += 1
        for a, (op, vals, vv) in enumerate(aggs):
            if vv is not None and not vv[i]:
                continue
            v = int(vals[i])
            e[2][a] += 1
            cur = e[1][a]
            if op == "sum":
                e[1][a] = v if cur is None else cur + v
            elif op == "min":
                e[1][a] = v if cur is None else min(cur, v)
            elif op == "max":
                e[1][a] = v if cur is None else max(cur, v)
    return {k: (v[0], v[1], v[2]) for k, v in out.items()}


def hashagg(keys, aggs, n, mask=None):
    """Serial group-by restatement (exact python-int arithmetic).

    keys: list of numpy arrays (any comparable dtype incl. i64 views of
    dec128 pairs); aggs: list of ("sum", int-array-or-python-int-list);
    mask: optional boolean numpy array.
    -> dict key_tuple -> (count, [exact sums])."""
    out = {}
    for i in range(n):
        if mask is not None and not mask[i]:
            continue
        kt = tuple(int(k[i]) if hasattr(k[i], "item") else k[i] for k in keys)
        e = out.setdefault(kt, [0] + [0] * len(aggs))
        e[0] += 1
        for a, (_, vals) in enumerate(aggs):
            e[a + 1] += int(vals[i])
    return {k: (v[0], v[1:]) for k, v in out.items()}
