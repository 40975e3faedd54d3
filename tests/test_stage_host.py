"""CPU-side checks of the C++ stage interpreter (no GPU):
  - the C++ RecordBatch metadata builder is byte-identical to the Python
    ipc.py builder (which tests/test_shuffle_format.py validates against
    pyarrow's own reader)
  - bg_stage_validate type-checks plan JSON and reports output schemas
    (the CPU-testable half of the plan grammar)
"""
import pyarrow as pa
import pytest

from datafusion_ballista_amd import ipc as bgipc
from datafusion_ballista_amd import stage


def test_rb_message_matches_python_builder():
    cases = [
        (8192, [(8192, 0)], [(0, 0), (0, 65536)], 65552, True),
        (5, [(5, 2), (5, 0)], [(0, 8), (8, 0), (8, 24), (32, 48)], 80, True),
        (1, [(1, 0)], [(0, 0), (0, 16)], 16, True),
        (100, [(100, 0), (100, 1), (100, 100)],
         [(0, 0), (0, 800), (800, 13), (816, 404), (1224, 3)], 1232, True),
        (7, [(7, 0)], [(0, 0), (0, 56)], 56, False),
    ]
    for n_rows, nodes, bufs, body_len, comp in cases:
        want = bgipc.record_batch_message(n_rows, nodes, bufs, body_len,
                                          compressed=comp)
        got = stage.debug_rb_message(n_rows, nodes, bufs, body_len, comp)
        assert got == want, (n_rows, nodes, bufs, body_len, comp)


def _scan(table_name="t", schema=None):
    schema = schema or [{"name": "k", "dtype": "int64"},
                        {"name": "v", "dtype": "decimal128",
                         "precision": 15, "scale": 2},
                        {"name": "d", "dtype": "date32"}]
    return {"op": "scan", "schema": schema,
            "source": {"kind": "device", "table": table_name}}


def _doc(plan, **kw):
    d = {"job_id": "j", "stage_id": 1, "task_id": 0, "work_dir": "/tmp/w",
         "plan": plan}
    d.update(kw)
    return d


def test_validate_collect_schema():
    res = stage.validate(_doc({"op": "collect", "input": _scan()}))
    assert res["ok"] is True
    assert [f["name"] for f in res["schema"]] == ["k", "v", "d"]
    assert [f["dtype"] for f in res["schema"]] == \
        ["int64", "decimal128", "date32"]


def test_validate_projection_and_filter():
    plan = {"op": "collect", "input": {
        "op": "filter",
        "predicates": [{"col": "d", "cmp": "ge_lt", "lo": 100, "hi": 200}],
        "input": {"op": "project", "exprs": [
            {"as": "rev", "expr": {"mul": [{"col": "v"}, {"col": "v"}]}},
            {"as": "d", "expr": {"col": "d"}},
        ], "input": _scan()}}}
    # filter references 'd' which project keeps; 'rev' is dec128
    res = stage.validate(_doc(plan))
    assert [f["name"] for f in res["schema"]] == ["rev", "d"]
    assert res["schema"][0]["dtype"] == "decimal128"


def test_validate_aggregate_modes():
    base = {"op": "hash_aggregate", "group_by": ["k"],
            "aggs": [{"fn": "sum", "as": "s",
                      "expr": {"col": "v"}},
                     {"fn": "count", "as": "c"},
                     {"fn": "avg", "as": "a", "expr": {"col": "v"}}],
            "input": _scan()}
    single = dict(base, mode="single")
    res = stage.validate(_doc({"op": "collect", "input": single}))
    assert [f["name"] for f in res["schema"]] == ["k", "s", "c", "a"]

    partial = dict(base, mode="partial")
    res = stage.validate(_doc({"op": "collect", "input": partial}))
    assert [f["name"] for f in res["schema"]] == \
        ["k", "s", "s$n", "c", "a$s", "a$n"]

    # final consumes the partial schema by name
    final_scan = {"op": "scan", "schema": [
        {"name": "k", "dtype": "int64"},
        {"name": "s", "dtype": "decimal128", "precision": 38, "scale": 2},
        {"name": "s$n", "dtype": "int64"},
        {"name": "c", "dtype": "int64"},
        {"name": "a$s", "dtype": "decimal128", "precision": 38, "scale": 2},
        {"name": "a$n", "dtype": "int64"}],
        "source": {"kind": "device", "table": "partials"}}
    final = {"op": "hash_aggregate", "mode": "final", "group_by": ["k"],
             "aggs": [{"fn": "sum", "as": "s"}, {"fn": "count", "as": "c"},
                      {"fn": "avg", "as": "a"}], "input": final_scan}
    res = stage.validate(_doc({"op": "collect", "input": final}))
    assert [f["name"] for f in res["schema"]] == ["k", "s", "c", "a"]


def test_validate_sort_shuffle_write_requires_schema_msg():
    plan = {"op": "sort_shuffle_write", "k": 16,
            "keys": [{"col": "k"}], "input": _scan()}
    msg = stage.validate_error(_doc(plan))
    assert "schema_msg_hex" in msg

    schema = pa.schema([("k", pa.int64()),
                        ("v", pa.decimal128(15, 2)), ("d", pa.date32())])
    res = stage.validate(_doc(plan, schema_msg_hex=stage.schema_msg_hex(
        schema)))
    assert res["ok"] is True


def test_validate_errors_name_the_problem():
    msg = stage.validate_error(_doc({"op": "collect", "input": {
        "op": "filter", "predicates": [
            {"col": "nope", "cmp": "lt", "hi": 3}],
        "input": _scan()}}))
    assert "nope" in msg

    msg = stage.validate_error(_doc({"op": "collect", "input": {
        "op": "frobnicate", "input": _scan()}}))
    assert "frobnicate" in msg

    msg = stage.validate_error(_doc({"op": "collect", "input": {
        "op": "hash_aggregate", "mode": "single", "group_by": [],
        "aggs": [{"fn": "median", "as": "m", "expr": {"col": "v"}}],
        "input": _scan()}}))
    assert "median" in msg


def test_validate_join_schema():
    build = _scan("c", [{"name": "c_custkey", "dtype": "int64"},
                        {"name": "seg", "dtype": "dict8"}])
    probe = _scan("o", [{"name": "o_orderkey", "dtype": "int64"},
                        {"name": "o_custkey", "dtype": "int64"}])
    plan = {"op": "collect", "input": {
        "op": "hash_join", "build": build, "probe": probe,
        "build_keys": ["c_custkey"], "probe_keys": ["o_custkey"],
        "join_type": "inner",
        "output": [{"side": "probe", "col": "o_orderkey"},
                   {"side": "build", "col": "seg", "as": "c_seg"}]}}
    res = stage.validate(_doc(plan))
    assert [f["name"] for f in res["schema"]] == ["o_orderkey", "c_seg"]


def test_validate_new_grammar():
    """Round-2 grammar: LIKE / IN / OR-groups / CASE / parquet scan
    validate host-side (no GPU)."""
    sch = [{"name": "s", "dtype": "utf8"}, {"name": "m", "dtype": "dict8"},
           {"name": "v", "dtype": "decimal128", "precision": 15,
            "scale": 2}]
    sc = {"op": "scan", "schema": sch,
          "source": {"kind": "device", "table": "t"}}
    res = stage.validate(_doc({"op": "collect", "input": {
        "op": "filter", "predicates": [
            {"col": "s", "like": "%green%"},
            {"col": "m", "in": [1, 2, 3]}], "input": sc}}))
    assert res["ok"] is True

    res = stage.validate(_doc({"op": "collect", "input": {
        "op": "filter", "any": [
            [{"col": "m", "in": [1]}],
            [{"col": "s", "like": "PROMO%"}]], "input": sc}}))
    assert res["ok"] is True

    res = stage.validate(_doc({"op": "collect", "input": {
        "op": "project", "exprs": [{"as": "c", "expr": {"case": {
            "when": [{"col": "s", "like": "PROMO%"}],
            "then": {"col": "v"}, "else": {"lit": 0}}}}],
        "input": sc}}))
    assert res["ok"] is True
    assert res["schema"][0]["dtype"] == "decimal128"

    # parquet scan source kind accepted
    res = stage.validate(_doc({"op": "collect", "input": {
        "op": "scan", "schema": [{"name": "a", "dtype": "int64"}],
        "source": {"kind": "parquet", "path": "/nonexistent",
                   "columns": []}}}))
    assert res["ok"] is True

    # unknown column inside an OR group still errors
    msg = stage.validate_error(_doc({"op": "collect", "input": {
        "op": "filter", "any": [[{"col": "nope", "in": [1]}]],
        "input": sc}}))
    assert "nope" in msg


def test_validate_bad_json():
    import ctypes
    import json
    L = stage._lib()
    out = ctypes.c_char_p()
    rc = L.bg_stage_validate(b'{"plan": [unterminated', ctypes.byref(out))
    assert rc != 0
    assert b"parse error" in L.bg_last_error()
