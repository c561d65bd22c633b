"""Helpers to load tests/golden/*.json fixtures into Chunks and drive a
gxop implementation (oracle or HIP) through the operator lifecycle."""
import json
import os

from galaxysql_amd import abi
from galaxysql_amd.chunk import Block, Chunk, I64, I32, F64, SLICE, multiset
from galaxysql_amd.operators import EquiJoinKey, JoinCond, run_join, run_agg

GOLDEN = os.path.join(os.path.dirname(os.path.abspath(__file__)), "golden")

TYPE_BY_NAME = {"i64": I64, "i32": I32, "f64": F64, "slice": SLICE}
JOIN_BY_NAME = {"INNER": abi.INNER, "LEFT": abi.LEFT, "RIGHT": abi.RIGHT,
                "SEMI": abi.SEMI, "ANTI": abi.ANTI}
CMP_BY_NAME = {"LT": abi.LT, "LE": abi.LE, "GT": abi.GT, "GE": abi.GE,
               "EQ": abi.EQ, "NE": abi.NE,
               "EQ_NULLSAFE": abi.EQ_NULLSAFE, "NE_NULLSAFE": abi.NE_NULLSAFE}
AGG_BY_NAME = {"COUNT_ROW": abi.COUNT_ROW, "COUNT_COL": abi.COUNT_COL,
               "SUM_I64": abi.SUM_I64, "SUM_I64N": abi.SUM_I64N,
               "SUM_F64": abi.SUM_F64,
               "MIN_I64": abi.MIN_I64, "MAX_I64": abi.MAX_I64,
               "MIN_F64": abi.MIN_F64, "MAX_F64": abi.MAX_F64}


def load_cases(fname):
    with open(os.path.join(GOLDEN, fname)) as f:
        return json.load(f)["cases"]


def chunks_of(spec):
    types = [TYPE_BY_NAME[t] for t in spec["types"]]
    out = []
    for ch in spec["chunks"]:
        out.append(Chunk([Block.of(t, col) for t, col in zip(types, ch)]))
    return out, types


def expected_rows(cols):
    if not cols or not cols[0]:
        return []
    n = len(cols[0])
    return [tuple(_canon(c[i]) for c in cols) for i in range(n)]


def _canon(v):
    if isinstance(v, str):
        return v.encode()
    return v


def run_join_case(lib, case, device=-1, stream=0):
    build_chunks, inner_types = chunks_of(case["inner"])
    probe_chunks, outer_types = chunks_of(case["outer"])
    keys = [EquiJoinKey(k[0], k[1], TYPE_BY_NAME[k[2]]) for k in case["keys"]]
    conds = [JoinCond(c[0], CMP_BY_NAME[c[1]], c[2], c[3])
             for c in case.get("conds", [])] or None
    out = run_join(lib, JOIN_BY_NAME[case["join_type"]], keys,
                   build_chunks, probe_chunks, outer_types, inner_types,
                   max_one_row=bool(case.get("single")),
                   anti_null_col=case.get("anti_null_col", -1),
                   device=device, stream=stream, conds=conds)
    rows = []
    for c in out:
        rows.extend(c.rows())
    return rows


def check_join_case(lib, case, **kw):
    if case.get("expect_error"):
        import pytest
        with pytest.raises(RuntimeError):
            run_join_case(lib, case, **kw)
        return
    rows = run_join_case(lib, case, **kw)
    exp = expected_rows(case.get("expected", []))
    assert multiset(rows) == multiset(exp), \
        f"{case['name']}: got {sorted(multiset(rows).items())[:20]} expected {sorted(multiset(exp).items())[:20]}"


def run_agg_case(lib, case, device=-1, stream=0):
    input_chunks, input_types = chunks_of(case["input"])
    aggs = [(AGG_BY_NAME[f], col) for f, col in case["aggs"]]
    out = run_agg(lib, case["group_cols"], aggs, input_types, input_chunks,
                  device=device, stream=stream)
    rows = []
    for c in out:
        rows.extend(c.rows())
    return rows


def check_agg_case(lib, case, **kw):
    rows = run_agg_case(lib, case, **kw)
    exp = expected_rows(case.get("expected", []))
    assert multiset(rows, f64_round=9) == multiset(exp, f64_round=9), \
        f"{case['name']}: got {sorted(multiset(rows).items())[:20]} expected {sorted(multiset(exp).items())[:20]}"
