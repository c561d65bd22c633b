"""Independent brute-force cross-check of the ORACLE's group-by
aggregation semantics (AggOpenHashMap.putChunk + calc/aggfunctions/*):
random multi-key groups with nulls vs a dict-based restatement of each
aggregator's init/accumulate rule — COUNT(*)/COUNT(col) never NULL,
SUM0 init 0, SUM (SUM_I64N) and MIN/MAX/AVG init NULL, BIT_AND init
all-ones, Java wrap-around on int64 sums."""
import numpy as np
import pytest

from galaxysql_amd import abi
from galaxysql_amd.chunk import Block, Chunk, I64, I32, F64, multiset, \
    rows_of, chunks_from_columns
from galaxysql_amd.operators import run_agg

FUNCS = [abi.COUNT_ROW, abi.COUNT_COL, abi.SUM_I64, abi.SUM_I64N,
         abi.SUM_F64, abi.MIN_I64, abi.MAX_I64, abi.MIN_F64, abi.MAX_F64,
         abi.AVG_F64, abi.BIT_AND, abi.BIT_OR, abi.BIT_XOR]


def _wrap(v):
    v &= (1 << 64) - 1
    return v - (1 << 64) if v >= (1 << 63) else v


def _brute(groups, ivals, inulls, fvals, fnulls, aggs):
    state = {}
    for r in range(len(groups)):
        g = groups[r]
        st = state.setdefault(g, [None] * len(aggs))
        for a, (func, _) in enumerate(aggs):
            iv = None if inulls[r] else int(ivals[r])
            fv = None if fnulls[r] else float(fvals[r])
            s = st[a]
            if func == abi.COUNT_ROW:
                st[a] = (s or 0) + 1
            elif func == abi.COUNT_COL:
                st[a] = (s or 0) + (iv is not None)
            elif func == abi.SUM_I64:
                st[a] = _wrap((s or 0) + (iv or 0))
            elif func == abi.SUM_I64N:
                if iv is not None:
                    st[a] = _wrap((s or 0) + iv)
            elif func == abi.SUM_F64:
                if fv is not None:
                    st[a] = (s or 0.0) + fv
            elif func == abi.MIN_I64:
                if iv is not None:
                    st[a] = iv if s is None else min(s, iv)
            elif func == abi.MAX_I64:
                if iv is not None:
                    st[a] = iv if s is None else max(s, iv)
            elif func == abi.MIN_F64:
                if fv is not None:
                    st[a] = fv if s is None else min(s, fv)
            elif func == abi.MAX_F64:
                if fv is not None:
                    st[a] = fv if s is None else max(s, fv)
            elif func == abi.AVG_F64:
                if fv is not None:
                    t = s or (0.0, 0)
                    st[a] = (t[0] + fv, t[1] + 1)
            elif func == abi.BIT_AND:
                if iv is not None:
                    st[a] = iv if s is None else (s & iv)
            elif func == abi.BIT_OR:
                if iv is not None:
                    st[a] = iv if s is None else (s | iv)
            else:  # BIT_XOR
                if iv is not None:
                    st[a] = iv if s is None else (s ^ iv)
    rows = []
    for g, st in state.items():
        out = list(g)
        for a, (func, _) in enumerate(aggs):
            s = st[a]
            if func == abi.AVG_F64:
                out.append(None if s is None else s[0] / s[1])
            elif func in (abi.BIT_AND, abi.BIT_OR, abi.BIT_XOR):
                # never NULL: AND init all-ones, OR/XOR init 0
                if s is None:
                    s = -1 if func == abi.BIT_AND else 0
                out.append(_wrap(s))
            elif func in (abi.COUNT_ROW, abi.COUNT_COL, abi.SUM_I64):
                out.append(s or 0)
            else:
                out.append(s)
        rows.append(tuple(out))
    return rows


@pytest.mark.parametrize("seed", range(32))
def test_oracle_agg_vs_brute(seed):
    rng = np.random.default_rng(8000 + seed)
    lib = abi.load_oracle()
    n = int(rng.integers(1, 3000))
    n_keys = int(rng.integers(1, 3))
    card = int(rng.choice([1, 7, 200]))
    gcols = []
    for _ in range(n_keys):
        kv = rng.integers(0, card, n).astype(np.int64)
        kn = (rng.random(n) < rng.choice([0.0, 0.2])).astype(np.uint8)
        gcols.append((kv, kn))
    ivals = rng.integers(-(1 << 62), 1 << 62, n) \
        if rng.random() < 0.3 else rng.integers(-100, 100, n)
    ivals = ivals.astype(np.int64)
    inulls = (rng.random(n) < rng.choice([0.0, 0.3, 1.0])).astype(np.uint8)
    fvals = np.round(rng.standard_normal(n) * 10, 3)
    fnulls = (rng.random(n) < 0.2).astype(np.uint8)

    n_aggs = int(rng.integers(1, 5))
    aggs = []
    for _ in range(n_aggs):
        f = FUNCS[rng.integers(0, len(FUNCS))]
        col = n_keys + (1 if f in (abi.SUM_F64, abi.MIN_F64, abi.MAX_F64,
                                   abi.AVG_F64) else 0)
        aggs.append((f, -1 if f == abi.COUNT_ROW else col))

    types = [I64] * n_keys + [I64, F64]
    cols = [(kv, kn if kn.any() else None) for kv, kn in gcols]
    cols += [(ivals, inulls if inulls.any() else None),
             (fvals, fnulls if fnulls.any() else None)]
    chunks = chunks_from_columns(types, cols,
                                 chunk_size=int(rng.integers(100, 1500)))
    out = run_agg(lib, list(range(n_keys)), aggs, types, chunks)
    got = rows_of(out)

    groups = [tuple(None if gcols[k][1][r] else int(gcols[k][0][r])
                    for k in range(n_keys)) for r in range(n)]
    want = _brute(groups, ivals, inulls, fvals, fnulls, aggs)
    # float compare with rounding (accumulation order differs)
    assert multiset(got, f64_round=6) == multiset(want, f64_round=6), \
        f"seed {seed}"


def _nan_case(lib, device):
    from galaxysql_amd.operators import run_agg
    nan = float("nan")
    keys = np.array([1, 1, 1, 2, 2, 3, 3, 4], dtype=np.int64)
    vals = np.array([1.5, nan, 0.5, -0.0, 0.0, 0.0, -0.0, 2.0])
    chunk = Chunk([Block(I64, values=keys), Block(F64, values=vals)])
    out = run_agg(lib, [0], [(abi.MIN_F64, 1), (abi.MAX_F64, 1)],
                  [I64, F64], [chunk], device=device)
    rows = {}
    for c in out:
        for r in c.rows():
            rows[r[0]] = (r[1], r[2])
    return rows


def test_agg_minmax_f64_java_nan_semantics_oracle():
    """Math.min/max semantics (Double2DoubleMin.java:40-44): NaN
    propagates; -0.0 < +0.0."""
    import math
    import struct as st
    rows = _nan_case(abi.load_oracle(), -1)
    assert math.isnan(rows[1][0]) and math.isnan(rows[1][1])
    # group 2: min = -0.0 (signbit set), max = +0.0
    assert rows[2][0] == 0.0 and st.pack("<d", rows[2][0])[7] == 0x80
    assert rows[2][1] == 0.0 and st.pack("<d", rows[2][1])[7] == 0x00
    # group 3 (order flipped): same answers
    assert st.pack("<d", rows[3][0])[7] == 0x80
    assert st.pack("<d", rows[3][1])[7] == 0x00
    assert rows[4] == (2.0, 2.0)


@pytest.mark.gpu
def test_agg_minmax_f64_java_nan_semantics_gpu():
    import math
    import struct as st
    rows = _nan_case(abi.load_hip(), 0)
    assert math.isnan(rows[1][0]) and math.isnan(rows[1][1])
    assert st.pack("<d", rows[2][0])[7] == 0x80
    assert st.pack("<d", rows[2][1])[7] == 0x00
    assert st.pack("<d", rows[3][0])[7] == 0x80
    assert st.pack("<d", rows[3][1])[7] == 0x00


def _global_agg_cases(lib, device):
    from galaxysql_amd.chunk import Chunk
    # (a) zero input chunks  (b) block-less chunks carrying only a row count
    a = run_agg(lib, group_cols=[], aggs=[(abi.COUNT_ROW, -1),
                                          (abi.SUM_I64N, 0)],
                input_types=[I64], input_chunks=[], device=device)
    b = run_agg(lib, group_cols=[], aggs=[(abi.COUNT_ROW, -1)],
                input_types=[], input_chunks=[Chunk([], n_rows=123),
                                              Chunk([], n_rows=77)],
                device=device)
    return rows_of(a), rows_of(b)


def test_global_agg_empty_input_oracle():
    """SQL global aggregate over empty input emits ONE row: COUNT(*)=0,
    null-init SUM stays NULL (HashAggExec no-group-by contract); a
    block-less chunk still carries positionCount (Chunk.java:55-89).
    Deep-fuzz seed 100229 caught the HIP side emitting nothing."""
    a, b = _global_agg_cases(abi.load_oracle(), -1)
    assert a == [(0, None)]
    assert b == [(200,)]


@pytest.mark.gpu
def test_global_agg_empty_input_gpu():
    a, b = _global_agg_cases(abi.load_hip(), 0)
    assert a == [(0, None)]
    assert b == [(200,)]


def test_global_agg_two_phase_with_empty_rank_partial():
    """Two-phase global aggregate: a rank with NO input emits the SQL
    partial row (COUNT(*)=0, SUM NULL); the final re-agg (COUNT->SUM0,
    SUM->SUM over partials, final_agg_specs) must absorb it without
    changing the total."""
    from galaxysql_amd.chunk import Chunk, Block
    from galaxysql_amd.exchange import final_agg_specs
    lib = abi.load_oracle()
    aggs = [(abi.COUNT_ROW, -1), (abi.SUM_I64N, 0)]
    # rank 0: rows [3, null, 4]; rank 1: empty input
    r0 = run_agg(lib, group_cols=[], aggs=aggs, input_types=[I64],
                 input_chunks=[Chunk([Block(I64,
                                            values=np.array([3, 0, 4],
                                                            dtype=np.int64),
                                            nulls=np.array([0, 1, 0],
                                                           dtype=np.uint8))])],
                 device=-1)
    r1 = run_agg(lib, group_cols=[], aggs=aggs, input_types=[I64],
                 input_chunks=[], device=-1)
    partials = rows_of(r0) + rows_of(r1)
    assert partials == [(3, 7), (0, None)]
    finals, _types = final_agg_specs(0, aggs)
    # partial output schema: [count, sum]
    cnt = np.array([p[0] for p in partials], dtype=np.int64)
    sm = np.array([0 if p[1] is None else p[1] for p in partials],
                  dtype=np.int64)
    smn = np.array([1 if p[1] is None else 0 for p in partials],
                   dtype=np.uint8)
    ch = Chunk([Block(I64, values=cnt), Block(I64, values=sm, nulls=smn)])
    out = run_agg(lib, group_cols=[], aggs=finals, input_types=[I64, I64],
                  input_chunks=[ch], device=-1)
    assert rows_of(out) == [(3, 7)]
