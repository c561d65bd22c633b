"""Independent brute-force cross-check of the ORACLE's running-window
semantics (NonFrameOverWindowExec.java:66-169): per-partition cumulative
accumulation over SORTED partitions with per-agg resetAccumulators,
vs a literal sequential restatement — COUNT never NULL, SUM0 init 0
with Java wrap-around, SUM (SUM_I64N) / MIN / MAX / AVG null-init,
f64 MIN/MAX via Java Math.min/max (NaN propagates, -0.0 < +0.0),
BIT_AND init all-ones.  Random chunk splits stress the cross-chunk
carry."""
import math
import numpy as np
import pytest

from galaxysql_amd import abi
from galaxysql_amd.chunk import I64, F64, rows_of, chunks_from_columns
from galaxysql_amd.operators import run_window

FUNCS = [abi.COUNT_ROW, abi.COUNT_COL, abi.SUM_I64, abi.SUM_I64N,
         abi.SUM_F64, abi.MIN_I64, abi.MAX_I64, abi.MIN_F64, abi.MAX_F64,
         abi.AVG_F64, abi.BIT_AND, abi.BIT_OR, abi.BIT_XOR]
F64_FUNCS = {abi.SUM_F64, abi.MIN_F64, abi.MAX_F64, abi.AVG_F64}


def _wrap(v):
    v &= (1 << 64) - 1
    return v - (1 << 64) if v >= (1 << 63) else v


def _jmin(a, b):
    if a != a:
        return a
    if b != b:
        return b
    if a == 0.0 and b == 0.0:
        return a if math.copysign(1.0, a) < 0 else b
    return a if a < b else b


def _jmax(a, b):
    if a != a:
        return a
    if b != b:
        return b
    if a == 0.0 and b == 0.0:
        return b if math.copysign(1.0, a) < 0 else a
    return a if a > b else b


def _brute(parts, ivals, inulls, fvals, fnulls, specs, resets):
    """One output row per input row: running value per agg."""
    out = []
    state = [None] * len(specs)
    cnt = [0] * len(specs)
    last = None
    for r in range(len(parts)):
        p = int(parts[r])
        if p != last:
            state = [None] * len(specs)
            cnt = [0] * len(specs)
            last = p
        row = []
        for a, func in enumerate(specs):
            if resets[a]:
                state[a] = None
                cnt[a] = 0
            iv = None if inulls[r] else int(ivals[r])
            fv = None if fnulls[r] else float(fvals[r])
            s = state[a]
            if func == abi.COUNT_ROW:
                cnt[a] += 1
                row.append(cnt[a])
                continue
            if func == abi.COUNT_COL:
                cnt[a] += iv is not None
                row.append(cnt[a])
                continue
            if func == abi.SUM_I64:
                state[a] = _wrap((s or 0) + (iv or 0))
                row.append(state[a])
                continue
            if func == abi.SUM_I64N:
                if iv is not None:
                    state[a] = _wrap((s or 0) + iv)
            elif func == abi.SUM_F64:
                if fv is not None:
                    state[a] = fv if s is None else s + fv
            elif func == abi.MIN_I64:
                if iv is not None:
                    state[a] = iv if s is None else min(s, iv)
            elif func == abi.MAX_I64:
                if iv is not None:
                    state[a] = iv if s is None else max(s, iv)
            elif func == abi.MIN_F64:
                if fv is not None:
                    state[a] = fv if s is None else _jmin(s, fv)
            elif func == abi.MAX_F64:
                if fv is not None:
                    state[a] = fv if s is None else _jmax(s, fv)
            elif func == abi.AVG_F64:
                if fv is not None:
                    t = s or (0.0, 0)
                    state[a] = (t[0] + fv, t[1] + 1)
            elif func in (abi.BIT_AND, abi.BIT_OR, abi.BIT_XOR):
                if iv is not None:
                    if s is None:
                        state[a] = iv
                    elif func == abi.BIT_AND:
                        state[a] = s & iv
                    elif func == abi.BIT_OR:
                        state[a] = s | iv
                    else:
                        state[a] = s ^ iv
            s = state[a]
            if func == abi.AVG_F64:
                row.append(None if s is None else s[0] / s[1])
            elif func in (abi.BIT_AND, abi.BIT_OR, abi.BIT_XOR):
                row.append(_wrap(-1 if func == abi.BIT_AND else 0)
                           if s is None else _wrap(s))
            else:
                row.append(s)
        out.append(row)
    return out


def _feq(a, b):
    if a is None or b is None:
        return a is None and b is None
    if a != a or b != b:
        return (a != a) == (b != b)
    if a == 0.0 and b == 0.0:
        return math.copysign(1.0, a) == math.copysign(1.0, b)
    return abs(a - b) <= 1e-9 * max(1.0, abs(a), abs(b))


@pytest.mark.parametrize("seed", range(24))
def test_oracle_window_vs_brute(seed):
    rng = np.random.default_rng(16000 + seed)
    lib = abi.load_oracle()
    n = int(rng.integers(1, 4000))
    parts = np.sort(rng.integers(0, max(n // 29, 1), n)).astype(np.int64)
    ivals = rng.integers(-100, 100, n).astype(np.int64) \
        if rng.random() < 0.7 else rng.integers(-(1 << 62), 1 << 62,
                                                n).astype(np.int64)
    inulls = (rng.random(n) < rng.choice([0.0, 0.25])).astype(np.uint8)
    pool = np.array([np.nan, -0.0, 0.0, 1.5, -2.25, 7.0])
    fvals = pool[rng.integers(0, len(pool), n)] if rng.random() < 0.4 \
        else np.round(rng.standard_normal(n) * 10, 3)
    fnulls = (rng.random(n) < 0.15).astype(np.uint8)

    n_aggs = int(rng.integers(1, 5))
    specs = [FUNCS[rng.integers(0, len(FUNCS))] for _ in range(n_aggs)]
    resets = [bool(rng.random() < 0.25) for _ in range(n_aggs)]
    aggs = [(f, -1 if f == abi.COUNT_ROW else (2 if f in F64_FUNCS else 1))
            for f in specs]

    types = [I64, I64, F64]
    chunks = chunks_from_columns(
        types, [(parts, None), (ivals, inulls if inulls.any() else None),
                (fvals, fnulls if fnulls.any() else None)],
        chunk_size=int(rng.integers(2, 1200)))
    got = rows_of(run_window(lib, [0], aggs, types, chunks, reset=resets,
                             device=-1))
    want = _brute(parts, ivals, inulls, fvals, fnulls, specs, resets)
    assert len(got) == len(want) == n, f"seed {seed}"
    for r in range(n):
        g = got[r][3:]  # input cols first, then one col per agg
        w = want[r]
        for a, func in enumerate(specs):
            if func in F64_FUNCS:
                ok = _feq(g[a], w[a])
            else:
                ok = g[a] == w[a]
            assert ok, (f"seed {seed} row {r} agg {a} func {func}: "
                        f"{g[a]} != {w[a]}")
