"""Randomized differential sweep: HIP vs oracle over random operator
configurations (join type matrix x key shapes x null densities x hybrid
spill x agg function subsets). Seeds are fixed — failures reproduce.

This is breadth insurance on top of the targeted parity tests: every
iteration builds a fresh random config, runs both implementations on the
same data, and compares row multisets (f64 rounded to 1e-9)."""
import numpy as np
import pytest

from galaxysql_amd import abi
from galaxysql_amd.chunk import Block, Chunk, I64, I32, F64, SLICE, \
    chunks_from_columns, multiset, rows_of
from galaxysql_amd.operators import (EquiJoinKey, run_join, run_agg,
                                     run_groupjoin, run_window)

pytestmark = pytest.mark.gpu


def _close(a, b):
    if a is None or b is None:
        return a is b
    return abs(a - b) <= 1e-9 * max(1.0, abs(a), abs(b))


def assert_rows_match(got, want, ctx):
    """Exact columns via multiset; float columns via RELATIVE tolerance
    (atomic-order fp drift on a 50K-value SUM(double) exceeds any absolute
    rounding — fuzz seed 2010 caught exactly that)."""
    from collections import defaultdict

    def split(rows):
        g = defaultdict(list)
        for r in rows:
            exact = tuple(v for v in r if not isinstance(v, float))
            floats = tuple(v for v in r if isinstance(v, float))
            g[exact].append(floats)
        return g

    def kkey(t):
        return tuple((0, "") if v is None else
                     (1, v.decode("latin1")) if isinstance(v, bytes) else
                     (2, str(v)) for v in t)

    gg, ww = split(got), split(want)
    assert sorted(gg.keys(), key=kkey) == sorted(ww.keys(), key=kkey), ctx
    for k in gg:
        a = sorted(gg[k], key=lambda t: [(-1e300 if x is None else x)
                                         for x in t])
        b = sorted(ww[k], key=lambda t: [(-1e300 if x is None else x)
                                         for x in t])
        assert len(a) == len(b), (ctx, k)
        for fa, fb in zip(a, b):
            for x, y in zip(fa, fb):
                assert _close(x, y), (ctx, k, fa, fb)

JOIN_TYPES = [abi.INNER, abi.LEFT, abi.RIGHT, abi.SEMI, abi.ANTI]
KEY_TYPES = [I64, I32, SLICE]
PAYLOADS = [I64, I32, F64, SLICE]
AGG_FUNCS = [(abi.COUNT_ROW, None), (abi.COUNT_COL, I64),
             (abi.SUM_I64, I64), (abi.SUM_F64, F64),
             (abi.MIN_I64, I64), (abi.MAX_I64, I64),
             (abi.MIN_F64, F64), (abi.MAX_F64, F64),
             (abi.AVG_F64, F64), (abi.AVG_F64, I64),
             (abi.BIT_AND, I64), (abi.BIT_OR, I64), (abi.BIT_XOR, I64)]


def _col(rng, t, n, card, null_frac):
    nulls = (rng.random(n) < null_frac).astype(np.uint8) \
        if null_frac > 0 else None
    if t == SLICE:
        # value 0 maps to the EMPTY string: "" is a real key, equal to
        # itself and distinct from NULL (Block.equals byte semantics)
        vals = [None if nulls is not None and nulls[i]
                else ("" if (x := int(rng.integers(0, card))) == 0
                      else f"v{x}") for i in range(n)]
        return Block.of(SLICE, vals)
    if t == I32:
        v = rng.integers(-card, card, n).astype(np.int32)
    elif t == F64:
        v = np.round(rng.standard_normal(n) * card, 3)
    else:
        v = rng.integers(-card, card, n)
    return Block(t, values=v, nulls=nulls)


def _side(rng, key_types, n, card, null_frac, extra_payloads):
    types = list(key_types)
    blocks = [_col(rng, t, n, card, null_frac) for t in key_types]
    for t in extra_payloads:
        types.append(t)
        blocks.append(_col(rng, t, n, 1000, null_frac / 2))
    chunk_size = int(rng.integers(8, 2000))
    chunks = []
    for start in range(0, n, chunk_size):
        sub = []
        for b in blocks:
            if b.type == SLICE:
                base = int(b.offsets[start - 1]) if start > 0 else 0
                end = min(start + chunk_size, n)
                off = (b.offsets[start:end] - base).astype(np.int32)
                dend = int(b.offsets[end - 1]) if end > 0 else 0
                sub.append(Block(SLICE,
                                 nulls=None if b.nulls is None
                                 else b.nulls[start:end],
                                 offsets=off, data=b.data[base:dend]))
            else:
                end = min(start + chunk_size, n)
                sub.append(Block(b.type, values=b.values[start:end],
                                 nulls=None if b.nulls is None
                                 else b.nulls[start:end]))
        chunks.append(Chunk(sub, n_rows=min(start + chunk_size, n) - start)
                      if not sub else Chunk(sub))
    return types, chunks


@pytest.mark.parametrize("seed", range(24))
def test_fuzz_join(seed):
    rng = np.random.default_rng(1000 + seed)
    hip = abi.load_hip()
    ora = abi.load_oracle()
    n_keys = int(rng.integers(1, 3))
    key_types = [KEY_TYPES[rng.integers(0, len(KEY_TYPES))]
                 for _ in range(n_keys)]
    jt = JOIN_TYPES[rng.integers(0, len(JOIN_TYPES))]
    null_frac = float(rng.choice([0.0, 0.05, 0.3]))
    card = int(rng.choice([5, 100, 5000]))
    n_build = int(rng.integers(0, 4000))
    n_probe = int(rng.integers(1, 8000))
    bp = [PAYLOADS[rng.integers(0, len(PAYLOADS))]]
    pp = [PAYLOADS[rng.integers(0, len(PAYLOADS))]]
    btypes, build = _side(rng, key_types, n_build, card, null_frac, bp)
    ptypes, probe = _side(rng, key_types, n_probe, card, null_frac, pp)
    keys = [EquiJoinKey(i, i, key_types[i]) for i in range(n_keys)]
    kw = {}
    anti_ok = jt == abi.ANTI and n_keys == 1 and len(btypes) == 1
    if anti_ok and rng.random() < 0.5:
        kw["anti_null_col"] = 0
    if anti_ok:
        btypes, build = _side(rng, key_types, n_build, card, null_frac, [])
    if rng.random() < 0.3 and jt in (abi.INNER, abi.LEFT):
        kw["build_outer"] = True
        # build side is the outer side: swap type roles
        got = run_join(hip, jt, keys, build, probe, btypes, ptypes,
                       device=0, expected_build_rows=n_build,
                       memory_budget_bytes=4096 if rng.random() < 0.5 else 0,
                       **kw)
        want = run_join(ora, jt, keys, build, probe, btypes, ptypes,
                        device=-1, **kw)
    else:
        got = run_join(hip, jt, keys, build, probe, ptypes, btypes,
                       device=0, expected_build_rows=n_build,
                       memory_budget_bytes=4096 if rng.random() < 0.5 else 0,
                       **kw)
        want = run_join(ora, jt, keys, build, probe, ptypes, btypes,
                        device=-1, **kw)
    assert_rows_match(rows_of(got), rows_of(want), f"join seed {seed}")


@pytest.mark.parametrize("seed", range(16))
def test_fuzz_agg(seed):
    rng = np.random.default_rng(2000 + seed)
    hip = abi.load_hip()
    ora = abi.load_oracle()
    n = int(rng.integers(1, 50000))
    n_group = int(rng.integers(0, 3))
    gtypes = [KEY_TYPES[rng.integers(0, len(KEY_TYPES))]
              for _ in range(n_group)]
    null_frac = float(rng.choice([0.0, 0.2]))
    card = int(rng.choice([3, 50, 2000]))
    n_aggs = int(rng.integers(1, 5))
    picks = [AGG_FUNCS[rng.integers(0, len(AGG_FUNCS))]
             for _ in range(n_aggs)]
    # input: group cols then one value col per agg that needs one
    types = list(gtypes)
    aggs = []
    extra = []
    for f, vt in picks:
        if vt is None:
            aggs.append((f, -1))
        else:
            aggs.append((f, len(gtypes) + len(extra)))
            extra.append(vt)
    types2, chunks = _side(rng, gtypes, n, card, null_frac, extra)
    kw = dict(group_cols=list(range(n_group)), aggs=aggs,
              expected_groups=int(rng.choice([0, 10, 100000])))
    got = run_agg(hip, input_types=types2, input_chunks=chunks, device=0,
                  **kw)
    want = run_agg(ora, input_types=types2, input_chunks=chunks, device=-1,
                   **kw)
    assert_rows_match(rows_of(got), rows_of(want), f"agg seed {seed}")


@pytest.mark.parametrize("seed", range(4))
def test_fuzz_groupjoin_window(seed):
    rng = np.random.default_rng(3000 + seed)
    hip = abi.load_hip()
    ora = abi.load_oracle()
    # group-join
    n_b, n_p = int(rng.integers(1, 3000)), int(rng.integers(1, 6000))
    card = int(rng.choice([10, 500]))
    btypes, build = _side(rng, [I64], n_b, card, 0.1, [I64])
    ptypes, probe = _side(rng, [I64], n_p, card, 0.1, [I64, F64])
    jt = abi.INNER if rng.random() < 0.5 else abi.LEFT
    kw = dict(group_cols=[0, 1],
              aggs=[(abi.COUNT_ROW, -1), (abi.SUM_I64, 1),
                    (abi.SUM_F64, 2)])
    got = run_groupjoin(hip, jt, [EquiJoinKey(0, 0, I64)], build, probe,
                        btypes, ptypes, device=0, **kw)
    want = run_groupjoin(ora, jt, [EquiJoinKey(0, 0, I64)], build, probe,
                         btypes, ptypes, device=-1, **kw)
    assert_rows_match(rows_of(got), rows_of(want), f"gj seed {seed}")

    # window (sorted partitions)
    n = int(rng.integers(1, 20000))
    parts = np.sort(rng.integers(0, max(n // 17, 1), n)).astype(np.int64)
    vals = rng.integers(-100, 100, n)
    nulls = (rng.random(n) < 0.15).astype(np.uint8)
    chunks = chunks_from_columns([I64, I64], [(parts, None), (vals, nulls)],
                                 chunk_size=int(rng.integers(2, 3000)))
    aggs = [(abi.COUNT_ROW, -1), (abi.SUM_I64, 1), (abi.MIN_I64, 1)]
    reset = [bool(rng.random() < 0.3) for _ in aggs]
    got = run_window(hip, [0], aggs, [I64, I64], chunks, reset=reset,
                     device=0)
    want = run_window(ora, [0], aggs, [I64, I64], chunks, reset=reset,
                      device=-1)
    assert rows_of(got) == rows_of(want), f"seed {seed} win"


@pytest.mark.parametrize("seed", range(16))
def test_fuzz_join_conditions(seed):
    """Random residual (non-equi) conditions on top of random joins:
    HIP vs oracle (checkJoinCondition semantics, gx_join_cond)."""
    from galaxysql_amd.operators import JoinCond
    rng = np.random.default_rng(3000 + seed)
    hip = abi.load_hip()
    ora = abi.load_oracle()
    key_types = [KEY_TYPES[rng.integers(0, len(KEY_TYPES))]]
    jt = [abi.INNER, abi.LEFT, abi.SEMI, abi.ANTI][rng.integers(0, 4)]
    null_frac = float(rng.choice([0.0, 0.2]))
    card = int(rng.choice([10, 300]))
    n_build = int(rng.integers(1, 3000))
    n_probe = int(rng.integers(1, 6000))
    bp = [PAYLOADS[rng.integers(0, 3)]]  # numeric payloads for conds
    pp = [PAYLOADS[rng.integers(0, 3)]]
    btypes, build = _side(rng, key_types, n_build, card, null_frac, bp)
    ptypes, probe = _side(rng, key_types, n_probe, card, null_frac, pp)
    keys = [EquiJoinKey(0, 0, key_types[0])]

    # condition row = outer (probe side) cols then inner (build side) cols
    row_types = ptypes + btypes
    numeric = [i for i, t in enumerate(row_types) if t in (I64, I32, F64)]
    conds = []
    for _ in range(int(rng.integers(1, 3))):
        a = int(rng.choice(numeric))
        cmp = int(rng.choice([abi.LT, abi.LE, abi.GT, abi.GE, abi.EQ,
                              abi.NE, abi.EQ_NULLSAFE, abi.NE_NULLSAFE]))
        same_t = [i for i in numeric if row_types[i] == row_types[a]
                  and i != a]
        if same_t and rng.random() < 0.4:
            conds.append(JoinCond(a, cmp, int(rng.choice(same_t))))
        else:
            v = float(rng.integers(-50, 400)) if row_types[a] == F64 \
                else int(rng.integers(-50, 400))
            if rng.random() < 0.1:
                v = None  # SQL NULL constant
            conds.append(JoinCond(a, cmp, -1, v))

    kw = dict(conds=conds, enable_bloom=bool(rng.random() < 0.5))
    if rng.random() < 0.3:
        # out-of-core hybrid join must honor residual conditions too
        kw["memory_budget_bytes"] = 4096
    got = run_join(hip, jt, keys, build, probe, ptypes, btypes,
                   device=0, expected_build_rows=n_build, **kw)
    want = run_join(ora, jt, keys, build, probe, ptypes, btypes,
                    device=-1, **kw)
    assert_rows_match(rows_of(got), rows_of(want), f"cond seed {seed}")


@pytest.mark.gpu
@pytest.mark.parametrize("seed", range(6))
def test_fuzz_f64_minmax_nan_signed_zero(seed):
    """MIN/MAX(F64) over payloads laced with NaN and -0.0: HIP vs oracle,
    both implementing Java Math.min/max (Double2DoubleMin.java:40-44 —
    NaN propagates, -0.0 < +0.0). multiset keeps -0.0 distinct from
    +0.0 and canonicalizes NaN; only the quiet NaN is injected so the
    propagated bits are order-independent."""
    from galaxysql_amd.operators import run_fwindow
    rng = np.random.default_rng(12000 + seed)
    hip = abi.load_hip()
    ora = abi.load_oracle()
    n = int(rng.integers(16, 20000))
    card = int(rng.choice([3, 50]))
    keys = rng.integers(0, card, n).astype(np.int64)
    pool = np.array([np.nan, -0.0, 0.0, 1.5, -2.25, 7.0, -0.0, np.nan])
    vals = pool[rng.integers(0, len(pool), n)]
    nulls = (rng.random(n) < 0.15).astype(np.uint8)
    types = [I64, F64]
    cols = [(keys, None), (vals, nulls if nulls.any() else None)]
    chunks = chunks_from_columns(types, cols,
                                 chunk_size=int(rng.integers(64, 4096)))
    kw = dict(group_cols=[0], aggs=[(abi.MIN_F64, 1), (abi.MAX_F64, 1)])
    got = run_agg(hip, input_types=types, input_chunks=chunks, device=0,
                  **kw)
    want = run_agg(ora, input_types=types, input_chunks=chunks, device=-1,
                   **kw)
    assert multiset(rows_of(got), f64_sign_zero=True) == \
        multiset(rows_of(want), f64_sign_zero=True), f"agg seed {seed}"

    # NaN-free run so every group exercises the -0.0 < +0.0 ordering
    # (with NaN in the pool nearly all groups collapse to NaN)
    zpool = np.array([-0.0, 0.0, -0.0, 0.0, 3.5])
    zvals = zpool[rng.integers(0, len(zpool), n)]
    zchunks = chunks_from_columns(types, [(keys, None), (zvals, None)],
                                  chunk_size=2048)
    gz = run_agg(hip, input_types=types, input_chunks=zchunks, device=0,
                 **kw)
    wz = run_agg(ora, input_types=types, input_chunks=zchunks, device=-1,
                 **kw)
    assert multiset(rows_of(gz), f64_sign_zero=True) == \
        multiset(rows_of(wz), f64_sign_zero=True), f"agg-z seed {seed}"

    # running window over sorted partitions: output is positional, so tag
    # each row with its index before the multiset compare
    sp = np.sort(keys)
    wchunks = chunks_from_columns(types, [(sp, None), cols[1]],
                                  chunk_size=int(rng.integers(64, 4096)))
    waggs = [(abi.MIN_F64, 1), (abi.MAX_F64, 1)]
    gw = run_window(hip, [0], waggs, types, wchunks, device=0)
    ww = run_window(ora, [0], waggs, types, wchunks, device=-1)
    tag = lambda rows: [(i,) + r for i, r in enumerate(rows)]
    assert multiset(tag(rows_of(gw)), f64_sign_zero=True) == \
        multiset(tag(rows_of(ww)), f64_sign_zero=True), f"win seed {seed}"

    # sliding frame MIN/MAX via the sparse table (jmin over tree combine)
    fr = [(abi.MIN_F64, 1, abi.FRAME_ROWS_SLIDING, 3, 2),
          (abi.MAX_F64, 1, abi.FRAME_ROWS_SLIDING, 2, 4)]
    gf = run_fwindow(hip, [0], fr, types, wchunks, device=0)
    wf = run_fwindow(ora, [0], fr, types, wchunks, device=-1)
    assert multiset(tag(rows_of(gf)), f64_sign_zero=True) == \
        multiset(tag(rows_of(wf)), f64_sign_zero=True), f"fw seed {seed}"

    # fused group-join MIN/MAX(F64) accumulates per build position
    nb = int(rng.integers(1, 800))
    bkeys = rng.integers(0, card, nb).astype(np.int64)
    bchunks = chunks_from_columns([I64], [(bkeys, None)], chunk_size=512)
    gjkw = dict(group_cols=[0], aggs=[(abi.MIN_F64, 1), (abi.MAX_F64, 1)])
    gj = run_groupjoin(hip, abi.LEFT, [EquiJoinKey(0, 0, I64)], bchunks,
                       chunks, [I64], types, device=0, **gjkw)
    wj = run_groupjoin(ora, abi.LEFT, [EquiJoinKey(0, 0, I64)], bchunks,
                       chunks, [I64], types, device=-1, **gjkw)
    assert multiset(rows_of(gj), f64_sign_zero=True) == \
        multiset(rows_of(wj), f64_sign_zero=True), f"gj seed {seed}"


@pytest.mark.gpu
@pytest.mark.parametrize("seed", range(4))
def test_fuzz_f64_join_conditions_nan(seed):
    """Residual f64 conditions (col-vs-const and col-vs-col) with NaN and
    -0.0 in the compared payloads: HIP vs oracle. Both use raw IEEE
    compares (Java primitive <, AbstractJoinExec.checkJoinCondition
    :227-250): NaN fails <,<=,>,>=,=; NaN != NaN is true; -0.0 == 0.0."""
    from galaxysql_amd.operators import JoinCond
    rng = np.random.default_rng(13000 + seed)
    hip = abi.load_hip()
    ora = abi.load_oracle()
    jt = [abi.INNER, abi.LEFT, abi.SEMI, abi.ANTI][rng.integers(0, 4)]
    n_b, n_p = int(rng.integers(1, 2000)), int(rng.integers(1, 5000))
    card = int(rng.choice([5, 100]))
    pool = np.array([np.nan, -0.0, 0.0, 1.5, -3.75, 10.0])

    def side(n):
        k = rng.integers(0, card, n).astype(np.int64)
        f = pool[rng.integers(0, len(pool), n)]
        fn = (rng.random(n) < 0.1).astype(np.uint8)
        return chunks_from_columns(
            [I64, F64], [(k, None), (f, fn if fn.any() else None)],
            chunk_size=int(rng.integers(100, 3000)))

    build, probe = side(n_b), side(n_p)
    # condition row: probe cols (0,1) then build cols (2,3)
    pick = rng.random()
    if pick < 0.4:
        conds = [JoinCond(1, abi.LT, -1, float(rng.choice([0.0, -0.0, 2.0])))]
    elif pick < 0.7:
        conds = [JoinCond(1, abi.NE, 3)]
    else:
        conds = [JoinCond(1, abi.GE, 3),
                 JoinCond(3, abi.LE, -1, 5.0)]
    kw = dict(conds=conds)
    keys = [EquiJoinKey(0, 0, I64)]
    got = run_join(hip, jt, keys, build, probe, [I64, F64], [I64, F64],
                   device=0, **kw)
    want = run_join(ora, jt, keys, build, probe, [I64, F64], [I64, F64],
                    device=-1, **kw)
    assert multiset(rows_of(got), f64_sign_zero=True) == \
        multiset(rows_of(want), f64_sign_zero=True), f"seed {seed} jt {jt}"


@pytest.mark.gpu
@pytest.mark.parametrize("seed", range(4))
def test_fuzz_f64_sum_frames_nonfinite(seed):
    """Sliding SUM/AVG(F64) frames with NaN and +-Inf in the column: the
    HIP path computes frames by segmented-prefix DIFFERENCE, which a
    non-finite value poisons for every later frame in the partition
    (NaN sticks, inf-inf=NaN) even when the frame excludes it; the
    reference rescans each frame. k_fw_diff_f64 falls back to a direct
    rescan whenever the diff is non-finite — a finite diff implies an
    all-finite frame. HIP vs oracle, ROWS and RANGE bounds."""
    from galaxysql_amd.operators import run_fwindow
    rng = np.random.default_rng(14000 + seed)
    hip = abi.load_hip()
    ora = abi.load_oracle()
    n = int(rng.integers(64, 12000))
    parts = np.sort(rng.integers(0, max(n // 40, 1), n)).astype(np.int64)
    pool = np.array([np.nan, np.inf, -np.inf, 1.5, -2.25, 3.0, 0.5, 4.0])
    # mostly-finite: non-finite ~6% so many frames are clean (fast path)
    # and partitions still get poisoned tails (rescue path)
    pick = rng.random(n)
    vals = pool[3 + rng.integers(0, 5, n)]
    vals[pick < 0.06] = pool[rng.integers(0, 3, (pick < 0.06).sum())]
    nulls = (rng.random(n) < 0.1).astype(np.uint8)
    order = np.arange(n, dtype=np.int64)  # strictly increasing per part
    types = [I64, I64, F64]
    chunks = chunks_from_columns(
        types, [(parts, None), (order, None),
                (vals, nulls if nulls.any() else None)],
        chunk_size=int(rng.integers(100, 4000)))
    fr = [(abi.SUM_F64, 2, abi.FRAME_ROWS_SLIDING, int(rng.integers(0, 6)),
           int(rng.integers(0, 6))),
          (abi.AVG_F64, 2, abi.FRAME_ROWS_SLIDING, n, 0),
          (abi.SUM_F64, 2, abi.FRAME_RANGE_SLIDING, 7, 7, 1, 1)]
    got = run_fwindow(hip, [0], fr, types, chunks, device=0)
    want = run_fwindow(ora, [0], fr, types, chunks, device=-1)
    tag = lambda rows: [(i,) + r for i, r in enumerate(rows)]
    assert multiset(tag(rows_of(got)), f64_round=6) == \
        multiset(tag(rows_of(want)), f64_round=6), f"seed {seed}"
