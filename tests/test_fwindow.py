"""Frame windows (OverWindowFramesExec subset) — hand cases + numpy
cross-check on the oracle, HIP-vs-oracle parity on GPU.

Semantics pinned: whole-partition totals on every row (UnboundedOverFrame),
ROWS BETWEEN p PRECEDING AND f FOLLOWING clamped to the partition
(RowSlidingOverFrame), CURRENT ROW..UNBOUNDED FOLLOWING
(RowUnboundedFollowingOverFrame). Emission preserves input order."""
import numpy as np
import pytest

from galaxysql_amd import abi
from galaxysql_amd.chunk import Block, Chunk, I64, F64, SLICE, \
    chunks_from_columns, \
    rows_of
from galaxysql_amd.operators import run_fwindow


def test_fwindow_hand_case():
    lib = abi.load_oracle()
    part = [1, 1, 1, 2, 2]
    val = [10, None, 5, 7, 1]
    chunks = [Chunk([Block.of(I64, part), Block.of(I64, val)])]
    out = run_fwindow(
        lib, [0],
        [(abi.SUM_I64, 1, abi.FRAME_WHOLE_PARTITION),
         (abi.MIN_I64, 1, abi.FRAME_WHOLE_PARTITION),
         (abi.SUM_I64, 1, abi.FRAME_ROWS_SLIDING, 1, 1),
         (abi.COUNT_ROW, -1, abi.FRAME_ROWS_SLIDING, 0, 1),
         (abi.SUM_I64, 1, abi.FRAME_ROWS_UNBOUNDED_FOLLOWING)],
        [I64, I64], chunks)
    rows = rows_of(out)
    assert rows == [
        # part, val, whole_sum, whole_min, slide+-1, cnt_0_1, to_end
        (1, 10, 15, 5, 10, 2, 15),
        (1, None, 15, 5, 15, 2, 5),
        (1, 5, 15, 5, 5, 1, 5),
        (2, 7, 8, 1, 8, 2, 8),
        (2, 1, 8, 1, 8, 1, 1),
    ]


def test_fwindow_rejects_unsupported():
    lib = abi.load_oracle()
    # (sliding SUM_F64 and RANGE frames became supported in round 2)
    with pytest.raises(RuntimeError):
        # MIN in an unbounded-following frame is still unsupported
        run_fwindow(lib, [0],
                    [(abi.MIN_I64, 1, abi.FRAME_ROWS_UNBOUNDED_FOLLOWING)],
                    [I64, I64], [])
    with pytest.raises(RuntimeError):
        # RANGE frame with a non-numeric order col
        # (RangeSlidingOverFrame.java:36: numeric types only)
        run_fwindow(lib, [0],
                    [(abi.SUM_I64N, 1, abi.FRAME_RANGE_SLIDING, 1, 1, 2, 1)],
                    [I64, I64, SLICE], [])


def _gen(rng, n):
    parts = np.sort(rng.integers(0, n // 23 + 1, n)).astype(np.int64)
    vals = rng.integers(-50, 50, n).astype(np.int64)
    nulls = (rng.random(n) < 0.1).astype(np.uint8)
    fvals = rng.random(n) * 7
    return chunks_from_columns([I64, I64, F64],
                               [(parts, None), (vals, nulls),
                                (fvals, None)], chunk_size=997), parts, \
        vals, nulls


def test_fwindow_oracle_vs_numpy():
    lib = abi.load_oracle()
    rng = np.random.default_rng(51)
    chunks, parts, vals, nulls = _gen(rng, 4000)
    out = run_fwindow(lib, [0],
                      [(abi.SUM_I64, 1, abi.FRAME_WHOLE_PARTITION),
                       (abi.SUM_I64, 1, abi.FRAME_ROWS_SLIDING, 2, 3)],
                      [I64, I64, F64], chunks)
    rows = rows_of(out)
    v = np.where(nulls == 1, 0, vals)
    for i, r in enumerate(rows):
        seg = parts == parts[i]
        idx = np.nonzero(seg)[0]
        assert r[3] == int(v[seg].sum()), i
        lo = max(idx[0], i - 2)
        hi = min(idx[-1], i + 3)
        assert r[4] == int(v[lo:hi + 1].sum()), i


FRAMES = [(abi.SUM_I64, 1, abi.FRAME_WHOLE_PARTITION),
          (abi.AVG_F64, 1, abi.FRAME_WHOLE_PARTITION),
          (abi.COUNT_COL, 1, abi.FRAME_WHOLE_PARTITION),
          (abi.SUM_F64, 2, abi.FRAME_WHOLE_PARTITION),
          (abi.MAX_I64, 1, abi.FRAME_WHOLE_PARTITION),
          (abi.SUM_I64, 1, abi.FRAME_ROWS_SLIDING, 3, 2),
          (abi.COUNT_ROW, -1, abi.FRAME_ROWS_SLIDING, 0, 0),
          (abi.MIN_I64, 1, abi.FRAME_ROWS_SLIDING, 4, 1),
          (abi.MAX_F64, 2, abi.FRAME_ROWS_SLIDING, 2, 6),
          (abi.MIN_F64, 2, abi.FRAME_ROWS_SLIDING, 0, 3),
          (abi.SUM_I64, 1, abi.FRAME_ROWS_UNBOUNDED_FOLLOWING)]


@pytest.mark.gpu
def test_gpu_fwindow_matches_oracle():
    hip = abi.load_hip()
    ora = abi.load_oracle()
    rng = np.random.default_rng(52)
    chunks, *_ = _gen(rng, 30000)
    got = run_fwindow(hip, [0], FRAMES, [I64, I64, F64], chunks, device=0)
    want = run_fwindow(ora, [0], FRAMES, [I64, I64, F64], chunks, device=-1)
    grows = rows_of(got)
    wrows = rows_of(want)
    assert len(grows) == len(wrows)
    for i, (g, w) in enumerate(zip(grows, wrows)):
        for a, b in zip(g, w):
            if isinstance(a, float) and b is not None:
                assert abs(a - b) < 1e-9, (i, g, w)
            else:
                assert a == b, (i, g, w)


def test_navigation_hand_case_oracle():
    """FIRST/LAST/NTH_VALUE, LAG/LEAD, NTILE, CUME_DIST, PERCENT_RANK
    (calc/aggfunctions/FirstValue|Lag|NTile|CumeDist|PercentRank.java),
    whole-partition navigation over buffered input — strings included
    (index gathers)."""
    from galaxysql_amd.chunk import SLICE
    lib = abi.load_oracle()
    part = [1] * 5 + [2] * 2
    order = [3, 3, 5, 5, 9, 1, 1]
    name = ["a", "b", "c", "d", "e", "f", "g"]
    chunks = [Chunk([Block.of(I64, part), Block.of(I64, order),
                     Block.of(SLICE, name)])]
    W = abi.FRAME_WHOLE_PARTITION
    out = rows_of(run_fwindow(
        lib, [0],
        [(abi.FIRST_VALUE, 2, W), (abi.LAST_VALUE, 2, W),
         (abi.NTH_VALUE, 2, W, 2), (abi.LAG, 1, W, 1),
         (abi.LEAD, 2, W, 2), (abi.NTILE, -1, W, 2),
         (abi.CUME_DIST, -1, W), (abi.PERCENT_RANK, -1, W)],
        [I64, I64, SLICE], chunks, order_cols=[1]))
    assert out == [
        (1, 3, b"a", b"a", b"e", b"b", None, b"c", 1, 0.4, 0.0),
        (1, 3, b"b", b"a", b"e", b"b", 3, b"d", 1, 0.4, 0.0),
        (1, 5, b"c", b"a", b"e", b"b", 3, b"e", 1, 0.8, 0.5),
        (1, 5, b"d", b"a", b"e", b"b", 5, None, 2, 0.8, 0.5),
        (1, 9, b"e", b"a", b"e", b"b", 5, None, 2, 1.0, 1.0),
        (2, 1, b"f", b"f", b"g", b"g", None, None, 1, 1.0, 0.0),
        (2, 1, b"g", b"f", b"g", b"g", 1, None, 2, 1.0, 0.0),
    ]


NAV_FRAMES = None


@pytest.mark.gpu
def test_gpu_navigation_matches_oracle():
    from galaxysql_amd.chunk import SLICE
    hip = abi.load_hip()
    ora = abi.load_oracle()
    rng = np.random.default_rng(53)
    n = 20000
    parts = np.sort(rng.integers(0, n // 31, n)).astype(np.int64)
    order = np.concatenate([np.sort(rng.integers(0, 7, (parts == p).sum()))
                            for p in np.unique(parts)]).astype(np.int64)
    vals = rng.integers(-50, 50, n)
    vnulls = (rng.random(n) < 0.1).astype(np.uint8)
    names = Block.of(SLICE, [f"n{int(v) % 13}" for v in vals])
    chunks = chunks_from_columns(
        [I64, I64, I64, SLICE],
        [(parts, None), (order, None), (vals, vnulls), names],
        chunk_size=777)
    W = abi.FRAME_WHOLE_PARTITION
    frames = [(abi.FIRST_VALUE, 2, W), (abi.LAST_VALUE, 3, W),
              (abi.NTH_VALUE, 3, W, 3), (abi.LAG, 2, W, 2),
              (abi.LEAD, 3, W, 1), (abi.NTILE, -1, W, 4),
              (abi.CUME_DIST, -1, W), (abi.PERCENT_RANK, -1, W)]
    got = rows_of(run_fwindow(hip, [0], frames, [I64, I64, I64, SLICE],
                              chunks, order_cols=[1], device=0))
    want = rows_of(run_fwindow(ora, [0], frames, [I64, I64, I64, SLICE],
                               chunks, order_cols=[1], device=-1))
    assert len(got) == len(want)
    for i, (g, w) in enumerate(zip(got, want)):
        for a, b in zip(g, w):
            if isinstance(a, float) and b is not None:
                assert abs(a - b) < 1e-12, (i, g, w)
            else:
                assert a == b, (i, g, w)


def test_ntile_cume_percent_vs_numpy():
    """NTILE/CUME_DIST/PERCENT_RANK against direct numpy formulas."""
    lib = abi.load_oracle()
    rng = np.random.default_rng(54)
    n = 3000
    parts = np.sort(rng.integers(0, 60, n)).astype(np.int64)
    order = np.concatenate([np.sort(rng.integers(0, 5, (parts == p).sum()))
                            for p in np.unique(parts)]).astype(np.int64)
    chunks = chunks_from_columns([I64, I64], [(parts, None), (order, None)])
    W = abi.FRAME_WHOLE_PARTITION
    rows = rows_of(run_fwindow(lib, [0],
                               [(abi.NTILE, -1, W, 3),
                                (abi.CUME_DIST, -1, W),
                                (abi.PERCENT_RANK, -1, W)],
                               [I64, I64], chunks, order_cols=[1]))
    for i, r in enumerate(rows):
        seg = np.nonzero(parts == parts[i])[0]
        size = len(seg)
        pos = i - seg[0]
        base, rem = divmod(size, 3)
        if base == 0:
            exp_tile = pos + 1
        elif pos < rem * (base + 1):
            exp_tile = pos // (base + 1) + 1
        else:
            exp_tile = rem + (pos - rem * (base + 1)) // base + 1
        run = seg[order[seg] == order[i]]
        exp_cume = (run[-1] - seg[0] + 1) / size
        exp_pct = 0.0 if size == 1 else (run[0] - seg[0]) / (size - 1)
        assert r[2] == exp_tile, i
        assert abs(r[3] - exp_cume) < 1e-12, i
        assert abs(r[4] - exp_pct) < 1e-12, i


@pytest.mark.parametrize("seed", range(12))
def test_navigation_oracle_vs_brute(seed):
    """FIRST/LAST/NTH_VALUE, LAG/LEAD over random partitions with NULLs
    and string payloads (index-gather semantics, FirstValue/Lag.java):
    NTH's k and LAG/LEAD offsets are random; out-of-partition indexes
    and NULL source values both emit NULL."""
    from galaxysql_amd.chunk import SLICE, chunks_from_columns
    rng = np.random.default_rng(17000 + seed)
    lib = abi.load_oracle()
    n = int(rng.integers(1, 2500))
    parts = np.sort(rng.integers(0, max(n // 31, 1), n)).astype(np.int64)
    raw = [None if rng.random() < 0.15 else f"s{rng.integers(0, 50)}"
           for _ in range(n)]
    k_nth = int(rng.integers(1, 6))
    k_lag = int(rng.integers(1, 4))
    k_lead = int(rng.integers(1, 4))
    W = abi.FRAME_WHOLE_PARTITION
    chunks = chunks_from_columns(
        [I64, SLICE], [(parts, None), Block.of(SLICE, raw)],
        chunk_size=int(rng.integers(2, 900)))
    rows = rows_of(run_fwindow(
        lib, [0],
        [(abi.FIRST_VALUE, 1, W), (abi.LAST_VALUE, 1, W),
         (abi.NTH_VALUE, 1, W, k_nth), (abi.LAG, 1, W, k_lag),
         (abi.LEAD, 1, W, k_lead)],
        [I64, SLICE], chunks))
    enc = [None if v is None else v.encode() for v in raw]
    for i in range(n):
        seg = np.nonzero(parts == parts[i])[0]
        s, e = int(seg[0]), int(seg[-1])
        got = rows[i][2:]
        want = (
            enc[s],
            enc[e],
            enc[s + k_nth - 1] if s + k_nth - 1 <= e else None,
            enc[i - k_lag] if i - k_lag >= s else None,
            enc[i + k_lead] if i + k_lead <= e else None,
        )
        assert got == want, (seed, i, got, want)
