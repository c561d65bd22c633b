"""Fused group-join (HashGroupJoinExec) — oracle semantics + HIP parity.

Key semantic differences from the plain hash join, all pinned here:
- one group per CONSUMED-side row position (duplicate keys stay separate
  groups: HashGroupJoinExec.buildOneChunk:296-311 puts every position);
- NULL join keys match NULL join keys (matching is Chunk.equals via
  ElementaryChunksIndex.equals:166-175 — null-safe — not
  ExecUtils.buildOneChunk's null skip);
- LEFT emits unmatched groups after one null-row accumulation
  (buildNullRow: COUNT(*)=1, COUNT(col)=0, SUM/MIN/MAX=NULL)."""
import numpy as np
import pytest

from galaxysql_amd import abi
from galaxysql_amd.chunk import Block, Chunk, I64, F64, SLICE, \
    chunks_from_columns, multiset, rows_of
from galaxysql_amd.operators import (EquiJoinKey, run_groupjoin, run_join,
                                     run_agg)


def _oracle():
    return abi.load_oracle()


def test_groupjoin_hand_case():
    lib = _oracle()
    # orders (okey, custkey); lineitem (okey, qty)
    build = [Chunk([Block.of(I64, [10, 20, 30]),
                    Block.of(I64, [1, 2, 3])])]
    probe = [Chunk([Block.of(I64, [10, 10, 30, 40]),
                    Block.of(I64, [5, 7, 9, 100])])]
    out = run_groupjoin(lib, abi.INNER, [EquiJoinKey(0, 0, I64)],
                        build, probe, [I64, I64], [I64, I64],
                        group_cols=[0, 1],
                        aggs=[(abi.COUNT_ROW, -1), (abi.SUM_I64, 1)])
    assert multiset(rows_of(out)) == multiset([(10, 1, 2, 12), (30, 3, 1, 9)])


def test_groupjoin_left_unmatched():
    lib = _oracle()
    build = [Chunk([Block.of(I64, [10, 20])])]
    probe = [Chunk([Block.of(I64, [10]), Block.of(I64, [5])])]
    out = run_groupjoin(lib, abi.LEFT, [EquiJoinKey(0, 0, I64)],
                        build, probe, [I64], [I64, I64],
                        group_cols=[0],
                        aggs=[(abi.COUNT_ROW, -1), (abi.COUNT_COL, 1),
                              (abi.SUM_I64, 1), (abi.SUM_F64, 1)])
    # matched group: count*=1, count(col)=1, sum=5, fsum=5.0
    # unmatched: null row -> count*=1, count(col)=0, sum=0 (Sum0), fsum=NULL
    assert multiset(rows_of(out)) == multiset([(10, 1, 1, 5, 5.0),
                                               (20, 1, 0, 0, None)])


def test_groupjoin_duplicate_build_keys_stay_separate_groups():
    lib = _oracle()
    build = [Chunk([Block.of(I64, [7, 7]), Block.of(I64, [100, 200])])]
    probe = [Chunk([Block.of(I64, [7, 7, 7]), Block.of(I64, [1, 2, 3])])]
    out = run_groupjoin(lib, abi.INNER, [EquiJoinKey(0, 0, I64)],
                        build, probe, [I64, I64], [I64, I64],
                        group_cols=[1], aggs=[(abi.SUM_I64, 1)])
    # BOTH build rows match all 3 probe rows independently
    assert multiset(rows_of(out)) == multiset([(100, 6), (200, 6)])


def test_groupjoin_null_keys_match_null_safe():
    lib = _oracle()
    build = [Chunk([Block.of(I64, [5, None]), Block.of(I64, [1, 2])])]
    probe = [Chunk([Block.of(I64, [5, None, None]),
                    Block.of(I64, [10, 20, 30])])]
    out = run_groupjoin(lib, abi.INNER, [EquiJoinKey(0, 0, I64)],
                        build, probe, [I64, I64], [I64, I64],
                        group_cols=[1], aggs=[(abi.COUNT_ROW, -1),
                                              (abi.SUM_I64, 1)])
    # build NULL key groups the two NULL probe rows (Chunk.equals null-safe)
    assert multiset(rows_of(out)) == multiset([(1, 1, 10), (2, 2, 50)])


def test_groupjoin_equals_join_then_agg_when_keys_unique():
    """With unique consumed-side keys, groupjoin == agg(join()) grouped by
    the key — the plan equivalence the reference's planner exploits."""
    lib = _oracle()
    rng = np.random.default_rng(21)
    n_orders, n_items = 700, 5000
    okeys = rng.permutation(10000)[:n_orders].astype(np.int64)  # unique
    custs = rng.integers(0, 50, n_orders)
    ikeys = rng.integers(0, 10000, n_items).astype(np.int64)
    qty = rng.integers(1, 100, n_items)
    build = chunks_from_columns([I64, I64], [(okeys, None), (custs, None)])
    probe = chunks_from_columns([I64, I64], [(ikeys, None), (qty, None)])

    gj = run_groupjoin(lib, abi.INNER, [EquiJoinKey(0, 0, I64)],
                       build, probe, [I64, I64], [I64, I64],
                       group_cols=[0], aggs=[(abi.COUNT_ROW, -1),
                                             (abi.SUM_I64, 1)])
    # reference composition: INNER join (probe outer) then agg by okey
    joined = run_join(lib, abi.INNER, [EquiJoinKey(0, 0, I64)], build, probe,
                      [I64, I64], [I64, I64])
    # joined schema: probe cols (ikey, qty) then build cols (okey, cust)
    agg = run_agg(lib, [2], [(abi.COUNT_ROW, -1), (abi.SUM_I64, 1)],
                  [I64, I64, I64, I64], joined)
    assert multiset(rows_of(gj)) == multiset(rows_of(agg))


def _random_case(rng, with_slice=False, null_frac=0.0):
    n_b, n_p = 3000, 20000
    bk = rng.integers(0, 2000, n_b).astype(np.int64)
    bn = (rng.random(n_b) < null_frac).astype(np.uint8) if null_frac else None
    pay = rng.integers(0, 10**6, n_b)
    pk = rng.integers(0, 2500, n_p).astype(np.int64)
    pn = (rng.random(n_p) < null_frac).astype(np.uint8) if null_frac else None
    val = rng.integers(-1000, 1000, n_p)
    fval = rng.random(n_p) * 100
    btypes = [I64, I64]
    bcols = [(bk, bn), (pay, None)]
    if with_slice:
        btypes.append(SLICE)
        bcols.append(Block.of(SLICE, [f"g{int(k) % 31}" for k in bk]))
    build = chunks_from_columns(btypes, bcols)
    probe = chunks_from_columns([I64, I64, F64],
                                [(pk, pn), (val, None), (fval, None)])
    return btypes, build, probe


AGGS = [(abi.COUNT_ROW, -1), (abi.COUNT_COL, 1), (abi.SUM_I64, 1),
        (abi.SUM_F64, 2), (abi.MIN_I64, 1), (abi.MAX_F64, 2)]


@pytest.mark.gpu
@pytest.mark.parametrize("join_type", [abi.INNER, abi.LEFT])
@pytest.mark.parametrize("null_frac", [0.0, 0.15])
def test_gpu_groupjoin_matches_oracle(join_type, null_frac):
    hip = abi.load_hip()
    ora = _oracle()
    rng = np.random.default_rng(31)
    btypes, build, probe = _random_case(rng, with_slice=True,
                                        null_frac=null_frac)
    kw = dict(group_cols=[1, 2], aggs=AGGS)
    got = run_groupjoin(hip, join_type, [EquiJoinKey(0, 0, I64)], build,
                        probe, btypes, [I64, I64, F64], device=0, **kw)
    want = run_groupjoin(ora, join_type, [EquiJoinKey(0, 0, I64)], build,
                         probe, btypes, [I64, I64, F64], device=-1, **kw)
    assert multiset(rows_of(got), f64_round=6) == \
        multiset(rows_of(want), f64_round=6)


@pytest.mark.gpu
def test_gpu_groupjoin_multikey_and_empty():
    hip = abi.load_hip()
    ora = _oracle()
    rng = np.random.default_rng(32)
    n_b, n_p = 1000, 8000
    b1 = rng.integers(0, 40, n_b).astype(np.int64)
    b2 = rng.integers(0, 40, n_b).astype(np.int32)
    p1 = rng.integers(0, 50, n_p).astype(np.int64)
    p2 = rng.integers(0, 50, n_p).astype(np.int32)
    from galaxysql_amd.chunk import I32
    build = chunks_from_columns([I64, I32], [(b1, None), (b2, None)])
    probe = chunks_from_columns([I64, I32, I64],
                                [(p1, None), (p2, None),
                                 (rng.integers(0, 100, n_p), None)])
    keys = [EquiJoinKey(0, 0, I64), EquiJoinKey(1, 1, I32)]
    kw = dict(group_cols=[0, 1], aggs=[(abi.COUNT_ROW, -1),
                                       (abi.SUM_I64, 2)])
    got = run_groupjoin(hip, abi.INNER, keys, build, probe, [I64, I32],
                        [I64, I32, I64], device=0, **kw)
    want = run_groupjoin(ora, abi.INNER, keys, build, probe, [I64, I32],
                         [I64, I32, I64], device=-1, **kw)
    assert multiset(rows_of(got)) == multiset(rows_of(want))

    # empty build INNER -> nothing
    got2 = run_groupjoin(hip, abi.INNER, [EquiJoinKey(0, 0, I64)],
                         [], probe, [I64, I32], [I64, I32, I64],
                         device=0, group_cols=[0],
                         aggs=[(abi.COUNT_ROW, -1)])
    assert rows_of(got2) == []
