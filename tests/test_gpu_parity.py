"""GPU parity: the HIP path (libgxhip.so on gfx950) must produce the same
result-row multisets as the CPU oracle on the same seeded inputs
(SURVEY.md §8c — bit-exact on INT/BIGINT and COUNT/SUM(BIGINT); DOUBLE
aggregates compared at rel-tol 1e-9 via rounding since GPU atomic add order
is nondeterministic, per the north_star's stated tolerance)."""
import numpy as np
import pytest

from galaxysql_amd import abi
from galaxysql_amd.chunk import Block, Chunk, I64, I32, F64, chunks_from_columns, multiset
from galaxysql_amd.operators import EquiJoinKey, run_join, run_agg, PartitioningExchanger

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def libs():
    import subprocess, os
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    subprocess.run(["make", "-C", os.path.join(repo, "oracle")], check=True,
                   capture_output=True)
    return abi.load_oracle(), abi.load_hip()


def rand_col(rng, btype, n, key_space=None, null_frac=0.0):
    if btype == I64:
        vals = rng.integers(0, key_space or 1 << 40, size=n, dtype=np.int64)
    elif btype == I32:
        vals = rng.integers(-(key_space or 1 << 20), key_space or 1 << 20,
                            size=n, dtype=np.int32)
    else:
        vals = rng.standard_normal(n)
    nulls = None
    if null_frac > 0:
        nulls = (rng.random(n) < null_frac).astype(np.uint8)
    return vals, nulls


def make_chunks(rng, types, n, key_space=None, null_frac=0.0, chunk_size=997):
    cols = [rand_col(rng, t, n, key_space, null_frac) for t in types]
    return chunks_from_columns(types, cols, chunk_size=chunk_size)


def both_join(libs, join_type, keys, build, probe, ot, it, **kw):
    oracle, hip = libs
    ref = run_join(oracle, join_type, keys, build, probe, ot, it, device=-1, **kw)
    got = run_join(hip, join_type, keys, build, probe, ot, it, device=0, **kw)
    r = []
    for c in ref:
        r.extend(c.rows())
    g = []
    for c in got:
        g.extend(c.rows())
    return r, g


JOIN_TYPES = [abi.INNER, abi.LEFT, abi.RIGHT, abi.SEMI, abi.ANTI]


@pytest.mark.parametrize("join_type", JOIN_TYPES)
def test_join_types_i64_key(libs, join_type):
    rng = np.random.default_rng(42)
    # duplicate keys on both sides + nulls on key columns
    build = make_chunks(rng, [I64, I64], 5000, key_space=2000, null_frac=0.02)
    probe = make_chunks(rng, [I64, I64, F64], 20000, key_space=2500, null_frac=0.02)
    keys = [EquiJoinKey(0, 0, I64)]
    ref, got = both_join(libs, join_type, keys, build, probe,
                         [I64, I64, F64], [I64, I64])
    assert multiset(got) == multiset(ref)


def test_join_multikey_generic(libs):
    rng = np.random.default_rng(7)
    build = make_chunks(rng, [I32, I64, F64], 3000, key_space=40, null_frac=0.05)
    probe = make_chunks(rng, [I64, I32, I64], 9000, key_space=40, null_frac=0.05)
    # keys: probe(1:i32,2:i64) == build(0:i32,1:i64)
    keys = [EquiJoinKey(1, 0, I32), EquiJoinKey(2, 1, I64)]
    ref, got = both_join(libs, abi.INNER, keys, build, probe,
                         [I64, I32, I64], [I32, I64, F64])
    assert multiset(got) == multiset(ref)


def test_join_left_multikey(libs):
    rng = np.random.default_rng(8)
    build = make_chunks(rng, [I32, I64], 1000, key_space=30, null_frac=0.1)
    probe = make_chunks(rng, [I32, I64], 5000, key_space=35, null_frac=0.1)
    keys = [EquiJoinKey(0, 0, I32), EquiJoinKey(1, 1, I64)]
    ref, got = both_join(libs, abi.LEFT, keys, build, probe,
                         [I32, I64], [I32, I64])
    assert multiset(got) == multiset(ref)


def test_join_empty_build(libs):
    rng = np.random.default_rng(9)
    probe = make_chunks(rng, [I64, I64], 500, key_space=100, null_frac=0.1)
    keys = [EquiJoinKey(0, 0, I64)]
    for jt in JOIN_TYPES:
        ref, got = both_join(libs, jt, keys, [], probe, [I64, I64], [I64])
        assert multiset(got) == multiset(ref), f"join_type={jt}"


def test_join_anti_not_in_null_build(libs):
    rng = np.random.default_rng(10)
    build = make_chunks(rng, [I64], 100, key_space=50, null_frac=0.1)
    probe = make_chunks(rng, [I64, I64], 500, key_space=60, null_frac=0.1)
    keys = [EquiJoinKey(0, 0, I64)]
    ref, got = both_join(libs, abi.ANTI, keys, build, probe,
                         [I64, I64], [I64], anti_null_col=0)
    assert multiset(got) == multiset(ref)


def test_join_single_ok_and_error(libs):
    rng = np.random.default_rng(11)
    # unique build keys -> ok
    bvals = np.arange(1000, dtype=np.int64)
    rng.shuffle(bvals)
    build = chunks_from_columns([I64, F64], [(bvals, None),
                                             (rng.standard_normal(1000), None)])
    probe = make_chunks(rng, [I64, I64], 3000, key_space=1500)
    keys = [EquiJoinKey(0, 0, I64)]
    ref, got = both_join(libs, abi.LEFT, keys, build, probe,
                         [I64, I64], [I64, F64], max_one_row=True)
    assert multiset(got) == multiset(ref)

    # duplicate build keys -> both sides must raise
    build_dup = chunks_from_columns([I64, F64],
                                    [(np.zeros(10, np.int64), None),
                                     (np.zeros(10), None)])
    probe0 = chunks_from_columns([I64, I64], [(np.zeros(5, np.int64), None),
                                              (np.arange(5, dtype=np.int64), None)])
    oracle, hip = libs
    for lib, dev in ((oracle, -1), (hip, 0)):
        with pytest.raises(RuntimeError):
            run_join(lib, abi.LEFT, keys, build_dup, probe0, [I64, I64],
                     [I64, F64], max_one_row=True, device=dev)


def test_join_build_outer(libs):
    """buildOuter: the build side is the preserved outer side; unmatched
    build rows drain via the tail (ParallelHashJoinExec.nextJoinNullRows)."""
    rng = np.random.default_rng(12)
    build = make_chunks(rng, [I64, I64], 2000, key_space=1000, null_frac=0.05)
    probe = make_chunks(rng, [I64, F64], 3000, key_space=800, null_frac=0.05)
    keys = [EquiJoinKey(0, 0, I64)]
    ref, got = both_join(libs, abi.LEFT, keys, build, probe,
                         [I64, I64], [I64, F64], build_outer=True)
    assert multiset(got) == multiset(ref)


def test_join_large_fast_path(libs):
    """Larger stress on the single-i64-key fast path with skewed duplicates."""
    rng = np.random.default_rng(13)
    n_build, n_probe = 100_000, 400_000
    bkeys = rng.integers(0, 60_000, size=n_build, dtype=np.int64)
    build = chunks_from_columns(
        [I64, I64], [(bkeys, None),
                     (rng.integers(0, 1 << 30, n_build, dtype=np.int64), None)],
        chunk_size=10_000)
    pkeys = rng.integers(0, 70_000, size=n_probe, dtype=np.int64)
    probe = chunks_from_columns(
        [I64, I64], [(pkeys, None),
                     (rng.integers(0, 1 << 30, n_probe, dtype=np.int64), None)],
        chunk_size=100_000)
    keys = [EquiJoinKey(0, 0, I64)]
    ref, got = both_join(libs, abi.INNER, keys, build, probe, [I64, I64], [I64, I64])
    # multiset compare on 4-col rows; sizes ~n_probe*avg_dup
    assert len(ref) == len(got)
    assert multiset(got) == multiset(ref)


# ---- agg ----

def both_agg(libs, group_cols, aggs, types, chunks, f64_round=9, **kw):
    oracle, hip = libs
    ref = run_agg(oracle, group_cols, aggs, types, chunks, device=-1)
    got = run_agg(hip, group_cols, aggs, types, chunks, device=0, **kw)
    r, g = [], []
    for c in ref:
        r.extend(c.rows())
    for c in got:
        g.extend(c.rows())
    assert multiset(g, f64_round=f64_round) == multiset(r, f64_round=f64_round)
    return r, g


def test_agg_all_funcs(libs):
    rng = np.random.default_rng(21)
    n = 50_000
    gk = rng.integers(0, 3000, size=n, dtype=np.int64)
    chunks = chunks_from_columns(
        [I64, I64, F64, I32],
        [(gk, (rng.random(n) < 0.02).astype(np.uint8)),
         (rng.integers(-1000, 1000, n, dtype=np.int64),
          (rng.random(n) < 0.1).astype(np.uint8)),
         (rng.standard_normal(n), (rng.random(n) < 0.1).astype(np.uint8)),
         (rng.integers(0, 100, n, dtype=np.int32), None)])
    aggs = [(abi.COUNT_ROW, -1), (abi.COUNT_COL, 1), (abi.SUM_I64, 1),
            (abi.SUM_F64, 2), (abi.MIN_I64, 1), (abi.MAX_I64, 1),
            (abi.MIN_F64, 2), (abi.MAX_F64, 2), (abi.SUM_I64, 3)]
    both_agg(libs, [0], aggs, [I64, I64, F64, I32], chunks)


def test_agg_multikey_with_rehash(libs):
    rng = np.random.default_rng(22)
    n = 120_000
    chunks = chunks_from_columns(
        [I32, I64, I64],
        [(rng.integers(0, 300, n, dtype=np.int32),
          (rng.random(n) < 0.05).astype(np.uint8)),
         (rng.integers(0, 200, n, dtype=np.int64), None),
         (rng.integers(0, 10, n, dtype=np.int64), None)],
        chunk_size=7000)
    # force tiny initial table -> several rehashes
    both_agg(libs, [0, 1], [(abi.COUNT_ROW, -1), (abi.SUM_I64, 2)],
             [I32, I64, I64], chunks, expected_groups=16)


def test_agg_no_group_by(libs):
    rng = np.random.default_rng(23)
    n = 30_000
    chunks = chunks_from_columns(
        [I64, F64],
        [(rng.integers(-50, 50, n, dtype=np.int64), None),
         (rng.standard_normal(n), (rng.random(n) < 0.5).astype(np.uint8))])
    both_agg(libs, [], [(abi.COUNT_ROW, -1), (abi.SUM_I64, 0),
                        (abi.MIN_I64, 0), (abi.MAX_I64, 0), (abi.SUM_F64, 1)],
             [I64, F64], chunks)


def test_agg_sum_i64_bitexact(libs):
    """COUNT/SUM(BIGINT) must be bit-exact incl. Java wrap-around overflow."""
    rng = np.random.default_rng(24)
    n = 10_000
    big = rng.integers(1 << 61, (1 << 62) - 1, size=n, dtype=np.int64)
    gk = rng.integers(0, 7, size=n, dtype=np.int64)
    chunks = chunks_from_columns([I64, I64], [(gk, None), (big, None)])
    ref, got = both_agg(libs, [0], [(abi.SUM_I64, 1)], [I64, I64], chunks)
    assert multiset(got) == multiset(ref)  # strict, no rounding


# ---- partition ----

def test_partition_routing_parity(libs):
    """Partition ROUTING must be Java-exact: each partition's row multiset
    must match the oracle's exactly (same rows on the same partition)."""
    oracle, hip = libs
    rng = np.random.default_rng(31)
    n = 40_000
    chunks = chunks_from_columns(
        [I64, I32, F64],
        [(rng.integers(-(1 << 40), 1 << 40, n, dtype=np.int64),
          (rng.random(n) < 0.03).astype(np.uint8)),
         (rng.integers(-1000, 1000, n, dtype=np.int32), None),
         (rng.standard_normal(n), None)],
        chunk_size=9000)
    for n_parts in (8, 5):
        exo = PartitioningExchanger(oracle, n_parts, [0], [I64, I32, F64], device=-1)
        exh = PartitioningExchanger(hip, n_parts, [0], [I64, I32, F64], device=0)
        for ch in chunks:
            po = exo.consume_chunk(ch)
            ph = exh.consume_chunk(ch)
            for p in range(n_parts):
                ro = po[p].rows() if po[p] else []
                rh = ph[p].rows() if ph[p] else []
                assert multiset(rh) == multiset(ro), f"partition {p}/{n_parts}"
        exo.close()
        exh.close()


def test_agg_small_chunk_stream(libs):
    """CHUNK_SIZE-style streaming (the CN's real 1000-row push cadence,
    ConnectionParams.java:1088): many small chunks take the inline-gid path,
    then a big chunk switches to the compaction path — gids must stay
    consistent across both."""
    rng = np.random.default_rng(25)
    n_small, n_big = 60_000, 400_000
    gk_small = rng.integers(0, 5000, size=n_small, dtype=np.int64)
    gk_big = rng.integers(0, 9000, size=n_big, dtype=np.int64)
    v_small = rng.integers(-100, 100, n_small, dtype=np.int64)
    v_big = rng.integers(-100, 100, n_big, dtype=np.int64)
    chunks = chunks_from_columns([I64, I64], [(gk_small, None), (v_small, None)],
                                 chunk_size=1000)
    chunks += chunks_from_columns([I64, I64], [(gk_big, None), (v_big, None)],
                                  chunk_size=n_big)
    both_agg(libs, [0], [(abi.SUM_I64, 1), (abi.COUNT_ROW, -1),
                         (abi.MIN_I64, 1)], [I64, I64], chunks,
             expected_groups=64)


def test_join_radix_staged_probe(libs):
    """Force the radix-staged probe path (bucket-range partitioning for
    L3-resident table slices) and check exact parity vs the oracle."""
    import os
    rng = np.random.default_rng(33)
    n_build, n_probe = 300_000, 900_000
    bkeys = rng.integers(0, 200_000, size=n_build, dtype=np.int64)
    build = chunks_from_columns(
        [I64, I64], [(bkeys, None),
                     (rng.integers(0, 1 << 30, n_build, dtype=np.int64), None)],
        chunk_size=50_000)
    pkeys = rng.integers(0, 220_000, size=n_probe, dtype=np.int64)
    nulls = (rng.random(n_probe) < 0.01).astype(np.uint8)
    probe = chunks_from_columns(
        [I64, I64], [(pkeys, nulls),
                     (rng.integers(0, 1 << 30, n_probe, dtype=np.int64), None)],
        chunk_size=300_000)
    keys = [EquiJoinKey(0, 0, I64)]
    os.environ["GX_RADIX_FORCE"] = "1"
    try:
        for jt in (abi.INNER, abi.LEFT, abi.SEMI, abi.ANTI):
            ref, got = both_join(libs, jt, keys, build, probe,
                                 [I64, I64], [I64, I64])
            assert multiset(got) == multiset(ref), f"join_type={jt}"
    finally:
        del os.environ["GX_RADIX_FORCE"]


def test_join_projection_pushdown(libs):
    """out_proj: the planner's Project-above-join collapse — output only a
    subset/permutation of the full join schema."""
    rng = np.random.default_rng(35)
    build = make_chunks(rng, [I64, I64, F64], 2000, key_space=900, null_frac=0.05)
    probe = make_chunks(rng, [I64, I32], 6000, key_space=1000, null_frac=0.05)
    keys = [EquiJoinKey(0, 0, I64)]
    # full schema: [probe0 I64, probe1 I32, build0 I64, build1 I64, build2 F64]
    proj = [3, 0, 4]
    for jt in (abi.INNER, abi.LEFT):
        ref, got = both_join(libs, jt, keys, build, probe,
                             [I64, I32], [I64, I64, F64], out_proj=proj)
        assert multiset(got, f64_round=9) == multiset(ref, f64_round=9), jt
    # buildOuter tail with projection: the BUILD side is the OUTER input,
    # so consume receives outer-typed chunks and probe inner-typed ones
    bo_build = make_chunks(rng, [I64, I32], 1500, key_space=700, null_frac=0.05)
    bo_probe = make_chunks(rng, [I64, I64, F64], 4000, key_space=800,
                           null_frac=0.05)
    ref, got = both_join(libs, abi.LEFT, keys, bo_build, bo_probe,
                         [I64, I32], [I64, I64, F64], out_proj=[0, 3],
                         build_outer=True)
    assert multiset(got) == multiset(ref)
