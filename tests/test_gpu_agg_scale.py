"""Large-scale agg properties, closed-form (no oracle at this size).

Exercises the slot-indexed FUSED insert+accumulate path at big-chunk scale
(tests/test_gpu_parity covers small sizes): with expected_groups=0 the
table starts tiny, so millions of groups force the overflow -> x2 rehash ->
only-failed-rows rerun cycle repeatedly — every row must still accumulate
exactly once (records migrate with their slots; GX_SLOT_FAIL sentinel
gates the rerun)."""
import numpy as np
import pytest

from galaxysql_amd import abi
from galaxysql_amd.chunk import I64
from galaxysql_amd.operators import HashAggExec

pytestmark = pytest.mark.gpu


def _run_agg_cols(lib, keys, vals, expected_groups):
    from galaxysql_amd.chunk import Block, Chunk
    op = HashAggExec(lib, [0], [(abi.SUM_I64, 1), (abi.COUNT_ROW, -1)],
                     [I64, I64], expected_groups=expected_groups, device=0)
    try:
        op.consume_chunk(Chunk([Block(I64, values=keys),
                                Block(I64, values=vals)]))
        op.build_consume()
        out_k, out_s, out_c = [], [], []
        for c in op.result_chunks():
            out_k.append(np.asarray(c.blocks[0].values))
            out_s.append(np.asarray(c.blocks[1].values))
            out_c.append(np.asarray(c.blocks[2].values))
        return (np.concatenate(out_k), np.concatenate(out_s),
                np.concatenate(out_c))
    finally:
        op.close()


@pytest.mark.parametrize("expected_groups", [0, 5_000_000])
def test_agg_20m_rows_5m_groups_closed_form(expected_groups):
    """keys i%G, vals i: group k -> count N/G, sum = k + (k+G) + ... ;
    expected_groups=0 forces ~12 rehash doublings mid-insert."""
    lib = abi.load_hip()
    N, G = 20_000_000, 5_000_000
    i = np.arange(N, dtype=np.int64)
    keys = i % G
    vals = i
    k, s, c = _run_agg_cols(lib, keys, vals, expected_groups)
    assert len(k) == G
    order = np.argsort(k)
    k, s, c = k[order], s[order], c[order]
    assert np.array_equal(k, np.arange(G, dtype=np.int64))
    reps = N // G
    assert np.all(c == reps)
    # sum over arithmetic sequence k, k+G, ..., k+(reps-1)G
    expect = k * reps + G * (reps * (reps - 1) // 2)
    assert np.array_equal(s, expect)


def test_agg_streamed_chunks_match_single_consume():
    """State must persist across consume calls (slot records survive
    rehash between chunks too)."""
    lib = abi.load_hip()
    rng = np.random.default_rng(77)
    N = 3_000_000
    keys = rng.integers(0, 400_000, N)
    vals = rng.integers(-1000, 1000, N)
    k1, s1, c1 = _run_agg_cols(lib, keys, vals, expected_groups=0)
    # same data in 7 chunks
    from galaxysql_amd.chunk import Block, Chunk
    op = HashAggExec(lib, [0], [(abi.SUM_I64, 1), (abi.COUNT_ROW, -1)],
                     [I64, I64], expected_groups=0, device=0)
    try:
        for part in np.array_split(np.arange(N), 7):
            op.consume_chunk(Chunk([Block(I64, values=keys[part]),
                                    Block(I64, values=vals[part])]))
        op.build_consume()
        ks, ss, cs = [], [], []
        for ch in op.result_chunks():
            ks.append(np.asarray(ch.blocks[0].values))
            ss.append(np.asarray(ch.blocks[1].values))
            cs.append(np.asarray(ch.blocks[2].values))
        k2 = np.concatenate(ks)
        s2 = np.concatenate(ss)
        c2 = np.concatenate(cs)
    finally:
        op.close()
    o1, o2 = np.argsort(k1), np.argsort(k2)
    assert np.array_equal(k1[o1], k2[o2])
    assert np.array_equal(s1[o1], s2[o2])
    assert np.array_equal(c1[o1], c2[o2])
