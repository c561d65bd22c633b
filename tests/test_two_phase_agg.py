"""Partial -> shuffle -> final aggregation equals single-phase aggregation
(the MPP two-phase plan the reference planner emits around an exchange;
re-agg mapping in galaxysql_amd.exchange.final_agg_specs: COUNT->SUM,
SUM->SUM, MIN/MAX->self). Simulated with 3 'ranks' on the oracle — the
RCCL exchange mechanics themselves are covered by test_exchange_cpu."""
import os
import numpy as np
import pytest

from galaxysql_amd import abi
from galaxysql_amd.chunk import I64, I32, F64, chunks_from_columns, \
    multiset, rows_of
from galaxysql_amd.exchange import final_agg_specs
from galaxysql_amd.operators import run_agg, PartitioningExchanger


AGGS = [(abi.COUNT_ROW, -1), (abi.COUNT_COL, 2), (abi.SUM_I64, 2),
        (abi.SUM_F64, 3), (abi.MIN_I64, 2), (abi.MAX_F64, 3)]
GROUP_COLS = [0, 1]
TYPES = [I64, I32, I64, F64]


def _data(rng, n):
    g1 = rng.integers(0, 40, n)
    g2 = rng.integers(0, 5, n).astype(np.int32)
    v = rng.integers(-100, 100, n)
    nulls = (rng.random(n) < 0.2).astype(np.uint8)
    f = np.round(rng.standard_normal(n), 3)
    return chunks_from_columns(TYPES, [(g1, None), (g2, None), (v, nulls),
                                       (f, None)])


def test_two_phase_equals_single_phase():
    lib = abi.load_oracle()
    rng = np.random.default_rng(61)
    chunks = _data(rng, 9000)

    # single phase
    want = run_agg(lib, GROUP_COLS, AGGS, TYPES, chunks)

    # phase 1: 3 ranks each aggregate a share of the chunks
    ranks = 3
    partials = [[] for _ in range(ranks)]
    for i, c in enumerate(chunks):
        partials[i % ranks].append(c)
    partial_out = [run_agg(lib, GROUP_COLS, AGGS, TYPES, share)
                   for share in partials]

    # shuffle partial rows by group-key hash (Java-exact routing)
    finals, _ = final_agg_specs(len(GROUP_COLS), AGGS)
    ptypes = [I64, I32] + [I64, I64, I64, F64, I64, F64]
    buckets = [[] for _ in range(ranks)]
    ex = PartitioningExchanger(lib, ranks, key_cols=GROUP_COLS,
                               input_types=ptypes)
    for rank_out in partial_out:
        for c in rank_out:
            for p, out in enumerate(ex.consume_chunk(c)):
                if out is not None and out.n_rows:
                    buckets[p].append(out)

    # phase 2: final agg per destination rank; union of results
    got = []
    for p in range(ranks):
        if buckets[p]:
            got.extend(run_agg(lib, GROUP_COLS, finals, ptypes, buckets[p]))

    assert multiset(rows_of(got), f64_round=6) == \
        multiset(rows_of(want), f64_round=6)


def test_two_phase_distributed_gloo(tmp_path):
    """The same partial->shuffle->final plan over the REAL exchange
    (all_to_all_single, gloo world 2): union of rank outputs == the
    single-process aggregation."""
    import os
    import subprocess
    import sys
    REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    WORKER = os.path.join(REPO, "tests", "_two_phase_worker.py")
    env = dict(os.environ)
    env.update(MASTER_ADDR="127.0.0.1", MASTER_PORT="29741", WORLD_SIZE="2",
               PYTHONPATH=REPO)
    procs = []
    for rank in range(2):
        e = dict(env, RANK=str(rank))
        procs.append(subprocess.Popen(
            [sys.executable, WORKER, str(tmp_path)], env=e,
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
    outs = [p.communicate(timeout=180)[0].decode() for p in procs]
    for rank, p in enumerate(procs):
        assert p.returncode == 0, f"rank {rank}:\n{outs[rank]}"

    d = np.concatenate([np.load(tmp_path / f"tp_{r}.npy") for r in range(2)])

    # single-process reference using the worker's generator
    sys.path.insert(0, os.path.join(REPO, "tests"))
    import _two_phase_worker as w
    lib = abi.load_oracle()
    rng = np.random.default_rng(71)
    chunks = chunks_from_columns(w.TYPES, w.gen(rng, 8000))
    want = run_agg(lib, w.GROUP_COLS, w.AGGS, w.TYPES, chunks)
    wrows = []
    SENT = -(10**18)
    for c in want:
        for r in c.rows():
            wrows.append((r[0], r[1], r[2], r[3],
                          SENT if r[4] is None else r[4],
                          np.nan if r[5] is None else r[5]))
    got = sorted(map(tuple, d.tolist()))
    wex = sorted([tuple(float(x) for x in r) for r in wrows])
    assert len(got) == len(wex)
    for g, e in zip(got, wex):
        for a, b in zip(g, e):
            if np.isnan(a) or np.isnan(b):
                assert np.isnan(a) and np.isnan(b)
            else:
                assert abs(a - b) < 1e-6, (g, e)


@pytest.mark.timeout(300)
def test_global_agg_distributed_empty_rank(tmp_path):
    """GLOBAL aggregate at world 2 where rank 1 consumed nothing: its
    SQL partial row (0, NULL) must flow through the exchange and leave
    the final totals equal to rank 0's data alone."""
    import subprocess
    import sys as _sys
    world = 2
    REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    env.update(MASTER_ADDR="127.0.0.1", MASTER_PORT="29733",
               WORLD_SIZE=str(world), PYTHONPATH=REPO, OMP_NUM_THREADS="1")
    worker = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                          "_global_agg_worker.py")
    procs = []
    for rank in range(world):
        e = dict(env, RANK=str(rank))
        procs.append(subprocess.Popen([_sys.executable, worker,
                                       str(tmp_path)], env=e,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    outs = [p.communicate(timeout=240)[0].decode() for p in procs]
    for rank, p in enumerate(procs):
        assert p.returncode == 0, f"rank {rank} failed:\n{outs[rank]}"
    import numpy as np
    vals = np.arange(1, 101)
    expect_cnt = 100
    expect_sum = int(vals[vals % 7 != 0].sum())
    for rank in range(world):
        got = np.load(tmp_path / f"ga_{rank}.npy")
        assert int(got[0]) == expect_cnt
        assert int(got[1]) == expect_sum
