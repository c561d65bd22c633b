"""C1 (BASELINE.md configs[0]): the CPU-only plumbing config — join build
1M unique int64 keys, probe 10M uniform [0,1M), COUNT(*) bit-exact. Runs
the oracle through the same operator lifecycle the GPU path uses; no GPU
required (the reference's own 'no GPU, plumbing' case)."""
import os
import subprocess

import numpy as np
import pytest

from galaxysql_amd import abi
from galaxysql_amd.chunk import I64, chunks_from_columns
from galaxysql_amd.operators import EquiJoinKey, run_join, run_agg

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_c1_join_count():
    subprocess.run(["make", "-C", os.path.join(REPO, "oracle")], check=True,
                   capture_output=True)
    lib = abi.load_oracle()
    rng = np.random.default_rng(42)
    n_build, n_probe = 1_000_000, 10_000_000
    bkeys = rng.permutation(n_build).astype(np.int64)
    build = chunks_from_columns([I64], [(bkeys, None)], chunk_size=1000)
    pkeys = rng.integers(0, n_build, n_probe).astype(np.int64)
    probe = chunks_from_columns([I64], [(pkeys, None)], chunk_size=100_000)
    out = run_join(lib, abi.INNER, [EquiJoinKey(0, 0, I64)], build, probe,
                   [I64], [I64], device=-1)
    count = sum(c.n_rows for c in out)
    # unique build keys cover the whole probe key space -> every probe
    # row matches exactly once
    assert count == n_probe

    # COUNT(*) through the aggregate operator as well
    agg = run_agg(lib, [], [(abi.COUNT_ROW, -1)], [I64, I64],
                  [c for c in out], device=-1)
    rows = [t for c in agg for t in c.rows()]
    assert rows == [(n_probe,)]


def test_buffered_probe_equals_per_chunk():
    """probe_push/probe_flush (the LocalBufferExec-style buffered cadence)
    must produce the same multiset as per-chunk gxop_join_probe."""
    import numpy as np
    from galaxysql_amd import abi
    from galaxysql_amd.chunk import Block, Chunk, I64, multiset
    from galaxysql_amd.operators import ParallelHashJoinExec, EquiJoinKey

    lib = abi.load_oracle()
    rng = np.random.default_rng(17)
    bk = np.arange(2000, dtype=np.int64)
    bv = bk * 3
    pk = rng.integers(0, 4000, 10_000).astype(np.int64)
    pv = np.arange(10_000, dtype=np.int64)

    def run(buffered):
        op = ParallelHashJoinExec(lib, abi.INNER, [EquiJoinKey(0, 0, I64)],
                                  [I64, I64], [I64, I64], device=-1)
        try:
            op.consume_chunk(Chunk([Block(I64, values=bk),
                                    Block(I64, values=bv)]))
            op.build_consume()
            rows = []
            for lo in range(0, 10_000, 1000):
                ch = Chunk([Block(I64, values=pk[lo:lo + 1000]),
                            Block(I64, values=pv[lo:lo + 1000])])
                if buffered:
                    op.probe_push(ch)
                    if lo == 4000:  # mid-stream flush
                        r = op.probe_flush()
                        if r:
                            rows.extend(r.rows())
                else:
                    r = op.probe_chunk(ch)
                    if r:
                        rows.extend(r.rows())
            if buffered:
                r = op.probe_flush()
                if r:
                    rows.extend(r.rows())
            return rows
        finally:
            op.close()

    assert multiset(run(True)) == multiset(run(False))
