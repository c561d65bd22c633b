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
