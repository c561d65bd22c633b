"""Worker for test_two_phase_agg.py's gloo case: each rank partial-aggs its
local shard, shuffles the partial rows by group-key hash over
all_to_all_single, final-aggs its received rows, and writes its groups."""
import os
import sys

import numpy as np
import torch
import torch.distributed as dist

sys.path.insert(0, os.environ.get("PYTHONPATH",
                os.path.dirname(os.path.dirname(
                    os.path.abspath(__file__)))))

from galaxysql_amd import abi
from galaxysql_amd.chunk import I64, I32, F64, chunks_from_columns
from galaxysql_amd.exchange import final_agg_specs, shuffle_columns
from galaxysql_amd.operators import run_agg

AGGS = [(abi.COUNT_ROW, -1), (abi.SUM_I64, 2), (abi.MIN_I64, 2),
        (abi.SUM_F64, 3)]
GROUP_COLS = [0, 1]
TYPES = [I64, I32, I64, F64]


def gen(rng, n):
    g1 = rng.integers(0, 37, n)
    g2 = rng.integers(0, 4, n).astype(np.int32)
    v = rng.integers(-100, 100, n)
    nulls = (rng.random(n) < 0.2).astype(np.uint8)
    f = np.round(rng.standard_normal(n), 3)
    return [(g1, None), (g2, None), (v, nulls), (f, None)]


def main():
    outdir = sys.argv[1]
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    dist.init_process_group("gloo", rank=rank, world_size=world)
    lib = abi.load_oracle()

    rng = np.random.default_rng(71)          # same data on both ranks
    cols = gen(rng, 8000)
    chunks = chunks_from_columns(TYPES, cols)
    my = [c for i, c in enumerate(chunks) if i % world == rank]

    partial = run_agg(lib, GROUP_COLS, AGGS, TYPES, my)

    # partial output schema: group cols + one col per agg
    ptypes = [I64, I32, I64, I64, I64, F64]
    # flatten partial chunks into full columns, then shuffle by group hash.
    # Nullable partial cols (MIN): encode null as sentinel OUTSIDE the value
    # domain before the tensor exchange, decode after.
    SENT = np.int64(-(10**18))
    cols_t = []
    for ci, t in enumerate(ptypes):
        parts = []
        for c in partial:
            b = c.blocks[ci]
            v = np.asarray(b.values).copy()
            if b.nulls is not None:
                v[np.asarray(b.nulls) == 1] = SENT if t != F64 else np.nan
            parts.append(v)
        cols_t.append(torch.from_numpy(np.concatenate(parts)))
    recv = shuffle_columns(lib, cols_t, ptypes, GROUP_COLS, device=-1)

    rv = [t.numpy() for t in recv]
    nulls_min = (rv[4] == SENT).astype(np.uint8)
    rv[4] = np.where(nulls_min == 1, 0, rv[4])
    finals, _ = final_agg_specs(len(GROUP_COLS), AGGS)
    fchunks = chunks_from_columns(
        ptypes, [(rv[0], None), (rv[1], None), (rv[2], None), (rv[3], None),
                 (rv[4], nulls_min), (rv[5], None)])
    out = run_agg(lib, GROUP_COLS, finals, ptypes, fchunks)
    rows = []
    for c in out:
        rows.extend(c.rows())
    np.save(os.path.join(outdir, f"tp_{rank}.npy"),
            np.array([(r[0], r[1], r[2], r[3],
                       SENT if r[4] is None else r[4],
                       np.nan if r[5] is None else r[5])
                      for r in rows], dtype=np.float64))
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
