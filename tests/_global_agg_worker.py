"""Worker for test_two_phase_agg.py's GLOBAL (no-group-by) gloo case:
rank 0 aggregates real rows, rank 1 has NO input and must still emit the
SQL partial row (COUNT(*)=0, null-init SUM NULL); partials are gathered
to every rank and final-aggregated — the totals must equal rank 0's
local totals."""
import os
import sys

import numpy as np
import torch
import torch.distributed as dist

sys.path.insert(0, os.environ.get("PYTHONPATH",
                os.path.dirname(os.path.dirname(
                    os.path.abspath(__file__)))))

from galaxysql_amd import abi
from galaxysql_amd.chunk import Block, Chunk, I64, rows_of
from galaxysql_amd.exchange import final_agg_specs
from galaxysql_amd.operators import run_agg

AGGS = [(abi.COUNT_ROW, -1), (abi.SUM_I64N, 0)]


def main():
    outdir = sys.argv[1]
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    dist.init_process_group("gloo", rank=rank, world_size=world)
    lib = abi.load_oracle()

    if rank == 0:
        vals = np.arange(1, 101, dtype=np.int64)
        nulls = (vals % 7 == 0).astype(np.uint8)
        chunks = [Chunk([Block(I64, values=vals, nulls=nulls)])]
    else:
        chunks = []  # this rank's scan produced nothing

    partial = rows_of(run_agg(lib, [], AGGS, [I64], chunks, device=-1))
    assert len(partial) == 1, partial
    cnt, sm = partial[0]
    # exchange the partial row (encode NULL sum as a flag int)
    t = torch.tensor([cnt, 0 if sm is None else sm,
                      1 if sm is None else 0], dtype=torch.int64)
    gathered = [torch.zeros_like(t) for _ in range(world)]
    dist.all_gather(gathered, t)

    cnts = np.array([int(g[0]) for g in gathered], dtype=np.int64)
    sums = np.array([int(g[1]) for g in gathered], dtype=np.int64)
    snul = np.array([int(g[2]) for g in gathered], dtype=np.uint8)
    finals, _ = final_agg_specs(0, AGGS)
    fchunks = [Chunk([Block(I64, values=cnts),
                      Block(I64, values=sums,
                            nulls=snul if snul.any() else None)])]
    out = rows_of(run_agg(lib, [], finals, [I64, I64], fchunks, device=-1))
    assert len(out) == 1, out
    np.save(os.path.join(outdir, f"ga_{rank}.npy"),
            np.array([out[0][0], -1 if out[0][1] is None else out[0][1]],
                     dtype=np.int64))
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
