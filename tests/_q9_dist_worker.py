"""Worker for test_q9_world8.py: the DISTRIBUTED Q9 two-phase chain
(per-stage hash shuffles + partial/final aggregate exchange) on gloo/CPU
with the oracle backend — the same run_q9 code path bench.py C5 runs over
RCCL at N>1, here at world size 8."""
import os
import sys

import numpy as np
import torch
import torch.distributed as dist

sys.path.insert(0, os.environ["PYTHONPATH"])

from galaxysql_amd import abi
from galaxysql_amd.chunk import Block, Chunk, I64, SLICE
from galaxysql_amd.queries import (run_q9, gen_q9_numpy, stage_table,
                                   Q9_PART_TYPES)


def main():
    outdir = sys.argv[1]
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    dist.init_process_group("gloo", rank=rank, world_size=world)
    lib = abi.load_oracle()

    # all ranks generate the SAME global data, then keep a slice each
    rng = np.random.default_rng(909)
    part, supplier, partsupp, orders, lineitem = gen_q9_numpy(
        rng, n_part=1200, n_supp=90, n_orders=4000, n_lineitem=24000)

    def myslice(cols):
        return [torch.from_numpy(np.ascontiguousarray(c[rank::world]))
                for c in cols]

    # part is string-keyed: slice rows then stage
    pk = part[0][rank::world]
    pn = [part[1][i] for i in range(rank, len(part[1]), world)]
    part_chunk = Chunk([Block(I64, values=np.ascontiguousarray(pk)),
                        Block.of(SLICE, pn)])
    part_res = stage_table(lib, part_chunk, Q9_PART_TYPES, -1)

    rows, info = run_q9(lib, -1, part_res, myslice(supplier),
                        myslice(partsupp), myslice(orders),
                        myslice(lineitem), world=world, local_rank=rank)
    lib.lib.gxop_result_release(part_res)
    total = 0
    cnt = 0
    n_rows = 0
    for r in rows:
        total += r[2]
        cnt += r[3]
        n_rows += 1
    np.savez(os.path.join(outdir, f"q9dist_{rank}.npz"),
             rows=n_rows, sum4=total, cnt=cnt, groups=info["groups"])
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
