"""Worker for test_q18_distributed: the EXACT C4 distributed step shape
(bench.py C4 at N>1): shuffle lineitem/orders/cust by orderkey, then
run_q18 locally (group key == shuffle key, so the local aggregate is
final). gloo/CPU with the oracle backend."""
import os
import sys

import numpy as np
import torch
import torch.distributed as dist

sys.path.insert(0, os.environ["PYTHONPATH"])

from galaxysql_amd import abi
from galaxysql_amd.exchange import shuffle_columns
from galaxysql_amd.queries import (run_q18, gen_q18_numpy,
                                   Q18_LINEITEM_TYPES, Q18_ORDERS_TYPES,
                                   Q18_CUST_TYPES)


def main():
    outdir = sys.argv[1]
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    dist.init_process_group("gloo", rank=rank, world_size=world)
    lib = abi.load_oracle()
    rng = np.random.default_rng(606)
    cust, orders, lineitem = gen_q18_numpy(rng, n_cust=3000, n_orders=30000,
                                           having_frac=0.002)

    def myslice(cols):
        return [torch.from_numpy(np.ascontiguousarray(c[rank::world]))
                for c in cols]

    li = shuffle_columns(lib, myslice(lineitem), Q18_LINEITEM_TYPES, [0],
                         device=-1)
    od = shuffle_columns(lib, myslice(orders), Q18_ORDERS_TYPES, [0],
                         device=-1)
    cu = shuffle_columns(lib, myslice(cust), Q18_CUST_TYPES, [0], device=-1)
    n_final, info = run_q18(lib, -1, cu, od, li,
                            reshuffle_by_custkey=True)
    np.savez(os.path.join(outdir, f"q18dist_{rank}.npz"),
             final=n_final, groups=info["groups"],
             survivors=info["survivors"])
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
