"""Worker for test_q3_distributed.py: run the DISTRIBUTED Q3 chain (shuffles
+ orderkey re-shuffle) on gloo/CPU with the oracle backend and write this
rank's group count + checksums."""
import os
import sys

import numpy as np
import torch
import torch.distributed as dist

sys.path.insert(0, os.environ["PYTHONPATH"])

from galaxysql_amd import abi
from galaxysql_amd.exchange import shuffle_columns
from galaxysql_amd.queries import (run_q3, run_q3_honest, gen_q3_numpy,
                                   gen_q3_raw_numpy, CUST_TYPES,
                                   ORDERS_TYPES, LINEITEM_TYPES)


def main():
    outdir = sys.argv[1]
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    dist.init_process_group("gloo", rank=rank, world_size=world)
    lib = abi.load_oracle()

    # both ranks generate the SAME global data, then each keeps a slice
    # (a stand-in for rank-local scan output)
    rng = np.random.default_rng(2024)
    data = gen_q3_numpy(rng, n_cust_total=4000, n_orders_total=40000,
                        n_lineitem=160000)
    t = [[torch.from_numpy(a) for a in cols] for cols in data]

    def myslice(cols):
        return [c[rank::world].contiguous() for c in cols]

    cust = shuffle_columns(lib, myslice(t[0]), CUST_TYPES, [0], device=-1)
    orders = shuffle_columns(lib, myslice(t[1]), ORDERS_TYPES, [0], device=-1)
    lineitem = shuffle_columns(lib, myslice(t[2]), LINEITEM_TYPES, [0],
                               device=-1)
    chunks, info = run_q3(lib, -1, cust, orders, lineitem, to_host=True,
                          reshuffle_by_orderkey=True)
    # checksum of the local groups: sum of cents-sums and count
    cents = 0
    rows = 0
    for c in chunks:
        for r in c.rows():
            cents += r[4]
            rows += 1
    np.savez(os.path.join(outdir, f"q3dist_{rank}.npz"),
             groups=info["groups"], rows=rows, cents=cents,
             joined=info["joined_rows"])

    # HONEST distributed path — exactly what bench.py C3 runs at N>1:
    # local scans -> as_tensors -> shuffles -> run_q3 with orderkey reshuffle
    rng2 = np.random.default_rng(777)
    raw = gen_q3_raw_numpy(rng2, n_cust=1500, n_orders=15000, n_lineitem=60000)
    raww = [[torch.from_numpy(a) for a in cols] for cols in raw]
    mycust = [c[rank::world].contiguous() for c in raww[0]]
    myord = [c[rank::world].contiguous() for c in raww[1]]
    myli = [c[rank::world].contiguous() for c in raww[2]]
    (cust2, orders2, lineitem2), scanned = run_q3_honest(
        lib, -1, mycust, myord, myli, as_tensors=True)
    cust2 = shuffle_columns(lib, cust2, CUST_TYPES, [0], device=-1)
    orders2 = shuffle_columns(lib, orders2, ORDERS_TYPES, [0], device=-1)
    lineitem2 = shuffle_columns(lib, lineitem2, LINEITEM_TYPES, [0], device=-1)
    chunks2, info2 = run_q3(lib, -1, cust2, orders2, lineitem2, to_host=True,
                            reshuffle_by_orderkey=True)
    cents2 = 0
    rows2 = 0
    for c in chunks2:
        for r in c.rows():
            cents2 += r[4]
            rows2 += 1
    np.savez(os.path.join(outdir, f"q3hdist_{rank}.npz"),
             groups=info2["groups"], rows=rows2, cents=cents2,
             joined=info2["joined_rows"])
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
