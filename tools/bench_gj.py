"""A/B: Q18's agg stage as plain hash-agg vs the fused group-join.

Plain C4 path: HashAgg(lineitem GROUP BY l_orderkey, SUM qty) -> 150M-group
table (insert + gid + accumulate). Fused path: the orders table itself is
the group list — groupjoin(build=orders, probe=lineitem) accumulates the
probe directly into per-order records; no agg hash table at all.

Run on an MI355X box:  python tools/bench_gj.py [scale]
"""
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from galaxysql_amd import abi
from galaxysql_amd.chunk import I64
from galaxysql_amd.operators import (EquiJoinKey, HashGroupJoinExec,
                                     HashAggExec)
from galaxysql_amd.queries import result_to_tensors


def main():
    scale = float(sys.argv[1]) if len(sys.argv) > 1 else 1.0
    n_orders = int(150_000_000 * scale)
    n_items = int(600_000_000 * scale)
    lib = abi.load_hip()
    dev = torch.device("cuda:0")
    g = torch.Generator(device=dev).manual_seed(7)
    okeys = torch.randperm(n_orders, device=dev, generator=g,
                           dtype=torch.int64)
    ikeys = torch.randint(0, n_orders, (n_items,), device=dev, generator=g,
                          dtype=torch.int64)
    qty = torch.randint(1, 51, (n_items,), device=dev, generator=g,
                        dtype=torch.int64)

    def dev_chunk(cols):
        return [{"type": I64, "values": c.data_ptr(), "n_rows": c.numel()}
                for c in cols]

    # ---- fused group-join -------------------------------------------------
    for trial in range(3):
        gj = HashGroupJoinExec(lib, abi.INNER, [EquiJoinKey(0, 0, I64)],
                               [I64], [I64, I64], group_cols=[0],
                               aggs=[(abi.SUM_I64, 1)], device=0,
                               expected_build_rows=n_orders)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        import ctypes as C
        ka = []
        gc = lib.to_gx_chunk(None, ka, device_ptrs=dev_chunk([okeys]))
        lib.check(lib.lib.gxop_groupjoin_consume(gj._op, C.byref(gc)),
                  "consume")
        lib.check(lib.lib.gxop_groupjoin_build(gj._op), "build")
        t1 = time.perf_counter()
        gc2 = lib.to_gx_chunk(None, ka, device_ptrs=dev_chunk([ikeys, qty]))
        lib.check(lib.lib.gxop_groupjoin_probe(gj._op, C.byref(gc2)), "probe")
        t2 = time.perf_counter()
        out = C.POINTER(abi.GxResult)()
        lib.check(lib.lib.gxop_groupjoin_next(gj._op, C.byref(out)), "next")
        n_groups = out.contents.chunk.n_rows if out else 0
        survivors = -1
        if out:
            cols = result_to_tensors(lib, out, [I64, I64], dev)
            survivors = int((cols[1] > 300).sum().item())
            lib.lib.gxop_result_release(out)
        torch.cuda.synchronize()
        t3 = time.perf_counter()
        gj.close()
        print(f"fused  trial{trial}: build {1e3*(t1-t0):8.2f} ms  "
              f"probe+accum {1e3*(t2-t1):8.2f} ms  emit {1e3*(t3-t2):8.2f} ms"
              f"  total {1e3*(t3-t0):8.2f} ms  groups={n_groups}"
              f" survivors={survivors}")

    # ---- plain hash agg ---------------------------------------------------
    for trial in range(3):
        agg = HashAggExec(lib, [0], [(abi.SUM_I64, 1)], [I64, I64],
                          expected_groups=n_orders, device=0)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        import ctypes as C
        ka = []
        gc = lib.to_gx_chunk(None, ka,
                             device_ptrs=dev_chunk([ikeys, qty]))
        lib.check(lib.lib.gxop_agg_consume(agg._op, C.byref(gc)),
                  "agg_consume")
        agg.build_consume()
        torch.cuda.synchronize()
        t1 = time.perf_counter()
        print(f"hashagg trial{trial}: consume+build {1e3*(t1-t0):8.2f} ms  "
              f"stats={agg.stats()}")
        agg.close()


if __name__ == "__main__":
    main()
