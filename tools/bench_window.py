"""Window-operator throughput evidence (not a BASELINE config — the
north_star's HashWindowExec has no reference benchmark): running
partition aggregates (segmented ScanByKey) and whole-partition frames at
100M rows. Run on an MI355X box: python tools/bench_window.py"""
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from galaxysql_amd import abi
from galaxysql_amd.chunk import I64
from galaxysql_amd.operators import NonFrameOverWindowExec, OverWindowFramesExec
import ctypes as C


def dev_chunk(cols):
    return [{"type": I64, "values": c.data_ptr(), "n_rows": c.numel()}
            for c in cols]


def main():
    n = 100_000_000
    lib = abi.load_hip()
    dev = torch.device("cuda:0")
    g = torch.Generator(device=dev).manual_seed(3)
    parts = torch.sort(torch.randint(0, n // 50, (n,), device=dev,
                                     generator=g, dtype=torch.int64)).values
    vals = torch.randint(0, 100, (n,), device=dev, generator=g,
                         dtype=torch.int64)

    for trial in range(3):
        op = NonFrameOverWindowExec(lib, [0],
                                    [(abi.COUNT_ROW, -1), (abi.SUM_I64, 1)],
                                    [I64, I64], device=0)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        ka = []
        gc = lib.to_gx_chunk(None, ka, device_ptrs=dev_chunk([parts, vals]))
        out = C.POINTER(abi.GxResult)()
        lib.check(lib.lib.gxop_window_consume(op._op, C.byref(gc),
                                              C.byref(out)), "win")
        torch.cuda.synchronize()
        t1 = time.perf_counter()
        if out:
            lib.lib.gxop_result_release(out)
        op.close()
        print(f"window(run cnt+sum) trial{trial}: {1e3*(t1-t0):7.2f} ms "
              f"= {n/(t1-t0)/1e9:6.2f} G rows/s")

    for trial in range(3):
        op = OverWindowFramesExec(
            lib, [0],
            [(abi.SUM_I64, 1, abi.FRAME_WHOLE_PARTITION),
             (abi.SUM_I64, 1, abi.FRAME_ROWS_SLIDING, 5, 5)],
            [I64, I64], device=0)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        ka = []
        gc = lib.to_gx_chunk(None, ka, device_ptrs=dev_chunk([parts, vals]))
        lib.check(lib.lib.gxop_fwindow_consume(op._op, C.byref(gc)), "c")
        lib.check(lib.lib.gxop_fwindow_finish(op._op), "f")
        out = C.POINTER(abi.GxResult)()
        lib.check(lib.lib.gxop_fwindow_next(op._op, C.byref(out)), "n")
        torch.cuda.synchronize()
        t1 = time.perf_counter()
        if out:
            lib.lib.gxop_result_release(out)
        op.close()
        print(f"fwindow(whole+slide5) trial{trial}: {1e3*(t1-t0):7.2f} ms "
              f"= {n/(t1-t0)/1e9:6.2f} G rows/s")


if __name__ == "__main__":
    main()
