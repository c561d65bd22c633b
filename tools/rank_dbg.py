import sys, numpy as np
sys.path.insert(0, "/root/repo")
from galaxysql_amd import abi
from galaxysql_amd.chunk import I64, F64, rows_of, chunks_from_columns
from galaxysql_amd.operators import run_fwindow
rng = np.random.default_rng(52)
n = 30000
parts = np.sort(rng.integers(0, n // 23 + 1, n)).astype(np.int64)
vals = rng.integers(-50, 50, n).astype(np.int64)
nulls = (rng.random(n) < 0.1).astype(np.uint8)
fvals = np.round(rng.random(n) * 7, 3)
chunks = chunks_from_columns([I64, I64, F64],
                             [(parts, None), (vals, nulls), (fvals, None)],
                             chunk_size=997)
W = abi.FRAME_WHOLE_PARTITION
FR = [(abi.SUM_F64, 2, W),
      (abi.MIN_I64, 1, abi.FRAME_ROWS_SLIDING, 4, 1),
      (abi.MAX_F64, 2, abi.FRAME_ROWS_SLIDING, 2, 6),
      (abi.MIN_F64, 2, abi.FRAME_ROWS_SLIDING, 0, 3)]
for lib, dev, nm in [(abi.load_oracle(), -1, "ora"), (abi.load_hip(), 0, "hip")]:
    out = rows_of(run_fwindow(lib, [0], FR, [I64, I64, F64], chunks, device=dev))
    print(nm, [tuple(round(x, 3) if isinstance(x, float) else x for x in r[3:])
               for r in out[:4]])
print("inputs", [(int(parts[i]), None if nulls[i] else int(vals[i]),
                  float(fvals[i])) for i in range(6)])
