import sys, numpy as np
sys.path.insert(0, "/root/repo")
from galaxysql_amd import abi
from galaxysql_amd.chunk import Block, Chunk, I64, rows_of, chunks_from_columns
from galaxysql_amd.operators import run_window
part = np.array([1]*6 + [2]*2, np.int64)
order = np.array([5,5,7,7,7,9,1,1], np.int64)
chunks = chunks_from_columns([I64, I64], [(part, None), (order, None)], chunk_size=3)
for lib, dev, name in [(abi.load_oracle(), -1, "ora"), (abi.load_hip(), 0, "hip")]:
    rows = rows_of(run_window(lib, [0], [(abi.RANK,-1),(abi.DENSE_RANK,-1)],
                              [I64, I64], chunks, order_cols=[1], device=dev))
    print(name, [(r[2], r[3]) for r in rows])
