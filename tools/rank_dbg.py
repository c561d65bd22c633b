import sys, numpy as np
sys.path.insert(0, "/root/repo")
from galaxysql_amd import abi
from galaxysql_amd.chunk import I64, rows_of, chunks_from_columns
from galaxysql_amd.operators import run_window
rng = np.random.default_rng(44)
n = 30000
parts = np.sort(rng.integers(0, n // 40, n)).astype(np.int64)
order = np.concatenate([np.sort(rng.integers(0, 9, (parts == p).sum()))
                        for p in np.unique(parts)]).astype(np.int64)
onulls = (rng.random(n) < 0.05).astype(np.uint8)
chunks = chunks_from_columns([I64, I64], [(parts, None), (order, onulls)],
                             chunk_size=13)
AGGS = [(abi.RANK, -1), (abi.DENSE_RANK, -1), (abi.COUNT_ROW, -1)]
got = rows_of(run_window(abi.load_hip(), [0], AGGS, [I64, I64], chunks,
                         order_cols=[1], device=0))
want = rows_of(run_window(abi.load_oracle(), [0], AGGS, [I64, I64], chunks,
                          order_cols=[1], device=-1))
bad = [i for i, (g, w) in enumerate(zip(got, want)) if g != w]
print("mismatches:", len(bad), "first:", bad[:5])
for i in bad[:5]:
    print(i, "got", got[i], "want", want[i], "ctx",
          [(int(parts[j]), None if onulls[j] else int(order[j]))
           for j in range(max(0, i - 3), min(n, i + 2))])
