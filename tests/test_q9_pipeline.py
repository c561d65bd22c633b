"""Q9 chain parity (scaled down): LIKE filter -> SEMI ->
2-key join -> exact scale-4 DECIMAL amount -> joins -> group-by. Oracle vs
independent numpy on CPU; HIP vs oracle (bit-exact sums) on GPU."""
import os
import subprocess

import numpy as np
import pytest
import torch

from galaxysql_amd import abi
from galaxysql_amd.chunk import Block, Chunk, I64, SLICE, multiset
from galaxysql_amd.queries import run_q9, gen_q9_numpy, stage_table, \
    Q9_PART_TYPES

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.fixture(scope="module")
def oracle():
    subprocess.run(["make", "-C", os.path.join(REPO, "oracle")], check=True,
                   capture_output=True)
    return abi.load_oracle()


def numpy_q9(part, supplier, partsupp, orders, lineitem, pattern="green"):
    pk, pn = part
    keep_part = set(int(k) for k, s in zip(pk, pn) if pattern in s)
    sk, sn = supplier
    nation = {int(k): int(v) for k, v in zip(sk, sn)}
    psk, pss, psc = partsupp
    cost = {(int(a), int(b)): int(c) for a, b, c in zip(psk, pss, psc)
            if int(a) in keep_part}
    ok, oy = orders
    year = {int(k): int(v) for k, v in zip(ok, oy)}
    groups = {}
    lp, ls, lo, lq, le, ld = lineitem
    for p, s, o, q, e, d in zip(lp, ls, lo, lq, le, ld):
        c = cost.get((int(p), int(s)))
        if c is None:
            continue
        amount = int(e) * (100 - int(d)) - c * int(q) * 100
        key = (nation[int(s)], year[int(o)])
        g = groups.setdefault(key, [0, 0])
        g[0] += amount
        g[1] += 1
    return [(k[0], k[1], v[0], v[1]) for k, v in groups.items()]


def run_chain(lib, device, data):
    part_np, supplier, partsupp, orders, lineitem = data
    part_chunk = Chunk([Block(I64, values=part_np[0]),
                        Block.of(SLICE, part_np[1])])
    part_res = stage_table(lib, part_chunk, Q9_PART_TYPES, device)
    try:
        t = []
        for cols in (supplier, partsupp, orders, lineitem):
            tc = [torch.from_numpy(a) for a in cols]
            if device >= 0:
                tc = [x.cuda(device) for x in tc]
            t.append(tc)
        return run_q9(lib, device, part_res, t[0], t[1], t[2], t[3])
    finally:
        lib.lib.gxop_result_release(part_res)


def test_q9_oracle_vs_numpy(oracle):
    rng = np.random.default_rng(61)
    data = gen_q9_numpy(rng, n_part=2000, n_supp=500, n_orders=8000,
                        n_lineitem=40000)
    rows, info = run_chain(oracle, -1, data)
    exp = numpy_q9(*data)
    assert multiset(rows) == multiset(exp)


@pytest.mark.gpu
def test_q9_hip_vs_oracle():
    oracle = abi.load_oracle()
    hip = abi.load_hip()
    rng = np.random.default_rng(62)
    data = gen_q9_numpy(rng, n_part=50_000, n_supp=5_000, n_orders=150_000,
                        n_lineitem=1_000_000)
    ref, ri = run_chain(oracle, -1, data)
    got, gi = run_chain(hip, 0, data)
    for k in ("part_kept", "partsupp_kept", "lineitem_joined",
              "after_orders", "final_rows", "groups"):
        assert gi[k] == ri[k], k
    assert multiset(got) == multiset(ref)  # bit-exact (i64 sums)
