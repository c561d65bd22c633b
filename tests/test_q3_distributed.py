"""Distributed Q3 chain on CPU (gloo, world 2): the same shuffle +
orderkey-reshuffle code path bench.py runs over RCCL at N>1. The union of
the two ranks' group results must equal the single-process result exactly
(groups are orderkey-colocated after the shuffle, so local aggregation is
global)."""
import os
import subprocess
import sys

import numpy as np
import torch
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
WORKER = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                      "_q3_dist_worker.py")


def test_q3_distributed_matches_single(tmp_path):
    subprocess.run(["make", "-C", os.path.join(REPO, "oracle")], check=True,
                   capture_output=True)
    env = dict(os.environ)
    env.update(MASTER_ADDR="127.0.0.1", MASTER_PORT="29631", WORLD_SIZE="2",
               PYTHONPATH=REPO)
    procs = []
    for rank in range(2):
        e = dict(env, RANK=str(rank))
        procs.append(subprocess.Popen([sys.executable, WORKER, str(tmp_path)],
                                      env=e, stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    outs = [p.communicate(timeout=180)[0].decode() for p in procs]
    for rank, p in enumerate(procs):
        assert p.returncode == 0, f"rank {rank} failed:\n{outs[rank]}"

    d0 = np.load(tmp_path / "q3dist_0.npz")
    d1 = np.load(tmp_path / "q3dist_1.npz")

    # single-process reference
    from galaxysql_amd import abi
    from galaxysql_amd.queries import run_q3, gen_q3_numpy
    lib = abi.load_oracle()
    rng = np.random.default_rng(2024)
    data = gen_q3_numpy(rng, n_cust_total=4000, n_orders_total=40000,
                        n_lineitem=160000)
    t = [[torch.from_numpy(a) for a in cols] for cols in data]
    chunks, info = run_q3(lib, -1, t[0], t[1], t[2], to_host=True)
    cents = sum(r[4] for c in chunks for r in c.rows())

    assert int(d0["joined"]) + int(d1["joined"]) == info["joined_rows"]
    assert int(d0["groups"]) + int(d1["groups"]) == info["groups"]
    assert int(d0["cents"]) + int(d1["cents"]) == cents

    # honest distributed (the bench's N>1 path) vs single-process honest
    from galaxysql_amd.queries import run_q3_honest, gen_q3_raw_numpy
    h0 = np.load(tmp_path / "q3hdist_0.npz")
    h1 = np.load(tmp_path / "q3hdist_1.npz")
    rng2 = np.random.default_rng(777)
    raw = gen_q3_raw_numpy(rng2, n_cust=1500, n_orders=15000, n_lineitem=60000)
    tr = [[torch.from_numpy(a) for a in cols] for cols in raw]
    chunks, hinfo = run_q3_honest(lib, -1, tr[0], tr[1], tr[2], to_host=True)
    hcents = sum(r[4] for c in chunks for r in c.rows())
    assert int(h0["joined"]) + int(h1["joined"]) == hinfo["joined_rows"]
    assert int(h0["groups"]) + int(h1["groups"]) == hinfo["groups"]
    assert int(h0["cents"]) + int(h1["cents"]) == hcents
