"""Distributed Q18 (the exact bench.py C4 N>1 step shape) on CPU/gloo at
world 2: shuffle lineitem/orders/customer by orderkey, run_q18 locally —
the group key IS the shuffle key so the local aggregate is final
(SURVEY.md §8e). Union of rank results must equal the single-process run
bit-exactly (integer sums)."""
import os
import subprocess
import sys

import numpy as np
import torch
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
WORKER = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                      "_q18_dist_worker.py")


@pytest.mark.timeout(300)
def test_q18_distributed_world2_matches_single(tmp_path):
    subprocess.run(["make", "-C", os.path.join(REPO, "oracle")], check=True,
                   capture_output=True)
    world = 2
    env = dict(os.environ)
    env.update(MASTER_ADDR="127.0.0.1", MASTER_PORT="29797",
               WORLD_SIZE=str(world), PYTHONPATH=REPO, OMP_NUM_THREADS="1")
    procs = []
    for rank in range(world):
        e = dict(env, RANK=str(rank))
        procs.append(subprocess.Popen([sys.executable, WORKER, str(tmp_path)],
                                      env=e, stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    outs = [p.communicate(timeout=240)[0].decode() for p in procs]
    for rank, p in enumerate(procs):
        assert p.returncode == 0, f"rank {rank} failed:\n{outs[rank]}"

    got_final = got_groups = got_surv = 0
    for rank in range(world):
        d = np.load(tmp_path / f"q18dist_{rank}.npz")
        got_final += int(d["final"])
        got_groups += int(d["groups"])
        got_surv += int(d["survivors"])

    from galaxysql_amd import abi
    from galaxysql_amd.queries import run_q18, gen_q18_numpy
    lib = abi.load_oracle()
    rng = np.random.default_rng(606)
    cust, orders, lineitem = gen_q18_numpy(rng, n_cust=3000, n_orders=30000,
                                           having_frac=0.002)
    t = [[torch.from_numpy(c) for c in cols]
         for cols in (cust, orders, lineitem)]
    n_final, info = run_q18(lib, -1, t[0], t[1], t[2])
    assert got_surv == info["survivors"]
    assert got_groups == info["groups"]
    assert got_final == n_final
