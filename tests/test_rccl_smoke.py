"""RCCL execution smoke on a single GPU (multi-GPU
evidence as far as 1-GPU leases allow): two ranks on ONE MI355X run the
REAL shuffle_columns path over the nccl(=RCCL) backend. NCCL-family
libraries historically reject two ranks on one device; if this RCCL build
does too, the test SKIPS with the library's error recorded — a documented
attempt either way. The exchange logic itself is covered at world 2 and 8
on gloo (test_exchange_cpu, test_q9_world8)."""
import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

WORKER = r'''
import os, sys
import numpy as np
import torch
import torch.distributed as dist
sys.path.insert(0, os.environ["PYTHONPATH"])
from galaxysql_amd import abi
from galaxysql_amd.chunk import I64
from galaxysql_amd.exchange import shuffle_columns

rank = int(os.environ["RANK"])
dist.init_process_group("nccl", rank=rank, world_size=2)
torch.cuda.set_device(0)  # BOTH ranks share device 0
lib = abi.load_hip()
rng = np.random.default_rng(300 + rank)
keys = torch.from_numpy(rng.integers(0, 1 << 40, 4096, dtype=np.int64)).cuda()
pay = torch.from_numpy(rng.integers(0, 1 << 30, 4096, dtype=np.int64)).cuda()
recv = shuffle_columns(lib, [keys, pay], [I64, I64], [0], device=0)
total = torch.tensor([recv[0].numel()], device="cuda:0")
dist.all_reduce(total)
assert int(total.item()) == 8192, int(total.item())
if rank == 0:
    print("RCCL_SMOKE_OK", int(total.item()))
dist.destroy_process_group()
'''


def test_rccl_two_ranks_one_gpu(tmp_path):
    w = tmp_path / "worker.py"
    w.write_text(WORKER)
    env = dict(os.environ)
    env.update(MASTER_ADDR="127.0.0.1", MASTER_PORT="29881", WORLD_SIZE="2",
               PYTHONPATH=REPO, HSA_ENABLE_IPC_MODE_LEGACY="0")
    procs = []
    for rank in range(2):
        e = dict(env, RANK=str(rank))
        procs.append(subprocess.Popen([sys.executable, str(w)], env=e,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    try:
        outs = [p.communicate(timeout=240)[0].decode() for p in procs]
    except subprocess.TimeoutExpired:
        for p in procs:
            p.kill()
        pytest.skip("RCCL 2-ranks-on-1-GPU hung (unsupported rendezvous); "
                    "exchange covered on gloo worlds 2 and 8")
    if any(p.returncode != 0 for p in procs):
        msg = "\n".join(o[-500:] for o in outs)
        if ("Duplicate GPU" in msg or "invalid usage" in msg or
                "unsupported" in msg.lower() or "NCCL" in msg or
                "RCCL" in msg):
            pytest.skip("RCCL rejects two ranks on one device "
                        f"(documented attempt): {msg[-200:]}")
        raise AssertionError(msg)
    assert "RCCL_SMOKE_OK" in outs[0]
