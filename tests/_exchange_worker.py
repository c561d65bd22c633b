"""Worker for test_exchange_cpu.py (runs under torch.distributed gloo)."""
import os
import sys

import numpy as np
import torch
import torch.distributed as dist

sys.path.insert(0, os.environ["PYTHONPATH"])

from galaxysql_amd import abi
from galaxysql_amd.chunk import I64
from galaxysql_amd.exchange import shuffle_columns


def java_i32(x):
    x &= 0xFFFFFFFF
    return x - (1 << 32) if x >= (1 << 31) else x


def murmur3(x):
    x &= 0xFFFFFFFF
    x ^= x >> 16
    x = (x * 0x85EBCA6B) & 0xFFFFFFFF
    x ^= x >> 13
    x = (x * 0xC2B2AE35) & 0xFFFFFFFF
    x ^= x >> 16
    return x


def expected_rank(key, world):
    h = java_i32((key & 0xFFFFFFFFFFFFFFFF) ^ ((key & 0xFFFFFFFFFFFFFFFF) >> 32))
    return murmur3(h & 0xFFFFFFFF) & (world - 1)


def main():
    outdir = sys.argv[1]
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    dist.init_process_group("gloo", rank=rank, world_size=world)

    rng = np.random.default_rng(100 + rank)
    n = 5000
    keys = torch.from_numpy(rng.integers(-(1 << 50), 1 << 50, n, dtype=np.int64))
    pay = torch.from_numpy(rng.integers(0, 1 << 31, n, dtype=np.int64))
    np.savez(os.path.join(outdir, f"sent_{rank}.npz"),
             keys=keys.numpy(), pay=pay.numpy())

    lib = abi.load_oracle()
    recv = shuffle_columns(lib, [keys, pay], [I64, I64], [0], device=-1)
    rkeys, rpay = recv[0].numpy(), recv[1].numpy()
    np.savez(os.path.join(outdir, f"recv_{rank}.npz"), keys=rkeys, pay=rpay)

    # routing: every received key must belong on this rank (Java-exact)
    for k in rkeys[:500]:
        assert expected_rank(int(k), world) == rank, \
            f"key {k} misrouted to rank {rank}"
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
