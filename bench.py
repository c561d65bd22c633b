#!/usr/bin/env python3
"""bench.py — headline benchmark for the MI355X operator pack.

Workload (BASELINE.json configs[1], the largest single-GPU judged config):
C2 = TPC-H SF10 join: orders 15M (build, o_orderkey = 4*i permuted) JOIN
lineitem ~60M (probe, 1-7 lines per order) on orderkey, full output
materialization. One "step" = one complete operator pass over the batch:
hash-table build + probe + payload gather, through the gxop C-ABI
(HIP/gfx950 kernels), inputs already resident in HBM.

Metric: probe rows/s, whole-job over all ranks (BASELINE.json: "probe
rows/s ... TPC-H ... 1/2/4/8 MI355X; %HBM roofline").

N>1: weak scaling with a REAL exchange step per the reference's hash
shuffle — each rank holds an SF10-sized shard, both sides are hash-
repartitioned (Java-exact murmur routing) and exchanged via RCCL
all-to-allv over xGMI, then joined locally.

CPU baseline (rank 0, N=1 only): the C++ oracle (oracle/, "port") timed
probe-only on a bounded sample on this box's host cores.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import numpy as np  # noqa: E402

HBM_PEAK_BYTES_PER_S = 8.0e12  # MI355X HBM3E spec peak (MI355X_MICROARCH.md)


def next_pow2(x):
    n = 2
    while n < x:
        n <<= 1
    return n


def gen_c2_device(device, seed, sf=10):
    """Synthetic TPC-H-shaped SF{sf} join inputs, generated directly in HBM."""
    import torch
    g = torch.Generator(device=device)
    g.manual_seed(seed)
    n_orders = sf * 1_500_000
    okeys = 4 * torch.randperm(n_orders, generator=g, device=device,
                               dtype=torch.int64)
    opay = torch.randint(0, 1 << 30, (n_orders,), generator=g, device=device,
                         dtype=torch.int64)
    # lineitem: 1-7 lines per order (TPC-H shape, mean 4 -> ~sf*6M rows)
    lines = torch.randint(1, 8, (n_orders,), generator=g, device=device,
                          dtype=torch.int64)
    lkeys = torch.repeat_interleave(okeys, lines)
    n_li = lkeys.numel()
    perm = torch.randperm(n_li, generator=g, device=device)
    lkeys = lkeys[perm].contiguous()
    lpay = torch.randint(0, 1 << 30, (n_li,), generator=g, device=device,
                         dtype=torch.int64)
    return (okeys, opay), (lkeys, lpay)


def run_join_step(lib, build_cols, probe_cols, device):
    """One full operator pass; returns (out_rows, stats dict)."""
    import ctypes as C
    from galaxysql_amd import abi
    from galaxysql_amd.abi import GxResult
    from galaxysql_amd.chunk import I64
    from galaxysql_amd.exchange import chunk_from_torch
    from galaxysql_amd.operators import ParallelHashJoinExec, EquiJoinKey

    op = ParallelHashJoinExec(
        lib, abi.INNER, [EquiJoinKey(0, 0, I64)], [I64, I64], [I64, I64],
        device=device, expected_build_rows=build_cols[0].numel())
    try:
        ka = []
        bc = chunk_from_torch(lib, list(build_cols), [I64, I64], ka)
        lib.check(lib.lib.gxop_join_consume(op._op, C.byref(bc)), "consume")
        op.build_consume()
        pc = chunk_from_torch(lib, list(probe_cols), [I64, I64], ka)
        out = C.POINTER(GxResult)()
        lib.check(lib.lib.gxop_join_probe(op._op, C.byref(pc), C.byref(out)),
                  "probe")
        n_out = out.contents.chunk.n_rows if out else 0
        if out:
            lib.lib.gxop_result_release(out)
        return n_out, op.stats()
    finally:
        op.close()


def cpu_baseline_leg(n_orders=15_000_000, n_probe_sample=8_000_000, seed=99):
    """Oracle ('port') timed on this box's host cores: full build table,
    probe-only timing on a bounded lineitem sample (~10-30 s of CPU work)."""
    import ctypes as C
    from galaxysql_amd import abi
    from galaxysql_amd.abi import GxResult
    from galaxysql_amd.chunk import I64
    from galaxysql_amd.operators import ParallelHashJoinExec, EquiJoinKey

    lib = abi.load_oracle()
    rng = np.random.default_rng(seed)
    okeys = 4 * rng.permutation(n_orders).astype(np.int64)
    opay = rng.integers(0, 1 << 30, n_orders, dtype=np.int64)
    pidx = rng.integers(0, n_orders, n_probe_sample)
    lkeys = okeys[pidx]
    lpay = rng.integers(0, 1 << 30, n_probe_sample, dtype=np.int64)

    op = ParallelHashJoinExec(lib, abi.INNER,
                              [EquiJoinKey(0, 0, I64)], [I64, I64], [I64, I64],
                              device=-1, expected_build_rows=n_orders)
    try:
        def mk_chunk(cols, ka):
            blocks = (abi.GxBlock * len(cols))()
            for i, a in enumerate(cols):
                blocks[i].type = I64
                blocks[i].mem = 0
                blocks[i].values = C.c_void_p(a.ctypes.data)
            ka.append(blocks)
            return abi.GxChunk(n_rows=len(cols[0]), n_blocks=len(cols),
                               blocks=blocks)

        ka = []
        bc = mk_chunk([okeys, opay], ka)
        lib.check(lib.lib.gxop_join_consume(op._op, C.byref(bc)), "consume")
        op.build_consume()
        pc = mk_chunk([lkeys, lpay], ka)
        t0 = time.perf_counter()
        out = C.POINTER(GxResult)()
        lib.check(lib.lib.gxop_join_probe(op._op, C.byref(pc), C.byref(out)), "probe")
        t1 = time.perf_counter()
        n_out = out.contents.chunk.n_rows if out else 0
        if out:
            lib.lib.gxop_result_release(out)
        assert n_out == n_probe_sample, f"cpu baseline self-check: {n_out}"
        return {
            "value": n_probe_sample / (t1 - t0),
            "unit": "rows/s",
            "cores": 1,
            "kind": "port",
            "sample": f"build {n_orders} orders (untimed) + {n_probe_sample} "
                      f"lineitem probe rows timed, single thread, -O3 -march=native",
        }
    finally:
        op.close()


def load_traffic(workload):
    """Per-launch HBM traffic measured by a separate rocprofv3 --pmc pass
    (profiles/); null when absent."""
    path = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                        "profiles", "pmc_traffic.json")
    if not os.path.exists(path):
        return None
    try:
        d = json.load(open(path))
        return d.get(workload)
    except Exception:
        return None


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--sf", type=int, default=10)
    ap.add_argument("--no-cpu-baseline", action="store_true")
    ap.add_argument("--cpu-probe-rows", type=int, default=8_000_000)
    args = ap.parse_args()

    import torch
    if not torch.cuda.is_available():
        print(json.dumps({"error": "no GPU available"}))
        sys.exit(1)

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    distributed = world > 1
    if distributed:
        import torch.distributed as dist
        torch.cuda.set_device(local_rank)
        dist.init_process_group("nccl")
    device = f"cuda:{local_rank}"
    torch.cuda.set_device(local_rank)

    from galaxysql_amd import abi
    from galaxysql_amd.chunk import I64
    from galaxysql_amd.exchange import shuffle_columns
    lib = abi.load_hip()  # the HIP extension — no CPU fallback exists

    (okeys, opay), (lkeys, lpay) = gen_c2_device(device, seed=1234 + rank,
                                                 sf=args.sf)
    n_build_local, n_probe_local = okeys.numel(), lkeys.numel()

    def barrier_sync():
        if distributed:
            import torch.distributed as dist
            dist.barrier()
        torch.cuda.synchronize()

    stats_acc = {"probe_kernel_ms": 0.0, "probe_launches": 0,
                 "probe_rows": 0, "matches": 0}
    out_rows_last = 0

    def step(accumulate):
        nonlocal out_rows_last
        if distributed:
            b = shuffle_columns(lib, [okeys, opay], [I64, I64], [0],
                                device=local_rank)
            p = shuffle_columns(lib, [lkeys, lpay], [I64, I64], [0],
                                device=local_rank)
        else:
            b, p = [okeys, opay], [lkeys, lpay]
        n_out, st = run_join_step(lib, b, p, local_rank)
        out_rows_last = n_out
        if accumulate:
            for k in stats_acc:
                stats_acc[k] += st[k]

    for _ in range(args.warmup):
        step(False)
    # self-check: every lineitem key exists in orders exactly once
    if not distributed:
        assert out_rows_last == n_probe_local, \
            f"join self-check failed: {out_rows_last} != {n_probe_local}"

    barrier_sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step(True)
    barrier_sync()
    t1 = time.perf_counter()

    elapsed = t1 - t0
    if distributed:
        import torch.distributed as dist
        t = torch.tensor([elapsed], device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())
        tot = torch.tensor([float(n_probe_local)], device=device)
        dist.all_reduce(tot)
        total_probe_rows_per_step = float(tot.item())
    else:
        total_probe_rows_per_step = float(n_probe_local)

    if rank != 0:
        return

    value = total_probe_rows_per_step * args.steps / elapsed

    # roofline for the dominant kernel (k_probe), HIP-event timed live.
    # Algorithmic bytes per probe row (DESIGN.md "Roofline accounting"):
    #   4 (hash) + 1 (nullflag) + 8 (bucket starts) + 8 (probe key)
    #   + 16*s (CSR entries scanned) + 8*m (emitted pair), with
    #   s = m + n_build/n_buckets (own match + expected colliders).
    n_build = n_build_local  # per-rank build size (weak scaling)
    n_buckets = next_pow2(max(2, n_build * 2))
    m_bar = (stats_acc["matches"] / stats_acc["probe_rows"]
             if stats_acc["probe_rows"] else 0.0)
    s_bar = m_bar + n_build / n_buckets
    bytes_per_row = 4 + 1 + 8 + 8 + 16 * s_bar + 8 * m_bar
    probe_ms = stats_acc["probe_kernel_ms"]
    achieved = (bytes_per_row * stats_acc["probe_rows"] / (probe_ms / 1e3)
                if probe_ms > 0 else 0.0)
    traffic = load_traffic("c2_sf%d" % args.sf)

    cpu_baseline = None
    if not args.no_cpu_baseline and world == 1:
        cpu_baseline = cpu_baseline_leg(n_orders=n_build_local,
                                        n_probe_sample=args.cpu_probe_rows)

    result = {
        "metric": "probe_rows_per_s",
        "value": value,
        "unit": "rows/s",
        "n_gpus": world,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": elapsed * 1000.0 / args.steps,
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "int64",
        "data": "synthetic",
        "config": {
            "workload": f"C2_tpch_sf{args.sf}_orders_join_lineitem",
            "orders_rows_per_gpu": n_build_local,
            "lineitem_rows_per_gpu": n_probe_local,
            "output": "full 4-column materialization",
            "exchange": "rccl_all_to_allv" if distributed else "none",
            "parallelism": f"hash_shuffle_dp{world}",
        },
        "roofline": {
            "bound": "hbm",
            "achieved": achieved / 1e9,
            "peak": HBM_PEAK_BYTES_PER_S / 1e9,
            "unit": "GB/s",
            "frac": achieved / HBM_PEAK_BYTES_PER_S,
            "traffic": traffic,
        },
        "cpu_baseline": cpu_baseline,
    }
    print(json.dumps(result))


if __name__ == "__main__":
    main()
