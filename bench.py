#!/usr/bin/env python3
"""bench.py — headline benchmark for the MI355X operator pack.

Metric (BASELINE.json): probe rows/s (+ agg groups/s) on TPC-H-shaped
synthetic workloads, whole-job over all ranks, %HBM roofline for the
dominant kernel.

Workloads:
  c3 (default) — HONEST SF100 Q3: UNFILTERED customer/orders/lineitem in
       HBM; the timed step runs the device-side vectorized filter +
       projection scans, the SEMI+INNER join chain and the 3-key GROUP BY
       SUM(revenue) as DOUBLE and as scaled-int64 cents (DECIMAL(15,2)-sum
       semantics). The config the metric is quoted on ("TPC-H SF100 Q3 ...
       1xMI355X"); fits one GPU.
  c2 — SF10 orders⋈lineitem full materialization (configs[1]).
  c2chunk — the CN's CHUNK_SIZE push cadence (per-chunk and buffered).
  c4 / c5 — SF100 Q18 and the per-GPU SF300/8 Q9 shard.

One "step" = one complete operator pass over the batch through the gxop
C-ABI (HIP/gfx950), inputs already resident in HBM. N>1: weak scaling with
the REAL hash-shuffle exchange (Java-exact murmur routing + RCCL
all-to-allv over xGMI) per the reference's FIXED shuffle; for Q3 the group
keys are orderkey-led, so rows group-colocate after the orderkey shuffle
and the local aggregate is final (the reference optimizer's own colocation
reasoning).

CPU baseline (rank 0, N=1 only): the C++ oracle ("port") on ALL host
cores, one thread per core (BASELINE.md protocol), bounded sample, core
count stated in the JSON.
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


def run_threaded_baseline(prep, n_threads):
    """All-cores CPU baseline runner (BASELINE.md protocol: one thread per
    core, all host cores, core count stated). prep(tid) does the UNTIMED
    data generation and returns a closure; the closures then run
    concurrently (the oracle's C calls release the GIL) and the timed
    region covers all of them. Returns (total_rows, wall_seconds)."""
    import concurrent.futures as cf
    with cf.ThreadPoolExecutor(n_threads) as ex:
        jobs = list(ex.map(prep, range(n_threads)))
        t0 = time.perf_counter()
        rows = list(ex.map(lambda f: f(), jobs))
        t1 = time.perf_counter()
    return sum(rows), t1 - t0


def probe_bytes_per_row(n_build, matches, probe_rows):
    """Algorithmic bytes per probe row (DESIGN.md 'Roofline accounting',
    inline-bucket table): 4 hash + 1 nullflag + 8 key + 16*s entry slots +
    8*m pair, where s = max(1, m + expected bucket colliders) — e0 (count +
    first entry, one 16-B slot) is always read; matches and colliders come
    from the same 64-B line."""
    n_buckets = next_pow2(max(2, n_build))  # 4-slot buckets
    if n_build > n_buckets * 3 // 4:
        n_buckets <<= 1                     # load cap 0.75 (gxhip do_build)
    m = matches / probe_rows if probe_rows else 0.0
    s = max(1.0, m + n_build / n_buckets)
    return 4 + 1 + 8 + 16 * s + 8 * m


def load_traffic(workload):
    path = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                        "profiles", "pmc_traffic.json")
    if not os.path.exists(path):
        return None
    try:
        return json.load(open(path)).get(workload)
    except Exception:
        return None


# --------------------------------------------------------------------------
# C2: SF10 orders ⋈ lineitem
# --------------------------------------------------------------------------

class C2:
    name = "c2"

    def __init__(self, args, device, rank):
        import torch
        self.sf = args.sf or 10
        g = torch.Generator(device=device)
        g.manual_seed(1234 + rank)
        n_orders = self.sf * 1_500_000
        okeys = 4 * torch.randperm(n_orders, generator=g, device=device,
                                   dtype=torch.int64)
        opay = torch.randint(0, 1 << 30, (n_orders,), generator=g,
                             device=device, dtype=torch.int64)
        lines = torch.randint(1, 8, (n_orders,), generator=g, device=device,
                              dtype=torch.int64)
        lkeys = torch.repeat_interleave(okeys, lines)
        perm = torch.randperm(lkeys.numel(), generator=g, device=device)
        lkeys = lkeys[perm].contiguous()
        lpay = torch.randint(0, 1 << 30, (lkeys.numel(),), generator=g,
                             device=device, dtype=torch.int64)
        self.build = [okeys, opay]
        self.probe = [lkeys, lpay]
        self.n_probe = lkeys.numel()
        self.n_build = okeys.numel()

    def step(self, lib, local_rank, distributed):
        import ctypes as C
        from galaxysql_amd import abi
        from galaxysql_amd.abi import GxResult
        from galaxysql_amd.chunk import I64
        from galaxysql_amd.exchange import chunk_from_torch, shuffle_columns
        from galaxysql_amd.operators import ParallelHashJoinExec, EquiJoinKey

        b, p = self.build, self.probe
        if distributed:
            b = shuffle_columns(lib, b, [I64, I64], [0], device=local_rank)
            p = shuffle_columns(lib, p, [I64, I64], [0], device=local_rank)
        op = ParallelHashJoinExec(
            lib, abi.INNER, [EquiJoinKey(0, 0, I64)], [I64, I64], [I64, I64],
            device=local_rank, expected_build_rows=b[0].numel())
        try:
            ka = []
            bc = chunk_from_torch(lib, b, [I64, I64], ka)
            lib.check(lib.lib.gxop_join_consume(op._op, C.byref(bc)), "consume")
            op.build_consume()
            pc = chunk_from_torch(lib, p, [I64, I64], ka)
            out = C.POINTER(GxResult)()
            lib.check(lib.lib.gxop_join_probe(op._op, C.byref(pc), C.byref(out)),
                      "probe")
            n_out = out.contents.chunk.n_rows if out else 0
            if out:
                lib.lib.gxop_result_release(out)
            st = op.stats()
            st["n_build"] = b[0].numel()
            st["groups"] = 0
            if not distributed:
                assert n_out == self.n_probe, \
                    f"C2 self-check: {n_out} != {self.n_probe}"
            return st
        finally:
            op.close()

    def config(self, world):
        return {"workload": f"C2_tpch_sf{self.sf}_orders_join_lineitem",
                "orders_rows_per_gpu": self.n_build,
                "lineitem_rows_per_gpu": self.n_probe,
                "output": "full 4-column materialization",
                "exchange": "rccl_all_to_allv" if world > 1 else "none",
                "parallelism": f"hash_shuffle_dp{world}"}

    def cpu_baseline(self, sample_rows, threads=1):
        return c2_cpu_baseline(self.n_build, sample_rows, threads=threads)

    traffic_key_fmt = "c2_sf%d"

    def traffic_key(self):
        return self.traffic_key_fmt % self.sf


class C2Chunk(C2):
    """C2 at the CN's real push cadence: the probe side arrives in
    CHUNK_SIZE-row chunks (ConnectionParams.java:1088 default 1000) through
    gxop_join_probe, one ABI call + kernel launch set per chunk — measures
    the per-chunk overhead the monolithic pass amortizes away (VERDICT r1
    weak #3). Build side stays one consume (the reference also accumulates
    the whole build before buildConsume)."""
    name = "c2chunk"

    def __init__(self, args, device, rank):
        # default to SF1 so a step (60k probe calls/SF) stays in seconds
        if args.sf is None:
            args.sf = 1
        super().__init__(args, device, rank)
        self.chunk_size = args.chunk_size
        self.flush_rows = args.flush_rows

    def step(self, lib, local_rank, distributed):
        import ctypes as C
        from galaxysql_amd import abi
        from galaxysql_amd.abi import GxResult
        from galaxysql_amd.chunk import I64
        from galaxysql_amd.exchange import chunk_from_torch
        from galaxysql_amd.operators import ParallelHashJoinExec, EquiJoinKey

        b, p = self.build, self.probe
        op = ParallelHashJoinExec(
            lib, abi.INNER, [EquiJoinKey(0, 0, I64)], [I64, I64], [I64, I64],
            device=local_rank, expected_build_rows=b[0].numel())
        try:
            ka = []
            bc = chunk_from_torch(lib, b, [I64, I64], ka)
            lib.check(lib.lib.gxop_join_consume(op._op, C.byref(bc)), "consume")
            op.build_consume()
            n_out = 0
            cs = self.chunk_size
            fr = self.flush_rows
            pushed = 0
            for lo in range(0, self.n_probe, cs):
                ka2 = []
                sl = [t[lo:lo + cs] for t in p]
                pc = chunk_from_torch(lib, sl, [I64, I64], ka2)
                if fr > 0:
                    # buffered cadence: push chunk, flush per `fr` rows
                    # (the shim's LocalBufferExec-style amortization)
                    lib.check(lib.lib.gxop_join_probe_push(op._op,
                                                           C.byref(pc)),
                              "push")
                    pushed += sl[0].numel()
                    if pushed >= fr:
                        out = C.POINTER(GxResult)()
                        lib.check(lib.lib.gxop_join_probe_flush(
                            op._op, C.byref(out)), "flush")
                        if out:
                            n_out += out.contents.chunk.n_rows
                            lib.lib.gxop_result_release(out)
                        pushed = 0
                else:
                    out = C.POINTER(GxResult)()
                    lib.check(lib.lib.gxop_join_probe(op._op, C.byref(pc),
                                                      C.byref(out)), "probe")
                    if out:
                        n_out += out.contents.chunk.n_rows
                        lib.lib.gxop_result_release(out)
            if fr > 0 and pushed:
                out = C.POINTER(GxResult)()
                lib.check(lib.lib.gxop_join_probe_flush(op._op, C.byref(out)),
                          "flush")
                if out:
                    n_out += out.contents.chunk.n_rows
                    lib.lib.gxop_result_release(out)
            st = op.stats()
            st["n_build"] = b[0].numel()
            st["groups"] = 0
            assert n_out == self.n_probe, (n_out, self.n_probe)
            return st
        finally:
            op.close()

    def config(self, world):
        c = super().config(world)
        c["workload"] = f"C2chunk_sf{self.sf}_probe_cadence"
        c["chunk_size"] = self.chunk_size
        c["flush_rows"] = self.flush_rows
        c["note"] = ("probe pushed in CHUNK_SIZE-row ABI calls "
                     "(flush_rows=0: one gxop_join_probe per chunk; >0: "
                     "probe_push per chunk + probe_flush per flush_rows); "
                     "compare value against the monolithic c2 line")
        return c


def c2_cpu_baseline(n_orders, n_probe_sample, threads=1, seed=99):
    """C2 baseline mirrors the reference's probe parallelism: ONE shared
    build (untimed, like the warmup-built Synchronizer table) probed by
    `threads` concurrent operator instances over disjoint probe slices
    (ParallelHashJoinExecutorFactory probeParallelism) — all probes timed
    together."""
    import ctypes as C
    from galaxysql_amd import abi
    from galaxysql_amd.abi import GxResult
    from galaxysql_amd.chunk import I64
    from galaxysql_amd.operators import ParallelHashJoinExec, EquiJoinKey

    lib = abi.load_oracle()
    rng = np.random.default_rng(seed)
    okeys = 4 * rng.permutation(n_orders).astype(np.int64)
    opay = rng.integers(0, 1 << 30, n_orders, dtype=np.int64)
    n_total = n_probe_sample * max(1, threads)
    lkeys = okeys[rng.integers(0, n_orders, n_total)]
    lpay = rng.integers(0, 1 << 30, n_total, dtype=np.int64)

    def mk_chunk(cols, ka):
        blocks = (abi.GxBlock * len(cols))()
        for i, a in enumerate(cols):
            blocks[i].type = I64
            blocks[i].mem = 0
            blocks[i].values = C.c_void_p(a.ctypes.data)
        ka.append(blocks)
        return abi.GxChunk(n_rows=len(cols[0]), n_blocks=len(cols),
                           blocks=blocks)

    op = ParallelHashJoinExec(lib, abi.INNER, [EquiJoinKey(0, 0, I64)],
                              [I64, I64], [I64, I64], device=-1,
                              expected_build_rows=n_orders)
    try:
        ka = []
        bc = mk_chunk([okeys, opay], ka)
        lib.check(lib.lib.gxop_join_consume(op._op, C.byref(bc)), "consume")
        op.build_consume()

        def prep(tid):
            lo = tid * n_probe_sample
            hi = lo + n_probe_sample
            ka2 = []
            pc = mk_chunk([np.ascontiguousarray(lkeys[lo:hi]),
                           np.ascontiguousarray(lpay[lo:hi])], ka2)

            def go():
                out = C.POINTER(GxResult)()
                lib.check(lib.lib.gxop_join_probe(op._op, C.byref(pc),
                                                  C.byref(out)), "probe")
                n_out = out.contents.chunk.n_rows if out else 0
                if out:
                    lib.lib.gxop_result_release(out)
                assert n_out == n_probe_sample
                _ = ka2  # keep chunk arrays alive
                return n_probe_sample

            return go

        rows, wall = run_threaded_baseline(prep, max(1, threads))
        return {"value": rows / wall, "unit": "rows/s",
                "cores": max(1, threads), "kind": "port",
                "sample": f"shared build {n_orders} orders (untimed) + "
                          f"{rows} probe rows timed across "
                          f"{max(1, threads)} threads (1/core), "
                          "-O3 -march=native"}
    finally:
        op.close()


# --------------------------------------------------------------------------
# C3: SF100 Q3 chain
# --------------------------------------------------------------------------

class C3:
    """HONEST SF100 Q3: unfiltered tables in HBM; the timed step runs the
    device-side vectorized filter+project scans (customer segment, orders
    date, lineitem shipdate; revenue = extendedprice*(1-discount) both as
    f64 and exact scaled-int), then the join chain + 3-key aggregate."""
    name = "c3"
    CUST_TOTAL = 15_000_000
    ORDERS_TOTAL = 150_000_000
    LINEITEM = 600_000_000

    def __init__(self, args, device, rank):
        import torch
        from galaxysql_amd.queries import Q3_SHIP_CUTOFF
        scale = args.c3_scale
        g = torch.Generator(device=device)
        g.manual_seed(4321 + rank)
        ct = int(self.CUST_TOTAL * scale)
        ot = int(self.ORDERS_TOTAL * scale)
        li = int(self.LINEITEM * scale)
        self.cust = [torch.arange(ct, dtype=torch.int64, device=device),
                     torch.randint(0, 5, (ct,), generator=g, device=device,
                                   dtype=torch.int32)]
        okeys_all = 4 * torch.randperm(ot, generator=g, device=device,
                                       dtype=torch.int64)
        self.orders = [
            torch.randint(0, ct, (ot,), generator=g, device=device,
                          dtype=torch.int64),
            okeys_all,
            torch.randint(8000, 9500, (ot,), generator=g, device=device,
                          dtype=torch.int32),
            torch.zeros(ot, dtype=torch.int32, device=device),
        ]
        lkeys = okeys_all[torch.randint(0, ot, (li,), generator=g,
                                        device=device)].contiguous()
        ship = torch.randint(8000, 9500, (li,), generator=g, device=device,
                             dtype=torch.int32)
        cents = torch.randint(100, 10_000_000, (li,), generator=g,
                              device=device, dtype=torch.int64)
        disc = torch.randint(0, 11, (li,), generator=g, device=device,
                             dtype=torch.int64)
        self.lineitem = [lkeys, ship, cents.double() / 100.0,
                         disc.double() / 100.0, cents, disc]
        # metric unit = join-probe rows = lineitem rows surviving the scan
        self.n_probe = int((ship > Q3_SHIP_CUTOFF).sum().item())
        self.n_scan = li
        self.expected_groups = int(ot * 0.486 * 0.2) + 1024
        self.last_info = None

    def step(self, lib, local_rank, distributed):
        from galaxysql_amd.exchange import shuffle_columns
        from galaxysql_amd.queries import (run_q3, run_q3_honest, CUST_TYPES,
                                           ORDERS_TYPES, LINEITEM_TYPES)
        if not distributed:
            _, info = run_q3_honest(lib, local_rank, self.cust, self.orders,
                                    self.lineitem,
                                    expected_groups=self.expected_groups,
                                    to_host=False)
        else:
            (cust, orders, lineitem), scanned = run_q3_honest(
                lib, local_rank, self.cust, self.orders, self.lineitem,
                as_tensors=True)
            cust = shuffle_columns(lib, cust, CUST_TYPES, [0],
                                   device=local_rank)
            orders = shuffle_columns(lib, orders, ORDERS_TYPES, [0],
                                     device=local_rank)
            lineitem = shuffle_columns(lib, lineitem, LINEITEM_TYPES, [0],
                                       device=local_rank)
            _, info = run_q3(lib, local_rank, cust, orders, lineitem,
                             expected_groups=self.expected_groups,
                             to_host=False, reshuffle_by_orderkey=True,
                             local_rank=local_rank)
            info.update(scanned)
        self.last_info = info
        st = dict(info["join2_stats"])
        st["n_build"] = info["orders_kept"]
        st["groups"] = info["groups"]
        return st

    def config(self, world):
        li = self.last_info or {}
        return {"workload": "C3_tpch_sf100_q3_join_chain_groupby",
                "customer_rows_per_gpu": self.cust[0].numel(),
                "orders_rows_per_gpu": self.orders[0].numel(),
                "lineitem_rows_scanned_per_gpu": self.n_scan,
                "lineitem_rows_joined_per_gpu": self.n_probe,
                "groups_last_step": li.get("groups"),
                "aggregates": "SUM(revenue) f64 + SUM(rev_scaled4) i64 + COUNT(*)",
                "inputs": "UNFILTERED; segment/date/shipdate filters + "
                          "revenue projection run on device in the timed step",
                "exchange": "rccl_all_to_allv" if world > 1 else "none",
                "parallelism": f"hash_shuffle_dp{world}"}

    def cpu_baseline(self, sample_rows, threads=1):
        return c3_cpu_baseline(sample_scale=1.0 / 128, threads=threads)

    def traffic_key(self):
        return "c3_sf100"


def c3_cpu_baseline(sample_scale, threads=1, seed=101):
    """Oracle honest-Q3 chain, all host cores: each thread runs the FULL
    chain (scans + builds + probes + aggregate, all timed) on its own
    1/128-of-SF100 shard — the MPP CN's own sharded data-parallel shape
    (each worker owns colocated shards). value = total joined rows / wall."""
    import torch
    from galaxysql_amd import abi
    from galaxysql_amd.queries import run_q3_honest, gen_q3_raw_numpy

    lib = abi.load_oracle()
    threads = max(1, threads)
    joined = [0] * threads

    def prep(tid):
        rng = np.random.default_rng(seed + tid)
        data = gen_q3_raw_numpy(rng,
                                n_cust=int(C3.CUST_TOTAL * sample_scale),
                                n_orders=int(C3.ORDERS_TOTAL * sample_scale),
                                n_lineitem=int(C3.LINEITEM * sample_scale))
        t = [[torch.from_numpy(a) for a in cols] for cols in data]

        def go():
            _, info = run_q3_honest(lib, -1, t[0], t[1], t[2], to_host=False)
            joined[tid] = info["lineitem_kept"]
            return info["lineitem_kept"]

        return go

    rows, wall = run_threaded_baseline(prep, threads)
    return {"value": rows / wall, "unit": "rows/s", "cores": threads,
            "kind": "port",
            "sample": f"honest Q3 chain (scans+joins+agg timed), "
                      f"{threads} threads (1/core) x 1/{int(1/sample_scale)}"
                      f"-SF100 shards, {rows} joined rows total, "
                      "-O3 -march=native"}


# --------------------------------------------------------------------------
# C4: SF100 Q18 (agg-dominant: ~150M groups)
# --------------------------------------------------------------------------

class C4:
    name = "c4"
    CUST = 15_000_000
    ORDERS = 150_000_000

    def __init__(self, args, device, rank):
        import torch
        scale = args.c4_scale
        g = torch.Generator(device=device)
        g.manual_seed(5678 + rank)
        nc = int(self.CUST * scale)
        no = int(self.ORDERS * scale)
        okeys = 4 * torch.randperm(no, generator=g, device=device,
                                   dtype=torch.int64)
        self.orders = [okeys,
                       torch.randint(0, nc, (no,), generator=g, device=device,
                                     dtype=torch.int64),
                       torch.randint(0, 1 << 30, (no,), generator=g,
                                     device=device, dtype=torch.int64)]
        lines = torch.randint(1, 8, (no,), generator=g, device=device,
                              dtype=torch.int64)
        lkeys = torch.repeat_interleave(okeys, lines)
        nl = lkeys.numel()
        perm = torch.randperm(nl, generator=g, device=device)
        lkeys = lkeys[perm].contiguous()
        lqty = torch.randint(1, 51, (nl,), generator=g, device=device,
                             dtype=torch.int64)
        hot = torch.rand(nl, generator=g, device=device) < 1e-5
        lqty = torch.where(hot, lqty + 400, lqty)
        self.lineitem = [lkeys, lqty]
        self.cust = [torch.arange(nc, dtype=torch.int64, device=device),
                     torch.randint(0, 1 << 30, (nc,), generator=g,
                                   device=device, dtype=torch.int64)]
        self.n_probe = nl  # metric unit: lineitem rows aggregated
        self.n_build = no
        self.expected_groups = no
        self.last_info = None

    def step(self, lib, local_rank, distributed):
        from galaxysql_amd.exchange import shuffle_columns
        from galaxysql_amd.queries import (run_q18, Q18_LINEITEM_TYPES,
                                           Q18_ORDERS_TYPES, Q18_CUST_TYPES)
        cust, orders, lineitem = self.cust, self.orders, self.lineitem
        if distributed:
            # Q18's group key IS the shuffle key: after the orderkey
            # all-to-allv the local aggregate is final (two-phase agg
            # degenerates to one local phase — SURVEY.md §8e).
            lineitem = shuffle_columns(lib, lineitem, Q18_LINEITEM_TYPES, [0],
                                       device=local_rank)
            orders = shuffle_columns(lib, orders, Q18_ORDERS_TYPES, [0],
                                     device=local_rank)
            cust = shuffle_columns(lib, cust, Q18_CUST_TYPES, [0],
                                   device=local_rank)
        _, info = run_q18(lib, local_rank, cust, orders, lineitem,
                          expected_groups=self.expected_groups,
                          reshuffle_by_custkey=distributed)
        self.last_info = info
        ast = info["agg_stats"]
        return {"probe_kernel_ms": ast["kernel_ms"], "probe_launches":
                ast["consumes"], "probe_rows": ast["rows"],
                "matches": ast["groups"], "n_build": ast["groups"],
                "groups": info["groups"]}

    def config(self, world):
        return {"workload": "C4_tpch_sf100_q18_groupby_having_join",
                "lineitem_rows_per_gpu": self.n_probe,
                "orders_rows_per_gpu": self.n_build,
                "groups_last_step": self.last_info["groups"] if self.last_info else None,
                "survivors_last_step": self.last_info["survivors"] if self.last_info else None,
                "aggregates": "SUM(l_quantity) i64, HAVING > 300",
                "exchange": "rccl_all_to_allv" if world > 1 else "none",
                "parallelism": f"hash_shuffle_dp{world}",
                "roofline_kernel": "agg insert+gid+accumulate chain"}

    def roofline_bytes_per_row(self, acc):
        # agg chain algorithmic bytes per input row (slot-indexed
        # records): keystore append 8 wr + hash 4 wr + 4 rd + insert slot 16
        # (+ keystore key compare 8 for the ~75% tie rows) + slot_of_row
        # 4 wr + 4 rd + record RMW 8 (records[slot] directly — no gid
        # indirection) + l_quantity 8 rd
        return 8 + 4 + 4 + 16 + 0.75 * 8 + 4 + 4 + 8 + 8

    def cpu_baseline(self, sample_rows, threads=1):
        return c4_cpu_baseline(sample_scale=1.0 / 128, threads=threads)

    def traffic_key(self):
        return "c4_sf100"


def c4_cpu_baseline(sample_scale, threads=1, seed=103):
    """Q18 chain, all host cores: one full chain per thread on its own
    1/128-SF100 shard (sharded data-parallel, value = rows / wall)."""
    import torch
    from galaxysql_amd import abi
    from galaxysql_amd.queries import run_q18, gen_q18_numpy

    lib = abi.load_oracle()
    threads = max(1, threads)

    def prep(tid):
        rng = np.random.default_rng(seed + tid)
        data = gen_q18_numpy(rng, n_cust=int(C4.CUST * sample_scale),
                             n_orders=int(C4.ORDERS * sample_scale),
                             having_frac=1e-5)
        t = [[torch.from_numpy(a) for a in cols] for cols in data]
        n_li = data[2][0].shape[0]

        def go():
            run_q18(lib, -1, t[0], t[1], t[2])
            return n_li

        return go

    rows, wall = run_threaded_baseline(prep, threads)
    return {"value": rows / wall, "unit": "rows/s", "cores": threads,
            "kind": "port",
            "sample": f"full Q18 chain (agg+joins timed), {threads} threads "
                      f"(1/core) x 1/{int(1/sample_scale)}-SF100 shards, "
                      f"{rows} lineitem rows total, -O3 -march=native"}


# --------------------------------------------------------------------------
# C5: SF300 Q9-shaped 6-way join chain, sharded 8-ways (per-GPU = SF37.5)
# --------------------------------------------------------------------------

class C5:
    name = "c5"
    # per-GPU shard of SF300/8 (weak scaling; BASELINE.md C5 = 8 GPUs)
    PART = 7_500_000
    SUPP = 375_000
    ORDERS = 56_250_000
    LINEITEM = 225_000_000
    N_NATION = 25

    def __init__(self, args, device, rank):
        import torch
        from galaxysql_amd.chunk import Block, Chunk, I64, SLICE
        from galaxysql_amd import abi as _abi
        from galaxysql_amd.queries import stage_table, Q9_PART_TYPES
        scale = args.c5_scale
        g = torch.Generator(device=device)
        g.manual_seed(6789 + rank)
        npart = int(self.PART * scale)
        nsupp = int(self.SUPP * scale)
        nord = int(self.ORDERS * scale)
        nli = int(self.LINEITEM * scale)

        # p_name: one 8-char word from a 92-word vocab, 5 of which contain
        # "green" -> ~5.4 % LIKE selectivity
        rng = np.random.default_rng(6789 + rank)
        vocab = [f"w{i:02d}filler"[:8] for i in range(87)] + \
                ["green00", "greenish", "00green0", "agreenxx", "greeny00"]
        vocab = [w.ljust(8, "x")[:8] for w in vocab]
        vb = np.frombuffer("".join(vocab).encode(), dtype=np.uint8).reshape(-1, 8)
        idx = rng.integers(0, len(vocab), npart)
        data = vb[idx].reshape(-1)
        offsets = (np.arange(1, npart + 1, dtype=np.int32) * 8)
        pkeys = np.arange(npart, dtype=np.int64)
        part_chunk = Chunk([Block(I64, values=pkeys),
                            Block(SLICE, offsets=offsets, data=data)])
        self.lib = None  # staged lazily (needs the lib handle)
        self._part_chunk = part_chunk
        self.part_res = None
        self.device = device

        self.supplier = [torch.arange(nsupp, dtype=torch.int64, device=device),
                         torch.randint(0, self.N_NATION, (nsupp,), generator=g,
                                       device=device, dtype=torch.int64)]
        ps_part = torch.repeat_interleave(
            torch.arange(npart, dtype=torch.int64, device=device), 4)
        first = torch.randint(0, nsupp, (npart,), generator=g, device=device,
                              dtype=torch.int64)
        step4 = torch.arange(4, device=device, dtype=torch.int64) * max(1, nsupp // 5)
        ps_supp = ((first.unsqueeze(1) + step4.unsqueeze(0)).reshape(-1)) % nsupp
        ps_cost = torch.randint(100, 100_000, (ps_part.numel(),), generator=g,
                                device=device, dtype=torch.int64)
        self.partsupp = [ps_part.contiguous(), ps_supp.contiguous(), ps_cost]

        okeys = 4 * torch.randperm(nord, generator=g, device=device,
                                   dtype=torch.int64)
        self.orders = [okeys,
                       torch.randint(1992, 1999, (nord,), generator=g,
                                     device=device, dtype=torch.int32)]
        pick = torch.randint(0, ps_part.numel(), (nli,), generator=g,
                             device=device)
        self.lineitem = [ps_part[pick].contiguous(), ps_supp[pick].contiguous(),
                         okeys[torch.randint(0, nord, (nli,), generator=g,
                                             device=device)].contiguous(),
                         torch.randint(1, 51, (nli,), generator=g,
                                       device=device, dtype=torch.int64),
                         torch.randint(100, 10_000_000, (nli,), generator=g,
                                       device=device, dtype=torch.int64),
                         torch.randint(0, 11, (nli,), generator=g,
                                       device=device, dtype=torch.int64)]
        self.n_probe = nli
        self.n_build = ps_part.numel()
        self.last_info = None

    def step(self, lib, local_rank, distributed):
        from galaxysql_amd.queries import run_q9, stage_table, Q9_PART_TYPES
        if self.part_res is None:
            # one-time device upload of the part table (strings), untimed
            # relative to steady state (first call is warmup)
            self.part_res = stage_table(lib, self._part_chunk, Q9_PART_TYPES,
                                        local_rank)
        world = int(os.environ.get("WORLD_SIZE", "1"))
        rows, info = run_q9(lib, local_rank, self.part_res, self.supplier,
                            self.partsupp, self.orders, self.lineitem,
                            world=world, local_rank=local_rank)
        self.last_info = info
        st = dict(info["join2_stats"])
        st["n_build"] = info["partsupp_kept"]
        st["groups"] = info["groups"]
        return st

    def config(self, world):
        li = self.last_info or {}
        return {"workload": "C5_tpch_sf300_q9_6way_join_like_decimal",
                "part_rows_per_gpu": self._part_chunk.n_rows,
                "partsupp_rows_per_gpu": self.n_build,
                "orders_rows_per_gpu": self.orders[0].numel(),
                "lineitem_rows_per_gpu": self.n_probe,
                "part_like_kept": li.get("part_kept"),
                "lineitem_joined": li.get("lineitem_joined"),
                "groups_last_step": li.get("groups"),
                "aggregates": "SUM(amount scale-4 scaled-int, exact DECIMAL) "
                              "GROUP BY (nation, year), two-phase at N>1",
                "exchange": "rccl_all_to_allv" if world > 1 else "none",
                "parallelism": f"hash_shuffle_dp{world}"}

    def cpu_baseline(self, sample_rows, threads=1):
        return c5_cpu_baseline(sample_scale=1.0 / 128, threads=threads)

    def traffic_key(self):
        return "c5_sf300"


def c5_cpu_baseline(sample_scale, threads=1, seed=107):
    """Q9 chain, all host cores: one full chain per thread on its own
    1/128 shard of the per-GPU SF300/8 slice."""
    import torch
    from galaxysql_amd import abi
    from galaxysql_amd.chunk import Block, Chunk, I64, SLICE
    from galaxysql_amd.queries import (run_q9, gen_q9_numpy, stage_table,
                                       Q9_PART_TYPES)

    lib = abi.load_oracle()
    threads = max(1, threads)
    frees = []

    def prep(tid):
        rng = np.random.default_rng(seed + tid)
        data = gen_q9_numpy(rng, n_part=int(C5.PART * sample_scale),
                            n_supp=int(C5.SUPP * sample_scale),
                            n_orders=int(C5.ORDERS * sample_scale),
                            n_lineitem=int(C5.LINEITEM * sample_scale))
        part_chunk = Chunk([Block(I64, values=data[0][0]),
                            Block.of(SLICE, data[0][1])])
        part_res = stage_table(lib, part_chunk, Q9_PART_TYPES, -1)
        frees.append(part_res)
        t = [[torch.from_numpy(a) for a in cols] for cols in data[1:]]
        n_li = data[4][0].shape[0]

        def go():
            run_q9(lib, -1, part_res, t[0], t[1], t[2], t[3])
            return n_li

        return go

    rows, wall = run_threaded_baseline(prep, threads)
    for r in frees:
        lib.lib.gxop_result_release(r)
    return {"value": rows / wall, "unit": "rows/s", "cores": threads,
            "kind": "port",
            "sample": f"full Q9 chain, {threads} threads (1/core) x "
                      f"1/{int(1/sample_scale)} per-GPU-shard samples, "
                      f"{rows} lineitem rows total, -O3 -march=native"}


WORKLOADS = {"c2": C2, "c2chunk": C2Chunk, "c3": C3, "c4": C4, "c5": C5}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--workload", choices=list(WORKLOADS), default="c3")
    ap.add_argument("--sf", type=int, default=None, help="C2 scale factor")
    ap.add_argument("--chunk-size", type=int, default=1000,
                    help="c2chunk probe cadence (CHUNK_SIZE default 1000)")
    ap.add_argument("--flush-rows", type=int, default=0,
                    help="c2chunk: 0 = probe per chunk; >0 = probe_push "
                         "per chunk + probe_flush per this many rows")
    ap.add_argument("--c3-scale", type=float, default=1.0,
                    help="C3 size fraction of SF100 (1.0 = full)")
    ap.add_argument("--c4-scale", type=float, default=1.0,
                    help="C4 size fraction of SF100 (1.0 = full)")
    ap.add_argument("--c5-scale", type=float, default=1.0,
                    help="C5 size fraction of the per-GPU SF300/8 shard")
    ap.add_argument("--no-cpu-baseline", action="store_true")
    ap.add_argument("--cpu-probe-rows", type=int, default=4_000_000,
                    help="C2 baseline probe rows PER THREAD")
    ap.add_argument("--cpu-threads", type=int, default=0,
                    help="CPU-baseline threads; 0 = one per host core "
                         "(BASELINE.md protocol)")
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
    lib = abi.load_hip()  # the HIP extension — no CPU fallback exists

    wl = WORKLOADS[args.workload](args, device, rank)

    def barrier_sync():
        if distributed:
            import torch.distributed as dist
            dist.barrier()
        torch.cuda.synchronize()

    acc = {"probe_kernel_ms": 0.0, "probe_launches": 0, "probe_rows": 0,
           "matches": 0, "n_build": 0, "groups": 0}

    for _ in range(args.warmup):
        wl.step(lib, local_rank, distributed)

    barrier_sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        st = wl.step(lib, local_rank, distributed)
        for k in acc:
            acc[k] += st.get(k, 0)
    barrier_sync()
    t1 = time.perf_counter()

    elapsed = t1 - t0
    if distributed:
        import torch.distributed as dist
        t = torch.tensor([elapsed], device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())
        tot = torch.tensor([float(wl.n_probe)], device=device)
        dist.all_reduce(tot)
        total_probe_rows_per_step = float(tot.item())
    else:
        total_probe_rows_per_step = float(wl.n_probe)

    if rank != 0:
        return

    value = total_probe_rows_per_step * args.steps / elapsed
    n_build_avg = acc["n_build"] / args.steps
    if hasattr(wl, "roofline_bytes_per_row"):
        bpr = wl.roofline_bytes_per_row(acc)
    else:
        bpr = probe_bytes_per_row(int(n_build_avg), acc["matches"],
                                  acc["probe_rows"])
    probe_ms = acc["probe_kernel_ms"]
    achieved = (bpr * acc["probe_rows"] / (probe_ms / 1e3)) if probe_ms else 0.0

    cpu_baseline = None
    if not args.no_cpu_baseline and world == 1:
        threads = args.cpu_threads or (os.cpu_count() or 1)
        cpu_baseline = wl.cpu_baseline(args.cpu_probe_rows, threads=threads)

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
        "config": dict(wl.config(world),
                       agg_groups_per_s=(acc["groups"] / elapsed)
                       if acc["groups"] else None,
                       probe_launches=acc["probe_launches"],
                       probe_event_ms_per_launch=(
                           acc["probe_kernel_ms"] / acc["probe_launches"])
                       if acc["probe_launches"] else None),
        "roofline": {
            "bound": "hbm",
            "achieved": achieved / 1e9,
            "peak": HBM_PEAK_BYTES_PER_S / 1e9,
            "unit": "GB/s",
            "frac": achieved / HBM_PEAK_BYTES_PER_S,
            "traffic": load_traffic(wl.traffic_key()),
        },
        "cpu_baseline": cpu_baseline,
    }
    print(json.dumps(result))


if __name__ == "__main__":
    main()
