"""Hash-repartition exchange across ranks — the MI355X stand-in for the
reference's cross-node MPP hash shuffle (PartitionedOutputCollector +
HTTP ExchangeClient): RCCL all-to-allv over xGMI, one
process per GPU (`torch.distributed` backend "nccl" IS RCCL on ROCm).

Routing is Java-exact (gxop partition: murmurHash3(rowHash) & (world-1),
utils/ExecUtils.java:1023-1033) so each rank receives exactly the keys the
reference's FIXED shuffle would send it. The all-to-all shape is chosen for
xGMI: 7 direct p2p links per GPU carry the 7 peer partitions concurrently
(SURVEY.md §5), vs ring collectives which are single-link bound.
"""
from __future__ import annotations

import ctypes as C

import torch
import torch.distributed as dist

from . import abi
from .abi import GxResult, GxPartCfg
from .chunk import I64, I32, F64

_TORCH_DTYPES = {I64: torch.int64, I32: torch.int32, F64: torch.float64}


def chunk_from_torch(lib, tensors, types, keepalive):
    """Build a GxChunk over torch tensors (cpu or cuda) without copying."""
    n = len(tensors)
    blocks = (abi.GxBlock * n)()
    n_rows = tensors[0].numel()
    for i, (t, ty) in enumerate(zip(tensors, types)):
        assert t.is_contiguous()
        gb = blocks[i]
        gb.type = ty
        gb.mem = 1 if t.is_cuda else 0
        gb.values = C.c_void_p(t.data_ptr())
        gb.nulls = None
        gb.offsets = None
        gb.data = None
        gb.data_len = 0
    gc = abi.GxChunk(n_rows=n_rows, n_blocks=n, blocks=blocks)
    keepalive.append(blocks)
    keepalive.extend(tensors)
    return gc


def partition_concat(lib, tensors, types, key_cols, n_parts, device=-1, stream=0):
    """Partition columns (torch tensors, no nulls) into ONE concatenated
    result grouped by partition id + split counts. Returns (torch tensors
    list, counts list)."""
    kc = (C.c_int32 * len(key_cols))(*key_cols)
    it = (C.c_int32 * len(types))(*types)
    cfg = GxPartCfg(n_parts=n_parts, n_key_cols=len(key_cols), key_cols=kc,
                    n_input_cols=len(types), input_types=it,
                    device=device, stream=stream)
    op = lib.lib.gxop_part_create(C.byref(cfg))
    if not op:
        raise RuntimeError(f"part_create: {lib.error()}")
    try:
        ka = []
        gc = chunk_from_torch(lib, tensors, types, ka)
        out = C.POINTER(GxResult)()
        counts = (C.c_int64 * n_parts)()
        lib.check(lib.lib.gxop_part_consume_concat(op, C.byref(gc),
                                                   C.byref(out), counts),
                  "part_consume_concat")
        n = out.contents.chunk.n_rows
        dev = tensors[0].device
        outs = []
        for ci, ty in enumerate(types):
            t = torch.empty(n, dtype=_TORCH_DTYPES[ty], device=dev)
            if n > 0:
                lib.check(lib.lib.gxop_result_copy_col(
                    out, ci, C.c_void_p(t.data_ptr()), None), "result_copy_col")
            outs.append(t)
        lib.lib.gxop_result_release(out)
        return outs, list(counts)
    finally:
        lib.lib.gxop_part_close(op)


def all_to_all_columns(send_cols, in_splits, group=None):
    """Exchange partition-grouped columns: every rank sends slice p to rank p.
    Returns received columns (concatenated over source ranks)."""
    world = dist.get_world_size(group)
    in_t = torch.tensor(in_splits, dtype=torch.int64)
    out_t = torch.empty(world, dtype=torch.int64)
    dist.all_to_all_single(out_t, in_t, group=group)
    out_splits = [int(x) for x in out_t]
    n_recv = sum(out_splits)
    recv_cols = []
    for col in send_cols:
        recv = torch.empty(n_recv, dtype=col.dtype, device=col.device)
        dist.all_to_all_single(recv, col, out_splits, in_splits, group=group)
        recv_cols.append(recv)
    return recv_cols


def shuffle_columns(lib, tensors, types, key_cols, device=-1, group=None):
    """Full hash-shuffle: Java-exact partition by key + all-to-allv.
    Each rank ends up owning rows with murmur(hash) % world == rank."""
    world = dist.get_world_size(group)
    send_cols, counts = partition_concat(lib, tensors, types, key_cols, world,
                                         device=device)
    return all_to_all_columns(send_cols, counts, group=group)


def final_agg_specs(n_group_cols, aggs):
    """Partial->final aggregation mapping for the MPP two-phase plan (the
    reference planner splits an exchanged aggregation into a partial
    HashAggExec below the shuffle and a final one above it; re-aggregation
    functions follow SQL: COUNT re-aggregates as SUM, SUM as SUM, MIN/MAX
    as themselves).

    Input: the PARTIAL phase's agg specs [(func, input_col)].
    Output: (final_specs, final_input_types_suffix) where final specs
    index the partial OUTPUT schema: group cols first, then one column
    per partial aggregator."""
    from . import abi
    finals = []
    out_types = []
    for i, (func, _col) in enumerate(aggs):
        col = n_group_cols + i
        if func == abi.AVG_F64:
            raise ValueError(
                "AVG must be planned as partial SUM+COUNT around an "
                "exchange (the reference planner does the same split); "
                "it never crosses a shuffle as AVG")
        if func in (abi.COUNT_ROW, abi.COUNT_COL):
            finals.append((abi.SUM_I64, col))
            out_types.append(0)  # I64
        elif func in (abi.SUM_I64, abi.SUM_I64N, abi.MIN_I64, abi.MAX_I64,
                      abi.BIT_AND, abi.BIT_OR, abi.BIT_XOR):
            # bit aggs re-aggregate as themselves over I64 partial values
            finals.append((func, col))
            out_types.append(0)
        elif func in (abi.SUM_F64, abi.MIN_F64, abi.MAX_F64):
            finals.append((func, col))
            out_types.append(2)  # F64
        else:
            raise ValueError(
                f"agg func {func} has no defined partial->final mapping "
                "for the two-phase exchanged plan")
    return finals, out_types
