"""ctypes bindings for the gxop C-ABI (include/gxop.h).

Used for BOTH libraries:
  - galaxysql_amd/csrc -> libgxhip.so   (the product: HIP/gfx950 kernels)
  - oracle/libgxoracle.so               (test-only CPU parity checker)

The product path must load libgxhip.so and FAIL LOUDLY if it is missing on a
GPU box; only tests/bench baseline code may load the oracle.
"""
from __future__ import annotations

import ctypes as C
import os

import numpy as np

from .chunk import (Block, Chunk, I64, I32, F64, SLICE, DECIMAL,
                    _NP_DTYPES)

_REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


class GxBlock(C.Structure):
    _fields_ = [
        ("type", C.c_int32),
        ("mem", C.c_int32),
        ("values", C.c_void_p),
        ("nulls", C.c_void_p),
        ("offsets", C.c_void_p),
        ("data", C.c_void_p),
        ("data_len", C.c_int64),
    ]


class GxChunk(C.Structure):
    _fields_ = [
        ("n_rows", C.c_int32),
        ("n_blocks", C.c_int32),
        ("blocks", C.POINTER(GxBlock)),
    ]


class GxResult(C.Structure):
    _fields_ = [("chunk", GxChunk), ("opaque", C.c_void_p)]


class GxEquiKey(C.Structure):
    _fields_ = [
        ("outer_index", C.c_int32),
        ("inner_index", C.c_int32),
        ("unified_type", C.c_int32),
        ("null_safe_equal", C.c_int32),
    ]


class GxJoinCond(C.Structure):
    """Residual (non-equi) join condition term (gx_join_cond in gxop.h):
    AND of terms over the reference's condition-row layout (leftSide cols
    then rightSide cols, JoinRelType.java:145-151)."""
    _fields_ = [
        ("col_a", C.c_int32),
        ("cmp", C.c_int32),
        ("col_b", C.c_int32),
        ("v_i64", C.c_int64),
        ("v_f64", C.c_double),
        ("v_bytes", C.POINTER(C.c_uint8)),
        ("v_len", C.c_int32),
        ("const_is_null", C.c_int32),
    ]


class GxJoinCfg(C.Structure):
    _fields_ = [
        ("join_type", C.c_int32),
        ("single_join", C.c_int32),
        ("build_outer", C.c_int32),
        ("n_keys", C.c_int32),
        ("keys", C.POINTER(GxEquiKey)),
        ("n_outer_cols", C.c_int32),
        ("outer_types", C.POINTER(C.c_int32)),
        ("n_inner_cols", C.c_int32),
        ("inner_types", C.POINTER(C.c_int32)),
        ("anti_null_col", C.c_int32),
        ("device", C.c_int32),
        ("stream", C.c_uint64),
        ("expected_build_rows", C.c_int64),
        ("n_out_proj", C.c_int32),
        ("out_proj", C.POINTER(C.c_int32)),
        ("memory_budget_bytes", C.c_int64),
        ("n_conds", C.c_int32),
        ("conds", C.POINTER(GxJoinCond)),
        ("enable_bloom", C.c_int32),
    ]


class GxAggSpec(C.Structure):
    _fields_ = [("func", C.c_int32), ("input_col", C.c_int32)]


class GxAggCfg(C.Structure):
    _fields_ = [
        ("n_group_cols", C.c_int32),
        ("group_cols", C.POINTER(C.c_int32)),
        ("n_aggs", C.c_int32),
        ("aggs", C.POINTER(GxAggSpec)),
        ("n_input_cols", C.c_int32),
        ("input_types", C.POINTER(C.c_int32)),
        ("expected_groups", C.c_int64),
        ("device", C.c_int32),
        ("stream", C.c_uint64),
    ]


class GxJoinStats(C.Structure):
    _fields_ = [
        ("probe_kernel_ms", C.c_double),
        ("probe_launches", C.c_int64),
        ("probe_rows", C.c_int64),
        ("matches", C.c_int64),
    ]


class GxAggStats(C.Structure):
    _fields_ = [
        ("kernel_ms", C.c_double),
        ("consumes", C.c_int64),
        ("rows", C.c_int64),
        ("groups", C.c_int64),
    ]


class GxGroupJoinCfg(C.Structure):
    _fields_ = [
        ("join_type", C.c_int32),
        ("n_keys", C.c_int32),
        ("keys", C.POINTER(GxEquiKey)),
        ("n_build_cols", C.c_int32),
        ("build_types", C.POINTER(C.c_int32)),
        ("n_probe_cols", C.c_int32),
        ("probe_types", C.POINTER(C.c_int32)),
        ("n_group_cols", C.c_int32),
        ("group_cols", C.POINTER(C.c_int32)),
        ("n_aggs", C.c_int32),
        ("aggs", C.POINTER(GxAggSpec)),
        ("device", C.c_int32),
        ("stream", C.c_uint64),
        ("expected_build_rows", C.c_int64),
    ]


class GxWindowCfg(C.Structure):
    _fields_ = [
        ("n_part_cols", C.c_int32),
        ("part_cols", C.POINTER(C.c_int32)),
        ("n_order_cols", C.c_int32),
        ("order_cols", C.POINTER(C.c_int32)),
        ("n_aggs", C.c_int32),
        ("aggs", C.POINTER(GxAggSpec)),
        ("reset", C.POINTER(C.c_uint8)),
        ("n_input_cols", C.c_int32),
        ("input_types", C.POINTER(C.c_int32)),
        ("device", C.c_int32),
        ("stream", C.c_uint64),
    ]


class GxFrameSpec(C.Structure):
    _fields_ = [
        ("func", C.c_int32),
        ("input_col", C.c_int32),
        ("kind", C.c_int32),
        ("preceding", C.c_int64),
        ("following", C.c_int64),
        ("order_col", C.c_int32),   # RANGE kinds: the ORDER BY column
        ("order_asc", C.c_int32),   # RANGE kinds: 1 = asc, 0 = desc
    ]


class GxFWindowCfg(C.Structure):
    _fields_ = [
        ("n_part_cols", C.c_int32),
        ("part_cols", C.POINTER(C.c_int32)),
        ("n_order_cols", C.c_int32),
        ("order_cols", C.POINTER(C.c_int32)),
        ("n_frames", C.c_int32),
        ("frames", C.POINTER(GxFrameSpec)),
        ("n_input_cols", C.c_int32),
        ("input_types", C.POINTER(C.c_int32)),
        ("device", C.c_int32),
        ("stream", C.c_uint64),
    ]


class GxPartCfg(C.Structure):
    _fields_ = [
        ("n_parts", C.c_int32),
        ("n_key_cols", C.c_int32),
        ("key_cols", C.POINTER(C.c_int32)),
        ("n_input_cols", C.c_int32),
        ("input_types", C.POINTER(C.c_int32)),
        ("device", C.c_int32),
        ("stream", C.c_uint64),
    ]


class GxPred(C.Structure):
    _fields_ = [("col", C.c_int32), ("cmp", C.c_int32),
                ("v_i64", C.c_int64), ("v_f64", C.c_double),
                ("v_bytes", C.c_void_p), ("v_len", C.c_int32)]


class GxProj(C.Structure):
    _fields_ = [("op", C.c_int32), ("a", C.c_int32), ("b", C.c_int32),
                ("c", C.c_int32), ("d", C.c_int32)]


class GxScanCfg(C.Structure):
    _fields_ = [
        ("n_preds", C.c_int32), ("preds", C.POINTER(GxPred)),
        ("n_projs", C.c_int32), ("projs", C.POINTER(GxProj)),
        ("n_input_cols", C.c_int32), ("input_types", C.POINTER(C.c_int32)),
        ("device", C.c_int32), ("stream", C.c_uint64),
    ]


# Join types (gx_join_type)
INNER, LEFT, RIGHT, SEMI, ANTI = 0, 1, 2, 3, 4
# Comparisons (gx_cmp)
LT, LE, GT, GE, EQ, NE = 0, 1, 2, 3, 4, 5
CONTAINS = 6  # SLICE LIKE '%pat%'
# join-condition-only null-safe compares (Objects.equals semantics)
EQ_NULLSAFE, NE_NULLSAFE = 7, 8
# Projections (gx_proj_op)
PROJ_COPY, PROJ_REV_F64, PROJ_REV_SCALED4, PROJ_Q9_AMOUNT4 = 0, 1, 2, 3
PROJ_DEC_TO_SCALED, PROJ_SCALED_TO_DEC = 4, 5  # gx_proj.c = decimal scale
FRAME_WHOLE_PARTITION, FRAME_ROWS_SLIDING, FRAME_ROWS_UNBOUNDED_FOLLOWING = \
    0, 1, 2
(FRAME_RANGE_SLIDING, FRAME_RANGE_UNBOUNDED_PRECEDING,
 FRAME_RANGE_UNBOUNDED_FOLLOWING) = 3, 4, 5
# Agg funcs (gx_agg_func)
(COUNT_ROW, COUNT_COL, SUM_I64, SUM_F64, MIN_I64, MAX_I64, MIN_F64,
 MAX_F64, AVG_F64, BIT_AND, BIT_OR, BIT_XOR, RANK, DENSE_RANK,
 FIRST_VALUE, LAST_VALUE, NTH_VALUE, LAG, LEAD, NTILE, CUME_DIST,
 PERCENT_RANK) = range(22)
# SQL SUM over integers, NULL-init (the Sum family AggregateUtils maps
# SqlKind.SUM to); SUM_I64 is SqlKind.SUM0 (Long2LongSum0, init 0)
SUM_I64N = 22

ORACLE_PATH = os.path.join(_REPO, "oracle", "libgxoracle.so")
HIP_PATH = os.path.join(_REPO, "galaxysql_amd", "csrc", "libgxhip.so")


def _np_ptr(arr):
    if arr is None:
        return None
    return arr.ctypes.data_as(C.c_void_p)


class GxLib:
    """One loaded gxop implementation."""

    def __init__(self, path):
        self.path = path
        self.lib = C.CDLL(path, mode=C.RTLD_LOCAL)
        L = self.lib
        L.gxop_join_create.restype = C.c_void_p
        L.gxop_join_create.argtypes = [C.POINTER(GxJoinCfg)]
        L.gxop_join_consume.argtypes = [C.c_void_p, C.POINTER(GxChunk)]
        L.gxop_join_build.argtypes = [C.c_void_p]
        L.gxop_join_probe.argtypes = [C.c_void_p, C.POINTER(GxChunk),
                                      C.POINTER(C.POINTER(GxResult))]
        L.gxop_join_tail.argtypes = [C.c_void_p, C.POINTER(C.POINTER(GxResult))]
        L.gxop_join_probe_push.argtypes = [C.c_void_p, C.POINTER(GxChunk)]
        L.gxop_join_probe_flush.argtypes = [
            C.c_void_p, C.POINTER(C.POINTER(GxResult))]
        L.gxop_join_close.argtypes = [C.c_void_p]
        L.gxop_agg_create.restype = C.c_void_p
        L.gxop_agg_create.argtypes = [C.POINTER(GxAggCfg)]
        L.gxop_agg_consume.argtypes = [C.c_void_p, C.POINTER(GxChunk)]
        L.gxop_agg_build.argtypes = [C.c_void_p]
        L.gxop_agg_next.argtypes = [C.c_void_p, C.POINTER(C.POINTER(GxResult))]
        L.gxop_agg_close.argtypes = [C.c_void_p]
        L.gxop_groupjoin_create.restype = C.c_void_p
        L.gxop_groupjoin_create.argtypes = [C.POINTER(GxGroupJoinCfg)]
        L.gxop_groupjoin_consume.argtypes = [C.c_void_p, C.POINTER(GxChunk)]
        L.gxop_groupjoin_build.argtypes = [C.c_void_p]
        L.gxop_groupjoin_probe.argtypes = [C.c_void_p, C.POINTER(GxChunk)]
        L.gxop_groupjoin_next.argtypes = [C.c_void_p,
                                          C.POINTER(C.POINTER(GxResult))]
        L.gxop_groupjoin_close.argtypes = [C.c_void_p]
        L.gxop_window_create.restype = C.c_void_p
        L.gxop_window_create.argtypes = [C.POINTER(GxWindowCfg)]
        L.gxop_window_consume.argtypes = [C.c_void_p, C.POINTER(GxChunk),
                                          C.POINTER(C.POINTER(GxResult))]
        L.gxop_window_close.argtypes = [C.c_void_p]
        L.gxop_fwindow_create.restype = C.c_void_p
        L.gxop_fwindow_create.argtypes = [C.POINTER(GxFWindowCfg)]
        L.gxop_fwindow_consume.argtypes = [C.c_void_p, C.POINTER(GxChunk)]
        L.gxop_fwindow_finish.argtypes = [C.c_void_p]
        L.gxop_fwindow_next.argtypes = [C.c_void_p,
                                        C.POINTER(C.POINTER(GxResult))]
        L.gxop_fwindow_close.argtypes = [C.c_void_p]
        L.gxop_part_create.restype = C.c_void_p
        L.gxop_part_create.argtypes = [C.POINTER(GxPartCfg)]
        L.gxop_part_consume.argtypes = [C.c_void_p, C.POINTER(GxChunk),
                                        C.POINTER(C.POINTER(GxResult))]
        L.gxop_part_consume_concat.argtypes = [C.c_void_p, C.POINTER(GxChunk),
                                               C.POINTER(C.POINTER(GxResult)),
                                               C.POINTER(C.c_int64)]
        L.gxop_part_close.argtypes = [C.c_void_p]
        L.gxop_scan_create.restype = C.c_void_p
        L.gxop_scan_create.argtypes = [C.POINTER(GxScanCfg)]
        L.gxop_scan_consume.argtypes = [C.c_void_p, C.POINTER(GxChunk),
                                        C.POINTER(C.POINTER(GxResult))]
        L.gxop_scan_close.argtypes = [C.c_void_p]
        L.gxop_result_copy_col.argtypes = [C.POINTER(GxResult), C.c_int32,
                                           C.c_void_p, C.c_void_p]
        L.gxop_join_get_stats.argtypes = [C.c_void_p, C.POINTER(GxJoinStats)]
        L.gxop_agg_get_stats.argtypes = [C.c_void_p, C.POINTER(GxAggStats)]
        L.gxop_result_to_host.argtypes = [C.POINTER(GxResult)]
        L.gxop_result_release.argtypes = [C.POINTER(GxResult)]
        L.gx_last_error.restype = C.c_char_p
        L.gxop_abi_version.restype = C.c_int

    def error(self):
        return (self.lib.gx_last_error() or b"").decode()

    def check(self, rc, what):
        if rc != 0:
            raise RuntimeError(f"{what} failed (rc={rc}): {self.error()}")

    # ---- chunk marshalling ----

    def to_gx_chunk(self, chunk: Chunk, keepalive, mem=0, device_ptrs=None):
        """Build a GxChunk over a host Chunk (mem=0) or over raw device
        pointers (mem=1, device_ptrs = list of dicts per block)."""
        n = len(chunk.blocks) if chunk is not None else len(device_ptrs)
        blocks = (GxBlock * n)()
        if chunk is not None:
            for i, b in enumerate(chunk.blocks):
                gb = blocks[i]
                gb.type = b.type
                gb.mem = mem
                gb.values = _np_ptr(b.values)
                gb.nulls = _np_ptr(b.nulls)
                gb.offsets = _np_ptr(b.offsets)
                gb.data = _np_ptr(b.data)
                gb.data_len = 0 if b.data is None else int(len(b.data))
            n_rows = chunk.n_rows
        else:
            for i, d in enumerate(device_ptrs):
                gb = blocks[i]
                gb.type = d["type"]
                gb.mem = 1
                gb.values = d.get("values")
                gb.nulls = d.get("nulls")
                gb.offsets = d.get("offsets")
                gb.data = d.get("data")
                gb.data_len = d.get("data_len", 0)
            n_rows = device_ptrs[0]["n_rows"]
        gc = GxChunk(n_rows=n_rows, n_blocks=n, blocks=blocks)
        keepalive.append(blocks)
        return gc

    def result_to_chunk(self, res_ptr) -> Chunk:
        """Copy a gx_result (host-resident) into numpy-backed Blocks and free it."""
        res = res_ptr.contents
        self.check(self.lib.gxop_result_to_host(res_ptr), "result_to_host")
        ch = res.chunk
        n = ch.n_rows
        blocks = []
        for i in range(ch.n_blocks):
            gb = ch.blocks[i]
            nulls = None
            if gb.nulls:
                nulls = np.ctypeslib.as_array(
                    C.cast(gb.nulls, C.POINTER(C.c_uint8)), shape=(n,)).copy()
            if gb.type == SLICE:
                offsets = np.ctypeslib.as_array(
                    C.cast(gb.offsets, C.POINTER(C.c_int32)), shape=(n,)).copy() \
                    if n else np.zeros(0, np.int32)
                dlen = int(gb.data_len)
                data = np.ctypeslib.as_array(
                    C.cast(gb.data, C.POINTER(C.c_uint8)), shape=(dlen,)).copy() \
                    if dlen else np.zeros(0, np.uint8)
                blocks.append(Block(SLICE, nulls=nulls, offsets=offsets, data=data))
            elif gb.type == DECIMAL:
                vals = np.ctypeslib.as_array(
                    C.cast(gb.values, C.POINTER(C.c_uint8)),
                    shape=(n, 40)).copy() \
                    if n else np.zeros((0, 40), np.uint8)
                blocks.append(Block(DECIMAL, values=vals, nulls=nulls))
            else:
                ctype = {I64: C.c_int64, I32: C.c_int32, F64: C.c_double}[gb.type]
                vals = np.ctypeslib.as_array(
                    C.cast(gb.values, C.POINTER(ctype)), shape=(n,)).copy() \
                    if n else np.zeros(0, _NP_DTYPES[gb.type])
                blocks.append(Block(gb.type, values=vals, nulls=nulls))
        self.lib.gxop_result_release(res_ptr)
        if not blocks:
            c = Chunk([])
            c.n_rows = n
            return c
        return Chunk(blocks)


_cached = {}


def load_oracle() -> GxLib:
    """TEST INFRASTRUCTURE ONLY — the CPU parity checker / cpu_baseline leg."""
    if "oracle" not in _cached:
        if not os.path.exists(ORACLE_PATH):
            raise RuntimeError(
                f"oracle library not built: {ORACLE_PATH} (run `make -C oracle`)")
        _cached["oracle"] = GxLib(ORACLE_PATH)
    return _cached["oracle"]


def load_hip() -> GxLib:
    """The product library. Raises if the HIP extension is missing — there is
    deliberately no CPU fallback (SURVEY.md tier rule: a product path routed
    through the oracle voids parity)."""
    if "hip" not in _cached:
        if not os.path.exists(HIP_PATH):
            raise RuntimeError(
                f"HIP extension not built: {HIP_PATH} "
                "(run python -m galaxysql_amd.build or __graft_entry__.build())")
        _cached["hip"] = GxLib(HIP_PATH)
    return _cached["hip"]
