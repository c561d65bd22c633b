"""Host-side operators mirroring the reference's Executor / ConsumerExecutor
lifecycle (operator/Executor.java, operator/ConsumerExecutor.java), driving
the gxop C-ABI.

Lifecycle exercised by Driver.processInternal (mpp/operator/Driver.java:
449-525): openConsume -> consumeChunk* -> buildConsume -> (operator flips to
producer) nextChunk* -> close. Names and argument meaning follow the
reference factories (mpp/operator/factory/ParallelHashJoinExecutorFactory
.java:77-117, HashAggExecutorFactory.java:71-104).
"""
from __future__ import annotations

import ctypes as C

from . import abi
from .abi import (GxJoinCfg, GxEquiKey, GxAggCfg, GxAggSpec, GxPartCfg, GxResult)
from .chunk import Chunk, CHUNK_SIZE


class EquiJoinKey:
    """Mirror of EquiJoinKey (polardbx-optimizer/.../core/join/EquiJoinKey
    .java:25-43). Key columns arrive unified to unified_type."""

    def __init__(self, outer_index, inner_index, unified_type):
        self.outer_index = outer_index
        self.inner_index = inner_index
        self.unified_type = unified_type


class JoinCond:
    """Residual (non-equi) condition term (gx_join_cond): the AND of terms
    is evaluated per matched candidate (AbstractJoinExec.checkJoinCondition
    :227-250). col_a/col_b index the condition row: leftSide cols then
    rightSide cols (JoinRelType.java:145-151). col_b = -1 compares against
    `value` (int/float/str/bytes, or None for SQL NULL)."""

    def __init__(self, col_a, cmp, col_b=-1, value=None):
        self.col_a = col_a
        self.cmp = cmp
        self.col_b = col_b
        self.value = value


def _conds_to_ctypes(conds, keep):
    arr = (abi.GxJoinCond * len(conds))()
    for i, c in enumerate(conds):
        t = abi.GxJoinCond(col_a=c.col_a, cmp=c.cmp, col_b=c.col_b)
        if c.col_b < 0:
            v = c.value
            if v is None:
                t.const_is_null = 1
            elif isinstance(v, float):
                t.v_f64 = v
            elif isinstance(v, (bytes, str)):
                raw = v.encode() if isinstance(v, str) else v
                buf = (C.c_uint8 * max(1, len(raw)))(*raw)
                keep.append(buf)
                t.v_bytes = buf
                t.v_len = len(raw)
            else:
                t.v_i64 = int(v)
        arr[i] = t
    keep.append(arr)
    return arr


class ParallelHashJoinExec:
    """One operator instance (the C side holds the shared build state).

    join_type: abi.INNER/LEFT/RIGHT/SEMI/ANTI; mirrors
    ParallelHashJoinExec.java:49 construction.
    """

    def __init__(self, lib, join_type, join_keys, outer_types, inner_types,
                 max_one_row=False, build_outer=False, anti_null_col=-1,
                 device=-1, stream=0, expected_build_rows=0, out_proj=None,
                 memory_budget_bytes=0, conds=None, enable_bloom=False):
        self._lib = lib
        self._keep = []
        keys = (GxEquiKey * len(join_keys))()
        for i, k in enumerate(join_keys):
            keys[i] = GxEquiKey(k.outer_index, k.inner_index, k.unified_type, 0)
        ot = (C.c_int32 * len(outer_types))(*outer_types)
        it = (C.c_int32 * len(inner_types))(*inner_types)
        op_arr = (C.c_int32 * max(1, len(out_proj or [])))(*(out_proj or [0]))
        cfg = GxJoinCfg(
            join_type=join_type, single_join=int(max_one_row),
            build_outer=int(build_outer), n_keys=len(join_keys), keys=keys,
            n_outer_cols=len(outer_types), outer_types=ot,
            n_inner_cols=len(inner_types), inner_types=it,
            anti_null_col=anti_null_col, device=device, stream=stream,
            expected_build_rows=expected_build_rows,
            n_out_proj=len(out_proj or []), out_proj=op_arr,
            memory_budget_bytes=memory_budget_bytes,
            enable_bloom=int(enable_bloom))
        if conds:
            cfg.n_conds = len(conds)
            cfg.conds = _conds_to_ctypes(conds, self._keep)
        self._keep += [keys, ot, it, op_arr, cfg]
        self._op = lib.lib.gxop_join_create(C.byref(cfg))
        if not self._op:
            raise RuntimeError(f"gxop_join_create: {lib.error()}")

    def consume_chunk(self, chunk: Chunk):
        ka = []
        gc = self._lib.to_gx_chunk(chunk, ka)
        self._lib.check(self._lib.lib.gxop_join_consume(self._op, C.byref(gc)),
                        "join_consume")

    def build_consume(self):
        self._lib.check(self._lib.lib.gxop_join_build(self._op), "join_build")

    def probe_chunk(self, chunk: Chunk) -> Chunk:
        ka = []
        gc = self._lib.to_gx_chunk(chunk, ka)
        out = C.POINTER(GxResult)()
        self._lib.check(self._lib.lib.gxop_join_probe(self._op, C.byref(gc),
                                                      C.byref(out)), "join_probe")
        if not out:
            return None
        return self._lib.result_to_chunk(out)

    def probe_push(self, chunk: Chunk):
        """Buffered probe (gxop.h: the LocalBufferExec pattern): stage a
        chunk device-side without launching kernels."""
        ka = []
        gc = self._lib.to_gx_chunk(chunk, ka)
        self._lib.check(self._lib.lib.gxop_join_probe_push(self._op,
                                                           C.byref(gc)),
                        "join_probe_push")

    def probe_flush(self) -> Chunk:
        """Probe the accumulated batch; None when nothing was buffered."""
        out = C.POINTER(GxResult)()
        self._lib.check(self._lib.lib.gxop_join_probe_flush(self._op,
                                                            C.byref(out)),
                        "join_probe_flush")
        if not out:
            return None
        return self._lib.result_to_chunk(out)

    def stats(self):
        """Cumulative probe-kernel stats (bench roofline leg)."""
        from .abi import GxJoinStats
        s = GxJoinStats()
        self._lib.check(self._lib.lib.gxop_join_get_stats(self._op, C.byref(s)),
                        "join_get_stats")
        return {"probe_kernel_ms": s.probe_kernel_ms,
                "probe_launches": s.probe_launches,
                "probe_rows": s.probe_rows, "matches": s.matches}

    def tail_chunks(self):
        chunks = []
        while True:
            out = C.POINTER(GxResult)()
            self._lib.check(self._lib.lib.gxop_join_tail(self._op, C.byref(out)),
                            "join_tail")
            if not out:
                break
            chunks.append(self._lib.result_to_chunk(out))
        return chunks

    def close(self):
        if self._op:
            self._lib.lib.gxop_join_close(self._op)
            self._op = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


class HashAggExec:
    def __init__(self, lib, group_cols, aggs, input_types,
                 expected_groups=0, device=-1, stream=0):
        """aggs: list of (abi.COUNT_ROW.., input_col_or_-1)."""
        self._lib = lib
        self._keep = []
        gc = (C.c_int32 * max(1, len(group_cols)))(*(group_cols or [0]))
        sp = (GxAggSpec * max(1, len(aggs)))()
        for i, (f, col) in enumerate(aggs):
            sp[i] = GxAggSpec(f, col)
        it = (C.c_int32 * len(input_types))(*input_types)
        cfg = GxAggCfg(n_group_cols=len(group_cols), group_cols=gc,
                       n_aggs=len(aggs), aggs=sp,
                       n_input_cols=len(input_types), input_types=it,
                       expected_groups=expected_groups, device=device,
                       stream=stream)
        self._keep += [gc, sp, it, cfg]
        self._op = lib.lib.gxop_agg_create(C.byref(cfg))
        if not self._op:
            raise RuntimeError(f"gxop_agg_create: {lib.error()}")

    def consume_chunk(self, chunk: Chunk):
        ka = []
        gch = self._lib.to_gx_chunk(chunk, ka)
        self._lib.check(self._lib.lib.gxop_agg_consume(self._op, C.byref(gch)),
                        "agg_consume")

    def stats(self):
        from .abi import GxAggStats
        s = GxAggStats()
        self._lib.check(self._lib.lib.gxop_agg_get_stats(self._op, C.byref(s)),
                        "agg_get_stats")
        return {"kernel_ms": s.kernel_ms, "consumes": s.consumes,
                "rows": s.rows, "groups": s.groups}

    def build_consume(self):
        self._lib.check(self._lib.lib.gxop_agg_build(self._op), "agg_build")

    def result_chunks(self):
        chunks = []
        while True:
            out = C.POINTER(GxResult)()
            self._lib.check(self._lib.lib.gxop_agg_next(self._op, C.byref(out)),
                            "agg_next")
            if not out:
                break
            chunks.append(self._lib.result_to_chunk(out))
        return chunks

    def close(self):
        if self._op:
            self._lib.lib.gxop_agg_close(self._op)
            self._op = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


class HashGroupJoinExec:
    """Fused join+agg (HashGroupJoinExec.java:186-447): consume the GROUP
    side, then feed probe chunks; one output row per (matched) consumed
    row: group_cols then agg values. join_keys: EquiJoinKey with
    outer_index = consumed-side col, inner_index = probe-side col."""

    def __init__(self, lib, join_type, join_keys, build_types, probe_types,
                 group_cols, aggs, device=-1, stream=0,
                 expected_build_rows=0):
        from .abi import GxGroupJoinCfg
        self._lib = lib
        self._keep = []
        keys = (GxEquiKey * len(join_keys))()
        for i, k in enumerate(join_keys):
            keys[i] = GxEquiKey(k.outer_index, k.inner_index, k.unified_type, 0)
        bt = (C.c_int32 * len(build_types))(*build_types)
        pt = (C.c_int32 * len(probe_types))(*probe_types)
        gc = (C.c_int32 * max(1, len(group_cols)))(*(group_cols or [0]))
        sp = (GxAggSpec * max(1, len(aggs)))()
        for i, (f, col) in enumerate(aggs):
            sp[i] = GxAggSpec(f, col)
        cfg = GxGroupJoinCfg(
            join_type=join_type, n_keys=len(join_keys), keys=keys,
            n_build_cols=len(build_types), build_types=bt,
            n_probe_cols=len(probe_types), probe_types=pt,
            n_group_cols=len(group_cols), group_cols=gc,
            n_aggs=len(aggs), aggs=sp, device=device, stream=stream,
            expected_build_rows=expected_build_rows)
        self._keep += [keys, bt, pt, gc, sp, cfg]
        self._op = lib.lib.gxop_groupjoin_create(C.byref(cfg))
        if not self._op:
            raise RuntimeError(f"gxop_groupjoin_create: {lib.error()}")

    def consume_chunk(self, chunk: Chunk):
        ka = []
        gc = self._lib.to_gx_chunk(chunk, ka)
        self._lib.check(
            self._lib.lib.gxop_groupjoin_consume(self._op, C.byref(gc)),
            "groupjoin_consume")

    def build_consume(self):
        self._lib.check(self._lib.lib.gxop_groupjoin_build(self._op),
                        "groupjoin_build")

    def probe_chunk(self, chunk: Chunk):
        ka = []
        gc = self._lib.to_gx_chunk(chunk, ka)
        self._lib.check(
            self._lib.lib.gxop_groupjoin_probe(self._op, C.byref(gc)),
            "groupjoin_probe")

    def result_chunks(self):
        chunks = []
        while True:
            out = C.POINTER(GxResult)()
            self._lib.check(
                self._lib.lib.gxop_groupjoin_next(self._op, C.byref(out)),
                "groupjoin_next")
            if not out:
                break
            chunks.append(self._lib.result_to_chunk(out))
        return chunks

    def close(self):
        if self._op:
            self._lib.lib.gxop_groupjoin_close(self._op)
            self._op = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


def run_groupjoin(lib, join_type, join_keys, build_chunks, probe_chunks,
                  build_types, probe_types, group_cols, aggs, **kw):
    op = HashGroupJoinExec(lib, join_type, join_keys, build_types,
                           probe_types, group_cols, aggs, **kw)
    try:
        for ch in build_chunks:
            op.consume_chunk(ch)
        op.build_consume()
        for ch in probe_chunks:
            op.probe_chunk(ch)
        return op.result_chunks()
    finally:
        op.close()


class NonFrameOverWindowExec:
    """Running window aggregates over partition-sorted input
    (NonFrameOverWindowExec.java:34-160). aggs: (abi.COUNT_ROW.., col);
    reset[a]=True = CURRENT ROW..CURRENT ROW mode. ROW_NUMBER() =
    (abi.COUNT_ROW, -1) cumulative; RANK/DENSE_RANK consult order_cols
    (null-safe run detection)."""

    def __init__(self, lib, part_cols, aggs, input_types, reset=None,
                 order_cols=None, device=-1, stream=0):
        from .abi import GxWindowCfg
        self._lib = lib
        self._keep = []
        pc = (C.c_int32 * max(1, len(part_cols)))(*(part_cols or [0]))
        oc_list = order_cols or []
        oc = (C.c_int32 * max(1, len(oc_list)))(*(oc_list or [0]))
        sp = (GxAggSpec * max(1, len(aggs)))()
        for i, (f, col) in enumerate(aggs):
            sp[i] = GxAggSpec(f, col)
        rs = (C.c_uint8 * max(1, len(aggs)))(
            *[1 if (reset and reset[i]) else 0 for i in range(len(aggs))])
        it = (C.c_int32 * len(input_types))(*input_types)
        cfg = GxWindowCfg(n_part_cols=len(part_cols), part_cols=pc,
                          n_order_cols=len(oc_list), order_cols=oc,
                          n_aggs=len(aggs), aggs=sp, reset=rs,
                          n_input_cols=len(input_types), input_types=it,
                          device=device, stream=stream)
        self._keep += [pc, oc, sp, rs, it, cfg]
        self._op = lib.lib.gxop_window_create(C.byref(cfg))
        if not self._op:
            raise RuntimeError(f"gxop_window_create: {lib.error()}")

    def consume_chunk(self, chunk: Chunk) -> Chunk:
        ka = []
        gc = self._lib.to_gx_chunk(chunk, ka)
        out = C.POINTER(GxResult)()
        self._lib.check(
            self._lib.lib.gxop_window_consume(self._op, C.byref(gc),
                                              C.byref(out)),
            "window_consume")
        return self._lib.result_to_chunk(out)

    def close(self):
        if self._op:
            self._lib.lib.gxop_window_close(self._op)
            self._op = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


def run_window(lib, part_cols, aggs, input_types, input_chunks, reset=None,
               order_cols=None, **kw):
    op = NonFrameOverWindowExec(lib, part_cols, aggs, input_types,
                                reset=reset, order_cols=order_cols, **kw)
    try:
        return [op.consume_chunk(ch) for ch in input_chunks]
    finally:
        op.close()


class OverWindowFramesExec:
    """Frame windows (OverWindowFramesExec.java + operator/frame/):
    consume* -> finish barrier -> result_chunks. frames: (func, input_col,
    kind, preceding, following); for NTH_VALUE/LAG/LEAD/NTILE `preceding`
    carries the parameter; CUME_DIST/PERCENT_RANK need order_cols."""

    def __init__(self, lib, part_cols, frames, input_types, device=-1,
                 order_cols=None, stream=0):
        from .abi import GxFrameSpec, GxFWindowCfg
        self._lib = lib
        self._keep = []
        pc = (C.c_int32 * max(1, len(part_cols)))(*(part_cols or [0]))
        oc_list = order_cols or []
        oc = (C.c_int32 * max(1, len(oc_list)))(*(oc_list or [0]))
        fr = (GxFrameSpec * len(frames))()
        for i, spec in enumerate(frames):
            # (func, input_col, kind[, preceding, following,
            #  order_col, order_asc]) — RANGE kinds need the last two
            spec = tuple(spec) + (0,) * (7 - len(spec))
            fr[i] = GxFrameSpec(*spec[:7])
        it = (C.c_int32 * len(input_types))(*input_types)
        cfg = GxFWindowCfg(n_part_cols=len(part_cols), part_cols=pc,
                           n_order_cols=len(oc_list), order_cols=oc,
                           n_frames=len(frames), frames=fr,
                           n_input_cols=len(input_types), input_types=it,
                           device=device, stream=stream)
        self._keep += [pc, oc, fr, it, cfg]
        self._op = lib.lib.gxop_fwindow_create(C.byref(cfg))
        if not self._op:
            raise RuntimeError(f"gxop_fwindow_create: {lib.error()}")

    def consume_chunk(self, chunk: Chunk):
        ka = []
        gc = self._lib.to_gx_chunk(chunk, ka)
        self._lib.check(
            self._lib.lib.gxop_fwindow_consume(self._op, C.byref(gc)),
            "fwindow_consume")

    def finish(self):
        self._lib.check(self._lib.lib.gxop_fwindow_finish(self._op),
                        "fwindow_finish")

    def result_chunks(self):
        chunks = []
        while True:
            out = C.POINTER(GxResult)()
            self._lib.check(
                self._lib.lib.gxop_fwindow_next(self._op, C.byref(out)),
                "fwindow_next")
            if not out:
                break
            chunks.append(self._lib.result_to_chunk(out))
        return chunks

    def close(self):
        if self._op:
            self._lib.lib.gxop_fwindow_close(self._op)
            self._op = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


def run_fwindow(lib, part_cols, frames, input_types, input_chunks,
                order_cols=None, **kw):
    op = OverWindowFramesExec(lib, part_cols, frames, input_types,
                              order_cols=order_cols, **kw)
    try:
        for ch in input_chunks:
            op.consume_chunk(ch)
        op.finish()
        return op.result_chunks()
    finally:
        op.close()


class PartitioningExchanger:
    """Mirrors mpp/operator/PartitioningExchanger.java:71-134."""

    def __init__(self, lib, n_parts, key_cols, input_types, device=-1, stream=0):
        self._lib = lib
        self._keep = []
        self.n_parts = n_parts
        kc = (C.c_int32 * len(key_cols))(*key_cols)
        it = (C.c_int32 * len(input_types))(*input_types)
        cfg = GxPartCfg(n_parts=n_parts, n_key_cols=len(key_cols), key_cols=kc,
                        n_input_cols=len(input_types), input_types=it,
                        device=device, stream=stream)
        self._keep += [kc, it, cfg]
        self._op = lib.lib.gxop_part_create(C.byref(cfg))
        if not self._op:
            raise RuntimeError(f"gxop_part_create: {lib.error()}")

    def consume_chunk(self, chunk: Chunk):
        """Returns list of n_parts chunks (None where empty)."""
        ka = []
        gc = self._lib.to_gx_chunk(chunk, ka)
        outs = (C.POINTER(GxResult) * self.n_parts)()
        self._lib.check(self._lib.lib.gxop_part_consume(self._op, C.byref(gc), outs),
                        "part_consume")
        res = []
        for p in range(self.n_parts):
            res.append(self._lib.result_to_chunk(outs[p]) if outs[p] else None)
        return res

    def close(self):
        if self._op:
            self._lib.lib.gxop_part_close(self._op)
            self._op = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


class ScanExec:
    """Vectorized filter + project (executor/vectorized/ stage,
    SURVEY.md §8f row 1). preds: (col, abi.LT.., const); projs:
    (abi.PROJ_COPY, a, -1) / (abi.PROJ_REV_F64, a, b) / ..."""

    def __init__(self, lib, preds, projs, input_types, device=-1, stream=0):
        from .abi import GxPred, GxProj, GxScanCfg
        self._lib = lib
        self._keep = []
        pa = (GxPred * max(1, len(preds)))()
        self._pats = []
        for i, (col, cmp, const) in enumerate(preds):
            if isinstance(const, (str, bytes)):
                pat = const.encode() if isinstance(const, str) else bytes(const)
                buf = C.create_string_buffer(pat, len(pat))
                self._pats.append(buf)
                pa[i] = GxPred(col, cmp, 0, 0.0,
                               C.cast(buf, C.c_void_p), len(pat))
            else:
                pa[i] = GxPred(col, cmp,
                               int(const) if not isinstance(const, float) else 0,
                               float(const), None, 0)
        pj = (GxProj * len(projs))()
        for i, spec in enumerate(projs):
            spec = tuple(spec) + (-1,) * (5 - len(spec))
            pj[i] = GxProj(*spec[:5])
        it = (C.c_int32 * len(input_types))(*input_types)
        cfg = GxScanCfg(n_preds=len(preds), preds=pa, n_projs=len(projs),
                        projs=pj, n_input_cols=len(input_types),
                        input_types=it, device=device, stream=stream)
        self._keep += [pa, pj, it, cfg]
        self._op = lib.lib.gxop_scan_create(C.byref(cfg))
        if not self._op:
            raise RuntimeError(f"gxop_scan_create: {lib.error()}")

    def consume_chunk(self, chunk: Chunk) -> Chunk:
        ka = []
        gc = self._lib.to_gx_chunk(chunk, ka)
        out = C.POINTER(GxResult)()
        self._lib.check(self._lib.lib.gxop_scan_consume(self._op, C.byref(gc),
                                                        C.byref(out)),
                        "scan_consume")
        return self._lib.result_to_chunk(out) if out else None

    def consume_raw(self, gx_chunk_ref):
        out = C.POINTER(GxResult)()
        self._lib.check(self._lib.lib.gxop_scan_consume(
            self._op, gx_chunk_ref, C.byref(out)), "scan_consume")
        return out

    def close(self):
        if self._op:
            self._lib.lib.gxop_scan_close(self._op)
            self._op = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


def run_join(lib, join_type, join_keys, build_chunks, probe_chunks,
             outer_types, inner_types, **kw):
    """Convenience: full consume -> build -> probe -> tail pass; returns the
    list of output Chunks (the shape SingleExecTest drives,
    polardbx-executor/src/test/.../SingleExecTest.java:72-110)."""
    op = ParallelHashJoinExec(lib, join_type, join_keys, outer_types,
                              inner_types, **kw)
    try:
        for ch in build_chunks:
            op.consume_chunk(ch)
        op.build_consume()
        out = []
        for ch in probe_chunks:
            r = op.probe_chunk(ch)
            if r is not None and r.n_rows > 0:
                out.append(r)
        out.extend(op.tail_chunks())
        return out
    finally:
        op.close()


def run_agg(lib, group_cols, aggs, input_types, input_chunks, **kw):
    op = HashAggExec(lib, group_cols, aggs, input_types, **kw)
    try:
        for ch in input_chunks:
            op.consume_chunk(ch)
        op.build_consume()
        return op.result_chunks()
    finally:
        op.close()
