"""Host-side mirror of the reference columnar batch format.

Chunk = list of Blocks (reference: polardbx-executor/.../chunk/Chunk.java:41-66);
Block physical layouts mirror LongBlock (long[] values + boolean[] valueIsNull,
chunk/LongBlock.java:41-55), IntegerBlock (chunk/IntegerBlock.java:38-73),
DoubleBlock, and SliceBlock (int[] end-offsets + byte data,
chunk/SliceBlock.java:40-57), re-expressed as numpy arrays so they map 1:1
onto the C-ABI `gx_block` (include/gxop.h).

CHUNK_SIZE default 1000 rows mirrors ConnectionParams.java:1088.
"""
from __future__ import annotations

import numpy as np

CHUNK_SIZE = 1000

I64, I32, F64, SLICE, DECIMAL = 0, 1, 2, 3, 4

_NP_DTYPES = {I64: np.int64, I32: np.int32, F64: np.float64}


class Block:
    """One column of a batch.

    values: numpy array (i64/i32/f64); (n, 40) uint8 for DECIMAL
    (DecimalBlock "simple" 40-B records, chunk/DecimalBlock.java /
    DecimalBox layout — see include/gxop.h); None for SLICE.
    nulls:  uint8 array (1 = NULL) or None.
    offsets/data: SLICE only — int32 END offsets per value + byte buffer.
    """

    __slots__ = ("type", "values", "nulls", "offsets", "data")

    def __init__(self, btype, values=None, nulls=None, offsets=None, data=None):
        self.type = btype
        self.values = values
        self.nulls = nulls
        self.offsets = offsets
        self.data = data

    @property
    def n_rows(self):
        if self.type == SLICE:
            return len(self.offsets)
        return len(self.values)

    @staticmethod
    def of(btype, pyvals):
        """Build from a python list with None for NULL (mirrors the reference
        test fixture helpers IntegerBlock.of / LongBlock.of /
        StringBlock.of)."""
        n = len(pyvals)
        nulls = np.array([1 if v is None else 0 for v in pyvals], dtype=np.uint8)
        has_null = bool(nulls.any())
        if btype == SLICE:
            data = bytearray()
            offsets = np.zeros(n, dtype=np.int32)
            for i, v in enumerate(pyvals):
                if v is not None:
                    data.extend(v.encode() if isinstance(v, str) else bytes(v))
                offsets[i] = len(data)
            return Block(SLICE, nulls=nulls if has_null else None,
                         offsets=offsets, data=np.frombuffer(bytes(data), dtype=np.uint8))
        if btype == DECIMAL:
            vals = np.zeros((n, 40), dtype=np.uint8)
            for i, v in enumerate(pyvals):
                if v is not None:
                    scaled, scale = v  # (scaled int, decimal scale)
                    vals[i] = np.frombuffer(dec40_encode(scaled, scale),
                                            dtype=np.uint8)
            return Block(DECIMAL, values=vals,
                         nulls=nulls if has_null else None)
        vals = np.array([0 if v is None else v for v in pyvals], dtype=_NP_DTYPES[btype])
        return Block(btype, values=vals, nulls=nulls if has_null else None)

    def get(self, i):
        if self.nulls is not None and self.nulls[i]:
            return None
        if self.type == SLICE:
            start = int(self.offsets[i - 1]) if i > 0 else 0
            return bytes(self.data[start:int(self.offsets[i])])
        if self.type == DECIMAL:
            return bytes(self.values[i])
        v = self.values[i]
        if self.type == F64:
            return float(v)
        return int(v)


class Chunk:
    """The reference's Chunk carries positionCount independently of its
    blocks (chunk/Chunk.java:55-89) — a block-less chunk can still have
    rows (COUNT(*)-only plans). n_rows overrides the derived count for
    that case."""
    __slots__ = ("blocks", "n_rows")

    def __init__(self, blocks, n_rows=None):
        self.blocks = list(blocks)
        self.n_rows = (n_rows if n_rows is not None
                       else self.blocks[0].n_rows if self.blocks else 0)
        for b in self.blocks:
            assert b.n_rows == self.n_rows, "ragged chunk"

    @property
    def types(self):
        return [b.type for b in self.blocks]

    def rows(self):
        """Materialize python row tuples (test/parity use only)."""
        return [tuple(b.get(i) for b in self.blocks) for i in range(self.n_rows)]


def rows_of(chunks):
    out = []
    for c in chunks:
        out.extend(c.rows())
    return out


def chunks_from_columns(btypes, columns, chunk_size=CHUNK_SIZE):
    """Split full-length numpy columns into CHUNK_SIZE-row Chunks.

    columns: list of (values, nulls_or_None) per column, or Block instances
    covering all rows.
    """
    blocks = []
    for t, col in zip(btypes, columns):
        if isinstance(col, Block):
            blocks.append(col)
        else:
            vals, nulls = col
            blocks.append(Block(t, values=np.asarray(vals, dtype=_NP_DTYPES[t]),
                                nulls=None if nulls is None else np.asarray(nulls, dtype=np.uint8)))
    n = blocks[0].n_rows
    out = []
    for start in range(0, n, chunk_size):
        end = min(start + chunk_size, n)
        sub = []
        for b in blocks:
            if b.type == SLICE:
                base = int(b.offsets[start - 1]) if start > 0 else 0
                off = (b.offsets[start:end] - base).astype(np.int32)
                dend = int(b.offsets[end - 1]) if end > 0 else 0
                sub.append(Block(SLICE,
                                 nulls=None if b.nulls is None else b.nulls[start:end],
                                 offsets=off, data=b.data[base:dend]))
            else:
                sub.append(Block(b.type, values=b.values[start:end],
                                 nulls=None if b.nulls is None else b.nulls[start:end]))
        out.append(Chunk(sub))
    return out


def dec40_encode(scaled: int, scale: int) -> bytes:
    """Python mirror of oracle scaled_to_dec40 (DecimalBox simple layout:
    base-1e9 words w0[,w1[,w2]], integers@36, fractions@37,
    derivedFractions@38, isNeg@39)."""
    import struct
    neg = scaled < 0
    a = -scaled if neg else scaled
    if a >= 10 ** 18:
        raise ValueError(
            "wide DECIMAL: >18 significant digits does not fit the "
            "DecimalBox simple layout / scaled-int64 fast path "
            "(reference falls back to full Decimal arithmetic, "
            "DecimalBox.java:43-71)")
    ip, rem = divmod(a, 10 ** scale)
    fr = rem * 10 ** (9 - scale)
    if ip >= 10 ** 9:
        w = (ip // 10 ** 9, ip % 10 ** 9, fr)
        integers = 18
    else:
        w = (ip, fr, 0)
        integers = 9
    p = bytearray(40)
    struct.pack_into("<iii", p, 0, *[int(x) for x in w])
    p[36] = integers
    p[37] = scale
    p[38] = scale
    p[39] = 1 if neg else 0
    return bytes(p)


def dec40_decode(p, scale: int) -> int:
    """Python mirror of oracle dec40_to_scaled."""
    import struct
    w0, w1, w2 = struct.unpack_from("<iii", bytes(p), 0)
    integers, isneg = p[36], p[39]
    if integers > 9:
        ip, fr = w0 * 10 ** 9 + w1, w2
    else:
        ip, fr = w0, w1
    if integers > 18 or ip >= 10 ** (18 - scale):
        raise ValueError(
            "wide DECIMAL: >18 significant digits does not fit the "
            "scaled-int64 fast path (DecimalBox.java:43-71)")
    v = ip * 10 ** scale + fr // 10 ** (9 - scale)
    return -v if isneg else v


def multiset(rows, f64_round=None, f64_sign_zero=False):
    """Canonical multiset of rows for order-insensitive parity compares
    (mirrors BaseExecTest.assertExecResultByRow, BaseExecTest.java:78-103).
    f64_round: decimal places to round floats (tolerance compare).
    NaN canonicalizes to a sentinel (all Java NaNs are equal); with
    f64_sign_zero, -0.0 and +0.0 are kept DISTINCT (Math.min ordering)."""
    import math
    from collections import Counter

    def canon(v):
        if isinstance(v, float):
            if v != v:
                return "NaN"
            if f64_sign_zero and v == 0.0 and math.copysign(1.0, v) < 0:
                return "-0.0"
            if f64_round is not None:
                return round(v, f64_round)
        return v

    return Counter(tuple(canon(v) for v in r) for r in rows)
