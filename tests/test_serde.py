"""PagesSerde wire format (galaxysql_amd/serde.py vs the reference's
PagesSerdeUtil/BlockEncoding byte layout).

Golden byte patterns are hand-derived from the encoding code cited in the
module doc (LongBlockEncoding.java:46-58, EncoderUtil.java:42-120,
PagesSerdeUtil.java:53-70) — every int little-endian, null bits MSB-first,
non-null values only."""
import struct

import numpy as np

from galaxysql_amd.chunk import (Block, Chunk, I64, I32, F64, SLICE, DECIMAL,
                                 dec40_encode, multiset, rows_of)
from galaxysql_amd.serde import (serialize_chunk, deserialize_chunk,
                                 serialize_chunks, deserialize_chunks)


def test_long_block_golden_bytes():
    # 3 rows [7, NULL, -1]: nullbits byte = 0b01000000 = 0x40
    c = Chunk([Block.of(I64, [7, None, -1])])
    buf = serialize_chunk(c)
    payload = (struct.pack("<i", 1)                 # block count
               + struct.pack("<i", 3)               # positionCount
               + b"\x40"                            # null bits
               + struct.pack("<q", 7)               # non-null values only
               + struct.pack("<q", -1))
    frame = struct.pack("<ibii", 3, 0, len(payload), len(payload))
    assert buf == frame + payload


def test_slice_block_golden_bytes():
    # 2 rows ["ab", NULL]: real end-offsets [2, 2]; nulls 0b0100.. = 0x40
    c = Chunk([Block.of(SLICE, ["ab", None])])
    buf = serialize_chunk(c)
    cs, co = b"UTF8MB4", b"UTF8MB4_GENERAL_CI"
    payload = (struct.pack("<i", 1)
               + struct.pack("<i", len(cs)) + cs
               + struct.pack("<i", len(co)) + co
               + b"\x01"                            # isCompatible
               + struct.pack("<i", 2)               # positionCount
               + b"\x40"                            # null bits
               + b"\x01"                            # existNonNull
               + struct.pack("<ii", 2, 2)           # end offsets
               + struct.pack("<i", 2) + b"ab")      # dataLen + bytes
    frame = struct.pack("<ibii", 2, 0, len(payload), len(payload))
    assert buf == frame + payload


def test_roundtrip_all_types():
    rng = np.random.default_rng(5)
    n = 1000
    nulls = (rng.random(n) < 0.1).astype(np.uint8)
    blocks = [
        Block(I64, values=rng.integers(-2**62, 2**62, n), nulls=nulls.copy()),
        Block(I32, values=rng.integers(-2**31, 2**31, n).astype(np.int32),
              nulls=None),
        Block(F64, values=rng.random(n), nulls=nulls.copy()),
        Block.of(SLICE, [None if nulls[i] else f"v{i}"[:1 + i % 7]
                         for i in range(n)]),
        Block.of(DECIMAL, [None if nulls[i] else (int(i) * 97 - 5000, 2)
                           for i in range(n)]),
    ]
    for b in blocks:
        if b.values is not None:
            b.values = np.asarray(b.values)
    c = Chunk(blocks)
    got, pos = deserialize_chunk(serialize_chunk(c),
                                 [I64, I32, F64, SLICE, DECIMAL])
    assert pos == len(serialize_chunk(c))
    assert got.n_rows == n
    assert rows_of([got]) == rows_of([c])


def test_roundtrip_edge_cases():
    # empty chunk, all-null slice, all-null decimal
    c = Chunk([Block.of(I64, []), Block.of(SLICE, []), Block.of(DECIMAL, [])])
    got, _ = deserialize_chunk(serialize_chunk(c), [I64, SLICE, DECIMAL])
    assert got.n_rows == 0

    c2 = Chunk([Block.of(SLICE, [None, None, None]),
                Block.of(DECIMAL, [None, None, None])])
    got2, _ = deserialize_chunk(serialize_chunk(c2), [SLICE, DECIMAL])
    assert rows_of([got2]) == [(None, None)] * 3


def test_frame_stream():
    cs = [Chunk([Block.of(I64, list(range(k, k + 5)))]) for k in range(3)]
    stream = serialize_chunks(cs)
    back = deserialize_chunks(stream, [I64])
    assert len(back) == 3
    assert rows_of(back) == rows_of(cs)


def test_operator_results_survive_the_wire():
    """GPU/oracle operator output -> wire -> back == original (the remote
    exchange hand-off shape)."""
    from galaxysql_amd import abi
    from galaxysql_amd.operators import EquiJoinKey, run_join
    from galaxysql_amd.chunk import chunks_from_columns
    lib = abi.load_oracle()
    rng = np.random.default_rng(6)
    bk = rng.integers(0, 500, 1000)
    pk = rng.integers(0, 800, 3000)
    build = chunks_from_columns([I64], [(bk, None)])
    probe = chunks_from_columns([I64, I64],
                                [(pk, None), (rng.integers(0, 9, 3000), None)])
    out = run_join(lib, abi.INNER, [EquiJoinKey(0, 0, I64)], build, probe,
                   [I64, I64], [I64], device=-1)
    types = [I64, I64, I64]
    back = deserialize_chunks(serialize_chunks(out), types)
    assert multiset(rows_of(back)) == multiset(rows_of(out))


def test_serde_random_roundtrip_fuzz():
    """Random chunks across all types/null densities: python and native
    writers agree byte-for-byte and roundtrip losslessly."""
    import ctypes as C
    from galaxysql_amd import abi
    from galaxysql_amd.chunk import Block, Chunk, DECIMAL, dec40_encode
    from galaxysql_amd.serde import deserialize_chunk
    lib = abi.load_oracle()
    L = lib.lib
    L.gxop_chunk_serialize.restype = C.c_int
    L.gxop_chunk_serialize.argtypes = [C.c_void_p,
                                       C.POINTER(C.POINTER(C.c_uint8)),
                                       C.POINTER(C.c_int64)]
    for seed in range(6):
        rng = np.random.default_rng(500 + seed)
        n = int(rng.integers(0, 700))
        types = [int(rng.choice([I64, I32, F64, SLICE, DECIMAL]))
                 for _ in range(int(rng.integers(1, 5)))]
        nf = float(rng.choice([0.0, 0.3, 1.0]))
        blocks = []
        for t in types:
            nulls = rng.random(n) < nf
            if t == SLICE:
                blocks.append(Block.of(SLICE, [
                    None if nulls[i] else "x" * int(rng.integers(0, 9))
                    for i in range(n)]))
            elif t == DECIMAL:
                blocks.append(Block.of(DECIMAL, [
                    None if nulls[i] else (int(rng.integers(-10**12, 10**12)), 4)
                    for i in range(n)]))
            else:
                dt = {I64: np.int64, I32: np.int32, F64: np.float64}[t]
                v = rng.integers(-1000, 1000, n).astype(dt)
                blocks.append(Block(t, values=v,
                                    nulls=nulls.astype(np.uint8)
                                    if nulls.any() else None))
        c = Chunk(blocks)
        buf = serialize_chunk(c)
        # native == python
        ka = []
        gc = lib.to_gx_chunk(c, ka)
        out = C.POINTER(C.c_uint8)()
        blen = C.c_int64()
        assert L.gxop_chunk_serialize(C.byref(gc), C.byref(out),
                                      C.byref(blen)) == 0
        nbuf = bytes(C.cast(out, C.POINTER(C.c_uint8 * blen.value)).contents)
        L.gxop_buf_free(out)
        assert nbuf == buf, f"seed {seed}"
        back, pos = deserialize_chunk(buf, types)
        assert pos == len(buf)
        assert rows_of([back]) == rows_of([c]), f"seed {seed}"
