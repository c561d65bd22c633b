"""PagesSerde-compatible chunk wire format.

Byte-exact restatement of the reference's remote-exchange serialization so
GPU operator results can enter the real CN's RemoteExchange buffers:

- frame: [i32 positionCount][u8 compression marker][i32 uncompressedSize]
  [i32 sizeInBytes][payload]  (PagesSerdeUtil.writeSerializedChunk /
  mpp/execution/buffer/PagesSerdeUtil.java:53-70); markers
  UNCOMPRESSED=0 / COMPRESSED=1 (buffer/ChunkCompression.java:24-25).
- payload ("raw page"): [i32 blockCount][block...]
  (PagesSerdeUtil.writeRawPage:36-42).
- LongBlock / IntegerBlock / DoubleBlock: [i32 n][null bits][value per
  NON-null position] (chunk/LongBlockEncoding.java:46-58, IntegerBlock~,
  DoubleBlockEncoding.java:47-55); null bits are MSB-first packed bytes,
  ceil(n/8), via EncoderUtil.encodeNullsAsBits (chunk/EncoderUtil.java:
  42-120).
- SliceBlock: [i32 charsetLen][charset][i32 collationLen][collation]
  [bool compatible][i32 n][null bits][bool existNonNull] then, if any
  non-null: [i32 end-offset x n (nulls add 0 length)][i32 dataLen][bytes
  of non-null values] (chunk/SliceBlockEncoding.java:48-70,
  SliceBlock.encoding:303-333).
- DecimalBlock: [i32 n][bool isSimple][i32 int1Pos][i32 int2Pos]
  [i32 fracPos][null bits][bool existNonNull] then, if any non-null:
  [i32 n*40][40-byte records x n, nulls included]
  (chunk/DecimalBlockEncoding.java:50-70, DecimalBlock.encoding:264-274).
  We always emit isSimple=false / UNSET(-1) positions — the reference
  accepts that for any decimal payload; simple-mode detection is a reader
  optimization, not a format change.

All integers little-endian (airlift Slice byte order). Compression is NOT
implemented: frames always carry marker 0, which every reference reader
accepts (PagesSerde.deserialize takes the UNCOMPRESSED branch,
PagesSerde.java:99-106); airlift LZ4 is an absent third-party dep
 and is a size optimization only.
"""
from __future__ import annotations

import struct

import numpy as np

from .chunk import Block, Chunk, I64, I32, F64, SLICE, DECIMAL

UNCOMPRESSED = 0
COMPRESSED = 1

_DEFAULT_CHARSET = b"UTF8MB4"
_DEFAULT_COLLATION = b"UTF8MB4_GENERAL_CI"


def _nulls_of(block):
    n = block.n_rows
    if block.nulls is None:
        return np.zeros(n, dtype=np.uint8)
    return np.asarray(block.nulls, dtype=np.uint8)


def _encode_null_bits(nulls):
    """EncoderUtil.encodeNullsAsBits: MSB-first bit packing, ceil(n/8)."""
    return np.packbits(nulls.astype(bool)).tobytes()


def _decode_null_bits(buf, pos, n):
    nb = (n + 7) // 8
    bits = np.unpackbits(np.frombuffer(buf, np.uint8, nb, pos))[:n]
    return bits.astype(np.uint8), pos + nb


def _write_fixed_block(out, block, dtype):
    n = block.n_rows
    out.append(struct.pack("<i", n))
    nulls = _nulls_of(block)
    out.append(_encode_null_bits(nulls))
    vals = np.ascontiguousarray(np.asarray(block.values, dtype=dtype))
    out.append(vals[nulls == 0].tobytes())


def _write_slice_block(out, block):
    n = block.n_rows
    out.append(struct.pack("<i", len(_DEFAULT_CHARSET)))
    out.append(_DEFAULT_CHARSET)
    out.append(struct.pack("<i", len(_DEFAULT_COLLATION)))
    out.append(_DEFAULT_COLLATION)
    out.append(b"\x01")  # isCompatible
    out.append(struct.pack("<i", n))
    nulls = _nulls_of(block)
    out.append(_encode_null_bits(nulls))
    exist = bool(n > int(nulls.sum()))
    out.append(b"\x01" if exist else b"\x00")
    if not exist:
        return
    offsets = np.asarray(block.offsets, dtype=np.int32)
    lens = np.diff(offsets, prepend=np.int32(0))
    lens = np.where(nulls == 1, 0, lens)  # nulls contribute zero length
    real = np.cumsum(lens, dtype=np.int32)
    out.append(real.tobytes())
    max_off = int(real[-1]) if n else 0
    if max_off > 0:
        out.append(struct.pack("<i", max_off))
        data = np.asarray(block.data, dtype=np.uint8)
        if block.nulls is None and max_off == len(data):
            out.append(data.tobytes())
        else:  # re-gather non-null extents
            parts = []
            for i in range(n):
                if nulls[i]:
                    continue
                b = int(offsets[i - 1]) if i > 0 else 0
                parts.append(data[b:int(offsets[i])].tobytes())
            out.append(b"".join(parts))


def _write_decimal_block(out, block):
    n = block.n_rows
    out.append(struct.pack("<i", n))
    out.append(b"\x00")                      # isSimple = false
    out.append(struct.pack("<iii", -1, -1, -1))  # UNSET word positions
    nulls = _nulls_of(block)
    out.append(_encode_null_bits(nulls))
    exist = bool(n > int(nulls.sum()))
    out.append(b"\x01" if exist else b"\x00")
    if exist:
        out.append(struct.pack("<i", n * 40))
        out.append(np.ascontiguousarray(
            np.asarray(block.values, dtype=np.uint8)).tobytes())


def serialize_chunk(chunk: Chunk) -> bytes:
    """One SerializedChunk frame (uncompressed), ready for the reference's
    readSerializedChunk."""
    out = [struct.pack("<i", len(chunk.blocks))]
    for b in chunk.blocks:
        if b.type == I64:
            _write_fixed_block(out, b, np.int64)
        elif b.type == I32:
            _write_fixed_block(out, b, np.int32)
        elif b.type == F64:
            _write_fixed_block(out, b, np.float64)
        elif b.type == SLICE:
            _write_slice_block(out, b)
        elif b.type == DECIMAL:
            _write_decimal_block(out, b)
        else:
            raise ValueError(f"unsupported block type {b.type}")
    payload = b"".join(out)
    frame = struct.pack("<ibii", chunk.n_rows, UNCOMPRESSED,
                        len(payload), len(payload))
    return frame + payload


def _read_fixed_block(buf, pos, dtype, btype):
    (n,) = struct.unpack_from("<i", buf, pos)
    pos += 4
    nulls, pos = _decode_null_bits(buf, pos, n)
    n_vals = int(n - nulls.sum())
    es = np.dtype(dtype).itemsize
    packed = np.frombuffer(buf, dtype, n_vals, pos)
    pos += n_vals * es
    if nulls.any():
        vals = np.zeros(n, dtype=dtype)
        vals[nulls == 0] = packed
        return Block(btype, values=vals, nulls=nulls), pos
    return Block(btype, values=packed.copy(), nulls=None), pos


def _read_slice_block(buf, pos):
    (clen,) = struct.unpack_from("<i", buf, pos); pos += 4 + clen
    (llen,) = struct.unpack_from("<i", buf, pos); pos += 4 + llen
    pos += 1  # isCompatible
    (n,) = struct.unpack_from("<i", buf, pos); pos += 4
    nulls, pos = _decode_null_bits(buf, pos, n)
    exist = buf[pos] != 0
    pos += 1
    if not exist:
        return Block(SLICE, nulls=nulls if nulls.any() else None,
                     offsets=np.zeros(n, np.int32),
                     data=np.zeros(0, np.uint8)), pos
    offsets = np.frombuffer(buf, np.int32, n, pos).copy()
    pos += 4 * n
    max_off = int(offsets[-1]) if n else 0
    data = np.zeros(0, np.uint8)
    if max_off > 0:
        (dlen,) = struct.unpack_from("<i", buf, pos)
        pos += 4
        data = np.frombuffer(buf, np.uint8, dlen, pos).copy()
        pos += dlen
    return Block(SLICE, nulls=nulls if nulls.any() else None,
                 offsets=offsets, data=data), pos


def _read_decimal_block(buf, pos):
    (n,) = struct.unpack_from("<i", buf, pos)
    pos += 4 + 1 + 12  # isSimple + int1Pos/int2Pos/fracPos
    nulls, pos = _decode_null_bits(buf, pos, n)
    exist = buf[pos] != 0
    pos += 1
    vals = np.zeros((n, 40), dtype=np.uint8)
    if exist:
        (blen,) = struct.unpack_from("<i", buf, pos)
        pos += 4
        got = np.frombuffer(buf, np.uint8, blen, pos).reshape(-1, 40)
        vals[:got.shape[0]] = got
        pos += blen
    return Block(DECIMAL, values=vals,
                 nulls=nulls if nulls.any() else None), pos


def deserialize_chunk(buf: bytes, types, pos: int = 0):
    """Read one frame; returns (Chunk, next_pos). `types` mirrors the
    reference's type-list-driven BlockEncodingBuilders.create."""
    n_rows, marker, _unc, size = struct.unpack_from("<ibii", buf, pos)
    pos += 13
    if marker != UNCOMPRESSED:
        raise ValueError("compressed frames not supported (see module doc)")
    end = pos + size
    (n_blocks,) = struct.unpack_from("<i", buf, pos)
    pos += 4
    if n_blocks != len(types):
        raise ValueError(
            f"frame declares {n_blocks} blocks but {len(types)} types given")
    blocks = []
    for t in types:
        if t == I64:
            b, pos = _read_fixed_block(buf, pos, np.int64, I64)
        elif t == I32:
            b, pos = _read_fixed_block(buf, pos, np.int32, I32)
        elif t == F64:
            b, pos = _read_fixed_block(buf, pos, np.float64, F64)
        elif t == SLICE:
            b, pos = _read_slice_block(buf, pos)
        elif t == DECIMAL:
            b, pos = _read_decimal_block(buf, pos)
        else:
            raise ValueError(f"unsupported block type {t}")
        blocks.append(b)
    if pos != end:
        raise ValueError(
            f"frame body misparsed: ended at {pos}, frame declares {end}")
    ch = Chunk(blocks) if blocks else Chunk([])
    ch.n_rows = n_rows
    return ch, pos


def serialize_chunks(chunks) -> bytes:
    """Frame stream (PagesSerdeUtil.writeSerializedChunks)."""
    return b"".join(serialize_chunk(c) for c in chunks)


def deserialize_chunks(buf: bytes, types):
    """Iterate every frame in a stream (readSerializedChunks)."""
    out = []
    pos = 0
    while pos < len(buf):
        c, pos = deserialize_chunk(buf, types, pos)
        out.append(c)
    return out
