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

All integers little-endian (airlift Slice byte order). Compression:
standard LZ4 block format, as airlift's Lz4Compressor/Lz4Decompressor
(PagesSerde.java:66-118). The reader accepts reference-compressed frames
(format-exact decompressor); the writer applies the reference's own
keep-if-saves->=20% policy with a greedy spec-conformant compressor whose
output any LZ4 decoder accepts — not byte-identical to airlift's match
choices, so frame BYTE-identity holds for uncompressed frames (and
value-identity always).
"""
from __future__ import annotations

import struct

import numpy as np

from .chunk import Block, Chunk, I64, I32, F64, SLICE, DECIMAL

UNCOMPRESSED = 0
COMPRESSED = 1  # ChunkCompression.java:24-25


# ---- LZ4 block codec (standard block format; what airlift's
# Lz4Compressor/Lz4Decompressor move through PagesSerde.java:66-118).
# _lz4_compress MIRRORS gx_serde.inc's greedy matcher EXACTLY (same hash,
# same single-probe table, same traversal) so python and native frames
# stay byte-identical; _lz4_decompress is format-exact and accepts
# reference-compressed frames.

def _lz4_decompress(src, unc):
    d = bytearray(unc)
    si, di, sl = 0, 0, len(src)
    while si < sl:
        tok = src[si]
        si += 1
        lit = tok >> 4
        if lit == 15:
            while True:
                b = src[si]
                si += 1
                lit += b
                if b != 255:
                    break
        if si + lit > sl or di + lit > unc:
            raise ValueError("corrupt LZ4 block (literals)")
        d[di:di + lit] = src[si:si + lit]
        si += lit
        di += lit
        if si >= sl:
            break
        off = src[si] | (src[si + 1] << 8)
        si += 2
        if off == 0 or off > di:
            raise ValueError("corrupt LZ4 block (offset)")
        ml = tok & 15
        if ml == 15:
            while True:
                b = src[si]
                si += 1
                ml += b
                if b != 255:
                    break
        ml += 4
        if di + ml > unc:
            raise ValueError("corrupt LZ4 block (match)")
        for _ in range(ml):
            d[di] = d[di - off]
            di += 1
    if di != unc:
        raise ValueError("LZ4 block inflated to %d, expected %d" % (di, unc))
    return bytes(d)


_LZ4_HBITS = 13


def _lz4_compress(s, cap):
    # greedy matcher identical to gx_serde.inc lz4_compress; returns
    # compressed bytes or None when they would not fit cap
    import struct as _st
    sl = len(s)
    out = bytearray()

    def put_lit_run(frm, to, ml4):
        lit = to - frm
        tok = (15 if lit >= 15 else lit) << 4
        tok |= 15 if ml4 >= 15 else ml4
        out.append(tok)
        if lit >= 15:
            v = lit - 15
            while v >= 255:
                out.append(255)
                v -= 255
            out.append(v)
        out.extend(s[frm:to])

    if sl >= 13:
        tab = [-1] * (1 << _LZ4_HBITS)
        si = anchor = 0
        mflimit = sl - 12
        matchlim = sl - 5
        while si < mflimit:
            (v,) = _st.unpack_from("<I", s, si)
            h = ((v * 2654435761) & 0xFFFFFFFF) >> (32 - _LZ4_HBITS)
            cand = tab[h]
            tab[h] = si
            if cand >= 0 and si - cand <= 65535 and \
                    s[cand:cand + 4] == s[si:si + 4]:
                ml = 4
                while si + ml < matchlim and s[cand + ml] == s[si + ml]:
                    ml += 1
                put_lit_run(anchor, si, ml - 4)
                off = si - cand
                out.append(off & 0xFF)
                out.append(off >> 8)
                if ml - 4 >= 15:
                    v2 = ml - 4 - 15
                    while v2 >= 255:
                        out.append(255)
                        v2 -= 255
                    out.append(v2)
                si += ml
                anchor = si
                if len(out) > cap:
                    return None
            else:
                si += 1
        put_lit_run(anchor, sl, 0)
    else:
        put_lit_run(0, sl, 0)
    return bytes(out) if len(out) <= cap else None

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
    # keep the LZ4 payload only when it saves >= 20%
    # (PagesSerde.serializeForce:70-90, MINIMUM_COMPRESSION_RATIO 0.8)
    comp = _lz4_compress(payload, len(payload) - 1) \
        if len(payload) >= 32 else None
    if comp is not None and len(comp) / len(payload) <= 0.8:
        frame = struct.pack("<ibii", chunk.n_rows, COMPRESSED,
                            len(payload), len(comp))
        return frame + comp
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
    if marker == COMPRESSED:
        raw = _lz4_decompress(bytes(buf[pos:pos + size]), _unc)
        inner, _ = deserialize_chunk(
            struct.pack("<ibii", n_rows, UNCOMPRESSED, _unc, _unc) + raw,
            types, 0)
        return inner, pos + size
    if marker != UNCOMPRESSED:
        raise ValueError("unknown compression marker %d" % marker)
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
