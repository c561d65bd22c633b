"""DECIMAL boundary format: the 40-byte DecimalBlock "simple" record.

Layout pinned against the reference's fast add paths, which read the words
at these exact offsets (optimizer/core/datatype/DecimalBox.java doAddToSum1
~:170-210 / doAddToSum2, and DecimalTypeBase word layout): 9 x int32
base-1e9 words at 0..35, integers@36, fractions@37, derivedFractions@38,
isNeg@39.  "Simple" 1-word: w0 = integer part, w1 = fraction digits x
10^(9-scale), integers = 9; 2-word: w0 = hi, w1 = lo (base 1e9), w2 = frac,
integers = 18.

Tests: (1) hand-computed byte patterns for the python/oracle encoders,
(2) oracle scan SCALED_TO_DEC -> bytes -> DEC_TO_SCALED round trip,
(3) GPU scan parity vs oracle (gpu-marked, in test_gpu_parity-style).
"""
import numpy as np
import pytest

from galaxysql_amd import abi
from galaxysql_amd.chunk import (Block, Chunk, I64, DECIMAL, dec40_encode,
                                 dec40_decode)
from galaxysql_amd.operators import ScanExec


def _w(p, i):
    return int.from_bytes(p[4 * i:4 * i + 4], "little", signed=True)


def test_dec40_hand_patterns():
    # 12.34 as scaled 1234, scale 2 -> w0=12, w1 = 34 * 10^(9-2) = 340000000
    p = dec40_encode(1234, 2)
    assert _w(p, 0) == 12 and _w(p, 1) == 340000000 and _w(p, 2) == 0
    assert p[36] == 9 and p[37] == 2 and p[38] == 2 and p[39] == 0
    assert dec40_decode(p, 2) == 1234

    # negative: -0.05 scale 2 -> w0=0, w1 = 5*10^7, isNeg=1
    p = dec40_encode(-5, 2)
    assert _w(p, 0) == 0 and _w(p, 1) == 50000000 and p[39] == 1
    assert dec40_decode(p, 2) == -5

    # 2-word: 1234567890.5 scale 1 (scaled 12345678905)
    p = dec40_encode(12345678905, 1)
    assert _w(p, 0) == 1 and _w(p, 1) == 234567890
    assert _w(p, 2) == 500000000 and p[36] == 18
    assert dec40_decode(p, 1) == 12345678905

    # integer-valued, scale 0
    p = dec40_encode(7, 0)
    assert _w(p, 0) == 7 and _w(p, 1) == 0 and p[36] == 9 and p[37] == 0
    assert dec40_decode(p, 0) == 7


SCALED = [1234, -5, 0, 99999999999, -12345678905, 100, 7, 2**53]
SCALE = 2


def _roundtrip(lib, device):
    vals = np.array(SCALED, dtype=np.int64)
    nulls = np.zeros(len(vals), dtype=np.uint8)
    nulls[3] = 1
    chunk = Chunk([Block(I64, values=vals, nulls=nulls)])

    to_dec = ScanExec(lib, preds=[],
                      projs=[(abi.PROJ_SCALED_TO_DEC, 0, -1, SCALE)],
                      input_types=[I64], device=device)
    dec_chunk = to_dec.consume_chunk(chunk)
    to_dec.close()
    assert dec_chunk.blocks[0].type == DECIMAL

    to_scaled = ScanExec(lib, preds=[],
                         projs=[(abi.PROJ_DEC_TO_SCALED, 0, -1, SCALE)],
                         input_types=[DECIMAL], device=device)
    back = to_scaled.consume_chunk(dec_chunk)
    to_scaled.close()
    return dec_chunk, back


def test_oracle_decimal_roundtrip():
    lib = abi.load_oracle()
    dec_chunk, back = _roundtrip(lib, device=-1)
    b = dec_chunk.blocks[0]
    for i, v in enumerate(SCALED):
        if i == 3:
            assert b.nulls is not None and b.nulls[i]
            continue
        assert bytes(b.values[i]) == dec40_encode(v, SCALE), f"row {i}"
    bb = back.blocks[0]
    for i, v in enumerate(SCALED):
        if i == 3:
            assert bb.nulls is not None and bb.nulls[i]
        else:
            assert int(bb.values[i]) == v


def _canon_dec(dec_chunk, back_chunk):
    """Order-insensitive canonical form (scan emit order is
    nondeterministic): sorted non-null decoded values + null count."""
    b = dec_chunk.blocks[0]
    n = dec_chunk.n_rows
    nulls = b.nulls if b.nulls is not None else np.zeros(n, np.uint8)
    decs = sorted(dec40_decode(b.values[i], SCALE)
                  for i in range(n) if not nulls[i])
    bb = back_chunk.blocks[0]
    bn = bb.nulls if bb.nulls is not None else np.zeros(n, np.uint8)
    backs = sorted(int(bb.values[i]) for i in range(n) if not bn[i])
    return decs, int(nulls.sum()), backs, int(bn.sum())


@pytest.mark.gpu
def test_gpu_decimal_roundtrip_matches_oracle():
    hip = abi.load_hip()
    ora = abi.load_oracle()
    hd, hb = _roundtrip(hip, device=0)
    od, ob = _roundtrip(ora, device=-1)
    assert _canon_dec(hd, hb) == _canon_dec(od, ob)


@pytest.mark.gpu
def test_gpu_decimal_copy_through_scan():
    """DECIMAL survives a filtered COPY projection (the gather/staging path)."""
    hip = abi.load_hip()
    keys = np.arange(100, dtype=np.int64)
    vals = np.zeros((100, 40), dtype=np.uint8)
    for k in range(100):
        vals[k] = np.frombuffer(dec40_encode(k * 7 - 50, 2), dtype=np.uint8)
    chunk = Chunk([Block(I64, values=keys),
                   Block(DECIMAL, values=vals)])
    sc = ScanExec(hip, preds=[(0, abi.LT, 40)],
                  projs=[(abi.PROJ_COPY, 0), (abi.PROJ_COPY, 1)],
                  input_types=[I64, DECIMAL], device=0)
    out = sc.consume_chunk(chunk)
    sc.close()
    assert out.n_rows == 40
    got = {int(out.blocks[0].values[i]): bytes(out.blocks[1].values[i])
           for i in range(out.n_rows)}
    for k in range(40):
        assert got[k] == dec40_encode(k * 7 - 50, 2)


# ---- wide-DECIMAL fence --------------------------------
# Values with >18 significant digits (or a 3rd integer word) exceed the
# DecimalBox "simple" layout the scaled-int64 fast path mirrors
# (DecimalBox.java:43-71); the reference falls back to full 9-limb Decimal
# arithmetic there. We must reject LOUDLY, not truncate silently.

def _wide_dec40(scale: int) -> bytes:
    """Raw 2-word record whose integer part has 18-scale+1 digits: decodes
    to >18 total significant digits at `scale`."""
    import struct
    ip = 10 ** (18 - scale)  # too big by exactly one digit
    p = bytearray(40)
    struct.pack_into("<iii", p, 0, ip // 10 ** 9, ip % 10 ** 9, 0)
    p[36] = 18
    p[37] = scale
    p[38] = scale
    return bytes(p)


def test_dec40_encode_rejects_wide():
    with pytest.raises(ValueError, match="wide DECIMAL"):
        dec40_encode(10 ** 18, 2)
    with pytest.raises(ValueError, match="wide DECIMAL"):
        dec40_decode(np.frombuffer(_wide_dec40(2), np.uint8), 2)
    # boundary value (exactly 18 digits) still fine
    assert dec40_decode(np.frombuffer(dec40_encode(10 ** 18 - 1, 2),
                                      np.uint8), 2) == 10 ** 18 - 1


def _wide_chunk():
    vals = np.zeros((3, 40), dtype=np.uint8)
    vals[0] = np.frombuffer(dec40_encode(1234, 2), np.uint8)
    vals[1] = np.frombuffer(_wide_dec40(2), np.uint8)
    vals[2] = np.frombuffer(dec40_encode(-5, 2), np.uint8)
    return Chunk([Block(DECIMAL, values=vals)])


def test_oracle_rejects_wide_decimal():
    lib = abi.load_oracle()
    sc = ScanExec(lib, preds=[],
                  projs=[(abi.PROJ_DEC_TO_SCALED, 0, -1, 2)],
                  input_types=[DECIMAL], device=-1)
    with pytest.raises(RuntimeError, match="wide DECIMAL"):
        sc.consume_chunk(_wide_chunk())
    sc.close()


@pytest.mark.gpu
def test_gpu_rejects_wide_decimal():
    lib = abi.load_hip()
    sc = ScanExec(lib, preds=[],
                  projs=[(abi.PROJ_DEC_TO_SCALED, 0, -1, 2)],
                  input_types=[DECIMAL], device=0)
    with pytest.raises(RuntimeError, match="wide DECIMAL"):
        sc.consume_chunk(_wide_chunk())
    sc.close()


@pytest.mark.parametrize("seed", range(10))
def test_decimal_roundtrip_fuzz(seed):
    """Random in-fence scaled values across scales 0..8: oracle
    SCALED->DEC40->SCALED identity, and the oracle's 40-byte records
    must be byte-identical to the python mirror's dec40_encode."""
    rng = np.random.default_rng(19000 + seed)
    lib = abi.load_oracle()
    scale = int(rng.integers(0, 9))
    n = int(rng.integers(1, 600))
    # keep within the 18-significant-digit "simple" fence
    hi = 10 ** 18 - 1
    vals = rng.integers(-hi, hi, n).astype(np.int64) \
        if rng.random() < 0.5 else rng.integers(-10**6, 10**6,
                                                n).astype(np.int64)
    nulls = (rng.random(n) < 0.1).astype(np.uint8)
    chunk = Chunk([Block(I64, values=vals,
                         nulls=nulls if nulls.any() else None)])

    to_dec = ScanExec(lib, preds=[],
                      projs=[(abi.PROJ_SCALED_TO_DEC, 0, -1, scale)],
                      input_types=[I64], device=-1)
    dec_chunk = to_dec.consume_chunk(chunk)
    to_dec.close()
    to_scaled = ScanExec(lib, preds=[],
                         projs=[(abi.PROJ_DEC_TO_SCALED, 0, -1, scale)],
                         input_types=[DECIMAL], device=-1)
    back = to_scaled.consume_chunk(dec_chunk)
    to_scaled.close()

    b, bb = dec_chunk.blocks[0], back.blocks[0]
    for i in range(n):
        if nulls[i]:
            assert b.nulls is not None and b.nulls[i], (seed, i)
            continue
        assert bytes(b.values[i]) == dec40_encode(int(vals[i]), scale), \
            (seed, i, int(vals[i]), scale)
        assert int(bb.values[i]) == int(vals[i]), (seed, i)
