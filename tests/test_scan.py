"""Scan (vectorized filter+project) parity: oracle vs numpy on CPU,
HIP vs oracle on GPU (bit-exact for ints / copies, f64 exact — per-row ops
have no reduction order)."""
import os
import subprocess

import numpy as np
import pytest

from galaxysql_amd import abi
from galaxysql_amd.chunk import Chunk, Block, I64, I32, F64, SLICE, chunks_from_columns, multiset
from galaxysql_amd.operators import ScanExec

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.fixture(scope="module")
def oracle():
    subprocess.run(["make", "-C", os.path.join(REPO, "oracle")], check=True,
                   capture_output=True)
    return abi.load_oracle()


def make_inputs(rng, n):
    date = rng.integers(8000, 9500, n).astype(np.int32)
    qty = rng.integers(0, 100, n).astype(np.int64)
    price = (rng.integers(100, 10_000_00, n)).astype(np.int64)  # cents
    disc = rng.integers(0, 11, n).astype(np.int64)              # hundredths
    pricef = price.astype(np.float64) / 100.0
    discf = disc.astype(np.float64) / 100.0
    nulls_d = (rng.random(n) < 0.05).astype(np.uint8)
    return chunks_from_columns(
        [I32, I64, I64, I64, F64, F64],
        [(date, nulls_d), (qty, None), (price, None), (disc, None),
         (pricef, None), (discf, None)], chunk_size=997)


PREDS = [(0, abi.GE, 8300), (0, abi.LT, 9000), (1, abi.GT, 10)]
PROJS = [(abi.PROJ_COPY, 1, -1), (abi.PROJ_REV_F64, 4, 5),
         (abi.PROJ_REV_SCALED4, 2, 3), (abi.PROJ_COPY, 0, -1)]


def run_scan(lib, chunks, device):
    op = ScanExec(lib, PREDS, PROJS, [I32, I64, I64, I64, F64, F64],
                  device=device)
    try:
        rows = []
        for ch in chunks:
            out = op.consume_chunk(ch)
            if out is not None:
                rows.extend(out.rows())
        return rows
    finally:
        op.close()


def numpy_scan(chunks):
    rows = []
    for ch in chunks:
        n = ch.n_rows
        for i in range(n):
            d = ch.blocks[0].get(i)
            q = ch.blocks[1].get(i)
            if d is None or not (8300 <= d < 9000) or not q > 10:
                continue
            price, disc = ch.blocks[2].get(i), ch.blocks[3].get(i)
            pf, df = ch.blocks[4].get(i), ch.blocks[5].get(i)
            rows.append((q, pf * (1.0 - df), price * (100 - disc), d))
    return rows


def test_scan_oracle_vs_numpy(oracle):
    rng = np.random.default_rng(91)
    chunks = make_inputs(rng, 20000)
    got = run_scan(oracle, chunks, -1)
    exp = numpy_scan(chunks)
    assert multiset(got, f64_round=9) == multiset(exp, f64_round=9)


@pytest.mark.gpu
def test_scan_hip_vs_oracle():
    oracle = abi.load_oracle()
    hip = abi.load_hip()
    rng = np.random.default_rng(92)
    chunks = make_inputs(rng, 500_000)
    ref = run_scan(oracle, chunks, -1)
    got = run_scan(hip, chunks, 0)
    # per-row arithmetic: fully exact, including the doubles
    assert multiset(got) == multiset(ref)


# ---- LIKE '%pat%' on SLICE columns (the Q9/C5 p_name filter) ----

def make_string_inputs(rng, n):
    words = ["green", "blue", "lime", "forest", "salmon", "puff", "dim"]
    vals = []
    for _ in range(n):
        if rng.random() < 0.03:
            vals.append(None)
        else:
            k = rng.integers(1, 4)
            vals.append(" ".join(words[int(i)]
                                 for i in rng.integers(0, len(words), k)))
    keys = rng.integers(0, 1 << 40, n).astype(np.int64)
    return chunks_from_columns(
        [I64, SLICE],
        [(keys, None), Block.of(SLICE, vals)], chunk_size=911)


def run_like_scan(lib, chunks, device):
    op = ScanExec(lib, [(1, abi.CONTAINS, "green")],
                  [(abi.PROJ_COPY, 0, -1), (abi.PROJ_COPY, 1, -1)],
                  [I64, SLICE], device=device)
    try:
        rows = []
        for ch in chunks:
            out = op.consume_chunk(ch)
            if out is not None:
                rows.extend(out.rows())
        return rows
    finally:
        op.close()


def test_like_scan_oracle_vs_numpy(oracle):
    rng = np.random.default_rng(95)
    chunks = make_string_inputs(rng, 12000)
    got = run_like_scan(oracle, chunks, -1)
    exp = []
    for ch in chunks:
        for i in range(ch.n_rows):
            s = ch.blocks[1].get(i)
            if s is not None and b"green" in s:
                exp.append((ch.blocks[0].get(i), s))
    assert multiset(got) == multiset(exp)


@pytest.mark.gpu
def test_like_scan_hip_vs_oracle():
    oracle = abi.load_oracle()
    hip = abi.load_hip()
    rng = np.random.default_rng(96)
    chunks = make_string_inputs(rng, 300_000)
    ref = run_like_scan(oracle, chunks, -1)
    got = run_like_scan(hip, chunks, 0)
    assert multiset(got) == multiset(ref)


@pytest.mark.parametrize("seed", range(14))
def test_scan_oracle_vs_brute_random(seed):
    """Random predicate sets (i32/i64/f64 compares + SLICE CONTAINS) and
    random projection mixes vs a literal per-row restatement (SQL: NULL
    fails every predicate; AND of all predicates)."""
    import subprocess as sp
    sp.run(["make", "-C", os.path.join(REPO, "oracle")], check=True,
           capture_output=True)
    lib = abi.load_oracle()
    rng = np.random.default_rng(18000 + seed)
    n = int(rng.integers(1, 4000))
    dates = rng.integers(0, 40, n).astype(np.int32)
    dn = (rng.random(n) < 0.1).astype(np.uint8)
    qty = rng.integers(0, 30, n).astype(np.int64)
    f = np.round(rng.standard_normal(n) * 5, 3)
    fn = (rng.random(n) < 0.1).astype(np.uint8)
    tags = [None if rng.random() < 0.1 else
            ("" if rng.random() < 0.1 else
             "".join(rng.choice(list("abc"), rng.integers(1, 6))))
            for _ in range(n)]
    types = [I32, I64, F64, SLICE]
    chunks = chunks_from_columns(
        types, [(dates, dn if dn.any() else None), (qty, None),
                (f, fn if fn.any() else None), Block.of(SLICE, tags)],
        chunk_size=int(rng.integers(2, 1500)))

    cmps = [abi.LT, abi.LE, abi.GT, abi.GE, abi.EQ, abi.NE]
    preds = []
    n_preds = int(rng.integers(0, 4))
    for _ in range(n_preds):
        pick = rng.random()
        if pick < 0.35:
            preds.append((0, cmps[rng.integers(0, 6)], int(rng.integers(0, 40))))
        elif pick < 0.6:
            preds.append((1, cmps[rng.integers(0, 6)], int(rng.integers(0, 30))))
        elif pick < 0.8:
            preds.append((2, cmps[rng.integers(0, 6)],
                          float(np.round(rng.standard_normal() * 5, 2))))
        else:
            preds.append((3, abi.CONTAINS,
                          "".join(rng.choice(list("abc"),
                                             rng.integers(0, 3)))))
    projs = [(abi.PROJ_COPY, 1, -1), (abi.PROJ_COPY, 3, -1),
             (abi.PROJ_COPY, 2, -1)]

    op = ScanExec(lib, preds, projs, types, device=-1)
    got = []
    try:
        for ch in chunks:
            out = op.consume_chunk(ch)
            if out is not None:
                got.extend(out.rows())
    finally:
        op.close()

    def cmp_ok(v, c, const):
        if v is None:
            return False
        if c == abi.LT: return v < const
        if c == abi.LE: return v <= const
        if c == abi.GT: return v > const
        if c == abi.GE: return v >= const
        if c == abi.EQ: return v == const
        return v != const

    want = []
    for i in range(n):
        vals = (None if dn[i] else int(dates[i]), int(qty[i]),
                None if fn[i] else float(f[i]), tags[i])
        ok = True
        for col, c, const in preds:
            if c == abi.CONTAINS:
                ok = vals[3] is not None and const in vals[3]
            else:
                ok = cmp_ok(vals[col], c, const)
            if not ok:
                break
        if ok:
            t = None if vals[3] is None else vals[3].encode()
            want.append((vals[1], t, vals[2]))
    assert got == want, f"seed {seed}: {len(got)} vs {len(want)}"


def test_contains_empty_pattern_oracle():
    """LIKE '%%' (empty CONTAINS pattern) keeps every non-NULL string,
    including the empty string; NULL still fails (SQL LIKE on NULL)."""
    from galaxysql_amd.chunk import SLICE
    lib = abi.load_oracle()
    ch = Chunk([Block.of(I64, [1, 2, 3, 4]),
                Block.of(SLICE, ["abc", "", None, "bcd"])])
    sc = ScanExec(lib, [(1, abi.CONTAINS, "")], [(abi.PROJ_COPY, 0, -1)],
                  [I64, SLICE], device=-1)
    out = sc.consume_chunk(ch)
    sc.close()
    assert [r[0] for r in out.rows()] == [1, 2, 4]
