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
