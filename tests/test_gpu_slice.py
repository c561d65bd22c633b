"""SLICE (string) columns on the device path: the reference's own golden
join vectors (which carry string keys AND payloads) run against the HIP
library, plus larger randomized slice-key/payload cases vs the oracle."""
import numpy as np
import pytest

from galaxysql_amd import abi
from galaxysql_amd.chunk import Block, Chunk, I64, I32, SLICE, multiset
from galaxysql_amd.operators import EquiJoinKey, run_join, run_agg, \
    PartitioningExchanger
from . import fixtures

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def libs():
    import subprocess, os
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    subprocess.run(["make", "-C", os.path.join(repo, "oracle")], check=True,
                   capture_output=True)
    return abi.load_oracle(), abi.load_hip()


@pytest.mark.parametrize("case", fixtures.load_cases("join_vectors.json"),
                         ids=lambda c: c["name"])
def test_golden_join_vectors_on_gpu(libs, case):
    """The transcribed HashJoinTest.java vectors, straight through the HIP
    path (string payloads, string keys, the full join-type matrix)."""
    _, hip = libs
    fixtures.check_join_case(hip, case, device=0)


def rand_strings(rng, n, null_frac=0.05, maxlen=12):
    vals = []
    for _ in range(n):
        if rng.random() < null_frac:
            vals.append(None)
        else:
            ln = int(rng.integers(0, maxlen))
            vals.append("".join(chr(97 + int(x))
                                for x in rng.integers(0, 26, ln)))
    return vals


def test_slice_key_join_random(libs):
    oracle, hip = libs
    rng = np.random.default_rng(71)
    keys_pool = rand_strings(rng, 200, null_frac=0.0, maxlen=40)
    bk = [keys_pool[i] for i in rng.integers(0, 200, 3000)]
    pk = [keys_pool[i] for i in rng.integers(0, 180, 9000)] + \
         rand_strings(rng, 500, null_frac=0.2)
    build = [Chunk([Block.of(SLICE, bk), Block.of(I64, list(range(len(bk))))])]
    probe = [Chunk([Block.of(SLICE, pk),
                    Block.of(I64, list(range(len(pk))))])]
    keys = [EquiJoinKey(0, 0, SLICE)]
    for jt in (abi.INNER, abi.LEFT, abi.SEMI, abi.ANTI):
        ref = run_join(oracle, jt, keys, build, probe, [SLICE, I64],
                       [SLICE, I64], device=-1)
        got = run_join(hip, jt, keys, build, probe, [SLICE, I64],
                       [SLICE, I64], device=0)
        r = [t for c in ref for t in c.rows()]
        g = [t for c in got for t in c.rows()]
        assert multiset(g) == multiset(r), f"join_type={jt}"


def test_slice_group_by(libs):
    oracle, hip = libs
    rng = np.random.default_rng(72)
    gk = rand_strings(rng, 5000, null_frac=0.05, maxlen=8)
    vals = rng.integers(-100, 100, 5000)
    chunks = [Chunk([Block.of(SLICE, gk),
                     Block.of(I64, [int(v) for v in vals])])]
    aggs = [(abi.SUM_I64, 1), (abi.COUNT_ROW, -1)]
    ref = run_agg(oracle, [0], aggs, [SLICE, I64], chunks, device=-1)
    got = run_agg(hip, [0], aggs, [SLICE, I64], chunks, device=0)
    r = [t for c in ref for t in c.rows()]
    g = [t for c in got for t in c.rows()]
    assert multiset(g) == multiset(r)


def test_slice_partition(libs):
    oracle, hip = libs
    rng = np.random.default_rng(73)
    sv = rand_strings(rng, 4000, null_frac=0.05)
    iv = [int(x) for x in rng.integers(0, 1 << 30, 4000)]
    ch = Chunk([Block.of(I64, iv), Block.of(SLICE, sv)])
    exo = PartitioningExchanger(oracle, 8, [0], [I64, SLICE], device=-1)
    exh = PartitioningExchanger(hip, 8, [0], [I64, SLICE], device=0)
    po, ph = exo.consume_chunk(ch), exh.consume_chunk(ch)
    for p in range(8):
        ro = po[p].rows() if po[p] else []
        rh = ph[p].rows() if ph[p] else []
        assert multiset(rh) == multiset(ro), f"partition {p}"
    exo.close()
    exh.close()
