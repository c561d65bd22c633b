"""Q3 operator-chain parity (scaled down):
 - CPU: the oracle chain vs an INDEPENDENT numpy recomputation of Q3
   (filter -> join -> groupby-sum), so the chain itself is pinned.
 - GPU: the HIP chain vs the oracle chain on the same seeded inputs —
   SUM(cents)/COUNT bit-exact, SUM(f64) at rel-tol 1e-9."""
import subprocess
import os

import numpy as np
import pytest
import torch

from galaxysql_amd import abi
from galaxysql_amd.chunk import multiset
from galaxysql_amd.queries import run_q3, gen_q3_numpy

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.fixture(scope="module")
def oracle():
    subprocess.run(["make", "-C", os.path.join(REPO, "oracle")], check=True,
                   capture_output=True)
    return abi.load_oracle()


def numpy_q3(cust, orders, lineitem):
    """Independent Q3 reference: plain numpy, no shared code with oracle."""
    cust_keys = cust[0]
    o_custkey, o_orderkey, o_date, o_prio = orders
    l_orderkey, revenue, cents = lineitem
    keep = np.isin(o_custkey, cust_keys)
    ok, od, op_ = o_orderkey[keep], o_date[keep], o_prio[keep]
    order_info = {int(k): (int(d), int(p)) for k, d, p in zip(ok, od, op_)}
    groups = {}
    for k, r, c in zip(l_orderkey, revenue, cents):
        info = order_info.get(int(k))
        if info is None:
            continue
        key = (int(k), info[0], info[1])
        s = groups.setdefault(key, [0.0, 0, 0])
        s[0] += float(r)
        s[1] += int(c)
        s[2] += 1
    return [(k[0], k[1], k[2], v[0], v[1], v[2]) for k, v in groups.items()]


def chain_rows(lib, device, data):
    cust, orders, lineitem = data
    t = [[torch.from_numpy(a) for a in cols] for cols in data]
    if device >= 0:
        t = [[x.cuda(device) for x in cols] for cols in t]
    chunks, info = run_q3(lib, device, t[0], t[1], t[2], to_host=True)
    rows = []
    for c in chunks:
        rows.extend(c.rows())
    return rows, info


def test_q3_oracle_vs_numpy(oracle):
    rng = np.random.default_rng(77)
    data = gen_q3_numpy(rng, n_cust_total=2000, n_orders_total=20000,
                        n_lineitem=80000)
    got, info = chain_rows(oracle, -1, data)
    exp = numpy_q3(*data)
    assert info["groups"] == len(exp)
    assert multiset(got, f64_round=6) == multiset(exp, f64_round=6)


@pytest.mark.gpu
def test_q3_hip_vs_oracle():
    oracle = abi.load_oracle()
    hip = abi.load_hip()
    rng = np.random.default_rng(78)
    data = gen_q3_numpy(rng, n_cust_total=20000, n_orders_total=300000,
                        n_lineitem=1_200_000)
    ref, ri = chain_rows(oracle, -1, data)
    got, gi = chain_rows(hip, 0, data)
    assert gi["orders_kept"] == ri["orders_kept"]
    assert gi["joined_rows"] == ri["joined_rows"]
    assert gi["groups"] == ri["groups"]
    # bit-exact on i64 group keys + SUM(cents) + COUNT; f64 rounded
    strip_f64 = lambda rows: [(a, b, c, e, f) for a, b, c, d, e, f in rows]
    assert multiset(strip_f64(got)) == multiset(strip_f64(ref))
    assert multiset(got, f64_round=4) == multiset(ref, f64_round=4)


# ---- honest Q3 (device-side filter+project) ----

def numpy_q3_honest(cust, orders, lineitem):
    from galaxysql_amd.queries import Q3_DATE_CUTOFF, Q3_SHIP_CUTOFF, Q3_SEGMENT
    ck, cseg = cust
    keep_c = set(int(k) for k, s in zip(ck, cseg) if s == Q3_SEGMENT)
    ocust, okey, odate, oprio = orders
    order_info = {}
    for c, k, d, p in zip(ocust, okey, odate, oprio):
        if d < Q3_DATE_CUTOFF and int(c) in keep_c:
            order_info[int(k)] = (int(d), int(p))
    lkey, ship, pf, df, cents, disc = lineitem
    groups = {}
    for k, sdt, a, b, cc, dd in zip(lkey, ship, pf, df, cents, disc):
        if not sdt > Q3_SHIP_CUTOFF:
            continue
        info = order_info.get(int(k))
        if info is None:
            continue
        key = (int(k), info[0], info[1])
        s = groups.setdefault(key, [0.0, 0, 0])
        s[0] += float(a) * (1.0 - float(b))
        s[1] += int(cc) * (100 - int(dd))
        s[2] += 1
    return [(k[0], k[1], k[2], v[0], v[1], v[2]) for k, v in groups.items()]


def honest_rows(lib, device, data):
    from galaxysql_amd.queries import run_q3_honest
    t = [[torch.from_numpy(a) for a in cols] for cols in data]
    if device >= 0:
        t = [[x.cuda(device) for x in cols] for cols in t]
    chunks, info = run_q3_honest(lib, device, t[0], t[1], t[2], to_host=True)
    rows = []
    for c in chunks:
        rows.extend(c.rows())
    return rows, info


def test_q3_honest_oracle_vs_numpy(oracle):
    from galaxysql_amd.queries import gen_q3_raw_numpy
    rng = np.random.default_rng(81)
    data = gen_q3_raw_numpy(rng, n_cust=3000, n_orders=25000, n_lineitem=100000)
    got, info = honest_rows(oracle, -1, data)
    exp = numpy_q3_honest(*data)
    assert info["groups"] == len(exp)
    assert multiset(got, f64_round=6) == multiset(exp, f64_round=6)


@pytest.mark.gpu
def test_q3_honest_hip_vs_oracle():
    from galaxysql_amd.queries import gen_q3_raw_numpy
    oracle = abi.load_oracle()
    hip = abi.load_hip()
    rng = np.random.default_rng(82)
    data = gen_q3_raw_numpy(rng, n_cust=30000, n_orders=400000,
                            n_lineitem=1_500_000)
    ref, ri = honest_rows(oracle, -1, data)
    got, gi = honest_rows(hip, 0, data)
    for k in ("cust_kept", "orders_kept_scan", "lineitem_kept",
              "orders_kept", "joined_rows", "groups"):
        assert gi[k] == ri[k], k
    strip_f64 = lambda rows: [(a, b, c, e, f) for a, b, c, d, e, f in rows]
    assert multiset(strip_f64(got)) == multiset(strip_f64(ref))
    assert multiset(got, f64_round=4) == multiset(ref, f64_round=4)
