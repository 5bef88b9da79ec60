"""GF(2^16) (jerasure reed_sol_van w=16) tests: gf-complete's 0x1100B
field, the same big-Vandermonde construction, u16 LE symbols
(galois_w16_region_multiply semantics, ErasureCodeJerasure.cc:316-319)."""
from itertools import combinations

import numpy as np
import pytest

import ceph_amd
import oracle


def test_gf16_field_kats():
    # alpha^16 reduces by 0x1100B -> 0x100B
    assert oracle.gf16_mul(2, 1 << 15) == 0x100B
    assert oracle.gf16_mul(3, 7) == 9
    assert oracle.gf16_mul(0, 0x1234) == 0


def test_matrix_structure_w16():
    k, m = 6, 3
    g = oracle.matrix_w16(k, m)
    assert (g[:k] == np.eye(k, dtype=np.uint16)).all()
    assert (g[k] == 1).all()  # parity0 == XOR, as in w=8


def test_round_trip_exhaustive_w16():
    k, m = 4, 3
    L = 1024
    rng = np.random.default_rng(16)
    data = [rng.integers(0, 256, L, dtype=np.uint8) for _ in range(k)]
    par = oracle.encode_w16(k, m, data)
    x = np.zeros(L, np.uint8)
    for d in data:
        x ^= d
    assert (par[0] == x).all()
    full = data + par
    for e in range(1, m + 1):
        for er in combinations(range(k + m), e):
            pres = np.ones(k + m, np.uint8)
            pres[list(er)] = 0
            test = [c.copy() if pres[i] else np.zeros(L, np.uint8)
                    for i, c in enumerate(full)]
            oracle.decode_w16(k, m, test, pres)
            for i in range(k + m):
                assert (test[i] == full[i]).all(), (er, i)


def test_w16_differs_from_w8():
    """Same inputs, different field: the parity bytes must differ (a
    mislabeled w would silently produce w=8 output)."""
    k, m = 4, 2
    L = 512
    rng = np.random.default_rng(7)
    data = [rng.integers(0, 256, L, dtype=np.uint8) for _ in range(k)]
    p16 = oracle.encode_w16(k, m, data)
    p8 = oracle.encode("jerasure_reed_sol_van", k, m, data)
    assert (p16[0] == p8[0]).all()          # XOR row identical
    assert not (p16[1] == p8[1]).all()      # GF rows differ


@pytest.mark.gpu
@pytest.mark.parametrize("k,m", [(4, 2), (8, 3), (6, 4)])
def test_gpu_w16_parity_vs_oracle(k, m):
    C = 64 * 1024
    rng = np.random.default_rng(0x16 ^ k)
    ctx = ceph_amd.EcContext(k, m, "jerasure_reed_sol_van_w16", device=0)
    try:
        data = [rng.integers(0, 256, C, dtype=np.uint8) for _ in range(k)]
        got = ctx.encode_chunks(data)
        want = oracle.encode_w16(k, m, data)
        for j in range(m):
            assert np.array_equal(got[j], want[j]), j
        # decode exhaustive e<=2
        full = data + got
        for e in range(1, min(m, 2) + 1):
            for er in combinations(range(k + m), e):
                present = [i not in er for i in range(k + m)]
                chunks = [c.copy() if present[i] else np.zeros(C, np.uint8)
                          for i, c in enumerate(full)]
                ctx.decode_chunks(chunks, present)
                for i in range(k + m):
                    assert np.array_equal(chunks[i], full[i]), (er, i)
    finally:
        ctx.close()


@pytest.mark.gpu
def test_gpu_w16_delta():
    k, m = 5, 2
    C = 16 * 1024
    rng = np.random.default_rng(3)
    ctx = ceph_amd.EcContext(k, m, "jerasure_reed_sol_van_w16", device=0)
    try:
        data = [rng.integers(0, 256, C, dtype=np.uint8) for _ in range(k)]
        par = ctx.encode_chunks(data)
        newc = rng.integers(0, 256, C, dtype=np.uint8)
        delta = ctx.encode_delta(data[1], newc)
        par2 = [p.copy() for p in par]
        for j in range(m):
            ctx.apply_delta(delta, 1, k + j, par2[j])
        data[1] = newc
        want = oracle.encode_w16(k, m, data)
        for j in range(m):
            assert np.array_equal(par2[j], want[j]), j
    finally:
        ctx.close()


@pytest.mark.gpu
def test_gpu_w16_via_plugin_cli():
    import os
    import subprocess
    ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    H = os.path.join(ROOT, "ceph_amd", "harness")
    r = subprocess.run(
        [os.path.join(H, "ec_benchmark"), "-d", H, "-p", "mi355x",
         "-P", "technique=jerasure_reed_sol_van", "-P", "w=16",
         "-P", "k=4", "-P", "m=2", "-s", str(4 * 65536), "-i", "2",
         "-w", "decode", "-e", "2", "-E", "exhaustive"],
        capture_output=True, text=True)
    assert r.returncode == 0, r.stderr + r.stdout
