"""cauchy_good (jerasure cauchy.c cauchy_good_general_coding_matrix,
general branch: cauchy_original + the n_ones-minimising improve pass;
selected by ErasureCodeJerasureCauchyGood::prepare,
ErasureCodeJerasure.cc:584-591). CPU tests: an independent numpy
replication of the transform pins the oracle, the core's gf.cpp copy is
pinned against the oracle via the ecx_gen_matrix_probe export, and the
bitmatrix round-trip/MDS sweeps run on the oracle. m == 2 (jerasure's
unsourceable cbest-table branch) must be refused everywhere."""
import ctypes
from itertools import combinations

import numpy as np
import pytest

import ceph_amd
import oracle


def np_n_ones(e):
    """Independent n_ones: popcount of the 8x8 companion bitmatrix."""
    bm = oracle.bitmatrix(np.array([[e]], dtype=np.uint8))
    return int(bm.sum())


def np_improve(coding):
    """Independent numpy replication of cauchy_improve_coding_matrix."""
    m, k = coding.shape
    a = coding.copy()
    for j in range(k):
        if a[0, j] != 1:
            t = oracle.gf_inv(a[0, j])
            for i in range(1, m):
                a[i, j] = oracle.gf_mul(a[i, j], t)
            a[0, j] = 1
    for i in range(1, m):
        bno = sum(np_n_ones(v) for v in a[i])
        bno_index = -1
        for j in range(k):
            if a[i, j] == 1:
                continue
            t = oracle.gf_inv(a[i, j])
            tno = sum(np_n_ones(oracle.gf_mul(v, t)) for v in a[i])
            if tno < bno:
                bno, bno_index = tno, j
        if bno_index != -1:
            t = oracle.gf_inv(a[i, bno_index])
            for j in range(k):
                a[i, j] = oracle.gf_mul(a[i, j], t)
    return a


@pytest.mark.parametrize("k,m", [(4, 3), (8, 3), (10, 4), (12, 4), (5, 1),
                                 (16, 3)])
def test_matrix_matches_independent_replication(k, m):
    orig = oracle.cauchy_orig_matrix(k, m)
    want = np_improve(orig)
    got = oracle.cauchy_good_matrix(k, m)
    assert np.array_equal(got, want), (k, m)


@pytest.mark.parametrize("k,m", [(4, 3), (8, 3), (10, 4), (12, 4)])
def test_row0_all_ones_and_fewer_ones(k, m):
    orig = oracle.cauchy_orig_matrix(k, m)
    good = oracle.cauchy_good_matrix(k, m)
    # step 1 normalises row 0 to all ones (pure-XOR first parity)
    assert (good[0] == 1).all()
    # the whole point: never more bitmatrix ones than cauchy_orig
    total = lambda a: sum(np_n_ones(v) for v in a.ravel())
    assert total(good) <= total(orig), (total(good), total(orig))
    assert total(good) < total(orig)  # strict for these shapes


def test_n_ones_probe_matches_everywhere():
    lib = ceph_amd.lib()
    lib.ecx_cauchy_n_ones_probe.restype = ctypes.c_int
    for e in range(256):
        want = np_n_ones(e)
        assert oracle.cauchy_n_ones(e) == want, e
        assert lib.ecx_cauchy_n_ones_probe(e) == want, e


@pytest.mark.parametrize("tech", ["reed_sol_van", "cauchy",
                                  "jerasure_reed_sol_van", "cauchy_orig",
                                  "cauchy_good"])
@pytest.mark.parametrize("k,m", [(4, 3), (8, 3), (10, 4)])
def test_core_matrix_probe_matches_oracle(tech, k, m):
    """gf.cpp's generator construction == the oracle's, via the CPU-only
    ecx_gen_matrix_probe export (no GPU context needed)."""
    lib = ceph_amd.lib()
    lib.ecx_gen_matrix_probe.restype = ctypes.c_int
    a = np.zeros((k + m, k), dtype=np.uint8)
    r = lib.ecx_gen_matrix_probe(ceph_amd.TECHNIQUES[tech], k, m,
                                 a.ctypes.data_as(ctypes.c_void_p))
    assert r == 0
    assert np.array_equal(a, oracle.matrix(tech, k, m)), tech


def test_m2_refused_everywhere():
    """jerasure's m==2 cauchy_good reads its precomputed cbest tables,
    which cannot be faithfully restated here — every layer must refuse
    rather than silently diverge."""
    with pytest.raises(ValueError):
        oracle.cauchy_good_matrix(6, 2)
    lib = ceph_amd.lib()
    a = np.zeros((8, 6), dtype=np.uint8)
    assert lib.ecx_gen_matrix_probe(ceph_amd.T_CAUCHY_GOOD_JERASURE, 6, 2,
                                    a.ctypes.data_as(ctypes.c_void_p)) == -22
    # ecx_create2 refuses with EINVAL before the GPU check, so this is
    # CPU-testable: EINVAL (-22), not ENODEV (-19)
    ctxp = ctypes.c_void_p()
    r = lib.ecx_create2(6, 2, ceph_amd.T_CAUCHY_GOOD_JERASURE, 8, 2048, 0, 1,
                        ctypes.byref(ctxp))
    assert r == -22


def test_m2_refused_by_plugin_parse():
    """The dlopen plugin's parse() rejects technique=cauchy_good m=2 with
    a sourcing message before touching the GPU."""
    import os
    import subprocess
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    harness = os.path.join(root, "ceph_amd", "harness")
    bench = os.path.join(harness, "ec_benchmark")
    if not os.path.exists(bench):
        pytest.skip("ec_benchmark not built")
    r = subprocess.run(
        [bench, "-p", "mi355x", "-P", "technique=cauchy_good", "-P", "k=6",
         "-P", "m=2", "-s", "65536", "-i", "1", "-d", harness],
        capture_output=True, text=True)
    assert r.returncode != 0
    assert "cbest" in (r.stderr + r.stdout)


@pytest.mark.parametrize("k,m", [(4, 3), (6, 4)])
def test_oracle_bitmatrix_roundtrip_exhaustive(k, m):
    """Oracle encode/decode round trip for cauchy_good over every erasure
    pattern (MDS check — row/column scaling preserves the Cauchy MDS
    property; this verifies it end to end)."""
    p = 64
    C = 8 * p * 2
    rng = np.random.default_rng(0x600D ^ (k << 8) ^ m)
    data = [rng.integers(0, 256, C, dtype=np.uint8) for _ in range(k)]
    par = oracle.bitmatrix_encode(k, m, data, p, technique="cauchy_good")
    full = [d.copy() for d in data] + [q.copy() for q in par]
    n = k + m
    for e in range(1, m + 1):
        for er in combinations(range(n), e):
            present = np.array([1 if i not in er else 0 for i in range(n)],
                               dtype=np.uint8)
            chunks = [c.copy() if present[i] else np.zeros(C, np.uint8)
                      for i, c in enumerate(full)]
            oracle.bitmatrix_decode(k, m, chunks, present, p,
                                    technique="cauchy_good")
            for i in range(n):
                assert np.array_equal(chunks[i], full[i]), (er, i)


def test_good_differs_from_orig_but_same_code_space():
    """Sanity: the improved matrix produces different parity bytes than
    cauchy_orig (it is a different generator), while both remain MDS over
    the same data."""
    k, m, p = 8, 3, 64
    C = 8 * p
    rng = np.random.default_rng(3)
    data = [rng.integers(0, 256, C, dtype=np.uint8) for _ in range(k)]
    a = oracle.bitmatrix_encode(k, m, data, p, technique="cauchy_orig")
    b = oracle.bitmatrix_encode(k, m, data, p, technique="cauchy_good")
    assert any(not np.array_equal(a[j], b[j]) for j in range(m))


@pytest.mark.parametrize("tech", ["reed_sol_van", "cauchy",
                                  "jerasure_reed_sol_van"])
def test_decode_rows_probe_matches_oracle(tech):
    """gf.cpp's decode-plan composition (survivor pick + inversion +
    per-erasure rows) == decoding with the oracle, pinned on CPU via
    ecx_decode_rows_probe over 40 random erasure patterns."""
    lib = ceph_amd.lib()
    fn = lib.ecx_decode_rows_probe
    fn.restype = ctypes.c_int
    k, m = 8, 3
    n = k + m
    rng = np.random.default_rng(0xD0)
    C = 512
    data = [rng.integers(0, 256, C, dtype=np.uint8) for _ in range(k)]
    par = oracle.encode(tech, k, m, data)
    full = data + par
    for _ in range(40):
        ne = int(rng.integers(1, m + 1))
        er = sorted(rng.choice(n, size=ne, replace=False).tolist())
        mask = sum(1 << i for i in range(n) if i not in er)
        sv = (ctypes.c_int * k)()
        eo = (ctypes.c_int * m)()
        rows = np.zeros((m, k), dtype=np.uint8)
        r = fn(ceph_amd.TECHNIQUES[tech], k, m, ctypes.c_uint64(mask), sv,
               eo, rows.ctypes.data_as(ctypes.c_void_p))
        assert r == ne, (er, r)
        assert list(eo[:ne]) == er
        # applying the composed rows to the survivors reconstructs the
        # erased chunks byte-exactly (checked with the oracle's encoder)
        srcs = [full[sv[i]] for i in range(k)]
        got = oracle.encode_with_rows(
            np.ascontiguousarray(rows[:ne]), srcs)
        for i, e in enumerate(er):
            assert np.array_equal(got[i], full[e]), (er, e)
