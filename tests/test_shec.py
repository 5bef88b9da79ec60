"""SHEC (shingled EC) tests. The construction is restated from the
reference's OWN in-tree code (src/erasure-code/shec/ErasureCodeShec.cc:
700-768) — this file pins it with an independent numpy replication, and
the CLI drives registry-level round trips (GPU tests in
test_harness_gpu-style live below under the gpu marker)."""
import os
import subprocess

import numpy as np
import pytest

import ceph_amd
import oracle

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
HARNESS = os.path.join(ROOT, "ceph_amd", "harness")


def numpy_shec_matrix(k, m, c, single=False):
    """Independent replication of shec_reedsolomon_coding_matrix."""
    def r_e1(k, m1, m2, c1, c2):
        if m1 < c1 or m2 < c2:
            return -1
        if (m1 == 0 and c1 != 0) or (m2 == 0 and c2 != 0):
            return -1
        r_eff_k = [10 ** 8] * k
        tot = 0.0
        for (mm, cc_) in ((m1, c1), (m2, c2)):
            for rr in range(mm):
                start = ((rr * k) // mm) % k
                end = (((rr + cc_) * k) // mm) % k
                cc = start
                first = True
                while first or cc != end:
                    first = False
                    r_eff_k[cc] = min(r_eff_k[cc],
                                      ((rr + cc_) * k) // mm -
                                      (rr * k) // mm)
                    cc = (cc + 1) % k
                tot += ((rr + cc_) * k) // mm - (rr * k) // mm
        return (tot + sum(r_eff_k)) / (k + m1 + m2)

    if single:
        m1, c1, m2, c2 = 0, 0, m, c
    else:
        best = None
        for c1 in range(c // 2 + 1):
            for m1 in range(m + 1):
                c2, m2 = c - c1, m - m1
                if m1 < c1 or m2 < c2:
                    continue
                if (m1 == 0) != (c1 == 0) or (m2 == 0) != (c2 == 0):
                    continue
                r = r_e1(k, m1, m2, c1, c2)
                if best is None or r < best[0] - np.finfo(float).eps:
                    best = (r, c1, m1)
        _, c1, m1 = best
        m2, c2 = m - m1, c - c1

    mat = oracle.matrix("jerasure_reed_sol_van", k, m)[k:].copy()
    for (base, mm, cc_) in ((0, m1, c1), (m1, m2, c2)):
        for rr in range(mm):
            end = ((rr * k) // mm) % k
            start = (((rr + cc_) * k) // mm) % k
            cc = start
            while cc != end:
                mat[base + rr, cc] = 0
                cc = (cc + 1) % k
    return mat


@pytest.mark.parametrize("k,m,c", [(4, 3, 2), (6, 4, 2), (8, 4, 3),
                                   (10, 3, 2), (12, 4, 2)])
@pytest.mark.parametrize("single", [False, True])
def test_matrix_matches_independent_replication(k, m, c, single):
    got = ceph_amd.shec_matrix(k, m, c, single)
    want = numpy_shec_matrix(k, m, c, single)
    assert np.array_equal(got, want), (k, m, c, single)


def test_matrix_shingle_structure():
    """Every data column must be covered by at least c parities (the SHEC
    durability property)."""
    for (k, m, c) in ((4, 3, 2), (8, 4, 3), (12, 4, 2)):
        mat = ceph_amd.shec_matrix(k, m, c)
        cover = (mat != 0).sum(axis=0)
        assert (cover >= c).all(), (k, m, c, cover)


def run_bench(*args):
    binp = os.path.join(HARNESS, "ec_benchmark")
    if not os.path.exists(binp):
        pytest.skip("harness not built")
    return subprocess.run([binp, "-d", HARNESS, *args],
                          capture_output=True, text=True)


def test_plugin_rejects_bad_params():
    r = run_bench("-p", "shec", "-P", "technique=nope", "-P", "k=4",
                  "-P", "m=3", "-P", "c=2", "-s", "65536", "-i", "1")
    assert r.returncode != 0
    r = run_bench("-p", "shec", "-P", "k=13", "-P", "m=3", "-P", "c=2",
                  "-s", "65536", "-i", "1")
    assert r.returncode != 0


@pytest.mark.gpu
def test_shec_encode_parity_vs_oracle():
    """GPU SHEC parity == oracle encode with the same shingled matrix."""
    k, m, c = 4, 3, 2
    C = 64 * 1024
    rng = np.random.default_rng(0x5EC)
    mat = ceph_amd.shec_matrix(k, m, c)
    ctx = ceph_amd.EcContext(k, m, "jerasure_reed_sol_van", device=0)
    try:
        ctx.set_matrix(mat)
        data = [rng.integers(0, 256, C, dtype=np.uint8) for _ in range(k)]
        got = ctx.encode_chunks(data)
        want = oracle.encode_with_rows(mat, data)
        for j in range(m):
            assert np.array_equal(got[j], want[j]), j
    finally:
        ctx.close()


@pytest.mark.gpu
def test_shec_plugin_single_erasure_exhaustive():
    """Any single erasure is recoverable; CLI verifies recovered bytes
    (benchmark.cc:211-258 style)."""
    r = run_bench("-p", "shec", "-P", "k=4", "-P", "m=3", "-P", "c=2",
                  "-s", str(4 * 4096), "-i", "2", "-w", "decode", "-e", "1",
                  "-E", "exhaustive")
    assert r.returncode == 0, r.stderr + r.stdout


@pytest.mark.gpu
def test_shec_plugin_double_erasure_exhaustive_c2():
    """With c=2 every two-chunk erasure the reference can recover must
    round-trip; k=4 m=3 c=2 is fully 2-recoverable."""
    r = run_bench("-p", "shec", "-P", "k=4", "-P", "m=3", "-P", "c=2",
                  "-s", str(4 * 4096), "-i", "1", "-w", "decode", "-e", "2",
                  "-E", "exhaustive")
    assert r.returncode == 0, r.stderr + r.stdout


@pytest.mark.gpu
def test_shec_plugin_encode_runs():
    r = run_bench("-p", "shec", "-P", "k=6", "-P", "m=4", "-P", "c=2",
                  "-s", str(1 << 20), "-i", "3")
    assert r.returncode == 0, r.stderr


# ---- minimum_to_decode (CPU, via the ecx_shec_minimum_probe export) ----
# The probe in plugin_shec.cc runs the exact minimum computation the
# plugin's _minimum_to_decode uses (ErasureCodeShec.cc:130-178 + :943-962).

def _shec_minimum(k, m, c, want_ids, avail_ids, single=False):
    import ctypes
    lib = ctypes.CDLL(os.path.join(HARNESS, "libec_shec.so"))
    fn = lib.ecx_shec_minimum_probe
    fn.restype = ctypes.c_int
    fn.argtypes = [ctypes.c_int, ctypes.c_int, ctypes.c_int, ctypes.c_int,
                   ctypes.c_uint64, ctypes.c_uint64,
                   ctypes.POINTER(ctypes.c_uint64)]
    want = sum(1 << i for i in want_ids)
    avail = sum(1 << i for i in avail_ids)
    out = ctypes.c_uint64(0)
    rc = fn(k, m, c, int(single), want, avail, ctypes.byref(out))
    return rc, {i for i in range(64) if out.value >> i & 1}


def test_minimum_all_wanted_available():
    rc, mn = _shec_minimum(4, 3, 2, {0, 2}, {0, 1, 2, 3, 4})
    assert rc == 0 and mn == {0, 2}


def test_minimum_includes_wanted_available_data():
    """Regression for the want&~avail masking bug: when chunk 1 is erased
    and chunks {0,1} are wanted, the minimum must still include the
    wanted AVAILABLE chunk 0 (ErasureCodeShec.cc:957-959 adds
    want[i]&&avails[i] chunks)."""
    k, m, c = 4, 3, 2
    avail = set(range(k + m)) - {1}
    rc, mn = _shec_minimum(k, m, c, {0, 1}, avail)
    assert rc == 0
    assert 0 in mn, mn
    # minimum is a read set: every member must be available
    assert mn <= avail, mn
    # and it must contain enough chunks to actually recover chunk 1
    assert len(mn - {0}) >= 1


@pytest.mark.parametrize("erased", range(7))
def test_minimum_single_erasure_subset_of_available(erased):
    k, m, c = 4, 3, 2
    avail = set(range(k + m)) - {erased}
    want = {erased} | ({0} if erased != 0 else {2})
    rc, mn = _shec_minimum(k, m, c, want, avail)
    assert rc == 0
    assert mn <= avail, (erased, mn)
    assert (want & avail) <= mn, (erased, mn)


def test_minimum_wanted_available_parity_kept():
    """A wanted available parity chunk is added to minimum when it covers
    a data chunk outside want (ErasureCodeShec.cc:943-953 branch)."""
    k, m, c = 4, 3, 2
    avail = set(range(k + m)) - {0}
    rc, mn = _shec_minimum(k, m, c, {0, k}, avail)  # want data 0 + parity k
    assert rc == 0
    assert k in mn, mn
    assert mn <= avail, mn
