"""Known-answer tests pinning the oracle's GF(2^8) arithmetic and matrix
constructions (no fixed-parity KATs exist in the reference tree;
these pin the *documented* algorithms by hand-computed facts)."""
import numpy as np
import pytest

import oracle


def test_gf_field_kats():
    # hand-computed facts in GF(2^8)/0x11d
    assert oracle.gf_mul(0, 5) == 0
    assert oracle.gf_mul(1, 123) == 123
    assert oracle.gf_mul(2, 2) == 4
    assert oracle.gf_mul(2, 0x80) == 0x1d          # overflow reduces by 0x11d
    assert oracle.gf_mul(3, 7) == 9                # (x+1)(x^2+x+1) = x^3+1
    assert oracle.gf_mul(2, 0x8e) == 0x01          # 2*0x8e = 0x11c ^ 0x11d = 1
    assert oracle.gf_inv(0) == 0                   # isa-l gf_inv(0) == 0
    for a in range(1, 256):
        assert oracle.gf_mul(a, oracle.gf_inv(a)) == 1


def test_gf_tables_consistent():
    logt, expt = oracle.gf_log_table(), oracle.gf_exp_table()
    assert expt[0] == 1 and expt[1] == 2 and expt[8] == 0x1d
    # log/exp are inverse on 1..255
    for v in (1, 2, 3, 0x1d, 0x80, 255):
        assert expt[logt[v]] == v
    # generator order 255: all of 1..255 appear
    assert len(set(expt[:255].tolist())) == 255


@pytest.mark.parametrize("tech", ["reed_sol_van", "cauchy",
                                  "jerasure_reed_sol_van"])
@pytest.mark.parametrize("k,m", [(2, 1), (4, 2), (8, 3), (10, 4), (12, 4)])
def test_matrix_structure(tech, k, m):
    g = oracle.matrix(tech, k, m)
    # systematic: identity on top (all constructions)
    assert (g[:k] == np.eye(k, dtype=np.uint8)).all()
    if tech in ("reed_sol_van", "jerasure_reed_sol_van"):
        # first coding row all ones => parity0 == XOR of data; the reference
        # relies on this (ErasureCodeIsa.cc:395-456 xor fast path,
        # jerasure row_k_ones=1 at ErasureCodeJerasure.cc:394)
        assert (g[k] == 1).all()


def test_isa_rs_matrix_values():
    # gf_gen_rs_matrix: row k+i, col j == (2^i)^j
    k, m = 6, 4
    g = oracle.matrix("reed_sol_van", k, m)
    for i in range(m):
        gen = 1
        for _ in range(i):
            gen = oracle.gf_mul(gen, 2)
        p = 1
        for j in range(k):
            assert g[k + i, j] == p
            p = oracle.gf_mul(p, gen)


def test_isa_cauchy_matrix_values():
    k, m = 5, 3
    g = oracle.matrix("cauchy", k, m)
    for i in range(m):
        for j in range(k):
            assert g[k + i, j] == oracle.gf_inv((k + i) ^ j)


@pytest.mark.parametrize("tech", ["reed_sol_van", "cauchy",
                                  "jerasure_reed_sol_van"])
def test_mds_property(tech):
    """Every k x k submatrix of the generator must be invertible (MDS) —
    checked over all erasure patterns for small (k,m), mirroring the
    reference's exhaustive sweeps (TestErasureCodeIsa.cc:400-650)."""
    from itertools import combinations
    k, m = 5, 3
    g = oracle.matrix(tech, k, m)
    n = k + m
    for rows in combinations(range(n), k):
        sub = g[list(rows)].astype(np.uint8)
        # invert via oracle decode on synthetic data: encode, erase
        # complement, decode must reproduce
        rng = np.random.default_rng(1)
        data = [rng.integers(0, 256, 64, dtype=np.uint8) for _ in range(k)]
        par = oracle.encode(tech, k, m, data)
        chunks = [d.copy() for d in data] + [p.copy() for p in par]
        ref = [c.copy() for c in chunks]
        present = np.zeros(n, np.uint8)
        present[list(rows)] = 1
        for i in range(n):
            if not present[i]:
                chunks[i][:] = 0
        oracle.decode(tech, k, m, chunks, present)
        for i in range(n):
            assert (chunks[i] == ref[i]).all(), (tech, rows, i)


def test_golden_vectors():
    """Self-pin: committed vectors generated once by the oracle
    (tests/golden/gen_golden.py). Guards against regressions in the GF
    tables, matrix derivations and encode semantics."""
    import os
    path = os.path.join(os.path.dirname(__file__), "golden",
                        "ec_golden.npz")
    gold = np.load(path)
    for tech in ("reed_sol_van", "cauchy", "jerasure_reed_sol_van"):
        for (k, m) in ((2, 1), (8, 3), (10, 4)):
            key = f"{tech}_k{k}m{m}"
            assert (oracle.matrix(tech, k, m) == gold[f"mat_{key}"]).all()
            C = 256
            rng = np.random.default_rng(0xEC)
            data = [rng.integers(0, 256, C, dtype=np.uint8)
                    for _ in range(k)]
            par = oracle.encode(tech, k, m, data)
            assert (np.stack(par) == gold[f"par_{key}"]).all()


def _gf_inv_matrix(M, mul, inv):
    """Gauss-Jordan inverse over a GF given mul/inv callables (test-local,
    independent of the oracle's implementation)."""
    n = M.shape[0]
    A = M.copy()
    I = np.eye(n, dtype=M.dtype)
    for i in range(n):
        if A[i, i] == 0:
            j = next(r for r in range(i + 1, n) if A[r, i] != 0)
            A[[i, j]] = A[[j, i]]
            I[[i, j]] = I[[j, i]]
        pinv = inv(int(A[i, i]))
        for c in range(n):
            A[i, c] = mul(pinv, int(A[i, c]))
            I[i, c] = mul(pinv, int(I[i, c]))
        for r in range(n):
            if r == i or A[r, i] == 0:
                continue
            f = int(A[r, i])
            for c in range(n):
                A[r, c] ^= mul(f, int(A[i, c]))
                I[r, c] ^= mul(f, int(I[i, c]))
    return I


def test_jerasure_vandermonde_uniqueness_crosscheck():
    """The jerasure RS-van coding matrix is UNIQUELY determined by two
    published facts, independent of elimination order: (1) the code of the
    extended Vandermonde matrix (reed_sol_extended_vandermonde_matrix: row
    0 = e0, rows 1..k+m-2 = powers of i, last row = e_{k-1}); (2) the
    normalisation 'systematic top identity + first coding row all ones'.
    Derivation: G_sys = G_ext * inv(G_ext_top); coding = P * diag(1/P[0]).
    This cross-check computes that closed form independently and compares
    with the oracle's elimination-based restatement — pinning the
    construction beyond 'same algorithm transcribed'."""
    for (k, m) in ((2, 1), (4, 3), (8, 3), (10, 4)):
        # independent extended Vandermonde
        n = k + m
        G = np.zeros((n, k), dtype=np.uint8)
        G[0, 0] = 1
        G[n - 1, k - 1] = 1
        for i in range(1, n - 1):
            v = 1
            for j in range(k):
                G[i, j] = v
                v = oracle.gf_mul(v, i)
        top_inv = _gf_inv_matrix(G[:k], oracle.gf_mul, oracle.gf_inv)
        # G_sys = G @ top_inv over GF(2^8)
        P = np.zeros((m, k), dtype=np.uint8)
        for r in range(m):
            for c in range(k):
                s = 0
                for t in range(k):
                    s ^= oracle.gf_mul(int(G[k + r, t]), int(top_inv[t, c]))
                P[r, c] = s
        assert (P[0] != 0).all()
        for c in range(k):
            sc = oracle.gf_inv(int(P[0, c]))
            for r in range(m):
                P[r, c] = oracle.gf_mul(int(P[r, c]), sc)
        got = oracle.matrix("jerasure_reed_sol_van", k, m)[k:]
        assert np.array_equal(got, P), (k, m)


def test_jerasure_vandermonde_w16_uniqueness_crosscheck():
    """Same closed-form cross-check in GF(2^16)."""
    def inv16(a):
        # brute via log tables through oracle.gf16_mul search is slow;
        # use Fermat: a^(2^16-2)
        r, e, b = 1, 65534, a
        while e:
            if e & 1:
                r = oracle.gf16_mul(r, b)
            b = oracle.gf16_mul(b, b)
            e >>= 1
        return r

    k, m = 5, 3
    n = k + m
    G = np.zeros((n, k), dtype=np.uint16)
    G[0, 0] = 1
    G[n - 1, k - 1] = 1
    for i in range(1, n - 1):
        v = 1
        for j in range(k):
            G[i, j] = v
            v = oracle.gf16_mul(v, i)
    top_inv = _gf_inv_matrix(G[:k], oracle.gf16_mul, inv16)
    P = np.zeros((m, k), dtype=np.uint16)
    for r in range(m):
        for c in range(k):
            s = 0
            for t in range(k):
                s ^= oracle.gf16_mul(int(G[k + r, t]), int(top_inv[t, c]))
            P[r, c] = s
    for c in range(k):
        sc = inv16(int(P[0, c]))
        for r in range(m):
            P[r, c] = oracle.gf16_mul(int(P[r, c]), sc)
    got = oracle.matrix_w16(k, m)[k:]
    assert np.array_equal(got, P)
