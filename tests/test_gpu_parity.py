"""GPU parity tests: the HIP path must match the CPU oracle bit-exactly on
the same inputs (BASELINE.md 'Parity'). All tests here need
an MI355X and run through the C-ABI (include/ec_mi355x.h)."""
import ctypes
from itertools import combinations

import numpy as np
import pytest

import ceph_amd
import oracle

pytestmark = pytest.mark.gpu

TECHS = ["reed_sol_van", "cauchy", "jerasure_reed_sol_van"]


def make_ctx(k, m, tech):
    return ceph_amd.EcContext(k, m, tech, device=0)


@pytest.mark.parametrize("tech", TECHS)
@pytest.mark.parametrize("k,m", [(2, 1), (4, 2), (8, 3), (10, 4), (12, 4)])
def test_encode_parity_vs_oracle(tech, k, m):
    C = 64 * 1024
    rng = np.random.default_rng(0xEC ^ (k << 8) ^ m)
    ctx = make_ctx(k, m, tech)
    try:
        data = [rng.integers(0, 256, C, dtype=np.uint8) for _ in range(k)]
        got = ctx.encode_chunks(data)
        want = oracle.encode(tech, k, m, data)
        for j in range(m):
            assert np.array_equal(got[j], want[j]), (tech, k, m, j)
    finally:
        ctx.close()


@pytest.mark.parametrize("tech", TECHS)
def test_decode_exhaustive_patterns(tech):
    """Every erasure pattern for (k=4, m=3), mirroring the reference's
    exhaustive sweeps (TestErasureCodeIsa.cc:400-650)."""
    k, m = 4, 3
    C = 16 * 1024
    rng = np.random.default_rng(5)
    ctx = make_ctx(k, m, tech)
    try:
        data = [rng.integers(0, 256, C, dtype=np.uint8) for _ in range(k)]
        par = ctx.encode_chunks(data)
        full = data + par
        for e in range(1, m + 1):
            for er in combinations(range(k + m), e):
                present = [i not in er for i in range(k + m)]
                chunks = [c.copy() if present[i] else np.zeros(C, np.uint8)
                          for i, c in enumerate(full)]
                ctx.decode_chunks(chunks, present)
                for i in range(k + m):
                    assert np.array_equal(chunks[i], full[i]), (tech, er, i)
    finally:
        ctx.close()


def test_zero_chunk_convention():
    """NULL data pointer == zeros chunk (ErasureCodeJerasure.cc:146-157)."""
    k, m = 6, 3
    C = 4096
    rng = np.random.default_rng(9)
    ctx = make_ctx(k, m, "reed_sol_van")
    try:
        data = [rng.integers(0, 256, C, dtype=np.uint8) for _ in range(k)]
        data[2] = None
        data[5] = None
        got = ctx.encode_chunks(data)
        want = oracle.encode("reed_sol_van", k, m, data)
        for j in range(m):
            assert np.array_equal(got[j], want[j])
        # all-None => all-zero parity (zero-in-zero-out flag behaviour)
        got = ctx.encode_chunks([None] * (k - 1) +
                                [np.zeros(C, np.uint8)])
        for j in range(m):
            assert not got[j].any()
    finally:
        ctx.close()


@pytest.mark.parametrize("tech", TECHS)
def test_batch_api_vs_oracle(tech):
    """Device-resident batch encode+decode (the bench hot path) vs the
    oracle's batch path on identical bytes."""
    k, m = 8, 3
    S, C = 16, 64 * 1024
    n = k + m
    ctx = make_ctx(k, m, tech)
    try:
        nbytes = S * n * C
        host = np.random.default_rng(0xEC).integers(
            0, 256, nbytes, dtype=np.uint8)
        ref = host.copy()
        oracle.cpu_encode_batch(tech, k, m, ref, S, C)

        d = ctx.dbuf_alloc(nbytes)
        ctx.upload(d, host)
        ctx.encode_batch(d, S, C)
        out = np.zeros(nbytes, np.uint8)
        ctx.download(out, d)
        assert np.array_equal(out, ref), "batch encode mismatch"

        # decode: wipe 3 chunks per stripe on device by re-uploading zeros
        erased = [1, 6, 9]
        present_mask = (1 << n) - 1
        zero = np.zeros(C, np.uint8)
        for e in erased:
            present_mask &= ~(1 << e)
            for s in range(S):
                dst = ctypes.c_void_p(d.value + (s * n + e) * C)
                ctx.upload(dst, zero)
        ctx.decode_batch(d, S, C, present_mask)
        ctx.download(out, d)
        assert np.array_equal(out, ref), "batch decode mismatch"
    finally:
        ctx.close()


def test_fill_random_matches_host_replica():
    """bench.py's parity self-check depends on reproducing the device fill
    on the host."""
    import bench
    ctx = make_ctx(8, 3, "reed_sol_van")
    try:
        nbytes = 1 << 20
        d = ctx.dbuf_alloc(nbytes)
        ctx.fill_random(d, nbytes, 0xEC)
        ctx.sync()
        out = np.zeros(nbytes, np.uint8)
        ctx.download(out, d)
        assert np.array_equal(out, bench.expected_fill(0, nbytes, 0xEC))
        ctx.dbuf_free(d)
    finally:
        ctx.close()


def test_delta_ops_vs_oracle():
    """encode_delta/apply_delta == full re-encode (ParityDelta conformance,
    TestErasureCodePlugins.cc:302-...)."""
    k, m = 5, 3
    C = 16 * 1024
    tech = "reed_sol_van"
    rng = np.random.default_rng(21)
    ctx = make_ctx(k, m, tech)
    try:
        data = [rng.integers(0, 256, C, dtype=np.uint8) for _ in range(k)]
        par = ctx.encode_chunks(data)
        newc = rng.integers(0, 256, C, dtype=np.uint8)
        delta = ctx.encode_delta(data[2], newc)
        assert np.array_equal(delta, data[2] ^ newc)
        par2 = [p.copy() for p in par]
        for j in range(m):
            ctx.apply_delta(delta, 2, k + j, par2[j])
        data[2] = newc
        want = oracle.encode(tech, k, m, data)
        for j in range(m):
            assert np.array_equal(par2[j], want[j]), j
    finally:
        ctx.close()


def test_large_chunk_and_odd_sizes():
    """1 MiB chunks (BASELINE shape) and a few non-power-of-2 sizes
    (multiples of 16 per the C-ABI contract)."""
    k, m = 8, 3
    tech = "reed_sol_van"
    rng = np.random.default_rng(2)
    ctx = make_ctx(k, m, tech)
    try:
        for C in (1 << 20, 4096 + 16, 31 * 16, 1 << 16):
            data = [rng.integers(0, 256, C, dtype=np.uint8)
                    for _ in range(k)]
            got = ctx.encode_chunks(data)
            want = oracle.encode(tech, k, m, data)
            for j in range(m):
                assert np.array_equal(got[j], want[j]), C
        # non-multiple-of-16 must be rejected loudly, not silently wrong
        bad = [rng.integers(0, 256, 24, dtype=np.uint8) for _ in range(k)]
        with pytest.raises(ceph_amd.EcError):
            ctx.encode_chunks(bad)
    finally:
        ctx.close()


def test_decode_lru_reuse():
    """Repeated decode with the same erasure signature must hit the cached
    plan and stay bit-exact (ErasureCodeIsaTableCache analogue)."""
    k, m = 8, 3
    C = 4096
    ctx = make_ctx(k, m, "reed_sol_van")
    rng = np.random.default_rng(4)
    try:
        for trial in range(5):
            data = [rng.integers(0, 256, C, dtype=np.uint8)
                    for _ in range(k)]
            par = ctx.encode_chunks(data)
            full = data + par
            present = [i not in (0, 4, 10) for i in range(k + m)]
            chunks = [c.copy() if present[i] else np.zeros(C, np.uint8)
                      for i, c in enumerate(full)]
            ctx.decode_chunks(chunks, present)
            for i in range(k + m):
                assert np.array_equal(chunks[i], full[i]), (trial, i)
    finally:
        ctx.close()


def test_concurrent_calls_one_ctx():
    """One codec instance must be safe from many threads (the reference's
    plugins are called from many PG threads; ErasureCodeInterface.h
    threading contract — our slots serialize internally)."""
    import threading
    k, m = 8, 3
    C = 256 * 1024
    tech = "reed_sol_van"
    rng = np.random.default_rng(33)
    ctx = make_ctx(k, m, tech)
    datasets = [[rng.integers(0, 256, C, dtype=np.uint8) for _ in range(k)]
                for _ in range(4)]
    wants = [oracle.encode(tech, k, m, d) for d in datasets]
    errors = []

    def worker(t):
        try:
            for _ in range(5):
                got = ctx.encode_chunks(datasets[t])
                for j in range(m):
                    assert np.array_equal(got[j], wants[t][j])
        except Exception as e:  # pragma: no cover
            errors.append((t, repr(e)))

    threads = [threading.Thread(target=worker, args=(t,)) for t in range(4)]
    for th in threads:
        th.start()
    for th in threads:
        th.join()
    ctx.close()
    assert not errors, errors


@pytest.mark.parametrize("tech", ["cauchy_orig", "cauchy_good"])
def test_cauchy_bitmatrix_vs_oracle(tech):
    """jerasure cauchy_orig/cauchy_good (bitmatrix/packet layout) on the
    GPU: the LDS-staged XOR kernel must match the oracle's packet
    semantics bit-exactly, encode and decode (host path + device batch)."""
    k, m, p = 4, 3, 2048
    C = 8 * p * 4  # 4 superwords
    rng = np.random.default_rng(0xCA)
    ctx = ceph_amd.EcContext(k, m, tech, device=0, packetsize=p)
    try:
        data = [rng.integers(0, 256, C, dtype=np.uint8) for _ in range(k)]
        got = ctx.encode_chunks(data)
        want = oracle.bitmatrix_encode(k, m, data, p, technique=tech)
        for j in range(m):
            assert np.array_equal(got[j], want[j]), j
        # decode exhaustive over all patterns
        full = data + got
        for e in range(1, m + 1):
            for er in combinations(range(k + m), e):
                present = [i not in er for i in range(k + m)]
                chunks = [c.copy() if present[i] else np.zeros(C, np.uint8)
                          for i, c in enumerate(full)]
                ctx.decode_chunks(chunks, present)
                for i in range(k + m):
                    assert np.array_equal(chunks[i], full[i]), (er, i)
    finally:
        ctx.close()


@pytest.mark.parametrize("tech", ["cauchy_orig", "cauchy_good"])
def test_cauchy_bitmatrix_batch_roundtrip(tech):
    k, m, p = 7, 3, 2048
    n = k + m
    C = 8 * p * 8
    S = 8
    ctx = ceph_amd.EcContext(k, m, tech, device=0, packetsize=p)
    try:
        nbytes = S * n * C
        d = ctx.dbuf_alloc(nbytes)
        ctx.fill_random(d, nbytes, 0xB1)
        ctx.sync()
        ctx.encode_batch(d, S, C)
        ref = np.zeros(nbytes, np.uint8)
        ctx.download(ref, d)
        # spot-check stripe 3 vs oracle
        st = ref[3 * n * C:(3 * n + n) * C]
        data = [st[i * C:(i + 1) * C].copy() for i in range(k)]
        want = oracle.bitmatrix_encode(k, m, data, p, technique=tech)
        for j in range(m):
            assert np.array_equal(st[(k + j) * C:(k + j + 1) * C], want[j])
        # decode round trip
        erased = [0, 5, 9]
        mask = (1 << n) - 1
        zero = np.zeros(C, np.uint8)
        for e in erased:
            mask &= ~(1 << e)
            for s in range(S):
                ctx.upload(ctypes.c_void_p(d.value + (s * n + e) * C), zero)
        ctx.decode_batch(d, S, C, mask)
        out = np.zeros(nbytes, np.uint8)
        ctx.download(out, d)
        assert np.array_equal(out, ref)
    finally:
        ctx.close()


@pytest.mark.parametrize("tech", ["cauchy_orig", "cauchy_good"])
def test_bitmatrix_delta_equals_reencode(tech):
    """Bitmatrix parity-delta (schedule_apply_delta semantics,
    ErasureCodeJerasure.cc:348-377): delta-apply per (data, coding) pair
    must equal a full re-encode with the new data chunk."""
    k, m, p = 5, 3, 2048
    C = 8 * p * 2
    rng = np.random.default_rng(0xDE17A)
    ctx = ceph_amd.EcContext(k, m, tech, device=0, packetsize=p)
    try:
        data = [rng.integers(0, 256, C, dtype=np.uint8) for _ in range(k)]
        par = ctx.encode_chunks(data)
        newc = rng.integers(0, 256, C, dtype=np.uint8)
        delta = ctx.encode_delta(data[2], newc)
        assert np.array_equal(delta, data[2] ^ newc)
        par2 = [q.copy() for q in par]
        for j in range(m):
            ctx.apply_delta(delta, 2, k + j, par2[j])
        data[2] = newc
        want = oracle.bitmatrix_encode(k, m, data, p, technique=tech)
        for j in range(m):
            assert np.array_equal(par2[j], want[j]), j
    finally:
        ctx.close()


@pytest.mark.parametrize("k,m", [(16, 6), (32, 4), (20, 8)])
def test_large_km_parity(k, m):
    """Upper end of the isa limits (MAX_K=MAX_M=32, ErasureCodeIsa.h:48):
    wide stripes exercise the n_out>4 launch splitting and 64-bit masks."""
    C = 16 * 1024
    tech = "cauchy"  # Vandermonde is MDS-limited to m<=4 (ErasureCodeIsa.cc:598-631)
    rng = np.random.default_rng(k * 100 + m)
    ctx = make_ctx(k, m, tech)
    try:
        data = [rng.integers(0, 256, C, dtype=np.uint8) for _ in range(k)]
        got = ctx.encode_chunks(data)
        want = oracle.encode(tech, k, m, data)
        for j in range(m):
            assert np.array_equal(got[j], want[j]), j
        # decode with m/2 random erasures
        full = data + got
        er = sorted(rng.choice(k + m, size=m // 2, replace=False).tolist())
        present = [i not in er for i in range(k + m)]
        chunks = [c.copy() if present[i] else np.zeros(C, np.uint8)
                  for i, c in enumerate(full)]
        ctx.decode_chunks(chunks, present)
        for i in range(k + m):
            assert np.array_equal(chunks[i], full[i]), (er, i)
    finally:
        ctx.close()


def test_decode_insufficient_chunks_fails_loudly():
    """Fewer than k survivors must fail with EIO, not fabricate data
    (ErasureCodeInterface.h:29-35 error conventions)."""
    k, m = 4, 2
    C = 4096
    ctx = make_ctx(k, m, "reed_sol_van")
    try:
        chunks = [np.zeros(C, np.uint8) for _ in range(k + m)]
        present = [True, True, True, False, False, False]  # 3 < k
        with pytest.raises(ceph_amd.EcError, match="EIO|too many"):
            ctx.decode_chunks(chunks, present)
    finally:
        ctx.close()


def test_minimum_to_decode_semantics():
    ctx = make_ctx(6, 3, "reed_sol_van")
    try:
        want = 0b000001
        avail = 0b111111111
        assert ctx.minimum_to_decode(want, avail) == want
        # chunk 0 lost: first k available in id order (ErasureCode.cc:154-170)
        avail = 0b111111110
        assert ctx.minimum_to_decode(want, avail) == 0b001111110
    finally:
        ctx.close()


def test_randomized_config_sweep():
    """Seeded fuzz over (technique, k, m, C, erasure pattern): every config
    bit-exact vs the oracle. One test, many configs — the long tail the
    parametrized cases miss."""
    rng = np.random.default_rng(0xF00D)
    for trial in range(18):
        tech = ["reed_sol_van", "cauchy",
                "jerasure_reed_sol_van"][trial % 3]
        k = int(rng.integers(2, 13))
        m = int(rng.integers(1, 5))
        if tech != "cauchy":
            m = min(m, 4)  # Vandermonde MDS bound (ErasureCodeIsa.cc:598)
        C = int(rng.integers(1, 64)) * 16
        ctx = make_ctx(k, m, tech)
        try:
            data = [rng.integers(0, 256, C, dtype=np.uint8)
                    for _ in range(k)]
            got = ctx.encode_chunks(data)
            want = oracle.encode(tech, k, m, data)
            for j in range(m):
                assert np.array_equal(got[j], want[j]), (trial, tech, k, m, C)
            e = int(rng.integers(1, m + 1))
            er = rng.choice(k + m, size=e, replace=False)
            full = data + got
            present = [i not in er for i in range(k + m)]
            chunks = [c.copy() if present[i] else np.zeros(C, np.uint8)
                      for i, c in enumerate(full)]
            ctx.decode_chunks(chunks, present)
            for i in range(k + m):
                assert np.array_equal(chunks[i], full[i]), (trial, i)
        finally:
            ctx.close()


@pytest.mark.parametrize("pkt", [256, 512, 2048, 4096])
def test_cauchy_orig_packetsize_variants(pkt):
    """packetsize is a live profile key for the bitmatrix technique
    (ErasureCodeJerasure.h DEFAULT_PACKETSIZE '2048'); every value changes
    the byte layout and must match the oracle exactly."""
    k, m = 5, 2
    C = 8 * pkt * 2
    rng = np.random.default_rng(pkt)
    ctx = ceph_amd.EcContext(k, m, "cauchy_orig", device=0, packetsize=pkt)
    try:
        data = [rng.integers(0, 256, C, dtype=np.uint8) for _ in range(k)]
        got = ctx.encode_chunks(data)
        want = oracle.bitmatrix_encode(k, m, data, pkt)
        for j in range(m):
            assert np.array_equal(got[j], want[j]), (pkt, j)
    finally:
        ctx.close()


def test_multi_context_concurrent_techniques():
    """An OSD hosts many EC profiles at once: several live contexts of
    different techniques driven from concurrent threads must not
    interfere (separate slot pools, shared device)."""
    import threading

    import ceph_amd
    import oracle

    shapes = [("reed_sol_van", 8, 3), ("cauchy", 6, 2),
              ("jerasure_reed_sol_van", 4, 2), ("reed_sol_van", 10, 4)]
    C = 256 * 1024
    ctxs = [ceph_amd.EcContext(k, m, t, device=0) for (t, k, m) in shapes]
    errs = []

    def worker(idx):
        t, k, m = shapes[idx]
        rng = np.random.default_rng(idx)
        try:
            for _ in range(8):
                data = [rng.integers(0, 256, C, dtype=np.uint8)
                        for _ in range(k)]
                got = ctxs[idx].encode_chunks(data)
                base = "reed_sol_van" if t == "reed_sol_van" else t
                want = oracle.encode(base, k, m, data)
                for j in range(m):
                    if not np.array_equal(got[j], want[j]):
                        errs.append((idx, j))
                        return
        except Exception as e:  # noqa: BLE001 - surface into main thread
            errs.append((idx, repr(e)))

    threads = [threading.Thread(target=worker, args=(i,))
               for i in range(len(shapes))]
    for th in threads:
        th.start()
    for th in threads:
        th.join()
    for c in ctxs:
        c.close()
    assert not errs, errs


def test_matmul_batch_arbitrary_rows():
    """ecx_matmul_batch (the LRC layer-composition primitive): arbitrary
    coefficient rows over arbitrary chunk-id subsets of a device-resident
    stripe batch, checked against a numpy/oracle GF(2^8) mat-mul."""
    import ctypes

    import ceph_amd
    import oracle

    k, m, C, S = 6, 3, 4096, 32
    n = k + m
    ctx = ceph_amd.EcContext(k, m, "reed_sol_van", device=0)
    rng = np.random.default_rng(0xAB)
    host = rng.integers(0, 256, S * n * C, dtype=np.uint8)
    d = ctx.dbuf_alloc(host.nbytes)
    try:
        ctx.upload(d, host)
        # layer: outputs 7, 8 from sources 1, 3, 4 with random coeffs
        src_ids, out_ids = [1, 3, 4], [7, 8]
        rows = rng.integers(0, 256, (2, 3), dtype=np.uint8)
        ctx.matmul_batch(d, S, C, src_ids, out_ids, rows)
        ctx.sync()
        got = np.zeros_like(host)
        ctx.download(got, d)
        stripes = host.reshape(S, n, C)
        out = got.reshape(S, n, C)
        for s in (0, 11, S - 1):
            for jo, oid in enumerate(out_ids):
                want = np.zeros(C, np.uint8)
                for ji, sid in enumerate(src_ids):
                    tab = np.array([oracle.gf_mul(int(rows[jo, ji]), v)
                                    for v in range(256)], np.uint8)
                    want ^= tab[stripes[s, sid]]
                assert np.array_equal(out[s, oid], want), (s, oid)
            # untouched chunks unchanged
            for cid in range(n):
                if cid not in out_ids:
                    assert np.array_equal(out[s, cid], stripes[s, cid])
    finally:
        ctx.dbuf_free(d)
        ctx.close()


@pytest.mark.parametrize("tech,k,m", [
    ("reed_sol_van", 12, 4), ("cauchy", 12, 4),
    ("jerasure_reed_sol_van", 12, 4), ("cauchy_orig", 12, 4),
    ("cauchy_good", 12, 4), ("cauchy_good", 10, 4),
])
def test_sampled_decode_headline_shapes(tech, k, m):
    """Decode parity at the BASELINE headline widths (k=12/10, m=4):
    exhaustive single-erasure plus 60 sampled multi-erasure patterns
    (the (16 choose e) spaces are too large to sweep on every run)."""
    from itertools import combinations
    bitm = tech in ("cauchy_orig", "cauchy_good")
    p = 512
    C = (8 * p * 2) if bitm else 64 * 1024
    n = k + m
    rng = np.random.default_rng(0xBEEF ^ (k << 8) ^ m)
    kw = {"packetsize": p} if bitm else {}
    ctx = ceph_amd.EcContext(k, m, tech, device=0, **kw)
    try:
        data = [rng.integers(0, 256, C, dtype=np.uint8) for _ in range(k)]
        par = ctx.encode_chunks(data)
        if bitm:
            want = oracle.bitmatrix_encode(k, m, data, p, technique=tech)
        else:
            want = oracle.encode(tech, k, m, data)
        for j in range(m):
            assert np.array_equal(par[j], want[j]), ("enc", j)
        full = data + par
        patterns = [(e,) for e in range(n)]
        for _ in range(60):
            e = int(rng.integers(2, m + 1))
            patterns.append(tuple(sorted(
                rng.choice(n, size=e, replace=False).tolist())))
        for er in patterns:
            present = [i not in er for i in range(n)]
            chunks = [c.copy() if present[i] else np.zeros(C, np.uint8)
                      for i, c in enumerate(full)]
            ctx.decode_chunks(chunks, present)
            for i in range(n):
                assert np.array_equal(chunks[i], full[i]), (er, i)
    finally:
        ctx.close()
