"""Host-pointer staging path edge cases (pipelined_matmul_host /
staged_host_call): the double-buffered tile loop only engages for chunks
above the 4 MiB default tile, and the zeros-chunk convention must survive
the gather. All parity is checked against the CPU oracle."""
import numpy as np
import pytest

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def mods():
    import ceph_amd
    import oracle
    return ceph_amd, oracle


def _roundtrip(ceph_amd, oracle, tech, k, m, C, erase, **kw):
    rng = np.random.default_rng(C ^ k)
    data = [rng.integers(0, 256, C, dtype=np.uint8) for _ in range(k)]
    ctx = ceph_amd.EcContext(k, m, tech, device=0, **kw)
    try:
        par = ctx.encode_chunks(data)
        if tech == "cauchy_orig":
            want = oracle.bitmatrix_encode(k, m, data, 2048)
        elif tech == "jerasure_reed_sol_van_w16":
            want = oracle.encode_w16(k, m, data)
        else:
            want = oracle.encode(tech, k, m, data)
        for j in range(m):
            assert np.array_equal(par[j], want[j]), ("parity", j)
        chunks = [d.copy() for d in data] + [p.copy() for p in par]
        present = [i not in erase for i in range(k + m)]
        for e in erase:
            chunks[e][:] = 0
        ctx.decode_chunks(chunks, present)
        ref = data + par
        for i in range(k + m):
            assert np.array_equal(chunks[i], ref[i]), ("decode", i)
    finally:
        ctx.close()


def test_tiled_pipeline_two_tiles_ragged(mods):
    """chunk > 4 MiB default tile with a ragged last tile: T=2, second
    tile 16 bytes — exercises the double-buffer loop and the drain."""
    ceph_amd, oracle = mods
    _roundtrip(ceph_amd, oracle, "reed_sol_van", 8, 3,
               4 * 1024 * 1024 + 16, erase=(0, 9))


def test_tiled_pipeline_three_tiles(mods):
    """chunk = 9 MiB: T=3, buffer 1 reused — the scatter of tile t-2
    inside the loop runs (t >= 2 branch)."""
    ceph_amd, oracle = mods
    _roundtrip(ceph_amd, oracle, "cauchy", 6, 3, 9 * 1024 * 1024,
               erase=(1, 6, 8))


def test_zeros_chunk_through_gather(mods):
    """data[i] = None is the zeros-chunk convention
    (ErasureCodeJerasure.cc:146-157): the pipelined gather must skip the
    slot and the kernel must treat the source as zeros."""
    ceph_amd, oracle = mods
    k, m, C = 6, 2, 512 * 1024
    rng = np.random.default_rng(7)
    data = [rng.integers(0, 256, C, dtype=np.uint8) for _ in range(k)]
    full = [d.copy() for d in data]
    data[2] = None
    full[2][:] = 0
    ctx = ceph_amd.EcContext(k, m, "reed_sol_van", device=0)
    try:
        par = ctx.encode_chunks(data)
        want = oracle.encode("reed_sol_van", k, m, full)
        for j in range(m):
            assert np.array_equal(par[j], want[j])
    finally:
        ctx.close()


def test_w16_staged_large(mods):
    """w=16 technique through staged_host_call at a size past 4 MiB
    (single-shot staging, no tiling for w16)."""
    ceph_amd, oracle = mods
    _roundtrip(ceph_amd, oracle, "jerasure_reed_sol_van_w16", 5, 3,
               6 * 1024 * 1024, erase=(0, 4, 6))


def test_bitmatrix_staged(mods):
    """cauchy_orig (bitmatrix) through staged_host_call, packetsize-
    aligned chunk."""
    ceph_amd, oracle = mods
    _roundtrip(ceph_amd, oracle, "cauchy_orig", 4, 2, 1024 * 1024,
               erase=(0, 5))


def test_legacy_path_still_correct(mods, monkeypatch):
    """ECX_HOSTPIPE=0 must keep the legacy per-chunk staging path alive
    (it is read once per process, so run it in a subprocess)."""
    import subprocess
    import sys
    code = (
        "import numpy as np, ceph_amd, oracle\n"
        "rng = np.random.default_rng(3)\n"
        "data = [rng.integers(0,256,65536,dtype=np.uint8) for _ in range(4)]\n"
        "ctx = ceph_amd.EcContext(4,2,'reed_sol_van',device=0)\n"
        "par = ctx.encode_chunks(data)\n"
        "want = oracle.encode('reed_sol_van',4,2,data)\n"
        "assert all(np.array_equal(p,w) for p,w in zip(par,want))\n"
        "print('LEGACY_OK')\n")
    r = subprocess.run([sys.executable, "-c", code], capture_output=True,
                       text=True, env={**__import__('os').environ,
                                        "ECX_HOSTPIPE": "0"})
    assert r.returncode == 0 and "LEGACY_OK" in r.stdout, r.stderr


def test_all_zeros_sentinels_with_explicit_chunk_bytes(mods):
    """Every chunk a zeros sentinel (None): the wrapper cannot infer the
    length, so the explicit chunk_bytes parameter carries it (edge found
    by tools/fuzz_gpu.py); parity of all-zeros data is all zeros
    (the zeroinout property)."""
    ceph_amd, _ = mods
    ctx = ceph_amd.EcContext(4, 2, "reed_sol_van", device=0)
    try:
        par = ctx.encode_chunks([None] * 4, chunk_bytes=65536)
        for p in par:
            assert p.nbytes == 65536 and not p.any()
    finally:
        ctx.close()


def test_graph_replay_across_size_changes(mods):
    """Regression for the round-2 GPU memory fault: single-tile host
    calls replay captured hipGraphs that bake in the pinned/device
    staging addresses; growing the pipe (a bigger call on the same slot)
    reallocates those buffers and must invalidate the cached graphs.
    Alternate small and large calls on ONE context (n_streams=1 pins
    every call to the same slot) and byte-check each against the
    oracle."""
    ceph_amd, oracle = mods
    k, m = 4, 2
    ctx = ceph_amd.EcContext(k, m, "reed_sol_van", device=0, n_streams=1)
    rng = np.random.default_rng(0x6F)
    try:
        for C in (64 << 10, 64 << 10, 1 << 20, 64 << 10, 2 << 20,
                  128 << 10, 64 << 10):
            data = [rng.integers(0, 256, C, dtype=np.uint8)
                    for _ in range(k)]
            par = ctx.encode_chunks(data)
            want = oracle.encode("reed_sol_van", k, m, data)
            for j in range(m):
                assert np.array_equal(par[j], want[j]), (C, j)
    finally:
        ctx.close()


def test_small_calls_many_shapes_one_slot(mods):
    """Graph cache keying: different (n_src, n_out, tl) combinations on
    one slot (decode plans vary n_src/n_out) must not cross-talk."""
    ceph_amd, oracle = mods
    rng = np.random.default_rng(0x51A)
    for (k, m) in ((4, 2), (6, 3), (4, 2)):
        ctx = ceph_amd.EcContext(k, m, "reed_sol_van", device=0,
                                 n_streams=1)
        try:
            C = 64 << 10
            data = [rng.integers(0, 256, C, dtype=np.uint8)
                    for _ in range(k)]
            par = ctx.encode_chunks(data)
            want = oracle.encode("reed_sol_van", k, m, data)
            for j in range(m):
                assert np.array_equal(par[j], want[j])
            # decode with 1..m erasures => varying (n_src, n_out) keys
            for ne in range(1, m + 1):
                chunks = [d.copy() for d in data] + [p.copy() for p in par]
                er = list(range(ne))
                present = [i not in er for i in range(k + m)]
                for e in er:
                    chunks[e][:] = 0
                ctx.decode_chunks(chunks, present)
                ref = data + par
                for i in range(k + m):
                    assert np.array_equal(chunks[i], ref[i]), (k, m, ne, i)
        finally:
            ctx.close()
