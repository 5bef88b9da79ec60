"""GPU tests for the variable-size slice batch API (the OSD
write path's many small variable-blocksize encode_chunks calls,
shard_extent_map_t::encode, src/osd/ECUtil.cc:485-514 — batched here into
one launch)."""
import ctypes

import numpy as np
import pytest

import ceph_amd
import oracle

pytestmark = pytest.mark.gpu


def carve(base, off, n):
    return ctypes.c_void_p(base.value + off)


def test_encode_slices_vs_oracle():
    k, m, tech = 8, 3, "reed_sol_van"
    n = k + m
    sizes = [4096, 64 * 1024, 1 << 20, 16 * 31, 4096 + 16]
    ctx = ceph_amd.EcContext(k, m, tech, device=0)
    try:
        total = sum(sizes) * n
        d = ctx.dbuf_alloc(total)
        ctx.fill_random(d, total, 0xEC)
        ctx.sync()

        ptrs, offs = [], []
        off = 0
        for sz in sizes:
            offs.append(off)
            for c in range(n):
                ptrs.append(d.value + off + c * sz)
            off += sz * n

        ctx.encode_slices(ptrs, sizes)
        ctx.sync()

        import bench
        for si, sz in enumerate(sizes):
            host = np.zeros(sz * n, dtype=np.uint8)
            ctx.download(host, carve(d, offs[si], 0))
            data = [host[i * sz:(i + 1) * sz] for i in range(k)]
            # data region must equal the deterministic fill
            exp = bench.expected_fill(offs[si], k * sz, 0xEC)
            assert np.array_equal(host[:k * sz], exp), si
            want = oracle.encode(tech, k, m, data)
            for j in range(m):
                got = host[(k + j) * sz:(k + j + 1) * sz]
                assert np.array_equal(got, want[j]), (si, j)
    finally:
        ctx.close()


def test_encode_slices_null_data_is_zeros():
    k, m = 4, 2
    n = k + m
    sz = 8192
    ctx = ceph_amd.EcContext(k, m, "reed_sol_van", device=0)
    try:
        d = ctx.dbuf_alloc(sz * n)
        ctx.fill_random(d, sz * n, 7)
        ctx.sync()
        ptrs = [d.value + c * sz for c in range(n)]
        ptrs[2] = None  # zeros chunk
        ctx.encode_slices(ptrs, [sz])
        ctx.sync()
        host = np.zeros(sz * n, dtype=np.uint8)
        ctx.download(host, d)
        data = [host[i * sz:(i + 1) * sz] for i in range(k)]
        data[2] = None
        want = oracle.encode("reed_sol_van", k, m, data, chunk_bytes=sz)
        for j in range(m):
            assert np.array_equal(host[(k + j) * sz:(k + j + 1) * sz],
                                  want[j]), j
    finally:
        ctx.close()


def test_decode_slices_round_trip():
    k, m, tech = 6, 3, "cauchy"
    n = k + m
    sizes = [16 * 1024, 4096, 256 * 1024]
    ctx = ceph_amd.EcContext(k, m, tech, device=0)
    try:
        total = sum(sizes) * n
        d = ctx.dbuf_alloc(total)
        ctx.fill_random(d, total, 0xD)
        ctx.sync()
        ptrs = []
        off = 0
        offs = []
        for sz in sizes:
            offs.append(off)
            for c in range(n):
                ptrs.append(d.value + off + c * sz)
            off += sz * n
        ctx.encode_slices(ptrs, sizes)
        ctx.sync()
        ref = np.zeros(total, dtype=np.uint8)
        ctx.download(ref, d)

        # erase chunks 1, 7, 8 in every slice (zero the regions)
        erased = [1, 7, 8]
        mask = (1 << n) - 1
        for e in erased:
            mask &= ~(1 << e)
        for si, sz in enumerate(sizes):
            z = np.zeros(sz, dtype=np.uint8)
            for e in erased:
                ctx.upload(ctypes.c_void_p(d.value + offs[si] + e * sz), z)
        ctx.decode_slices(ptrs, sizes, mask)
        ctx.sync()
        out = np.zeros(total, dtype=np.uint8)
        ctx.download(out, d)
        assert np.array_equal(out, ref)
    finally:
        ctx.close()


def test_slices_reject_bad_sizes():
    ctx = ceph_amd.EcContext(4, 2, "reed_sol_van", device=0)
    try:
        d = ctx.dbuf_alloc(4096)
        ptrs = [d.value] * 6
        with pytest.raises(ceph_amd.EcError):
            ctx.encode_slices(ptrs, [24])  # not a multiple of 16
    finally:
        ctx.close()
