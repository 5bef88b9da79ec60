"""ceph_amd — Python bindings for the MI355X-native erasure-coding core.

Product path: thin ctypes over the C-ABI of include/ec_mi355x.h
(libec_mi355x_core.so, hand-written HIP for gfx950). There is NO CPU
fallback anywhere in this package: if the extension is missing or no GPU is
visible, calls raise, loudly.
"""
import ctypes
import os

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "libec_mi355x_core.so")

T_RS_VAN_ISA = 0
T_CAUCHY_ISA = 1
T_RS_VAN_JERASURE = 2
T_CAUCHY_ORIG_JERASURE = 3
T_RS_VAN_JERASURE_W16 = 4
T_CAUCHY_GOOD_JERASURE = 5
TECHNIQUES = {
    "reed_sol_van": T_RS_VAN_ISA,
    "cauchy": T_CAUCHY_ISA,
    "jerasure_reed_sol_van": T_RS_VAN_JERASURE,
    "cauchy_orig": T_CAUCHY_ORIG_JERASURE,
    "jerasure_reed_sol_van_w16": T_RS_VAN_JERASURE_W16,
    "cauchy_good": T_CAUCHY_GOOD_JERASURE,
}

_ERR = {
    -22: "EINVAL", -12: "ENOMEM", -5: "EIO (too many erasures / singular)",
    -19: "ENODEV (no GPU — the mi355x EC core has no CPU fallback)",
    -71: "EPROTO (HIP runtime failure)",
}


class EcError(RuntimeError):
    pass


def _lib():
    if not os.path.exists(_SO):
        raise EcError(
            f"{_SO} not built. Run `make core` at the repo root (or "
            "python -c 'import __graft_entry__; __graft_entry__.build()').")
    lib = ctypes.CDLL(_SO)
    lib.ecx_version.restype = ctypes.c_char_p
    lib.ecx_device_count.restype = ctypes.c_int
    lib.ecx_create.argtypes = [ctypes.c_int] * 5 + [ctypes.POINTER(ctypes.c_void_p)]
    lib.ecx_create2.argtypes = [ctypes.c_int] * 7 + [ctypes.POINTER(ctypes.c_void_p)]
    lib.ecx_chunk_size.restype = ctypes.c_uint
    lib.ecx_chunk_size.argtypes = [ctypes.c_void_p, ctypes.c_uint]
    lib.ecx_minimum_to_decode.argtypes = [
        ctypes.c_void_p, ctypes.c_uint64, ctypes.c_uint64,
        ctypes.POINTER(ctypes.c_uint64)]
    lib.ecx_dbuf_alloc.argtypes = [ctypes.c_void_p, ctypes.c_size_t,
                                   ctypes.POINTER(ctypes.c_void_p)]
    lib.ecx_dbuf_free.argtypes = [ctypes.c_void_p, ctypes.c_void_p]
    lib.ecx_upload.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                               ctypes.c_void_p, ctypes.c_size_t,
                               ctypes.c_int, ctypes.c_int]
    lib.ecx_download.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                 ctypes.c_void_p, ctypes.c_size_t,
                                 ctypes.c_int, ctypes.c_int]
    lib.ecx_dbuf_fill_random.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                         ctypes.c_size_t, ctypes.c_uint64,
                                         ctypes.c_int]
    lib.ecx_encode_batch.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                     ctypes.c_long, ctypes.c_size_t,
                                     ctypes.c_int]
    lib.ecx_decode_batch.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                     ctypes.c_long, ctypes.c_size_t,
                                     ctypes.c_uint64, ctypes.c_int]
    lib.ecx_encode_delta_dev.argtypes = [ctypes.c_void_p] + [ctypes.c_void_p] * 3 + [
        ctypes.c_size_t, ctypes.c_int]
    lib.ecx_apply_delta_dev.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                        ctypes.c_int, ctypes.c_int,
                                        ctypes.c_void_p, ctypes.c_size_t,
                                        ctypes.c_int]
    lib.ecx_encode_slices.argtypes = [ctypes.c_void_p,
                                      ctypes.POINTER(ctypes.c_void_p),
                                      ctypes.POINTER(ctypes.c_size_t),
                                      ctypes.c_int, ctypes.c_int]
    lib.ecx_decode_slices.argtypes = [ctypes.c_void_p,
                                      ctypes.POINTER(ctypes.c_void_p),
                                      ctypes.POINTER(ctypes.c_size_t),
                                      ctypes.c_int, ctypes.c_uint64,
                                      ctypes.c_int]
    lib.ecx_set_matrix.argtypes = [ctypes.c_void_p, ctypes.c_void_p]
    lib.ecx_matmul_batch.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                     ctypes.c_long, ctypes.c_size_t,
                                     ctypes.POINTER(ctypes.c_int),
                                     ctypes.c_int,
                                     ctypes.POINTER(ctypes.c_int),
                                     ctypes.c_int, ctypes.c_void_p,
                                     ctypes.c_int]
    lib.ecx_shec_matrix.argtypes = [ctypes.c_int, ctypes.c_int, ctypes.c_int,
                                    ctypes.c_int, ctypes.c_void_p]
    lib.ecx_sync.argtypes = [ctypes.c_void_p, ctypes.c_int]
    lib.ecx_last_kernel_ms.argtypes = [ctypes.c_void_p, ctypes.c_int,
                                       ctypes.POINTER(ctypes.c_double)]
    lib.ecx_get_matrix.argtypes = [ctypes.c_void_p, ctypes.c_void_p]
    lib.ecx_get_matrix16.argtypes = [ctypes.c_void_p, ctypes.c_void_p]
    lib.ecx_encode_chunks_host.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                           ctypes.c_void_p, ctypes.c_size_t]
    lib.ecx_decode_chunks_host.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                           ctypes.c_uint64, ctypes.c_size_t]
    lib.ecx_encode_delta_host.argtypes = [ctypes.c_void_p] + [ctypes.c_void_p] * 3 + [
        ctypes.c_size_t]
    lib.ecx_apply_delta_host.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                         ctypes.c_int, ctypes.c_int,
                                         ctypes.c_void_p, ctypes.c_size_t]
    return lib


_cached_lib = None


def lib():
    global _cached_lib
    if _cached_lib is None:
        _cached_lib = _lib()
    return _cached_lib


def version():
    return lib().ecx_version().decode()


def device_count():
    return lib().ecx_device_count()


def shec_matrix(k, m, c, single=False):
    """SHEC shingled coding matrix (host-side; no GPU needed)."""
    out = np.zeros((m, k), dtype=np.uint8)
    _ck(lib().ecx_shec_matrix(k, m, c, int(single),
                              out.ctypes.data_as(ctypes.c_void_p)),
        "ecx_shec_matrix")
    return out


def _ck(r, what):
    if isinstance(r, int) and r < 0:
        raise EcError(f"{what}: {_ERR.get(r, r)}")
    return r


def _ptr_array(bufs):
    arr = (ctypes.c_void_p * len(bufs))()
    for i, b in enumerate(bufs):
        arr[i] = None if b is None else b.ctypes.data_as(ctypes.c_void_p).value
    return arr


class EcContext:
    """One (k, m, technique) codec bound to one GPU, mirroring a plugin
    instance after init()/prepare() (ErasureCodeIsa.cc:637-697)."""

    def __init__(self, k, m, technique="reed_sol_van", device=0,
                 n_streams=2, packetsize=2048, w=8):
        t = TECHNIQUES[technique] if isinstance(technique, str) else technique
        if t == T_RS_VAN_JERASURE_W16:
            w = 16
        self._h = ctypes.c_void_p()
        self.k, self.m, self.technique = k, m, technique
        self.packetsize = packetsize
        _ck(lib().ecx_create2(k, m, t, w, packetsize, device, n_streams,
                              ctypes.byref(self._h)), "ecx_create2")

    def close(self):
        if self._h:
            lib().ecx_destroy(self._h)
            self._h = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass

    def matrix(self):
        if TECHNIQUES.get(self.technique) == T_RS_VAN_JERASURE_W16:
            a = np.zeros((self.k + self.m, self.k), dtype=np.uint16)
            _ck(lib().ecx_get_matrix16(
                self._h, a.ctypes.data_as(ctypes.c_void_p)),
                "ecx_get_matrix16")
            return a
        a = np.zeros((self.k + self.m, self.k), dtype=np.uint8)
        _ck(lib().ecx_get_matrix(self._h, a.ctypes.data_as(ctypes.c_void_p)),
            "ecx_get_matrix")
        return a

    def chunk_size(self, stripe_width):
        return lib().ecx_chunk_size(self._h, stripe_width)

    def minimum_to_decode(self, want_mask, avail_mask):
        out = ctypes.c_uint64()
        _ck(lib().ecx_minimum_to_decode(self._h, want_mask, avail_mask,
                                        ctypes.byref(out)),
            "ecx_minimum_to_decode")
        return out.value

    # ---- device-resident batch API ----
    def dbuf_alloc(self, nbytes):
        p = ctypes.c_void_p()
        _ck(lib().ecx_dbuf_alloc(self._h, nbytes, ctypes.byref(p)),
            "ecx_dbuf_alloc")
        return p

    def dbuf_free(self, dptr):
        _ck(lib().ecx_dbuf_free(self._h, dptr), "ecx_dbuf_free")

    def upload(self, dptr, arr, slot=0, blocking=True):
        _ck(lib().ecx_upload(self._h, dptr,
                             arr.ctypes.data_as(ctypes.c_void_p), arr.nbytes,
                             slot, int(blocking)), "ecx_upload")

    def download(self, arr, dptr, slot=0, blocking=True):
        _ck(lib().ecx_download(self._h,
                               arr.ctypes.data_as(ctypes.c_void_p), dptr,
                               arr.nbytes, slot, int(blocking)),
            "ecx_download")

    def fill_random(self, dptr, nbytes, seed, slot=0):
        _ck(lib().ecx_dbuf_fill_random(self._h, dptr, nbytes, seed, slot),
            "ecx_dbuf_fill_random")

    def encode_batch(self, dptr, n_stripes, chunk_bytes, slot=0):
        _ck(lib().ecx_encode_batch(self._h, dptr, n_stripes, chunk_bytes,
                                   slot), "ecx_encode_batch")

    def decode_batch(self, dptr, n_stripes, chunk_bytes, present_mask, slot=0):
        _ck(lib().ecx_decode_batch(self._h, dptr, n_stripes, chunk_bytes,
                                   present_mask, slot), "ecx_decode_batch")

    def _slice_args(self, chunk_ptrs, sizes):
        n = len(sizes)
        assert len(chunk_ptrs) == n * (self.k + self.m)
        arr = (ctypes.c_void_p * len(chunk_ptrs))()
        for i, p in enumerate(chunk_ptrs):
            arr[i] = None if p in (None, 0) else int(p)
        sz = (ctypes.c_size_t * n)(*sizes)
        return arr, sz, n

    def encode_slices(self, chunk_ptrs, sizes, slot=0):
        """Variable-size slice batch: chunk_ptrs is a flat list
        of n*(k+m) device addresses (None => zeros data chunk); sizes is
        the per-slice byte length (multiples of 16). One kernel launch."""
        arr, sz, n = self._slice_args(chunk_ptrs, sizes)
        _ck(lib().ecx_encode_slices(self._h, arr, sz, n, slot),
            "ecx_encode_slices")

    def decode_slices(self, chunk_ptrs, sizes, present_mask, slot=0):
        arr, sz, n = self._slice_args(chunk_ptrs, sizes)
        _ck(lib().ecx_decode_slices(self._h, arr, sz, n, present_mask,
                                    slot), "ecx_decode_slices")

    def matmul_batch(self, dptr, n_stripes, chunk_bytes, src_ids, out_ids,
                     rows, slot=0):
        """Generic device-batch GF matmul over chunk ids (LRC layer
        composition primitive)."""
        rows = np.ascontiguousarray(rows, dtype=np.uint8)
        assert rows.shape == (len(out_ids), len(src_ids))
        sa = (ctypes.c_int * len(src_ids))(*src_ids)
        oa = (ctypes.c_int * len(out_ids))(*out_ids)
        _ck(lib().ecx_matmul_batch(self._h, dptr, n_stripes, chunk_bytes,
                                   sa, len(src_ids), oa, len(out_ids),
                                   rows.ctypes.data_as(ctypes.c_void_p),
                                   slot), "ecx_matmul_batch")

    def set_matrix(self, coding_rows):
        """Replace the coding rows (custom-matrix codecs, e.g. SHEC)."""
        rows = np.ascontiguousarray(coding_rows, dtype=np.uint8)
        assert rows.shape == (self.m, self.k)
        _ck(lib().ecx_set_matrix(self._h,
                                 rows.ctypes.data_as(ctypes.c_void_p)),
            "ecx_set_matrix")

    def sync(self, slot=0):
        _ck(lib().ecx_sync(self._h, slot), "ecx_sync")

    def last_kernel_ms(self, slot=0):
        ms = ctypes.c_double()
        _ck(lib().ecx_last_kernel_ms(self._h, slot, ctypes.byref(ms)),
            "ecx_last_kernel_ms")
        return ms.value

    # ---- host-pointer (plugin) path ----
    def encode_chunks(self, data, chunk_bytes=None, out=None):
        """data: list of k uint8 arrays (None => zeros chunk). Returns m
        parity arrays. Mirrors encode_chunks marshalling
        (ErasureCodeJerasure.cc:121-164)."""
        sizes = {d.nbytes for d in data if d is not None}
        if chunk_bytes is not None:
            sizes.add(chunk_bytes)
        assert len(sizes) == 1, "equal-length chunks required (pass " \
            "chunk_bytes when every chunk is a zeros sentinel)"
        n = sizes.pop()
        if out is None:
            out = [np.zeros(n, dtype=np.uint8) for _ in range(self.m)]
        _ck(lib().ecx_encode_chunks_host(self._h, _ptr_array(data),
                                         _ptr_array(out), n),
            "ecx_encode_chunks_host")
        return out

    def decode_chunks(self, chunks, present):
        """chunks: list of k+m uint8 arrays; erased entries (present[i]
        false) are filled in place. Mirrors decode_chunks
        (ErasureCodeIsa.cc:167-243)."""
        n = chunks[0].nbytes
        mask = 0
        for i, p in enumerate(present):
            if p:
                mask |= 1 << i
        _ck(lib().ecx_decode_chunks_host(self._h, _ptr_array(chunks), mask, n),
            "ecx_decode_chunks_host")
        return chunks

    def encode_delta(self, old, new):
        delta = np.zeros_like(old)
        _ck(lib().ecx_encode_delta_host(
            self._h, old.ctypes.data_as(ctypes.c_void_p),
            new.ctypes.data_as(ctypes.c_void_p),
            delta.ctypes.data_as(ctypes.c_void_p), old.nbytes),
            "ecx_encode_delta_host")
        return delta

    def apply_delta(self, delta, data_shard, coding_shard, parity):
        _ck(lib().ecx_apply_delta_host(
            self._h, delta.ctypes.data_as(ctypes.c_void_p), data_shard,
            coding_shard, parity.ctypes.data_as(ctypes.c_void_p),
            delta.nbytes), "ecx_apply_delta_host")
        return parity
