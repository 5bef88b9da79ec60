"""oracle — ctypes wrapper over the CPU oracle (TEST INFRASTRUCTURE ONLY).

Scope (see ec_ref.h): this package is the parity checker for the MI355X EC
backend. It may be imported only from tests/, __graft_entry__.smoke() (as the
checker) and bench.py's cpu_baseline leg. It is NOT the product path.
"""
import ctypes
import os
import subprocess

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))

# Technique ids (must match ec_ref.h / include/ec_mi355x.h)
T_RS_VAN_ISA = 0
T_CAUCHY_ISA = 1
T_RS_VAN_JERASURE = 2
T_CAUCHY_ORIG_JERASURE = 3  # bitmatrix/packet layout: use bitmatrix_* fns
T_CAUCHY_GOOD_JERASURE = 5  # bitmatrix/packet layout (orig + improve pass)
TECHNIQUES = {
    "reed_sol_van": T_RS_VAN_ISA,
    "cauchy": T_CAUCHY_ISA,
    "jerasure_reed_sol_van": T_RS_VAN_JERASURE,
    "cauchy_orig": T_CAUCHY_ORIG_JERASURE,
    "cauchy_good": T_CAUCHY_GOOD_JERASURE,
}
BITMATRIX_TECHNIQUES = (T_CAUCHY_ORIG_JERASURE, T_CAUCHY_GOOD_JERASURE)


def _build_if_needed():
    for so, srcs in (("libec_ref.so", ["ec_ref.c"]),
                     ("libec_cpu.so", ["ec_cpu.c", "ec_ref.c"])):
        sop = os.path.join(_DIR, so)
        if not os.path.exists(sop) or any(
                os.path.getmtime(os.path.join(_DIR, s)) > os.path.getmtime(sop)
                for s in srcs):
            subprocess.run(["make", "-C", _DIR], check=True,
                           capture_output=True)
            break


def _load(name):
    _build_if_needed()
    return ctypes.CDLL(os.path.join(_DIR, name))


_ref = _load("libec_ref.so")
_cpu = _load("libec_cpu.so")

_ref.ecref_gf_init()
_ref.ecref_gf_mul.restype = ctypes.c_uint8
_ref.ecref_gf_mul.argtypes = [ctypes.c_uint8, ctypes.c_uint8]
_ref.ecref_gf_inv.restype = ctypes.c_uint8
_ref.ecref_gf_inv.argtypes = [ctypes.c_uint8]
_ref.ecref_gf_log_table.restype = ctypes.POINTER(ctypes.c_uint8 * 256)
_ref.ecref_gf_exp_table.restype = ctypes.POINTER(ctypes.c_uint8 * 256)
_ref.ecref_matrix.restype = ctypes.c_int
_ref.ecref_matrix.argtypes = [ctypes.c_int, ctypes.c_void_p, ctypes.c_int,
                              ctypes.c_int]
_ref.ecref_decode.restype = ctypes.c_int
_ref.ecref_chunk_size_isa.restype = ctypes.c_uint
_ref.ecref_chunk_size_isa.argtypes = [ctypes.c_int, ctypes.c_uint]
_ref.ecref_chunk_size_jerasure.restype = ctypes.c_uint
_ref.ecref_chunk_size_jerasure.argtypes = [ctypes.c_int, ctypes.c_int,
                                           ctypes.c_uint]
_ref.ecref_matrix_cauchy_orig_jerasure.restype = ctypes.c_int
_ref.ecref_matrix_cauchy_good_jerasure.restype = ctypes.c_int
_ref.ecref_cauchy_n_ones.restype = ctypes.c_int
_ref.ecref_bitmatrix_encode.restype = ctypes.c_int
_ref.ecref_bitmatrix_decode.restype = ctypes.c_int
_ref.ecref_matrix_rs_vandermonde_jerasure_w16.restype = ctypes.c_int
_ref.ecref_decode16.restype = ctypes.c_int
_cpu.eccpu_encode_batch.restype = ctypes.c_int
_cpu.eccpu_decode_batch.restype = ctypes.c_int
_cpu.eccpu_threads.restype = ctypes.c_int


def gf_mul(a, b):
    return _ref.ecref_gf_mul(a, b)


def gf_inv(a):
    return _ref.ecref_gf_inv(a)


def gf_log_table():
    return np.frombuffer(bytes(_ref.ecref_gf_log_table().contents), np.uint8)


def gf_exp_table():
    return np.frombuffer(bytes(_ref.ecref_gf_exp_table().contents), np.uint8)


def matrix(technique, k, m):
    """Full (k+m) x k generator (identity top), as uint8 ndarray."""
    t = TECHNIQUES[technique] if isinstance(technique, str) else technique
    if t in BITMATRIX_TECHNIQUES:
        a = np.zeros((k + m, k), dtype=np.uint8)
        a[:k] = np.eye(k, dtype=np.uint8)
        a[k:] = (cauchy_orig_matrix(k, m) if t == T_CAUCHY_ORIG_JERASURE
                 else cauchy_good_matrix(k, m))
        return a
    a = np.zeros((k + m, k), dtype=np.uint8)
    r = _ref.ecref_matrix(t, a.ctypes.data_as(ctypes.c_void_p), k, m)
    if r != 0:
        raise ValueError(f"ecref_matrix failed: {r}")
    return a


def _ptr_array(bufs):
    arr = (ctypes.c_void_p * len(bufs))()
    for i, b in enumerate(bufs):
        arr[i] = None if b is None else b.ctypes.data_as(ctypes.c_void_p).value
    return arr


def encode(technique, k, m, data, chunk_bytes=None):
    """data: list of k uint8 arrays (or None for zeros). Returns list of m
    parity arrays. Scalar oracle (ecref_encode)."""
    t = TECHNIQUES[technique] if isinstance(technique, str) else technique
    if t in BITMATRIX_TECHNIQUES:
        raise ValueError("bitmatrix techniques use the packet layout: "
                         "call bitmatrix_encode/bitmatrix_decode")
    lens = {d.nbytes for d in data if d is not None}
    assert len(lens) == 1 or (not lens and chunk_bytes)
    length = lens.pop() if lens else chunk_bytes
    gen = matrix(technique, k, m)
    rows = np.ascontiguousarray(gen[k:])
    parity = [np.zeros(length, dtype=np.uint8) for _ in range(m)]
    _ref.ecref_encode(k, m, rows.ctypes.data_as(ctypes.c_void_p),
                      _ptr_array(data), _ptr_array(parity),
                      ctypes.c_size_t(length))
    return parity


def encode_with_rows(rows, data, chunk_bytes=None):
    """Encode with an explicit m x k coding matrix (e.g. a SHEC shingled
    matrix) — the checker for custom-matrix codecs."""
    m, k = rows.shape
    lens = {d.nbytes for d in data if d is not None}
    assert len(lens) == 1 or (not lens and chunk_bytes)
    length = lens.pop() if lens else chunk_bytes
    rows = np.ascontiguousarray(rows)
    parity = [np.zeros(length, dtype=np.uint8) for _ in range(m)]
    _ref.ecref_encode(k, m, rows.ctypes.data_as(ctypes.c_void_p),
                      _ptr_array(data), _ptr_array(parity),
                      ctypes.c_size_t(length))
    return parity


def decode(technique, k, m, chunks, present):
    """chunks: list of k+m uint8 arrays (erased entries are overwritten in
    place with the reconstruction). present: list/array of 0/1 flags."""
    t = TECHNIQUES[technique] if isinstance(technique, str) else technique
    length = chunks[0].nbytes
    pres = np.asarray(present, dtype=np.uint8)
    r = _ref.ecref_decode(t, k, m, _ptr_array(chunks),
                          pres.ctypes.data_as(ctypes.c_void_p),
                          ctypes.c_size_t(length))
    if r != 0:
        raise ValueError(f"ecref_decode failed: {r}")
    return chunks


def cauchy_orig_matrix(k, m):
    """jerasure cauchy_orig coding matrix (m x k): inv(i XOR (m+j))."""
    a = np.zeros((m, k), dtype=np.uint8)
    r = _ref.ecref_matrix_cauchy_orig_jerasure(
        a.ctypes.data_as(ctypes.c_void_p), k, m)
    if r != 0:
        raise ValueError(f"cauchy_orig matrix failed: {r}")
    return a


def cauchy_good_matrix(k, m):
    """jerasure cauchy_good coding matrix (m x k): cauchy_orig + the
    n_ones-minimising improve pass (cauchy.c general branch; m==2 would
    need jerasure's unsourceable cbest tables and raises)."""
    a = np.zeros((m, k), dtype=np.uint8)
    r = _ref.ecref_matrix_cauchy_good_jerasure(
        a.ctypes.data_as(ctypes.c_void_p), k, m)
    if r != 0:
        raise ValueError(f"cauchy_good matrix failed: {r}")
    return a


def cauchy_n_ones(e):
    """Ones in the 8x8 companion bitmatrix of e (cauchy.c cauchy_n_ones)."""
    return int(_ref.ecref_cauchy_n_ones(ctypes.c_uint8(e)))


def _bit_coding_matrix(technique, k, m):
    t = TECHNIQUES[technique] if isinstance(technique, str) else technique
    if t == T_CAUCHY_GOOD_JERASURE:
        return cauchy_good_matrix(k, m)
    return cauchy_orig_matrix(k, m)


def bitmatrix(coding, w=8):
    m, k = coding.shape
    bm = np.zeros((m * w, k * w), dtype=np.uint8)
    _ref.ecref_matrix_to_bitmatrix(
        np.ascontiguousarray(coding).ctypes.data_as(ctypes.c_void_p), k, m,
        w, bm.ctypes.data_as(ctypes.c_void_p))
    return bm


def bitmatrix_encode(k, m, data, packetsize, w=8,
                     technique=T_CAUCHY_ORIG_JERASURE):
    """jerasure bitmatrix/packet-layout encode (cauchy_orig by default,
    technique="cauchy_good" for the improved matrix); data entries may be
    None for zeros chunks."""
    lens = {d.nbytes for d in data if d is not None}
    assert len(lens) == 1
    size = lens.pop()
    bm = bitmatrix(_bit_coding_matrix(technique, k, m), w)
    parity = [np.zeros(size, dtype=np.uint8) for _ in range(m)]
    r = _ref.ecref_bitmatrix_encode(
        k, m, w, bm.ctypes.data_as(ctypes.c_void_p), _ptr_array(data),
        _ptr_array(parity), ctypes.c_size_t(size), packetsize)
    if r != 0:
        raise ValueError(f"bitmatrix_encode failed: {r}")
    return parity


def bitmatrix_decode(k, m, chunks, present, packetsize, w=8,
                     technique=T_CAUCHY_ORIG_JERASURE):
    bm = bitmatrix(_bit_coding_matrix(technique, k, m), w)
    pres = np.asarray(present, dtype=np.uint8)
    r = _ref.ecref_bitmatrix_decode(
        k, m, w, bm.ctypes.data_as(ctypes.c_void_p), _ptr_array(chunks),
        pres.ctypes.data_as(ctypes.c_void_p),
        ctypes.c_size_t(chunks[0].nbytes), packetsize)
    if r != 0:
        raise ValueError(f"bitmatrix_decode failed: {r}")
    return chunks


def gf16_mul(a, b):
    _ref.ecref_gf16_init()
    _ref.ecref_gf16_mul.restype = ctypes.c_uint16
    return _ref.ecref_gf16_mul(ctypes.c_uint16(a), ctypes.c_uint16(b))


def matrix_w16(k, m):
    _ref.ecref_gf16_init()
    a = np.zeros((k + m, k), dtype=np.uint16)
    r = _ref.ecref_matrix_rs_vandermonde_jerasure_w16(
        a.ctypes.data_as(ctypes.c_void_p), k, m)
    if r != 0:
        raise ValueError(f"matrix_w16 failed: {r}")
    return a


def encode_w16(k, m, data, chunk_bytes=None):
    """jerasure reed_sol_van w=16 (GF(2^16), u16 LE symbols)."""
    lens = {d.nbytes for d in data if d is not None}
    assert len(lens) == 1 or (not lens and chunk_bytes)
    length = lens.pop() if lens else chunk_bytes
    rows = np.ascontiguousarray(matrix_w16(k, m)[k:])
    parity = [np.zeros(length, dtype=np.uint8) for _ in range(m)]
    _ref.ecref_encode16(k, m, rows.ctypes.data_as(ctypes.c_void_p),
                        _ptr_array(data), _ptr_array(parity),
                        ctypes.c_size_t(length))
    return parity


def decode_w16(k, m, chunks, present):
    pres = np.asarray(present, dtype=np.uint8)
    r = _ref.ecref_decode16(_ptr_array(chunks),
                            pres.ctypes.data_as(ctypes.c_void_p), k, m,
                            ctypes.c_size_t(chunks[0].nbytes))
    if r != 0:
        raise ValueError(f"decode_w16 failed: {r}")
    return chunks


def xor_region(a, b):
    out = np.zeros_like(a)
    _ref.ecref_xor_region(a.ctypes.data_as(ctypes.c_void_p),
                          b.ctypes.data_as(ctypes.c_void_p),
                          out.ctypes.data_as(ctypes.c_void_p),
                          ctypes.c_size_t(a.nbytes))
    return out


def region_mul_xor(coeff, delta, parity):
    _ref.ecref_region_mul_xor(ctypes.c_uint8(coeff),
                              delta.ctypes.data_as(ctypes.c_void_p),
                              parity.ctypes.data_as(ctypes.c_void_p),
                              ctypes.c_size_t(delta.nbytes))
    return parity


def chunk_size(technique, k, stripe_width, w=8):
    t = TECHNIQUES[technique] if isinstance(technique, str) else technique
    if t == T_RS_VAN_JERASURE:
        return _ref.ecref_chunk_size_jerasure(k, w, stripe_width)
    return _ref.ecref_chunk_size_isa(k, stripe_width)


# ---- fast CPU-baseline batch ops (AVX2/OpenMP; kind="port" in bench) ----

def cpu_first_touch(batch):
    """NUMA-spread the batch's pages (parallel first-touch) BEFORE filling
    content — a single-thread numpy fill places every page on one node
    and the OpenMP encode then starves the other socket(s)."""
    _cpu.eccpu_first_touch(batch.ctypes.data_as(ctypes.c_void_p),
                           ctypes.c_size_t(batch.nbytes))


def cpu_threads():
    return _cpu.eccpu_threads()


def cpu_encode_batch(technique, k, m, batch, n_stripes, chunk_bytes):
    """batch: uint8 ndarray of n_stripes*(k+m)*chunk_bytes laid out like the
    GPU device buffer. Encodes parity in place."""
    t = TECHNIQUES[technique] if isinstance(technique, str) else technique
    r = _cpu.eccpu_encode_batch(t, k, m,
                                batch.ctypes.data_as(ctypes.c_void_p),
                                ctypes.c_long(n_stripes),
                                ctypes.c_size_t(chunk_bytes))
    if r != 0:
        raise ValueError(f"eccpu_encode_batch failed: {r}")


def cpu_decode_batch(technique, k, m, batch, present, n_stripes, chunk_bytes):
    t = TECHNIQUES[technique] if isinstance(technique, str) else technique
    pres = np.asarray(present, dtype=np.uint8)
    r = _cpu.eccpu_decode_batch(t, k, m,
                                batch.ctypes.data_as(ctypes.c_void_p),
                                pres.ctypes.data_as(ctypes.c_void_p),
                                ctypes.c_long(n_stripes),
                                ctypes.c_size_t(chunk_bytes))
    if r != 0:
        raise ValueError(f"eccpu_decode_batch failed: {r}")
