"""oracle/pyoracle.py — ctypes wrapper over liboracle.so.

TEST INFRASTRUCTURE ONLY (see oracle.h).  Importable from tests/,
__graft_entry__.smoke() and bench.py's cpu_baseline leg exclusively.
"""
import ctypes
import os
import subprocess

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_LIB = os.path.join(_DIR, "liboracle.so")


def build():
    subprocess.check_call(["make", "-C", _DIR, "-s"])


_lib = None


def lib():
    global _lib
    if _lib is None:
        if not os.path.exists(_LIB):
            build()
        L = ctypes.CDLL(_LIB)
        u8p = ctypes.POINTER(ctypes.c_uint8)
        L.orc_gf_tables.argtypes = [u8p, ctypes.c_size_t]
        L.orc_gf_mul.restype = ctypes.c_uint8
        L.orc_gf_mul.argtypes = [ctypes.c_uint8, ctypes.c_uint8]
        L.orc_gf_exp.restype = ctypes.c_uint8
        L.orc_gf_exp.argtypes = [ctypes.c_uint8, ctypes.c_int]
        L.orc_vandermonde.argtypes = [ctypes.c_int, ctypes.c_int, u8p]
        L.orc_invert_matrix.argtypes = [u8p, ctypes.c_int, u8p]
        L.orc_build_matrix.argtypes = [ctypes.c_int, ctypes.c_int, u8p]
        L.orc_crc32.restype = ctypes.c_uint32
        L.orc_crc32.argtypes = [ctypes.c_uint32, u8p, ctypes.c_size_t]
        L.orc_crc32_combine.restype = ctypes.c_uint32
        L.orc_crc32_combine.argtypes = [ctypes.c_uint32, ctypes.c_uint32, ctypes.c_int64]
        L.orc_crc32_shift.restype = ctypes.c_uint32
        L.orc_crc32_shift.argtypes = [ctypes.c_uint32, ctypes.c_int64]
        for f in ("orc_crc32b_encode_size", "orc_crc32b_decode_size"):
            getattr(L, f).restype = ctypes.c_int64
            getattr(L, f).argtypes = [ctypes.c_int64, ctypes.c_int64]
        L.orc_crc32b_encode.restype = ctypes.c_int64
        L.orc_crc32b_encode.argtypes = [u8p, u8p, ctypes.c_int64, ctypes.c_int64]
        L.orc_crc32b_verify.restype = ctypes.c_int64
        L.orc_crc32b_verify.argtypes = [u8p, ctypes.c_int64, ctypes.c_int64]
        L.orc_crc32b_decode.restype = ctypes.c_int64
        L.orc_crc32b_decode.argtypes = [u8p, u8p, ctypes.c_int64, ctypes.c_int64]
        L.orc_partial_encode_size.restype = ctypes.c_int64
        L.orc_partial_encode_size.argtypes = [ctypes.c_int64, ctypes.c_int64,
                                              ctypes.c_int64,
                                              ctypes.POINTER(ctypes.c_int64)]
        L.orc_partial_decode_size.restype = ctypes.c_int64
        L.orc_partial_decode_size.argtypes = [ctypes.c_int64] * 4
        L.orc_sized_encode.restype = ctypes.c_int64
        L.orc_sized_encode.argtypes = [u8p, u8p, ctypes.c_int64, ctypes.c_int64]
        L.orc_sized_verify.restype = ctypes.c_int64
        L.orc_sized_verify.argtypes = [u8p, ctypes.c_int64, ctypes.c_int64, ctypes.c_int64]
        L.orc_sized_decode.restype = ctypes.c_int64
        L.orc_sized_decode.argtypes = [u8p, u8p, ctypes.c_int64, ctypes.c_int64, ctypes.c_int64]
        L.orc_shard_disk_size.restype = ctypes.c_int64
        L.orc_shard_disk_size.argtypes = [ctypes.c_int64, ctypes.c_int64]
        L.orc_shard_write.restype = ctypes.c_int64
        L.orc_shard_write.argtypes = [u8p, u8p, ctypes.c_int64, ctypes.c_int64,
                                      ctypes.c_uint64, ctypes.c_uint64]
        L.orc_shard_parse.argtypes = [u8p, ctypes.c_int64, ctypes.c_int64,
                                      ctypes.POINTER(ctypes.c_uint64),
                                      ctypes.POINTER(ctypes.c_uint64),
                                      ctypes.POINTER(ctypes.c_uint32)]
        _lib = L
    return _lib


def _ptr(a):
    return a.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8))


def _shard_ptrs(shards):
    arr = (ctypes.POINTER(ctypes.c_uint8) * len(shards))()
    for i, s in enumerate(shards):
        assert s.dtype == np.uint8 and s.flags["C_CONTIGUOUS"]
        arr[i] = _ptr(s)
    return arr


def gf_mul(a, b):
    return lib().orc_gf_mul(a, b)


def gf_exp(a, n):
    return lib().orc_gf_exp(a, n)


def gf_tables():
    out = np.zeros(74494, dtype=np.uint8)
    lib().orc_gf_tables(_ptr(out), out.size)
    return out


def build_matrix(k, total):
    out = np.zeros((total, k), dtype=np.uint8)
    rc = lib().orc_build_matrix(k, total, _ptr(out))
    assert rc == 0, rc
    return out


def invert_matrix(m):
    n = m.shape[0]
    out = np.zeros((n, n), dtype=np.uint8)
    rc = lib().orc_invert_matrix(_ptr(np.ascontiguousarray(m)), n, out.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)))
    if rc != 0:
        raise ValueError("singular" if rc == -8 else "err %d" % rc)
    return out


def rs_encode(k, m, shards):
    rc = lib().orc_rs_encode(k, m, _shard_ptrs(shards), len(shards[0]))
    assert rc == 0, rc


def rs_verify(k, m, shards):
    rc = lib().orc_rs_verify(k, m, _shard_ptrs(shards), len(shards[0]))
    assert rc >= 0, rc
    return bool(rc)


def rs_reconstruct(k, m, shards, present, data_only=False):
    pres = np.asarray(present, dtype=np.uint8)
    rc = lib().orc_rs_reconstruct(k, m, _shard_ptrs(shards), len(shards[0]), _ptr(pres), int(data_only))
    return rc


def rs_decode_matrix(k, m, present):
    pres = np.asarray(present, dtype=np.uint8)
    rows = np.zeros((k, k), dtype=np.uint8)
    valid = np.zeros(k, dtype=np.int32)
    rc = lib().orc_rs_decode_matrix(k, m, _ptr(pres), _ptr(rows), valid.ctypes.data_as(ctypes.POINTER(ctypes.c_int)))
    assert rc == 0, rc
    return rows, valid


def lrc_encode(n, mm, l, az, shards):
    rc = lib().orc_lrc_encode(n, mm, l, az, _shard_ptrs(shards), len(shards[0]))
    assert rc == 0, rc


def lrc_reconstruct(n, mm, l, az, shards, present, data_only=False):
    pres = np.asarray(present, dtype=np.uint8)
    return lib().orc_lrc_reconstruct(n, mm, l, az, _shard_ptrs(shards), len(shards[0]), _ptr(pres), int(data_only))


def lrc_local_stripe(n, mm, l, az, az_idx):
    cnt = (n + mm + l) // az
    out = np.zeros(cnt, dtype=np.int32)
    rc = lib().orc_lrc_local_stripe(n, mm, l, az, az_idx, out.ctypes.data_as(ctypes.POINTER(ctypes.c_int)))
    assert rc == cnt, rc
    return out.tolist()


def buffer_sizes(n, mm, l, min_shard_size, data_size):
    ss = ctypes.c_longlong()
    eds = ctypes.c_longlong()
    es = ctypes.c_longlong()
    rc = lib().orc_buffer_sizes(n, mm, l, min_shard_size, data_size, ctypes.byref(ss), ctypes.byref(eds), ctypes.byref(es))
    if rc != 0:
        raise ValueError(rc)
    return ss.value, eds.value, es.value


def crc32(data, crc=0):
    a = np.ascontiguousarray(np.frombuffer(bytes(data), dtype=np.uint8)) if not isinstance(data, np.ndarray) else data
    return lib().orc_crc32(crc, _ptr(a), a.size)


def crc32_combine(c1, c2, len2):
    return lib().orc_crc32_combine(c1, c2, len2)


def crc32_shift(c, n):
    return lib().orc_crc32_shift(c, n)


def crc32b_encode_size(size, block_len=65536):
    return lib().orc_crc32b_encode_size(size, block_len)


def crc32b_decode_size(size, block_len=65536):
    return lib().orc_crc32b_decode_size(size, block_len)


def crc32b_encode(src, block_len=65536):
    n = src.size
    out = np.zeros(crc32b_encode_size(n, block_len), dtype=np.uint8)
    w = lib().orc_crc32b_encode(_ptr(out), _ptr(src), n, block_len)
    assert w == out.size, (w, out.size)
    return out


def crc32b_verify(framed, block_len=65536):
    return lib().orc_crc32b_verify(_ptr(framed), framed.size, block_len)


def crc32b_decode(framed, block_len=65536):
    out = np.zeros(crc32b_decode_size(framed.size, block_len), dtype=np.uint8)
    w = lib().orc_crc32b_decode(_ptr(out), _ptr(framed), framed.size, block_len)
    if w < 0:
        raise ValueError("crc mismatch" if w == -9 else "err %d" % w)
    return out


def partial_encode_size(actual, stable=0, block_len=65536):
    t = ctypes.c_int64()
    total = lib().orc_partial_encode_size(actual, stable, block_len, ctypes.byref(t))
    return total, t.value


def partial_decode_size(total, tail, stable=0, block_len=65536):
    return lib().orc_partial_decode_size(total, tail, stable, block_len)


def sized_encode(src, block_len=65536):
    total, tail = partial_encode_size(src.size, 0, block_len)
    out = np.zeros(total, dtype=np.uint8)
    w = lib().orc_sized_encode(_ptr(out), _ptr(src), src.size, block_len)
    assert w == total, (w, total)
    return out, tail


def sized_verify(framed, tail, block_len=65536):
    return lib().orc_sized_verify(_ptr(framed), framed.size, tail, block_len)


def sized_decode(framed, tail, block_len=65536):
    n = partial_decode_size(framed.size, tail, 0, block_len)
    out = np.zeros(n, dtype=np.uint8)
    w = lib().orc_sized_decode(_ptr(out), _ptr(framed), framed.size, tail, block_len)
    if w < 0:
        raise ValueError("err %d" % w)
    return out


def shard_disk_size(size, block_len=65536):
    return lib().orc_shard_disk_size(size, block_len)


def shard_write(src, bid, vuid, block_len=65536):
    n = src.size
    out = np.zeros(shard_disk_size(n, block_len), dtype=np.uint8)
    w = lib().orc_shard_write(_ptr(out), _ptr(src), n, block_len, bid, vuid)
    assert w == out.size, (w, out.size)
    return out


def shard_parse(img, block_len=65536):
    bid = ctypes.c_uint64()
    vuid = ctypes.c_uint64()
    size = ctypes.c_uint32()
    rc = lib().orc_shard_parse(_ptr(img), img.size, block_len,
                               ctypes.byref(bid), ctypes.byref(vuid),
                               ctypes.byref(size))
    if rc != 0:
        raise ValueError("shard parse err %d" % rc)
    return bid.value, vuid.value, size.value


def rs_encode_mt(k, m, stripes, nthreads=0):
    """stripes: list of lists of shards (np.uint8).  Baseline timing path."""
    flat = [s for st in stripes for s in st]
    rc = lib().orc_rs_encode_mt(k, m, _shard_ptrs(flat), len(flat[0]), len(stripes), nthreads)
    assert rc == 0, rc


def threads_avail():
    return lib().orc_threads_avail()
