"""cubefs_amd.runtime — ctypes binding of the product C ABI (include/gfrs.h).

Loads cubefs_amd/libgfrs.so (the hand-written HIP/CDNA4 engine).  There is
deliberately NO CPU fallback: if the library is missing or no GPU is
visible, compute calls raise.  PyTorch is used only for device memory and
streams (plumbing, not the compute path).
"""
import ctypes
import os

_DIR = os.path.dirname(os.path.abspath(__file__))
_LIB_PATH = os.path.join(_DIR, "libgfrs.so")

MEM_DEVICE = 0
MEM_HOST = 1

ERRORS = {
    -1: "ErrInvalidCodeMode",
    -2: "ErrTooFewShards",
    -3: "ErrShardSize",
    -4: "ErrShardNoData",
    -5: "ErrVerify",
    -6: "ErrInvalidShards",
    -7: "ErrShortData",
    -8: "ErrSingular",
    -9: "ErrMismatchedCrc",
    -10: "ErrInvalidBlock",
    -11: "ErrReadOnClosed",
    -100: "ErrHIP",
    -101: "ErrNoGPU",
    -102: "ErrNoMem",
    -103: "ErrUnsupported",
}


class GfrsError(RuntimeError):
    def __init__(self, code, detail=""):
        self.code = code
        name = ERRORS.get(code, "Err%d" % code)
        super().__init__("%s%s" % (name, (": " + detail) if detail else ""))


class Tactic(ctypes.Structure):
    _fields_ = [
        ("n", ctypes.c_int32),
        ("m", ctypes.c_int32),
        ("l", ctypes.c_int32),
        ("az_count", ctypes.c_int32),
        ("put_quorum", ctypes.c_int32),
        ("get_quorum", ctypes.c_int32),
        ("min_shard_size", ctypes.c_int32),
    ]


_lib = None


def lib():
    global _lib
    if _lib is None:
        if not os.path.exists(_LIB_PATH):
            raise ImportError(
                "cubefs_amd/libgfrs.so not built — run `make -C cubefs_amd` "
                "(or __graft_entry__.build()).  The gfrs engine has no CPU "
                "fallback by design.")
        # torch's wheel bundles its own ROCm runtime (unversioned
        # libamdhip64.so / libhsa-runtime64.so sonames), while libgfrs
        # links /opt/rocm's libamdhip64.so.7 — two HSA stacks coexist in
        # the process and only the "torch's stack initializes first"
        # order works (the second stack to init still enumerates the
        # device; the other order leaves gfrs with hipGetDeviceCount==0).
        # Enforce that order here so it doesn't depend on the caller's
        # import sequence.
        try:
            import torch
            if torch.cuda.is_available():
                torch.cuda.init()
        except ImportError:
            pass  # pure-ctypes consumers (no torch) manage init order
        L = ctypes.CDLL(_LIB_PATH)
        vp, vpp = ctypes.c_void_p, ctypes.POINTER(ctypes.c_void_p)
        i32p = ctypes.POINTER(ctypes.c_int32)
        i64, i64p = ctypes.c_int64, ctypes.POINTER(ctypes.c_int64)
        u64p = ctypes.POINTER(ctypes.c_uint64)

        L.gfrs_last_error.restype = ctypes.c_char_p
        L.gfrs_version.restype = ctypes.c_char_p
        L.gfrs_create.restype = vp
        L.gfrs_create.argtypes = [ctypes.POINTER(Tactic), ctypes.c_int]
        L.gfrs_destroy.argtypes = [vp]
        L.gfrs_set_stream.argtypes = [vp, vp]
        L.gfrs_synchronize.argtypes = [vp]
        L.gfrs_encode.argtypes = [vp, vpp, ctypes.c_size_t, ctypes.c_int, ctypes.c_int]
        L.gfrs_verify.argtypes = [vp, vpp, ctypes.c_size_t, ctypes.c_int, ctypes.c_int,
                                  ctypes.POINTER(ctypes.c_int)]
        L.gfrs_reconstruct.argtypes = [vp, vpp, ctypes.c_size_t, ctypes.c_int,
                                       ctypes.c_int, i32p, ctypes.c_int, ctypes.c_int]
        L.gfrs_encode_batch.argtypes = [vp, vp, ctypes.c_size_t, ctypes.c_size_t, ctypes.c_int]
        L.gfrs_verify_batch.argtypes = [vp, vp, ctypes.c_size_t, ctypes.c_size_t,
                                        ctypes.c_int, u64p]
        L.gfrs_reconstruct_batch.argtypes = [vp, vp, ctypes.c_size_t, ctypes.c_size_t,
                                             ctypes.c_int, i32p, ctypes.c_int, ctypes.c_int]
        L.gfrs_crc32_host.restype = ctypes.c_uint32
        L.gfrs_crc32_host.argtypes = [ctypes.c_uint32, vp, i64]
        L.gfrs_crc32b_encode_size.restype = i64
        L.gfrs_crc32b_encode_size.argtypes = [i64, i64]
        L.gfrs_crc32b_decode_size.restype = i64
        L.gfrs_crc32b_decode_size.argtypes = [i64, i64]
        L.gfrs_crc32b_encode.restype = i64
        L.gfrs_crc32b_encode.argtypes = [vp, vp, vp, i64, i64]
        L.gfrs_crc32b_verify.argtypes = [vp, vp, i64, i64, i64p]
        L.gfrs_crc32b_decode.restype = i64
        L.gfrs_crc32b_decode.argtypes = [vp, vp, vp, i64, i64]
        L.gfrs_crc32b_encode_batch.argtypes = [vp, vp, ctypes.c_size_t, vp,
                                               ctypes.c_size_t, i64, i64, ctypes.c_int]
        L.gfrs_crc32b_verify_batch.argtypes = [vp, vp, ctypes.c_size_t, i64, i64,
                                               ctypes.c_int, i64p]
        L.gfrs_buffer_sizes.argtypes = [ctypes.POINTER(Tactic), i64, i64p, i64p, i64p]
        L.gfrs_sized_encode_size.argtypes = [i64, i64, i64p, i64p]
        L.gfrs_sized_decode_size.restype = i64
        L.gfrs_sized_decode_size.argtypes = [i64, i64, i64]
        L.gfrs_sized_encode.restype = i64
        L.gfrs_sized_encode.argtypes = [vp, vp, vp, i64, i64]
        L.gfrs_sized_verify.argtypes = [vp, vp, i64, i64, i64, i64p]
        L.gfrs_sized_decode.restype = i64
        L.gfrs_sized_decode.argtypes = [vp, vp, vp, i64, i64, i64]
        L.gfrs_shard_disk_size.restype = i64
        L.gfrs_shard_disk_size.argtypes = [i64, i64]
        L.gfrs_shard_write_batch.argtypes = [vp, vp, ctypes.c_size_t, vp,
                                             ctypes.c_size_t, i64, i64, u64p,
                                             u64p, ctypes.c_int]
        L.gfrs_shard_parse_batch.argtypes = [vp, vp, ctypes.c_size_t, i64, i64,
                                             u64p, i64p, ctypes.c_int]
        L.gfrs_encode_idx.argtypes = [vp, vp, ctypes.c_int, vpp,
                                      ctypes.c_size_t, ctypes.c_int]
        L.gfrs_encode_frame_batch.argtypes = [vp, vp, ctypes.c_size_t, vp,
                                              ctypes.c_size_t, ctypes.c_size_t,
                                              ctypes.c_int, i64]
        L.gfrs_reconstruct_verify_batch.argtypes = [vp, vp, ctypes.c_size_t,
                                                    ctypes.c_size_t,
                                                    ctypes.c_int, i32p,
                                                    ctypes.c_int, u64p]
        L.gfrs_update_idx.argtypes = [vp, vp, vp, ctypes.c_int, vpp,
                                      ctypes.c_size_t, ctypes.c_int]
        L.gfrs_repair_batch.argtypes = [vp, vp, ctypes.c_size_t, ctypes.c_size_t,
                                        ctypes.c_int, i32p, ctypes.c_int, vp,
                                        ctypes.c_size_t, i64, u64p, u64p, u64p]
        L.gfrs_encode_matrix.argtypes = [vp, ctypes.POINTER(ctypes.c_uint8)]
        _lib = L
    return _lib


def check(rc, what=""):
    if rc < 0:
        raise GfrsError(rc, "%s: %s" % (what, lib().gfrs_last_error().decode()))
    return rc


def available():
    try:
        return lib().gfrs_device_count() > 0
    except ImportError:
        return False


def version():
    return lib().gfrs_version().decode()
