"""cubefs_amd.crc32block — blobstore/common/crc32block surface over the
gfrs C ABI.

Frame format (block.go:22-49): 4 B LE CRC32-IEEE ‖ payload, block length a
positive multiple of 4096 (util.go:40), default 64 KiB.  The compute path
is the HIP kernel; this module is size math + call plumbing.
"""
import ctypes

from .runtime import GfrsError, Tactic, check, lib

DEFAULT_BLOCK = 64 * 1024


def encode_size(size, block_len=DEFAULT_BLOCK):
    """EncodeSize (util.go:56-62)."""
    return check(lib().gfrs_crc32b_encode_size(size, block_len), "encode_size")


def decode_size(size, block_len=DEFAULT_BLOCK):
    """DecodeSize (util.go:65-71)."""
    return check(lib().gfrs_crc32b_decode_size(size, block_len), "decode_size")


def sized_encode_size(actual_size, block_len=DEFAULT_BLOCK):
    """PartialEncodeSize with stableSize=0 (util.go:73-84): returns
    (total incl. 512-B tail pad, tail)."""
    size = ctypes.c_int64()
    tail = ctypes.c_int64()
    check(lib().gfrs_sized_encode_size(actual_size, block_len,
                                       ctypes.byref(size), ctypes.byref(tail)),
          "sized_encode_size")
    return size.value, tail.value


def sized_decode_size(total, tail, block_len=DEFAULT_BLOCK):
    return check(lib().gfrs_sized_decode_size(total, tail, block_len),
                 "sized_decode_size")


class Codec:
    """Device-side encoder/decoder; one gfrs context (any valid tactic works,
    the CRC path ignores it)."""

    def __init__(self, device=-1):
        t = Tactic(4, 2, 0, 1, 5, 0, 2048)
        self._ctx = lib().gfrs_create(ctypes.byref(t), device)
        if not self._ctx:
            raise GfrsError(-100, lib().gfrs_last_error().decode())
        self._destroy = lib().gfrs_destroy

    def __del__(self):
        # interpreter shutdown can clear module globals before __del__
        # runs, so use the function handle captured at construction
        ctx = getattr(self, "_ctx", None)
        destroy = getattr(self, "_destroy", None)
        if ctx and destroy is not None:
            try:
                destroy(ctx)
            except TypeError:
                pass
            self._ctx = None

    def encode(self, dst, src, block_len=DEFAULT_BLOCK):
        """Frame src (device uint8 tensor) into dst; returns bytes written."""
        n = src.numel()
        need = encode_size(n, block_len)
        assert dst.numel() >= need, (dst.numel(), need)
        return check(lib().gfrs_crc32b_encode(self._ctx, dst.data_ptr(),
                                              src.data_ptr(), n, block_len),
                     "crc_encode")

    def verify(self, framed, block_len=DEFAULT_BLOCK):
        """Returns -1 if all frames pass, else first bad block index."""
        bad = ctypes.c_int64(-1)
        check(lib().gfrs_crc32b_verify(self._ctx, framed.data_ptr(),
                                       framed.numel(), block_len,
                                       ctypes.byref(bad)), "crc_verify")
        return bad.value

    def decode(self, dst, framed, block_len=DEFAULT_BLOCK):
        """Strip frames into dst, checking CRCs; returns payload bytes."""
        return check(lib().gfrs_crc32b_decode(self._ctx, dst.data_ptr(),
                                              framed.data_ptr(),
                                              framed.numel(), block_len),
                     "crc_decode")

    def encode_batch(self, dst, src, block_len=DEFAULT_BLOCK):
        """src: [nshards, n] device tensor; dst: [nshards, encode_size(n)]."""
        ns, n = src.shape
        check(lib().gfrs_crc32b_encode_batch(
            self._ctx, dst.data_ptr(), dst.stride(0), src.data_ptr(),
            src.stride(0), n, block_len, ns), "crc_encode_batch")

    def verify_batch(self, framed, block_len=DEFAULT_BLOCK):
        ns, fl = framed.shape
        bad = (ctypes.c_int64 * ns)()
        check(lib().gfrs_crc32b_verify_batch(self._ctx, framed.data_ptr(),
                                             framed.stride(0), fl, block_len,
                                             ns, bad), "crc_verify_batch")
        return list(bad)

    def sized_encode(self, dst, src, block_len=DEFAULT_BLOCK):
        """rpc2 body framing (sized_coder.go ModeEncode): payload ‖ CRC(BE)
        frames + 512-B zero tail.  Returns total bytes written."""
        return check(lib().gfrs_sized_encode(self._ctx, dst.data_ptr(),
                                             src.data_ptr(), src.numel(),
                                             block_len), "sized_encode")

    def sized_verify(self, framed, tail, block_len=DEFAULT_BLOCK):
        bad = ctypes.c_int64(-1)
        check(lib().gfrs_sized_verify(self._ctx, framed.data_ptr(),
                                      framed.numel(), tail, block_len,
                                      ctypes.byref(bad)), "sized_verify")
        return bad.value

    def sized_decode(self, dst, framed, tail, block_len=DEFAULT_BLOCK):
        return check(lib().gfrs_sized_decode(self._ctx, dst.data_ptr(),
                                             framed.data_ptr(), framed.numel(),
                                             tail, block_len), "sized_decode")

    def synchronize(self):
        check(lib().gfrs_synchronize(self._ctx), "synchronize")
