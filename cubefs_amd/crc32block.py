"""cubefs_amd.crc32block — blobstore/common/crc32block surface over the
gfrs C ABI.

Frame format (block.go:22-49): 4 B LE CRC32-IEEE ‖ payload, block length a
positive multiple of 4096 (util.go:40), default 64 KiB.  The compute path
is the HIP kernel; this module is size math + call plumbing.
"""
import ctypes

from .runtime import GfrsError, Tactic, check, lib

DEFAULT_BLOCK = 64 * 1024
_CRC_LEN = 4


def crc32_host(data, crc=0):
    """Host CRC32-IEEE in hash/crc32.Update semantics (gfrs_crc32_host):
    backs the host-side streaming wrappers below, exactly as the
    reference's streaming forms run Go stdlib crc32 on the host."""
    if not data:
        return crc
    buf = (ctypes.c_char * len(data)).from_buffer_copy(data)
    return lib().gfrs_crc32_host(crc, buf, len(data))


def _read_full_or_to_end(rc, want):
    """readFullOrToEnd (util.go:105-117): fill up to `want` bytes,
    stopping early only at EOF."""
    parts = []
    n = 0
    while n < want:
        chunk = rc.read(want - n)
        if not chunk:
            break
        parts.append(chunk)
        n += len(chunk)
    return b"".join(parts)


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


def encode_to(reader, limit_size, writer, block_len=DEFAULT_BLOCK):
    """Encoder.Encode (encode.go:47-57): frame exactly limit_size bytes
    from `reader` into `writer`; a short source is an error (the
    reference's io.ReadFull → ReaderError).  Returns bytes written.
    Host-side streaming plumbing; bulk framing is Codec.encode."""
    if block_len <= 0 or block_len % 4096:
        raise GfrsError(-10, "block_len must be a positive 4096-multiple")
    payload_len = block_len - _CRC_LEN
    remain = limit_size
    written = 0
    while remain > 0:
        need = min(remain, payload_len)
        payload = _read_full_or_to_end(reader, need)
        if len(payload) < need:
            raise GfrsError(-7, "short source: want %d got %d"
                            % (need, len(payload)))
        crc = crc32_host(payload)
        writer.write(crc.to_bytes(4, "little") + payload)
        written += _CRC_LEN + len(payload)
        remain -= len(payload)
    return written


class _BlockReader:
    """blockReader (decode.go:54-108): framed stream -> payload stream
    with the remaining payload size known, so the final short block is
    read exactly (io.ReadFull semantics: a truncated journal errors)."""

    def __init__(self, reader, limit, block_len):
        self._r = reader
        self._remain = limit
        self._block_len = block_len
        self._block = b""
        self._i = 0
        self._err = None

    def read(self, size=-1):
        if self._err is not None:
            raise self._err  # sticky, like blockReader.err (decode.go:58)
        out = []
        got = 0
        while (size < 0 or got < size) and not (
                self._i == len(self._block) and self._remain == 0):
            if self._i == len(self._block):
                try:
                    self._next_block()
                except GfrsError as e:
                    self._err = e
                    raise
            take = len(self._block) - self._i
            if size >= 0:
                take = min(take, size - got)
            out.append(self._block[self._i:self._i + take])
            self._i += take
            self._remain -= take
            got += take
        return b"".join(out)

    def _next_block(self):
        payload_len = self._block_len - _CRC_LEN
        want = self._block_len
        if self._remain < payload_len:
            want = self._remain + _CRC_LEN
        raw = _read_full_or_to_end(self._r, want)
        if len(raw) < want:
            raise GfrsError(-7, "truncated journal: want %d got %d"
                            % (want, len(raw)))
        if crc32_host(raw[_CRC_LEN:]) != int.from_bytes(raw[:_CRC_LEN],
                                                        "little"):
            raise GfrsError(-9, "mismatched checksum")
        self._block = raw[_CRC_LEN:]
        self._i = 0


class _RangeReader:
    """rangeReader (decode.go:46-119): skip into the first block, then
    limit to the requested span."""

    def __init__(self, reader, skip, limit):
        self._r = reader
        self._skip = skip
        self._limit = limit

    def read(self, size=-1):
        if self._skip:
            self._r.read(self._skip)
            self._skip = 0
        if size < 0 or size > self._limit:
            size = self._limit
        out = self._r.read(size)
        self._limit -= len(out)
        return out


class Decoder:
    """Decoder (decode.go:32-135): random-access ranged decode over a
    crc32block-framed journal — the datanode range-GET shape.  `read_at`
    is pread semantics: read_at(offset, n) -> bytes (a callable, or any
    object with a .read_at method or seek+read file API)."""

    def __init__(self, read_at, off, size, block_len=DEFAULT_BLOCK):
        if block_len <= 0 or block_len % 4096:
            raise GfrsError(-10, "block_len must be a positive 4096-multiple")
        if callable(read_at):
            self._read_at = read_at
        elif hasattr(read_at, "read_at"):
            self._read_at = read_at.read_at
        else:  # seekable file object
            def _pread(o, n, _f=read_at):
                _f.seek(o)
                return _f.read(n)
            self._read_at = _pread
        self._off = off
        self._limit = size       # decoded (payload) size of the journal
        self._block_len = block_len

    def reader(self, from_, to):
        """Decoder.Reader (decode.go:121-145): a file-like yielding
        payload bytes [from_, to) of the decoded journal, touching only
        the blocks that overlap the range."""
        payload_len = self._block_len - _CRC_LEN
        block_off = (from_ // payload_len) * self._block_len
        encoded_size = encode_size(self._limit, self._block_len) - block_off

        class _Section:
            def __init__(s):
                s.pos = 0

            def read(s, n):
                n = min(n, encoded_size - s.pos)
                if n <= 0:
                    return b""
                out = self._read_at(self._off + block_off + s.pos, n)
                s.pos += len(out)
                return out

        payload = _BlockReader(_Section(),
                               decode_size(encoded_size, self._block_len),
                               self._block_len)
        return _RangeReader(payload, from_ % payload_len, to - from_)
