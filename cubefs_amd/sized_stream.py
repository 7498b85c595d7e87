"""cubefs_amd.sized_stream — streaming sized/partial coder
(blobstore/common/crc32block/sized_coder.go, sized_coder_block.go).

File-like mirrors of the rpc2-body coders, byte-identical to the bulk
device path (`Codec.sized_encode` / `gfrs_sized_*`):

  - sized frames: payload ‖ 4 B BIG-endian CRC32-IEEE per block
    (sized_coder.go:269-276), whole stream padded to the 512-B transport
    alignment (transport.Alignment, allocator.go:12) with a zero tail;
    a nonzero stableSize opens the stream mid-block with the head pad
    `(stable %% payload) %% 512` (sized_coder.go:166-170).
  - block frames (ModeBlockEncode/Decode, sized_coder_block.go:43-104):
    4 B LITTLE-endian CRC32 at the block HEAD, no alignment pads.

Like request_body.py this is host-side streaming plumbing exactly as in
the reference (Go stdlib crc32 on the host); the CRC is libgfrs's
`gfrs_crc32_host`.  The WriterTo-side modes (ModeAppend/Check/Fix,
sized_coder.go:185-214,446-) are Go transport machinery operating on
rpc2 frame buffers and stay in the shim — their frame bytes are the same
sized frames produced here.

Section semantics: the Go decoder returns transport.ErrFrameContinue
together with the data at every interior block boundary
(sized_coder.go:330-332) so the rpc2 layer can re-frame per block; the
Python mirror expresses the same boundary by returning from read() at
each block edge when section=True (a read never spans two sections).
"""
from .crc32block import _read_full_or_to_end, crc32_host, encode_size
from .runtime import GfrsError

_CRC_LEN = 4
_ALIGN = 512  # rpc2 transport.Alignment

MODE_ENCODE = 1
MODE_DECODE = 4
MODE_LOAD = 5
MODE_BLOCK_ENCODE = 11
MODE_BLOCK_DECODE = 12

DEFAULT_BLOCK = 64 * 1024


def _partial_tail(actual, stable, block_len):
    """PartialEncodeSizeWith's tail (util.go:73-84)."""
    payload = block_len - _CRC_LEN
    part = (stable % payload) & ~(_ALIGN - 1)
    pad = (stable % payload) % _ALIGN
    size = encode_size(actual + part + pad, block_len) - part
    tail = (_ALIGN - (size & (_ALIGN - 1))) % _ALIGN
    return size + tail, tail


class SizedCoder:
    """sizedCoder (sized_coder.go:76-93), Read side."""

    def __init__(self, rc, actual_size, stable_size, block_len, mode,
                 section):
        if block_len <= 0 or block_len % 4096:
            raise GfrsError(-10, "block_len must be a positive 4096-multiple")
        if mode not in (MODE_ENCODE, MODE_DECODE, MODE_LOAD,
                        MODE_BLOCK_ENCODE, MODE_BLOCK_DECODE):
            # ModeAppend/Check/Fix are WriterTo-side transport machinery
            # (sized_coder.go:446-) and stay in the Go shim
            raise GfrsError(-103, "unsupported streaming mode %d" % mode)
        if mode == MODE_LOAD and not section:
            raise GfrsError(-103, "load mode requires section")
        payload = block_len - _CRC_LEN
        _, tail = _partial_tail(actual_size, stable_size, block_len)
        self._rc = rc
        self._payload = payload
        self._mode = mode
        self._section = section
        self._padhead = (stable_size % payload) % _ALIGN
        self._remain = actual_size
        self._padtail = tail
        self._nx = (stable_size % payload) & ~(_ALIGN - 1)
        self._crc = 0
        self._buf = b""   # produced-but-unread output
        self._off = 0
        self._err = None
        self._eof = False
        self._closed = False

    # ---- production of one output burst per mode -------------------
    def _produce_encode(self):
        out = []
        if self._padhead:
            # alignment pad: length-only in the reference (content is
            # never examined); emitted as zeros so the stream is
            # deterministic and matches the device/oracle images
            out.append(b"\x00" * self._padhead)
            self._nx += self._padhead
            self._padhead = 0
        if self._remain <= 0:
            if self._padtail:
                out.append(b"\x00" * self._padtail)
                self._padtail = 0
            if not out:
                self._eof = True
            return b"".join(out)
        want = min(self._payload - self._nx, self._remain)
        data = _read_full_or_to_end(self._rc, want)
        if len(data) < want:
            raise GfrsError(-7, "short source: want %d got %d"
                            % (want, len(data)))
        self._crc = crc32_host(data, self._crc)
        self._nx += len(data)
        self._remain -= len(data)
        out.append(data)
        if self._nx == self._payload or self._remain == 0:
            out.append(self._crc.to_bytes(4, "big"))
            self._crc = 0
            self._nx = 0
        return b"".join(out)

    def _produce_decode(self):
        if self._remain <= 0:
            self._eof = True
            return b""
        if self._padhead:
            pad = _read_full_or_to_end(self._rc, self._padhead)
            if len(pad) < self._padhead:
                raise GfrsError(-7, "short head pad")
            self._nx += self._padhead
            self._padhead = 0
        want = min(self._payload - self._nx, self._remain)
        data = _read_full_or_to_end(self._rc, want)
        if len(data) < want:
            raise GfrsError(-7, "short block: want %d got %d"
                            % (want, len(data)))
        self._crc = crc32_host(data, self._crc)
        self._nx += len(data)
        self._remain -= len(data)
        if self._nx == self._payload or self._remain == 0:
            cell = _read_full_or_to_end(self._rc, _CRC_LEN)
            if len(cell) < _CRC_LEN:
                raise GfrsError(-7, "short checksum cell")
            if int.from_bytes(cell, "big") != self._crc:
                raise GfrsError(-9, "mismatched checksum")
            self._crc = 0
            self._nx = 0
            if self._remain == 0 and self._padtail:
                pad = _read_full_or_to_end(self._rc, self._padtail)
                if len(pad) < self._padtail:
                    raise GfrsError(-7, "short tail pad")
                self._padtail = 0
        return data

    def _produce_load(self):
        """decodeLoad (sized_coder.go:349-413): yields [head-pad ‖ data]
        per block — the pads the rpc2 server keeps in its aligned
        buffers — stripping and checking the CRC cell and discarding the
        final tail pad.  (The Go form additionally demands a 512-aligned
        caller buffer of exactly one block; a Python bytes return has no
        such constraint.)"""
        if self._remain <= 0:
            self._eof = True
            return b""
        want = self._payload + _CRC_LEN - self._nx
        extra = want - self._padhead - _CRC_LEN - self._padtail - self._remain
        last = extra >= 0
        if last:
            want -= extra
        raw = _read_full_or_to_end(self._rc, want)
        if len(raw) < want:
            raise GfrsError(-7, "short load block: want %d got %d"
                            % (want, len(raw)))
        out_head = b""
        if self._padhead:
            out_head = raw[:self._padhead]
            self._nx += self._padhead
            raw = raw[self._padhead:]
            self._padhead = 0
        n = len(raw) - _CRC_LEN
        if last:
            n -= self._padtail
            self._padtail = 0
        cell = raw[n:n + _CRC_LEN]
        data = raw[:n]
        self._crc = crc32_host(data, self._crc)
        self._nx += n
        self._remain -= n
        if self._nx == self._payload or self._remain == 0:
            if int.from_bytes(cell, "big") != self._crc:
                raise GfrsError(-9, "mismatched checksum")
            self._crc = 0
            self._nx = 0
        return out_head + data

    def _produce_block(self, decode):
        if self._remain <= 0:
            self._eof = True
            return b""
        want = min(self._payload, self._remain)
        if decode:
            cell = _read_full_or_to_end(self._rc, _CRC_LEN)
            data = _read_full_or_to_end(self._rc, want)
            if len(cell) < _CRC_LEN or len(data) < want:
                raise GfrsError(-7, "short block")
            if int.from_bytes(cell, "little") != crc32_host(data):
                raise GfrsError(-9, "mismatched checksum")
            self._remain -= want
            # decodeBlock hands the caller crc ‖ payload verbatim
            return cell + data
        data = _read_full_or_to_end(self._rc, want)
        if len(data) < want:
            raise GfrsError(-7, "short source")
        self._remain -= want
        return crc32_host(data).to_bytes(4, "little") + data

    # ---- io.Reader -------------------------------------------------
    def read(self, size=-1):
        if self._closed:
            raise GfrsError(-11, "read on closed coder")
        if self._err is not None:
            raise self._err
        out = []
        got = 0
        boundary = False
        while (size < 0 or got < size) and not self._eof and not boundary:
            if self._off == len(self._buf):
                try:
                    if self._mode == MODE_ENCODE:
                        self._buf = self._produce_encode()
                    elif self._mode == MODE_DECODE:
                        self._buf = self._produce_decode()
                        # a completed interior block is a section edge
                        boundary = (self._section and self._nx == 0
                                    and self._remain > 0)
                    elif self._mode == MODE_LOAD:
                        self._buf = self._produce_load()
                        boundary = (self._section and self._nx == 0
                                    and self._remain > 0)
                    else:
                        self._buf = self._produce_block(
                            self._mode == MODE_BLOCK_DECODE)
                        boundary = self._section and self._remain > 0
                except GfrsError as e:
                    self._err = e  # sticky, like the reference's r.err
                    if got:
                        break
                    raise
                self._off = 0
                if self._eof:
                    break
            take = len(self._buf) - self._off
            if size >= 0:
                take = min(take, size - got)
            out.append(self._buf[self._off:self._off + take])
            self._off += take
            got += take
            if boundary and self._off < len(self._buf):
                boundary = False  # caller asked for less; keep the rest
        return b"".join(out)

    def close(self):
        if not self._closed:
            self._closed = True
            close = getattr(self._rc, "close", None)
            if close is not None:
                close()

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()


# ---- constructors (sized_coder.go:104-214, sized_coder_block.go) -----

def new_sized_coder(rc, actual_size, stable_size=0,
                    block_len=DEFAULT_BLOCK, mode=MODE_ENCODE,
                    section=False):
    """NewSizedCoder (sized_coder.go:153-183)."""
    return SizedCoder(rc, actual_size, stable_size, block_len, mode, section)


def new_sized_encoder(rc, actual_size):
    """NewSizedEncoder (sized_coder.go:104-106)."""
    return new_sized_coder(rc, actual_size, 0, DEFAULT_BLOCK, MODE_ENCODE)


def new_sized_decoder(rc, actual_size):
    """NewSizedDecoder (sized_coder.go:109-111)."""
    return new_sized_coder(rc, actual_size, 0, DEFAULT_BLOCK, MODE_DECODE)


def new_sized_section_decoder(rc, actual_size):
    """NewSizedSectionDecoder (sized_coder.go:114-116)."""
    return new_sized_coder(rc, actual_size, 0, DEFAULT_BLOCK, MODE_DECODE,
                           section=True)


def new_partial_encoder(rc, actual_size, stable_size):
    """NewPartialEncoder (sized_coder.go:143-145)."""
    return new_sized_coder(rc, actual_size, stable_size, DEFAULT_BLOCK,
                           MODE_ENCODE)


def new_partial_decoder(rc, actual_size, stable_size):
    """NewPartialDecoder (sized_coder.go:148-150)."""
    return new_sized_coder(rc, actual_size, stable_size, DEFAULT_BLOCK,
                           MODE_DECODE)


def _ranged(rc, actual_size, from_, to, block_len, mode, section):
    """newSizedRangeDecoder (sized_coder.go:124-141): widen [from, to) to
    payload-block edges; returns (head, tail, coder) — the caller drops
    `head` leading and `tail` trailing payload bytes."""
    payload = block_len - _CRC_LEN
    head = from_ % payload                       # util.AlignedHead
    tail = (payload - to % payload) % payload    # util.AlignedTail
    more = to + tail
    if more > actual_size:
        tail -= more - actual_size
    actual = to - from_ + head + tail
    return head, tail, new_sized_coder(rc, actual, 0, block_len, mode,
                                       section)


def new_sized_range_decoder(rc, actual_size, from_, to,
                            block_len=DEFAULT_BLOCK):
    """NewSizedRangeDecoder (sized_coder.go:120-122)."""
    return _ranged(rc, actual_size, from_, to, block_len, MODE_DECODE, True)


def new_sized_block_encoder(rc, actual_size, block_len=DEFAULT_BLOCK):
    """NewSizedBlockEncoder (sized_coder_block.go:25-27)."""
    return new_sized_coder(rc, actual_size, 0, block_len, MODE_BLOCK_ENCODE)


def new_sized_block_decoder(rc, actual_size, block_len=DEFAULT_BLOCK):
    """NewSizedBlockDecoder (sized_coder_block.go:30-32)."""
    return new_sized_coder(rc, actual_size, 0, block_len, MODE_BLOCK_DECODE)


def new_sized_range_block_decoder(rc, actual_size, from_, to,
                                  block_len=DEFAULT_BLOCK):
    """NewSizedRangeBlockDecoder (sized_coder_block.go:36-38)."""
    return _ranged(rc, actual_size, from_, to, block_len, MODE_BLOCK_DECODE,
                   False)
