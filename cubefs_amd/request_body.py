"""cubefs_amd.request_body — streaming HTTP-body framing
(blobstore/common/crc32block/request_body.go).

`body_encoder` / `body_decoder` mirror NewBodyEncoder/NewBodyDecoder
(request_body.go:180-200): file-like wrappers that frame or strip
[4 B LE CRC32-IEEE ‖ payload] blocks one block at a time as a body
streams through (request_body.go:57-127).  This path is host-side in the
reference too — the Go stdlib CRC on the client/server CPU — so the CRC
here is libgfrs's host fold (`gfrs_crc32_host`), not an oracle or torch
path.  Bulk framing belongs to `crc32block.Codec` and the fused
encode+frame kernel; this is per-request plumbing.

Semantics kept from the reference:
  - encode: payload read in (block_len-4)-byte units, each emitted as
    crc ‖ payload; a short final read emits a short final block
    (request_body.go:111-115).
  - decode: blocks read in block_len units; a final unit of <= 4 bytes
    or any CRC mismatch raises ErrMismatchedCrc (request_body.go:117-126).
  - read after close raises ErrReadOnClosed (request_body.go:104).
  - code_size(size): encoded size for encoding, origin size for decoding
    (request_body.go:141-146); rc=None gives a size-only body
    (request_body.go:155-163).
"""
from .crc32block import (DEFAULT_BLOCK, _read_full_or_to_end, crc32_host,
                         decode_size, encode_size)
from .runtime import GfrsError

_CRC_LEN = 4


def _crc32(data):
    return crc32_host(data)


class _RequestBody:
    """requestBody (request_body.go:45-146), minus the goroutine close
    machinery — Python readers are synchronous, so Close just poisons
    the next read as the reference's closeCh does."""

    def __init__(self, rc, encode, block_len):
        if block_len <= 0 or block_len % 4096:
            raise GfrsError(-10, "block_len must be a positive 4096-multiple")
        self._rc = rc
        self._encode = encode
        self._block_len = block_len
        self._block = b""
        self._off = 0
        self._err = None     # sticky error (reference r.err)
        self._eof = False
        self._closed = False

    # -- io.Reader ----------------------------------------------------
    def read(self, size=-1):
        if self._closed:
            raise GfrsError(-11, "read on closed body")
        if self._err is not None:
            raise self._err
        out = []
        got = 0
        while size < 0 or got < size:
            if self._off == len(self._block):
                try:
                    if not self._next_block():
                        break  # clean EOF
                except GfrsError as e:
                    # errors are sticky (reference r.err, request_body.go:46)
                    self._err = e
                    if got:
                        # reference Read returns (n, nil) and surfaces the
                        # error on the next call (request_body.go:64-68)
                        break
                    raise
            take = len(self._block) - self._off
            if size >= 0:
                take = min(take, size - got)
            out.append(self._block[self._off:self._off + take])
            self._off += take
            got += take
        return b"".join(out)

    def _next_block(self):
        if self._eof:
            return False
        if self._encode:
            payload = _read_full_or_to_end(self._rc, self._block_len - _CRC_LEN)
            if not payload:
                self._eof = True
                return False
            crc = _crc32(payload)
            self._block = crc.to_bytes(4, "little") + payload
        else:
            block = _read_full_or_to_end(self._rc, self._block_len)
            if not block:
                self._eof = True
                return False
            if len(block) <= _CRC_LEN:
                raise GfrsError(-9, "short final block")
            want = int.from_bytes(block[:_CRC_LEN], "little")
            if _crc32(block[_CRC_LEN:]) != want:
                raise GfrsError(-9, "mismatched checksum")
            self._block = block[_CRC_LEN:]
        self._off = 0
        return True

    # -- io.Closer ----------------------------------------------------
    def close(self):
        if not self._closed:
            self._closed = True
            self._block = b""
            close = getattr(self._rc, "close", None)
            if close is not None:
                close()

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()

    # -- RequestBody --------------------------------------------------
    def code_size(self, size):
        if self._encode:
            return encode_size(size, self._block_len)
        return decode_size(size, self._block_len)


class _CodeSizeBody:
    """codeSizeBody (request_body.go:148-163): size math only."""

    def __init__(self, encode, block_len):
        self._encode = encode
        self._block_len = block_len

    def read(self, size=-1):
        return b""

    def close(self):
        pass

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()

    def code_size(self, size):
        if self._encode:
            return encode_size(size, self._block_len)
        return decode_size(size, self._block_len)


def body_encoder(rc=None, block_len=DEFAULT_BLOCK):
    """NewBodyEncoder (request_body.go:185-190)."""
    if rc is None:
        return _CodeSizeBody(True, block_len)
    return _RequestBody(rc, True, block_len)


def body_decoder(rc=None, block_len=DEFAULT_BLOCK):
    """NewBodyDecoder (request_body.go:192-200)."""
    if rc is None:
        return _CodeSizeBody(False, block_len)
    return _RequestBody(rc, False, block_len)
