"""CPU tests for the streaming request-body wrapper
(crc32block/request_body.go analog), mirroring request_body_test.go and
pinning the framed bytes against the oracle's crc32b framing."""
import io

import numpy as np
import pytest

from cubefs_amd import crc32block, request_body
from cubefs_amd.runtime import GfrsError

SIZES = [0, 64 * 1024 - 4, 64 * 1024, 64 * 1024 + 4, 1024 * 1024]


def _rand(n, seed):
    return np.random.default_rng(seed).integers(0, 256, n,
                                                dtype=np.uint8).tobytes()


def test_body_encoder_decoder_roundtrip():
    """TestBodyEncoderDecoder (request_body_test.go:26-71): decoder chained
    directly on the encoder, io.ReadFull-style."""
    for idx, size in enumerate(SIZES):
        wb = _rand(size, idx)
        enc = request_body.body_encoder(io.BytesIO(wb))
        dec = request_body.body_decoder(enc)
        assert dec.code_size(enc.code_size(size)) == size
        rb = dec.read(size)
        assert rb == wb, "index %d" % idx
        assert dec.read(1) == b""  # clean EOF after the body
        dec.close()
        with pytest.raises(GfrsError, match="ReadOnClosed"):
            dec.read(1)


def test_encoded_bytes_match_oracle(oracle):
    """The streamed frames are byte-identical to the bulk framing
    (encode.go) the oracle and the GPU kernels produce."""
    for size in (1, 4092, 65532, 65533, 200000):
        wb = _rand(size, size)
        enc = request_body.body_encoder(io.BytesIO(wb))
        got = enc.read()
        want = oracle.crc32b_encode(np.frombuffer(wb, dtype=np.uint8).copy())
        assert got == bytes(want)
        assert len(got) == crc32block.encode_size(size)


def test_chunked_reads():
    """Odd-sized reads cross block boundaries without losing bytes."""
    wb = _rand(65532 * 2 + 7, 99)
    enc = request_body.body_encoder(io.BytesIO(wb))
    dec = request_body.body_decoder(enc)
    out = bytearray()
    while True:
        chunk = dec.read(1237)
        if not chunk:
            break
        out += chunk
    assert bytes(out) == wb


def test_nil_bodies():
    """TestNilEncoderDecoder (request_body_test.go:74-92): size-only."""
    enc = request_body.body_encoder(None)
    dec = request_body.body_decoder(None)
    for size in SIZES:
        assert dec.code_size(enc.code_size(size)) == size
    assert dec.read(1024) == b""
    enc.close()


def test_decoder_mismatch():
    """TestBodyDecoderMissmatch (request_body_test.go:104-117): one
    flipped byte in the stream raises ErrMismatchedCrc."""
    wb = _rand(1 << 12, 7)
    enc = request_body.body_encoder(io.BytesIO(wb))
    framed = bytearray(enc.read())
    framed[10] ^= 1
    dec = request_body.body_decoder(io.BytesIO(bytes(framed)))
    with pytest.raises(GfrsError, match="MismatchedCrc"):
        dec.read(len(wb))


def test_decoder_short_final_block():
    """A trailing fragment of <= 4 bytes is ErrMismatchedCrc
    (request_body.go:117-119)."""
    wb = _rand(65532, 3)  # exactly one full 64 KiB frame
    enc = request_body.body_encoder(io.BytesIO(wb))
    framed = enc.read()
    assert len(framed) == 65536
    dec = request_body.body_decoder(io.BytesIO(framed + b"\x00\x00\x00"))
    got = dec.read(65532)  # first block fine
    assert got == wb
    with pytest.raises(GfrsError, match="MismatchedCrc"):
        dec.read(1)


def test_trickle_source():
    """readFullOrToEnd (util.go:105) keeps reading a slow source until the
    block fills; 1-byte-at-a-time underlying reads still frame correctly."""
    class Trickle:
        def __init__(self, data):
            self.b = io.BytesIO(data)

        def read(self, n):
            return self.b.read(min(n, 1))

    wb = _rand(70000, 5)
    enc = request_body.body_encoder(Trickle(wb))
    dec = request_body.body_decoder(enc)
    assert dec.read() == wb


def test_nonstandard_block_len():
    wb = _rand(30000, 11)
    enc = request_body.body_encoder(io.BytesIO(wb), block_len=8192)
    dec = request_body.body_decoder(enc, block_len=8192)
    assert dec.read() == wb
    with pytest.raises(GfrsError, match="InvalidBlock"):
        request_body.body_encoder(io.BytesIO(wb), block_len=1000)


def test_host_crc_matches_oracle(oracle):
    """gfrs_crc32_host is CRC32-IEEE with Update chaining semantics."""
    buf = _rand(5000, 13)
    a = np.frombuffer(buf, dtype=np.uint8)
    assert request_body._crc32(buf) == oracle.crc32(a)
    from cubefs_amd.runtime import lib
    c = lib().gfrs_crc32_host(0, buf[:2000], 2000)
    c = lib().gfrs_crc32_host(c, buf[2000:], 3000)
    assert c == oracle.crc32(a)
    assert request_body._crc32(b"123456789") == 0xCBF43926


def test_errors_are_sticky():
    """After a decode error every later read raises the same error
    (requestBody.err, request_body.go:46,58-60)."""
    wb = _rand(1 << 12, 31)
    enc = request_body.body_encoder(io.BytesIO(wb))
    framed = bytearray(enc.read())
    framed[10] ^= 1
    dec = request_body.body_decoder(io.BytesIO(bytes(framed)))
    with pytest.raises(GfrsError, match="MismatchedCrc"):
        dec.read()
    with pytest.raises(GfrsError, match="MismatchedCrc"):
        dec.read(1)
