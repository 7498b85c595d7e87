"""CPU tests for the streaming sized/partial coder
(sized_coder.go / sized_coder_block.go analogs), pinned against the
oracle's bulk sized framing and zlib's CRC32-IEEE."""
import io
import zlib

import numpy as np
import pytest

from cubefs_amd import sized_stream as ss
from cubefs_amd.runtime import GfrsError

PAYLOAD = ss.DEFAULT_BLOCK - 4


def _rand(n, seed):
    return np.random.default_rng(seed).integers(0, 256, n,
                                                dtype=np.uint8).tobytes()


def test_sized_encoder_matches_oracle(oracle):
    for size in (1, 511, 512, 4092, PAYLOAD, PAYLOAD + 1, 200000):
        wb = _rand(size, size)
        enc = ss.new_sized_encoder(io.BytesIO(wb), size)
        got = enc.read()
        want, tail = oracle.sized_encode(
            np.frombuffer(wb, dtype=np.uint8).copy())
        assert got == bytes(want), size
        assert len(got) % 512 == 0


def test_sized_roundtrip_chunked():
    size = PAYLOAD * 2 + 777
    wb = _rand(size, 1)
    enc = ss.new_sized_encoder(io.BytesIO(wb), size)
    dec = ss.new_sized_decoder(enc, size)
    out = bytearray()
    while True:
        c = dec.read(997)
        if not c:
            break
        out += c
    assert bytes(out) == wb


def test_sized_decoder_detects_corruption(oracle):
    size = 100000
    wb = _rand(size, 2)
    framed, _ = oracle.sized_encode(np.frombuffer(wb, dtype=np.uint8).copy())
    bad = bytearray(bytes(framed))
    bad[50] ^= 1
    dec = ss.new_sized_decoder(io.BytesIO(bytes(bad)), size)
    with pytest.raises(GfrsError, match="MismatchedCrc"):
        dec.read()


def test_section_decoder_stops_at_block_edges():
    """ErrFrameContinue semantics (sized_coder.go:330-332): a section
    read never spans two blocks."""
    size = PAYLOAD * 2 + 100
    wb = _rand(size, 3)
    enc = ss.new_sized_encoder(io.BytesIO(wb), size)
    dec = ss.new_sized_section_decoder(enc, size)
    parts = []
    while True:
        c = dec.read()
        if not c:
            break
        parts.append(c)
    assert [len(p) for p in parts] == [PAYLOAD, PAYLOAD, 100]
    assert b"".join(parts) == wb


def test_partial_coder_roundtrip(oracle):
    """TestSizedCoderPartial (sized_coder_test.go:463-509): nonzero
    stableSize opens the stream mid-block with the 512-B head pad
    (sized_coder.go:166-170); the emitted length equals
    PartialEncodeSize(actual, stable) — pinned here against the oracle's
    independent C implementation — and the stream round-trips."""
    size = (1 << 20) + 17
    cases = [(0, size), (size - 1, 1), (817374, 1999), (11223344, 2000),
             (100, 5000), (PAYLOAD - 7, 10000), (PAYLOAD + 513, 3),
             (512, PAYLOAD * 2)]
    rng = np.random.default_rng(0x51AB1E)
    cases += [(int(rng.integers(0, 1 << 20)), int(rng.integers(1, size)))
              for _ in range(30)]
    for stable, actual in cases:
        wb = _rand(actual, stable & 0xFFFF)
        enc = ss.new_partial_encoder(io.BytesIO(wb), actual, stable)
        framed = enc.read()
        want_total, _ = oracle.partial_encode_size(actual, stable)
        nx0 = (stable % PAYLOAD) & ~511
        # PartialEncodeSize counts the stable part already on the wire
        # (the `- part` in util.go:78): the fresh stream emits total
        assert len(framed) == want_total, (stable, actual)
        assert (nx0 + len(framed)) % 512 == 0  # transport alignment
        dec = ss.new_partial_decoder(io.BytesIO(framed), actual, stable)
        assert dec.read() == wb, (stable, actual)


def test_range_decoder(oracle):
    size = PAYLOAD * 3 + 4000
    wb = _rand(size, 5)
    framed, _ = oracle.sized_encode(np.frombuffer(wb, dtype=np.uint8).copy())
    framed = bytes(framed)
    for from_, to in [(0, size), (10, 20), (PAYLOAD - 5, PAYLOAD + 5),
                      (PAYLOAD, PAYLOAD * 2), (PAYLOAD * 3, size),
                      (size - 1, size), (5, PAYLOAD * 3 + 1)]:
        blk0 = from_ // PAYLOAD
        rc = io.BytesIO(framed[blk0 * ss.DEFAULT_BLOCK:])
        head, tail, dec = ss.new_sized_range_decoder(rc, size, from_, to)
        parts = []
        while True:  # the ranged decoder is sectioned: one block per read
            c = dec.read()
            if not c:
                break
            parts.append(c)
        data = b"".join(parts)
        assert len(data) == head + (to - from_) + tail, (from_, to)
        assert data[head:len(data) - tail if tail else None] == wb[from_:to]


def test_block_coder_roundtrip_and_format():
    """ModeBlockEncode/Decode (sized_coder_block.go:43-104): LE CRC at
    the block head, no pads; the decoder hands crc ‖ payload through."""
    size = PAYLOAD + 1234
    wb = _rand(size, 7)
    enc = ss.new_sized_block_encoder(io.BytesIO(wb), size)
    framed = enc.read()
    assert len(framed) == size + 8  # two blocks, 4 B crc each
    # independent format pin via zlib
    c0 = int.from_bytes(framed[:4], "little")
    assert c0 == zlib.crc32(wb[:PAYLOAD])
    dec = ss.new_sized_block_decoder(io.BytesIO(framed), size)
    out = dec.read()
    # decoder output is crc ‖ payload per block
    assert out[4:4 + PAYLOAD] == wb[:PAYLOAD]
    assert out[PAYLOAD + 8:] == wb[PAYLOAD:]
    bad = bytearray(framed)
    bad[10] ^= 1
    dec = ss.new_sized_block_decoder(io.BytesIO(bytes(bad)), size)
    with pytest.raises(GfrsError, match="MismatchedCrc"):
        dec.read()


def test_range_block_decoder():
    size = PAYLOAD * 2 + 99
    wb = _rand(size, 9)
    enc = ss.new_sized_block_encoder(io.BytesIO(wb), size)
    framed = enc.read()
    from_, to = PAYLOAD + 3, PAYLOAD * 2 + 50
    blk0 = from_ // PAYLOAD
    rc = io.BytesIO(framed[blk0 * ss.DEFAULT_BLOCK:])
    head, tail, dec = ss.new_sized_range_block_decoder(rc, size, from_, to)
    out = dec.read()
    # strip the 4-B LE crc heads per block, then head/tail
    payloads = []
    pos = 0
    want_payload = head + (to - from_) + tail
    while pos < len(out):
        n = min(PAYLOAD, want_payload - sum(len(p) for p in payloads))
        payloads.append(out[pos + 4:pos + 4 + n])
        pos += 4 + n
    data = b"".join(payloads)
    assert data[head:len(data) - tail if tail else None] == wb[from_:to]


def test_short_source_raises():
    with pytest.raises(GfrsError, match="ShortData"):
        ss.new_sized_encoder(io.BytesIO(b"xy"), 100).read()


def test_invalid_mode_and_block():
    with pytest.raises(GfrsError, match="InvalidBlock"):
        ss.new_sized_coder(io.BytesIO(b""), 0, block_len=1000)
    with pytest.raises(GfrsError, match="Unsupported"):
        ss.new_sized_coder(io.BytesIO(b""), 0, mode=2)


def test_load_mode():
    """ModeLoad (sized_coder.go:349-413): per-block [head-pad ‖ data]
    with CRC cells stripped+checked and the tail pad discarded; requires
    section, one block per read."""
    size = PAYLOAD * 2 + 333
    wb = _rand(size, 21)
    enc = ss.new_sized_encoder(io.BytesIO(wb), size)
    ld = ss.new_sized_coder(enc, size, mode=ss.MODE_LOAD, section=True)
    parts = []
    while True:
        c = ld.read()
        if not c:
            break
        parts.append(c)
    assert [len(p) for p in parts] == [PAYLOAD, PAYLOAD, 333]
    assert b"".join(parts) == wb

    # nonzero stable: the first returned block carries the head pad
    stable, actual = PAYLOAD - 7, 10000
    wb = _rand(actual, 22)
    framed = ss.new_partial_encoder(io.BytesIO(wb), actual, stable).read()
    padhead = (stable % PAYLOAD) % 512
    ld = ss.new_sized_coder(io.BytesIO(framed), actual, stable,
                            mode=ss.MODE_LOAD, section=True)
    out = bytearray()
    while True:
        c = ld.read()
        if not c:
            break
        out += c
    assert bytes(out[padhead:]) == wb

    # corruption caught
    bad = bytearray(ss.new_sized_encoder(io.BytesIO(wb), actual).read())
    bad[100] ^= 1
    ld = ss.new_sized_coder(io.BytesIO(bytes(bad)), actual,
                            mode=ss.MODE_LOAD, section=True)
    with pytest.raises(GfrsError, match="MismatchedCrc"):
        while ld.read():
            pass
    with pytest.raises(GfrsError, match="Unsupported"):
        ss.new_sized_coder(io.BytesIO(b""), 0, mode=ss.MODE_LOAD,
                           section=False)
