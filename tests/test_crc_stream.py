"""CPU tests for the encode.go/decode.go streaming forms: the limit-size
Encoder.Encode analog (encode_to) and the random-access ranged Decoder
(decode.go:32-145) over crc32block-framed journals."""
import io

import numpy as np
import pytest

from cubefs_amd import crc32block
from cubefs_amd.runtime import GfrsError


def _rand(n, seed):
    return np.random.default_rng(seed).integers(0, 256, n,
                                                dtype=np.uint8).tobytes()


def _journal(payload, block_len=crc32block.DEFAULT_BLOCK):
    out = io.BytesIO()
    crc32block.encode_to(io.BytesIO(payload), len(payload), out, block_len)
    return out.getvalue()


def test_encode_to_matches_oracle(oracle):
    for size in (1, 4092, 65532, 65536, 200001):
        wb = _rand(size, size)
        got = _journal(wb)
        want = oracle.crc32b_encode(np.frombuffer(wb, dtype=np.uint8).copy())
        assert got == bytes(want)
        assert len(got) == crc32block.encode_size(size)


def test_encode_to_short_source():
    """Encoder.Encode wraps a short source in an error (encode.go:97-99)."""
    wb = _rand(100, 1)
    with pytest.raises(GfrsError, match="ShortData"):
        crc32block.encode_to(io.BytesIO(wb), 200, io.BytesIO())


def test_decoder_ranges():
    """Decoder.Reader(from, to) yields exactly the requested payload span
    (decode.go:121-145), for ranges crossing block boundaries, block-
    aligned, inside one block, at the tail, and empty."""
    payload_len = crc32block.DEFAULT_BLOCK - 4
    size = payload_len * 2 + 1234
    wb = _rand(size, 42)
    framed = _journal(wb)
    dec = crc32block.Decoder(io.BytesIO(framed), 0, size)
    cases = [(0, size), (0, 1), (payload_len - 3, payload_len + 7),
             (payload_len, payload_len * 2), (5, 5),
             (payload_len * 2, size), (size - 1, size),
             (payload_len * 2 + 100, size - 7)]
    for from_, to in cases:
        r = dec.reader(from_, to)
        got = r.read()
        assert got == wb[from_:to], (from_, to)
        assert r.read(16) == b""


def test_decoder_chunked_range_read():
    payload_len = crc32block.DEFAULT_BLOCK - 4
    size = payload_len * 3
    wb = _rand(size, 7)
    dec = crc32block.Decoder(io.BytesIO(_journal(wb)), 0, size)
    r = dec.reader(1000, size - 1000)
    out = bytearray()
    while True:
        c = r.read(999)
        if not c:
            break
        out += c
    assert bytes(out) == wb[1000:size - 1000]


def test_decoder_nonzero_offset():
    """The journal can sit at an offset inside a larger region
    (NewDecoder's off parameter)."""
    size = 100000
    wb = _rand(size, 9)
    blob = b"\xee" * 777 + _journal(wb) + b"\xdd" * 33
    dec = crc32block.Decoder(io.BytesIO(blob), 777, size)
    assert dec.reader(0, size).read() == wb


def test_decoder_corruption_in_range():
    payload_len = crc32block.DEFAULT_BLOCK - 4
    size = payload_len * 2
    wb = _rand(size, 11)
    framed = bytearray(_journal(wb))
    framed[crc32block.DEFAULT_BLOCK + 100] ^= 1  # inside block 1
    dec = crc32block.Decoder(io.BytesIO(bytes(framed)), 0, size)
    with pytest.raises(GfrsError, match="MismatchedCrc"):
        dec.reader(payload_len + 5, size).read()
    # block 0 is untouched by the corruption and readable on its own
    assert dec.reader(0, payload_len).read() == wb[:payload_len]


def test_decoder_skips_blocks_before_range():
    """Only blocks overlapping the range are read at all
    (decode.go:125-129 blockOff seek) — corruption before the range must
    not matter."""
    payload_len = crc32block.DEFAULT_BLOCK - 4
    size = payload_len * 3
    wb = _rand(size, 13)
    framed = bytearray(_journal(wb))
    framed[10] ^= 1  # corrupt block 0
    dec = crc32block.Decoder(io.BytesIO(bytes(framed)), 0, size)
    got = dec.reader(payload_len * 2, size).read()
    assert got == wb[payload_len * 2:]


def test_decoder_truncated_journal():
    size = 100000
    wb = _rand(size, 17)
    framed = _journal(wb)
    dec = crc32block.Decoder(io.BytesIO(framed[:-10]), 0, size)
    with pytest.raises(GfrsError, match="ShortData"):
        dec.reader(0, size).read()


def test_decoder_read_at_callable():
    size = 70000
    wb = _rand(size, 19)
    framed = _journal(wb)
    calls = []

    def read_at(off, n):
        calls.append((off, n))
        return framed[off:off + n]

    dec = crc32block.Decoder(read_at, 0, size)
    assert dec.reader(0, size).read() == wb
    assert calls
