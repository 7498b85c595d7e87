#!/usr/bin/env python3
"""Generate frozen RS/LRC/CRC golden vectors (tests/golden/rs_vectors.npz).

The oracle generates these once; committing them freezes the bytes so any
later change to oracle or HIP engine that alters parity output is caught
even if both change together.  The oracle itself is pinned independently by
the literal GF tables extracted from the reference (gf_tables.bin) and the
CRC32 universal KAT — see oracle/oracle.h.

Inputs follow SURVEY.md §8d: seeded PCG64, seed 0xB10B5703 ^ case-index,
uniform random bytes.
"""
import os
import sys

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))
from oracle import pyoracle as po  # noqa: E402

OUT = os.path.join(os.path.dirname(__file__), "rs_vectors.npz")

CASES = [
    # (name, n, m, l, az, shard_len)
    ("EC4P2_1KiB", 4, 2, 0, 1, 1024),
    ("EC6P3_2048", 6, 3, 0, 1, 2048),
    ("EC6P3_4093", 6, 3, 0, 1, 4093),   # ragged length
    ("EC12P4_2048", 12, 4, 0, 1, 2048),
    ("EC16P20L2_1024", 16, 20, 2, 2, 1024),
    ("LRC12P2L2_2048", 12, 2, 2, 2, 2048),  # Azure-LRC(12,2,2), AZ=2
    ("EC6P10L2_1024", 6, 10, 2, 2, 1024),
]


def main():
    out = {}
    for idx, (name, n, m, l, az, slen) in enumerate(CASES):
        rng = np.random.default_rng(np.random.PCG64(0xB10B5703 ^ idx))
        shards = [rng.integers(0, 256, slen, dtype=np.uint8) for _ in range(n)]
        shards += [np.zeros(slen, np.uint8) for _ in range(m + l)]
        po.lrc_encode(n, m, l, az, shards)
        out[name + "/data"] = np.stack(shards[:n])
        out[name + "/parity"] = np.stack(shards[n:])
    # crc32block vectors
    for idx, size in enumerate([100, 4096, 65532, 65533, 200000]):
        rng = np.random.default_rng(np.random.PCG64(0xC2C32B10 ^ idx))
        raw = rng.integers(0, 256, size, dtype=np.uint8)
        framed = po.crc32b_encode(raw)
        out["crc%d/raw" % size] = raw
        out["crc%d/framed" % size] = framed
    np.savez_compressed(OUT, **out)
    print("wrote %s (%d arrays)" % (OUT, len(out)))


if __name__ == "__main__":
    main()
