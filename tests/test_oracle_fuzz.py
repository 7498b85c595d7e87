"""CPU-only randomized oracle self-consistency fuzz (seeded): round-trip
and algebraic invariants across random geometries — guards future oracle
refactors independently of the GPU suite."""
import numpy as np


def test_oracle_fuzz_roundtrip(oracle):
    rng = np.random.default_rng(0x0AC1E)
    for it in range(30):
        k = int(rng.integers(1, 17))
        m = int(rng.integers(1, 9))
        slen = int(rng.integers(1, 20001))
        shards = [rng.integers(0, 256, slen, dtype=np.uint8) for _ in range(k)]
        shards += [np.zeros(slen, np.uint8) for _ in range(m)]
        oracle.rs_encode(k, m, shards)
        ref = [s.copy() for s in shards]
        assert oracle.rs_verify(k, m, shards), (it, k, m)
        # encode is deterministic/idempotent
        oracle.rs_encode(k, m, shards)
        for i in range(k + m):
            assert np.array_equal(shards[i], ref[i])
        # lose a random recoverable set
        nbad = int(rng.integers(1, m + 1))
        bad = rng.choice(k + m, size=nbad, replace=False)
        present = np.ones(k + m, np.uint8)
        for i in bad:
            present[i] = 0
            shards[i][:] = 0
        assert oracle.rs_reconstruct(k, m, shards, present) == 0, (it, k, m)
        for i in range(k + m):
            assert np.array_equal(shards[i], ref[i]), (it, k, m, i)
        # GF linearity: encode(a XOR b) == encode(a) XOR encode(b)
        if slen <= 4096:
            a = [rng.integers(0, 256, slen, dtype=np.uint8) for _ in range(k)]
            b = [rng.integers(0, 256, slen, dtype=np.uint8) for _ in range(k)]
            ea = a + [np.zeros(slen, np.uint8) for _ in range(m)]
            eb = b + [np.zeros(slen, np.uint8) for _ in range(m)]
            ex = [x ^ y for x, y in zip(a, b)] + \
                 [np.zeros(slen, np.uint8) for _ in range(m)]
            oracle.rs_encode(k, m, ea)
            oracle.rs_encode(k, m, eb)
            oracle.rs_encode(k, m, ex)
            for r in range(m):
                assert np.array_equal(ex[k + r], ea[k + r] ^ eb[k + r])


def test_oracle_fuzz_crc(oracle):
    rng = np.random.default_rng(0xC4C)
    for it in range(20):
        n = int(rng.integers(1, 300001))
        raw = rng.integers(0, 256, n, dtype=np.uint8)
        bl = int(rng.choice([4096, 8192, 65536]))
        framed = oracle.crc32b_encode(raw, block_len=bl)
        assert framed.size == oracle.crc32b_encode_size(n, bl)
        assert oracle.crc32b_verify(framed, block_len=bl) == -1
        assert np.array_equal(oracle.crc32b_decode(framed, block_len=bl), raw)
        # flip one random bit -> detected in the right block
        pos = int(rng.integers(0, framed.size))
        framed[pos] ^= 1 << int(rng.integers(0, 8))
        assert oracle.crc32b_verify(framed, block_len=bl) == pos // bl
        # sized coder the same way
        sz, tail = oracle.partial_encode_size(n, 0, bl)
        sf, t2 = oracle.sized_encode(raw, block_len=bl)
        assert sf.size == sz and t2 == tail
        assert oracle.sized_verify(sf, tail, block_len=bl) == -1
        assert np.array_equal(oracle.sized_decode(sf, tail, block_len=bl), raw)
