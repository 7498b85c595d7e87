"""CPU oracle tests: GF tables pinned to the reference's literal tables,
RS/LRC round-trip properties mirroring blobstore/common/ec/encoder_test.go,
and the frozen golden vectors."""
import os

import numpy as np
import pytest


def test_tables_match_reference_literals(oracle, golden_dir):
    """oracle-generated tables byte-identical to galois.go:28,70,81,340,596."""
    golden = np.fromfile(os.path.join(golden_dir, "gf_tables.bin"), dtype=np.uint8)
    assert golden.size == 74494
    assert np.array_equal(oracle.gf_tables(), golden)


def test_gf_mul_properties(oracle):
    # mul/exp agree with the log/exp definition (galois.go:855-906)
    g = oracle.gf_tables()
    mul = g[766:766 + 65536].reshape(256, 256)
    assert oracle.gf_mul(0, 77) == 0 and oracle.gf_mul(77, 0) == 0
    for a in (1, 2, 3, 0x1d, 200, 255):
        assert oracle.gf_mul(1, a) == a
        for b in (1, 5, 91, 254):
            assert oracle.gf_mul(a, b) == mul[a][b] == oracle.gf_mul(b, a)
    # galExp: a^1 = a; 2^8 = 0x1d (poly reduction)
    assert oracle.gf_exp(2, 8) == 0x1d
    assert oracle.gf_exp(5, 0) == 1 and oracle.gf_exp(0, 3) == 0


def test_build_matrix_systematic(oracle):
    """Top k×k of the encode matrix is identity (reedsolomon.go:226-231)."""
    for k, m in [(4, 2), (6, 3), (12, 4), (16, 20), (24, 8), (1, 1), (10, 4)]:
        em = oracle.build_matrix(k, k + m)
        assert np.array_equal(em[:k], np.eye(k, dtype=np.uint8))


def test_invert_roundtrip(oracle):
    rng = np.random.default_rng(7)
    em = oracle.build_matrix(6, 9)
    sub = em[[0, 2, 4, 6, 7, 8]]  # any k rows are invertible (Vandermonde)
    inv = oracle.invert_matrix(sub)
    # sub × inv == I over GF
    prod = np.zeros((6, 6), dtype=np.uint8)
    for r in range(6):
        for c in range(6):
            v = 0
            for i in range(6):
                v ^= oracle.gf_mul(sub[r, i], inv[i, c])
            prod[r, c] = v
    assert np.array_equal(prod, np.eye(6, dtype=np.uint8))
    with pytest.raises(ValueError):
        oracle.invert_matrix(np.zeros((3, 3), dtype=np.uint8))


def test_golden_vectors(oracle, golden_dir):
    z = np.load(os.path.join(golden_dir, "rs_vectors.npz"))
    cases = sorted({k.split("/")[0] for k in z.files if k.startswith("EC") or k.startswith("LRC")})
    assert cases
    for name in cases:
        data = z[name + "/data"]
        parity = z[name + "/parity"]
        import re
        mm = re.match(r"(?:EC|LRC)(\d+)P(\d+)(?:L(\d+))?_(\d+)", name)
        n, m = int(mm.group(1)), int(mm.group(2))
        l = int(mm.group(3) or 0)
        az = 2 if l else 1
        shards = [data[i].copy() for i in range(n)] + \
                 [np.zeros(data.shape[1], np.uint8) for _ in range(m + l)]
        oracle.lrc_encode(n, m, l, az, shards)
        got = np.stack(shards[n:])
        assert np.array_equal(got, parity), name


@pytest.mark.parametrize("k,m,slen", [(4, 2, 1024), (6, 3, 2048), (6, 3, 1000),
                                      (12, 4, 4096), (3, 3, 512), (24, 8, 777),
                                      (1, 1, 100), (16, 4, 2048)])
def test_roundtrip(oracle, k, m, slen):
    """Mirrors encoder_test.go:53-106: encode → verify → corrupt/lose →
    reconstruct → identical."""
    rng = np.random.default_rng(k * 1000 + m)
    shards = [rng.integers(0, 256, slen, dtype=np.uint8) for _ in range(k)] + \
             [np.zeros(slen, np.uint8) for _ in range(m)]
    oracle.rs_encode(k, m, shards)
    assert oracle.rs_verify(k, m, shards)
    ref = [s.copy() for s in shards]

    # lose up to m arbitrary shards (data+parity mix)
    lose = list(rng.choice(k + m, size=min(m, k + m - 1), replace=False))
    present = np.ones(k + m, np.uint8)
    for i in lose:
        present[i] = 0
        shards[i][:] = 0
    rc = oracle.rs_reconstruct(k, m, shards, present)
    assert rc == 0
    for i in range(k + m):
        assert np.array_equal(shards[i], ref[i]), i

    # corrupt a parity byte → verify false
    shards[k][0] ^= 0xA5
    assert not oracle.rs_verify(k, m, shards)
    shards[k][0] ^= 0xA5

    # too few shards
    present = np.zeros(k + m, np.uint8)
    present[:k - 1] = 1
    rc = oracle.rs_reconstruct(k, m, shards, present)
    assert rc == -2  # ErrTooFewShards


def test_reconstruct_data_only(oracle):
    k, m, slen = 6, 3, 2048
    rng = np.random.default_rng(5)
    shards = [rng.integers(0, 256, slen, dtype=np.uint8) for _ in range(k)] + \
             [np.zeros(slen, np.uint8) for _ in range(m)]
    oracle.rs_encode(k, m, shards)
    ref = [s.copy() for s in shards]
    present = np.ones(k + m, np.uint8)
    present[[1, 7]] = 0
    shards[1][:] = 0
    shards[7][:] = 0
    assert oracle.rs_reconstruct(k, m, shards, present, data_only=True) == 0
    assert np.array_equal(shards[1], ref[1])      # data restored
    assert not np.array_equal(shards[7], ref[7])  # parity untouched


def test_decode_matrix_selection(oracle):
    """Valid-row selection must be 'first k present in index order'
    (reedsolomon.go:1453-1466)."""
    k, m = 6, 3
    present = np.ones(k + m, np.uint8)
    present[[0, 4]] = 0
    rows, valid = oracle.rs_decode_matrix(k, m, present)
    assert list(valid) == [1, 2, 3, 5, 6, 7]


@pytest.mark.parametrize("n,m,l,az", [(12, 2, 2, 2), (6, 10, 2, 2),
                                      (16, 20, 2, 2), (6, 3, 3, 3), (4, 4, 2, 2)])
def test_lrc_roundtrip(oracle, n, m, l, az):
    """Mirrors encoder_test.go LRC cases: global + local encode, local
    stripe self-consistency, reconstruct incl. local parities."""
    slen = 1024
    rng = np.random.default_rng(n * 100 + m)
    shards = [rng.integers(0, 256, slen, dtype=np.uint8) for _ in range(n)] + \
             [np.zeros(slen, np.uint8) for _ in range(m + l)]
    oracle.lrc_encode(n, m, l, az, shards)
    ref = [s.copy() for s in shards]

    ln, lm = (n + m) // az, l // az
    for a in range(az):
        idx = oracle.lrc_local_stripe(n, m, l, az, a)
        loc = [shards[i].copy() for i in idx]
        assert oracle.rs_verify(ln, lm, loc), "local stripe %d" % a

    # lose a data shard, a global parity and a local parity
    lose = [1, n, n + m]
    present = np.ones(n + m + l, np.uint8)
    for i in lose:
        present[i] = 0
        shards[i][:] = 0
    assert oracle.lrc_reconstruct(n, m, l, az, shards, present) == 0
    for i in range(n + m + l):
        assert np.array_equal(shards[i], ref[i]), i


def test_local_stripe_layout(oracle):
    """EC6P10L2 layout from the reference's own comment
    (codemode.go:152-158): local stripe1 = [0,1,2, 6..10, 16]."""
    assert oracle.lrc_local_stripe(6, 10, 2, 2, 0) == [0, 1, 2, 6, 7, 8, 9, 10, 16]
    assert oracle.lrc_local_stripe(6, 10, 2, 2, 1) == [3, 4, 5, 11, 12, 13, 14, 15, 17]


def test_buffer_sizes(oracle):
    """buf_test.go:45-174 semantics."""
    # 8 MiB data, EC6P3, min 2 KiB
    ss, eds, es = oracle.buffer_sizes(6, 3, 0, 2048, 8 << 20)
    assert ss == (8 * 1024 * 1024 + 5) // 6
    assert eds == ss * 6 and es == ss * 9
    # tiny data aligns up to MinShardSize
    ss, eds, es = oracle.buffer_sizes(6, 3, 0, 2048, 100)
    assert ss == 2048 and eds == 12288 and es == 18432
    ss, _, _ = oracle.buffer_sizes(6, 3, 0, 0, 100)
    assert ss == 17  # ceil(100/6), Align0


def test_rs_encode_mt_matches_scalar(oracle):
    """The chunked OpenMP/AVX2 baseline path is bit-identical to the
    scalar oracle across ragged lengths and stripe counts."""
    import numpy as np
    rng = np.random.default_rng(77)
    for k, m, slen, ns in [(6, 3, 200_000, 3), (6, 3, 131_072, 1),
                           (12, 4, 50_001, 5), (4, 2, 131_073, 2)]:
        stripes = []
        ref = []
        for _ in range(ns):
            st = [rng.integers(0, 256, slen, dtype=np.uint8)
                  for _ in range(k)]
            st += [np.zeros(slen, np.uint8) for _ in range(m)]
            stripes.append(st)
            r = [x.copy() for x in st]
            oracle.rs_encode(k, m, r)
            ref.append(r)
        oracle.rs_encode_mt(k, m, stripes)
        for s in range(ns):
            for i in range(k + m):
                assert np.array_equal(stripes[s][i], ref[s][i]), (k, m, s, i)
