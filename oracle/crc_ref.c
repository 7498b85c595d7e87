/* oracle/crc_ref.c — CRC32-IEEE + crc32block framing oracle.
 *
 * TEST INFRASTRUCTURE ONLY (see oracle.h header).
 *
 * Restates:
 *   Go hash/crc32 IEEE (poly 0xEDB88320 reflected; Update/ChecksumIEEE)
 *   blobstore/common/crc32block/block.go:22-49   (frame = 4 B LE crc ‖ payload)
 *   blobstore/common/crc32block/util.go:28-71    (size math; block multiple of 4096)
 *   blobstore/common/crc32block/encode.go:86-106 (per-block framing, short tail)
 *   blobstore/common/crc32block/decode.go:84-107 (per-block check)
 *   zlib crc32_combine (GF(2) x^(8len) operator) — not in the reference, but
 *   needed to pin the GPU's parallel per-chunk CRC fold; proven against
 *   whole-buffer CRCs in tests.
 */
#include "oracle.h"

#include <string.h>

#define POLY 0xEDB88320u
#define CRC_LEN 4
#define BASE_BLOCK_LEN 4096

static uint32_t crc_tab[256];
static uint32_t crc_tab8[8][256]; /* slice-by-8 */
static int crc_ready = 0;

static void crc_init(void) {
    if (crc_ready) return;
    for (uint32_t i = 0; i < 256; i++) {
        uint32_t c = i;
        for (int j = 0; j < 8; j++) c = (c & 1) ? (c >> 1) ^ POLY : c >> 1;
        crc_tab[i] = c;
        crc_tab8[0][i] = c;
    }
    for (int t = 1; t < 8; t++)
        for (int i = 0; i < 256; i++)
            crc_tab8[t][i] =
                crc_tab[crc_tab8[t - 1][i] & 0xFF] ^ (crc_tab8[t - 1][i] >> 8);
    crc_ready = 1;
}

/* Slice-by-8 (the reference's stdlib CRC uses CLMUL assembly; bytewise
 * tables under-reported the CPU baseline's CRC leg ~8x). */
uint32_t orc_crc32(uint32_t crc, const uint8_t *buf, size_t len) {
    crc_init();
    uint32_t c = ~crc;
    size_t i = 0;
    for (; i + 8 <= len; i += 8) {
        uint32_t w0, w1;
        memcpy(&w0, buf + i, 4);
        memcpy(&w1, buf + i + 4, 4);
        w0 ^= c;
        c = crc_tab8[7][w0 & 0xFF] ^ crc_tab8[6][(w0 >> 8) & 0xFF] ^
            crc_tab8[5][(w0 >> 16) & 0xFF] ^ crc_tab8[4][w0 >> 24] ^
            crc_tab8[3][w1 & 0xFF] ^ crc_tab8[2][(w1 >> 8) & 0xFF] ^
            crc_tab8[1][(w1 >> 16) & 0xFF] ^ crc_tab8[0][w1 >> 24];
    }
    for (; i < len; i++) c = crc_tab[(c ^ buf[i]) & 0xFF] ^ (c >> 8);
    return ~c;
}

/* GF(2) carry-less multiply modulo P in the reflected domain.
 * Reflected rep: bit i = coefficient of x^(31-i), so the identity element
 * (x^0) is 0x80000000 and multiply-by-x is a right shift with reduction.
 * prod = a(x)·b(x) mod P: walk a's bits from bit 31 (x^0) down, adding
 * b·x^e for each set coefficient. */
static uint32_t gf2_mulmod(uint32_t a, uint32_t b) {
    uint32_t prod = 0;
    for (int i = 31; i >= 0; i--) {
        if ((a >> i) & 1) prod ^= b;
        b = (b & 1) ? (b >> 1) ^ POLY : b >> 1;
    }
    return prod;
}

/* x^(8*len) mod P (reflected).  x^1 in reflected form is 1<<30 (bit index
 * 31-1).  Computed by binary exponentiation on len. */
static uint32_t x8n_mod_p(int64_t len_bytes) {
    uint32_t op = 0x80000000u; /* identity: x^0 */
    uint32_t sq = 0x00800000u; /* x^8 reflected: bit 31-8 */
    int64_t n = len_bytes;
    while (n) {
        if (n & 1) op = gf2_mulmod(op, sq);
        sq = gf2_mulmod(sq, sq);
        n >>= 1;
    }
    return op;
}

uint32_t orc_crc32_shift(uint32_t crc, int64_t len_bytes) {
    return gf2_mulmod(x8n_mod_p(len_bytes), crc);
}

uint32_t orc_crc32_combine(uint32_t crc1, uint32_t crc2, int64_t len2) {
    /* zlib semantics: both inputs finalized, result finalized. */
    return orc_crc32_shift(crc1, len2) ^ crc2;
}

/* ---------------- crc32block ---------------- */

static int valid_block_len(int64_t block_len) {
    return block_len > 0 && block_len % BASE_BLOCK_LEN == 0; /* util.go:40 */
}

static int64_t aligned_blocks(int64_t size, int64_t unit) {
    return (size + unit - 1) / unit; /* util.AlignedBlocks */
}

int64_t orc_crc32b_encode_size(int64_t size, int64_t block_len) {
    if (!valid_block_len(block_len)) return ORC_ERR_INVALID_BLOCK;
    int64_t payload = block_len - CRC_LEN;
    return size + CRC_LEN * aligned_blocks(size, payload); /* util.go:56-62 */
}

int64_t orc_crc32b_decode_size(int64_t total, int64_t block_len) {
    if (!valid_block_len(block_len)) return ORC_ERR_INVALID_BLOCK;
    return total - CRC_LEN * aligned_blocks(total, block_len); /* util.go:65-71 */
}

int64_t orc_crc32b_encode(uint8_t *dst, const uint8_t *src, int64_t n,
                          int64_t block_len) {
    if (!valid_block_len(block_len)) return ORC_ERR_INVALID_BLOCK;
    int64_t payload = block_len - CRC_LEN;
    int64_t w = 0;
    for (int64_t off = 0; off < n; off += payload) {
        int64_t take = n - off < payload ? n - off : payload;
        uint32_t crc = orc_crc32(0, src + off, (size_t)take);
        dst[w + 0] = (uint8_t)(crc);
        dst[w + 1] = (uint8_t)(crc >> 8);
        dst[w + 2] = (uint8_t)(crc >> 16);
        dst[w + 3] = (uint8_t)(crc >> 24);
        memcpy(dst + w + CRC_LEN, src + off, (size_t)take);
        w += CRC_LEN + take;
    }
    return w;
}

int64_t orc_crc32b_verify(const uint8_t *framed, int64_t framed_len,
                          int64_t block_len) {
    if (!valid_block_len(block_len)) return ORC_ERR_INVALID_BLOCK;
    int64_t idx = 0;
    for (int64_t off = 0; off < framed_len; off += block_len, idx++) {
        int64_t blk = framed_len - off < block_len ? framed_len - off : block_len;
        if (blk <= CRC_LEN) return idx; /* torn frame */
        uint32_t want = (uint32_t)framed[off] | ((uint32_t)framed[off + 1] << 8) |
                        ((uint32_t)framed[off + 2] << 16) |
                        ((uint32_t)framed[off + 3] << 24);
        uint32_t got = orc_crc32(0, framed + off + CRC_LEN, (size_t)(blk - CRC_LEN));
        if (want != got) return idx;
    }
    return -1;
}

int64_t orc_crc32b_decode(uint8_t *dst, const uint8_t *framed,
                          int64_t framed_len, int64_t block_len) {
    int64_t bad = orc_crc32b_verify(framed, framed_len, block_len);
    if (bad == ORC_ERR_INVALID_BLOCK) return bad;
    if (bad >= 0) return ORC_ERR_MISMATCHED_CRC;
    int64_t w = 0;
    for (int64_t off = 0; off < framed_len; off += block_len) {
        int64_t blk = framed_len - off < block_len ? framed_len - off : block_len;
        memcpy(dst + w, framed + off + CRC_LEN, (size_t)(blk - CRC_LEN));
        w += blk - CRC_LEN;
    }
    return w;
}

/* ---------------- sized coder (sized_coder.go, util.go:73-94) ----------
 * ModeEncode with stableSize==0: frame = payload (<= block_len-4) ‖
 * 4-byte BIG-endian CRC32-IEEE of the payload (sized_coder.go:256-279,
 * be.Uint32 check :323-326); the whole encoded stream is zero-padded to
 * the 512-byte transport alignment (PartialEncodeSizeWith, util.go:73-80,
 * rpc2/transport/allocator.go:12-15). */

#define SIZED_ALIGN 512

int64_t orc_partial_encode_size(int64_t actual, int64_t stable,
                                int64_t block_len, int64_t *tail) {
    if (!valid_block_len(block_len)) return ORC_ERR_INVALID_BLOCK;
    int64_t payload = block_len - CRC_LEN;
    int64_t part = (stable % payload) & ~(int64_t)(SIZED_ALIGN - 1);
    int64_t pad = (stable % payload) % SIZED_ALIGN;
    int64_t size = orc_crc32b_encode_size(actual + part + pad, block_len) - part;
    int64_t t = (SIZED_ALIGN - (size & (SIZED_ALIGN - 1))) % SIZED_ALIGN;
    if (tail) *tail = t;
    return size + t;
}

int64_t orc_partial_decode_size(int64_t total, int64_t tail, int64_t stable,
                                int64_t block_len) {
    if (!valid_block_len(block_len)) return ORC_ERR_INVALID_BLOCK;
    int64_t payload = block_len - CRC_LEN;
    int64_t part = (stable % payload) & ~(int64_t)(SIZED_ALIGN - 1);
    int64_t pad = (stable % payload) % SIZED_ALIGN;
    return orc_crc32b_decode_size(total - tail + part, block_len) - part - pad;
}

int64_t orc_sized_encode(uint8_t *dst, const uint8_t *src, int64_t n,
                         int64_t block_len) {
    if (!valid_block_len(block_len)) return ORC_ERR_INVALID_BLOCK;
    int64_t payload = block_len - CRC_LEN;
    int64_t w = 0;
    for (int64_t off = 0; off < n; off += payload) {
        int64_t take = n - off < payload ? n - off : payload;
        memcpy(dst + w, src + off, (size_t)take);
        uint32_t crc = orc_crc32(0, src + off, (size_t)take);
        dst[w + take + 0] = (uint8_t)(crc >> 24); /* big-endian */
        dst[w + take + 1] = (uint8_t)(crc >> 16);
        dst[w + take + 2] = (uint8_t)(crc >> 8);
        dst[w + take + 3] = (uint8_t)crc;
        w += take + CRC_LEN;
    }
    int64_t t = (SIZED_ALIGN - (w & (SIZED_ALIGN - 1))) % SIZED_ALIGN;
    memset(dst + w, 0, (size_t)t);
    return w + t;
}

int64_t orc_sized_verify(const uint8_t *framed, int64_t total, int64_t tail,
                         int64_t block_len) {
    if (!valid_block_len(block_len)) return ORC_ERR_INVALID_BLOCK;
    int64_t body = total - tail;
    int64_t idx = 0;
    for (int64_t off = 0; off < body; off += block_len, idx++) {
        int64_t blk = body - off < block_len ? body - off : block_len;
        if (blk <= CRC_LEN) return idx;
        int64_t plen = blk - CRC_LEN;
        uint32_t want = ((uint32_t)framed[off + plen] << 24) |
                        ((uint32_t)framed[off + plen + 1] << 16) |
                        ((uint32_t)framed[off + plen + 2] << 8) |
                        (uint32_t)framed[off + plen + 3];
        if (want != orc_crc32(0, framed + off, (size_t)plen)) return idx;
    }
    return -1;
}

int64_t orc_sized_decode(uint8_t *dst, const uint8_t *framed, int64_t total,
                         int64_t tail, int64_t block_len) {
    int64_t bad = orc_sized_verify(framed, total, tail, block_len);
    if (bad == ORC_ERR_INVALID_BLOCK) return bad;
    if (bad >= 0) return ORC_ERR_MISMATCHED_CRC;
    int64_t body = total - tail;
    int64_t w = 0;
    for (int64_t off = 0; off < body; off += block_len) {
        int64_t blk = body - off < block_len ? body - off : block_len;
        memcpy(dst + w, framed + off, (size_t)(blk - CRC_LEN));
        w += blk - CRC_LEN;
    }
    return w;
}
