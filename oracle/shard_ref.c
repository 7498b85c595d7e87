/* oracle/shard_ref.c — blobnode on-disk shard image oracle.
 *
 * TEST INFRASTRUCTURE ONLY (see oracle.h header).
 *
 * Restates core/shard.go:42-111 (header/footer layout, big-endian fields,
 * magics ab cd ef cc / cc ef cd ab) and datafile.go:330-407 (header ‖
 * crc32block body ‖ footer; footer crc = CRC32-IEEE of the raw unframed
 * data via the TeeReader at datafile.go:338-340).
 */
#include "oracle.h"

#include <string.h>

static void be32(uint8_t *p, uint32_t v) {
    p[0] = (uint8_t)(v >> 24); p[1] = (uint8_t)(v >> 16);
    p[2] = (uint8_t)(v >> 8); p[3] = (uint8_t)v;
}
static uint32_t rd_be32(const uint8_t *p) {
    return ((uint32_t)p[0] << 24) | ((uint32_t)p[1] << 16) |
           ((uint32_t)p[2] << 8) | p[3];
}
static void be64(uint8_t *p, uint64_t v) {
    be32(p, (uint32_t)(v >> 32));
    be32(p + 4, (uint32_t)v);
}
static uint64_t rd_be64(const uint8_t *p) {
    return ((uint64_t)rd_be32(p) << 32) | rd_be32(p + 4);
}

int64_t orc_shard_disk_size(int64_t size, int64_t block_len) {
    int64_t body = orc_crc32b_encode_size(size, block_len);
    if (body < 0) return body;
    return 32 + body + 8;
}

int64_t orc_shard_write(uint8_t *dst, const uint8_t *src, int64_t size,
                        int64_t block_len, uint64_t bid, uint64_t vuid) {
    int64_t body = orc_crc32b_encode(dst + 32, src, size, block_len);
    if (body < 0) return body;
    /* header (shard.go:241-261) */
    memset(dst, 0, 32);
    dst[4] = 0xab; dst[5] = 0xcd; dst[6] = 0xef; dst[7] = 0xcc;
    be64(dst + 8, bid);
    be64(dst + 16, vuid);
    be32(dst + 24, (uint32_t)size);
    be32(dst, orc_crc32(0, dst + 4, 28));
    /* footer (shard.go:264-275): crc of raw data */
    uint8_t *ftr = dst + 32 + body;
    ftr[0] = 0xcc; ftr[1] = 0xef; ftr[2] = 0xcd; ftr[3] = 0xab;
    be32(ftr + 4, orc_crc32(0, src, (size_t)size));
    return 32 + body + 8;
}

int orc_shard_parse(const uint8_t *img, int64_t img_len, int64_t block_len,
                    uint64_t *bid, uint64_t *vuid, uint32_t *psize) {
    if (img_len < 40) return ORC_ERR_INVALID_ARG;
    if (!(img[4] == 0xab && img[5] == 0xcd && img[6] == 0xef &&
          img[7] == 0xcc))
        return ORC_ERR_MISMATCHED_CRC;
    if (rd_be32(img) != orc_crc32(0, img + 4, 28))
        return ORC_ERR_MISMATCHED_CRC;
    *bid = rd_be64(img + 8);
    *vuid = rd_be64(img + 16);
    *psize = rd_be32(img + 24);
    int64_t body = orc_crc32b_encode_size(*psize, block_len);
    if (body < 0 || 32 + body + 8 != img_len) return ORC_ERR_INVALID_ARG;
    if (orc_crc32b_verify(img + 32, body, block_len) != -1)
        return ORC_ERR_MISMATCHED_CRC;
    const uint8_t *ftr = img + 32 + body;
    if (!(ftr[0] == 0xcc && ftr[1] == 0xef && ftr[2] == 0xcd &&
          ftr[3] == 0xab))
        return ORC_ERR_MISMATCHED_CRC;
    /* footer crc vs decoded payload */
    uint32_t want = rd_be32(ftr + 4);
    /* recompute raw crc by combining the per-frame payload CRCs */
    uint32_t crc = 0;
    int64_t payload = block_len - 4, remain = *psize;
    int first = 1;
    for (int64_t off = 32; remain > 0; off += block_len) {
        int64_t plen = remain < payload ? remain : payload;
        uint32_t fc = (uint32_t)img[off] | ((uint32_t)img[off + 1] << 8) |
                      ((uint32_t)img[off + 2] << 16) |
                      ((uint32_t)img[off + 3] << 24);
        crc = first ? fc : orc_crc32_combine(crc, fc, plen);
        first = 0;
        remain -= plen;
    }
    if (want != crc) return ORC_ERR_MISMATCHED_CRC;
    return ORC_OK;
}
