/* oracle/oracle.h — CPU oracle for the CubeFS blobstore EC/CRC hot path.
 *
 * TEST INFRASTRUCTURE ONLY.  This library is a scalar C restatement of the
 * reference algorithms (klauspost/reedsolomon v1.11.7 as vendored by
 * cubefs/cubefs v3.5.3, plus blobstore/common/{ec,codemode,crc32block}).
 * It exists so the HIP engine can be checked bit-for-bit.  Only tests/,
 * __graft_entry__.smoke() and bench.py's cpu_baseline leg may link or load
 * it.  The product path (cubefs_amd/libgfrs.so) must never call into it.
 *
 * Parity pinning: the reference ships no golden RS parity vectors (SURVEY.md
 * §4); RS bytes are pinned algorithmically by the vendored source and its
 * literal GF tables.  This oracle is pinned by (a) byte-identity of its
 * generated tables against tests/golden/gf_tables.bin (extracted verbatim
 * from the vendored galois.go), (b) the universal CRC32-IEEE known-answer
 * test, and (c) round-trip property tests mirroring the reference's own
 * ec/encoder_test.go and crc32block tests.
 */
#ifndef GFRS_ORACLE_H
#define GFRS_ORACLE_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* Error codes (shared numbering with include/gfrs.h). */
#define ORC_OK 0
#define ORC_ERR_TOO_FEW_SHARDS (-2)   /* reedsolomon.ErrTooFewShards */
#define ORC_ERR_SHARD_SIZE (-3)       /* reedsolomon.ErrShardSize */
#define ORC_ERR_INVALID_ARG (-6)
#define ORC_ERR_SHORT_DATA (-7)       /* reedsolomon.ErrShortData */
#define ORC_ERR_SINGULAR (-8)         /* matrix.errSingular */
#define ORC_ERR_MISMATCHED_CRC (-9)   /* crc32block.ErrMismatchedCrc */
#define ORC_ERR_INVALID_BLOCK (-10)   /* crc32block.ErrInvalidBlock */

/* ---- GF(2^8), poly 0x11D (generator 29; galois.go:25) ---- */

/* Writes the generated tables in the exact layout of
 * tests/golden/gf_tables.bin: logTable(256) ‖ expTable(510) ‖
 * mulTable(65536) ‖ mulTableLow(4096) ‖ mulTableHigh(4096) = 74,494 B. */
void orc_gf_tables(uint8_t *out, size_t out_len);
uint8_t orc_gf_mul(uint8_t a, uint8_t b);
uint8_t orc_gf_exp(uint8_t a, int n); /* galois.go:892 galExp */

/* ---- matrix algebra (matrix.go) ---- */

/* Vandermonde(rows, cols): m[r][c] = galExp(r, c) (matrix.go:271-282). */
void orc_vandermonde(int rows, int cols, uint8_t *out);
/* Gauss-Jordan inverse over GF(2^8) (matrix.go:193-266).
 * in/out are n*n row-major.  Returns 0 or ORC_ERR_SINGULAR. */
int orc_invert_matrix(const uint8_t *in, int n, uint8_t *out);
/* klauspost default encode matrix (reedsolomon.go:220-244):
 * vandermonde(k+m, k) × inverse(top k×k).  out is (k+m)*k row-major. */
int orc_build_matrix(int k, int total, uint8_t *out);

/* ---- Reed-Solomon on shard arrays (reedsolomon.go) ---- */

/* shards: array of k+m pointers, each shard len bytes.  Parity rows are
 * overwritten (reedsolomon.go:609 Encode). */
int orc_rs_encode(int k, int m, uint8_t **shards, size_t len);
/* Returns 1 when parity matches, 0 when not, <0 on error
 * (reedsolomon.go:770 Verify). */
int orc_rs_verify(int k, int m, uint8_t *const *shards, size_t len);
/* present[i] != 0 means shard i is intact.  Missing shards' buffers must be
 * allocated by the caller (len bytes each); they are filled in.
 * data_only != 0 mirrors ReconstructData (reedsolomon.go:1375,1407). */
int orc_rs_reconstruct(int k, int m, uint8_t **shards, size_t len,
                       const uint8_t *present, int data_only);
/* Expose the k×k decode matrix for a missing pattern: out_rows is k*k
 * row-major inverse of the valid-row submatrix; out_valid lists the k valid
 * shard indices used (reedsolomon.go:1446-1501). */
int orc_rs_decode_matrix(int k, int m, const uint8_t *present,
                         uint8_t *out_rows, int *out_valid);

/* ---- blobstore/common/ec layer ---- */

/* LRC encode (lrcencoder.go:35-82): global RS(n,mm) over shards[0..n+mm),
 * then per-AZ local RS((n+mm)/az, l/az) over the AZ's local stripe. */
int orc_lrc_encode(int n, int mm, int l, int az, uint8_t **shards, size_t len);
/* LRC reconstruct, full shard set form (lrcencoder.go:133-186). */
int orc_lrc_reconstruct(int n, int mm, int l, int az, uint8_t **shards,
                        size_t len, const uint8_t *present, int data_only);
/* Local stripe global-indices for one AZ (codemode.go:301-318,365-372).
 * out_idx must hold (n+mm+l)/az ints.  Returns that count. */
int orc_lrc_local_stripe(int n, int mm, int l, int az, int az_idx,
                         int *out_idx);
/* ec.Buffer size math (buf.go:67-133). */
int orc_buffer_sizes(int n, int mm, int l, int min_shard_size,
                     long long data_size, long long *shard_size,
                     long long *ec_data_size, long long *ec_size);

/* ---- crc32block (blobstore/common/crc32block) ---- */

/* Go hash/crc32 Update semantics on finalized values:
 * ChecksumIEEE(p) == orc_crc32(0, p, n). */
uint32_t orc_crc32(uint32_t crc, const uint8_t *buf, size_t len);
/* zlib-style combine: crc32(A‖B) == orc_crc32_combine(crc32(A), crc32(B), lenB). */
uint32_t orc_crc32_combine(uint32_t crc1, uint32_t crc2, int64_t len2);
/* x^(8*len) mod P in the reflected domain — the per-chunk fold operator the
 * GPU kernel uses.  orc_crc32_combine(c1,c2,l) ==
 * orc_crc32_shift(c1, l) ^ c2 must hold (tested). */
uint32_t orc_crc32_shift(uint32_t crc, int64_t len_bytes);

int64_t orc_crc32b_encode_size(int64_t size, int64_t block_len); /* util.go:56 */
int64_t orc_crc32b_decode_size(int64_t size, int64_t block_len); /* util.go:65 */
/* Frame src (n bytes) into dst: per block, 4 B LE CRC32-IEEE ‖ payload
 * (block.go:22-49, encode.go:86-106).  Returns bytes written or <0. */
int64_t orc_crc32b_encode(uint8_t *dst, const uint8_t *src, int64_t n,
                          int64_t block_len);
/* Check every frame.  Returns -1 if all blocks pass, else the index of the
 * first bad block (decode.go:84-107). */
int64_t orc_crc32b_verify(const uint8_t *framed, int64_t framed_len,
                          int64_t block_len);
/* Strip frames into dst, checking CRCs.  Returns payload bytes written or
 * ORC_ERR_MISMATCHED_CRC. */
int64_t orc_crc32b_decode(uint8_t *dst, const uint8_t *framed,
                          int64_t framed_len, int64_t block_len);

/* ---- sized coder (crc32block/sized_coder.go: payload ‖ CRC32 BE per
 * block; stream zero-padded to the 512-B transport alignment;
 * util.go:73-94 Partial sizes) ---- */
int64_t orc_partial_encode_size(int64_t actual, int64_t stable,
                                int64_t block_len, int64_t *tail);
int64_t orc_partial_decode_size(int64_t total, int64_t tail, int64_t stable,
                                int64_t block_len);
int64_t orc_sized_encode(uint8_t *dst, const uint8_t *src, int64_t n,
                         int64_t block_len);
int64_t orc_sized_verify(const uint8_t *framed, int64_t total, int64_t tail,
                         int64_t block_len);
int64_t orc_sized_decode(uint8_t *dst, const uint8_t *framed, int64_t total,
                         int64_t tail, int64_t block_len);

/* ---- blobnode on-disk shard image (core/shard.go:42-111,
 * datafile.go:342-407): 32 B header ‖ crc32block body ‖ 8 B footer. ---- */
int64_t orc_shard_disk_size(int64_t size, int64_t block_len);
int64_t orc_shard_write(uint8_t *dst, const uint8_t *src, int64_t size,
                        int64_t block_len, uint64_t bid, uint64_t vuid);
/* Returns 0 and fills bid/vuid/psize, or ORC_ERR_MISMATCHED_CRC. */
int orc_shard_parse(const uint8_t *img, int64_t img_len, int64_t block_len,
                    uint64_t *bid, uint64_t *vuid, uint32_t *psize);

/* ---- multithreaded CPU baseline (bench.py cpu_baseline leg) ----
 * Identical nibble-table algorithm (galois_amd64.go:37-52 semantics),
 * OpenMP across stripes; AVX2 pshufb inner loop when compiled in.
 * shards_flat holds nstripes*(k+m) pointers.  Returns 0/err. */
int orc_rs_encode_mt(int k, int m, uint8_t **shards_flat, size_t len,
                     int nstripes, int nthreads);
int orc_crc32b_encode_mt(uint8_t **dst, uint8_t **src, int64_t n,
                         int64_t block_len, int nshards, int nthreads);
int orc_threads_avail(void);

#ifdef __cplusplus
}
#endif
#endif /* GFRS_ORACLE_H */
