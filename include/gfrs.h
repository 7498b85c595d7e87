/* include/gfrs.h — C ABI of the MI355X-native erasure-coding + checksum
 * engine for CubeFS blobstore ("gfrs" = GF(2^8) Reed-Solomon).
 *
 * This is the drop-in boundary replacing the compute underneath
 * blobstore/common/ec.Encoder (encoder.go:41-62) and
 * blobstore/common/crc32block.  The reference has no FFI of its own (the
 * arithmetic is reached through Go interfaces), so each entry point below
 * names the Go interface method it replaces; INTEGRATION.md shows the cgo
 * shim a CubeFS maintainer would add so access/stream and blobnode/worker
 * call this engine unchanged.
 *
 * Memory model: shard buffers are caller-owned, equal-length; data shards
 * are read-only during Encode (reedsolomon.go:30-32).  Pointers may be HIP
 * device pointers (GFRS_MEM_DEVICE) or plain host pointers
 * (GFRS_MEM_HOST, staged through pinned buffers internally — the cgo path).
 * A context is thread-safe and may be shared, like one ec.Encoder shared by
 * many goroutines (encoder.go:114-151).
 */
#ifndef GFRS_H
#define GFRS_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* Error codes: one per Go sentinel the cgo shim maps back.
 * 0 = success; negative = failure. */
enum {
  GFRS_OK = 0,
  GFRS_ERR_INVALID_CODEMODE = -1, /* ec.ErrInvalidCodeMode (encoder.go:35) */
  GFRS_ERR_TOO_FEW_SHARDS = -2,   /* reedsolomon.ErrTooFewShards (:598) */
  GFRS_ERR_SHARD_SIZE = -3,       /* reedsolomon.ErrShardSize */
  GFRS_ERR_SHARD_NO_DATA = -4,    /* reedsolomon.ErrShardNoData */
  GFRS_ERR_VERIFY = -5,           /* ec.ErrVerify (encoder.go:36) */
  GFRS_ERR_INVALID_SHARDS = -6,   /* ec.ErrInvalidShards (encoder.go:37) */
  GFRS_ERR_SHORT_DATA = -7,       /* ec.ErrShortData / rs.ErrShortData */
  GFRS_ERR_SINGULAR = -8,         /* matrix.errSingular (matrix.go:186) */
  GFRS_ERR_MISMATCHED_CRC = -9,   /* crc32block.ErrMismatchedCrc */
  GFRS_ERR_INVALID_BLOCK = -10,   /* crc32block.ErrInvalidBlock */
  GFRS_ERR_READ_ON_CLOSED = -11,  /* crc32block.ErrReadOnClosed
                                     (request_body.go streaming bodies) */
  GFRS_ERR_HIP = -100,            /* HIP runtime failure (see gfrs_last_error) */
  GFRS_ERR_NO_GPU = -101,         /* no MI355X visible; the engine never
                                     falls back to CPU — by design */
  GFRS_ERR_NOMEM = -102,
  GFRS_ERR_UNSUPPORTED = -103,
};

/* Where shard pointers live. */
enum { GFRS_MEM_DEVICE = 0, GFRS_MEM_HOST = 1 };

/* codemode.Tactic (codemode.go:156-190). */
typedef struct gfrs_tactic {
  int32_t n;              /* data shards */
  int32_t m;              /* global parity shards */
  int32_t l;              /* local (AZ) parity shards, 0 for plain RS */
  int32_t az_count;       /* AZ count; n,m,l all divisible by it */
  int32_t put_quorum;     /* carried for the shim; unused by compute */
  int32_t get_quorum;
  int32_t min_shard_size; /* ec.Buffer alignment floor (buf.go:79-83) */
} gfrs_tactic;

typedef struct gfrs_ctx gfrs_ctx;

/* ---- lifecycle ---- */

/* ec.NewEncoder(ec.Config{CodeMode: t}) (encoder.go:78-112).
 * device < 0 selects the current HIP device.  Returns NULL on failure
 * (gfrs_last_error() has the reason). */
gfrs_ctx *gfrs_create(const gfrs_tactic *t, int device);
void gfrs_destroy(gfrs_ctx *ctx);
const char *gfrs_last_error(void);
const char *gfrs_version(void);
int gfrs_device_count(void);

/* Optional: run on an existing HIP stream (e.g. torch's). stream==NULL
 * reverts to the context's own stream. */
int gfrs_set_stream(gfrs_ctx *ctx, void *hip_stream);
int gfrs_synchronize(gfrs_ctx *ctx);

/* ---- single stripe (the ec.Encoder surface) ----
 * shards: n+m+l pointers, each shard_len bytes, all in `memloc` memory.
 * Single-stripe calls BLOCK until the result is ready (ec.Encoder
 * semantics: the caller owns the buffers on return).  The batch APIs
 * below are async on the ctx stream; synchronize with
 * gfrs_synchronize().  When the shards form one contiguous ec.Buffer
 * layout (buf.go:24-35) the pointer-table upload is skipped. */

/* Encoder.Encode (encoder.go:114 / lrcencoder.go:35): fills parity (and
 * local parity when l > 0) from the data shards. */
int gfrs_encode(gfrs_ctx *ctx, void *const *shards, size_t shard_len,
                int nshards, int memloc);

/* Encoder.Verify (encoder.go:133): *ok=1 when parity matches. */
int gfrs_verify(gfrs_ctx *ctx, void *const *shards, size_t shard_len,
                int nshards, int memloc, int *ok);

/* Encoder.Reconstruct / ReconstructData (encoder.go:139-151,
 * lrcencoder.go:133-207): bad_idx lists missing shard indices; their
 * buffers must be allocated (shard_len bytes) and are overwritten. */
int gfrs_reconstruct(gfrs_ctx *ctx, void *const *shards, size_t shard_len,
                     int nshards, int memloc, const int32_t *bad_idx,
                     int nbad, int data_only);

/* ---- stripe batches (blobnode repair/migrate bulk path,
 * worker_slice_recover.go:804-888) ----
 * Stripes laid out contiguously: stripe s shard i starts at
 * base + s*stripe_stride + i*shard_len (the ec.Buffer layout, buf.go:24-35).
 * base is always a device pointer. */

int gfrs_encode_batch(gfrs_ctx *ctx, void *base, size_t shard_len,
                      size_t stripe_stride, int nstripes);
/* fail_bitmap (host, optional): bit s set when stripe s fails verify. */
int gfrs_verify_batch(gfrs_ctx *ctx, const void *base, size_t shard_len,
                      size_t stripe_stride, int nstripes,
                      uint64_t *fail_bitmap);
/* One missing pattern shared by the whole batch (one repair tasklet). */
int gfrs_reconstruct_batch(gfrs_ctx *ctx, void *base, size_t shard_len,
                           size_t stripe_stride, int nstripes,
                           const int32_t *bad_idx, int nbad, int data_only);

/* ---- crc32block (blobstore/common/crc32block) ----
 * block_len must be a positive multiple of 4096 (util.go:40); frame =
 * 4 B LE CRC32-IEEE ‖ payload (block.go:22-49).  dst/src are device
 * pointers; *_host variants stage host memory. */

int64_t gfrs_crc32b_encode_size(int64_t size, int64_t block_len);
int64_t gfrs_crc32b_decode_size(int64_t size, int64_t block_len);

/* Host-side CRC32-IEEE in hash/crc32 Update semantics: pass the previous
 * finalized crc (0 for a fresh start) and get the finalized crc of the
 * concatenation.  This exists for the per-block STREAMING request-body
 * wrapper (crc32block/request_body.go:57-127), which in the reference
 * runs on the client/server host too — every bulk path computes CRCs on
 * device.  Needs no ctx and no GPU. */
uint32_t gfrs_crc32_host(uint32_t crc, const void *data, int64_t n);

/* Frame n raw bytes from src into dst (encode.go:86-106).
 * dst capacity must be gfrs_crc32b_encode_size(n).  Returns bytes
 * written or a negative error. */
int64_t gfrs_crc32b_encode(gfrs_ctx *ctx, void *dst, const void *src,
                           int64_t n, int64_t block_len);
/* Check all frames (decode.go:84-107).  *bad_block = first failing block
 * index or -1.  Returns GFRS_OK even when CRCs mismatch (inspect
 * *bad_block); negative only on launch/arg errors. */
int gfrs_crc32b_verify(gfrs_ctx *ctx, const void *framed, int64_t framed_len,
                       int64_t block_len, int64_t *bad_block);
/* Strip frames into dst, checking CRCs.  Returns payload bytes or
 * GFRS_ERR_MISMATCHED_CRC. */
int64_t gfrs_crc32b_decode(gfrs_ctx *ctx, void *dst, const void *framed,
                           int64_t framed_len, int64_t block_len);

/* Batched framing of many equal-length shards (datafile.go:342-445 write
 * path): shard i raw at src + i*src_stride, framed at dst + i*dst_stride. */
int gfrs_crc32b_encode_batch(gfrs_ctx *ctx, void *dst, size_t dst_stride,
                             const void *src, size_t src_stride, int64_t n,
                             int64_t block_len, int nshards);
int gfrs_crc32b_verify_batch(gfrs_ctx *ctx, const void *framed,
                             size_t stride, int64_t framed_len,
                             int64_t block_len, int nshards,
                             int64_t *bad_block_per_shard);

/* ---- fused encode+frame (the PUT pipeline: stream_put.go:146 encode +
 * datafile.go:342 framing in ONE pass) ----
 * Reads the data shards once, computes parity in registers and writes
 * ONLY the k+m crc32block-framed shard images (frame image j of stripe s
 * at framed + (s*(n+m)+j)*framed_stride); the unframed parity never
 * touches HBM.  Requires block_len 65536, 1<=m<=4, n+m<=16, l==0 and
 * framed_stride % 4 == 0; other shapes fall back to
 * encode_batch + crc32b_encode_batch (which needs
 * stripe_stride == (n+m)*shard_len so the shard rows form one strided
 * array).  Framed bytes are identical either way. */
int gfrs_encode_frame_batch(gfrs_ctx *ctx, void *framed,
                            size_t framed_stride, void *base,
                            size_t shard_len, size_t stripe_stride,
                            int nstripes, int64_t block_len);

/* ---- sized coder (crc32block/sized_coder.go, the rpc2 body framing) ----
 * Frame = payload (block_len-4) ‖ CRC32-IEEE big-endian (ModeEncode,
 * sized_coder.go:256-279); the encoded stream is zero-padded to the
 * transport alignment of 512 (PartialEncodeSizeWith, util.go:73-80).
 * *size returns total bytes incl. tail pad; *tail the pad length. */
int gfrs_sized_encode_size(int64_t actual_size, int64_t block_len,
                           int64_t *size, int64_t *tail);
int64_t gfrs_sized_decode_size(int64_t total, int64_t tail, int64_t block_len);
/* Encode n raw bytes into dst (device); writes the tail pad.  Returns
 * total bytes written (incl. pad) or negative error. */
int64_t gfrs_sized_encode(gfrs_ctx *ctx, void *dst, const void *src,
                          int64_t n, int64_t block_len);
/* Verify all frames of a sized body (total includes tail pad). */
int gfrs_sized_verify(gfrs_ctx *ctx, const void *framed, int64_t total,
                      int64_t tail, int64_t block_len, int64_t *bad_block);
int64_t gfrs_sized_decode(gfrs_ctx *ctx, void *dst, const void *framed,
                          int64_t total, int64_t tail, int64_t block_len);

/* ---- blobnode on-disk shard image (core/shard.go:42-111,
 * datafile.go:342-445) ----
 * image = 32 B header (crc|magic|bid|vuid|size|reserved, big-endian) ‖
 * crc32block body ‖ 8 B footer (magic | crc32 of the raw data).  Writing
 * a repaired shard through this produces bytes directly pwrite()-able by
 * blobnode. */
int64_t gfrs_shard_disk_size(int64_t size, int64_t block_len);
/* Frame shard j (raw at src + j*src_stride, size bytes) into a full disk
 * image at dst + j*dst_stride; bids/vuids are per-shard metadata. */
int gfrs_shard_write_batch(gfrs_ctx *ctx, void *dst, size_t dst_stride,
                           const void *src, size_t src_stride, int64_t size,
                           int64_t block_len, const uint64_t *bids,
                           const uint64_t *vuids, int nshards);
/* Parse + fully verify images (header crc+magic, every body block crc,
 * footer magic + whole-shard crc).  out_meta: per shard
 * {bid, vuid, size, err} as 4 uint64 (err 0 or GFRS_ERR_MISMATCHED_CRC
 * cast); bad_block_per_shard as in crc32b_verify_batch. */
int gfrs_shard_parse_batch(gfrs_ctx *ctx, const void *img, size_t stride,
                           int64_t size, int64_t block_len,
                           uint64_t *out_meta, int64_t *bad_block_per_shard,
                           int nshards);

/* Fused reconstruct+verify (worker_slice_recover.go:804-888: the
 * mandatory Verify after Reconstruct, :865-874) in ONE data pass: the
 * decode rows are substituted into the parity-check rows so every output
 * — reconstructed shard or parity comparison — is a matrix apply over the
 * same k valid inputs.  Stored parity is compared in-register; bad_idx
 * shards are rewritten.  fail_bitmap bit s set when stripe s's parity
 * does not match.  Plain RS only (l == 0); LRC falls back to
 * reconstruct_batch + verify_batch. */
int gfrs_reconstruct_verify_batch(gfrs_ctx *ctx, void *base,
                                  size_t shard_len, size_t stripe_stride,
                                  int nstripes, const int32_t *bad_idx,
                                  int nbad, uint64_t *fail_bitmap);

/* ---- incremental parity (reedsolomon.go:631-668 EncodeIdx) ----
 * Adds data shard idx's contribution into the m parity shards:
 * parity[r] ^= coeff[r][idx]*data.  Parity must be zeroed before the
 * first call; each data shard delivered exactly once.  Device pointers. */
int gfrs_encode_idx(gfrs_ctx *ctx, const void *data_shard, int idx,
                    void *const *parity, size_t shard_len, int nparity);

/* Update (reedsolomon.go:676-766): replace data shard idx's content and
 * patch the parity in place: parity[r] ^= coeff[r][idx]*(old ^ new) —
 * applied as two accumulate passes by GF(2^8) linearity.  old_shard and
 * new_shard are device pointers; the caller swaps its own data buffer. */
int gfrs_update_idx(gfrs_ctx *ctx, const void *old_shard,
                    const void *new_shard, int idx, void *const *parity,
                    size_t shard_len, int nparity);

/* ---- fused repair pipeline (worker_slice_recover.go:804-888 +
 * datafile.go:342-407) ----
 * One stream-ordered call per repair tasklet: reconstruct the bad shards
 * of every stripe, verify all parity (mandatory Verify, :865-874), then
 * frame each repaired shard into a pwrite()-able disk image.  Intermediate
 * data never leaves HBM; on the fused path the raw reconstruction is
 * never materialized, so the bad shards' regions of `base` are left
 * UNSPECIFIED after the call (the reference worker also only consumes
 * the repaired bytes through the written bids).  For LRC the surviving
 * local parities serve as extra verify equations, so corruption is
 * detected even when the global stripe has no spare parity.
 *   disk_dst: nstripes*nbad images, image (s,b) at
 *             disk_dst + (s*nbad + b)*dst_stride
 *   bids/vuids: one per (stripe, bad shard), same order
 *   fail_bitmap: bit s set when stripe s failed verify (those images are
 *             still written; the caller drops them like the reference
 *             drops failed bids). */
int gfrs_repair_batch(gfrs_ctx *ctx, void *base, size_t shard_len,
                      size_t stripe_stride, int nstripes,
                      const int32_t *bad_idx, int nbad, void *disk_dst,
                      size_t dst_stride, int64_t block_len,
                      const uint64_t *bids, const uint64_t *vuids,
                      uint64_t *fail_bitmap);

/* ---- ec.Buffer size math (buf.go:67-133) ---- */
int gfrs_buffer_sizes(const gfrs_tactic *t, int64_t data_size,
                      int64_t *shard_size, int64_t *ec_data_size,
                      int64_t *ec_size);

/* ---- introspection for tests ---- */
/* Copy the (n+m)×n encode matrix into out (row-major). */
int gfrs_encode_matrix(gfrs_ctx *ctx, uint8_t *out);
/* GPU-free: compute the klauspost default encode matrix (total×k) so CPU
 * tests can pin the host math without a device. */
int gfrs_compute_encode_matrix(int k, int total, uint8_t *out);
/* Device probe used by tests: returns 1 when v_perm byte-select semantics
 * match the kernel's assumption (sel>=8 yields 0), 0 otherwise, <0 error. */
int gfrs_probe_perm(void);

#ifdef __cplusplus
}
#endif
#endif /* GFRS_H */
