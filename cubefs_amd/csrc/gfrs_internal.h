/* cubefs_amd/csrc/gfrs_internal.h — internal declarations shared between
 * the host runtime (gfrs_host.cpp) and kernel launchers (gfrs_kernels.hip).
 * Product code: independent of oracle/.
 */
#ifndef GFRS_INTERNAL_H
#define GFRS_INTERNAL_H

#include <cstddef>
#include <cstdint>
#include <vector>

#include <hip/hip_runtime.h>

namespace gfrs {

/* ---- GF(2^8) field, poly 0x11D (generator 29; galois.go:25) ---- */
struct GfTables {
  uint8_t log_t[256];
  uint8_t exp_t[510];
  uint8_t mul[256][256];
  /* nibble tables: lo[c][x] = mul[c][x], hi[c][x] = mul[c][x<<4]
   * (galois.go:340,596) */
  uint8_t lo[256][16];
  uint8_t hi[256][16];
  GfTables();
  uint8_t gexp(uint8_t a, int n) const; /* galois.go:892 */
  uint8_t div(uint8_t a, uint8_t b) const;
};
const GfTables &gft();

/* Row-major byte matrix helpers (matrix.go). */
bool gf_invert(const uint8_t *in, int n, uint8_t *out); /* false = singular */
/* klauspost default encode matrix: vandermonde(total,k) × inv(top k×k)
 * (reedsolomon.go:220-244).  out is total×k. */
bool gf_build_matrix(int k, int total, uint8_t *out);

/* ---- kernel launch interface ---- */

/* Matrix-apply over a stripe batch (serves encode and reconstruct).
 * For each stripe s (0..nstripes), output r (0..nout), byte i:
 *   out[r][i] = XOR_c coeff[r*k+c] * in[c][i]
 * where in/out shard pointers come from ptrs[s*nptr + idx]:
 *   in  c -> ptrs[s*nptr + in_idx[c]]
 *   out r -> ptrs[s*nptr + out_idx[r]]
 * tabs = device array [nout*k][32]: per-coefficient lo|hi nibble tables.
 * All index/table arrays are device memory. */
void launch_rs_apply(const uint64_t *ptrs, int nptr, const int32_t *in_idx,
                     int k, const int32_t *out_idx, int nout,
                     const uint8_t *tabs, size_t shard_len, int nstripes,
                     hipStream_t s);

/* Verify: recompute parity from tabs and OR a nonzero flag into
 * fail[s] for any mismatching stripe. */
void launch_rs_verify(const uint64_t *ptrs, int nptr, const int32_t *in_idx,
                      int k, const int32_t *out_idx, int nout,
                      const uint8_t *tabs, size_t shard_len, int nstripes,
                      uint32_t *fail, hipStream_t s);

/* crc32block kernels.  Shards framed independently; shard j raw bytes at
 * src + j*src_stride (n bytes), framed at dst + j*dst_stride.
 * suffix_ops: device array of fold operators (see gfrs_kernels.hip). */
void launch_crc_encode(uint8_t *dst, size_t dst_stride, const uint8_t *src,
                       size_t src_stride, int64_t n, int64_t block_len,
                       int nshards, hipStream_t s);
/* bad[j] = first failing block index in shard j, or -1 (preset by host). */
void launch_crc_verify(const uint8_t *framed, size_t stride,
                       int64_t framed_len, int64_t block_len, int nshards,
                       int64_t *bad, hipStream_t s);
void launch_crc_decode(uint8_t *dst, size_t dst_stride, const uint8_t *framed,
                       size_t src_stride, int64_t framed_len,
                       int64_t block_len, int nshards, int64_t *bad,
                       hipStream_t s);

/* fused encode+frame: parity + crc32block framed images in one pass
 * (PUT/repair pipeline; data read once, framed written once). */
void launch_rs_encode_frame_small(uint8_t *dst, size_t dst_stride,
                                  uint64_t base, uint64_t stripe_stride,
                                  size_t shard_len, int k, int m,
                                  const uint8_t *tabs, int nstripes,
                                  hipStream_t s);

void launch_rs_repair_frame_small(uint8_t *dst, size_t dst_stride,
                                  uint64_t base, uint64_t stripe_stride,
                                  size_t shard_len, int k, int gm, int nw,
                                  const int32_t *imap, const uint8_t *tabs,
                                  uint32_t colpack, uint32_t *fail,
                                  int nstripes, hipStream_t s);

void launch_rs_repair_frame(uint8_t *dst, size_t dst_stride, uint64_t base,
                            uint64_t stripe_stride, size_t shard_len, int k,
                            int gm, int nw, const int32_t *imap,
                            const uint8_t *tabs, uint32_t colpack,
                            uint32_t *fail, int nstripes, hipStream_t s);

void launch_rs_encode_frame(uint8_t *dst, size_t dst_stride, uint64_t base,
                            uint64_t stripe_stride, size_t shard_len, int k,
                            int m, const uint8_t *tabs, int nstripes,
                            hipStream_t s);

/* sized coder (crc32block/sized_coder.go): payload ‖ CRC32(BE) frames,
 * 512-B tail alignment handled by the host. */
void launch_sized_encode(uint8_t *dst, size_t dst_stride, const uint8_t *src,
                         size_t src_stride, int64_t n, int64_t block_len,
                         int nshards, hipStream_t s);
void launch_sized_verify(const uint8_t *framed, size_t stride,
                         int64_t body_len, int64_t block_len, int nshards,
                         int64_t *bad, hipStream_t s);
void launch_sized_decode(uint8_t *dst, size_t dst_stride,
                         const uint8_t *framed, size_t src_stride,
                         int64_t body_len, int64_t block_len, int nshards,
                         int64_t *bad, hipStream_t s);

/* Mixed write/compare strided apply (reconstruct+verify in one pass). */
void launch_rs_apply_mixed_strided(uint64_t base, uint64_t stripe_stride,
                                   const int32_t *in_idx, int k,
                                   const int32_t *out_idx, int nout,
                                   const uint8_t *tabs, uint32_t cmp_mask,
                                   size_t shard_len, int nstripes,
                                   uint32_t *fail, hipStream_t s);

/* EncodeIdx-style accumulate apply (reedsolomon.go:631-668):
 * out[r] ^= coeff[r]*in for one input shard. */
void launch_rs_apply_xor(const uint64_t *ptrs, int nptr,
                         const int32_t *in_idx, int k,
                         const int32_t *out_idx, int nout,
                         const uint8_t *tabs, size_t shard_len, int nstripes,
                         hipStream_t s);

/* blobnode on-disk shard codec (core/shard.go, datafile.go:342-445). */
void launch_shard_finalize(uint8_t *dst, size_t dst_stride,
                           const uint64_t *bids, const uint64_t *vuids,
                           int64_t raw_size,
                           int64_t block_len, int nshards, hipStream_t s);
void launch_shard_parse(const uint8_t *img, size_t stride, int64_t raw_size,
                        int64_t block_len, int nshards, uint64_t *out,
                        hipStream_t s);

/* v_perm semantics probe: returns 1 when `sel byte >= 8 -> 0x00` holds. */
int probe_perm_device(void);

/* One-time CRC table upload for the current device. */
int crc_device_init_current(void);

/* Strided-layout variants (batch APIs; stripe s shard i at
 * base + s*stripe_stride + i*shard_len). */
void launch_rs_apply_strided(uint64_t base, uint64_t stripe_stride,
                             const int32_t *in_idx, int k,
                             const int32_t *out_idx, int nout,
                             const uint8_t *tabs, size_t shard_len,
                             int nstripes, hipStream_t s);
void launch_rs_verify_strided(uint64_t base, uint64_t stripe_stride,
                              const int32_t *in_idx, int k,
                              const int32_t *out_idx, int nout,
                              const uint8_t *tabs, size_t shard_len,
                              int nstripes, uint32_t *fail, hipStream_t s);

}  // namespace gfrs

#endif
