/* oracle/gf_fast.c — multithreaded CPU baseline for bench.py's cpu_baseline
 * leg ("kind": "port").
 *
 * TEST INFRASTRUCTURE ONLY (see oracle.h header).  This is the oracle's
 * fast build of the *identical* algorithm the reference's amd64 assembler
 * implements: the low/high nibble-table GF(2^8) multiply
 * (galois_amd64.go:37-52, galois_amd64.s galMulAVX2Xor) via AVX2 pshufb
 * when available, OpenMP across stripes like the reference's goroutine
 * split (reedsolomon.go:897-985).  Results are bit-identical to the scalar
 * oracle (tested).
 */
#include "oracle.h"

#include <stdlib.h>
#include <string.h>

#ifdef _OPENMP
#include <omp.h>
#endif

#ifdef __AVX2__
#include <immintrin.h>
#endif

int orc_threads_avail(void) {
#ifdef _OPENMP
    return omp_get_max_threads();
#else
    return 1;
#endif
}

/* out ^= mul_c(in) over len bytes using c's 16-entry low/high nibble tables
 * (xor_mode=0: assign).  lo/hi are 16 B each. */
static void mul_slice(const uint8_t *lo, const uint8_t *hi, const uint8_t *in,
                      uint8_t *out, size_t len, int xor_mode) {
    size_t i = 0;
#ifdef __AVX2__
    __m256i tlo = _mm256_broadcastsi128_si256(_mm_loadu_si128((const __m128i *)lo));
    __m256i thi = _mm256_broadcastsi128_si256(_mm_loadu_si128((const __m128i *)hi));
    __m256i mask = _mm256_set1_epi8(0x0F);
    for (; i + 32 <= len; i += 32) {
        __m256i v = _mm256_loadu_si256((const __m256i *)(in + i));
        __m256i vlo = _mm256_and_si256(v, mask);
        __m256i vhi = _mm256_and_si256(_mm256_srli_epi16(v, 4), mask);
        __m256i r = _mm256_xor_si256(_mm256_shuffle_epi8(tlo, vlo),
                                     _mm256_shuffle_epi8(thi, vhi));
        if (xor_mode)
            r = _mm256_xor_si256(r, _mm256_loadu_si256((const __m256i *)(out + i)));
        _mm256_storeu_si256((__m256i *)(out + i), r);
    }
#endif
    for (; i < len; i++) {
        uint8_t r = lo[in[i] & 0x0F] ^ hi[in[i] >> 4];
        out[i] = xor_mode ? (uint8_t)(out[i] ^ r) : r;
    }
}

int orc_rs_encode_mt(int k, int m, uint8_t **shards_flat, size_t len,
                     int nstripes, int nthreads) {
    if (k <= 0 || m <= 0 || k + m > 256) return ORC_ERR_INVALID_ARG;
    uint8_t *em = (uint8_t *)malloc((size_t)(k + m) * k);
    if (!em) return ORC_ERR_INVALID_ARG;
    int rc = orc_build_matrix(k, k + m, em);
    if (rc != ORC_OK) { free(em); return rc; }
    /* nibble tables for the m×k parity coefficients */
    uint8_t *tabs = (uint8_t *)malloc((size_t)m * k * 32);
    uint8_t full[74494];
    orc_gf_tables(full, sizeof(full));
    const uint8_t *mul_lo_all = full + 256 + 510 + 65536;
    const uint8_t *mul_hi_all = mul_lo_all + 4096;
    for (int r = 0; r < m; r++)
        for (int c = 0; c < k; c++) {
            uint8_t coef = em[(size_t)(k + r) * k + c];
            memcpy(tabs + ((size_t)r * k + c) * 32, mul_lo_all + coef * 16, 16);
            memcpy(tabs + ((size_t)r * k + c) * 32 + 16, mul_hi_all + coef * 16, 16);
        }
    /* Parallelize over (stripe, 128 KiB block) so all cores stay busy
     * even when nstripes < ncores, and chunk so each data block is read
     * from DRAM once and reused from L2 for all m parity rows - the
     * same perRound chunking the reference uses (reedsolomon.go:478-511).
     * Without this the baseline under-reported the CPU by >10x. */
    {
        const size_t CH = 128 * 1024;
        const long nblocks = (long)((len + CH - 1) / CH);
        const long total = (long)nstripes * nblocks;
#ifdef _OPENMP
        if (nthreads <= 0) nthreads = omp_get_max_threads();
#pragma omp parallel for num_threads(nthreads) schedule(static)
#endif
        for (long w = 0; w < total; w++) {
            const int s = (int)(w / nblocks);
            const size_t off = (size_t)(w % nblocks) * CH;
            const size_t blen = len - off < CH ? len - off : CH;
            uint8_t **sh = shards_flat + (size_t)s * (k + m);
            for (int r = 0; r < m; r++) {
                for (int c = 0; c < k; c++) {
                    const uint8_t *t = tabs + ((size_t)r * k + c) * 32;
                    mul_slice(t, t + 16, sh[c] + off, sh[k + r] + off, blen,
                              c != 0);
                }
            }
        }
    }
    free(tabs);
    free(em);
    return ORC_OK;
}

int orc_crc32b_encode_mt(uint8_t **dst, uint8_t **src, int64_t n,
                         int64_t block_len, int nshards, int nthreads) {
    int rc = 0;
#ifdef _OPENMP
    if (nthreads <= 0) nthreads = omp_get_max_threads();
#pragma omp parallel for num_threads(nthreads) schedule(static)
#endif
    for (int s = 0; s < nshards; s++) {
        int64_t w = orc_crc32b_encode(dst[s], src[s], n, block_len);
        if (w < 0) rc = (int)w;
    }
    return rc;
}
