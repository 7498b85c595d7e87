/* oracle/ec_ref.c — blobstore/common/ec layer oracle: LRC layering and
 * buffer size math.
 *
 * TEST INFRASTRUCTURE ONLY (see oracle.h header).
 *
 * Restates:
 *   codemode.go:301-318   GetECLayoutByAZ (local stripe index layout)
 *   codemode.go:365-372   LocalStripeInAZ
 *   lrcencoder.go:35-82   lrcEncoder.Encode (global then per-AZ local)
 *   lrcencoder.go:133-186 lrcEncoder.Reconstruct
 *   buf.go:67-133         ec.Buffer size math (shardSize/ECDataSize/ECSize)
 */
#include "oracle.h"

#include <stdlib.h>
#include <string.h>

int orc_lrc_local_stripe(int n, int mm, int l, int az, int az_idx,
                         int *out_idx) {
    if (az <= 0 || az_idx < 0 || az_idx >= az) return ORC_ERR_INVALID_ARG;
    if (n % az || mm % az || l % az) return ORC_ERR_INVALID_ARG;
    int ln = n / az, lm = mm / az, ll = l / az;
    int c = 0;
    for (int i = 0; i < ln; i++) out_idx[c++] = az_idx * ln + i;
    for (int i = 0; i < lm; i++) out_idx[c++] = n + az_idx * lm + i;
    for (int i = 0; i < ll; i++) out_idx[c++] = n + mm + az_idx * ll + i;
    return c;
}

int orc_lrc_encode(int n, int mm, int l, int az, uint8_t **shards, size_t len) {
    if (l == 0) return orc_rs_encode(n, mm, shards, len);
    if (az <= 0 || n % az || mm % az || l % az) return ORC_ERR_INVALID_ARG;
    /* global engine RS(n, mm) over shards[0..n+mm) (lrcencoder.go:44) */
    int rc = orc_rs_encode(n, mm, shards, len);
    if (rc != ORC_OK) return rc;
    /* local engine RS((n+mm)/az, l/az) per AZ (encoder.go:92-104,
     * lrcencoder.go:57-77) */
    int local_n = (n + mm) / az, local_m = l / az;
    int cnt = local_n + local_m;
    int *idx = (int *)malloc(sizeof(int) * cnt);
    uint8_t **local = (uint8_t **)malloc(sizeof(uint8_t *) * cnt);
    if (!idx || !local) { free(idx); free(local); return ORC_ERR_INVALID_ARG; }
    for (int a = 0; a < az; a++) {
        orc_lrc_local_stripe(n, mm, l, az, a, idx);
        for (int i = 0; i < cnt; i++) local[i] = shards[idx[i]];
        rc = orc_rs_encode(local_n, local_m, local, len);
        if (rc != ORC_OK) break;
    }
    free(idx);
    free(local);
    return rc;
}

int orc_lrc_reconstruct(int n, int mm, int l, int az, uint8_t **shards,
                        size_t len, const uint8_t *present, int data_only) {
    if (l == 0) return orc_rs_reconstruct(n, mm, shards, len, present, data_only);
    if (az <= 0 || n % az || mm % az || l % az) return ORC_ERR_INVALID_ARG;
    /* Full-set path (lrcencoder.go:155-186): global reconstruct over the
     * first n+mm, then recompute bad local parities per AZ. */
    int rc = orc_rs_reconstruct(n, mm, shards, len, present, data_only);
    if (rc != ORC_OK) return rc;
    if (data_only) return ORC_OK; /* lrcencoder.go:190-203 ReconstructData */
    int local_n = (n + mm) / az, local_m = l / az;
    int cnt = local_n + local_m;
    int *idx = (int *)malloc(sizeof(int) * cnt);
    uint8_t **local = (uint8_t **)malloc(sizeof(uint8_t *) * cnt);
    uint8_t *lpresent = (uint8_t *)malloc((size_t)cnt);
    if (!idx || !local || !lpresent) { rc = ORC_ERR_INVALID_ARG; goto out; }
    for (int a = 0; a < az; a++) {
        int need = 0;
        orc_lrc_local_stripe(n, mm, l, az, a, idx);
        for (int i = 0; i < cnt; i++) {
            local[i] = shards[idx[i]];
            /* after global reconstruct everything below n+mm is intact */
            lpresent[i] = idx[i] < n + mm ? 1 : present[idx[i]];
            if (!lpresent[i]) need = 1;
        }
        if (need) {
            rc = orc_rs_reconstruct(local_n, local_m, local, len, lpresent, 0);
            if (rc != ORC_OK) goto out;
        }
    }
out:
    free(idx);
    free(local);
    free(lpresent);
    return rc;
}

int orc_buffer_sizes(int n, int mm, int l, int min_shard_size,
                     long long data_size, long long *shard_size,
                     long long *ec_data_size, long long *ec_size) {
    if (n <= 0 || data_size <= 0) return ORC_ERR_SHORT_DATA; /* buf.go:69-77 */
    long long ss = (data_size + n - 1) / n;
    if (ss < min_shard_size) ss = min_shard_size; /* buf.go:79-83 */
    *shard_size = ss;
    *ec_data_size = ss * n;
    *ec_size = ss * (n + mm + l);
    return ORC_OK;
}
