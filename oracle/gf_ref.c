/* oracle/gf_ref.c — GF(2^8) field, matrix algebra and Reed-Solomon codec.
 *
 * TEST INFRASTRUCTURE ONLY (see oracle.h header).
 *
 * Restates, in order:
 *   galois.go:13-26        field constants (generating polynomial 29 → 0x11D)
 *   galois.go:28,70,81     logTable / expTable / mulTable (generated here,
 *                          asserted byte-identical to the vendored literals
 *                          via tests/golden/gf_tables.bin)
 *   galois.go:340,596      mulTableLow/High nibble tables
 *   galois.go:855-906      galMultiply / galDivide / galExp
 *   matrix.go:193-282      Invert (Gauss-Jordan), vandermonde
 *   reedsolomon.go:220-244 buildMatrix (Vandermonde × inv(top square) —
 *                          the only construction CubeFS reaches,
 *                          reedsolomon.go:471-472)
 *   reedsolomon.go:609-625,807-896  Encode / codeSomeShards
 *   reedsolomon.go:770-784,1287-1301 Verify / checkSomeShards
 *   reedsolomon.go:1375-1552 reconstruct (+ decode-matrix selection)
 */
#include "oracle.h"

#include <stdlib.h>
#include <string.h>

/* ---------------- field tables ---------------- */

static uint8_t log_t[256];
static uint8_t exp_t[510];
static uint8_t mul_t[256][256];
static uint8_t mul_lo[256][16];
static uint8_t mul_hi[256][16];
static int tables_ready = 0;

static void init_tables(void) {
    if (tables_ready) return;
    /* generating polynomial 29 → full reduction poly 0x11D (galois.go:25) */
    int e = 1;
    for (int i = 0; i < 255; i++) {
        exp_t[i] = (uint8_t)e;
        log_t[e] = (uint8_t)i;
        e <<= 1;
        if (e & 0x100) e ^= 0x11D;
    }
    for (int i = 0; i < 255; i++) exp_t[255 + i] = exp_t[i];
    log_t[0] = 0; /* unused entry, zero like the Go literal */
    for (int a = 0; a < 256; a++)
        for (int b = 0; b < 256; b++)
            mul_t[a][b] = (a == 0 || b == 0)
                              ? 0
                              : exp_t[(int)log_t[a] + (int)log_t[b]];
    for (int c = 0; c < 256; c++)
        for (int n = 0; n < 16; n++) {
            mul_lo[c][n] = mul_t[c][n];
            mul_hi[c][n] = mul_t[c][n << 4];
        }
    tables_ready = 1;
}

void orc_gf_tables(uint8_t *out, size_t out_len) {
    init_tables();
    size_t need = 256 + 510 + 65536 + 4096 + 4096;
    if (out_len < need) return;
    uint8_t *p = out;
    memcpy(p, log_t, 256); p += 256;
    memcpy(p, exp_t, 510); p += 510;
    memcpy(p, mul_t, 65536); p += 65536;
    memcpy(p, mul_lo, 4096); p += 4096;
    memcpy(p, mul_hi, 4096);
}

uint8_t orc_gf_mul(uint8_t a, uint8_t b) {
    init_tables();
    return mul_t[a][b];
}

/* galois.go:873 galDivide */
static uint8_t gf_div(uint8_t a, uint8_t b) {
    if (a == 0) return 0;
    int lr = (int)log_t[a] - (int)log_t[b];
    if (lr < 0) lr += 255;
    return exp_t[lr];
}

/* galois.go:892 galExp */
uint8_t orc_gf_exp(uint8_t a, int n) {
    init_tables();
    if (n == 0) return 1;
    if (a == 0) return 0;
    long lr = (long)log_t[a] * n;
    while (lr >= 255) lr -= 255;
    return exp_t[lr];
}

/* ---------------- matrix algebra ---------------- */

void orc_vandermonde(int rows, int cols, uint8_t *out) {
    init_tables();
    for (int r = 0; r < rows; r++)
        for (int c = 0; c < cols; c++) out[r * cols + c] = orc_gf_exp((uint8_t)r, c);
}

/* matrix.go:210-266 gaussianElimination on [m | I], then take right half. */
int orc_invert_matrix(const uint8_t *in, int n, uint8_t *out) {
    init_tables();
    int cols = 2 * n;
    uint8_t *w = (uint8_t *)malloc((size_t)n * cols);
    if (!w) return ORC_ERR_INVALID_ARG;
    for (int r = 0; r < n; r++) {
        memcpy(w + r * cols, in + r * n, n);
        memset(w + r * cols + n, 0, n);
        w[r * cols + n + r] = 1;
    }
    for (int r = 0; r < n; r++) {
        if (w[r * cols + r] == 0) {
            for (int rb = r + 1; rb < n; rb++)
                if (w[rb * cols + r] != 0) {
                    for (int c = 0; c < cols; c++) {
                        uint8_t t = w[r * cols + c];
                        w[r * cols + c] = w[rb * cols + c];
                        w[rb * cols + c] = t;
                    }
                    break;
                }
        }
        if (w[r * cols + r] == 0) { free(w); return ORC_ERR_SINGULAR; }
        if (w[r * cols + r] != 1) {
            uint8_t scale = gf_div(1, w[r * cols + r]);
            for (int c = 0; c < cols; c++)
                w[r * cols + c] = mul_t[w[r * cols + c]][scale];
        }
        for (int rb = r + 1; rb < n; rb++) {
            uint8_t scale = w[rb * cols + r];
            if (scale != 0)
                for (int c = 0; c < cols; c++)
                    w[rb * cols + c] ^= mul_t[scale][w[r * cols + c]];
        }
    }
    for (int d = 0; d < n; d++)
        for (int ra = 0; ra < d; ra++) {
            uint8_t scale = w[ra * cols + d];
            if (scale != 0)
                for (int c = 0; c < cols; c++)
                    w[ra * cols + c] ^= mul_t[scale][w[d * cols + c]];
        }
    for (int r = 0; r < n; r++) memcpy(out + r * n, w + r * cols + n, n);
    free(w);
    return ORC_OK;
}

/* GF matrix multiply: a(rows×inner) × b(inner×cols) → out(rows×cols). */
static void mat_mul(const uint8_t *a, const uint8_t *b, int rows, int inner,
                    int cols, uint8_t *out) {
    for (int r = 0; r < rows; r++)
        for (int c = 0; c < cols; c++) {
            uint8_t v = 0;
            for (int i = 0; i < inner; i++) v ^= mul_t[a[r * inner + i]][b[i * cols + c]];
            out[r * cols + c] = v;
        }
}

int orc_build_matrix(int k, int total, uint8_t *out) {
    init_tables();
    if (k <= 0 || total < k || total > 256) return ORC_ERR_INVALID_ARG;
    uint8_t *vm = (uint8_t *)malloc((size_t)total * k);
    uint8_t *topinv = (uint8_t *)malloc((size_t)k * k);
    if (!vm || !topinv) { free(vm); free(topinv); return ORC_ERR_INVALID_ARG; }
    orc_vandermonde(total, k, vm);
    int rc = orc_invert_matrix(vm, k, topinv); /* top square = first k rows */
    if (rc == ORC_OK) mat_mul(vm, topinv, total, k, k, out);
    free(vm);
    free(topinv);
    return rc;
}

/* ---------------- RS codec ---------------- */

/* codeSomeShards (reedsolomon.go:807): out[r][i] = ⊕_c rows[r][c]·in[c][i].
 * Scalar loop identical to galois_noasm.go:10-37 semantics. */
static void code_shards(const uint8_t *rows, int nrows, int k,
                        uint8_t *const *inputs, uint8_t **outputs, size_t len) {
    for (int r = 0; r < nrows; r++) {
        const uint8_t *mrow = rows + r * k;
        uint8_t *out = outputs[r];
        const uint8_t *mt0 = mul_t[mrow[0]];
        const uint8_t *in0 = inputs[0];
        for (size_t i = 0; i < len; i++) out[i] = mt0[in0[i]];
        for (int c = 1; c < k; c++) {
            const uint8_t *mt = mul_t[mrow[c]];
            const uint8_t *in = inputs[c];
            for (size_t i = 0; i < len; i++) out[i] ^= mt[in[i]];
        }
    }
}

int orc_rs_encode(int k, int m, uint8_t **shards, size_t len) {
    init_tables();
    if (k <= 0 || m < 0 || k + m > 256) return ORC_ERR_INVALID_ARG;
    if (m == 0) return ORC_OK;
    uint8_t *em = (uint8_t *)malloc((size_t)(k + m) * k);
    if (!em) return ORC_ERR_INVALID_ARG;
    int rc = orc_build_matrix(k, k + m, em);
    if (rc == ORC_OK)
        code_shards(em + (size_t)k * k, m, k, shards, shards + k, len);
    free(em);
    return rc;
}

int orc_rs_verify(int k, int m, uint8_t *const *shards, size_t len) {
    init_tables();
    if (k <= 0 || m <= 0 || k + m > 256) return ORC_ERR_INVALID_ARG;
    uint8_t *em = (uint8_t *)malloc((size_t)(k + m) * k);
    uint8_t **scratch = (uint8_t **)malloc(sizeof(uint8_t *) * m);
    if (!em || !scratch) { free(em); free(scratch); return ORC_ERR_INVALID_ARG; }
    int rc = orc_build_matrix(k, k + m, em);
    int ok = 1;
    if (rc == ORC_OK) {
        for (int r = 0; r < m; r++) scratch[r] = (uint8_t *)malloc(len);
        code_shards(em + (size_t)k * k, m, k, shards, scratch, len);
        for (int r = 0; r < m; r++) {
            if (memcmp(scratch[r], shards[k + r], len) != 0) ok = 0;
            free(scratch[r]);
        }
    }
    free(em);
    free(scratch);
    return rc == ORC_OK ? ok : rc;
}

int orc_rs_decode_matrix(int k, int m, const uint8_t *present,
                         uint8_t *out_rows, int *out_valid) {
    init_tables();
    int total = k + m;
    uint8_t *em = (uint8_t *)malloc((size_t)total * k);
    uint8_t *sub = (uint8_t *)malloc((size_t)k * k);
    if (!em || !sub) { free(em); free(sub); return ORC_ERR_INVALID_ARG; }
    int rc = orc_build_matrix(k, total, em);
    if (rc != ORC_OK) { free(em); free(sub); return rc; }
    /* first k valid rows in index order (reedsolomon.go:1453-1466) */
    int nvalid = 0;
    for (int i = 0; i < total && nvalid < k; i++)
        if (present[i]) out_valid[nvalid++] = i;
    if (nvalid < k) { free(em); free(sub); return ORC_ERR_TOO_FEW_SHARDS; }
    for (int r = 0; r < k; r++) memcpy(sub + r * k, em + out_valid[r] * k, k);
    rc = orc_invert_matrix(sub, k, out_rows);
    free(em);
    free(sub);
    return rc;
}

int orc_rs_reconstruct(int k, int m, uint8_t **shards, size_t len,
                       const uint8_t *present, int data_only) {
    init_tables();
    int total = k + m;
    int npresent = 0, dpresent = 0;
    for (int i = 0; i < total; i++)
        if (present[i]) { npresent++; if (i < k) dpresent++; }
    if (npresent == total || (data_only && dpresent == k)) return ORC_OK;
    if (npresent < k) return ORC_ERR_TOO_FEW_SHARDS;

    uint8_t *dec = (uint8_t *)malloc((size_t)k * k);
    int *valid = (int *)malloc(sizeof(int) * k);
    if (!dec || !valid) { free(dec); free(valid); return ORC_ERR_INVALID_ARG; }
    int rc = orc_rs_decode_matrix(k, m, present, dec, valid);
    if (rc != ORC_OK) { free(dec); free(valid); return rc; }

    uint8_t **sub = (uint8_t **)malloc(sizeof(uint8_t *) * k);
    uint8_t **outs = (uint8_t **)malloc(sizeof(uint8_t *) * m);
    uint8_t *rows = (uint8_t *)malloc((size_t)m * k);
    for (int r = 0; r < k; r++) sub[r] = shards[valid[r]];

    /* missing data shards from the decode matrix (reedsolomon.go:1503-1524) */
    int nout = 0;
    for (int i = 0; i < k; i++)
        if (!present[i]) {
            outs[nout] = shards[i];
            memcpy(rows + nout * k, dec + i * k, k);
            nout++;
        }
    if (nout > 0) code_shards(rows, nout, k, sub, outs, len);

    if (!data_only) {
        /* missing parity from all data shards (reedsolomon.go:1536-1551) */
        uint8_t *em = (uint8_t *)malloc((size_t)total * k);
        rc = orc_build_matrix(k, total, em);
        if (rc == ORC_OK) {
            nout = 0;
            for (int i = k; i < total; i++)
                if (!present[i]) {
                    outs[nout] = shards[i];
                    memcpy(rows + nout * k, em + i * k, k);
                    nout++;
                }
            if (nout > 0) code_shards(rows, nout, k, shards, outs, len);
        }
        free(em);
    }
    free(dec); free(valid); free(sub); free(outs); free(rows);
    return rc;
}
