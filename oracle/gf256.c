/* ORACLE — TEST INFRASTRUCTURE ONLY (see oracle.h header).
 *
 * GF(2^8) arithmetic and systematic-Vandermonde Reed-Solomon, restating
 * klauspost/reedsolomon v1.12.4's default matrix construction (buildMatrix:
 * Vandermonde vm[r][c] = r^c over GF(2^8)/0x11D, then vm * inv(vm[0:d][0:d])
 * so the top d rows become identity).  Reference call sites:
 * cmd/erasure-coding.go:63 (reedsolomon.New), :85 (Encode),
 * :106 (ReconstructData), :112 (Reconstruct).
 * Pinned by cmd/erasure-coding.go:160 golden fingerprints (tests/golden/).
 */
#include "oracle.h"
#include <string.h>

static uint8_t gf_exp_tbl[512];
static uint8_t gf_log_tbl[256];
static int gf_ready = 0;

static void gf_init(void) {
    if (gf_ready) return;
    /* generator 2, poly 0x11D */
    int x = 1;
    for (int i = 0; i < 255; i++) {
        gf_exp_tbl[i] = (uint8_t)x;
        gf_log_tbl[x] = (uint8_t)i;
        x <<= 1;
        if (x & 0x100) x ^= 0x11D;
    }
    for (int i = 255; i < 512; i++) gf_exp_tbl[i] = gf_exp_tbl[i - 255];
    gf_log_tbl[0] = 0; /* undefined; guarded by callers */
    gf_ready = 1;
}

uint8_t mo_gf_mul(uint8_t a, uint8_t b) {
    gf_init();
    if (a == 0 || b == 0) return 0;
    return gf_exp_tbl[gf_log_tbl[a] + gf_log_tbl[b]];
}

static uint8_t gf_div(uint8_t a, uint8_t b) {
    gf_init();
    if (a == 0) return 0;
    /* b == 0 is a caller bug; mirror klauspost's panic by returning 0 */
    int diff = (int)gf_log_tbl[a] - (int)gf_log_tbl[b];
    if (diff < 0) diff += 255;
    return gf_exp_tbl[diff];
}

uint8_t mo_gf_exp(uint8_t a, int n) {
    gf_init();
    if (n == 0) return 1;
    if (a == 0) return 0;
    int l = (int)gf_log_tbl[a] * n % 255;
    return gf_exp_tbl[l];
}

/* ---- dense matrix ops over GF(2^8); dims <= 256 ---- */

typedef struct {
    int rows, cols;
    uint8_t m[MO_MAX_SHARDS][MO_MAX_SHARDS];
} gmat;

static void gm_mul(const gmat *a, const gmat *b, gmat *out) {
    out->rows = a->rows;
    out->cols = b->cols;
    for (int r = 0; r < a->rows; r++) {
        for (int c = 0; c < b->cols; c++) {
            uint8_t acc = 0;
            for (int k = 0; k < a->cols; k++)
                acc ^= mo_gf_mul(a->m[r][k], b->m[k][c]);
            out->m[r][c] = acc;
        }
    }
}

/* Gauss-Jordan inversion, mirrors klauspost matrix.Invert semantics.
 * Returns 0 ok, -1 singular. */
static int gm_invert(const gmat *in, gmat *out) {
    int n = in->rows;
    /* augmented [in | I] */
    static _Thread_local uint8_t w[MO_MAX_SHARDS][2 * MO_MAX_SHARDS];
    for (int r = 0; r < n; r++) {
        memcpy(w[r], in->m[r], (size_t)n);
        memset(w[r] + n, 0, (size_t)n);
        w[r][n + r] = 1;
    }
    for (int r = 0; r < n; r++) {
        if (w[r][r] == 0) {
            int swap = -1;
            for (int rb = r + 1; rb < n; rb++)
                if (w[rb][r] != 0) { swap = rb; break; }
            if (swap < 0) return -1;
            for (int c = 0; c < 2 * n; c++) {
                uint8_t t = w[r][c];
                w[r][c] = w[swap][c];
                w[swap][c] = t;
            }
        }
        uint8_t piv = w[r][r];
        if (piv != 1) {
            uint8_t inv = gf_div(1, piv);
            for (int c = 0; c < 2 * n; c++) w[r][c] = mo_gf_mul(w[r][c], inv);
        }
        for (int rb = 0; rb < n; rb++) {
            if (rb == r || w[rb][r] == 0) continue;
            uint8_t f = w[rb][r];
            for (int c = 0; c < 2 * n; c++)
                w[rb][c] ^= mo_gf_mul(f, w[r][c]);
        }
    }
    out->rows = out->cols = n;
    for (int r = 0; r < n; r++) memcpy(out->m[r], w[r] + n, (size_t)n);
    return 0;
}

int mo_rs_init(mo_rs *rs, int d, int p) {
    gf_init();
    if (d <= 0 || p < 0 || d + p > MO_MAX_SHARDS) return -1;
    rs->d = d;
    rs->p = p;
    int total = d + p;
    static _Thread_local gmat vm, top, topinv, enc;
    vm.rows = total;
    vm.cols = d;
    for (int r = 0; r < total; r++)
        for (int c = 0; c < d; c++) vm.m[r][c] = mo_gf_exp((uint8_t)r, c);
    top.rows = top.cols = d;
    for (int r = 0; r < d; r++) memcpy(top.m[r], vm.m[r], (size_t)d);
    if (gm_invert(&top, &topinv) != 0) return -1;
    gm_mul(&vm, &topinv, &enc);
    for (int r = 0; r < total; r++)
        memcpy(rs->matrix + (size_t)r * d, enc.m[r], (size_t)d);
    return 0;
}

/* out[j] ^= coef * in[j] over shard_len bytes (table-driven scalar). */
static void gal_mul_xor(uint8_t coef, const uint8_t *in, uint8_t *out,
                        size_t n) {
    if (coef == 0) return;
    if (coef == 1) {
        for (size_t j = 0; j < n; j++) out[j] ^= in[j];
        return;
    }
    const uint8_t *ex = gf_exp_tbl + gf_log_tbl[coef];
    for (size_t j = 0; j < n; j++) {
        uint8_t b = in[j];
        if (b) out[j] ^= ex[gf_log_tbl[b]];
    }
}

void mo_rs_encode(const mo_rs *rs, uint8_t *const *shards, size_t shard_len) {
    for (int i = 0; i < rs->p; i++) {
        uint8_t *out = shards[rs->d + i];
        memset(out, 0, shard_len);
        const uint8_t *row = rs->matrix + (size_t)(rs->d + i) * rs->d;
        for (int k = 0; k < rs->d; k++)
            gal_mul_xor(row[k], shards[k], out, shard_len);
    }
}

int mo_rs_reconstruct(const mo_rs *rs, uint8_t *const *shards,
                      const uint8_t *present, size_t shard_len,
                      int data_only) {
    int d = rs->d, total = rs->d + rs->p;
    int n_present = 0;
    for (int i = 0; i < total; i++)
        if (present[i]) n_present++;
    if (n_present == total) return 0;
    if (n_present < d) return -2;

    /* sub-matrix of the first d present rows (klauspost reconstruct order) */
    static _Thread_local gmat sub, dec;
    int src_idx[MO_MAX_SHARDS];
    sub.rows = sub.cols = d;
    int r = 0;
    for (int i = 0; i < total && r < d; i++) {
        if (!present[i]) continue;
        memcpy(sub.m[r], rs->matrix + (size_t)i * d, (size_t)d);
        src_idx[r] = i;
        r++;
    }
    if (gm_invert(&sub, &dec) != 0) return -1;

    /* missing data shards: decode rows applied to the d collected shards */
    for (int t = 0; t < d; t++) {
        if (present[t]) continue;
        uint8_t *out = shards[t];
        memset(out, 0, shard_len);
        for (int k = 0; k < d; k++)
            gal_mul_xor(dec.m[t][k], shards[src_idx[k]], out, shard_len);
    }
    if (data_only) return 0;
    /* missing parity: encode rows applied to (now complete) data shards */
    for (int t = d; t < total; t++) {
        if (present[t]) continue;
        uint8_t *out = shards[t];
        memset(out, 0, shard_len);
        const uint8_t *row = rs->matrix + (size_t)t * d;
        for (int k = 0; k < d; k++)
            gal_mul_xor(row[k], shards[k], out, shard_len);
    }
    return 0;
}

/* BENCH LEG ONLY (simd.c dispatch): reconstruct with the SIMD
 * constant-multiply.  The checker path above stays scalar; bit-equality
 * of the two is pinned by test_simd_matches_scalar (the GF op) plus the
 * round-trip identity. */
int mo_rs_reconstruct_fast(const mo_rs *rs, uint8_t *const *shards,
                           const uint8_t *present, size_t shard_len,
                           int data_only) {
    int d = rs->d, total = rs->d + rs->p;
    int n_present = 0;
    for (int i = 0; i < total; i++)
        if (present[i]) n_present++;
    if (n_present == total) return 0;
    if (n_present < d) return -2;
    static _Thread_local gmat sub, dec;
    int src_idx[MO_MAX_SHARDS];
    sub.rows = sub.cols = d;
    int r = 0;
    for (int i = 0; i < total && r < d; i++) {
        if (!present[i]) continue;
        memcpy(sub.m[r], rs->matrix + (size_t)i * d, (size_t)d);
        src_idx[r] = i;
        r++;
    }
    if (gm_invert(&sub, &dec) != 0) return -1;
    for (int t = 0; t < d; t++) {
        if (present[t]) continue;
        uint8_t *out = shards[t];
        memset(out, 0, shard_len);
        for (int k = 0; k < d; k++)
            mo_gal_mul_xor_fast(dec.m[t][k], shards[src_idx[k]], out,
                                shard_len);
    }
    if (data_only) return 0;
    for (int t = d; t < total; t++) {
        if (present[t]) continue;
        uint8_t *out = shards[t];
        memset(out, 0, shard_len);
        const uint8_t *row = rs->matrix + (size_t)t * d;
        for (int k = 0; k < d; k++)
            mo_gal_mul_xor_fast(row[k], shards[k], out, shard_len);
    }
    return 0;
}
