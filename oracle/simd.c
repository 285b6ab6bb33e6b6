/* ORACLE — TEST INFRASTRUCTURE ONLY (see oracle.h header).
 *
 * SIMD CPU-baseline leg (BASELINE.md's "scalar + AVX2 builds" plan; VERDICT
 * r1 item 5): the reference's hot path runs klauspost/reedsolomon's
 * GFNI/AVX2 assembly and minio/highwayhash's AVX2 assembly on the host, so
 * a scalar-table oracle understates the CPU baseline by ~an order of
 * magnitude.  This file provides runtime-dispatched equivalents:
 *   - GF(2^8) constant-multiply-xor: GFNI gf2p8affineqb (the same
 *     instruction klauspost's _gfni kernels use), AVX2 pshufb split-nibble
 *     (klauspost/ISA-L galMulSlicedXor shape), scalar fallback;
 *   - HighwayHash-256: 4-lane AVX2 packet loop (the reference
 *     highwayhash_amd64 shape), scalar remainder/finalization reused from
 *     hh256.c so only the main loop differs.
 * These are BENCH legs only — the parity checker stays the scalar
 * restatement — and are pinned bit-exact against it by
 * test_oracle_properties.py::test_simd_matches_scalar.
 */
#include "oracle.h"
#include "hh_internal.h"
#include <immintrin.h>
#include <cpuid.h>
#include <string.h>

/* ---- CPU feature detection --------------------------------------------- */

static int cpu_has(unsigned leaf, unsigned subleaf, int reg, unsigned bit) {
    unsigned a, b, c, d;
    if (!__get_cpuid_count(leaf, subleaf, &a, &b, &c, &d)) return 0;
    unsigned v = reg == 1 ? b : (reg == 2 ? c : d);
    return (v >> bit) & 1;
}

static int g_isa = -1; /* 0 scalar, 1 avx2, 2 gfni+avx2 */

static void isa_detect(void);

const char *mo_cpu_isa(void) {
    if (g_isa < 0) isa_detect();
    switch (g_isa) {
    case 2: return "gfni+avx2";
    case 1: return "avx2";
    default: return "scalar";
    }
}

/* ---- GF(2^8)/0x11D constant-multiply tables ----------------------------- */

/* GFNI: y = A(c) . x over GF(2)^8 where A(c) is the bit-matrix of the
 * linear map x -> c*x in GF(2^8)/0x11D.  gf2p8affineqb's bit convention is
 * verified EMPIRICALLY at init against mo_gf_mul for all 256 inputs (the
 * two plausible row orders are tried); mismatch degrades to AVX2. */
static uint64_t gfni_mat[256];
/* AVX2 split-nibble: lo[c][x] = c*x, hi[c][x] = c*(x<<4) */
static uint8_t nib_lo[256][16], nib_hi[256][16];
static int tables_ready = 0;

__attribute__((target("gfni,avx2")))
static int gfni_selftest(const uint64_t *mats) {
    /* all 256 inputs for a few representative coefficients */
    const uint8_t coefs[5] = {2, 3, 0x1d, 0x8e, 0xff};
    uint8_t in[32], out[32];
    for (int ci = 0; ci < 5; ci++) {
        uint8_t c = coefs[ci];
        __m256i A = _mm256_set1_epi64x((long long)mats[c]);
        for (int base = 0; base < 256; base += 32) {
            for (int j = 0; j < 32; j++) in[j] = (uint8_t)(base + j);
            __m256i x = _mm256_loadu_si256((const __m256i *)in);
            __m256i y = _mm256_gf2p8affine_epi64_epi8(x, A, 0);
            _mm256_storeu_si256((__m256i *)out, y);
            for (int j = 0; j < 32; j++)
                if (out[j] != mo_gf_mul(c, in[j])) return 0;
        }
    }
    return 1;
}

static void build_tables(void) {
    if (tables_ready) return;
    for (int c = 0; c < 256; c++) {
        for (int x = 0; x < 16; x++) {
            nib_lo[c][x] = mo_gf_mul((uint8_t)c, (uint8_t)x);
            nib_hi[c][x] = mo_gf_mul((uint8_t)c, (uint8_t)(x << 4));
        }
    }
    /* candidate conventions: rowmask m_i collects input-bit coefficients of
     * output bit i; A.byte[k] = m_(7-k) (Intel doc order) or m_k */
    for (int conv = 0; conv < 2 && g_isa == 2; conv++) {
        for (int c = 0; c < 256; c++) {
            uint64_t A = 0;
            for (int i = 0; i < 8; i++) {
                uint8_t m = 0;
                for (int k = 0; k < 8; k++)
                    if ((mo_gf_mul((uint8_t)c, (uint8_t)(1 << k)) >> i) & 1)
                        m |= (uint8_t)(1 << k);
                int byte = conv == 0 ? (7 - i) : i;
                A |= (uint64_t)m << (8 * byte);
            }
            gfni_mat[c] = A;
        }
        if (gfni_selftest(gfni_mat)) goto done;
    }
    if (g_isa == 2) g_isa = 1; /* GFNI convention mismatch: degrade */
done:
    tables_ready = 1;
}

static void isa_detect(void) {
    int avx2 = cpu_has(7, 0, 1, 5);
    int gfni = cpu_has(7, 0, 2, 8);
    const char *force = getenv("MO_FORCE_ISA");
    g_isa = avx2 ? (gfni ? 2 : 1) : 0;
    if (force) {
        if (!strcmp(force, "scalar")) g_isa = 0;
        else if (!strcmp(force, "avx2") && g_isa >= 1) g_isa = 1;
    }
    build_tables();
}

/* ---- out[j] ^= c * in[j] ------------------------------------------------ */

__attribute__((target("gfni,avx2")))
static void gal_mul_xor_gfni(uint8_t c, const uint8_t *in, uint8_t *out,
                             size_t n) {
    __m256i A = _mm256_set1_epi64x((long long)gfni_mat[c]);
    size_t j = 0;
    for (; j + 64 <= n; j += 64) {
        __m256i x0 = _mm256_loadu_si256((const __m256i *)(in + j));
        __m256i x1 = _mm256_loadu_si256((const __m256i *)(in + j + 32));
        __m256i y0 = _mm256_gf2p8affine_epi64_epi8(x0, A, 0);
        __m256i y1 = _mm256_gf2p8affine_epi64_epi8(x1, A, 0);
        __m256i o0 = _mm256_loadu_si256((const __m256i *)(out + j));
        __m256i o1 = _mm256_loadu_si256((const __m256i *)(out + j + 32));
        _mm256_storeu_si256((__m256i *)(out + j), _mm256_xor_si256(o0, y0));
        _mm256_storeu_si256((__m256i *)(out + j + 32),
                            _mm256_xor_si256(o1, y1));
    }
    for (; j < n; j++) out[j] ^= mo_gf_mul(c, in[j]);
}

__attribute__((target("avx2")))
static void gal_mul_xor_avx2(uint8_t c, const uint8_t *in, uint8_t *out,
                             size_t n) {
    __m256i lo = _mm256_broadcastsi128_si256(
        _mm_loadu_si128((const __m128i *)nib_lo[c]));
    __m256i hi = _mm256_broadcastsi128_si256(
        _mm_loadu_si128((const __m128i *)nib_hi[c]));
    __m256i m0f = _mm256_set1_epi8(0x0f);
    size_t j = 0;
    for (; j + 32 <= n; j += 32) {
        __m256i x = _mm256_loadu_si256((const __m256i *)(in + j));
        __m256i xl = _mm256_and_si256(x, m0f);
        __m256i xh = _mm256_and_si256(_mm256_srli_epi64(x, 4), m0f);
        __m256i y = _mm256_xor_si256(_mm256_shuffle_epi8(lo, xl),
                                     _mm256_shuffle_epi8(hi, xh));
        __m256i o = _mm256_loadu_si256((const __m256i *)(out + j));
        _mm256_storeu_si256((__m256i *)(out + j), _mm256_xor_si256(o, y));
    }
    for (; j < n; j++) out[j] ^= mo_gf_mul(c, in[j]);
}

__attribute__((target("avx2")))
static void xor_only_avx2(const uint8_t *in, uint8_t *out, size_t n) {
    size_t j = 0;
    for (; j + 32 <= n; j += 32) {
        __m256i x = _mm256_loadu_si256((const __m256i *)(in + j));
        __m256i o = _mm256_loadu_si256((const __m256i *)(out + j));
        _mm256_storeu_si256((__m256i *)(out + j), _mm256_xor_si256(o, x));
    }
    for (; j < n; j++) out[j] ^= in[j];
}

void mo_gal_mul_xor_fast(uint8_t c, const uint8_t *in, uint8_t *out,
                         size_t n) {
    if (g_isa < 0) isa_detect();
    if (c == 0) return;
    if (c == 1 && g_isa >= 1) {
        xor_only_avx2(in, out, n);
        return;
    }
    if (g_isa == 2) gal_mul_xor_gfni(c, in, out, n);
    else if (g_isa == 1) gal_mul_xor_avx2(c, in, out, n);
    else {
        for (size_t j = 0; j < n; j++) out[j] ^= mo_gf_mul(c, in[j]);
    }
}

/* Single-pass encode: all p parity accumulators live in registers while
 * each input chunk is read ONCE (the row-at-a-time form re-reads the
 * inputs p times and re-reads/writes each parity row d times — ~8 B of
 * traffic per input byte vs 1.5 here; same structure as the GPU encode
 * kernel).  p <= 8 keeps the accumulators in ymm registers. */
#define MO_FAST_MAXP 8
#define MO_FAST_MAXD 32

__attribute__((target("gfni,avx2")))
static void rs_encode_fused_gfni(const mo_rs *rs, uint8_t *const *shards,
                                 size_t shard_len) {
    const int d = rs->d, p = rs->p;
    /* hoist the per-(parity,input) affine matrices out of the chunk loop */
    __m256i A[MO_FAST_MAXD][MO_FAST_MAXP];
    for (int k = 0; k < d; k++)
        for (int i = 0; i < p; i++)
            A[k][i] = _mm256_set1_epi64x(
                (long long)gfni_mat[rs->matrix[(size_t)(d + i) * d + k]]);
    size_t j = 0;
    for (; j + 32 <= shard_len; j += 32) {
        __m256i acc[MO_FAST_MAXP];
        for (int i = 0; i < p; i++) acc[i] = _mm256_setzero_si256();
        for (int k = 0; k < d; k++) {
            __m256i x =
                _mm256_loadu_si256((const __m256i *)(shards[k] + j));
            for (int i = 0; i < p; i++)
                acc[i] = _mm256_xor_si256(
                    acc[i], _mm256_gf2p8affine_epi64_epi8(x, A[k][i], 0));
        }
        for (int i = 0; i < p; i++)
            _mm256_storeu_si256((__m256i *)(shards[d + i] + j), acc[i]);
    }
    if (j < shard_len) { /* scalar tail */
        for (int i = 0; i < p; i++) {
            const uint8_t *row = rs->matrix + (size_t)(d + i) * d;
            for (size_t t = j; t < shard_len; t++) {
                uint8_t a = 0;
                for (int k = 0; k < d; k++)
                    a ^= mo_gf_mul(row[k], shards[k][t]);
                shards[d + i][t] = a;
            }
        }
    }
}

__attribute__((target("avx2")))
static void rs_encode_fused_avx2(const mo_rs *rs, uint8_t *const *shards,
                                 size_t shard_len) {
    const int d = rs->d, p = rs->p;
    const __m256i m0f = _mm256_set1_epi8(0x0f);
    __m256i LO[MO_FAST_MAXD][MO_FAST_MAXP], HI[MO_FAST_MAXD][MO_FAST_MAXP];
    for (int k = 0; k < d; k++)
        for (int i = 0; i < p; i++) {
            uint8_t c = rs->matrix[(size_t)(d + i) * d + k];
            LO[k][i] = _mm256_broadcastsi128_si256(
                _mm_loadu_si128((const __m128i *)nib_lo[c]));
            HI[k][i] = _mm256_broadcastsi128_si256(
                _mm_loadu_si128((const __m128i *)nib_hi[c]));
        }
    size_t j = 0;
    for (; j + 32 <= shard_len; j += 32) {
        __m256i acc[MO_FAST_MAXP];
        for (int i = 0; i < p; i++) acc[i] = _mm256_setzero_si256();
        for (int k = 0; k < d; k++) {
            __m256i x =
                _mm256_loadu_si256((const __m256i *)(shards[k] + j));
            __m256i xl = _mm256_and_si256(x, m0f);
            __m256i xh = _mm256_and_si256(_mm256_srli_epi64(x, 4), m0f);
            for (int i = 0; i < p; i++)
                acc[i] = _mm256_xor_si256(
                    acc[i],
                    _mm256_xor_si256(_mm256_shuffle_epi8(LO[k][i], xl),
                                     _mm256_shuffle_epi8(HI[k][i], xh)));
        }
        for (int i = 0; i < p; i++)
            _mm256_storeu_si256((__m256i *)(shards[d + i] + j), acc[i]);
    }
    if (j < shard_len) {
        for (int i = 0; i < p; i++) {
            const uint8_t *row = rs->matrix + (size_t)(d + i) * d;
            for (size_t t = j; t < shard_len; t++) {
                uint8_t a = 0;
                for (int k = 0; k < d; k++)
                    a ^= mo_gf_mul(row[k], shards[k][t]);
                shards[d + i][t] = a;
            }
        }
    }
}

void mo_rs_encode_fast(const mo_rs *rs, uint8_t *const *shards,
                       size_t shard_len) {
    if (g_isa < 0) isa_detect();
    if (rs->p <= MO_FAST_MAXP && rs->d <= MO_FAST_MAXD && g_isa == 2) {
        rs_encode_fused_gfni(rs, shards, shard_len);
        return;
    }
    if (rs->p <= MO_FAST_MAXP && rs->d <= MO_FAST_MAXD && g_isa == 1) {
        rs_encode_fused_avx2(rs, shards, shard_len);
        return;
    }
    for (int i = 0; i < rs->p; i++) {
        uint8_t *out = shards[rs->d + i];
        memset(out, 0, shard_len);
        const uint8_t *row = rs->matrix + (size_t)(rs->d + i) * rs->d;
        for (int k = 0; k < rs->d; k++)
            mo_gal_mul_xor_fast(row[k], shards[k], out, shard_len);
    }
}

/* ---- HighwayHash-256, AVX2 4-lane main loop ----------------------------- */

__attribute__((target("avx2")))
static void hh256_avx2(const uint8_t key32[32], const uint8_t *msg,
                       size_t len, uint8_t out[32]) {
    mo_hh_state st;
    mo_hh_reset_(&st, key32);
    __m256i v0 = _mm256_loadu_si256((const __m256i *)st.v0);
    __m256i v1 = _mm256_loadu_si256((const __m256i *)st.v1);
    __m256i mul0 = _mm256_loadu_si256((const __m256i *)st.mul0);
    __m256i mul1 = _mm256_loadu_si256((const __m256i *)st.mul1);
    /* zipper merge is a byte shuffle within each 128-bit pair
     * (hh256.c hh_zipper_merge_add's byte map, same derivation as the
     * GPU kernel's v_perm form: low u64 = pair bytes [3,12,2,5,14,1,15,0],
     * high u64 = [11,4,10,13,9,6,8,7]) */
    const __m256i zctl = _mm256_setr_epi8(
        3, 12, 2, 5, 14, 1, 15, 0, 11, 4, 10, 13, 9, 6, 8, 7,
        3, 12, 2, 5, 14, 1, 15, 0, 11, 4, 10, 13, 9, 6, 8, 7);
    while (len >= 32) {
        __m256i pkt = _mm256_loadu_si256((const __m256i *)msg);
        v1 = _mm256_add_epi64(v1, _mm256_add_epi64(mul0, pkt));
        mul0 = _mm256_xor_si256(
            mul0, _mm256_mul_epu32(v1, _mm256_srli_epi64(v0, 32)));
        v0 = _mm256_add_epi64(v0, mul1);
        mul1 = _mm256_xor_si256(
            mul1, _mm256_mul_epu32(v0, _mm256_srli_epi64(v1, 32)));
        v0 = _mm256_add_epi64(v0, _mm256_shuffle_epi8(v1, zctl));
        v1 = _mm256_add_epi64(v1, _mm256_shuffle_epi8(v0, zctl));
        msg += 32;
        len -= 32;
    }
    _mm256_storeu_si256((__m256i *)st.v0, v0);
    _mm256_storeu_si256((__m256i *)st.v1, v1);
    _mm256_storeu_si256((__m256i *)st.mul0, mul0);
    _mm256_storeu_si256((__m256i *)st.mul1, mul1);
    mo_hh_finish_(&st, msg, len, out);
}

void mo_hh256_fast(const uint8_t key32[32], const uint8_t *msg, size_t len,
                   uint8_t out[32]) {
    if (g_isa < 0) isa_detect();
    if (g_isa >= 1) hh256_avx2(key32, msg, len, out);
    else mo_hh256(key32, msg, len, out);
}
