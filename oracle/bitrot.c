/* ORACLE — TEST INFRASTRUCTURE ONLY (see oracle.h header).
 *
 * Bitrot algorithm dispatch mirroring reference cmd/bitrot.go:47-64
 * (BitrotAlgorithm.New) with the magic HighwayHash key (cmd/bitrot.go:37),
 * plus the timed CPU-baseline legs used by bench.py (kind "port": this
 * oracle timed on host cores — the reference Go path cannot run here, no Go
 * toolchain and its arithmetic deps are not vendored).
 */
#include "oracle.h"
#include <stdlib.h>
#include <string.h>
#include <time.h>
#ifdef _OPENMP
#include <omp.h>
#endif

/* cmd/bitrot.go:37 */
static const uint8_t MAGIC_HH_KEY[32] = {
    0x4b, 0xe7, 0x34, 0xfa, 0x8e, 0x23, 0x8a, 0xcd, 0x26, 0x3e, 0x83,
    0xe6, 0xbb, 0x96, 0x85, 0x52, 0x04, 0x0f, 0x93, 0x5d, 0xa3, 0x9f,
    0x44, 0x14, 0x97, 0xe0, 0x9d, 0x13, 0x22, 0xde, 0x36, 0xa0};

int mo_bitrot_size(int algo) {
    switch (algo) {
    case MO_BITROT_SHA256:
    case MO_BITROT_HIGHWAYHASH256:
    case MO_BITROT_HIGHWAYHASH256S:
        return 32;
    case MO_BITROT_BLAKE2B512:
        return 64;
    default:
        return 0;
    }
}

void mo_bitrot_sum(int algo, const uint8_t *msg, size_t len, uint8_t *out) {
    switch (algo) {
    case MO_BITROT_SHA256:
        mo_sha256(msg, len, out);
        break;
    case MO_BITROT_HIGHWAYHASH256:
    case MO_BITROT_HIGHWAYHASH256S:
        mo_hh256(MAGIC_HH_KEY, msg, len, out);
        break;
    case MO_BITROT_BLAKE2B512:
        mo_blake2b512(msg, len, out);
        break;
    }
}

/* bench-leg dispatch: SIMD hash where it exists (HighwayHash), scalar
 * otherwise (SHA-256/BLAKE2b are not on the bench legs' hot configs) */
void mo_bitrot_sum_fast(int algo, const uint8_t *msg, size_t len,
                        uint8_t *out) {
    switch (algo) {
    case MO_BITROT_HIGHWAYHASH256:
    case MO_BITROT_HIGHWAYHASH256S:
        mo_hh256_fast(MAGIC_HH_KEY, msg, len, out);
        break;
    default:
        mo_bitrot_sum(algo, msg, len, out);
    }
}

/* xoshiro256** — seeded synthetic inputs (seed stated in bench output) */
typedef struct { uint64_t s[4]; } xo_state;

static uint64_t xo_rotl(uint64_t x, int k) { return (x << k) | (x >> (64 - k)); }

static uint64_t splitmix64(uint64_t *x) {
    uint64_t z = (*x += 0x9e3779b97f4a7c15ull);
    z = (z ^ (z >> 30)) * 0xbf58476d1ce4e5b9ull;
    z = (z ^ (z >> 27)) * 0x94d049bb133111ebull;
    return z ^ (z >> 31);
}

static void xo_seed(xo_state *st, uint64_t seed) {
    for (int i = 0; i < 4; i++) st->s[i] = splitmix64(&seed);
}

static uint64_t xo_next(xo_state *st) {
    uint64_t *s = st->s;
    uint64_t result = xo_rotl(s[1] * 5, 7) * 9;
    uint64_t t = s[1] << 17;
    s[2] ^= s[0];
    s[3] ^= s[1];
    s[1] ^= s[2];
    s[0] ^= s[3];
    s[2] ^= t;
    s[3] = xo_rotl(s[3], 45);
    return result;
}

void mo_fill_random(uint8_t *buf, size_t n, uint64_t seed) {
    xo_state st;
    xo_seed(&st, seed);
    size_t i = 0;
    for (; i + 8 <= n; i += 8) {
        uint64_t v = xo_next(&st);
        memcpy(buf + i, &v, 8);
    }
    if (i < n) {
        uint64_t v = xo_next(&st);
        memcpy(buf + i, &v, n - i);
    }
}

static double now_sec(void) {
    struct timespec ts;
    clock_gettime(CLOCK_MONOTONIC, &ts);
    return (double)ts.tv_sec + 1e-9 * ts.tv_nsec;
}

/* Fused encode + per-shard bitrot over n_blocks independent blocks.
 * Mirrors the per-block work of Erasure.EncodeData (cmd/erasure-coding.go:77)
 * + streamingBitrotWriter.Write's hash (cmd/bitrot-streaming.go:57-59). */
double mo_cpu_encode_bench(int d, int p, size_t block_len, int n_blocks,
                           int algo, int threads, uint64_t seed) {
    mo_rs rs;
    if (mo_rs_init(&rs, d, p) != 0) return -1.0;
    size_t shard_len = (block_len + (size_t)d - 1) / (size_t)d;
    int total = d + p;
    int hsz = mo_bitrot_size(algo);
#ifdef _OPENMP
    omp_set_num_threads(threads);
#else
    (void)threads;
#endif
    /* pre-generate inputs outside the timed region */
    uint8_t *data = (uint8_t *)malloc((size_t)n_blocks * shard_len * (size_t)d);
    uint8_t *parity = (uint8_t *)malloc((size_t)n_blocks * shard_len * (size_t)p);
    uint8_t *sums = (uint8_t *)malloc((size_t)n_blocks * (size_t)total * (size_t)hsz);
    if (!data || !parity || !sums) { free(data); free(parity); free(sums); return -1.0; }
    memset(data, 0, (size_t)n_blocks * shard_len * (size_t)d);
#ifdef _OPENMP
#pragma omp parallel for schedule(static)
#endif
    for (int b = 0; b < n_blocks; b++)
        mo_fill_random(data + (size_t)b * shard_len * d, block_len,
                       seed + (uint64_t)b);
    double t0 = now_sec();
#ifdef _OPENMP
#pragma omp parallel for schedule(dynamic)
#endif
    for (int b = 0; b < n_blocks; b++) {
        uint8_t *shards[MO_MAX_SHARDS];
        for (int k = 0; k < d; k++)
            shards[k] = data + ((size_t)b * d + k) * shard_len;
        for (int i = 0; i < p; i++)
            shards[d + i] = parity + ((size_t)b * p + i) * shard_len;
        mo_rs_encode_fast(&rs, shards, shard_len);
        for (int s = 0; s < total; s++)
            mo_bitrot_sum_fast(algo, shards[s], shard_len,
                               sums + ((size_t)b * total + s) * hsz);
    }
    double el = now_sec() - t0;
    /* keep the compiler honest */
    volatile uint8_t sink = sums[0] ^ parity[0];
    (void)sink;
    free(data);
    free(parity);
    free(sums);
    return el;
}

double mo_cpu_reconstruct_bench(int d, int p, size_t block_len, int n_blocks,
                                int n_erased, int threads, uint64_t seed) {
    mo_rs rs;
    if (mo_rs_init(&rs, d, p) != 0) return -1.0;
    size_t shard_len = (block_len + (size_t)d - 1) / (size_t)d;
    int total = d + p;
#ifdef _OPENMP
    omp_set_num_threads(threads);
#else
    (void)threads;
#endif
    uint8_t *bufs = (uint8_t *)malloc((size_t)n_blocks * shard_len * (size_t)total);
    if (!bufs) return -1.0;
    memset(bufs, 0, (size_t)n_blocks * shard_len * (size_t)total);
#ifdef _OPENMP
#pragma omp parallel for schedule(static)
#endif
    for (int b = 0; b < n_blocks; b++) {
        uint8_t *shards[MO_MAX_SHARDS];
        for (int s = 0; s < total; s++)
            shards[s] = bufs + ((size_t)b * total + s) * shard_len;
        mo_fill_random(shards[0], block_len, seed + (uint64_t)b); /* contiguous d shards */
        mo_rs_encode(&rs, shards, shard_len);
        for (int e = 0; e < n_erased; e++) memset(shards[e], 0, shard_len);
    }
    uint8_t present[MO_MAX_SHARDS];
    for (int s = 0; s < total; s++) present[s] = s >= n_erased;
    double t0 = now_sec();
#ifdef _OPENMP
#pragma omp parallel for schedule(dynamic)
#endif
    for (int b = 0; b < n_blocks; b++) {
        uint8_t *shards[MO_MAX_SHARDS];
        for (int s = 0; s < total; s++)
            shards[s] = bufs + ((size_t)b * total + s) * shard_len;
        mo_rs_reconstruct_fast(&rs, shards, present, shard_len, 1);
    }
    double el = now_sec() - t0;
    volatile uint8_t sink = bufs[0];
    (void)sink;
    free(bufs);
    return el;
}
