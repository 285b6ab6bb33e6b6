/* ORACLE — TEST INFRASTRUCTURE ONLY.
 *
 * CPU restatement of the reference algorithms for MinIO's erasure-coding +
 * bitrot hot path.  This library is the parity checker: only tests/,
 * __graft_entry__.smoke() and bench.py's cpu_baseline leg may call it.
 * It is never the product path and never ships.
 *
 * Restates:
 *  - GF(2^8) Reed-Solomon (poly 0x11D, generator 2), systematic-Vandermonde
 *    matrix per klauspost/reedsolomon v1.12.4 (go.mod:49 of the reference;
 *    dependency source absent from /root/reference — restated from its
 *    published algorithm, pinned by the reference's own boot self-test
 *    vectors at cmd/erasure-coding.go:160).  MinIO names this construction
 *    "rs-vandermonde" (cmd/erasure-metadata.go:40).
 *  - HighwayHash-256 per minio/highwayhash v1.0.3 (go.mod:58; restated from
 *    the published HighwayHash portable reference, pinned by
 *    cmd/bitrot.go:225-230 chained vectors with the magic key at
 *    cmd/bitrot.go:37).
 *  - SHA-256 (FIPS 180-4) and BLAKE2b-512 (RFC 7693), the bitrot algorithms
 *    at cmd/bitrot.go:49-53.
 *  - xxHash64 (cespare/xxhash/v2), used only to check the self-test
 *    fingerprints (cmd/erasure-coding.go:177).
 *
 * Parity status: erasure matrix + HH256/SHA256/BLAKE2b pinned bit-exactly by
 * the reference's own golden vectors (tests/golden/).  HighwayHash ragged
 * tails (len % 32 != 0) are pinned transitively: the golden vectors cover
 * only multiples of 32; ragged behaviour follows the published portable
 * reference and is cross-checked oracle-vs-HIP in tests.
 * r2 hardening (external ragged vectors remain unobtainable offline —
 * DESIGN.md §2): the GPU cross-check covers EVERY len%32 residue, and a
 * second independently-written AVX2 main loop (simd.c) agrees with this
 * restatement for lengths 0..63 (test_simd_matches_scalar); the two share
 * only the UpdateRemainder code, which stays single-sourced.
 */
#ifndef MINIO_AMD_ORACLE_H
#define MINIO_AMD_ORACLE_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- GF(2^8) Reed-Solomon ---- */

#define MO_MAX_SHARDS 256

typedef struct {
    int d, p; /* data, parity shard counts; d+p <= 256 */
    /* systematic encode matrix, (d+p) x d, row-major: top d rows identity */
    uint8_t matrix[MO_MAX_SHARDS * MO_MAX_SHARDS];
} mo_rs;

/* Build the encode matrix.  Returns 0, or -1 on invalid (d,p). */
int mo_rs_init(mo_rs *rs, int d, int p);

/* shards: d+p pointers, each shard_len bytes; data shards 0..d-1 are inputs,
 * parity shards d..d+p-1 are outputs (caller-allocated). */
void mo_rs_encode(const mo_rs *rs, uint8_t *const *shards, size_t shard_len);

/* Reconstruct missing shards.  present[i] != 0 marks shard i as intact.
 * Missing shards' buffers must be caller-allocated (they are outputs).
 * data_only != 0 reconstructs only shards 0..d-1 (ReconstructData,
 * cmd/erasure-coding.go:106); otherwise all (Reconstruct, :112).
 * Returns 0 ok, -2 too few shards present. */
int mo_rs_reconstruct(const mo_rs *rs, uint8_t *const *shards,
                      const uint8_t *present, size_t shard_len, int data_only);

/* Scalar GF helpers (exposed for tests). */
uint8_t mo_gf_mul(uint8_t a, uint8_t b);
uint8_t mo_gf_exp(uint8_t a, int n);

/* ---- Hashes ---- */

void mo_hh256(const uint8_t key[32], const uint8_t *msg, size_t len,
              uint8_t out[32]);
void mo_sha256(const uint8_t *msg, size_t len, uint8_t out[32]);
void mo_blake2b512(const uint8_t *msg, size_t len, uint8_t out[64]);
uint64_t mo_xxh64(const uint8_t *msg, size_t len, uint64_t seed);

/* Bitrot algorithm ids — mirror cmd/xl-storage-format-v1.go:146-153 */
enum {
    MO_BITROT_SHA256 = 1,
    MO_BITROT_HIGHWAYHASH256 = 2,
    MO_BITROT_HIGHWAYHASH256S = 3,
    MO_BITROT_BLAKE2B512 = 4,
};

/* Digest size for an algorithm id (32 or 64), 0 if unknown. */
int mo_bitrot_size(int algo);
/* One-shot digest with MinIO semantics (HH algorithms use the magic key,
 * cmd/bitrot.go:37). out must hold mo_bitrot_size(algo) bytes. */
void mo_bitrot_sum(int algo, const uint8_t *msg, size_t len, uint8_t *out);

/* Seeded synthetic input generator (xoshiro256**), shared by tests/bench. */
void mo_fill_random(uint8_t *buf, size_t n, uint64_t seed);

/* ---- CPU baseline timing (bench.py cpu_baseline leg) ----
 * Fused encode+bitrot of n_blocks blocks of block_len random bytes
 * (seeded xoshiro256**), EC d+p, per-shard digest with `algo`.
 * Runs on `threads` OpenMP threads.  Returns elapsed seconds. */
/* ---- SIMD bench legs (simd.c): runtime-dispatched GFNI/AVX2/scalar.
 * BENCH ONLY — the parity checker stays the scalar restatement; these are
 * pinned bit-exact against it by test_simd_matches_scalar. */
const char *mo_cpu_isa(void);
void mo_gal_mul_xor_fast(uint8_t c, const uint8_t *in, uint8_t *out,
                         size_t n);
void mo_rs_encode_fast(const mo_rs *rs, uint8_t *const *shards,
                       size_t shard_len);
int mo_rs_reconstruct_fast(const mo_rs *rs, uint8_t *const *shards,
                           const uint8_t *present, size_t shard_len,
                           int data_only);
void mo_hh256_fast(const uint8_t key[32], const uint8_t *msg, size_t len,
                   uint8_t out[32]);
void mo_bitrot_sum_fast(int algo, const uint8_t *msg, size_t len,
                        uint8_t *out);

double mo_cpu_encode_bench(int d, int p, size_t block_len, int n_blocks,
                           int algo, int threads, uint64_t seed);
/* Same for reconstruction with the first n_erased shards erased. */
double mo_cpu_reconstruct_bench(int d, int p, size_t block_len, int n_blocks,
                                int n_erased, int threads, uint64_t seed);

#ifdef __cplusplus
}
#endif
#endif
