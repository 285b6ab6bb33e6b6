/* example.c — plain-C client of the minio_ec C-ABI (include/minio_ec.h).
 *
 * This is exactly the call sequence a cgo shim drives
 * behind MinIO's reedsolomon.Encoder / bitrotWriter / bitrotReader
 * interfaces, written in C to show the boundary is language-neutral:
 *
 *   1. mec_ctx_create for an EC4+2 geometry
 *   2. mec_encode_batch: parity + per-shard HighwayHash-256 sums
 *   3. mec_reconstruct_batch with two shards erased -> bit-compare
 *   4. mec_encode_stream -> corrupt one drive -> mec_decode_stream
 *      (verify-on-read drops the corrupt drive, reconstructs, returns
 *      the original object bytes)
 *
 * Build:  gcc -O2 -I include tools/example.c -L minio_amd -lminio_ec_hip \
 *             -Wl,-rpath,'$ORIGIN/../minio_amd' -o tools/example
 * Run (MI355X box): ./tools/example   — prints PASS and exits 0.
 * Without a GPU it fails loudly with MEC_ERR_NO_GPU (no CPU fallback).
 */
#include <stdio.h>
#include <stdlib.h>
#include <string.h>

#include "minio_ec.h"

#define CHECK(call)                                                          \
    do {                                                                     \
        mec_status _s = (call);                                              \
        if (_s != MEC_OK) {                                                  \
            fprintf(stderr, "%s -> %d (%s)\n", #call, (int)_s,               \
                    mec_last_error());                                       \
            return 1;                                                        \
        }                                                                    \
    } while (0)

/* xoshiro-ish deterministic filler (seeded; independence from libc rand) */
static void fill(uint8_t *p, size_t n, uint64_t seed) {
    uint64_t s = seed * 0x9e3779b97f4a7c15ull + 1;
    for (size_t i = 0; i < n; i++) {
        s ^= s << 13;
        s ^= s >> 7;
        s ^= s << 17;
        p[i] = (uint8_t)(s >> 33);
    }
}

int main(void) {
    printf("mec_version=%d devices=%d\n", mec_version(), mec_device_count());

    const int d = 4, p = 2, total = d + p;
    const int64_t bs = 8192;
    mec_ctx *ctx = NULL;
    CHECK(mec_ctx_create(d, p, bs, 0, &ctx));

    const int64_t S = mec_shard_size(bs, d);

    /* ---- 2. encode one block, fused parity + HH256S sums ---- */
    uint8_t *data = malloc(bs), *parity = malloc((size_t)p * S);
    uint8_t sums[6 * 32];
    fill(data, (size_t)bs, 42);
    CHECK(mec_encode_batch(ctx, 1, data, bs, parity, MEC_BITROT_HIGHWAYHASH256S,
                           sums));

    /* ---- 3. erase 2 shards (one data, one parity), reconstruct ---- */
    uint8_t *shards = malloc((size_t)total * S);
    memcpy(shards, data, (size_t)d * S); /* Split aliases the block */
    memcpy(shards + (size_t)d * S, parity, (size_t)p * S);
    uint8_t *ref = malloc((size_t)total * S);
    memcpy(ref, shards, (size_t)total * S);
    uint8_t present[6] = {1, 0, 1, 1, 0, 1};
    memset(shards + 1 * S, 0, (size_t)S);
    memset(shards + 4 * S, 0, (size_t)S);
    CHECK(mec_reconstruct_batch(ctx, 1, shards, present, S, 0));
    if (memcmp(shards, ref, (size_t)total * S) != 0) {
        fprintf(stderr, "reconstruct mismatch\n");
        return 1;
    }

    /* ---- 4. streaming format round trip with a corrupt drive ---- */
    const int64_t obj_len = 3 * bs + 1234; /* ragged last block */
    uint8_t *obj = malloc((size_t)obj_len);
    fill(obj, (size_t)obj_len, 7);
    int64_t fsz = mec_bitrot_shard_file_size(
        mec_shard_file_size(bs, d, obj_len), S, MEC_BITROT_HIGHWAYHASH256S);
    uint8_t *drives[6];
    for (int s = 0; s < total; s++) drives[s] = malloc((size_t)fsz);
    CHECK(mec_encode_stream(ctx, obj, obj_len, MEC_BITROT_HIGHWAYHASH256S,
                            drives, NULL));
    drives[2][40] ^= 0xff; /* flip a byte inside drive 2's first shard */
    uint8_t *out = malloc((size_t)obj_len);
    CHECK(mec_decode_stream(ctx, (const uint8_t *const *)drives, NULL,
                            MEC_BITROT_HIGHWAYHASH256S, obj_len, 0, obj_len,
                            out));
    if (memcmp(out, obj, (size_t)obj_len) != 0) {
        fprintf(stderr, "stream round-trip mismatch\n");
        return 1;
    }

    mec_ctx_destroy(ctx);
    printf("PASS\n");
    return 0;
}
