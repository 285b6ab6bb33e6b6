/* ORACLE — TEST INFRASTRUCTURE ONLY (see oracle.h header).
 *
 * HighwayHash-256, portable restatement of the published HighwayHash
 * reference algorithm, as used by minio/highwayhash v1.0.3 (reference
 * go.mod:58; call site cmd/bitrot.go:55-59 with the magic key at
 * cmd/bitrot.go:37).  Pinned bit-exactly for lengths that are multiples of
 * 32 by the cmd/bitrot.go:225-230 chained vectors (tests/golden/).
 */
#include "oracle.h"
#include "hh_internal.h"
#include <string.h>

typedef mo_hh_state hh_state;  /* shared with simd.c (hh_internal.h) */

static const uint64_t hh_init0[4] = {0xdbe6d5d5fe4cce2full, 0xa4093822299f31d0ull,
                                     0x13198a2e03707344ull, 0x243f6a8885a308d3ull};
static const uint64_t hh_init1[4] = {0x3bd39e10cb0ef593ull, 0xc0acf169b5f18a8cull,
                                     0xbe5466cf34e90c6cull, 0x452821e638d01377ull};

static uint64_t le64(const uint8_t *p) {
    uint64_t v;
    memcpy(&v, p, 8); /* little-endian host (x86-64 / gfx950 host) */
    return v;
}

static void hh_reset(hh_state *s, const uint64_t key[4]) {
    for (int i = 0; i < 4; i++) {
        s->mul0[i] = hh_init0[i];
        s->mul1[i] = hh_init1[i];
        s->v0[i] = hh_init0[i] ^ key[i];
        s->v1[i] = hh_init1[i] ^ ((key[i] >> 32) | (key[i] << 32));
    }
}

static void hh_zipper_merge_add(uint64_t v1, uint64_t v0, uint64_t *add1,
                                uint64_t *add0) {
    *add0 += (((v0 & 0xff000000ull) | (v1 & 0xff00000000ull)) >> 24) |
             (((v0 & 0xff0000000000ull) | (v1 & 0xff000000000000ull)) >> 16) |
             (v0 & 0xff0000ull) | ((v0 & 0xff00ull) << 32) |
             ((v1 & 0xff00000000000000ull) >> 8) | (v0 << 56);
    *add1 += (((v1 & 0xff000000ull) | (v0 & 0xff00000000ull)) >> 24) |
             (v1 & 0xff0000ull) | ((v1 & 0xff0000000000ull) >> 16) |
             ((v1 & 0xff00ull) << 24) | ((v0 & 0xff000000000000ull) >> 8) |
             ((v1 & 0xffull) << 48) | (v0 & 0xff00000000000000ull);
}

static void hh_update(hh_state *s, const uint64_t lanes[4]) {
    for (int i = 0; i < 4; i++) {
        s->v1[i] += s->mul0[i] + lanes[i];
        s->mul0[i] ^= (s->v1[i] & 0xffffffffull) * (s->v0[i] >> 32);
        s->v0[i] += s->mul1[i];
        s->mul1[i] ^= (s->v0[i] & 0xffffffffull) * (s->v1[i] >> 32);
    }
    hh_zipper_merge_add(s->v1[1], s->v1[0], &s->v0[1], &s->v0[0]);
    hh_zipper_merge_add(s->v1[3], s->v1[2], &s->v0[3], &s->v0[2]);
    hh_zipper_merge_add(s->v0[1], s->v0[0], &s->v1[1], &s->v1[0]);
    hh_zipper_merge_add(s->v0[3], s->v0[2], &s->v1[3], &s->v1[2]);
}

static void hh_update_packet(hh_state *s, const uint8_t *packet) {
    uint64_t lanes[4];
    for (int i = 0; i < 4; i++) lanes[i] = le64(packet + 8 * i);
    hh_update(s, lanes);
}

static void hh_rotate32by(uint64_t count, hh_state *s) {
    for (int i = 0; i < 4; i++) {
        uint32_t half0 = (uint32_t)(s->v1[i] & 0xffffffffull);
        uint32_t half1 = (uint32_t)(s->v1[i] >> 32);
        s->v1[i] = (uint32_t)((half0 << count) | (half0 >> (32 - count)));
        s->v1[i] |= (uint64_t)((half1 << count) | (half1 >> (32 - count))) << 32;
    }
}

static void hh_update_remainder(hh_state *s, const uint8_t *bytes,
                                size_t size_mod32) {
    const size_t size_mod4 = size_mod32 & 3;
    const uint8_t *remainder = bytes + (size_mod32 & ~(size_t)3);
    uint8_t packet[32] = {0};
    for (int i = 0; i < 4; i++)
        s->v0[i] += ((uint64_t)size_mod32 << 32) + size_mod32;
    hh_rotate32by(size_mod32, s);
    for (size_t i = 0; i < (size_mod32 & ~(size_t)3); i++) packet[i] = bytes[i];
    if (size_mod32 & 16) {
        for (int i = 0; i < 4; i++)
            packet[28 + i] = remainder[i + (ptrdiff_t)size_mod4 - 4];
    } else if (size_mod4) {
        packet[16 + 0] = remainder[0];
        packet[16 + 1] = remainder[size_mod4 >> 1];
        packet[16 + 2] = remainder[size_mod4 - 1];
    }
    hh_update_packet(s, packet);
}

static void hh_permute_update(hh_state *s) {
    uint64_t permuted[4];
    permuted[0] = (s->v0[2] >> 32) | (s->v0[2] << 32);
    permuted[1] = (s->v0[3] >> 32) | (s->v0[3] << 32);
    permuted[2] = (s->v0[0] >> 32) | (s->v0[0] << 32);
    permuted[3] = (s->v0[1] >> 32) | (s->v0[1] << 32);
    hh_update(s, permuted);
}

static void hh_modular_reduction(uint64_t a3_unmasked, uint64_t a2,
                                 uint64_t a1, uint64_t a0, uint64_t *m1,
                                 uint64_t *m0) {
    uint64_t a3 = a3_unmasked & 0x3fffffffffffffffull;
    *m1 = a1 ^ ((a3 << 1) | (a2 >> 63)) ^ ((a3 << 2) | (a2 >> 62));
    *m0 = a0 ^ (a2 << 1) ^ (a2 << 2);
}

void mo_hh256(const uint8_t key32[32], const uint8_t *msg, size_t len,
              uint8_t out[32]) {
    uint64_t key[4];
    for (int i = 0; i < 4; i++) key[i] = le64(key32 + 8 * i);
    hh_state s;
    hh_reset(&s, key);
    while (len >= 32) {
        hh_update_packet(&s, msg);
        msg += 32;
        len -= 32;
    }
    if (len > 0) hh_update_remainder(&s, msg, len);
    for (int i = 0; i < 10; i++) hh_permute_update(&s);
    uint64_t hash[4];
    hh_modular_reduction(s.v1[1] + s.mul1[1], s.v1[0] + s.mul1[0],
                         s.v0[1] + s.mul0[1], s.v0[0] + s.mul0[0], &hash[1],
                         &hash[0]);
    hh_modular_reduction(s.v1[3] + s.mul1[3], s.v1[2] + s.mul1[2],
                         s.v0[3] + s.mul0[3], s.v0[2] + s.mul0[2], &hash[3],
                         &hash[2]);
    memcpy(out, hash, 32);
}

/* ---- shared entry points for the SIMD bench leg (simd.c) ----------------
 * The SIMD main loop must agree bit-for-bit with this restatement; it
 * reuses the scalar reset/remainder/finalization so only the 32-B packet
 * loop differs (pinned by test_oracle_properties.py::test_simd_matches_scalar). */

void mo_hh_reset_(mo_hh_state *s, const uint8_t key32[32]) {
    uint64_t key[4];
    for (int i = 0; i < 4; i++) key[i] = le64(key32 + 8 * i);
    hh_reset(s, key);
}

void mo_hh_update_packet_(mo_hh_state *s, const uint8_t *packet) {
    hh_update_packet(s, packet);
}

void mo_hh_finish_(mo_hh_state *s, const uint8_t *tail, size_t tail_len,
                   uint8_t out[32]) {
    if (tail_len > 0) hh_update_remainder(s, tail, tail_len);
    for (int i = 0; i < 10; i++) hh_permute_update(s);
    uint64_t hash[4];
    hh_modular_reduction(s->v1[1] + s->mul1[1], s->v1[0] + s->mul1[0],
                         s->v0[1] + s->mul0[1], s->v0[0] + s->mul0[0],
                         &hash[1], &hash[0]);
    hh_modular_reduction(s->v1[3] + s->mul1[3], s->v1[2] + s->mul1[2],
                         s->v0[3] + s->mul0[3], s->v0[2] + s->mul0[2],
                         &hash[3], &hash[2]);
    memcpy(out, hash, 32);
}
