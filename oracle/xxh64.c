/* ORACLE — TEST INFRASTRUCTURE ONLY (see oracle.h header).
 *
 * xxHash64 (cespare/xxhash/v2, reference go.mod:23) — used only to check the
 * erasureSelfTest fingerprints (reference cmd/erasure-coding.go:177-184).
 */
#include "oracle.h"
#include <string.h>

#define P1 0x9E3779B185EBCA87ull
#define P2 0xC2B2AE3D27D4EB4Full
#define P3 0x165667B19E3779F9ull
#define P4 0x85EBCA77C2B2AE63ull
#define P5 0x27D4EB2F165667C5ull

static uint64_t rol(uint64_t x, int n) { return (x << n) | (x >> (64 - n)); }

static uint64_t rd64(const uint8_t *p) {
    uint64_t v;
    memcpy(&v, p, 8);
    return v;
}
static uint32_t rd32(const uint8_t *p) {
    uint32_t v;
    memcpy(&v, p, 4);
    return v;
}

static uint64_t round1(uint64_t acc, uint64_t input) {
    acc += input * P2;
    return rol(acc, 31) * P1;
}
static uint64_t merge_round(uint64_t acc, uint64_t val) {
    acc ^= round1(0, val);
    return acc * P1 + P4;
}

uint64_t mo_xxh64(const uint8_t *p, size_t len, uint64_t seed) {
    const uint8_t *end = p + len;
    uint64_t h;
    if (len >= 32) {
        uint64_t v1 = seed + P1 + P2, v2 = seed + P2, v3 = seed,
                 v4 = seed - P1;
        const uint8_t *limit = end - 32;
        do {
            v1 = round1(v1, rd64(p));
            v2 = round1(v2, rd64(p + 8));
            v3 = round1(v3, rd64(p + 16));
            v4 = round1(v4, rd64(p + 24));
            p += 32;
        } while (p <= limit);
        h = rol(v1, 1) + rol(v2, 7) + rol(v3, 12) + rol(v4, 18);
        h = merge_round(h, v1);
        h = merge_round(h, v2);
        h = merge_round(h, v3);
        h = merge_round(h, v4);
    } else {
        h = seed + P5;
    }
    h += (uint64_t)len;
    while (p + 8 <= end) {
        h ^= round1(0, rd64(p));
        h = rol(h, 27) * P1 + P4;
        p += 8;
    }
    if (p + 4 <= end) {
        h ^= (uint64_t)rd32(p) * P1;
        h = rol(h, 23) * P2 + P3;
        p += 4;
    }
    while (p < end) {
        h ^= (uint64_t)(*p) * P5;
        h = rol(h, 11) * P1;
        p++;
    }
    h ^= h >> 33;
    h *= P2;
    h ^= h >> 29;
    h *= P3;
    h ^= h >> 32;
    return h;
}
