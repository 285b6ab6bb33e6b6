/* ORACLE — TEST INFRASTRUCTURE ONLY (see oracle.h header).
 *
 * BLAKE2b-512 unkeyed (RFC 7693), bitrot algorithm at reference
 * cmd/bitrot.go:51-53 (golang.org/x/crypto/blake2b.New512(nil)).  Pinned by
 * the cmd/bitrot.go:227 chained vector (tests/golden/).
 */
#include "oracle.h"
#include <string.h>

static const uint64_t IV[8] = {
    0x6a09e667f3bcc908ull, 0xbb67ae8584caa73bull, 0x3c6ef372fe94f82bull,
    0xa54ff53a5f1d36f1ull, 0x510e527fade682d1ull, 0x9b05688c2b3e6c1full,
    0x1f83d9abfb41bd6bull, 0x5be0cd19137e2179ull};

static const uint8_t SIGMA[12][16] = {
    {0, 1, 2, 3, 4, 5, 6, 7, 8, 9, 10, 11, 12, 13, 14, 15},
    {14, 10, 4, 8, 9, 15, 13, 6, 1, 12, 0, 2, 11, 7, 5, 3},
    {11, 8, 12, 0, 5, 2, 15, 13, 10, 14, 3, 6, 7, 1, 9, 4},
    {7, 9, 3, 1, 13, 12, 11, 14, 2, 6, 5, 10, 4, 0, 15, 8},
    {9, 0, 5, 7, 2, 4, 10, 15, 14, 1, 11, 12, 6, 8, 3, 13},
    {2, 12, 6, 10, 0, 11, 8, 3, 4, 13, 7, 5, 15, 14, 1, 9},
    {12, 5, 1, 15, 14, 13, 4, 10, 0, 7, 6, 3, 9, 2, 8, 11},
    {13, 11, 7, 14, 12, 1, 3, 9, 5, 0, 15, 4, 8, 6, 2, 10},
    {6, 15, 14, 9, 11, 3, 0, 8, 12, 2, 13, 7, 1, 4, 10, 5},
    {10, 2, 8, 4, 7, 6, 1, 5, 15, 11, 9, 14, 3, 12, 13, 0},
    {0, 1, 2, 3, 4, 5, 6, 7, 8, 9, 10, 11, 12, 13, 14, 15},
    {14, 10, 4, 8, 9, 15, 13, 6, 1, 12, 0, 2, 11, 7, 5, 3}};

static uint64_t rotr64(uint64_t x, int n) { return (x >> n) | (x << (64 - n)); }

static void G(uint64_t *v, int a, int b, int c, int d, uint64_t x, uint64_t y) {
    v[a] = v[a] + v[b] + x;
    v[d] = rotr64(v[d] ^ v[a], 32);
    v[c] = v[c] + v[d];
    v[b] = rotr64(v[b] ^ v[c], 24);
    v[a] = v[a] + v[b] + y;
    v[d] = rotr64(v[d] ^ v[a], 16);
    v[c] = v[c] + v[d];
    v[b] = rotr64(v[b] ^ v[c], 63);
}

static void b2b_compress(uint64_t h[8], const uint8_t *block, uint64_t t,
                         int last) {
    uint64_t m[16], v[16];
    memcpy(m, block, 128); /* little-endian host */
    for (int i = 0; i < 8; i++) v[i] = h[i];
    for (int i = 0; i < 8; i++) v[8 + i] = IV[i];
    v[12] ^= t; /* t0; t1 = 0 for messages < 2^64 bytes */
    if (last) v[14] = ~v[14];
    for (int r = 0; r < 12; r++) {
        const uint8_t *s = SIGMA[r];
        G(v, 0, 4, 8, 12, m[s[0]], m[s[1]]);
        G(v, 1, 5, 9, 13, m[s[2]], m[s[3]]);
        G(v, 2, 6, 10, 14, m[s[4]], m[s[5]]);
        G(v, 3, 7, 11, 15, m[s[6]], m[s[7]]);
        G(v, 0, 5, 10, 15, m[s[8]], m[s[9]]);
        G(v, 1, 6, 11, 12, m[s[10]], m[s[11]]);
        G(v, 2, 7, 8, 13, m[s[12]], m[s[13]]);
        G(v, 3, 4, 9, 14, m[s[14]], m[s[15]]);
    }
    for (int i = 0; i < 8; i++) h[i] ^= v[i] ^ v[8 + i];
}

void mo_blake2b512(const uint8_t *msg, size_t len, uint8_t out[64]) {
    uint64_t h[8];
    memcpy(h, IV, sizeof(h));
    h[0] ^= 0x01010000ull ^ 64; /* digest_length=64, key_length=0, fanout=depth=1 */
    size_t off = 0;
    /* all full blocks except the last one (last block is always compressed
     * with the finalization flag, even when len is a multiple of 128) */
    while (len - off > 128) {
        b2b_compress(h, msg + off, (uint64_t)(off + 128), 0);
        off += 128;
    }
    uint8_t tail[128] = {0};
    size_t rem = len - off; /* 0..128; 0 only when len == 0 */
    memcpy(tail, msg + off, rem);
    b2b_compress(h, tail, (uint64_t)len, 1);
    memcpy(out, h, 64);
}
