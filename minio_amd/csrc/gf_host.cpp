/* See gf_host.h. */
#include "gf_host.h"
#include <cstring>
#include <mutex>
#include <vector>

namespace mec {

static uint8_t exp_tbl[512];
static uint8_t log_tbl[256];
static std::once_flag gf_once;

static void gf_init() {
    std::call_once(gf_once, [] {
        int x = 1;
        for (int i = 0; i < 255; i++) {
            exp_tbl[i] = (uint8_t)x;
            log_tbl[x] = (uint8_t)i;
            x <<= 1;
            if (x & 0x100) x ^= 0x11D;
        }
        for (int i = 255; i < 512; i++) exp_tbl[i] = exp_tbl[i - 255];
        log_tbl[0] = 0;
    });
}

uint8_t gf_mul(uint8_t a, uint8_t b) {
    gf_init();
    if (a == 0 || b == 0) return 0;
    return exp_tbl[log_tbl[a] + log_tbl[b]];
}

static uint8_t gf_inv(uint8_t a) {
    gf_init();
    if (a == 0) return 0;
    return exp_tbl[255 - log_tbl[a]];
}

uint8_t gf_exp(uint8_t a, int n) {
    gf_init();
    if (n == 0) return 1;
    if (a == 0) return 0;
    return exp_tbl[(int)log_tbl[a] * n % 255];
}

/* Gauss-Jordan inversion of an n x n matrix (row-major). */
static bool invert(const uint8_t *in, int n, uint8_t *out) {
    std::vector<uint8_t> w((size_t)n * 2 * n, 0);
    auto W = [&](int r, int c) -> uint8_t & { return w[(size_t)r * 2 * n + c]; };
    for (int r = 0; r < n; r++) {
        memcpy(&W(r, 0), in + (size_t)r * n, (size_t)n);
        W(r, n + r) = 1;
    }
    for (int r = 0; r < n; r++) {
        if (W(r, r) == 0) {
            int swap = -1;
            for (int rb = r + 1; rb < n; rb++)
                if (W(rb, r) != 0) { swap = rb; break; }
            if (swap < 0) return false;
            for (int c = 0; c < 2 * n; c++) std::swap(W(r, c), W(swap, c));
        }
        uint8_t piv = W(r, r);
        if (piv != 1) {
            uint8_t iv = gf_inv(piv);
            for (int c = 0; c < 2 * n; c++) W(r, c) = gf_mul(W(r, c), iv);
        }
        for (int rb = 0; rb < n; rb++) {
            if (rb == r || W(rb, r) == 0) continue;
            uint8_t f = W(rb, r);
            for (int c = 0; c < 2 * n; c++)
                W(rb, c) ^= gf_mul(f, W(r, c));
        }
    }
    for (int r = 0; r < n; r++) memcpy(out + (size_t)r * n, &W(r, n), (size_t)n);
    return true;
}

bool build_encode_matrix(int d, int p, uint8_t *out) {
    gf_init();
    if (d <= 0 || p < 0 || d + p > kMaxShards) return false;
    int total = d + p;
    std::vector<uint8_t> vm((size_t)total * d), top((size_t)d * d),
        topinv((size_t)d * d);
    for (int r = 0; r < total; r++)
        for (int c = 0; c < d; c++)
            vm[(size_t)r * d + c] = gf_exp((uint8_t)r, c);
    memcpy(top.data(), vm.data(), (size_t)d * d);
    if (!invert(top.data(), d, topinv.data())) return false;
    for (int r = 0; r < total; r++)
        for (int c = 0; c < d; c++) {
            uint8_t acc = 0;
            for (int k = 0; k < d; k++)
                acc ^= gf_mul(vm[(size_t)r * d + k], topinv[(size_t)k * d + c]);
            out[(size_t)r * d + c] = acc;
        }
    return true;
}

bool build_decode_plan(const uint8_t *enc_matrix, int d, int p,
                       const uint8_t *present, int data_only, int *src_idx,
                       int *dst_idx, int *n_dst, uint8_t *dec) {
    int total = d + p;
    int n_present = 0;
    for (int i = 0; i < total; i++)
        if (present[i]) n_present++;
    if (n_present < d) return false;

    std::vector<uint8_t> sub((size_t)d * d), subinv((size_t)d * d);
    int r = 0;
    for (int i = 0; i < total && r < d; i++) {
        if (!present[i]) continue;
        memcpy(sub.data() + (size_t)r * d, enc_matrix + (size_t)i * d,
               (size_t)d);
        src_idx[r] = i;
        r++;
    }
    if (!invert(sub.data(), d, subinv.data())) return false;

    int nd = 0;
    /* missing data shards: decode-matrix rows */
    for (int t = 0; t < d; t++) {
        if (present[t]) continue;
        dst_idx[nd] = t;
        memcpy(dec + (size_t)nd * d, subinv.data() + (size_t)t * d, (size_t)d);
        nd++;
    }
    /* missing parity shards (full Reconstruct): encode rows composed with
     * the decode matrix so parity is expressed over the SAME src shards:
     * parity_t = enc[t] . data = enc[t] . (subinv . src) */
    if (!data_only) {
        for (int t = d; t < total; t++) {
            if (present[t]) continue;
            dst_idx[nd] = t;
            for (int c = 0; c < d; c++) {
                uint8_t acc = 0;
                for (int k = 0; k < d; k++)
                    acc ^= gf_mul(enc_matrix[(size_t)t * d + k],
                                  subinv[(size_t)k * d + c]);
                dec[(size_t)nd * d + c] = acc;
            }
            nd++;
        }
    }
    *n_dst = nd;
    return true;
}

} // namespace mec
