/* Single-pass fused erasure-encode + HighwayHash-256 kernel (v3, r2).
 *
 * Semantics of cmd/erasure-coding.go:85 (GF parity) +
 * cmd/bitrot-streaming.go:57-59 (per-shard HH256) in ONE launch.
 *
 * Design history (all measured on MI355X):
 *  - fused2 (r1, LDS data ring): 0.935 ms/step — the pair-lane hash's
 *    per-packet dependency chain dominated every tile.
 *  - fused3-LDS (r2 first cut): producers copied data AND parity through
 *    an LDS ring; 1.39 ms/step.  Root cause: ds_write_b128 costs ~13
 *    cycles of WAVE ISSUE per 16 B (LDS store transfer path,
 *    MI355X_MICROARCH.md §LDS) — 12 writes/task = ~12.5k cycles/tile per
 *    producer wave, 3.4x the GF math itself.  This is also r1 fused2's
 *    "unexplained 3.5k cycles/tile".
 *  - fused3 (this version): NO data plane in LDS at all.
 *      * consumer waves hash DATA shards straight from HBM (the
 *        hh256_batch4 shape: 4 lanes/chain, DPP zipper) — data is read
 *        twice (GF + hash), which is cheap next to LDS store issue;
 *      * consumer waves hash PARITY from the producers' nontemporal
 *        global stores, tile-paced by an LDS done-counter: nt stores
 *        KEEP the line in the writing XCD's L2 (microarch table), the
 *        consumer shares the producer's CU, and its L1 never held those
 *        addresses — so parity re-reads are same-XCD L2 hits, not HBM;
 *      * producers never wait (each tile writes fresh addresses): one
 *        monotonic done-counter, no ring, no backpressure.  If
 *        consumers lag, parity reads fall back to HBM — graceful.
 *    HBM traffic: data 2x read + parity 1x write = 2.5 B/input byte
 *    (vs 3.0 for the kernel pair: the parity re-read stays in L2).
 *
 * Wave layout per WG: NPROD producer waves (first half — the older,
 * arbitration-winning half) + consumer waves at static s_setprio(1)
 * (younger half, latency-critical chains: microarch "Two waves per SIMD"
 * items 2/4).  All consumer lanes poll the tile counter (uniform, LDS),
 * so data and parity lanes run the SAME code path — no divergence; the
 * per-lane base pointer is the only difference.
 *
 * Eligibility: shard_len % 1024 == 0 and a compiled (d,p) specialization;
 * anything else falls back to the kernel pair.
 */
#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstdlib>

#include "kernels.h"
#include "ec_matrices_gen.h"
#include "gf_bs.h"

namespace fused3 {

__device__ __forceinline__ uint32_t gf2x(uint32_t x) {
    uint32_t sel = (x & 0x80808080u) >> 7;
    uint32_t m = __builtin_amdgcn_perm(0u, 0x00001d00u, sel);
    return (uint32_t)__builtin_amdgcn_bitop3_b32(x << 1, 0xfefefefeu, m,
                                                 0x6a); /* (a&b)^c */
}

__device__ __forceinline__ void gf2x4(uint4 &v) {
    v.x = gf2x(v.x); v.y = gf2x(v.y); v.z = gf2x(v.z); v.w = gf2x(v.w);
}

__device__ __forceinline__ uint32_t xor3(uint32_t a, uint32_t b, uint32_t c) {
    return (uint32_t)__builtin_amdgcn_bitop3_b32(a, b, c, 0x96);
}

__device__ __forceinline__ void xor4(uint4 &a, const uint4 &b) {
    a.x ^= b.x; a.y ^= b.y; a.z ^= b.z; a.w ^= b.w;
}

__device__ __forceinline__ void xor34(uint4 &a, const uint4 &b,
                                      const uint4 &c) {
    a.x = xor3(a.x, b.x, c.x);
    a.y = xor3(a.y, b.y, c.y);
    a.z = xor3(a.z, b.z, c.z);
    a.w = xor3(a.w, b.w, c.w);
}

__device__ __forceinline__ uint32_t permb(uint32_t hi, uint32_t lo,
                                          uint32_t sel) {
    return __builtin_amdgcn_perm(hi, lo, sel);
}

__device__ __forceinline__ uint32_t dpp_swap1(uint32_t v) {
    return (uint32_t)__builtin_amdgcn_mov_dpp((int)v, 0xB1, 0xF, 0xF, true);
}
__device__ __forceinline__ uint32_t dpp_swap2(uint32_t v) {
    return (uint32_t)__builtin_amdgcn_mov_dpp((int)v, 0x4E, 0xF, 0xF, true);
}

struct HH1 {
    uint64_t v0, v1, mul0, mul1;
};

__device__ __forceinline__ void hh1_update(HH1 &s, uint64_t w, uint32_t S3) {
    s.v1 += s.mul0 + w;
    s.mul0 ^= (s.v1 & 0xffffffffull) * (s.v0 >> 32);
    s.v0 += s.mul1;
    s.mul1 ^= (s.v0 & 0xffffffffull) * (s.v1 >> 32);
    {
        uint32_t own_lo = (uint32_t)s.v1, own_hi = (uint32_t)(s.v1 >> 32);
        uint32_t p_hi = dpp_swap1(own_hi);
        uint32_t lo = permb(own_hi, own_lo, 0x05020C03u) |
                      permb(0u, p_hi, 0x0C0C000Cu);
        uint32_t hi = permb(p_hi, own_lo, S3);
        s.v0 += ((uint64_t)hi << 32) | lo;
    }
    {
        uint32_t own_lo = (uint32_t)s.v0, own_hi = (uint32_t)(s.v0 >> 32);
        uint32_t p_hi = dpp_swap1(own_hi);
        uint32_t lo = permb(own_hi, own_lo, 0x05020C03u) |
                      permb(0u, p_hi, 0x0C0C000Cu);
        uint32_t hi = permb(p_hi, own_lo, S3);
        s.v1 += ((uint64_t)hi << 32) | lo;
    }
}

template <int NPROD>
__device__ __forceinline__ int lds_poll_min_ge(int *prog, int want) {
    /* bounded relaxed poll over NPROD per-wave monotonic progress
     * counters (no shared arrival counter: producer waves never wait, so
     * a conflated counter publishes early/loses publishes when waves run
     * ahead of each other — r2 bug).  Acquire pairs with each wave's
     * release on success. */
    for (int spin = 0; spin < (1 << 24); spin++) {
        int mn = 1 << 30;
#pragma unroll
        for (int w = 0; w < NPROD; w++) {
            int v = __hip_atomic_load(&prog[w], __ATOMIC_RELAXED,
                                      __HIP_MEMORY_SCOPE_WORKGROUP);
            mn = v < mn ? v : mn;
        }
        if (mn >= want) {
            (void)__hip_atomic_load(&prog[0], __ATOMIC_ACQUIRE,
                                    __HIP_MEMORY_SCOPE_WORKGROUP);
            return 0;
        }
        __builtin_amdgcn_s_sleep(2);
    }
    return 1;
}

} // namespace fused3

template <int D, int P, const uint8_t (&MAT)[P][D], int WAVES = 8,
          int NPROD_T = 4, int PRODBS = 0>
__global__ void __launch_bounds__(WAVES * 64) fused3_encode_hh_kernel(
    FusedArgs a) {
    using namespace fused3;
    constexpr int TOT = D + P;
    /* blocks per workgroup: G = NPROD makes the producer side EXACTLY one
     * 16-B task per lane per tile (and, at the headline batch, a grid
     * that covers all 256 CUs), capped by consumer lanes (4 per chain).
     * The 8-wave config's cyclic wave->SIMD placement then puts exactly
     * one producer + one consumer wave on every SIMD (waves i and i+4
     * share a SIMD) — deterministic balance; the 4-wave config relies on
     * the dispatcher mixing two WGs per CU. */
    constexpr int GCAP = (WAVES - NPROD_T) * 16 / TOT; /* consumer lanes */
    /* byte-ladder producer: 16-B tasks, G = NPROD lanes-exact;
     * bit-sliced producer: 32-B tasks, G = 2*NPROD lanes-exact */
    constexpr int GWANT = PRODBS ? 2 * NPROD_T : NPROD_T;
    constexpr int G = GWANT < GCAP ? GWANT : GCAP;
    constexpr int TILE = 1024; /* bytes per shard per pacing tile */
    constexpr int PUBK = 8;    /* publish cadence (tiles): a vmcnt(0)
        store-drain costs ~1-3 us under full-chip load (microarch
        publish-large row) — draining EVERY tile was 0.62 ms of the r2
        fused step (probe1).  Publishing every PUBK tiles cuts that 8x;
        consumers simply run up to PUBK tiles behind. */
    constexpr int NPROD = NPROD_T;
    __shared__ int prog[NPROD_T]; /* per-producer-wave tiles-done */

    const int tid = threadIdx.x;
    const int wid = tid >> 6;
    const int64_t b0 = (int64_t)blockIdx.x * G;
    const int64_t S = a.shard_len; /* multiple of TILE (launcher) */
    const int64_t stride = a.row_stride;
    const int64_t n_iter = S / TILE;

    if (tid < NPROD_T) prog[tid] = 0;
    __syncthreads(); /* the ONLY workgroup barrier: counter init */
    if (a.probe >= 4) { if (wid >= NPROD_T) return; }
    else if (a.probe == 1 && wid >= NPROD_T) return;

    if (PRODBS && wid < NPROD) {
        /* ---- bit-sliced producer: (block g, 32-B column o) tasks ----
         * Same plane-transpose + constexpr xor3 network as
         * gf_encode_bs_kernel (gf_bs.h): ~2.2x fewer VALU slots than the
         * ladder, which matters because the fused kernel is VALU-pipe-
         * bound (r2 SQ counters: int VALU ~4 cyc/instr). */
        constexpr int TASKS = G * (TILE / 32);
        const int lane_g = wid * 64 + (tid & 63);
        for (int64_t it = 0; it < n_iter; it++) {
            for (int task = lane_g; task < TASKS; task += NPROD * 64) {
                const int g = task / (TILE / 32);
                const int o = task % (TILE / 32);
                if (b0 + g >= a.n) continue;
                const int64_t off = it * TILE + (int64_t)o * 32;
                uint32_t accp[P][8];
#pragma unroll
                for (int i = 0; i < P; i++)
#pragma unroll
                    for (int pb = 0; pb < 8; pb++) accp[i][pb] = 0;
                uint32_t xc[8], xn[8];
                {
                    const uint8_t *row = a.data + ((b0 + g) * D) * stride + off;
                    uint4 lo = *(const uint4 *)row;
                    uint4 hi = *(const uint4 *)(row + 16);
                    xc[0] = lo.x; xc[1] = lo.y; xc[2] = lo.z; xc[3] = lo.w;
                    xc[4] = hi.x; xc[5] = hi.y; xc[6] = hi.z; xc[7] = hi.w;
                }
#pragma unroll
                for (int k = 0; k < D; k++) {
                    if (k + 1 < D) {
                        const uint8_t *row =
                            a.data + ((b0 + g) * D + k + 1) * stride + off;
                        uint4 lo = *(const uint4 *)row;
                        uint4 hi = *(const uint4 *)(row + 16);
                        xn[0] = lo.x; xn[1] = lo.y; xn[2] = lo.z;
                        xn[3] = lo.w; xn[4] = hi.x; xn[5] = hi.y;
                        xn[6] = hi.z; xn[7] = hi.w;
                    }
                    bs_transpose(xc);
                    bs_acc_k<D, P, MAT>(k, xc, accp);
#pragma unroll
                    for (int w = 0; w < 8; w++) xc[w] = xn[w];
                }
#pragma unroll
                for (int i = 0; i < P; i++) {
                    bs_transpose(accp[i]);
                    uint8_t *orow =
                        a.parity + ((b0 + g) * P + i) * stride + off;
                    typedef unsigned int v4u
                        __attribute__((ext_vector_type(4)));
                    v4u vlo = {accp[i][0], accp[i][1], accp[i][2],
                               accp[i][3]};
                    v4u vhi = {accp[i][4], accp[i][5], accp[i][6],
                               accp[i][7]};
                    __builtin_nontemporal_store(vlo, (v4u *)orow);
                    __builtin_nontemporal_store(vhi, (v4u *)(orow + 16));
                }
            }
            if (a.probe != 4 &&
                ((it + 1) % PUBK == 0 || it + 1 == n_iter)) {
                __builtin_amdgcn_s_waitcnt(0x0f70); /* vmcnt(0) */
                if ((tid & 63) == 0)
                    __hip_atomic_store(&prog[wid], (int)it + 1,
                                       __ATOMIC_RELEASE,
                                       __HIP_MEMORY_SCOPE_WORKGROUP);
            }
        }
        return;
    }
    if (wid < NPROD) {
        /* ---- producer: (block g, 16-B column o) tasks per tile ----
         * All of a task's D row loads are issued before any ladder math
         * (the consume-as-you-load form exposes one HBM latency per row
         * — r03 PMC; r2 profile showed 49% of fused wave-cycles parked
         * for exactly this reason), and when each lane owns exactly one
         * task (EXACT), the NEXT tile's loads are issued before the
         * current tile's ladder so the latency rides under compute. */
        constexpr bool EXACT = (G * (TILE / 16) == NPROD * 64);
        const int lane_g = wid * 64 + (tid & 63);
        if (EXACT) {
            const int g = lane_g / (TILE / 16);
            const int o = lane_g % (TILE / 16);
            const bool pact = b0 + g < a.n;
            const uint8_t *rows[D];
#pragma unroll
            for (int k = 0; k < D; k++)
                rows[k] = a.data + ((b0 + g) * D + k) * stride +
                          (int64_t)o * 16;
            uint4 pwsA[D], pwsB[D];
            if (pact) {
#pragma unroll
                for (int k = 0; k < D; k++) pwsA[k] = *(const uint4 *)rows[k];
            }
            for (int64_t it = 0; it < n_iter; it++) {
                if (pact && it + 1 < n_iter) {
#pragma unroll
                    for (int k = 0; k < D; k++)
                        pwsB[k] =
                            *(const uint4 *)(rows[k] + (it + 1) * TILE);
                }
                if (pact) {
                    const int64_t off = it * TILE + (int64_t)o * 16;
                    uint4 acc[P];
#pragma unroll
                    for (int i = 0; i < P; i++) acc[i] = uint4{0, 0, 0, 0};
#pragma unroll
                    for (int k = 0; k < D; k++) {
                        uint4 cur = pwsA[k], nxt;
#pragma unroll
                        for (int bit = 0; bit < 8; bit += 2) {
                            uint32_t needCur = 0, needHi = 0;
#pragma unroll
                            for (int i = 0; i < P; i++) {
                                needCur |= (uint32_t)MAT[i][k] >> bit;
                                needHi |= (uint32_t)MAT[i][k] >> (bit + 1);
                            }
                            if (!needCur) break;
                            if (needHi) {
                                nxt = cur;
                                gf2x4(nxt);
                            }
#pragma unroll
                            for (int i = 0; i < P; i++) {
                                const int b0i = (MAT[i][k] >> bit) & 1;
                                const int b1i = (MAT[i][k] >> (bit + 1)) & 1;
                                if (b0i && b1i) xor34(acc[i], cur, nxt);
                                else if (b0i) xor4(acc[i], cur);
                                else if (b1i) xor4(acc[i], nxt);
                            }
                            if (needHi >> 1) {
                                cur = nxt;
                                gf2x4(cur);
                            } else {
                                break;
                            }
                        }
                    }
#pragma unroll
                    for (int i = 0; i < P; i++) {
                        typedef unsigned int v4u
                            __attribute__((ext_vector_type(4)));
                        v4u v = {acc[i].x, acc[i].y, acc[i].z, acc[i].w};
                        __builtin_nontemporal_store(
                            v, (v4u *)(a.parity +
                                       ((b0 + g) * P + i) * stride + off));
                    }
#pragma unroll
                    for (int k = 0; k < D; k++) pwsA[k] = pwsB[k];
                }
                /* parity stores to L2 before publishing (PUBK cadence) */
                if (a.probe != 4 &&
                    ((it + 1) % PUBK == 0 || it + 1 == n_iter)) {
                    __builtin_amdgcn_s_waitcnt(0x0f70); /* vmcnt(0) */
                    if ((tid & 63) == 0)
                        __hip_atomic_store(&prog[wid], (int)it + 1,
                                           __ATOMIC_RELEASE,
                                           __HIP_MEMORY_SCOPE_WORKGROUP);
                }
            }
            return;
        }
        /* generic task loop (lane count != task count): loads hoisted
         * per task, no cross-tile prefetch */
        for (int64_t it = 0; it < n_iter; it++) {
            for (int task = lane_g; task < G * (TILE / 16);
                 task += NPROD * 64) {
                const int g = task / (TILE / 16);
                const int o = task % (TILE / 16);
                if (b0 + g >= a.n) continue;
                const int64_t off = it * TILE + (int64_t)o * 16;
                uint4 pws[D];
#pragma unroll
                for (int k = 0; k < D; k++)
                    pws[k] = *(const uint4 *)(a.data +
                                              ((b0 + g) * D + k) * stride +
                                              off);
                uint4 acc[P];
#pragma unroll
                for (int i = 0; i < P; i++) acc[i] = uint4{0, 0, 0, 0};
#pragma unroll
                for (int k = 0; k < D; k++) {
                    uint4 cur = pws[k], nxt;
#pragma unroll
                    for (int bit = 0; bit < 8; bit += 2) {
                        uint32_t needCur = 0, needHi = 0;
#pragma unroll
                        for (int i = 0; i < P; i++) {
                            needCur |= (uint32_t)MAT[i][k] >> bit;
                            needHi |= (uint32_t)MAT[i][k] >> (bit + 1);
                        }
                        if (!needCur) break;
                        if (needHi) {
                            nxt = cur;
                            gf2x4(nxt);
                        }
#pragma unroll
                        for (int i = 0; i < P; i++) {
                            const int b0i = (MAT[i][k] >> bit) & 1;
                            const int b1i = (MAT[i][k] >> (bit + 1)) & 1;
                            if (b0i && b1i) xor34(acc[i], cur, nxt);
                            else if (b0i) xor4(acc[i], cur);
                            else if (b1i) xor4(acc[i], nxt);
                        }
                        if (needHi >> 1) {
                            cur = nxt;
                            gf2x4(cur);
                        } else {
                            break;
                        }
                    }
                }
#pragma unroll
                for (int i = 0; i < P; i++) {
                    typedef unsigned int v4u
                        __attribute__((ext_vector_type(4)));
                    v4u v = {acc[i].x, acc[i].y, acc[i].z, acc[i].w};
                    __builtin_nontemporal_store(
                        v,
                        (v4u *)(a.parity + ((b0 + g) * P + i) * stride +
                                off));
                }
            }
            if ((it + 1) % PUBK == 0 || it + 1 == n_iter) {
                __builtin_amdgcn_s_waitcnt(0x0f70); /* vmcnt(0) */
                if ((tid & 63) == 0)
                    __hip_atomic_store(&prog[wid], (int)it + 1,
                                       __ATOMIC_RELEASE,
                                       __HIP_MEMORY_SCOPE_WORKGROUP);
            }
        }
        return;
    }

    /* ---- consumer: 4 lanes per chain (hh256_batch4 shape), tile-paced
     * by the done-counter; data and parity lanes share the code path —
     * only the base pointer differs ---- */
    const int ln = (wid - NPROD) * 64 + (tid & 63);
    const int cp = ln >> 2; /* chain index in WG */
    const int j = ln & 3;   /* HighwayHash lane */
    const int cg = cp / TOT;
    const int cs = cp % TOT;
    const bool act = (cp < G * TOT) && (b0 + cg < a.n);
    const uint32_t S3 = (j & 1) ? 0x07000601u : 0x00070106u;
    const uint8_t *base =
        (cs < D) ? a.data + ((b0 + cg) * D + cs) * stride
                 : a.parity + ((b0 + cg) * P + (cs - D)) * stride;
    const uint8_t *mp = base + 8 * j;

    /* consumers are the younger (arbitration-losing) dispatch half and
     * sit on the latency-critical chains: one static priority raise */
    __builtin_amdgcn_s_setprio(1);

    HH1 s;
    {
        const uint64_t init0[4] = {0xdbe6d5d5fe4cce2full,
                                   0xa4093822299f31d0ull,
                                   0x13198a2e03707344ull,
                                   0x243f6a8885a308d3ull};
        const uint64_t init1[4] = {0x3bd39e10cb0ef593ull,
                                   0xc0acf169b5f18a8cull,
                                   0xbe5466cf34e90c6cull,
                                   0x452821e638d01377ull};
        s.mul0 = init0[j];
        s.mul1 = init1[j];
        s.v0 = init0[j] ^ a.key[j];
        s.v1 = init1[j] ^ ((a.key[j] >> 32) | (a.key[j] << 32));
    }

    /* run ONE TILE BEHIND the producers: tile t+1's 32 loads are issued
     * before tile t's serial chain, so the (L2-or-HBM) load latency rides
     * under ~1.7k cycles of hashing instead of stalling every tile */
    {
        uint64_t qA[TILE / 32], qB[TILE / 32];
        if (n_iter > 0) {
            if (a.probe < 2 && lds_poll_min_ge<NPROD>(prog, 1)) return;
            if (act) {
#pragma unroll
                for (int t = 0; t < TILE / 32; t++)
                    qA[t] = *(const uint64_t *)(mp + 32 * t);
                mp += TILE;
            }
        }
        for (int64_t it = 0; it < n_iter; it++) {
            if (it + 1 < n_iter) {
                if (a.probe < 2 &&
                    lds_poll_min_ge<NPROD>(prog, (int)it + 2))
                    return; /* timeout */
                if (act) {
#pragma unroll
                    for (int t = 0; t < TILE / 32; t++)
                        qB[t] = *(const uint64_t *)(mp + 32 * t);
                    mp += TILE;
                }
            }
            if (act) {
#pragma unroll
                for (int t = 0; t < TILE / 32; t++)
                    hh1_update(s, qA[t], S3);
#pragma unroll
                for (int t = 0; t < TILE / 32; t++) qA[t] = qB[t];
            }
        }
    }

    /* finalize + store sums (hh256_batch4 epilogue) */
    if (act) {
#pragma unroll 1
        for (int r = 0; r < 10; r++) {
            uint32_t p_lo = dpp_swap2((uint32_t)s.v0);
            uint32_t p_hi = dpp_swap2((uint32_t)(s.v0 >> 32));
            hh1_update(s, ((uint64_t)p_lo << 32) | p_hi, S3);
        }
        uint64_t t = s.v0 + s.mul0;
        uint64_t sv = s.v1 + s.mul1;
        uint32_t se_lo = dpp_swap1((uint32_t)sv);
        uint32_t se_hi = dpp_swap1((uint32_t)(sv >> 32));
        uint64_t se = ((uint64_t)se_hi << 32) | se_lo;
        uint64_t out;
        if ((j & 1) == 0) {
            out = t ^ (sv << 1) ^ (sv << 2);
        } else {
            uint64_t a3 = sv & 0x3fffffffffffffffull;
            out = t ^ ((a3 << 1) | (se >> 63)) ^ ((a3 << 2) | (se >> 62));
        }
        *(uint64_t *)(a.sums + ((b0 + cg) * TOT + cs) * 32 + 8 * j) = out;
    }
}

extern "C" hipError_t mec_launch_fused3_encode_hh(int d, int p,
                                                  const FusedArgs *args,
                                                  hipStream_t stream) {
    /* Default OFF (r2): the workload is VALU-pipe-bound once the GF leg
     * is bit-sliced — the kernel pair runs each leg at ~92% of its
     * resource floor with 7-8 waves/SIMD of latency hiding, while the
     * fused persistent layout caps at 1-2 waves/SIMD and pays 40-70%
     * stall overhead (DESIGN.md r2 notes).  Kept in-tree, bit-exact and
     * measured; enable with MEC_FUSED3=1. */
    static const char *env = getenv("MEC_FUSED3");
    static const bool enabled = env && atoi(env) != 0;
    if (!enabled) return hipErrorNotSupported;
    if (args->shard_len % 1024 != 0) return hipErrorNotSupported;
    /* MEC_F3_CFG: force 8- or 4-wave config (perf sweeps); MEC_F3_MIN:
     * minimum grid (WGs) below which the pair path is used instead */
    static const int f3cfg = [] {
        const char *v = getenv("MEC_F3_CFG");
        return v ? atoi(v) : 0;
    }();
    static const int f3min = [] {
        const char *v = getenv("MEC_F3_MIN");
        return v ? atoi(v) : 200;
    }();
    static const int f3probe = [] {
        const char *v = getenv("MEC_F3_PROBE");
        return v ? atoi(v) : 0;
    }();
    FusedArgs aa = *args;
    aa.probe = f3probe;
    args = &aa;
    /* grid = n/G persistent workgroups (1/CU at the 8-wave config's 122
     * KiB LDS).  Small batches under-fill the chip (the r2 canary caught
     * 52 WGs at batch 256 running 6x slow), so: 8-wave config when it
     * yields a near-full grid, 4-wave config (half the LDS, 2 WGs/CU)
     * for mid batches, kernel-pair fallback below a floor. */
#define X(D, P)                                                              \
    if (d == D && p == P) {                                                  \
        constexpr int G8 = 4 < 64 / (D + P) ? 4 : 64 / (D + P);             \
        constexpr int G4 = 2 < 32 / (D + P) ? 2 : 32 / (D + P);             \
        constexpr int G6 = (2 * 2) < (4 * 16 / (D + P))                     \
                               ? (2 * 2) : (4 * 16 / (D + P));               \
        if (G6 >= 1 && (f3cfg == 0 || f3cfg == 6) &&                         \
            (args->n / G6 >= f3min || f3cfg == 6)) {                         \
            dim3 grid((uint32_t)((args->n + G6 - 1) / G6));                  \
            hipLaunchKernelGGL(                                              \
                (fused3_encode_hh_kernel<D, P, MAT_##D##_##P, 6, 2, 1>),     \
                grid, dim3(384), 0, stream, *args);                          \
            return hipGetLastError();                                        \
        }                                                                    \
        if (G8 >= 1 && f3cfg != 4 && (args->n / G8 >= f3min ||            \
                                       f3cfg == 8)) {                                \
            dim3 grid((uint32_t)((args->n + G8 - 1) / G8));                  \
            hipLaunchKernelGGL(                                              \
                (fused3_encode_hh_kernel<D, P, MAT_##D##_##P, 8, 4>), grid,  \
                dim3(512), 0, stream, *args);                                \
            return hipGetLastError();                                        \
        }                                                                    \
        if (G4 >= 1 && f3cfg != 8 && (args->n / G4 >= f3min ||            \
                                       f3cfg == 4)) {                                \
            dim3 grid((uint32_t)((args->n + G4 - 1) / G4));                  \
            hipLaunchKernelGGL(                                              \
                (fused3_encode_hh_kernel<D, P, MAT_##D##_##P, 4, 2>), grid,  \
                dim3(256), 0, stream, *args);                                \
            return hipGetLastError();                                        \
        }                                                                    \
        return hipErrorNotSupported;                                         \
    }
    MEC_SPECIALIZED_GEOS(X)
#undef X
    return hipErrorNotSupported;
}
