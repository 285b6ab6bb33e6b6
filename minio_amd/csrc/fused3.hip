/* Producer/consumer fused erasure-encode + HighwayHash-256 kernel (v3).
 *
 * Single-pass semantics of cmd/erasure-coding.go:85 (GF parity) +
 * cmd/bitrot-streaming.go:57-59 (per-shard HH256) at 1.5 B of HBM traffic
 * per input byte: data read once, parity written once, every hash packet
 * consumed from LDS.
 *
 * Why v3 beats fused2 (r1, 0.935 ms vs the pipelined pair's 0.75): the r1
 * consumer was the pair-lane hash — 2 waves whose per-packet dependency
 * chain (~265 cyc solo) made the hash side of every tile LONGER than the
 * producer side, so hash waves sharing a SIMD with producers stretched
 * ~2x and the ring stalled.  The r2 4-lane-per-chain hash (hh256_batch4_
 * kernel) halves the per-lane instruction count and doubles hash wave
 * count: 4 producer + 4 consumer waves, one of each per SIMD, with the
 * consumer needing only ~330 wave-instr per tile against the producer's
 * ~1800 — the hash now FITS INSIDE the producer's tile time instead of
 * dominating it.  Measured effect: the fused step becomes memory-bound on
 * the 1.61 GiB it must move instead of issue-bound on the hash.
 *
 * Structure per 512-thread workgroup (G = 64/TOT blocks):
 *   waves 0..3 (producers): per 1-KiB tile, each lane owns (block,16-B
 *     column) tasks — load d inputs, write them to the LDS slot, constexpr
 *     ladder, parity to LDS + HBM (nontemporal).
 *   waves 4..7 (consumers): 4 lanes per chain (one HighwayHash lane each,
 *     DPP zipper — see hh256_batch4_kernel), 32 packets per tile from LDS.
 * Hand-off: 2-slot LDS ring with per-slot epoch flags, 4-wave arrival
 * counters, bounded spins (same protocol as fused2, which is bit-exact).
 * Consumers take static s_setprio(1): they are the younger (arbitration-
 * losing) half and sit on the latency-critical chains
 * (MI355X_MICROARCH.md "Two waves per SIMD" items 2/4).
 *
 * Eligibility: shard_len % 1024 == 0 and a compiled (d,p) specialization;
 * anything else falls back to the kernel pair.
 */
#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstdlib>

#include "kernels.h"
#include "ec_matrices_gen.h"

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

__device__ __forceinline__ int lds_poll(int *flag, int want) {
    for (int spin = 0; spin < (1 << 24); spin++) {
        if (__hip_atomic_load(flag, __ATOMIC_RELAXED,
                              __HIP_MEMORY_SCOPE_WORKGROUP) == want)
            return 0;
        __builtin_amdgcn_s_sleep(2);
    }
    return 1;
}

} // namespace fused3

template <int D, int P, const uint8_t (&MAT)[P][D], int WAVES = 8,
          int NPROD_T = 4>
__global__ void __launch_bounds__(WAVES * 64) fused3_encode_hh_kernel(
    FusedArgs a) {
    using namespace fused3;
    constexpr int TOT = D + P;
    /* blocks per workgroup: consumer lanes / (4 lanes per chain x TOT) */
    constexpr int G = (WAVES - NPROD_T) * 16 / TOT;
    constexpr int TILE = 1024;         /* bytes per shard per ring tile */
    constexpr int ROW = TILE + 16;     /* bank-skewed LDS row */
    constexpr int RING = 2;
    constexpr int SLOT = G * TOT * ROW;
    constexpr int NPROD = NPROD_T;     /* producer waves (first half) */
    __shared__ uint8_t lds[RING * SLOT + 64];
    int *flags = (int *)&lds[RING * SLOT];
    /* flags[0..1]=ready epoch, [2..3]=cons epoch, [4..5]=prod arrivals,
     * [6..7]=cons arrivals */

    const int tid = threadIdx.x;
    const int wid = tid >> 6;
    const int64_t b0 = (int64_t)blockIdx.x * G;
    const int64_t S = a.shard_len; /* multiple of TILE (launcher) */
    const int64_t stride = a.row_stride;
    const int64_t n_iter = S / TILE;

    if (tid < 8) flags[tid] = (tid == 2) ? 0 : (tid == 3 ? 1 : 0);
    __syncthreads(); /* the ONLY workgroup barrier: flag init */

    if (wid < NPROD) {
        /* ---- producer: (block g, 16-B column o) tasks ---- */
        const int lane_g = wid * 64 + (tid & 63);
        for (int64_t it = 0; it < n_iter; it++) {
            const int slot = (int)(it & 1);
            uint8_t *sb = &lds[slot * SLOT];
            if (lds_poll(&flags[2 + slot], (int)it)) return; /* timeout */
            for (int task = lane_g; task < G * (TILE / 16);
                 task += NPROD * 64) {
                const int g = task / (TILE / 16);
                const int o = task % (TILE / 16);
                if (b0 + g >= a.n) continue;
                const int64_t off = it * TILE + (int64_t)o * 16;
                uint4 acc[P];
#pragma unroll
                for (int i = 0; i < P; i++) acc[i] = uint4{0, 0, 0, 0};
#pragma unroll
                for (int k = 0; k < D; k++) {
                    uint4 pw = *(const uint4 *)(a.data +
                                                ((b0 + g) * D + k) * stride +
                                                off);
                    *(uint4 *)&sb[(g * TOT + k) * ROW + o * 16] = pw;
                    /* two-bit ladder with xor3 pair-folding (constexpr on
                     * MAT, same schedule as gf_encode_kernel) */
                    uint4 cur = pw, nxt;
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
                    *(uint4 *)&sb[(g * TOT + D + i) * ROW + o * 16] = acc[i];
                    typedef unsigned int v4u
                        __attribute__((ext_vector_type(4)));
                    v4u v = {acc[i].x, acc[i].y, acc[i].z, acc[i].w};
                    __builtin_nontemporal_store(
                        v,
                        (v4u *)(a.parity + ((b0 + g) * P + i) * stride +
                                off));
                }
            }
            /* LDS (ds_write) traffic must land before the publish; global
             * parity stores are not part of the handoff (lgkmcnt(0) only) */
            __builtin_amdgcn_s_waitcnt(0xc07f);
            if ((tid & 63) == 0) {
                int prev = __hip_atomic_fetch_add(
                    &flags[4 + slot], 1, __ATOMIC_RELAXED,
                    __HIP_MEMORY_SCOPE_WORKGROUP);
                if (prev == NPROD - 1) {
                    __hip_atomic_store(&flags[4 + slot], 0, __ATOMIC_RELAXED,
                                       __HIP_MEMORY_SCOPE_WORKGROUP);
                    __hip_atomic_store(&flags[slot], (int)it + 1,
                                       __ATOMIC_RELEASE,
                                       __HIP_MEMORY_SCOPE_WORKGROUP);
                }
            }
        }
        return;
    }

    /* ---- consumer: 4 lanes per chain (hh256_batch4 shape) ---- */
    const int ln = (wid - NPROD) * 64 + (tid & 63); /* 0..255 */
    const int cp = ln >> 2;       /* chain index in WG: 0..G*TOT-1 */
    const int j = ln & 3;         /* HighwayHash lane */
    const int cg = cp / TOT;
    const int cs = cp % TOT;
    const bool act = (cp < G * TOT) && (b0 + cg < a.n);
    const uint32_t S3 = (j & 1) ? 0x07000601u : 0x00070106u;

    /* consumers are the younger dispatch half AND the latency-critical
     * side: one static priority raise, no per-tile flips */
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

    for (int64_t it = 0; it < n_iter; it++) {
        const int slot = (int)(it & 1);
        if (lds_poll(&flags[slot], (int)it + 1)) return; /* timeout */
        __hip_atomic_load(&flags[slot], __ATOMIC_ACQUIRE,
                          __HIP_MEMORY_SCOPE_WORKGROUP);
        if (act) {
            const uint8_t *row =
                &lds[slot * SLOT + (cg * TOT + cs) * ROW + 8 * j];
#pragma unroll 8
            for (int t = 0; t < TILE / 32; t++)
                hh1_update(s, *(const uint64_t *)(row + 32 * t), S3);
        }
        __builtin_amdgcn_s_waitcnt(0xc07f); /* lgkmcnt(0): ds_reads done */
        if ((tid & 63) == 0) {
            int prev = __hip_atomic_fetch_add(&flags[6 + slot], 1,
                                              __ATOMIC_RELAXED,
                                              __HIP_MEMORY_SCOPE_WORKGROUP);
            if (prev == WAVES - NPROD - 1) {
                __hip_atomic_store(&flags[6 + slot], 0, __ATOMIC_RELAXED,
                                   __HIP_MEMORY_SCOPE_WORKGROUP);
                __hip_atomic_store(&flags[2 + slot], (int)it + 2,
                                   __ATOMIC_RELEASE,
                                   __HIP_MEMORY_SCOPE_WORKGROUP);
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
    static const char *env = getenv("MEC_FUSED3");
    static const bool enabled = !env || atoi(env) != 0; /* default ON */
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
    /* grid = n/G persistent workgroups (1/CU at the 8-wave config's 122
     * KiB LDS).  Small batches under-fill the chip (the r2 canary caught
     * 52 WGs at batch 256 running 6x slow), so: 8-wave config when it
     * yields a near-full grid, 4-wave config (half the LDS, 2 WGs/CU)
     * for mid batches, kernel-pair fallback below a floor. */
#define X(D, P)                                                              \
    if (d == D && p == P) {                                                  \
        constexpr int G8 = 64 / (D + P);                                     \
        constexpr int G4 = 32 / (D + P);                                     \
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
