/* Producer/consumer fused erasure-encode + HighwayHash-256 kernel (v2).
 *
 * Same semantics as the two-kernel pair behind mec_encode_batch (GF parity
 * per cmd/erasure-coding.go:85 + per-shard HH256 per
 * cmd/bitrot-streaming.go:57-59) with HBM traffic of 1.5 B per input byte:
 * data is read once, parity written once, the hash consumes both from LDS.
 *
 * Why not the barrier-lockstep fused kernel (fused.hip): phase barriers
 * force the GF ladder and the latency-bound hash chains onto the same
 * critical path (measured 2.3x slower).  Here the two run CONCURRENTLY:
 *
 *   waves 0..5 (producers): per 1-KiB tile, each lane owns one (block,
 *     16-B column) task — loads its d inputs from HBM, writes them to the
 *     LDS slot, computes the constexpr-matrix ladder, writes parity to LDS
 *     + HBM.  No producer-side cross-wave dependency at all.
 *   waves 6..7 (consumers): the 2x(d+p)xG hash pair-lanes advance their
 *     chains 32 packets per tile from LDS.
 *
 * Hand-off is a 2-slot LDS ring with per-slot epoch flags (single
 * __shared__ array — a second LDS object would de-pipeline hipcc's waits):
 *   producer of tile t waits cons[t&1]==t, fills the slot,
 *   6-wave arrival counter, last arrival publishes ready[t&1]=t+1;
 *   consumers poll ready, hash, 2-wave arrival, last publishes
 *   cons[t&1]=t+2.  All flag traffic is intra-workgroup LDS (one CU);
 *   every spin is bounded so a logic bug aborts instead of hanging the
 *   device.
 *
 * Eligibility: shard_len % 1024 == 0 and a compiled (d,p) specialization
 * (the headline geometries; anything else falls back to the kernel pair).
 */
#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstdlib>

#include "kernels.h"
#include "ec_matrices_gen.h"

namespace fused2 {

__device__ __forceinline__ uint32_t gf2x(uint32_t x) {
    /* full-rate v_perm reduction select; see kernels.hip gf2x */
    uint32_t sel = (x & 0x80808080u) >> 7;
    return ((x << 1) & 0xfefefefeu) ^
           __builtin_amdgcn_perm(0u, 0x00001d00u, sel);
}

__device__ __forceinline__ void gf2x4(uint4 &v) {
    v.x = gf2x(v.x); v.y = gf2x(v.y); v.z = gf2x(v.z); v.w = gf2x(v.w);
}

__device__ __forceinline__ void xor4(uint4 &a, const uint4 &b) {
    a.x ^= b.x; a.y ^= b.y; a.z ^= b.z; a.w ^= b.w;
}

__device__ __forceinline__ uint32_t permb(uint32_t hi, uint32_t lo,
                                          uint32_t sel) {
    return __builtin_amdgcn_perm(hi, lo, sel);
}

__device__ __forceinline__ uint64_t zip_even(uint64_t A, uint64_t B) {
    uint32_t a_lo = (uint32_t)A, a_hi = (uint32_t)(A >> 32);
    uint32_t b_hi = (uint32_t)(B >> 32);
    uint32_t lo = permb(a_hi, a_lo, 0x05020C03u) | permb(0u, b_hi, 0x0C0C000Cu);
    uint32_t hi = permb(b_hi, a_lo, 0x00070106u);
    return ((uint64_t)hi << 32) | lo;
}

__device__ __forceinline__ uint64_t zip_odd(uint64_t A, uint64_t B) {
    uint32_t a_hi = (uint32_t)(A >> 32);
    uint32_t b_lo = (uint32_t)B, b_hi = (uint32_t)(B >> 32);
    uint32_t lo = permb(b_hi, b_lo, 0x05020C03u) | permb(0u, a_hi, 0x0C0C000Cu);
    uint32_t hi = permb(a_hi, b_lo, 0x07000601u);
    return ((uint64_t)hi << 32) | lo;
}

struct HH2 {
    uint64_t v0[2], v1[2], mul0[2], mul1[2];
};

__device__ __forceinline__ void hh2_update(HH2 &s, uint64_t w0, uint64_t w1) {
    uint64_t w[2] = {w0, w1};
#pragma unroll
    for (int j = 0; j < 2; j++) {
        s.v1[j] += s.mul0[j] + w[j];
        s.mul0[j] ^= (s.v1[j] & 0xffffffffull) * (s.v0[j] >> 32);
        s.v0[j] += s.mul1[j];
        s.mul1[j] ^= (s.v0[j] & 0xffffffffull) * (s.v1[j] >> 32);
    }
    uint64_t t0 = zip_even(s.v1[0], s.v1[1]);
    uint64_t t1 = zip_odd(s.v1[0], s.v1[1]);
    s.v0[0] += t0;
    s.v0[1] += t1;
    uint64_t u0 = zip_even(s.v0[0], s.v0[1]);
    uint64_t u1 = zip_odd(s.v0[0], s.v0[1]);
    s.v1[0] += u0;
    s.v1[1] += u1;
}

__device__ __forceinline__ uint64_t shfl_x(uint64_t v, int mask) {
    return __shfl_xor((unsigned long long)v, mask, 64);
}

__device__ __forceinline__ int lds_poll(int *flag, int want) {
    /* bounded relaxed poll; returns 0 on success, 1 on timeout */
    for (int spin = 0; spin < (1 << 24); spin++) {
        if (__hip_atomic_load(flag, __ATOMIC_RELAXED,
                              __HIP_MEMORY_SCOPE_WORKGROUP) == want)
            return 0;
        __builtin_amdgcn_s_sleep(2);
    }
    return 1;
}

} // namespace fused2

template <int D, int P, const uint8_t (&MAT)[P][D]>
__global__ void __launch_bounds__(512) fused2_encode_hh_kernel(FusedArgs a) {
    using namespace fused2;
    constexpr int TOT = D + P;
    constexpr int G = 128 / (2 * TOT);  /* blocks per workgroup (2 hash waves) */
    constexpr int TILE = 1024;          /* bytes per shard per ring tile */
    constexpr int ROW = TILE + 16;      /* bank-skewed LDS row */
    constexpr int RING = 2;
    constexpr int SLOT = G * TOT * ROW;
    constexpr int NPROD = 6;            /* producer waves */
    /* one __shared__ object: ring data then flags (16-B aligned tail) */
    __shared__ uint8_t lds[RING * SLOT + 64];
    int *flags = (int *)&lds[RING * SLOT];
    /* flags[0..1]=ready epoch, flags[2..3]=cons epoch, [4..5]=prod arrivals,
     * [6..7]=cons arrivals */

    const int tid = threadIdx.x;
    const int wid = tid >> 6;
    const int64_t b0 = (int64_t)blockIdx.x * G;
    const int64_t S = a.shard_len;       /* multiple of TILE (launcher) */
    const int64_t stride = a.row_stride;
    const int64_t n_iter = S / TILE;

    if (tid < 8) flags[tid] = (tid == 2) ? 0 : (tid == 3 ? 1 : 0);
    /* cons epochs start "slot consumable": cons[0]=0 lets tile 0 in,
     * cons[1]=1 lets tile 1 in */
    __syncthreads(); /* the ONLY workgroup barrier: flag init */

    if (wid < NPROD) {
        /* ---- producer: one (block g, 16-B column o) task per lane ---- */
        const int lane_g = wid * 64 + (tid & 63);
        for (int64_t it = 0; it < n_iter; it++) {
            const int slot = (int)(it & 1);
            uint8_t *sb = &lds[slot * SLOT];
            if (lds_poll(&flags[2 + slot], (int)it)) return; /* timeout */
            for (int task = lane_g; task < G * (TILE / 16); task += NPROD * 64) {
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
#pragma unroll
                    for (int bit = 0; bit < 8; bit++) {
                        uint32_t need = 0;
#pragma unroll
                        for (int i = 0; i < P; i++)
                            need |= (uint32_t)MAT[i][k] >> bit;
                        if (!need) break; /* compile-time folded */
                        if (bit) gf2x4(pw);
#pragma unroll
                        for (int i = 0; i < P; i++)
                            if ((MAT[i][k] >> bit) & 1) xor4(acc[i], pw);
                    }
                }
#pragma unroll
                for (int i = 0; i < P; i++) {
                    *(uint4 *)&sb[(g * TOT + D + i) * ROW + o * 16] = acc[i];
                    typedef unsigned int v4u __attribute__((ext_vector_type(4)));
                    v4u v = {acc[i].x, acc[i].y, acc[i].z, acc[i].w};
                    __builtin_nontemporal_store(
                        v, (v4u *)(a.parity + ((b0 + g) * P + i) * stride +
                                   off));
                }
            }
            /* this wave's LDS (ds_write) traffic must land before the
             * publish; global parity stores are not part of the handoff,
             * so leave vmcnt unconstrained (lgkmcnt(0) only) */
            __builtin_amdgcn_s_waitcnt(0xc07f); /* lgkmcnt(0) */
            if ((tid & 63) == 0) {
                int prev = __hip_atomic_fetch_add(&flags[4 + slot], 1,
                                                  __ATOMIC_RELAXED,
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

    /* ---- consumer: hash pair-lanes ---- */
    const int ln = (wid - NPROD) * 64 + (tid & 63); /* 0..127 */
    const int cp = ln >> 1;
    const int h = ln & 1;
    const int cg = cp / TOT;
    const int cs = cp % TOT;
    const bool act = (ln < 2 * TOT * G) && (b0 + cg < a.n);

    HH2 s;
    {
        const uint64_t init0[4] = {0xdbe6d5d5fe4cce2full, 0xa4093822299f31d0ull,
                                   0x13198a2e03707344ull, 0x243f6a8885a308d3ull};
        const uint64_t init1[4] = {0x3bd39e10cb0ef593ull, 0xc0acf169b5f18a8cull,
                                   0xbe5466cf34e90c6cull, 0x452821e638d01377ull};
#pragma unroll
        for (int j = 0; j < 2; j++) {
            int li = 2 * h + j;
            s.mul0[j] = init0[li];
            s.mul1[j] = init1[li];
            s.v0[j] = init0[li] ^ a.key[li];
            s.v1[j] = init1[li] ^ ((a.key[li] >> 32) | (a.key[li] << 32));
        }
    }

    for (int64_t it = 0; it < n_iter; it++) {
        const int slot = (int)(it & 1);
        if (lds_poll(&flags[slot], (int)it + 1)) return; /* timeout */
        /* acquire pairs with the producer's release */
        __hip_atomic_load(&flags[slot], __ATOMIC_ACQUIRE,
                          __HIP_MEMORY_SCOPE_WORKGROUP);
        if (act) {
            /* chains are the latency wall: outprioritize the producer
             * waves sharing this SIMD (T5) */
            __builtin_amdgcn_s_setprio(3);
            const uint8_t *row =
                &lds[slot * SLOT + (cg * TOT + cs) * ROW + 16 * h];
#pragma unroll 4
            for (int t = 0; t < TILE / 32; t++) {
                uint4 q = *(const uint4 *)(row + 32 * t);
                hh2_update(s, (uint64_t)q.x | ((uint64_t)q.y << 32),
                           (uint64_t)q.z | ((uint64_t)q.w << 32));
            }
            __builtin_amdgcn_s_setprio(0);
        }
        __builtin_amdgcn_s_waitcnt(0xc07f); /* lgkmcnt(0): ds_reads done */
        if ((tid & 63) == 0) {
            int prev = __hip_atomic_fetch_add(&flags[6 + slot], 1,
                                              __ATOMIC_RELAXED,
                                              __HIP_MEMORY_SCOPE_WORKGROUP);
            if (prev == 1) {
                __hip_atomic_store(&flags[6 + slot], 0, __ATOMIC_RELAXED,
                                   __HIP_MEMORY_SCOPE_WORKGROUP);
                __hip_atomic_store(&flags[2 + slot], (int)it + 2,
                                   __ATOMIC_RELEASE,
                                   __HIP_MEMORY_SCOPE_WORKGROUP);
            }
        }
    }

    /* finalize + store sums */
    if (ln < 2 * TOT * G) {
#pragma unroll 1
        for (int r = 0; r < 10; r++) {
            uint64_t p0 = shfl_x(s.v0[0], 1);
            uint64_t p1 = shfl_x(s.v0[1], 1);
            hh2_update(s, (p0 >> 32) | (p0 << 32), (p1 >> 32) | (p1 << 32));
        }
        if (act) {
            uint64_t a2 = s.v1[0] + s.mul1[0];
            uint64_t a3 = (s.v1[1] + s.mul1[1]) & 0x3fffffffffffffffull;
            uint64_t o0 = (s.v0[0] + s.mul0[0]) ^ (a2 << 1) ^ (a2 << 2);
            uint64_t o1 = (s.v0[1] + s.mul0[1]) ^ ((a3 << 1) | (a2 >> 63)) ^
                          ((a3 << 2) | (a2 >> 62));
            uint4 out;
            out.x = (uint32_t)o0;
            out.y = (uint32_t)(o0 >> 32);
            out.z = (uint32_t)o1;
            out.w = (uint32_t)(o1 >> 32);
            *(uint4 *)(a.sums + ((b0 + cg) * TOT + cs) * 32 + 16 * h) = out;
        }
    }
}

extern "C" hipError_t mec_launch_fused2_encode_hh(int d, int p,
                                                  const FusedArgs *args,
                                                  hipStream_t stream) {
    static const char *env = getenv("MEC_FUSED2");
    static const bool enabled = env && atoi(env) != 0; /* opt-in for now */
    if (!enabled) return hipErrorNotSupported;
    if (args->shard_len % 1024 != 0) return hipErrorNotSupported;
    dim3 blk(512);
#define X(D, P)                                                              \
    if (d == D && p == P) {                                                  \
        constexpr int G = 128 / (2 * (D + P));                               \
        if (G < 1) return hipErrorNotSupported;                              \
        dim3 grid((uint32_t)((args->n + G - 1) / G));                        \
        hipLaunchKernelGGL((fused2_encode_hh_kernel<D, P, MAT_##D##_##P>),   \
                           grid, blk, 0, stream, *args);                     \
        return hipGetLastError();                                            \
    }
    MEC_SPECIALIZED_GEOS(X)
#undef X
    return hipErrorNotSupported;
}
