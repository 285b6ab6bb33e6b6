/* Single-pass fused erasure-encode + HighwayHash-256 kernel.
 *
 * Replaces, in one kernel, the hot loop of Erasure.Encode
 * (cmd/erasure-encode.go:76-108): per block, GF(2^8) parity
 * (reedsolomon.Encoder.Encode, cmd/erasure-coding.go:85) AND the per-shard
 * HighwayHash-256 of streamingBitrotWriter.Write
 * (cmd/bitrot-streaming.go:57-59) — data crosses HBM exactly once:
 * read blockLen, write p parity shards, write (d+p) 32-B sums
 * (1.5 B moved per input byte at EC8+4 vs 3.0 for the two-kernel pair).
 *
 * Structure (per 256-thread workgroup, G = 256/(2*(d+p)) blocks):
 *   loop over 256-B tiles of the shard:
 *     A) staged registers -> LDS (data tile, G*d rows), then ISSUE the next
 *        tile's global loads so they fly during B+C (register staging
 *        pipeline — plain-load latency would otherwise stall every tile)
 *     B) all 256 lanes: GF parity for the tile (constexpr matrix ladder,
 *        same schedule as gf_encode_kernel), parity -> LDS + global
 *     C) the 2*(d+p)*G hash lanes advance their chains 8 packets from LDS
 * Every lane works in phases A+B; hashing stays spread over all blocks in
 * flight, so the serial chains never serialize behind one block (the trap
 * measured in the stream-split experiment).
 *
 * LDS rows are padded to 272 B so consecutive shard rows start 4 banks
 * apart (256-B rows would put every chain's ds_read_b128 on one bank pair).
 */
#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstdlib>

#include "kernels.h"
#include "ec_matrices_gen.h"

/* shared device helpers (defined in kernels.hip, usable here because both
 * TUs are compiled into one .so; redeclare as inline copies) */
namespace fused {

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

} // namespace fused

template <int D, int P, const uint8_t (&MAT)[P][D], int TILE>
__global__ void __launch_bounds__(256) fused_encode_hh_kernel(FusedArgs a) {
    using namespace fused;
    constexpr int TOT = D + P;
    constexpr int G = 256 / (2 * TOT); /* blocks per workgroup */
    constexpr int ROW = TILE + 16;     /* tile row + 16-B bank skew */
    constexpr int TC = TILE / 16;      /* 16-B slots per tile row */
    constexpr int NL = (G * D * TC + 255) / 256; /* staged loads per lane */
    __shared__ uint8_t lds[G * TOT * ROW];

    const int tid = threadIdx.x;
    const int64_t b0 = (int64_t)blockIdx.x * G;
    const int64_t S = a.shard_len;
    const int64_t stride = a.row_stride;
    const int64_t n_iter = (S + TILE - 1) / TILE;
    const int64_t full_pkts = S / 32;          /* whole 32-B packets */
    const int mod32 = (int)(S % 32);
    const int64_t tail_tile = mod32 ? (S - mod32) / TILE : -1;

    /* hash role */
    const int cp = tid >> 1;
    const int h = tid & 1;
    const int cg = cp / TOT;
    const int cs = cp % TOT;
    const bool chain_act = (tid < 2 * TOT * G) && (b0 + cg < a.n);

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

    /* register staging of data tiles; element e -> (block g, shard k,
     * 16-B slot o) */
    uint4 st[NL];
    auto load_tile = [&](int64_t tbase) {
#pragma unroll
        for (int l = 0; l < NL; l++) {
            const int e = tid + 256 * l;
            if (NL * 256 > G * D * TC && e >= G * D * TC) continue;
            const int g = e / (D * TC);
            const int r = e % (D * TC);
            const int k = r / TC;
            const int o = r % TC;
            const int64_t off = tbase + (int64_t)o * 16;
            if (b0 + g < a.n && off + 16 <= stride) {
                st[l] = *(const uint4 *)(a.data +
                                         ((b0 + g) * D + k) * stride + off);
            } else {
                st[l] = uint4{0, 0, 0, 0};
            }
        }
    };

    load_tile(0);
    for (int64_t it = 0; it < n_iter; it++) {
        const int64_t tbase = it * TILE;
        __syncthreads(); /* previous tile fully consumed */
        /* A: staged registers -> LDS */
#pragma unroll
        for (int l = 0; l < NL; l++) {
            const int e = tid + 256 * l;
            if (NL * 256 > G * D * TC && e >= G * D * TC) continue;
            const int g = e / (D * TC);
            const int r = e % (D * TC);
            const int k = r / TC;
            const int o = r % TC;
            *(uint4 *)&lds[(g * TOT + k) * ROW + o * 16] = st[l];
        }
        if (it + 1 < n_iter) load_tile(tbase + TILE); /* fly during B+C */
        __syncthreads(); /* data tile visible */

        /* B: GF parity for the tile.  One thread per 16-B column computes
         * ALL P parity rows (the doubling ladder is shared across rows and
         * every matrix index stays compile-time — a per-(row,column)
         * split would re-run the ladder P times and turn MAT[i][k] into a
         * runtime load + branch). */
        for (int col = tid; col < G * TC; col += 256) {
            const int g = col / TC;
            const int o = col % TC;
            if (b0 + g < a.n) {
                uint4 acc[P];
#pragma unroll
                for (int i = 0; i < P; i++) acc[i] = uint4{0, 0, 0, 0};
#pragma unroll
                for (int k = 0; k < D; k++) {
                    uint4 pw =
                        *(const uint4 *)&lds[(g * TOT + k) * ROW + o * 16];
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
                const int64_t off = tbase + (int64_t)o * 16;
#pragma unroll
                for (int i = 0; i < P; i++) {
                    *(uint4 *)&lds[(g * TOT + D + i) * ROW + o * 16] = acc[i];
                    if (off + 16 <= stride) {
                        typedef unsigned int v4u
                            __attribute__((ext_vector_type(4)));
                        v4u v = {acc[i].x, acc[i].y, acc[i].z, acc[i].w};
                        __builtin_nontemporal_store(
                            v, (v4u *)(a.parity +
                                       ((b0 + g) * P + i) * stride + off));
                    }
                }
            }
        }
        __syncthreads(); /* parity tile visible */

        /* C: hash lanes advance 8 packets from LDS */
        if (chain_act) {
            const uint8_t *row = &lds[(cg * TOT + cs) * ROW + 16 * h];
            int pk = (int)(full_pkts - tbase / 32);
            if (pk > TILE / 32) pk = TILE / 32;
            for (int t = 0; t < pk; t++) {
                uint4 q = *(const uint4 *)(row + 32 * t);
                hh2_update(s, (uint64_t)q.x | ((uint64_t)q.y << 32),
                           (uint64_t)q.z | ((uint64_t)q.w << 32));
            }
            if (it == tail_tile) {
                /* UpdateRemainder from the LDS tile bytes */
                const uint8_t *tail_msg =
                    &lds[(cg * TOT + cs) * ROW] + (int)(S - mod32 - tbase);
                const int mod4 = mod32 & 3;
#pragma unroll
                for (int j = 0; j < 2; j++) {
                    s.v0[j] += ((uint64_t)mod32 << 32) + (uint64_t)mod32;
                    uint32_t h0 = (uint32_t)s.v1[j];
                    uint32_t h1 = (uint32_t)(s.v1[j] >> 32);
                    s.v1[j] = (uint32_t)((h0 << mod32) | (h0 >> (32 - mod32)));
                    s.v1[j] |=
                        (uint64_t)((h1 << mod32) | (h1 >> (32 - mod32)))
                        << 32;
                }
                uint8_t packet[32];
#pragma unroll
                for (int i = 0; i < 32; i++) packet[i] = 0;
                for (int i = 0; i < (mod32 & ~3); i++) packet[i] = tail_msg[i];
                const uint8_t *rem = tail_msg + (mod32 & ~3);
                if (mod32 & 16) {
                    for (int i = 0; i < 4; i++)
                        packet[28 + i] = rem[i + mod4 - 4];
                } else if (mod4) {
                    packet[16] = rem[0];
                    packet[17] = rem[mod4 >> 1];
                    packet[18] = rem[mod4 - 1];
                }
                uint64_t w[2];
#pragma unroll
                for (int j = 0; j < 2; j++) {
                    uint64_t v = 0;
                    for (int bt = 7; bt >= 0; bt--)
                        v = (v << 8) | packet[16 * h + 8 * j + bt];
                    w[j] = v;
                }
                hh2_update(s, w[0], w[1]);
            }
        }
    }

    /* finalize + store sums */
    if (tid < 2 * TOT * G) {
#pragma unroll 1
        for (int r = 0; r < 10; r++) {
            uint64_t p0 = shfl_x(s.v0[0], 1);
            uint64_t p1 = shfl_x(s.v0[1], 1);
            hh2_update(s, (p0 >> 32) | (p0 << 32), (p1 >> 32) | (p1 << 32));
        }
        if (chain_act) {
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

extern "C" hipError_t mec_launch_fused_encode_hh(int d, int p,
                                                 const FusedArgs *args,
                                                 hipStream_t stream) {
    /* opt-in: the barrier-lockstep fused structure measured 2.3x slower
     * than the kernel pair at EC8+4/1MiB/1024 (GF phase at 1 wave/SIMD);
     * kept for experimentation, off by default */
    static const char *env = getenv("MEC_FUSED");
    static const bool enabled = env && atoi(env) != 0;
    static const char *envt = getenv("MEC_FUSED_TILE");
    static const int tile = envt ? atoi(envt) : 512;
    if (!enabled) return hipErrorNotSupported;
    dim3 blk(256);
#define X(D, P)                                                              \
    if (d == D && p == P) {                                                  \
        constexpr int G = 256 / (2 * (D + P));                               \
        dim3 grid((uint32_t)((args->n + G - 1) / G));                        \
        if (tile >= 512)                                                     \
            hipLaunchKernelGGL(                                              \
                (fused_encode_hh_kernel<D, P, MAT_##D##_##P, 512>), grid,    \
                blk, 0, stream, *args);                                      \
        else                                                                 \
            hipLaunchKernelGGL(                                              \
                (fused_encode_hh_kernel<D, P, MAT_##D##_##P, 256>), grid,    \
                blk, 0, stream, *args);                                      \
        return hipGetLastError();                                            \
    }
    MEC_SPECIALIZED_GEOS(X)
#undef X
    return hipErrorNotSupported;
}
