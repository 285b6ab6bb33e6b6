/* MI355X (gfx950, CDNA4) kernels for the MinIO erasure+bitrot hot path.
 *
 * Replaces the compute of:
 *  - reedsolomon.Encoder.Encode / Reconstruct (GF(2^8) shard arithmetic;
 *    reference call sites cmd/erasure-coding.go:85,106,112) — r2 default:
 *    BIT-SLICED kernels (gf_encode_bs_kernel constexpr networks,
 *    gf_matmul_bs_kernel runtime bit-matrices; gf_bs.h); the r1 SWAR
 *    xtime-ladder kernels remain behind MEC_GF_BS=0 / MEC_GFM_BS=0.
 *    No MFMA: this is byte/integer work.
 *  - streamingBitrotWriter's per-shard hash (cmd/bitrot-streaming.go:57-59)
 *    and BitrotAlgorithm.New digests (cmd/bitrot.go:47-64) — HighwayHash
 *    at 4 GPU lanes per chain (hh256_batch4_kernel, r2 default; r1
 *    pair-lane kernel behind MEC_HH4=0), SHA-256 (16-round body),
 *    BLAKE2b-512; hashing is sequential per shard, parallelism comes
 *    from shards x blocks x lanes-per-chain (SURVEY.md §7 hard part (b)).
 *
 * Shipped structure (r2): encode and hash run as two kernels back to
 * back — the bit-sliced GF encode at ~90% of the HBM floor and the
 * 4-lane HighwayHash at ~97% of its memory floor while simultaneously at
 * the per-wave VALU issue cadence (DESIGN.md §4,
 * profiles/probes_r2_valu.txt).  r1's cross-batch pipelining is kept
 * behind MEC_PIPE=1 (overlap stopped paying once both legs reached their
 * resource floors), and the single-pass fused variants (fused.hip
 * lockstep, fused2.hip LDS ring, fused3.hip L2 handoff) are in-tree,
 * bit-exact, measured, and opt-in; the C-ABI signature treats the pair
 * as one fused operation either way.
 */
#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstdlib>

#include "kernels.h"

/* ---- GF(2^8) SWAR helpers --------------------------------------------- */

/* gfx950 3-input boolean LUT op (1 full-rate VALU slot) */
__device__ __forceinline__ uint32_t xor3(uint32_t a, uint32_t b, uint32_t c) {
    return __builtin_amdgcn_bitop3_b32(a, b, c, 0x96); /* a^b^c */
}
__device__ __forceinline__ uint32_t andxor(uint32_t a, uint32_t b,
                                           uint32_t c) {
    return __builtin_amdgcn_bitop3_b32(a, b, c, 0x6a); /* (a&b)^c */
}

__device__ __forceinline__ uint32_t gf2x(uint32_t x) {
    /* multiply 4 packed GF(2^8) bytes by 2 (poly 0x11D) in 5 full-rate
     * slots: the reduction byte (0x00 or 0x1d per sign bit) comes from a
     * v_perm_b32 byte-select instead of a multiply, and the shifted-out
     * cross-byte bits are cleared by the same bitop3 that applies the
     * reduction ((a&b)^c). */
    uint32_t sel = (x & 0x80808080u) >> 7;
    uint32_t m = __builtin_amdgcn_perm(0u, 0x00001d00u, sel);
    return andxor(x << 1, 0xfefefefeu, m);
}

__device__ __forceinline__ void gf2x4(uint4 &v) {
    v.x = gf2x(v.x);
    v.y = gf2x(v.y);
    v.z = gf2x(v.z);
    v.w = gf2x(v.w);
}

__device__ __forceinline__ void xor4(uint4 &a, const uint4 &b) {
    a.x ^= b.x;
    a.y ^= b.y;
    a.z ^= b.z;
    a.w ^= b.w;
}

__device__ __forceinline__ void xor34(uint4 &a, const uint4 &b,
                                      const uint4 &c) {
    a.x = xor3(a.x, b.x, c.x);
    a.y = xor3(a.y, b.y, c.y);
    a.z = xor3(a.z, b.z, c.z);
    a.w = xor3(a.w, b.w, c.w);
}

/* ---- specialized encode kernels (constexpr matrix) ---------------------
 *
 * For the common geometries the encode matrix is baked in at compile time
 * (ec_matrices_gen.h, generated from the product's own matrix builder), so
 * the inner loop is a straight-line XOR schedule: no per-bit branches (the
 * generic kernel's 60% front-end issue-stall), no coefficient loads, and
 * doubling steps stop at each column's highest set bit. */
#include "ec_matrices_gen.h"

/* W = uint4 columns per thread (16 or 32 B); NT = nontemporal parity
 * stores (parity is written once, never re-read by this kernel).
 * Variant selection via MEC_GF_W / MEC_GF_NT env (perf sweeps). */
template <int D, int P, const uint8_t (&MAT)[P][D], int W, bool NT,
          int LDSPAD = 0>
__global__ void __launch_bounds__(256) gf_encode_kernel(GfEncArgs a) {
    /* LDSPAD > 0 caps workgroups/CU (160 KiB / LDSPAD) so co-resident
     * hash waves of the pipelined previous batch keep issue share; the
     * pad must be touched or it is elided */
    if (LDSPAD > 0) {
        __shared__ uint8_t pad[LDSPAD > 0 ? LDSPAD : 1];
        if (threadIdx.x == 0xFFFFFFFF) pad[0] = 1; /* never true; keeps pad */
        (void)pad;
    }
    const int b = blockIdx.y;
    const int64_t cols = (a.shard_len + 16 * W - 1) / (16 * W);
    const uint8_t *__restrict__ sbase = a.data + (int64_t)b * D * a.row_stride;
    uint8_t *__restrict__ obase = a.parity + (int64_t)b * P * a.row_stride;

    for (int64_t c = blockIdx.x * blockDim.x + threadIdx.x; c < cols;
         c += (int64_t)gridDim.x * blockDim.x) {
        const int64_t j = c * (16 * W);
        uint4 acc[P][W];
#pragma unroll
        for (int i = 0; i < P; i++)
#pragma unroll
            for (int w = 0; w < W; w++) acc[i][w] = uint4{0, 0, 0, 0};
        /* issue ALL row loads before any ladder math: the consume-as-you-
         * load form left one load in flight at a time (ISA: LOAD WAIT
         * ladder LOAD WAIT ... — the 36% memory-wait in the r03 PMC) */
        uint4 pws[D][W];
#pragma unroll
        for (int k = 0; k < D; k++) {
            const uint8_t *row = sbase + (int64_t)k * a.row_stride + j;
#pragma unroll
            for (int w = 0; w < W; w++)
                pws[k][w] = *(const uint4 *)(row + 16 * w);
        }
#pragma unroll
        for (int k = 0; k < D; k++) {
            /* xtime chain walked two bits at a time so each parity's
             * accumulate can fold bit-pairs into one v_bitop3 xor3; all
             * bit tests are constexpr on MAT and fold at compile time */
            uint4 cur[W], nxt[W];
#pragma unroll
            for (int w = 0; w < W; w++) cur[w] = pws[k][w];
#pragma unroll
            for (int bit = 0; bit < 8; bit += 2) {
                uint32_t needCur = 0, needHi = 0;
#pragma unroll
                for (int i = 0; i < P; i++) {
                    needCur |= (uint32_t)MAT[i][k] >> bit;
                    needHi |= (uint32_t)MAT[i][k] >> (bit + 1);
                }
                if (!needCur) break; /* compile-time folded */
                if (needHi)
#pragma unroll
                    for (int w = 0; w < W; w++) {
                        nxt[w] = cur[w];
                        gf2x4(nxt[w]);
                    }
#pragma unroll
                for (int i = 0; i < P; i++) {
                    const int b0 = (MAT[i][k] >> bit) & 1;
                    const int b1 = (MAT[i][k] >> (bit + 1)) & 1;
                    if (b0 && b1)
#pragma unroll
                        for (int w = 0; w < W; w++)
                            xor34(acc[i][w], cur[w], nxt[w]);
                    else if (b0)
#pragma unroll
                        for (int w = 0; w < W; w++) xor4(acc[i][w], cur[w]);
                    else if (b1)
#pragma unroll
                        for (int w = 0; w < W; w++) xor4(acc[i][w], nxt[w]);
                }
                if (needHi >> 1)
#pragma unroll
                    for (int w = 0; w < W; w++) {
                        cur[w] = nxt[w];
                        gf2x4(cur[w]);
                    }
                else
                    break; /* compile-time folded */
            }
        }
#pragma unroll
        for (int i = 0; i < P; i++) {
            uint8_t *orow = obase + (int64_t)i * a.row_stride + j;
#pragma unroll
            for (int w = 0; w < W; w++) {
                if (NT) {
                    typedef unsigned int v4u __attribute__((ext_vector_type(4)));
                    v4u v = {acc[i][w].x, acc[i][w].y, acc[i][w].z,
                             acc[i][w].w};
                    __builtin_nontemporal_store(v, (v4u *)(orow + 16 * w));
                } else {
                    *(uint4 *)(orow + 16 * w) = acc[i][w];
                }
            }
        }
    }
}

#include "gf_bs.h"

/* PF: rows of load prefetch ahead of the transpose+fold (1 = r2 default;
 * 2 = deeper cover now that bit-slicing cut the compute between loads) */
template <int D, int P, const uint8_t (&MAT)[P][D], bool NT, int PF = 1>
__global__ void __launch_bounds__(256) gf_encode_bs_kernel(GfEncArgs a) {
    const int b = blockIdx.y;
    const int64_t cols = (a.shard_len + 31) / 32;
    const uint8_t *__restrict__ sbase = a.data + (int64_t)b * D * a.row_stride;
    uint8_t *__restrict__ obase = a.parity + (int64_t)b * P * a.row_stride;

    for (int64_t c = blockIdx.x * blockDim.x + threadIdx.x; c < cols;
         c += (int64_t)gridDim.x * blockDim.x) {
        const int64_t j = c * 32;
        uint32_t accp[P][8];
#pragma unroll
        for (int i = 0; i < P; i++)
#pragma unroll
            for (int pb = 0; pb < 8; pb++) accp[i][pb] = 0;
        /* software pipeline: the next PF rows' loads in flight during
         * this row's transpose + plane xors */
        uint32_t xq[PF + 1][8];
#pragma unroll
        for (int pf = 0; pf < PF && pf < D; pf++) {
            const uint8_t *row = sbase + (int64_t)pf * a.row_stride + j;
            uint4 lo = *(const uint4 *)row;
            uint4 hi = *(const uint4 *)(row + 16);
            xq[pf][0] = lo.x; xq[pf][1] = lo.y;
            xq[pf][2] = lo.z; xq[pf][3] = lo.w;
            xq[pf][4] = hi.x; xq[pf][5] = hi.y;
            xq[pf][6] = hi.z; xq[pf][7] = hi.w;
        }
#pragma unroll
        for (int k = 0; k < D; k++) {
            const int cur = k % (PF + 1);
            if (k + PF < D) {
                const int nxt = (k + PF) % (PF + 1);
                const uint8_t *row =
                    sbase + (int64_t)(k + PF) * a.row_stride + j;
                uint4 lo = *(const uint4 *)row;
                uint4 hi = *(const uint4 *)(row + 16);
                xq[nxt][0] = lo.x; xq[nxt][1] = lo.y;
                xq[nxt][2] = lo.z; xq[nxt][3] = lo.w;
                xq[nxt][4] = hi.x; xq[nxt][5] = hi.y;
                xq[nxt][6] = hi.z; xq[nxt][7] = hi.w;
            }
            bs_transpose(xq[cur]);
            bs_acc_k<D, P, MAT>(k, xq[cur], accp);
        }
#pragma unroll
        for (int i = 0; i < P; i++) {
            bs_transpose(accp[i]);
            uint8_t *orow = obase + (int64_t)i * a.row_stride + j;
            uint4 lo{accp[i][0], accp[i][1], accp[i][2], accp[i][3]};
            uint4 hi{accp[i][4], accp[i][5], accp[i][6], accp[i][7]};
            if (NT) {
                typedef unsigned int v4u __attribute__((ext_vector_type(4)));
                v4u vlo = {lo.x, lo.y, lo.z, lo.w};
                v4u vhi = {hi.x, hi.y, hi.z, hi.w};
                __builtin_nontemporal_store(vlo, (v4u *)orow);
                __builtin_nontemporal_store(vhi, (v4u *)(orow + 16));
            } else {
                *(uint4 *)orow = lo;
                *(uint4 *)(orow + 16) = hi;
            }
        }
    }
}

/* ---- generic GF matrix-multiply over shard rows ------------------------
 *
 * out[t][j] = sum_k mat[t][k] * src[k][j]  (GF(2^8)), per batch item.
 * Works for encode (src = data rows, out = parity rows) and reconstruct
 * (src = surviving rows, out = missing rows) via row index lists.
 * E = number of output rows (template so accumulators stay in registers).
 */
template <int E, bool MASKED>
__global__ void __launch_bounds__(256) gf_matmul_kernel(GfMatmulArgs a) {
    const int b = blockIdx.y; /* batch item */
    const int64_t cols = (a.shard_len + 15) >> 4;

    const uint8_t *__restrict__ sbase = a.src + (int64_t)b * a.src_item_stride;
    uint8_t *__restrict__ obase = a.dst + (int64_t)b * a.dst_item_stride;

    for (int64_t c = blockIdx.x * blockDim.x + threadIdx.x; c < cols;
         c += (int64_t)gridDim.x * blockDim.x) {
        const int64_t j = c << 4;
        uint4 acc[E];
#pragma unroll
        for (int i = 0; i < E; i++) acc[i] = uint4{0, 0, 0, 0};

        for (int k = 0; k < a.d; k++) {
            uint4 pw = *(const uint4 *)(sbase +
                                        (int64_t)a.src_rows[k] * a.row_stride +
                                        j);
            uint32_t cb[E];
#pragma unroll
            for (int i = 0; i < E; i++) cb[i] = a.mat[i * MEC_KMAX_D + k];
            uint32_t any = 0;
#pragma unroll
            for (int i = 0; i < E; i++) any |= cb[i];
#pragma unroll
            for (int bit = 0; bit < 8; bit++) {
                if (!(any >> bit)) break; /* uniform: no higher bits set */
                if (bit) gf2x4(pw);
                if (MASKED) {
                    /* branch-free: broadcast scalar mask per (row,bit) */
#pragma unroll
                    for (int i = 0; i < E; i++) {
                        uint32_t m = 0u - ((cb[i] >> bit) & 1u);
                        acc[i].x ^= pw.x & m;
                        acc[i].y ^= pw.y & m;
                        acc[i].z ^= pw.z & m;
                        acc[i].w ^= pw.w & m;
                    }
                } else {
                    /* wave-uniform coefficient bits -> scalar branches */
#pragma unroll
                    for (int i = 0; i < E; i++)
                        if (cb[i] & (1u << bit)) xor4(acc[i], pw);
                }
            }
        }
#pragma unroll
        for (int i = 0; i < E; i++)
            *(uint4 *)(obase + (int64_t)a.dst_rows[i] * a.row_stride + j) =
                acc[i];
    }
}

/* ---- bit-sliced generic matmul (reconstruct; runtime matrices) ---------
 *
 * Same plane-transpose trick as gf_encode_bs_kernel, but the decode
 * matrix is only known at run time, so the per-(output,plane) XOR
 * selection uses 0/~0 masks ((x & m) ^ acc as one v_bitop3) instead of
 * compile-time folds: 64 VALU per (src,dst) pair per 32 B regardless of
 * matrix density, vs the ladder's ~90-160.  Masks are built host-side
 * (mec::bs_build_masks) and read through L1 (8 dwords per (t,k), hot
 * after the first column). */
template <int E, int BRANCHY = 0>
__global__ void __launch_bounds__(256) gf_matmul_bs_kernel(GfMatmulArgs a) {
    const int b = blockIdx.y;
    const int64_t cols = (a.shard_len + 31) / 32;
    const uint8_t *__restrict__ sbase = a.src + (int64_t)b * a.src_item_stride;
    uint8_t *__restrict__ obase = a.dst + (int64_t)b * a.dst_item_stride;

    for (int64_t c = blockIdx.x * blockDim.x + threadIdx.x; c < cols;
         c += (int64_t)gridDim.x * blockDim.x) {
        const int64_t j = c * 32;
        uint32_t accp[E][8];
#pragma unroll
        for (int i = 0; i < E; i++)
#pragma unroll
            for (int pb = 0; pb < 8; pb++) accp[i][pb] = 0;
        uint32_t xc[8], xn[8];
        {
            const uint8_t *row = sbase + (int64_t)a.src_rows[0] * a.row_stride + j;
            uint4 lo = *(const uint4 *)row;
            uint4 hi = *(const uint4 *)(row + 16);
            xc[0] = lo.x; xc[1] = lo.y; xc[2] = lo.z; xc[3] = lo.w;
            xc[4] = hi.x; xc[5] = hi.y; xc[6] = hi.z; xc[7] = hi.w;
        }
        for (int k = 0; k < a.d; k++) {
            if (k + 1 < a.d) {
                const uint8_t *row =
                    sbase + (int64_t)a.src_rows[k + 1] * a.row_stride + j;
                uint4 lo = *(const uint4 *)row;
                uint4 hi = *(const uint4 *)(row + 16);
                xn[0] = lo.x; xn[1] = lo.y; xn[2] = lo.z; xn[3] = lo.w;
                xn[4] = hi.x; xn[5] = hi.y; xn[6] = hi.z; xn[7] = hi.w;
            }
            bs_transpose(xc);
#pragma unroll
            for (int i = 0; i < E; i++) {
                /* the 8x8 bit matrix of mat[i][k], 2 dwords; loaded once
                 * per (i,k) and moved to SGPRs so the per-element mask
                 * expansion runs on the SCALAR pipe concurrently with the
                 * VALU — leaving exactly ONE v_bitop3 per matrix element.
                 * BRANCHY=1 instead skips zero bits with wave-uniform
                 * scalar branches (~half the VALU at ~density 0.5, paid
                 * in s_cbranch overhead — measured) */
                const uint32_t *m = a.bs_masks + ((int64_t)i * a.d + k) * 2;
                const uint32_t mlo = __builtin_amdgcn_readfirstlane(m[0]);
                const uint32_t mhi = __builtin_amdgcn_readfirstlane(m[1]);
#pragma unroll
                for (int pb = 0; pb < 8; pb++) {
                    uint32_t acc = accp[i][pb];
                    const uint32_t mk =
                        (pb < 4) ? (mlo >> (8 * pb)) : (mhi >> (8 * (pb - 4)));
                    if (BRANCHY) {
#pragma unroll
                        for (int pa = 0; pa < 8; pa++)
                            if ((mk >> pa) & 1) acc ^= xc[pa];
                    } else {
#pragma unroll
                        for (int pa = 0; pa < 8; pa++) {
                            const uint32_t mm = 0u - ((mk >> pa) & 1u);
                            acc = (uint32_t)__builtin_amdgcn_bitop3_b32(
                                xc[pa], mm, acc, 0x6a); /* (x & m) ^ acc */
                        }
                    }
                    accp[i][pb] = acc;
                }
            }
#pragma unroll
            for (int w = 0; w < 8; w++) xc[w] = xn[w];
        }
#pragma unroll
        for (int i = 0; i < E; i++) {
            bs_transpose(accp[i]);
            uint8_t *orow = obase + (int64_t)a.dst_rows[i] * a.row_stride + j;
            /* nontemporal: rebuilt rows are written once and never
             * re-read by this kernel (reads and writes hit the SAME
             * in-place buffer — keeping stores out of the cache reduces
             * the read/write turnaround that parks 42% of wave-cycles,
             * gpurun decsq SQ pass) */
            typedef unsigned int v4u __attribute__((ext_vector_type(4)));
            v4u vlo = {accp[i][0], accp[i][1], accp[i][2], accp[i][3]};
            v4u vhi = {accp[i][4], accp[i][5], accp[i][6], accp[i][7]};
            __builtin_nontemporal_store(vlo, (v4u *)orow);
            __builtin_nontemporal_store(vhi, (v4u *)(orow + 16));
        }
    }
}

/* ---- HighwayHash-256 (one chain per lane) ------------------------------
 * Portable algorithm as published (minio/highwayhash v1.0.3 semantics,
 * magic key passed from host; pinned by tests/golden/bitrot_selftest.json).
 */

/* Map chain index -> (shard pointer, sum slot) for the fused encode
 * layout.  Sums always use the n x (d+p) layout regardless of mode, so
 * data-only + parity-only launches compose to the single-launch result. */
__device__ __forceinline__ const uint8_t *chain_ptr(const HashArgs &a,
                                                    int64_t chain,
                                                    int64_t &sum_idx) {
    if (a.parity == nullptr) {
        sum_idx = chain;
        return a.data + chain * a.row_stride;
    }
    const int total = a.d + a.p;
    if (a.mode == MEC_HASH_DATA) {
        const int64_t b = chain / a.d;
        const int s = (int)(chain % a.d);
        sum_idx = b * total + s;
        return a.data + (b * a.d + s) * a.row_stride;
    }
    if (a.mode == MEC_HASH_PARITY) {
        const int64_t b = chain / a.p;
        const int s = (int)(chain % a.p);
        sum_idx = b * total + a.d + s;
        return a.parity + (b * a.p + s) * a.row_stride;
    }
    const int64_t b = chain / total;
    const int s = (int)(chain % total);
    sum_idx = chain;
    if (s < a.d)
        return a.data + (b * a.d + s) * a.row_stride;
    return a.parity + (b * a.p + (s - a.d)) * a.row_stride;
}

/* The zipper merge is a pure byte permutation of the pair (the reference
 * implements it with pshufb); on CDNA4 that is v_perm_b32: 3 perms + 1 or
 * per output word instead of a ~17-op shift/mask tree.
 *   zip_even out bytes (LSB..MSB): [A3 B4 A2 A5 | B6 A1 B7 A0]
 *   zip_odd  out bytes:            [B3 A4 B2 B5 | B1 A6 B0 A7]
 * where A = v1[even], B = v1[odd] of the pair.  v_perm pool = {hi:lo},
 * selector byte 12 yields 0x00. */
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
    /* this GPU lane's zipper PAIR of HighwayHash lanes: even lane owns HH
     * lanes {0,1}, odd lane owns {2,3}.  The zipper merge operates on
     * exactly these pairs, so the packet loop is fully lane-local — no
     * cross-lane ops on the serial dependency chain. */
    uint64_t v0[2], v1[2], mul0[2], mul1[2];
};

__device__ __forceinline__ uint64_t shfl_x(uint64_t v, int mask) {
    return __shfl_xor((unsigned long long)v, mask, 64);
}

__device__ __forceinline__ void hh2_update(HH2 &s, uint64_t w0, uint64_t w1) {
    uint64_t w[2] = {w0, w1};
#pragma unroll
    for (int j = 0; j < 2; j++) {
        s.v1[j] += s.mul0[j] + w[j];
        s.mul0[j] ^= (s.v1[j] & 0xffffffffull) * (s.v0[j] >> 32);
        s.v0[j] += s.mul1[j];
        s.mul1[j] ^= (s.v0[j] & 0xffffffffull) * (s.v1[j] >> 32);
    }
    /* ZipperMergeAndAdd(v1[hi], v1[lo], &v0[hi], &v0[lo]) — pair-local */
    uint64_t t0 = zip_even(s.v1[0], s.v1[1]);
    uint64_t t1 = zip_odd(s.v1[0], s.v1[1]);
    s.v0[0] += t0;
    s.v0[1] += t1;
    uint64_t u0 = zip_even(s.v0[0], s.v0[1]);
    uint64_t u1 = zip_odd(s.v0[0], s.v0[1]);
    s.v1[0] += u0;
    s.v1[1] += u1;
}

/* NC = independent chains per lane (ILP): the hash chain is serial, so a
 * single chain per lane leaves every dependent-op latency exposed (measured
 * ~340 cyc/packet vs ~104 issue-bound).  Interleaving NC independent
 * chains' packet updates in one lane fills those stalls. */
template <int NC, bool RAGGED>
__global__ void __launch_bounds__(512) hh256_batch_kernel(HashArgs a) {
    const int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    const int64_t slot = tid >> 1;
    const int h = (int)(tid & 1); /* 0: HH lanes {0,1}; 1: HH lanes {2,3} */
    const int64_t c0 = slot * NC;
    if (c0 >= a.n_chains) return;

    bool act[NC];
    const uint8_t *mp[NC];
    int64_t sum_idx[NC];
#pragma unroll
    for (int u = 0; u < NC; u++) {
        act[u] = c0 + u < a.n_chains;
        mp[u] = chain_ptr(a, act[u] ? c0 + u : c0, sum_idx[u]) + 16 * h;
    }

    const uint64_t init0[4] = {0xdbe6d5d5fe4cce2full, 0xa4093822299f31d0ull,
                               0x13198a2e03707344ull, 0x243f6a8885a308d3ull};
    const uint64_t init1[4] = {0x3bd39e10cb0ef593ull, 0xc0acf169b5f18a8cull,
                               0xbe5466cf34e90c6cull, 0x452821e638d01377ull};
    HH2 s[NC];
#pragma unroll
    for (int u = 0; u < NC; u++)
#pragma unroll
        for (int j = 0; j < 2; j++) {
            int li = 2 * h + j;
            s[u].mul0[j] = init0[li];
            s[u].mul1[j] = init1[li];
            s[u].v0[j] = init0[li] ^ a.key[li];
            s[u].v1[j] = init1[li] ^ ((a.key[li] >> 32) | (a.key[li] << 32));
        }

    int64_t len = a.msg_len;
    /* double-buffered packet prefetch: compute tile A while tile B's loads
     * are in flight (a single-buffer loop leaves ~36% of wave-cycles in
     * memory waits — r03 PMC).  Buffers alternate by explicit two-phase
     * unroll: a runtime-selected array reference would spill to scratch. */
    constexpr int DP = (NC == 1) ? 16 : 8; /* prefetch depth (packets) */
    uint4 qa[NC][DP], qb[NC][DP];
#define HH_LOAD(Q)                                                           \
    {                                                                        \
        _Pragma("unroll") for (int t = 0; t < DP; t++)                       \
            _Pragma("unroll") for (int u = 0; u < NC; u++)                   \
                Q[u][t] = *(const uint4 *)(mp[u] + 32 * t);                  \
        _Pragma("unroll") for (int u = 0; u < NC; u++) mp[u] += 32 * DP;     \
    }
#define HH_COMP(Q)                                                           \
    {                                                                        \
        _Pragma("unroll") for (int t = 0; t < DP; t++)                       \
            _Pragma("unroll") for (int u = 0; u < NC; u++)                   \
                hh2_update(s[u],                                             \
                           (uint64_t)Q[u][t].x | ((uint64_t)Q[u][t].y << 32),\
                           (uint64_t)Q[u][t].z | ((uint64_t)Q[u][t].w << 32));\
    }
    /* raise wave priority: when a GF kernel shares the CU (pipelined
     * steps), the latency-critical chains must win issue slots (T5) */
    __builtin_amdgcn_s_setprio(1);
    if (len >= 32 * DP) {
        HH_LOAD(qa)
        len -= 32 * DP;
        while (len >= 2 * 32 * DP) {
            HH_LOAD(qb)
            HH_COMP(qa)
            HH_LOAD(qa)
            HH_COMP(qb)
            len -= 2 * 32 * DP;
        }
        if (len >= 32 * DP) {
            HH_LOAD(qb)
            HH_COMP(qa)
            HH_COMP(qb)
            len -= 32 * DP;
        } else {
            HH_COMP(qa)
        }
    }
#undef HH_LOAD
#undef HH_COMP
    __builtin_amdgcn_s_setprio(0);
    while (len >= 32) {
#pragma unroll
        for (int u = 0; u < NC; u++) {
            uint4 q = *(const uint4 *)mp[u];
            hh2_update(s[u], (uint64_t)q.x | ((uint64_t)q.y << 32),
                       (uint64_t)q.z | ((uint64_t)q.w << 32));
            mp[u] += 32;
        }
        len -= 32;
    }
    if (RAGGED && len > 0) {
        /* UpdateRemainder (published portable semantics); tail-only cost.
         * Compiled out for 32-aligned launches: the byte-wise packet
         * builder costs ~900 B of SGPR spill when duplicated per chain
         * (the NC=2 regression), so aligned lengths get the lean kernel. */
        const int mod32 = (int)len;
        const int mod4 = mod32 & 3;
#pragma unroll
        for (int u = 0; u < NC; u++) {
            const uint8_t *tail_msg = mp[u] - 16 * h; /* row + body offset */
#pragma unroll
            for (int j = 0; j < 2; j++) {
                s[u].v0[j] += ((uint64_t)mod32 << 32) + (uint64_t)mod32;
                uint32_t h0 = (uint32_t)s[u].v1[j];
                uint32_t h1 = (uint32_t)(s[u].v1[j] >> 32);
                s[u].v1[j] =
                    (uint32_t)((h0 << mod32) | (h0 >> (32 - mod32)));
                s[u].v1[j] |=
                    (uint64_t)((h1 << mod32) | (h1 >> (32 - mod32))) << 32;
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
            hh2_update(s[u], w[0], w[1]);
        }
    }
    /* finalization: 10 permute-update rounds; permuted word for HH lane l
     * is rot32(v0[l ^ 2]) — the partner lane's same-position word. */
#pragma unroll 1
    for (int r = 0; r < 10; r++) {
#pragma unroll
        for (int u = 0; u < NC; u++) {
            uint64_t p0 = shfl_x(s[u].v0[0], 1);
            uint64_t p1 = shfl_x(s[u].v0[1], 1);
            hh2_update(s[u], (p0 >> 32) | (p0 << 32),
                       (p1 >> 32) | (p1 << 32));
        }
    }
    /* modular reduction — pair-local: even lane emits hash[0..1], odd lane
     * hash[2..3] */
#pragma unroll
    for (int u = 0; u < NC; u++) {
        if (!act[u]) continue;
        uint64_t a2 = s[u].v1[0] + s[u].mul1[0];
        uint64_t a3 = (s[u].v1[1] + s[u].mul1[1]) & 0x3fffffffffffffffull;
        uint64_t o0 = (s[u].v0[0] + s[u].mul0[0]) ^ (a2 << 1) ^ (a2 << 2);
        uint64_t o1 = (s[u].v0[1] + s[u].mul0[1]) ^
                      ((a3 << 1) | (a2 >> 63)) ^ ((a3 << 2) | (a2 >> 62));
        uint4 out;
        out.x = (uint32_t)o0;
        out.y = (uint32_t)(o0 >> 32);
        out.z = (uint32_t)o1;
        out.w = (uint32_t)(o1 >> 32);
        *(uint4 *)(a.sums + sum_idx[u] * 32 + 16 * h) = out;
    }
}

/* ---- HighwayHash-256, 4 lanes per chain (round-2 default candidate) ----
 *
 * Why 4 GPU lanes per chain instead of the r1 pair-lane split: at the
 * headline chain count (12288) the pair kernel is 384 waves — only 96 CUs
 * at WG 256 — and r1's ISA shows ~69 VALU/packet per lane (64-bit adds,
 * zipper perms, word re-packing), so each wave alone on its SIMD pays
 * dependency latency (~265 cyc/packet measured vs 138 issue floor) while
 * only ~96 CUs' worth of load bandwidth is engaged (the WG-512 "2
 * waves/SIMD" variant measured WORSE because it halved the engaged CUs —
 * per-CU global-load bandwidth ~10 B/cyc is the real wall).  Splitting
 * each chain over 4 lanes (one HighwayHash lane each):
 *  - doubles the wave count (768 waves -> ~192 CUs engaged),
 *  - cuts per-lane work to ~27 VALU/packet (zipper needs only the
 *    partner's v1/v0 HIGH word: one quad-perm DPP mov per phase, and the
 *    even/odd zipper variants collapse to one code path with a per-lane
 *    v_perm selector),
 *  - loads 8 B/lane/packet (dwordx2), same coalescing (a chain's 4 lanes
 *    cover its 32-B packet contiguously).
 */
__device__ __forceinline__ uint32_t dpp_swap1(uint32_t v) {
    /* value from lane ^ 1 (zipper pair partner); quad_perm [1,0,3,2] */
    return (uint32_t)__builtin_amdgcn_mov_dpp((int)v, 0xB1, 0xF, 0xF, true);
}
__device__ __forceinline__ uint32_t dpp_swap2(uint32_t v) {
    /* value from lane ^ 2 (finalization permute); quad_perm [2,3,0,1] */
    return (uint32_t)__builtin_amdgcn_mov_dpp((int)v, 0x4E, 0xF, 0xF, true);
}

struct HH1 {
    uint64_t v0, v1, mul0, mul1; /* this lane's single HighwayHash lane */
};

/* One packet update for HH lane j (lane parity selects the zipper hi
 * selector; S1/S2 are parity-independent — see zip_even/zip_odd above). */
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

template <bool RAGGED, int WG = 256>
__global__ void __launch_bounds__(WG) hh256_batch4_kernel(HashArgs a) {
    const int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    const int64_t chain = tid >> 2;
    const int j = (int)(tid & 3); /* this lane's HighwayHash lane index */
    if (chain >= a.n_chains) return;
    int64_t sum_idx;
    const uint8_t *mp = chain_ptr(a, chain, sum_idx) + 8 * j;
    const uint32_t S3 = (j & 1) ? 0x07000601u : 0x00070106u;

    const uint64_t init0[4] = {0xdbe6d5d5fe4cce2full, 0xa4093822299f31d0ull,
                               0x13198a2e03707344ull, 0x243f6a8885a308d3ull};
    const uint64_t init1[4] = {0x3bd39e10cb0ef593ull, 0xc0acf169b5f18a8cull,
                               0xbe5466cf34e90c6cull, 0x452821e638d01377ull};
    HH1 s;
    s.mul0 = init0[j];
    s.mul1 = init1[j];
    s.v0 = init0[j] ^ a.key[j];
    s.v1 = init1[j] ^ ((a.key[j] >> 32) | (a.key[j] << 32));

    int64_t len = a.msg_len;
    constexpr int DP = 16; /* prefetch depth (packets); 8 B each */
    uint64_t qa[DP], qb[DP];
#define HH4_LOAD(Q)                                                          \
    {                                                                        \
        _Pragma("unroll") for (int t = 0; t < DP; t++)                       \
            Q[t] = *(const uint64_t *)(mp + 32 * t);                         \
        mp += 32 * DP;                                                       \
    }
#define HH4_COMP(Q)                                                          \
    {                                                                        \
        _Pragma("unroll") for (int t = 0; t < DP; t++)                       \
            hh1_update(s, Q[t], S3);                                         \
    }
    __builtin_amdgcn_s_setprio(1);
    if (len >= 32 * DP) {
        HH4_LOAD(qa)
        len -= 32 * DP;
        while (len >= 2 * 32 * DP) {
            HH4_LOAD(qb)
            HH4_COMP(qa)
            HH4_LOAD(qa)
            HH4_COMP(qb)
            len -= 2 * 32 * DP;
        }
        if (len >= 32 * DP) {
            HH4_LOAD(qb)
            HH4_COMP(qa)
            HH4_COMP(qb)
            len -= 32 * DP;
        } else {
            HH4_COMP(qa)
        }
    }
#undef HH4_LOAD
#undef HH4_COMP
    __builtin_amdgcn_s_setprio(0);
    while (len >= 32) {
        uint64_t w = *(const uint64_t *)mp;
        hh1_update(s, w, S3);
        mp += 32;
        len -= 32;
    }
    if (RAGGED && len > 0) {
        /* UpdateRemainder (same semantics as hh256_batch_kernel's tail);
         * each lane builds the padded 32-B packet privately and consumes
         * its own 8-B word — tail-only cost */
        const int mod32 = (int)len;
        const int mod4 = mod32 & 3;
        const uint8_t *tail_msg = mp - 8 * j;
        s.v0 += ((uint64_t)mod32 << 32) + (uint64_t)mod32;
        {
            uint32_t h0 = (uint32_t)s.v1;
            uint32_t h1 = (uint32_t)(s.v1 >> 32);
            s.v1 = (uint32_t)((h0 << mod32) | (h0 >> (32 - mod32)));
            s.v1 |= (uint64_t)((h1 << mod32) | (h1 >> (32 - mod32))) << 32;
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
        uint64_t w = 0;
        for (int bt = 7; bt >= 0; bt--) w = (w << 8) | packet[8 * j + bt];
        hh1_update(s, w, S3);
    }
    /* 10 permute-update rounds: permuted[j] = rot32(v0[j ^ 2]) — rot32 of
     * the cross-pair partner is just its halves swapped */
#pragma unroll 1
    for (int r = 0; r < 10; r++) {
        uint32_t p_lo = dpp_swap2((uint32_t)s.v0);
        uint32_t p_hi = dpp_swap2((uint32_t)(s.v0 >> 32));
        hh1_update(s, ((uint64_t)p_lo << 32) | p_hi, S3);
    }
    /* modular reduction: lanes {0,1} emit hash[0..1], {2,3} hash[2..3].
     * m for the pair needs a2 = s0 = partner-even's (v1+mul1):
     *   even lane j: out = t_j ^ (s_j << 1) ^ (s_j << 2)
     *   odd lane j:  out = t_j ^ ((s_j' << 1) | (s_e >> 63))
     *                          ^ ((s_j' << 2) | (s_e >> 62)),
     *   s_j' = s_j & 0x3fff..., s_e = even partner's s. */
    {
        uint64_t t = s.v0 + s.mul0;
        uint64_t sv = s.v1 + s.mul1;
        uint32_t se_lo = dpp_swap1((uint32_t)sv);
        uint32_t se_hi = dpp_swap1((uint32_t)(sv >> 32));
        uint64_t se = ((uint64_t)se_hi << 32) | se_lo; /* partner's s */
        uint64_t out;
        if ((j & 1) == 0) {
            out = t ^ (sv << 1) ^ (sv << 2);
        } else {
            uint64_t a3 = sv & 0x3fffffffffffffffull;
            out = t ^ ((a3 << 1) | (se >> 63)) ^ ((a3 << 2) | (se >> 62));
        }
        *(uint64_t *)(a.sums + sum_idx * 32 + 8 * j) = out;
    }
}

/* ---- LDS-staged HighwayHash (MEC_HH_LDS=1) -----------------------------
 *
 * Same chain math as hh256_batch_kernel (pair-lanes, v_perm zipper), but
 * the packet stream is staged through LDS with direct global->LDS loads
 * (global_load_lds_dwordx4) instead of per-lane strided VGPR prefetch:
 *  - each wave stages ITS OWN 32 chains' next 512-B tiles (wave-private:
 *    no barriers, only counted vmcnt), so HBM sees 64 long coalesced
 *    512-B bursts per tile instead of 32-B scatters from 24k lanes;
 *  - the 128-KiB static LDS footprint forces 1 WG/CU, and leaves <32 KiB
 *    free, so a co-scheduled GF kernel built with the MEC_GF_CAP=4 pad
 *    (36 KiB) CANNOT share the CU: hash waves get exclusive SIMDs during
 *    pipelined encode without CU masks (which gfx950 ignores).
 */
__device__ __forceinline__ void hh_lds_stage16(const uint8_t *const *src,
                                               int64_t off, uint8_t *lds) {
#pragma unroll
    for (int k = 0; k < 16; k++)
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) uint32_t
                 *)(src[k] + off),
            (__attribute__((address_space(3))) uint32_t
                 *)(lds + (int64_t)k * 1024),
            16, 0, 0);
}

template <bool RAGGED>
__global__ void __launch_bounds__(256) hh256_lds_kernel(HashArgs a) {
    constexpr int T = 512;   /* bytes per chain per tile */
    constexpr int CPW = 32;  /* chains per wave (pair-lanes) */
    __shared__ uint8_t stage[4][2][CPW * T]; /* 128 KiB, wave-private */

    const int wv = (int)(threadIdx.x >> 6), lane = (int)(threadIdx.x & 63);
    const int64_t wave_c0 = ((int64_t)blockIdx.x * 4 + wv) * CPW;
    const int rc = lane >> 1;
    const int h = lane & 1;
    const int64_t chain = wave_c0 + rc;
    const bool act = chain < a.n_chains;
    int64_t sum_idx;
    const uint8_t *msg = chain_ptr(a, act ? chain : 0, sum_idx);

    /* per-lane glds sources: glds k stages chains 2k (lanes 0-31) and
     * 2k+1 (lanes 32-63), 16 B per lane = 1024 B contiguous LDS */
    const uint8_t *src[CPW / 2];
#pragma unroll
    for (int k = 0; k < CPW / 2; k++) {
        int64_t c = wave_c0 + 2 * k + (lane >> 5);
        if (c >= a.n_chains) c = 0; /* clamp: data unused */
        int64_t si;
        src[k] = chain_ptr(a, c, si) + (int64_t)(lane & 31) * 16;
    }

    const uint64_t init0[4] = {0xdbe6d5d5fe4cce2full, 0xa4093822299f31d0ull,
                               0x13198a2e03707344ull, 0x243f6a8885a308d3ull};
    const uint64_t init1[4] = {0x3bd39e10cb0ef593ull, 0xc0acf169b5f18a8cull,
                               0xbe5466cf34e90c6cull, 0x452821e638d01377ull};
    HH2 s;
#pragma unroll
    for (int j = 0; j < 2; j++) {
        int li = 2 * h + j;
        s.mul0[j] = init0[li];
        s.mul1[j] = init1[li];
        s.v0[j] = init0[li] ^ a.key[li];
        s.v1[j] = init1[li] ^ ((a.key[li] >> 32) | (a.key[li] << 32));
    }

    int64_t len = a.msg_len;
    const int64_t ntiles = len / T;
    __builtin_amdgcn_s_setprio(1);
    if (ntiles > 0) {
        hh_lds_stage16(src, 0, &stage[wv][0][0]);
        for (int64_t it = 0; it < ntiles; it++) {
            const int buf = (int)(it & 1);
            if (it + 1 < ntiles)
                hh_lds_stage16(src, (it + 1) * T, &stage[wv][buf ^ 1][0]);
            /* wait for THIS buffer's 16 glds (the next tile's 16 may stay
             * in flight): vmcnt(16), lgkm/exp unconstrained */
            if (it + 1 < ntiles)
                __builtin_amdgcn_s_waitcnt(0x4f70); /* vmcnt(16) */
            else
                __builtin_amdgcn_s_waitcnt(0x0f70); /* vmcnt(0) */
            const uint8_t *row = &stage[wv][buf][rc * T + 16 * h];
#pragma unroll
            for (int t = 0; t < T / 32; t++) {
                uint4 q = *(const uint4 *)(row + 32 * t);
                hh2_update(s, (uint64_t)q.x | ((uint64_t)q.y << 32),
                           (uint64_t)q.z | ((uint64_t)q.w << 32));
            }
        }
        len -= ntiles * T;
    }
    /* sub-tile remainder straight from global (0..511 B) */
    const uint8_t *mp = msg + ntiles * T + 16 * h;
    while (len >= 32) {
        uint4 q = *(const uint4 *)mp;
        hh2_update(s, (uint64_t)q.x | ((uint64_t)q.y << 32),
                   (uint64_t)q.z | ((uint64_t)q.w << 32));
        mp += 32;
        len -= 32;
    }
    __builtin_amdgcn_s_setprio(0);
    if (RAGGED && len > 0) {
        /* UpdateRemainder — identical to hh256_batch_kernel's tail */
        const int mod32 = (int)len;
        const int mod4 = mod32 & 3;
        const uint8_t *tail_msg = mp - 16 * h;
#pragma unroll
        for (int j = 0; j < 2; j++) {
            s.v0[j] += ((uint64_t)mod32 << 32) + (uint64_t)mod32;
            uint32_t h0 = (uint32_t)s.v1[j];
            uint32_t h1 = (uint32_t)(s.v1[j] >> 32);
            s.v1[j] = (uint32_t)((h0 << mod32) | (h0 >> (32 - mod32)));
            s.v1[j] |= (uint64_t)((h1 << mod32) | (h1 >> (32 - mod32)))
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
        *(uint4 *)(a.sums + sum_idx * 32 + 16 * h) = out;
    }
}

/* ---- SHA-256 (one chain per lane), FIPS 180-4 -------------------------- */

/* constexpr (not __constant__): the rounds are fully unrolled with
 * literal indices, so K folds to inline literal operands — as a
 * __constant__ buffer the compiler preloaded round constants into SGPRs
 * (measured 44-88 spilled SGPRs in the NC=2 kernels, r2) */
__device__ constexpr uint32_t SHA_K[64] = {
    0x428a2f98, 0x71374491, 0xb5c0fbcf, 0xe9b5dba5, 0x3956c25b, 0x59f111f1,
    0x923f82a4, 0xab1c5ed5, 0xd807aa98, 0x12835b01, 0x243185be, 0x550c7dc3,
    0x72be5d74, 0x80deb1fe, 0x9bdc06a7, 0xc19bf174, 0xe49b69c1, 0xefbe4786,
    0x0fc19dc6, 0x240ca1cc, 0x2de92c6f, 0x4a7484aa, 0x5cb0a9dc, 0x76f988da,
    0x983e5152, 0xa831c66d, 0xb00327c8, 0xbf597fc7, 0xc6e00bf3, 0xd5a79147,
    0x06ca6351, 0x14292967, 0x27b70a85, 0x2e1b2138, 0x4d2c6dfc, 0x53380d13,
    0x650a7354, 0x766a0abb, 0x81c2c92e, 0x92722c85, 0xa2bfe8a1, 0xa81a664b,
    0xc24b8b70, 0xc76c51a3, 0xd192e819, 0xd6990624, 0xf40e3585, 0x106aa070,
    0x19a4c116, 0x1e376c08, 0x2748774c, 0x34b0bcb5, 0x391c0cb3, 0x4ed8aa4a,
    0x5b9cca4f, 0x682e6ff3, 0x748f82ee, 0x78a5636f, 0x84c87814, 0x8cc70208,
    0x90befffa, 0xa4506ceb, 0xbef9a3f7, 0xc67178f2};

__device__ __forceinline__ uint32_t rotr32(uint32_t x, int n) {
    return (x >> n) | (x << (32 - n));
}

__device__ __forceinline__ uint32_t bswap32(uint32_t v) {
    return __builtin_bswap32(v);
}

__device__ void sha256_block(uint32_t h[8], const uint32_t w_in[16]) {
    uint32_t w[16];
#pragma unroll
    for (int i = 0; i < 16; i++) w[i] = w_in[i];
    uint32_t a = h[0], b = h[1], c = h[2], d = h[3], e = h[4], f = h[5],
             g = h[6], hh = h[7];
    /* full unroll: every w[] index must be compile-time, or the 16-word
     * rolling schedule spills to scratch (measured 144 B/lane, 20x slower) */
#pragma unroll
    for (int i = 0; i < 64; i++) {
        uint32_t wi;
        if (i < 16) {
            wi = w[i];
        } else {
            uint32_t w15 = w[(i - 15) & 15], w2 = w[(i - 2) & 15];
            uint32_t s0 = xor3(rotr32(w15, 7), rotr32(w15, 18), w15 >> 3);
            uint32_t s1 = xor3(rotr32(w2, 17), rotr32(w2, 19), w2 >> 10);
            wi = w[i & 15] + s0 + w[(i - 7) & 15] + s1;
            w[i & 15] = wi;
        }
        /* Ch/Maj/Sigma as single v_bitop3_b32 LUT ops (gfx950): Ch and
         * Maj sit ON the round's dependency chain, so this shortens the
         * critical path as well as the issue count */
        uint32_t S1 = xor3(rotr32(e, 6), rotr32(e, 11), rotr32(e, 25));
        uint32_t ch = __builtin_amdgcn_bitop3_b32(e, f, g, 0xca);
        uint32_t t1 = hh + S1 + ch + SHA_K[i] + wi;
        uint32_t S0 = xor3(rotr32(a, 2), rotr32(a, 13), rotr32(a, 22));
        uint32_t maj = __builtin_amdgcn_bitop3_b32(a, b, c, 0xe8);
        uint32_t t2 = S0 + maj;
        hh = g; g = f; f = e; e = d + t1;
        d = c; c = b; b = a; a = t1 + t2;
    }
    h[0] += a; h[1] += b; h[2] += c; h[3] += d;
    h[4] += e; h[5] += f; h[6] += g; h[7] += hh;
}

/* 16-round-body variant: the fully-unrolled 64-round loop at NC=2 is a
 * ~12 KiB body — big enough that co-resident waves thrash the 32 KiB
 * I-cache (r2: NC2 at WG512 measured 1.8x SLOWER than 1 wave/SIMD).
 * Rolling to a 16-round body (schedule indices stay static mod 16, the
 * a..h renaming closes every 8 rounds) cuts the body 4x; K for rounds
 * 16..63 comes from scalar loads of the rodata table. */
template <int NC, int LB = 256>
__global__ void __launch_bounds__(LB, 1) sha256_batch_r16_kernel(HashArgs a) {
    const int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    const int64_t c0 = tid * NC;
    if (c0 >= a.n_chains) return;

    bool act[NC];
    const uint8_t *mp[NC];
    int64_t sum_idx[NC];
    uint32_t h[NC][8];
#pragma unroll
    for (int u = 0; u < NC; u++) {
        act[u] = c0 + u < a.n_chains;
        mp[u] = chain_ptr(a, act[u] ? c0 + u : c0, sum_idx[u]);
        const uint32_t iv[8] = {0x6a09e667, 0xbb67ae85, 0x3c6ef372,
                                0xa54ff53a, 0x510e527f, 0x9b05688c,
                                0x1f83d9ab, 0x5be0cd19};
#pragma unroll
        for (int i = 0; i < 8; i++) h[u][i] = iv[i];
    }
    int64_t len = a.msg_len;
    /* message double-buffer: the next 64-B block's loads ride under this
     * block's 64 rounds instead of stalling at each block boundary */
    uint4 mb[NC][4], mbn[NC][4];
    if (len >= 64) {
#pragma unroll
        for (int u = 0; u < NC; u++) {
            const uint4 *p = (const uint4 *)mp[u];
#pragma unroll
            for (int q = 0; q < 4; q++) mb[u][q] = p[q];
            mp[u] += 64;
        }
    }
    while (len >= 64) {
        if (len >= 128) {
#pragma unroll
            for (int u = 0; u < NC; u++) {
                const uint4 *p = (const uint4 *)mp[u];
#pragma unroll
                for (int q = 0; q < 4; q++) mbn[u][q] = p[q];
                mp[u] += 64;
            }
        }
        uint32_t w[NC][16];
#pragma unroll
        for (int u = 0; u < NC; u++) {
#pragma unroll
            for (int q = 0; q < 4; q++) {
                w[u][4 * q + 0] = bswap32(mb[u][q].x);
                w[u][4 * q + 1] = bswap32(mb[u][q].y);
                w[u][4 * q + 2] = bswap32(mb[u][q].z);
                w[u][4 * q + 3] = bswap32(mb[u][q].w);
            }
        }
        uint32_t A[NC], B[NC], C[NC], D[NC], E[NC], F[NC], G[NC], H[NC];
#pragma unroll
        for (int u = 0; u < NC; u++) {
            A[u] = h[u][0]; B[u] = h[u][1]; C[u] = h[u][2]; D[u] = h[u][3];
            E[u] = h[u][4]; F[u] = h[u][5]; G[u] = h[u][6]; H[u] = h[u][7];
        }
#define SHA_ROUND(u, wi, kk)                                                 \
        {                                                                    \
            uint32_t S1 = xor3(rotr32(E[u], 6), rotr32(E[u], 11),            \
                               rotr32(E[u], 25));                            \
            uint32_t ch = (uint32_t)__builtin_amdgcn_bitop3_b32(             \
                E[u], F[u], G[u], 0xca);                                     \
            uint32_t t1 = H[u] + S1 + ch + (kk) + (wi);                      \
            uint32_t S0 = xor3(rotr32(A[u], 2), rotr32(A[u], 13),            \
                               rotr32(A[u], 22));                            \
            uint32_t maj = (uint32_t)__builtin_amdgcn_bitop3_b32(            \
                A[u], B[u], C[u], 0xe8);                                     \
            uint32_t t2 = S0 + maj;                                          \
            H[u] = G[u]; G[u] = F[u]; F[u] = E[u]; E[u] = D[u] + t1;         \
            D[u] = C[u]; C[u] = B[u]; B[u] = A[u]; A[u] = t1 + t2;           \
        }
        /* rounds 0..15: message words direct, K folds to literals */
#pragma unroll
        for (int r = 0; r < 16; r++)
#pragma unroll
            for (int u = 0; u < NC; u++) SHA_ROUND(u, w[u][r], SHA_K[r]);
        /* rounds 16..63: three passes of a 16-round body */
#pragma unroll 1
        for (int q = 1; q < 4; q++) {
#pragma unroll
            for (int r = 0; r < 16; r++) {
                const uint32_t kk = SHA_K[(q << 4) + r];
#pragma unroll
                for (int u = 0; u < NC; u++) {
                    uint32_t w15 = w[u][(r + 1) & 15], w2 = w[u][(r + 14) & 15];
                    uint32_t s0 = xor3(rotr32(w15, 7), rotr32(w15, 18),
                                       w15 >> 3);
                    uint32_t s1 = xor3(rotr32(w2, 17), rotr32(w2, 19),
                                       w2 >> 10);
                    uint32_t wi = w[u][r] + s0 + w[u][(r + 9) & 15] + s1;
                    w[u][r] = wi;
                    SHA_ROUND(u, wi, kk);
                }
            }
        }
#undef SHA_ROUND
#pragma unroll
        for (int u = 0; u < NC; u++) {
            h[u][0] += A[u]; h[u][1] += B[u]; h[u][2] += C[u]; h[u][3] += D[u];
            h[u][4] += E[u]; h[u][5] += F[u]; h[u][6] += G[u]; h[u][7] += H[u];
        }
        if (len >= 128) {
#pragma unroll
            for (int u = 0; u < NC; u++)
#pragma unroll
                for (int q = 0; q < 4; q++) mb[u][q] = mbn[u][q];
        }
        len -= 64;
    }
    /* the double-buffer advanced mp one block past the remainder */
#pragma unroll
    for (int u = 0; u < NC; u++)
        mp[u] = chain_ptr(a, act[u] ? c0 + u : c0, sum_idx[u]) +
                (a.msg_len - len);
    /* tail: same as sha256_batch_kernel */
#pragma unroll
    for (int u = 0; u < NC; u++) {
        uint8_t tail[128];
#pragma unroll
        for (int i = 0; i < 128; i++) tail[i] = 0;
        for (int i = 0; i < (int)len; i++) tail[i] = mp[u][i];
        tail[(int)len] = 0x80;
        const int tlen = (len < 56) ? 64 : 128;
        uint64_t bits = (uint64_t)a.msg_len * 8;
        for (int i = 0; i < 8; i++)
            tail[tlen - 1 - i] = (uint8_t)(bits >> (8 * i));
        for (int blk = 0; blk < tlen; blk += 64) {
            uint32_t wt[16];
            for (int i = 0; i < 16; i++) {
                const uint8_t *q = tail + blk + 4 * i;
                wt[i] = ((uint32_t)q[0] << 24) | ((uint32_t)q[1] << 16) |
                        ((uint32_t)q[2] << 8) | q[3];
            }
            sha256_block(h[u], wt);
        }
        if (act[u]) {
            uint8_t *out = a.sums + sum_idx[u] * 32;
            for (int i = 0; i < 8; i++)
                *(uint32_t *)(out + 4 * i) = bswap32(h[u][i]);
        }
    }
}

/* NC independent chains per lane: SHA-256's round chain is strictly serial
 * (measured ~17 cyc/instr effective at 1 chain/lane, 1 wave/SIMD — pure
 * dependency latency); interleaving NC chains fills the stalls. */
template <int NC, int LB = 256>
__global__ void __launch_bounds__(LB, 1) sha256_batch_kernel(HashArgs a) {
    const int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    const int64_t c0 = tid * NC;
    if (c0 >= a.n_chains) return;

    bool act[NC];
    const uint8_t *mp[NC];
    int64_t sum_idx[NC];
    uint32_t h[NC][8];
#pragma unroll
    for (int u = 0; u < NC; u++) {
        act[u] = c0 + u < a.n_chains;
        mp[u] = chain_ptr(a, act[u] ? c0 + u : c0, sum_idx[u]);
        const uint32_t iv[8] = {0x6a09e667, 0xbb67ae85, 0x3c6ef372,
                                0xa54ff53a, 0x510e527f, 0x9b05688c,
                                0x1f83d9ab, 0x5be0cd19};
#pragma unroll
        for (int i = 0; i < 8; i++) h[u][i] = iv[i];
    }
    int64_t len = a.msg_len;
    /* message double-buffer: the next 64-B block's loads ride under this
     * block's 64 rounds instead of stalling at each block boundary */
    uint4 mb[NC][4], mbn[NC][4];
    if (len >= 64) {
#pragma unroll
        for (int u = 0; u < NC; u++) {
            const uint4 *p = (const uint4 *)mp[u];
#pragma unroll
            for (int q = 0; q < 4; q++) mb[u][q] = p[q];
            mp[u] += 64;
        }
    }
    while (len >= 64) {
        if (len >= 128) {
#pragma unroll
            for (int u = 0; u < NC; u++) {
                const uint4 *p = (const uint4 *)mp[u];
#pragma unroll
                for (int q = 0; q < 4; q++) mbn[u][q] = p[q];
                mp[u] += 64;
            }
        }
        uint32_t w[NC][16];
#pragma unroll
        for (int u = 0; u < NC; u++) {
#pragma unroll
            for (int q = 0; q < 4; q++) {
                w[u][4 * q + 0] = bswap32(mb[u][q].x);
                w[u][4 * q + 1] = bswap32(mb[u][q].y);
                w[u][4 * q + 2] = bswap32(mb[u][q].z);
                w[u][4 * q + 3] = bswap32(mb[u][q].w);
            }
        }
        uint32_t A[NC], B[NC], C[NC], D[NC], E[NC], F[NC], G[NC], H[NC];
#pragma unroll
        for (int u = 0; u < NC; u++) {
            A[u] = h[u][0]; B[u] = h[u][1]; C[u] = h[u][2]; D[u] = h[u][3];
            E[u] = h[u][4]; F[u] = h[u][5]; G[u] = h[u][6]; H[u] = h[u][7];
        }
#pragma unroll
        for (int i = 0; i < 64; i++) {
#pragma unroll
            for (int u = 0; u < NC; u++) {
                uint32_t wi;
                if (i < 16) {
                    wi = w[u][i];
                } else {
                    uint32_t w15 = w[u][(i - 15) & 15], w2 = w[u][(i - 2) & 15];
                    uint32_t s0 = rotr32(w15, 7) ^ rotr32(w15, 18) ^ (w15 >> 3);
                    uint32_t s1 = rotr32(w2, 17) ^ rotr32(w2, 19) ^ (w2 >> 10);
                    wi = w[u][i & 15] + s0 + w[u][(i - 7) & 15] + s1;
                    w[u][i & 15] = wi;
                }
                uint32_t S1 = rotr32(E[u], 6) ^ rotr32(E[u], 11) ^ rotr32(E[u], 25);
                uint32_t ch = (E[u] & F[u]) ^ (~E[u] & G[u]);
                uint32_t t1 = H[u] + S1 + ch + SHA_K[i] + wi;
                uint32_t S0 = rotr32(A[u], 2) ^ rotr32(A[u], 13) ^ rotr32(A[u], 22);
                uint32_t maj = (A[u] & B[u]) ^ (A[u] & C[u]) ^ (B[u] & C[u]);
                uint32_t t2 = S0 + maj;
                H[u] = G[u]; G[u] = F[u]; F[u] = E[u]; E[u] = D[u] + t1;
                D[u] = C[u]; C[u] = B[u]; B[u] = A[u]; A[u] = t1 + t2;
            }
        }
#pragma unroll
        for (int u = 0; u < NC; u++) {
            h[u][0] += A[u]; h[u][1] += B[u]; h[u][2] += C[u]; h[u][3] += D[u];
            h[u][4] += E[u]; h[u][5] += F[u]; h[u][6] += G[u]; h[u][7] += H[u];
        }
        if (len >= 128) {
#pragma unroll
            for (int u = 0; u < NC; u++)
#pragma unroll
                for (int q = 0; q < 4; q++) mb[u][q] = mbn[u][q];
        }
        len -= 64;
    }
    /* tail: rem bytes + 0x80 pad + 8-byte big-endian bit length */
#pragma unroll
    for (int u = 0; u < NC; u++) {
        uint8_t tail[128];
#pragma unroll
        for (int i = 0; i < 128; i++) tail[i] = 0;
        for (int i = 0; i < (int)len; i++) tail[i] = mp[u][i];
        tail[(int)len] = 0x80;
        const int tlen = (len < 56) ? 64 : 128;
        uint64_t bits = (uint64_t)a.msg_len * 8;
        for (int i = 0; i < 8; i++)
            tail[tlen - 1 - i] = (uint8_t)(bits >> (8 * i));
        for (int blk = 0; blk < tlen; blk += 64) {
            uint32_t wt[16];
            for (int i = 0; i < 16; i++) {
                const uint8_t *q = tail + blk + 4 * i;
                wt[i] = ((uint32_t)q[0] << 24) | ((uint32_t)q[1] << 16) |
                        ((uint32_t)q[2] << 8) | q[3];
            }
            sha256_block(h[u], wt);
        }
        if (act[u]) {
            uint8_t *out = a.sums + sum_idx[u] * 32;
            for (int i = 0; i < 8; i++)
                *(uint32_t *)(out + 4 * i) = bswap32(h[u][i]);
        }
    }
}

/* ---- BLAKE2b-512 (one chain per lane), RFC 7693 ------------------------ */

__constant__ uint64_t B2B_IV[8] = {
    0x6a09e667f3bcc908ull, 0xbb67ae8584caa73bull, 0x3c6ef372fe94f82bull,
    0xa54ff53a5f1d36f1ull, 0x510e527fade682d1ull, 0x9b05688c2b3e6c1full,
    0x1f83d9abfb41bd6bull, 0x5be0cd19137e2179ull};

__constant__ uint8_t B2B_SIGMA[12][16] = {
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

__device__ __forceinline__ uint64_t rotr64d(uint64_t x, int n) {
    return (x >> n) | (x << (64 - n));
}

__device__ void b2b_compress(uint64_t h[8], const uint64_t m[16], uint64_t t,
                             bool last) {
    uint64_t v[16];
#pragma unroll
    for (int i = 0; i < 8; i++) v[i] = h[i];
#pragma unroll
    for (int i = 0; i < 8; i++) v[8 + i] = B2B_IV[i];
    v[12] ^= t;
    if (last) v[14] = ~v[14];
#pragma unroll 1
    for (int r = 0; r < 12; r++) {
        const uint8_t *sg = B2B_SIGMA[r];
#define B2B_G(ai, bi, ci, di, x, y)                                          \
    {                                                                        \
        v[ai] = v[ai] + v[bi] + (x);                                         \
        v[di] = rotr64d(v[di] ^ v[ai], 32);                                  \
        v[ci] = v[ci] + v[di];                                               \
        v[bi] = rotr64d(v[bi] ^ v[ci], 24);                                  \
        v[ai] = v[ai] + v[bi] + (y);                                         \
        v[di] = rotr64d(v[di] ^ v[ai], 16);                                  \
        v[ci] = v[ci] + v[di];                                               \
        v[bi] = rotr64d(v[bi] ^ v[ci], 63);                                  \
    }
        B2B_G(0, 4, 8, 12, m[sg[0]], m[sg[1]]);
        B2B_G(1, 5, 9, 13, m[sg[2]], m[sg[3]]);
        B2B_G(2, 6, 10, 14, m[sg[4]], m[sg[5]]);
        B2B_G(3, 7, 11, 15, m[sg[6]], m[sg[7]]);
        B2B_G(0, 5, 10, 15, m[sg[8]], m[sg[9]]);
        B2B_G(1, 6, 11, 12, m[sg[10]], m[sg[11]]);
        B2B_G(2, 7, 8, 13, m[sg[12]], m[sg[13]]);
        B2B_G(3, 4, 9, 14, m[sg[14]], m[sg[15]]);
#undef B2B_G
    }
#pragma unroll
    for (int i = 0; i < 8; i++) h[i] ^= v[i] ^ v[8 + i];
}

__global__ void __launch_bounds__(256) blake2b512_batch_kernel(HashArgs a) {
    const int64_t chain = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (chain >= a.n_chains) return;
    int64_t sum_idx;
    const uint8_t *msg = chain_ptr(a, chain, sum_idx);

    uint64_t h[8];
#pragma unroll
    for (int i = 0; i < 8; i++) h[i] = B2B_IV[i];
    h[0] ^= 0x01010000ull ^ 64;
    int64_t off = 0;
    /* message double-buffer (r2, same treatment as SHA r16): the next
     * 128-B block's loads ride under this block's 12 rounds */
    uint64_t mA[16], mB[16];
#define B2B_LOAD(M, OFF)                                                     \
    {                                                                        \
        const uint4 *p = (const uint4 *)(msg + (OFF));                       \
        _Pragma("unroll") for (int q = 0; q < 8; q++) {                      \
            uint4 v = p[q];                                                  \
            M[2 * q] = (uint64_t)v.x | ((uint64_t)v.y << 32);                \
            M[2 * q + 1] = (uint64_t)v.z | ((uint64_t)v.w << 32);            \
        }                                                                    \
    }
    if (a.msg_len - off > 128) B2B_LOAD(mA, off)
    while (a.msg_len - off > 128) {
        if (a.msg_len - (off + 128) > 128) B2B_LOAD(mB, off + 128)
        b2b_compress(h, mA, (uint64_t)(off + 128), false);
        if (a.msg_len - (off + 128) > 128) {
#pragma unroll
            for (int i = 0; i < 16; i++) mA[i] = mB[i];
        }
        off += 128;
    }
#undef B2B_LOAD
    uint64_t m[16];
    {
        const int rem = (int)(a.msg_len - off);
        uint8_t tail[128];
#pragma unroll
        for (int i = 0; i < 128; i++) tail[i] = 0;
        for (int i = 0; i < rem; i++) tail[i] = msg[off + i];
        for (int i = 0; i < 16; i++) {
            uint64_t v = 0;
            for (int bt = 7; bt >= 0; bt--) v = (v << 8) | tail[8 * i + bt];
            m[i] = v;
        }
        b2b_compress(h, m, (uint64_t)a.msg_len, true);
    }
    uint8_t *out = a.sums + sum_idx * 64;
    for (int i = 0; i < 8; i++) *(uint64_t *)(out + 8 * i) = h[i];
}


/* ---- streaming-format device assembly -------------------
 * scatter_rows: packed object bytes -> padded strided shard rows (Split
 * semantics done on-device, cmd/erasure-coding.go:81).
 * stream_interleave: rows + sums -> per-drive on-disk [hash||shard]*
 * streams (cmd/bitrot-streaming.go:57-75) in one device buffer. */

__global__ void __launch_bounds__(256) scatter_rows_kernel(ScatterArgs a) {
    /* one 16-B slot per thread over n*d*ceil(S/16) */
    const int64_t per_row = (a.S + 15) >> 4;
    const int64_t total = a.n * a.d * per_row;
    for (int64_t u = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         u < total; u += (int64_t)gridDim.x * blockDim.x) {
        const int64_t b = u / (a.d * per_row);
        const int64_t r = u % (a.d * per_row);
        const int k = (int)(r / per_row);
        const int64_t o = (r % per_row) * 16;
        const int64_t src_off = (int64_t)k * a.S + o;
        uint4 v = uint4{0, 0, 0, 0};
        const uint8_t *sp = a.src + b * a.block_len + src_off;
        if (src_off + 16 <= a.block_len) {
            /* interior slot: byte-wise safe unaligned gather is avoided by
             * the packed layout being arbitrary — use 4-B loads (packed
             * source has no 16-B alignment guarantee at shard boundaries) */
            const uint32_t *p4 = (const uint32_t *)sp;
            if (((uintptr_t)sp & 3) == 0) {
                v.x = p4[0]; v.y = p4[1]; v.z = p4[2]; v.w = p4[3];
            } else {
                alignas(16) uint8_t tmp[16];
                for (int i = 0; i < 16; i++) tmp[i] = sp[i];
                v = *(uint4 *)tmp;
            }
        } else if (src_off < a.block_len) {
            alignas(16) uint8_t tmp[16] = {0};
            const int have = (int)(a.block_len - src_off);
            for (int i = 0; i < have; i++) tmp[i] = sp[i];
            v = *(uint4 *)tmp;
        }
        *(uint4 *)&a.rows[(b * a.d + k) * a.row_stride + o] = v;
    }
}

__global__ void __launch_bounds__(256) stream_interleave_kernel(
    InterleaveArgs a) {
    /* One workgroup per [hash||shard] entry (drive s, block b).  The
     * entry's destination base (s*n+b)*pitch is byte-arbitrary whenever
     * pitch = 32+S is odd (every ragged-S geometry, e.g. EC12+4), so the
     * r1 version fell back to byte-wise copies for exactly those
     * geometries.  Here stores are always aligned
     * 16-B dwordx4 to the dst: a head/tail handles the odd edges
     * byte-wise, and the body gathers each 16 output bytes from two
     * aligned source loads sheared with v_alignbyte (constant shift per
     * entry). */
    const int total = a.d + a.p;
    const int64_t pitch = 32 + a.S;
    const int64_t e = blockIdx.x;          /* entry: s = e/n, b = e%n */
    const int s = (int)(e / a.n);
    const int64_t b = e % a.n;
    if (s >= total) return;
    uint8_t *dst = a.out + e * pitch;
    const uint8_t *h = a.sums + (b * total + s) * 32;
    const uint8_t *row = (s < a.d)
        ? a.data + (b * a.d + s) * a.row_stride
        : a.parity + (b * a.p + (s - a.d)) * a.row_stride;
    const int t = (int)threadIdx.x;
    /* hash half: 32 B, byte-wise (arbitrary dst alignment, tiny) */
    if (t < 32) dst[t] = h[t];
    /* shard: dst offset 32, src row 16-B aligned */
    uint8_t *sd = dst + 32;
    const int head = (int)((16 - ((uintptr_t)sd & 15)) & 15);
    const int64_t S = a.S;
    /* head + tail bytes */
    for (int64_t i = t; i < head && i < S; i += blockDim.x) sd[i] = row[i];
    const int64_t units = (S - head) / 16;   /* full 16-B dst units */
    for (int64_t i = head + units * 16 + t; i < S; i += blockDim.x)
        sd[i] = row[i];
    /* body: dst 16-B aligned; src at constant misalignment sh */
    const int sh = head & 15;
    const int dw = sh >> 2, bo = sh & 3;
    for (int64_t u = t; u < units; u += blockDim.x) {
        const int64_t off = head + u * 16;
        const int64_t base = off & ~(int64_t)15;
        if (base + 32 <= a.row_stride) {
            uint4 q0 = *(const uint4 *)(row + base);
            uint4 q1 = *(const uint4 *)(row + base + 16);
            uint32_t d8[8] = {q0.x, q0.y, q0.z, q0.w, q1.x, q1.y, q1.z,
                              q1.w};
            uint4 o;
            /* out dword i = bytes [off+4i, off+4i+4): alignbyte shears
             * the concat {hi:lo} right by bo bytes */
            o.x = __builtin_amdgcn_alignbyte(d8[dw + 1], d8[dw + 0], bo);
            o.y = __builtin_amdgcn_alignbyte(d8[dw + 2], d8[dw + 1], bo);
            o.z = __builtin_amdgcn_alignbyte(d8[dw + 3], d8[dw + 2], bo);
            o.w = __builtin_amdgcn_alignbyte(d8[dw + 4], d8[dw + 3], bo);
            *(uint4 *)(sd + off) = o;
        } else {
            /* last unit of the last row may not have 32 readable source
             * bytes within the stride: byte-wise */
            for (int i2 = 0; i2 < 16; i2++) sd[off + i2] = row[off + i2];
        }
    }
}

/* ---- launch wrappers (called from ec_abi.cpp) -------------------------- */

extern "C" {

/* Specialized encode launch; returns hipErrorNotSupported when (d,p) has no
 * compiled specialization (caller falls back to the generic kernel). */
static int gf_env_int(const char *name, int dflt) {
    const char *v = getenv(name);
    return v ? atoi(v) : dflt;
}

hipError_t mec_launch_gf_encode_spec(int d, int p, const GfEncArgs *args,
                                     int n, hipStream_t stream) {
    static const int env_bs = gf_env_int("MEC_GF_BS", 1);
    if (env_bs) {
        /* bit-sliced encode (r2 default): ~2.4x fewer VALU slots than the
         * xtime ladder; 32 B per lane */
        const int64_t cols = (args->shard_len + 31) / 32;
        static const int bswgx = gf_env_int("MEC_GF_BSWGX", 8);
        int64_t max_x = (cols + 255) / 256;
        int64_t want_x = ((int64_t)2048 * bswgx + n - 1) / n;
        int64_t blocks_x = want_x < max_x ? want_x : max_x;
        if (blocks_x < 1) blocks_x = 1;
        dim3 grid((uint32_t)blocks_x, n);
        dim3 blk(256);
        static const int bspf = gf_env_int("MEC_GF_BSPF", 2);
#define XBS(D, P)                                                            \
        if (d == D && p == P) {                                              \
            if (bspf >= 4)                                                   \
                hipLaunchKernelGGL(                                          \
                    (gf_encode_bs_kernel<D, P, MAT_##D##_##P, true, 4>),     \
                    grid, blk, 0, stream, *args);                            \
            else if (bspf == 2)                                              \
                hipLaunchKernelGGL(                                          \
                    (gf_encode_bs_kernel<D, P, MAT_##D##_##P, true, 2>),     \
                    grid, blk, 0, stream, *args);                            \
            else                                                             \
                hipLaunchKernelGGL(                                          \
                    (gf_encode_bs_kernel<D, P, MAT_##D##_##P, true, 1>),     \
                    grid, blk, 0, stream, *args);                            \
            return hipGetLastError();                                        \
        }
        MEC_SPECIALIZED_GEOS(XBS)
#undef XBS
    }
    static const int env_w = gf_env_int("MEC_GF_W", 1);
    static const int env_nt = gf_env_int("MEC_GF_NT", 1);
    static const int env_wgx = gf_env_int("MEC_GF_WGX", 4);
    static const int env_cap = gf_env_int("MEC_GF_CAP", 0); /* wg/CU cap */
    static const int env_lds = 0; /* dynamic-LDS cap was a no-op; see CAP */
    const int W = (env_w == 1) ? 1 : 2;
    const int64_t cols = (args->shard_len + 16 * W - 1) / (16 * W);
    int64_t max_x = (cols + 255) / 256;
    int64_t want_x = ((int64_t)2048 * env_wgx + n - 1) / n;
    int64_t blocks_x = want_x < max_x ? want_x : max_x;
    if (blocks_x < 1) blocks_x = 1;
    dim3 grid((uint32_t)blocks_x, n);
    dim3 blk(256);
#define X(D, P)                                                              \
    if (d == D && p == P) {                                                  \
        if (env_cap == 4 && W == 1 && env_nt)                                \
            hipLaunchKernelGGL(                                              \
                (gf_encode_kernel<D, P, MAT_##D##_##P, 1, true, 36864>),     \
                grid, blk, env_lds, stream, *args);                          \
        else if (env_cap == 2 && W == 1 && env_nt)                           \
            hipLaunchKernelGGL(                                              \
                (gf_encode_kernel<D, P, MAT_##D##_##P, 1, true, 65536>),     \
                grid, blk, env_lds, stream, *args);                          \
        else if (W == 1 && !env_nt)                                          \
            hipLaunchKernelGGL((gf_encode_kernel<D, P, MAT_##D##_##P, 1,     \
                                                 false>),                    \
                               grid, blk, env_lds, stream, *args);           \
        else if (W == 1)                                                     \
            hipLaunchKernelGGL((gf_encode_kernel<D, P, MAT_##D##_##P, 1,     \
                                                 true>),                     \
                               grid, blk, env_lds, stream, *args);           \
        else if (!env_nt)                                                    \
            hipLaunchKernelGGL((gf_encode_kernel<D, P, MAT_##D##_##P, 2,     \
                                                 false>),                    \
                               grid, blk, env_lds, stream, *args);           \
        else                                                                 \
            hipLaunchKernelGGL((gf_encode_kernel<D, P, MAT_##D##_##P, 2,     \
                                                 true>),                     \
                               grid, blk, env_lds, stream, *args);           \
        return hipGetLastError();                                            \
    }
    MEC_SPECIALIZED_GEOS(X)
#undef X
    return hipErrorNotSupported;
}

hipError_t mec_launch_gf_matmul(const GfMatmulArgs *args, int n_dst, int n,
                                hipStream_t stream) {
    static const int env_bs = gf_env_int("MEC_GFM_BS", 1);
    if (env_bs && args->bs_masks != nullptr) {
        const int64_t cols32 = (args->shard_len + 31) / 32;
        int64_t max_x = (cols32 + 255) / 256;
        int64_t want_x = ((int64_t)2048 * 4 + n - 1) / n;
        int64_t blocks_x = want_x < max_x ? want_x : max_x;
        if (blocks_x < 1) blocks_x = 1;
        dim3 grid((uint32_t)blocks_x, n);
        dim3 blk(256);
#define CASEB(E)                                                             \
    case E:                                                                  \
        if (env_bs >= 2)                                                     \
            hipLaunchKernelGGL((gf_matmul_bs_kernel<E, 1>), grid, blk, 0,    \
                               stream, *args);                               \
        else                                                                 \
            hipLaunchKernelGGL((gf_matmul_bs_kernel<E, 0>), grid, blk, 0,    \
                               stream, *args);                               \
        return hipGetLastError();
        switch (n_dst) {
            CASEB(1) CASEB(2) CASEB(3) CASEB(4) CASEB(5) CASEB(6) CASEB(7)
            CASEB(8)
        default:
            return hipErrorInvalidValue;
        }
#undef CASEB
    }
    const int64_t cols = (args->shard_len + 15) >> 4;
    /* >=2048 workgroups total to fill 256 CUs, but never more than the work */
    int64_t max_x = (cols + 255) / 256;
    int64_t want_x = (2048 + n - 1) / n;
    int64_t blocks_x = want_x < max_x ? want_x : max_x;
    if (blocks_x < 1) blocks_x = 1;
    dim3 grid((uint32_t)blocks_x, n);
    dim3 blk(256);
    static const char *envm = getenv("MEC_GFM_MASKED");
    static const bool masked = envm && atoi(envm) != 0;
#define CASE(E)                                                              \
    case E:                                                                  \
        if (masked)                                                          \
            hipLaunchKernelGGL((gf_matmul_kernel<E, true>), grid, blk, 0,    \
                               stream, *args);                               \
        else                                                                 \
            hipLaunchKernelGGL((gf_matmul_kernel<E, false>), grid, blk, 0,   \
                               stream, *args);                               \
        break;
    switch (n_dst) {
        CASE(1) CASE(2) CASE(3) CASE(4) CASE(5) CASE(6) CASE(7) CASE(8)
    default:
        return hipErrorInvalidValue;
    }
#undef CASE
    return hipGetLastError();
}

hipError_t mec_launch_scatter_rows(const ScatterArgs *args,
                                   hipStream_t stream) {
    const int64_t per_row = (args->S + 15) >> 4;
    int64_t units = args->n * args->d * per_row;
    int64_t blocks = (units + 255) / 256;
    if (blocks > 16384) blocks = 16384;
    hipLaunchKernelGGL(scatter_rows_kernel, dim3((uint32_t)blocks), dim3(256),
                       0, stream, *args);
    return hipGetLastError();
}

hipError_t mec_launch_stream_interleave(const InterleaveArgs *args,
                                        hipStream_t stream) {
    int64_t entries = (int64_t)(args->d + args->p) * args->n;
    hipLaunchKernelGGL(stream_interleave_kernel, dim3((uint32_t)entries),
                       dim3(256), 0, stream, *args);
    return hipGetLastError();
}

hipError_t mec_launch_hash(int algo, const HashArgs *args,
                           hipStream_t stream) {
    const int64_t blocks = (args->n_chains + 255) / 256;
    dim3 grid((uint32_t)blocks);
    dim3 blk(256);
    switch (algo) {
    case 1: /* SHA256: serial rounds -> fill dependency-latency stalls
               with NC interleaved chains.  NC=2 at WG 256 is the default.
               MEC_SHA_NC=4 (64-thread launch-bounds variant) measured
               2.7x SLOWER (29.3 vs 10.7 ms at config #3) and NC=3 at
               LB 128 measured 2.2x slower (23.2 ms) — both spill the
               message schedule to scratch (272/208 B) and thrash; they
               are kept only as recorded negatives.  Ch/Maj/sigma
               run as single v_bitop3 LUT ops; that changed nothing
               either (the chain is latency-bound, not issue-bound). */
        {
            static const int nc = gf_env_int("MEC_SHA_NC", 2);
            /* MEC_SHA_WG=512: 8-wave workgroups -> 2 waves/SIMD on half
             * the CUs.  SHA's round chain is dependency-latency-bound
             * (~17 cyc/instr at 1 wave/SIMD, r1) and its bandwidth need
             * is low (~0.5 TB/s at config #3), so trading engaged CUs
             * for co-resident waves that fill each other's stalls is the
             * remaining occupancy lever (r2 probe). */
            static const int swg = gf_env_int("MEC_SHA_WG", 256);
            static const int r16 = gf_env_int("MEC_SHA_R16", 1);
            if (r16) {
                if (swg >= 512 && nc == 2) {
                    dim3 b512(512);
                    grid.x =
                        (uint32_t)(((args->n_chains + 1) / 2 + 511) / 512);
                    hipLaunchKernelGGL((sha256_batch_r16_kernel<2, 512>),
                                       grid, b512, 0, stream, *args);
                    break;
                }
                grid.x = (uint32_t)(((args->n_chains + 1) / 2 + 255) / 256);
                hipLaunchKernelGGL((sha256_batch_r16_kernel<2>), grid, blk,
                                   0, stream, *args);
                break;
            }
            if (swg >= 512 && nc == 2) {
                dim3 b512(512);
                grid.x = (uint32_t)(((args->n_chains + 1) / 2 + 511) / 512);
                hipLaunchKernelGGL((sha256_batch_kernel<2, 512>), grid,
                                   b512, 0, stream, *args);
                break;
            }
            if (swg >= 512 && nc == 1) {
                dim3 b512(512);
                grid.x = (uint32_t)((args->n_chains + 511) / 512);
                hipLaunchKernelGGL((sha256_batch_kernel<1, 512>), grid,
                                   b512, 0, stream, *args);
                break;
            }
            if (nc >= 4) {
                dim3 b64(64);
                grid.x =
                    (uint32_t)(((args->n_chains + 3) / 4 + 63) / 64);
                hipLaunchKernelGGL((sha256_batch_kernel<4, 64>), grid, b64,
                                   0, stream, *args);
            } else if (nc == 3) {
                dim3 b128(128);
                grid.x =
                    (uint32_t)(((args->n_chains + 2) / 3 + 127) / 128);
                hipLaunchKernelGGL((sha256_batch_kernel<3, 128>), grid,
                                   b128, 0, stream, *args);
            } else {
                grid.x = (uint32_t)(((args->n_chains + 1) / 2 + 255) / 256);
                hipLaunchKernelGGL((sha256_batch_kernel<2>), grid, blk, 0,
                                   stream, *args);
            }
        }
        break;
    case 2: /* HighwayHash256 */
    case 3: /* HighwayHash256S: 2 lanes/chain (zipper pairs).  32-aligned
               lengths run 2 chains/lane (ILP over the serial chain's
               dependency stalls) with the ragged-tail code compiled out;
               ragged lengths take the 1-chain/lane full kernel. */
        /* NC=2 measured 2.3x slower even with the tail compiled out (the
         * doubled stream reaches 256 VGPR and the interleave does not
         * cover HH's dependency chains the way it does SHA's) — NC=1 for
         * both cases; aligned lengths still skip the tail code.
         * Workgroup size via MEC_HH_WG: 512 packs 2 hash waves per SIMD
         * (a wave's dependency stalls are covered by its co-resident
         * partner); 256 spreads 1 wave/SIMD. */
        {
            static const int use_lds = gf_env_int("MEC_HH_LDS", 0);
            static const int wg = gf_env_int("MEC_HH_WG", 256);
            static const int hh4 = gf_env_int("MEC_HH4", 1);
            if (hh4 && !use_lds) {
                /* 4-lane-per-chain kernel (see hh256_batch4_kernel):
                 * 2x the waves of the pair kernel -> ~2x engaged CUs.
                 * MEC_HH4_WG: workgroup size — at the headline chain
                 * count a 256-thread WG yields 192 WGs = 192 CUs (64
                 * idle); 64-thread WGs spread the same waves over all
                 * 256 CUs (r2 sweep). */
                static const int wg4 = gf_env_int("MEC_HH4_WG", 256);
                const int64_t lanes = args->n_chains * 4;
#define HH4L(WGV)                                                            \
                {                                                            \
                    dim3 hblk(WGV);                                          \
                    grid.x = (uint32_t)((lanes + WGV - 1) / WGV);            \
                    if (args->msg_len % 32 == 0)                             \
                        hipLaunchKernelGGL(                                  \
                            (hh256_batch4_kernel<false, WGV>), grid, hblk,   \
                            0, stream, *args);                               \
                    else                                                     \
                        hipLaunchKernelGGL(                                  \
                            (hh256_batch4_kernel<true, WGV>), grid, hblk,    \
                            0, stream, *args);                               \
                }
                if (wg4 <= 64) HH4L(64)
                else if (wg4 <= 128) HH4L(128)
                else HH4L(256)
#undef HH4L
                break;
            }
            if (use_lds && args->msg_len >= 512) {
                dim3 hblk(256);
                grid.x = (uint32_t)((args->n_chains + 127) / 128);
                if (args->msg_len % 32 == 0)
                    hipLaunchKernelGGL((hh256_lds_kernel<false>), grid,
                                       hblk, 0, stream, *args);
                else
                    hipLaunchKernelGGL((hh256_lds_kernel<true>), grid, hblk,
                                       0, stream, *args);
                break;
            }
            dim3 hblk((uint32_t)(wg >= 512 ? 512 : 256));
            grid.x = (uint32_t)((args->n_chains * 2 + hblk.x - 1) / hblk.x);
            if (args->msg_len % 32 == 0)
                hipLaunchKernelGGL((hh256_batch_kernel<1, false>), grid,
                                   hblk, 0, stream, *args);
            else
                hipLaunchKernelGGL((hh256_batch_kernel<1, true>), grid, hblk,
                                   0, stream, *args);
        }
        break;
    case 4: /* BLAKE2b512 */
        hipLaunchKernelGGL(blake2b512_batch_kernel, grid, blk, 0, stream,
                           *args);
        break;
    default:
        return hipErrorInvalidValue;
    }
    return hipGetLastError();
}
}
