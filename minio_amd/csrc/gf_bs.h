/* Bit-sliced GF(2^8) device helpers shared by the standalone encode
 * kernel (kernels.hip) and the fused producer (fused3.hip).  See
 * gf_encode_bs_kernel's comment for the design and measured numbers. */
#ifndef MEC_GF_BS_H
#define MEC_GF_BS_H
#include <hip/hip_runtime.h>
#include <cstdint>

/* ---- bit-sliced specialized encode (r2) --------------------------------
 *
 * The SWAR xtime ladder costs ~1.46 VALU per input byte (and the int-VALU
 * pipe measures ~4 cyc/instr on gfx950, making the GF leg VALU-PIPE-bound
 * at ~0.32 ms for the headline batch — r2 SQ counters).  Bit-slicing cuts
 * the instruction count ~2.4x: each lane takes a 32-byte column per row,
 * transposes it to 8 bit-planes (3-stage delta-swap network, an
 * involution — verified exhaustively on CPU), and then EVERY GF(2^8)
 * constant-multiply-accumulate is a straight-line XOR of planes with the
 * coefficient's bit-matrix rows folded at compile time (constexpr over
 * MAT): ~16 xors per (input,parity) pair instead of the ladder's ~90
 * slots.  Transposes: 60 ops per 32 B, amortized over P outputs.
 */
__device__ __forceinline__ void bs_pair(uint32_t &A, uint32_t &B, int s,
                                        uint32_t m) {
    /* delta-swap: exchanges bit groups between regs A and B */
    uint32_t t = (uint32_t)__builtin_amdgcn_bitop3_b32(B << s, A, m,
                                                       0x28); /* (b^a)&m */
    A ^= t;
    B ^= t >> s;
}

__device__ __forceinline__ void bs_transpose(uint32_t r[8]) {
    /* stages d=1,2,4: after this, r[b] holds bit b of all 32 bytes
     * (slot order is a fixed byte permutation, identical across planes;
     * the same network inverts it — involution) */
    bs_pair(r[0], r[1], 1, 0xAAAAAAAAu);
    bs_pair(r[2], r[3], 1, 0xAAAAAAAAu);
    bs_pair(r[4], r[5], 1, 0xAAAAAAAAu);
    bs_pair(r[6], r[7], 1, 0xAAAAAAAAu);
    bs_pair(r[0], r[2], 2, 0xCCCCCCCCu);
    bs_pair(r[1], r[3], 2, 0xCCCCCCCCu);
    bs_pair(r[4], r[6], 2, 0xCCCCCCCCu);
    bs_pair(r[5], r[7], 2, 0xCCCCCCCCu);
    bs_pair(r[0], r[4], 4, 0xF0F0F0F0u);
    bs_pair(r[1], r[5], 4, 0xF0F0F0F0u);
    bs_pair(r[2], r[6], 4, 0xF0F0F0F0u);
    bs_pair(r[3], r[7], 4, 0xF0F0F0F0u);
}

/* constexpr GF(2^8)/0x11D multiply and bit-matrix row masks */
constexpr uint8_t bs_gfmul(uint8_t a, uint8_t b) {
    uint32_t r = 0, x = a;
    for (int i = 0; i < 8; i++) {
        if ((b >> i) & 1) r ^= x << i;
    }
    /* reduce 15-bit poly product mod 0x11D */
    for (int i = 14; i >= 8; i--)
        if ((r >> i) & 1) r ^= 0x11Du << (i - 8);
    return (uint8_t)r;
}
/* rowmask(c, b) bit a: output bit b of c*x depends on input bit a */
constexpr uint8_t bs_rowmask(uint8_t c, int b) {
    uint8_t m = 0;
    for (int a = 0; a < 8; a++)
        if ((bs_gfmul(c, (uint8_t)(1u << a)) >> b) & 1)
            m |= (uint8_t)(1u << a);
    return m;
}

/* explicit xor3 pair-folding of a plane-XOR set (the compiler leaves
 * these as chains of v_xor otherwise — measured 1267 plain xors/loop) */
template <uint8_t M>
__device__ __forceinline__ uint32_t bs_fold(const uint32_t x[8],
                                            uint32_t acc) {
    if constexpr (M == 0) {
        return acc;
    } else {
        constexpr int a0 = __builtin_ctz(M);
        constexpr uint8_t M1 = M & (M - 1);
        if constexpr (M1 == 0) {
            return acc ^ x[a0];
        } else {
            constexpr int a1 = __builtin_ctz(M1);
            constexpr uint8_t M2 = M1 & (M1 - 1);
            return bs_fold<M2>(
                x, (uint32_t)__builtin_amdgcn_bitop3_b32(acc, x[a0], x[a1],
                                                         0x96));
        }
    }
}

/* compile-time iteration over (parity row, plane) so the fold masks are
 * constant expressions */
template <int D, int P, const uint8_t (&MAT)[P][D], int K, int I, int PB>
__device__ __forceinline__ void bs_acc_all(const uint32_t xc[8],
                                           uint32_t accp[P][8]) {
    if constexpr (I < P) {
        accp[I][PB] = bs_fold<bs_rowmask(MAT[I][K], PB)>(xc, accp[I][PB]);
        if constexpr (PB < 7)
            bs_acc_all<D, P, MAT, K, I, PB + 1>(xc, accp);
        else
            bs_acc_all<D, P, MAT, K, I + 1, 0>(xc, accp);
    }
}

template <int D, int P, const uint8_t (&MAT)[P][D], int K = 0>
__device__ __forceinline__ void bs_acc_k(int k, const uint32_t xc[8],
                                         uint32_t accp[P][8]) {
    if constexpr (K < D) {
        if (k == K)
            bs_acc_all<D, P, MAT, K, 0, 0>(xc, accp);
        else
            bs_acc_k<D, P, MAT, K + 1>(k, xc, accp);
    }
}


#endif /* MEC_GF_BS_H */
