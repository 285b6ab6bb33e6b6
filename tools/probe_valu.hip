/* Probe: integer-VALU issue rate on gfx950, per instruction kind.
 *
 * The r2 design notes claim the erasure+bitrot workload is bounded by the
 * integer-VALU pipe at ~4 cycles/wave-instruction (inferred from SQ
 * counters on the production kernels).  This probe measures the rate
 * DIRECTLY: 8 independent accumulator streams per lane (ILP 8, chain
 * depth 1 per stream per iteration — issue-limited, not latency-limited),
 * one wave or two waves per SIMD, no memory traffic in the timed loop.
 *
 * cyc/instr = time * 2.4e9 / (iters * 8 * waves_per_simd)   [per SIMD]
 * (clock assumed 2.4 GHz; DVFS wobble ±10% — compare kinds, not digits.)
 */
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdint>
#include <cstdlib>

#define CK(x) do { hipError_t e = (x); if (e != hipSuccess) { \
    fprintf(stderr, "HIP err %s @%d\n", hipGetErrorString(e), __LINE__); \
    exit(1); } } while (0)

enum Kind { K_XOR, K_BITOP3, K_PERM, K_ADD64, K_MAD64, K_MIX };

template <int KIND, int WG>
__global__ void __launch_bounds__(WG) valu_probe(uint32_t *out, int iters) {
    uint32_t a[8];
    uint64_t b[4];
#pragma unroll
    for (int j = 0; j < 8; j++)
        a[j] = threadIdx.x * 2654435761u + j * 40503u + blockIdx.x;
#pragma unroll
    for (int j = 0; j < 4; j++)
        b[j] = ((uint64_t)a[j] << 32) | a[j + 4];
#pragma unroll 1
    for (int i = 0; i < iters; i++) {
#pragma unroll
        for (int j = 0; j < 8; j++) {
            if (KIND == K_XOR) {
                a[j] ^= a[(j + 1) & 7];
            } else if (KIND == K_BITOP3) {
                a[j] = (uint32_t)__builtin_amdgcn_bitop3_b32(
                    a[j], a[(j + 1) & 7], a[(j + 2) & 7], 0x96);
            } else if (KIND == K_PERM) {
                a[j] = __builtin_amdgcn_perm(a[j], a[(j + 1) & 7],
                                             a[(j + 2) & 7]);
            } else if (KIND == K_MIX) {
                /* the production kernels' mix: bitop3 / perm / xor / shift */
                if ((j & 3) == 0)
                    a[j] = (uint32_t)__builtin_amdgcn_bitop3_b32(
                        a[j], a[(j + 1) & 7], a[(j + 2) & 7], 0x96);
                else if ((j & 3) == 1)
                    a[j] = __builtin_amdgcn_perm(a[j], a[(j + 1) & 7],
                                                 a[(j + 2) & 7]);
                else if ((j & 3) == 2)
                    a[j] ^= a[(j + 1) & 7];
                else
                    a[j] = (a[j] << 1) ^ a[(j + 1) & 7];
            }
        }
        if (KIND == K_ADD64 || KIND == K_MAD64) {
#pragma unroll
            for (int j = 0; j < 4; j++) {
                if (KIND == K_ADD64) {
                    b[j] += b[(j + 1) & 3];
                    b[j] += b[(j + 2) & 3];
                } else {
                    b[j] ^= (b[j] & 0xffffffffull) * (b[(j + 1) & 3] >> 32);
                }
            }
        }
    }
    uint32_t acc = a[0] ^ a[1] ^ a[2] ^ a[3] ^ a[4] ^ a[5] ^ a[6] ^ a[7] ^
                   (uint32_t)b[0] ^ (uint32_t)b[1] ^ (uint32_t)b[2] ^
                   (uint32_t)b[3];
    if (acc == 0xDEADBEEFu) *out = acc;
}

template <int KIND, int WG>
static void run(const char *name, int ops_per_iter) {
    const int iters = 200000;
    uint32_t *out;
    CK(hipMalloc(&out, 4));
    hipEvent_t e0, e1;
    CK(hipEventCreate(&e0));
    CK(hipEventCreate(&e1));
    dim3 grid(256), blk(WG);
    hipLaunchKernelGGL((valu_probe<KIND, WG>), grid, blk, 0, 0, out, iters);
    CK(hipDeviceSynchronize());
    CK(hipEventRecord(e0));
    for (int r = 0; r < 3; r++)
        hipLaunchKernelGGL((valu_probe<KIND, WG>), grid, blk, 0, 0, out,
                           iters);
    CK(hipEventRecord(e1));
    CK(hipEventSynchronize(e1));
    float ms;
    CK(hipEventElapsedTime(&ms, e0, e1));
    double sec = ms / 1e3 / 3;
    /* grid 256 WGs = 1 per CU; WG/64 waves on 4 SIMDs */
    double waves_per_simd = (WG / 64) / 4.0;
    double instr = (double)iters * ops_per_iter * waves_per_simd;
    double cyc = sec * 2.4e9;
    printf("%-22s WG=%-3d waves/SIMD=%.0f  %6.2f cyc/instr/SIMD\n", name,
           WG, waves_per_simd, cyc / instr);
    CK(hipFree(out));
}

int main() {
    /* ops_per_iter = VALU instructions per loop iteration FROM OBJDUMP
     * (includes the register-rotation movs the compiler emits):
     * xor/bitop3/perm: 8 ops + 4 mov + 2 add = 14; mix: 15;
     * lshl_add_u64: 8 + 4 mov64 + 2 = 14; mad64: 4 mad + 8 xor + 4 mov64
     * = 16.  64-bit movs/adds may count double in the pipe — compare
     * KINDS, not absolute digits. */
    run<K_XOR, 256>("v_xor_b32", 14);
    run<K_XOR, 512>("v_xor_b32", 14);
    run<K_BITOP3, 256>("v_bitop3_b32", 14);
    run<K_BITOP3, 512>("v_bitop3_b32", 14);
    run<K_PERM, 256>("v_perm_b32", 14);
    run<K_PERM, 512>("v_perm_b32", 14);
    run<K_MIX, 256>("mix(bitop3/perm/xor/shl)", 15);
    run<K_MIX, 512>("mix(bitop3/perm/xor/shl)", 15);
    run<K_ADD64, 256>("v_lshl_add_u64", 14);
    run<K_ADD64, 512>("v_lshl_add_u64", 14);
    run<K_MAD64, 256>("mad_u64+xor", 16);
    run<K_MAD64, 512>("mad_u64+xor", 16);
    return 0;
}
