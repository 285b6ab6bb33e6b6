/* probe_l2chase.hip — can a consumer workgroup chase a producer's write
 * stream through L2 on gfx950, given correct agent-scope release/acquire?
 *
 * This is the go/no-go measurement for DESIGN.md §8 (round-2 candidate:
 * L2-chasing workgroup specialization).  One kernel, workgroup-
 * specialized: 128 producer WGs write 1024 blocks of 512 KiB (512 MiB
 * total, ≫ 32 MiB L2) and publish a per-block done flag with agent-scope
 * release; 128 consumer WGs spin (bounded) on their block's flag with
 * agent-scope acquire, then read the block.  If the acquire leaves the
 * producer's lines readable from cache, the consumer's reads mostly hit
 * L2 and the kernel's HBM FETCH stays near zero; if agent-scope acquire
 * invalidates the reader's L2 (or lines land remote), FETCH ≈ 512 MiB.
 *
 * Run both a "chase" pass (one kernel, interleaved) and a "cold" control
 * (producer kernel, then consumer kernel) and compare wall time; collect
 * FETCH_SIZE with rocprofv3 --pmc in a separate pass for the traffic
 * verdict (gfx950: double FETCH_SIZE before comparing).
 *
 * Build: hipcc -O3 -std=c++17 --offload-arch=gfx950 probe_l2chase.hip -o probe_l2chase
 */
#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstdio>

#define CK(x)                                                                \
    do {                                                                     \
        hipError_t e = (x);                                                  \
        if (e != hipSuccess) {                                               \
            printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__);  \
            return 1;                                                        \
        }                                                                    \
    } while (0)

constexpr int NPROD = 128, NCONS = 128;
constexpr int64_t BLK = 512 << 10;  /* bytes per block */
constexpr int64_t NB = 1024;        /* blocks: 512 MiB total */

__global__ void __launch_bounds__(256) chase_kernel(uint8_t *buf, int *done,
                                                    uint64_t *sink,
                                                    int *timeout_flag,
                                                    int mode) {
    /* mode 0: produce+consume (chase); 1: produce only; 2: consume only
     * (flags assumed pre-set) */
    const int wg = blockIdx.x;
    const bool producer = wg < NPROD;
    const int lane = threadIdx.x;
    if (producer) {
        if (mode == 2) return;
        for (int64_t b = wg; b < NB; b += NPROD) {
            uint8_t *base = buf + b * BLK;
            for (int64_t o = lane * 16; o < BLK; o += 256 * 16) {
                uint4 v = uint4{(uint32_t)(b + o), (uint32_t)o, 7u, 9u};
                *(uint4 *)(base + o) = v;
            }
            /* all stores visible before the publish */
            __builtin_amdgcn_s_waitcnt(0);
            __syncthreads();
            if (lane == 0)
                __hip_atomic_store(&done[b], 1, __ATOMIC_RELEASE,
                                   __HIP_MEMORY_SCOPE_AGENT);
        }
        return;
    }
    if (mode == 1) return;
    const int cw = wg - NPROD;
    uint64_t acc = 0;
    for (int64_t b = cw; b < NB; b += NCONS) {
        if (lane == 0) {
            int spins = 0;
            while (__hip_atomic_load(&done[b], __ATOMIC_ACQUIRE,
                                     __HIP_MEMORY_SCOPE_AGENT) == 0) {
                __builtin_amdgcn_s_sleep(8);
                if (++spins > (1 << 22)) {
                    *timeout_flag = 1;
                    break;
                }
            }
        }
        __syncthreads();
        if (__hip_atomic_load(timeout_flag, __ATOMIC_RELAXED,
                              __HIP_MEMORY_SCOPE_AGENT))
            return; /* bail everywhere on timeout */
        const uint8_t *base = buf + b * BLK;
        for (int64_t o = lane * 16; o < BLK; o += 256 * 16) {
            uint4 v = *(const uint4 *)(base + o);
            acc += v.x + v.y + v.z + v.w;
        }
    }
    sink[wg * 256 + lane] = acc;
}

static float run(uint8_t *buf, int *done, uint64_t *sink, int *tf, int mode,
                 const char *label) {
    hipEvent_t e0, e1;
    (void)hipEventCreate(&e0);
    (void)hipEventCreate(&e1);
    (void)hipMemset(tf, 0, 4);
    if (mode != 2) (void)hipMemset(done, 0, NB * 4);
    (void)hipEventRecord(e0);
    hipLaunchKernelGGL(chase_kernel, dim3(NPROD + NCONS), dim3(256), 0, 0,
                       buf, done, sink, tf, mode);
    (void)hipEventRecord(e1);
    (void)hipEventSynchronize(e1);
    float ms = 0;
    (void)hipEventElapsedTime(&ms, e0, e1);
    int host_tf = 0;
    (void)hipMemcpy(&host_tf, tf, 4, hipMemcpyDeviceToHost);
    printf("%s: %.3f ms (%.2f TB/s consumer-read)%s\n", label, ms,
           (double)NB * BLK / (ms * 1e-3) / 1e12,
           host_tf ? "  ** SPIN TIMEOUT **" : "");
    (void)hipEventDestroy(e0);
    (void)hipEventDestroy(e1);
    return ms;
}

int main() {
    uint8_t *buf;
    int *done, *tf;
    uint64_t *sink;
    CK(hipMalloc(&buf, NB * BLK));
    CK(hipMalloc(&done, NB * 4));
    CK(hipMalloc(&tf, 4));
    CK(hipMalloc(&sink, (NPROD + NCONS) * 256 * 8));
    /* warm */
    run(buf, done, sink, tf, 0, "warmup(chase)");
    run(buf, done, sink, tf, 0, "chase    (produce || consume, 512 MiB)");
    run(buf, done, sink, tf, 1, "produce-only");
    /* cold control: flags stay set from the produce-only pass; L2 holds at
     * most the tail 32 MiB of 512 MiB */
    run(buf, done, sink, tf, 2, "consume-cold (after full produce)");
    return 0;
}
