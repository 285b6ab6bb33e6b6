/* Round-2 design probes (measurement tooling, not product):
 *
 *  hh_reg:  the HighwayHash pair-lane chain run from REGISTERS (no memory)
 *           at the exact bench geometry — isolates pure chain issue/latency
 *           cost from load effects.  If ~= the real kernel's 0.47 ms, the
 *           hash is chain-bound and only a cheaper update helps; if much
 *           lower, the residual is the memory path.
 *  gf_mem:  the gf_encode access pattern (read d 16-B rows, write p rows,
 *           1 XOR) with the math removed — the memory-path ceiling of the
 *           12-stream layout.  Distance from the real kernel's 0.45 ms =
 *           the ladder's issue cost.
 *
 * Build: hipcc -O3 -std=c++17 --offload-arch=gfx950 probe.hip -o probe
 * Run (GPU box): ./probe
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

__global__ void __launch_bounds__(256) hh_reg_probe(uint64_t *sink,
                                                    int packets) {
    const int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    HH2 s;
#pragma unroll
    for (int j = 0; j < 2; j++) {
        s.v0[j] = 0x1111111111111111ull * (j + 1) + tid;
        s.v1[j] = 0x2222222222222222ull * (j + 1) ^ tid;
        s.mul0[j] = 0x3333333333333333ull * (j + 1);
        s.mul1[j] = 0x4444444444444444ull * (j + 1);
    }
    uint64_t w0 = tid, w1 = ~tid;
#pragma unroll 16
    for (int t = 0; t < packets; t++) {
        hh2_update(s, w0, w1);
        w0 += 0x9e3779b97f4a7c15ull; /* register-only packet stream */
        w1 ^= w0;
    }
    sink[tid] = s.v0[0] ^ s.v1[1] ^ s.mul0[0] ^ s.mul1[1];
}

/* 2 independent chains per lane, register-only: if per-chain cost drops
 * toward the issue floor, the product NC=2 failures were memory/VGPR-side
 * and a leaner NC=2 is worth building; if not, the chain stalls are not
 * coverable by lane-local ILP. */
__global__ void __launch_bounds__(256) hh_reg_probe2(uint64_t *sink,
                                                     int packets) {
    const int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    HH2 s[2];
#pragma unroll
    for (int u = 0; u < 2; u++)
#pragma unroll
        for (int j = 0; j < 2; j++) {
            s[u].v0[j] = 0x1111111111111111ull * (j + 1) + tid + u;
            s[u].v1[j] = 0x2222222222222222ull * (j + 1) ^ (tid - u);
            s[u].mul0[j] = 0x3333333333333333ull * (j + 1);
            s[u].mul1[j] = 0x4444444444444444ull * (j + 1);
        }
    uint64_t w0 = tid, w1 = ~tid;
#pragma unroll 8
    for (int t = 0; t < packets; t++) {
        hh2_update(s[0], w0, w1);
        hh2_update(s[1], w1, w0);
        w0 += 0x9e3779b97f4a7c15ull;
        w1 ^= w0;
    }
    sink[tid] = s[0].v0[0] ^ s[0].v1[1] ^ s[1].mul0[0] ^ s[1].mul1[1];
}

__global__ void __launch_bounds__(256) gf_mem_probe(const uint8_t *data,
                                                    uint8_t *parity,
                                                    int64_t stride,
                                                    int64_t shard_len, int d,
                                                    int p) {
    const int b = blockIdx.y;
    const int64_t cols = shard_len >> 4;
    const uint8_t *sbase = data + (int64_t)b * d * stride;
    uint8_t *obase = parity + (int64_t)b * p * stride;
    for (int64_t c = blockIdx.x * blockDim.x + threadIdx.x; c < cols;
         c += (int64_t)gridDim.x * blockDim.x) {
        const int64_t j = c << 4;
        uint4 acc = uint4{0, 0, 0, 0};
        for (int k = 0; k < d; k++) {
            uint4 v = *(const uint4 *)(sbase + (int64_t)k * stride + j);
            acc.x ^= v.x;
            acc.y ^= v.y;
            acc.z ^= v.z;
            acc.w ^= v.w;
        }
        for (int i = 0; i < p; i++) {
            typedef unsigned int v4u __attribute__((ext_vector_type(4)));
            v4u v = {acc.x + (unsigned)i, acc.y, acc.z, acc.w};
            __builtin_nontemporal_store(
                v, (v4u *)(obase + (int64_t)i * stride + j));
        }
    }
}

int main() {
    /* bench geometry: EC8+4, 1 MiB, batch 1024 */
    const int n = 1024, d = 8, p = 4;
    const int64_t S = 131072, stride = 131072;
    hipEvent_t e0, e1;
    CK(hipEventCreate(&e0));
    CK(hipEventCreate(&e1));
    float ms;

    /* ---- hh_reg: 24576 lanes x 4096 packets (the bench chain count) */
    uint64_t *sink;
    CK(hipMalloc(&sink, 24576 * 8));
    dim3 g1(24576 / 256), b1(256);
    hipLaunchKernelGGL(hh_reg_probe, g1, b1, 0, 0, sink, 4096);
    CK(hipDeviceSynchronize());
    CK(hipEventRecord(e0));
    for (int r = 0; r < 5; r++)
        hipLaunchKernelGGL(hh_reg_probe, g1, b1, 0, 0, sink, 4096);
    CK(hipEventRecord(e1));
    CK(hipEventSynchronize(e1));
    CK(hipEventElapsedTime(&ms, e0, e1));
    printf("hh_reg_probe (no memory): %.3f ms/launch (real hash kernel "
           "~0.47)\n", ms / 5);

    /* ---- hh_reg2: 12288 lanes x 2 chains x 4096 packets (same total) */
    dim3 g1b(12288 / 256);
    hipLaunchKernelGGL(hh_reg_probe2, g1b, b1, 0, 0, sink, 4096);
    CK(hipDeviceSynchronize());
    CK(hipEventRecord(e0));
    for (int r = 0; r < 5; r++)
        hipLaunchKernelGGL(hh_reg_probe2, g1b, b1, 0, 0, sink, 4096);
    CK(hipEventRecord(e1));
    CK(hipEventSynchronize(e1));
    CK(hipEventElapsedTime(&ms, e0, e1));
    printf("hh_reg_probe2 (2 chains/lane, half lanes): %.3f ms/launch\n",
           ms / 5);

    /* ---- gf_mem: the 12-stream layout with no ladder */
    uint8_t *data, *par;
    CK(hipMalloc(&data, (int64_t)n * d * stride));
    CK(hipMalloc(&par, (int64_t)n * p * stride));
    CK(hipMemset(data, 5, (int64_t)n * d * stride));
    int64_t cols = S >> 4;
    int64_t bx = (2048 * 4 + n - 1) / n;
    if (bx > (cols + 255) / 256) bx = (cols + 255) / 256;
    dim3 g2((uint32_t)bx, n), b2(256);
    hipLaunchKernelGGL(gf_mem_probe, g2, b2, 0, 0, data, par, stride, S, d, p);
    CK(hipDeviceSynchronize());
    CK(hipEventRecord(e0));
    for (int r = 0; r < 5; r++)
        hipLaunchKernelGGL(gf_mem_probe, g2, b2, 0, 0, data, par, stride, S,
                           d, p);
    CK(hipEventRecord(e1));
    CK(hipEventSynchronize(e1));
    CK(hipEventElapsedTime(&ms, e0, e1));
    double bytes = (double)n * (d + p) * S;
    printf("gf_mem_probe (no ladder): %.3f ms/launch = %.2f TB/s moved "
           "(real gf kernel ~0.45)\n", ms / 5, bytes / (ms / 5 * 1e-3) / 1e12);
    return 0;
}
