/* Kernel argument structs shared between ec_abi.cpp (host) and kernels.hip.
 * Internal to the library — the public boundary is include/minio_ec.h. */
#ifndef MEC_KERNELS_H
#define MEC_KERNELS_H

#include <hip/hip_runtime.h>
#include <cstdint>

/* Kernel-side geometry caps: MinIO erasure sets are 2..16 drives
 * (docs/distributed/DESIGN.md:44-52); the self-test sweeps d<=14, p<=8
 * (cmd/erasure-coding.go:152-156).  We allow headroom. */
#define MEC_KMAX_D 32
#define MEC_KMAX_E 8  /* output rows per launch; larger p uses multiple */
#define MEC_KMAX_TOTAL 40

struct GfMatmulArgs {
    const uint8_t *src;      /* batch base of source rows */
    uint8_t *dst;            /* batch base of destination rows */
    int64_t src_item_stride; /* bytes between batch items in src */
    int64_t dst_item_stride;
    int64_t row_stride;      /* bytes between shard rows (64-aligned) */
    int64_t shard_len;       /* valid bytes per shard */
    int d;                   /* number of source rows */
    uint8_t src_rows[MEC_KMAX_D];
    uint8_t dst_rows[MEC_KMAX_E];
    uint8_t mat[MEC_KMAX_E * MEC_KMAX_D]; /* row t: coefficients over src */
    /* bit-sliced path (r2): device pointer to E*d*2 u32 —
     * bs_masks[(t*d+k)*2 + w] = dword w of the 8x8 bit matrix of
     * mat[t][k] (byte b = rowmask(c,b): bit a set iff output bit b of
     * c*x depends on input bit a).  NULL -> ladder kernel. */
    const uint32_t *bs_masks;
};

/* mode: which shard rows this launch hashes.  Sums always land in the
 * fused n x (d+p) x digest layout so a data-only and a parity-only launch
 * together produce exactly the single-launch result. */
enum MecHashMode { MEC_HASH_ALL = 0, MEC_HASH_DATA = 1, MEC_HASH_PARITY = 2 };

struct HashArgs {
    const uint8_t *data;   /* data-shard rows, n*d*row_stride */
    const uint8_t *parity; /* parity rows, n*p*row_stride; NULL => simple
                              strided layout: chain i at data+i*row_stride */
    uint8_t *sums;         /* n * (d+p) * digest_size (fused layout) */
    int64_t row_stride;
    int64_t msg_len;       /* bytes hashed per chain (the padded shard) */
    int64_t n_chains;      /* chains in THIS launch: n*(d+p), n*d or n*p */
    int d, p;
    int mode;              /* MecHashMode */
    uint64_t key[4];       /* HighwayHash key (little-endian words) */
};

/* Specialized-encode args (constexpr-matrix kernels) */
struct GfEncArgs {
    const uint8_t *data;
    uint8_t *parity;
    int64_t row_stride;
    int64_t shard_len;
};

/* Fused single-pass encode+HighwayHash (fused.hip) */
struct FusedArgs {
    const uint8_t *data; /* n * d * row_stride */
    uint8_t *parity;     /* n * p * row_stride (output) */
    uint8_t *sums;       /* n * (d+p) * 32 (output) */
    int64_t row_stride;
    int64_t shard_len;
    int64_t n;
    uint64_t key[4];
    int probe; /* 0 normal; perf-isolation probes (results INVALID):
                  1 producers only, 2 consumers free-run (no poll),
                  3 both free-run */
};

struct ScatterArgs {
    const uint8_t *src;
    uint8_t *rows;
    int64_t block_len;
    int64_t S;
    int64_t row_stride;
    int64_t n;
    int d;
};

struct InterleaveArgs {
    const uint8_t *data;
    const uint8_t *parity;
    const uint8_t *sums;
    uint8_t *out;
    int64_t S;
    int64_t row_stride;
    int64_t n;
    int d, p;
};

extern "C" {
hipError_t mec_launch_scatter_rows(const ScatterArgs *args,
                                   hipStream_t stream);
hipError_t mec_launch_stream_interleave(const InterleaveArgs *args,
                                        hipStream_t stream);
hipError_t mec_launch_fused_encode_hh(int d, int p, const FusedArgs *args,
                                      hipStream_t stream);
hipError_t mec_launch_fused3_encode_hh(int d, int p, const FusedArgs *args,
                                       hipStream_t stream);
hipError_t mec_launch_fused2_encode_hh(int d, int p, const FusedArgs *args,
                                       hipStream_t stream);
hipError_t mec_launch_gf_matmul(const GfMatmulArgs *args, int n_dst, int n,
                                hipStream_t stream);
hipError_t mec_launch_gf_encode_spec(int d, int p, const GfEncArgs *args,
                                     int n, hipStream_t stream);
hipError_t mec_launch_hash(int algo, const HashArgs *args, hipStream_t stream);
}

#endif
