/* C-ABI implementation for the MI355X erasure+bitrot hot path.
 * Public surface: include/minio_ec.h (see the header for the Go-interface
 * mapping this library drops in behind).  Host driver logic mirrors the
 * reference's Go drivers:
 *   - Erasure.Encode loop          cmd/erasure-encode.go:76-108
 *   - streamingBitrotWriter        cmd/bitrot-streaming.go:44-75
 *   - streamingBitrotReader.ReadAt cmd/bitrot-streaming.go:161-200
 *   - Erasure.Decode / Heal        cmd/erasure-decode.go:239-364
 *   - writeDataBlocks              cmd/erasure-utils.go:42-105
 *   - bitrotVerify                 cmd/bitrot.go:164-216
 * All GF arithmetic and hashing runs in the HIP kernels (kernels.hip);
 * there is no CPU compute fallback — a missing GPU fails loudly
 * (MEC_ERR_NO_GPU / MEC_ERR_HIP).
 */
#include "../../include/minio_ec.h"
#include "gf_host.h"
#include "kernels.h"

#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstring>
#include <mutex>
#include <string>
#include <vector>

namespace {

thread_local std::string g_last_error;

void set_err(const char *what, hipError_t e) {
    g_last_error = std::string(what) + ": " + hipGetErrorString(e);
}

#define HIP_TRY(expr)                                                        \
    do {                                                                     \
        hipError_t _e = (expr);                                              \
        if (_e != hipSuccess) {                                              \
            set_err(#expr, _e);                                              \
            return MEC_ERR_HIP;                                              \
        }                                                                    \
    } while (0)

int64_t ceil_frac(int64_t num, int64_t den) {
    /* cmd/utils.go:689 (positive operands on this path) */
    if (den == 0) return 0;
    return (num + den - 1) / den;
}

/* magic HighwayHash key, cmd/bitrot.go:37, as 4 little-endian u64 words */
const uint8_t kMagicHHKey[32] = {
    0x4b, 0xe7, 0x34, 0xfa, 0x8e, 0x23, 0x8a, 0xcd, 0x26, 0x3e, 0x83,
    0xe6, 0xbb, 0x96, 0x85, 0x52, 0x04, 0x0f, 0x93, 0x5d, 0xa3, 0x9f,
    0x44, 0x14, 0x97, 0xe0, 0x9d, 0x13, 0x22, 0xde, 0x36, 0xa0};

int hash_size(int algo) {
    switch (algo) {
    case MEC_BITROT_SHA256:
    case MEC_BITROT_HIGHWAYHASH256:
    case MEC_BITROT_HIGHWAYHASH256S:
        return 32;
    case MEC_BITROT_BLAKE2B512:
        return 64;
    default:
        return 0;
    }
}

/* 8x8 bit matrix of the GF(2^8)/0x11D linear map x -> c*x, packed as two
 * dwords (byte b = rowmask: bit a set iff output bit b depends on input
 * bit a) — the r2 bit-sliced matmul's runtime-matrix form (kernels.h). */
void bs_pack_matrix(uint8_t c, uint32_t out[2]) {
    uint64_t m = 0;
    for (int b = 0; b < 8; b++) {
        uint8_t row = 0;
        for (int a = 0; a < 8; a++)
            if ((mec::gf_mul(c, (uint8_t)(1u << a)) >> b) & 1)
                row |= (uint8_t)(1u << a);
        m |= (uint64_t)row << (8 * b);
    }
    out[0] = (uint32_t)m;
    out[1] = (uint32_t)(m >> 32);
}

} // namespace

struct mec_ctx {
    int device = 0;
    int d = 0, p = 0;
    int64_t block_size = 0;
    int64_t S = 0;      /* shard size  = ceil(block_size / d) */
    int64_t stride = 0; /* device row stride = align64(S) */
    std::vector<uint8_t> enc_matrix; /* (d+p) x d */
    hipStream_t stream = nullptr;
    hipStream_t stream2 = nullptr; /* hash(data) overlap lane */
    hipEvent_t ev_start = nullptr, ev_stop = nullptr;
    hipEvent_t ev_gf = nullptr, ev_h = nullptr, ev_fork = nullptr;
    hipEvent_t ev_pipe[2] = {nullptr, nullptr}; /* hash-done per buffer slot */
    int pipe_idx = 0;
    std::mutex mu;

    /* grow-only scratch (device + pinned host) for host-pointer calls */
    void *dev_a = nullptr, *dev_b = nullptr, *dev_c = nullptr;
    void *dev_d = nullptr; /* stream-assembly output */
    void *dev_m = nullptr;  /* bit-matrix masks for the BS matmul (r2) */
    std::vector<uint32_t> m_cached; /* host copy of what dev_m holds — the
        chunked reconstruct path re-sends identical masks per chunk, and
        re-uploading would need a stream sync that serializes the
        overlapped ingest */
    size_t cap_a = 0, cap_b = 0, cap_c = 0, cap_d = 0, cap_m = 0;
    void *pin = nullptr;
    size_t cap_pin = 0;

    mec_status ensure(void **buf, size_t *cap, size_t need) {
        if (*cap >= need) return MEC_OK;
        if (*buf) (void)hipFree(*buf);
        *buf = nullptr;
        *cap = 0;
        HIP_TRY(hipMalloc(buf, need));
        *cap = need;
        return MEC_OK;
    }
    mec_status ensure_pin(size_t need) {
        if (cap_pin >= need) return MEC_OK;
        if (pin) (void)hipHostFree(pin);
        pin = nullptr;
        cap_pin = 0;
        HIP_TRY(hipHostMalloc(&pin, need));
        cap_pin = need;
        return MEC_OK;
    }
};

extern "C" {

int mec_version(void) { return 10000; /* 1.0.0 */ }

const char *mec_last_error(void) { return g_last_error.c_str(); }

int mec_device_count(void) {
    int n = 0;
    if (hipGetDeviceCount(&n) != hipSuccess) return 0;
    return n;
}

/* ---- shard-size math (exact mirrors) ---------------------------------- */

int64_t mec_shard_size(int64_t block_size, int d) {
    /* cmd/erasure-coding.go:116-118 */
    return ceil_frac(block_size, d);
}

int64_t mec_shard_file_size(int64_t block_size, int d, int64_t total_length) {
    /* cmd/erasure-coding.go:121-132 */
    if (total_length == 0) return 0;
    if (total_length == -1) return -1;
    int64_t num = total_length / block_size;
    int64_t last = total_length % block_size;
    int64_t last_shard = ceil_frac(last, d);
    return num * mec_shard_size(block_size, d) + last_shard;
}

int64_t mec_shard_file_offset(int64_t block_size, int d, int64_t start_offset,
                              int64_t length, int64_t total_length) {
    /* cmd/erasure-coding.go:135-141 */
    int64_t shard_size = mec_shard_size(block_size, d);
    int64_t shard_file_size = mec_shard_file_size(block_size, d, total_length);
    int64_t end_shard = (start_offset + length) / block_size;
    int64_t till = end_shard * shard_size + shard_size;
    if (till > shard_file_size) till = shard_file_size;
    return till;
}

int64_t mec_bitrot_shard_file_size(int64_t size, int64_t shard_size,
                                   int algo) {
    /* cmd/bitrot.go:156-161 */
    if (algo != MEC_BITROT_HIGHWAYHASH256S) return size;
    return ceil_frac(size, shard_size) * hash_size(algo) + size;
}

int64_t mec_shard_stride(int64_t block_size, int d) {
    return (mec_shard_size(block_size, d) + 63) & ~int64_t(63);
}

/* ---- context ----------------------------------------------------------- */

mec_status mec_ctx_create(int d, int p, int64_t block_size, int device,
                          mec_ctx **out) {
    /* sanity mirrors NewErasure, cmd/erasure-coding.go:43-50 */
    if (d <= 0 || p < 0) return MEC_ERR_INV_SHARD_NUM;
    if (d + p > 256) return MEC_ERR_MAX_SHARD_NUM;
    if (d > MEC_KMAX_D || d + p > MEC_KMAX_TOTAL) return MEC_ERR_MAX_SHARD_NUM;
    if (block_size <= 0) return MEC_ERR_INVALID_ARG;
    int ndev = mec_device_count();
    if (ndev == 0) {
        g_last_error = "no HIP device visible (MI355X required; this library "
                       "has no CPU fallback)";
        return MEC_ERR_NO_GPU;
    }
    if (device < 0 || device >= ndev) return MEC_ERR_INVALID_ARG;

    auto ctx = new mec_ctx();
    ctx->device = device;
    ctx->d = d;
    ctx->p = p;
    ctx->block_size = block_size;
    ctx->S = mec_shard_size(block_size, d);
    ctx->stride = mec_shard_stride(block_size, d);
    ctx->enc_matrix.resize((size_t)(d + p) * d);
    if (!mec::build_encode_matrix(d, p, ctx->enc_matrix.data())) {
        delete ctx;
        return MEC_ERR_INVALID_ARG;
    }
    hipError_t e = hipSetDevice(device);
    /* MEC_CU_SPLIT=H: partition the chip so the latency-bound hash lane
     * (stream2) gets H CUs EXCLUSIVELY and the memory-bound GF lane
     * (stream) the rest.  Overlapped gf/hash waves sharing a SIMD stretch
     * the hash ~1.5x (issue-slot contention); an exclusive partition
     * removes that.  CUs are spread round-robin so both lanes touch every
     * XCD's L2/HBM channels.  0/unset = classic shared streams. */
    int cu_split = 0;
    if (const char *s = getenv("MEC_CU_SPLIT")) cu_split = atoi(s);
    if (cu_split > 0) {
        int ncu = 0;
        (void)hipDeviceGetAttribute(&ncu,
                                    hipDeviceAttributeMultiprocessorCount,
                                    device);
        if (ncu <= 0 || cu_split >= ncu) cu_split = 0;
        if (cu_split > 0) {
            uint32_t mh[8] = {0}, mg[8] = {0};
            int given = 0;
            for (int i = 0; i < ncu && i < 256; i++) {
                /* spread hash CUs evenly across the index space (and so
                 * across XCDs, which interleave in the physical mapping) */
                bool h = ((int64_t)(i + 1) * cu_split / ncu) >
                         ((int64_t)i * cu_split / ncu);
                if (h) {
                    mh[i >> 5] |= 1u << (i & 31);
                    given++;
                } else {
                    mg[i >> 5] |= 1u << (i & 31);
                }
            }
            (void)given;
            /* note: gfx950 silently IGNORES CU masks (measured flat in the
             * r1 sweep) — the knob is a recorded no-op there.  If masked
             * stream creation fails, fall back to plain streams: this is a
             * tuning knob, never a correctness dependency. */
            if (e == hipSuccess) {
                hipError_t em =
                    hipExtStreamCreateWithCUMask(&ctx->stream, 8, mg);
                if (em == hipSuccess)
                    em = hipExtStreamCreateWithCUMask(&ctx->stream2, 8, mh);
                if (em != hipSuccess) {
                    if (ctx->stream) (void)hipStreamDestroy(ctx->stream);
                    ctx->stream = nullptr;
                    ctx->stream2 = nullptr;
                    cu_split = 0; /* plain-stream fallback below */
                }
            }
        }
    }
    if (cu_split <= 0) {
        if (e == hipSuccess) e = hipStreamCreate(&ctx->stream);
        if (e == hipSuccess) e = hipStreamCreate(&ctx->stream2);
    }
    if (e == hipSuccess) e = hipEventCreate(&ctx->ev_start);
    if (e == hipSuccess) e = hipEventCreate(&ctx->ev_stop);
    if (e == hipSuccess) e = hipEventCreateWithFlags(&ctx->ev_gf, hipEventDisableTiming);
    if (e == hipSuccess) e = hipEventCreateWithFlags(&ctx->ev_h, hipEventDisableTiming);
    if (e == hipSuccess) e = hipEventCreateWithFlags(&ctx->ev_fork, hipEventDisableTiming);
    if (e == hipSuccess) e = hipEventCreateWithFlags(&ctx->ev_pipe[0], hipEventDisableTiming);
    if (e == hipSuccess) e = hipEventCreateWithFlags(&ctx->ev_pipe[1], hipEventDisableTiming);
    if (e != hipSuccess) {
        set_err("ctx_create", e);
        delete ctx;
        return MEC_ERR_HIP;
    }
    *out = ctx;
    return MEC_OK;
}

int mec_ctx_d(mec_ctx *ctx) { return ctx->d; }
int mec_ctx_p(mec_ctx *ctx) { return ctx->p; }
int64_t mec_ctx_block_size(mec_ctx *ctx) { return ctx->block_size; }

void mec_ctx_destroy(mec_ctx *ctx) {
    if (!ctx) return;
    (void)hipSetDevice(ctx->device);
    if (ctx->dev_a) (void)hipFree(ctx->dev_a);
    if (ctx->dev_b) (void)hipFree(ctx->dev_b);
    if (ctx->dev_c) (void)hipFree(ctx->dev_c);
    if (ctx->dev_d) (void)hipFree(ctx->dev_d);
    if (ctx->dev_m) (void)hipFree(ctx->dev_m);
    if (ctx->pin) (void)hipHostFree(ctx->pin);
    if (ctx->ev_start) (void)hipEventDestroy(ctx->ev_start);
    if (ctx->ev_stop) (void)hipEventDestroy(ctx->ev_stop);
    if (ctx->ev_gf) (void)hipEventDestroy(ctx->ev_gf);
    if (ctx->ev_h) (void)hipEventDestroy(ctx->ev_h);
    if (ctx->ev_fork) (void)hipEventDestroy(ctx->ev_fork);
    if (ctx->ev_pipe[0]) (void)hipEventDestroy(ctx->ev_pipe[0]);
    if (ctx->ev_pipe[1]) (void)hipEventDestroy(ctx->ev_pipe[1]);
    if (ctx->stream2) (void)hipStreamDestroy(ctx->stream2);
    if (ctx->stream) (void)hipStreamDestroy(ctx->stream);
    delete ctx;
}

/* ---- batch encode ------------------------------------------------------ */

static mec_status encode_dev_locked(mec_ctx *ctx, int n, const void *data_dev,
                                    int64_t block_len, void *parity_dev,
                                    int algo, void *sums_dev) {
    if (n <= 0 || block_len <= 0 || block_len > ctx->block_size)
        return MEC_ERR_INVALID_ARG;
    HIP_TRY(hipSetDevice(ctx->device));

    const int d = ctx->d, p = ctx->p;
    /* per-call shard size mirrors Split: ceil(block_len/d)
     * (cmd/erasure-coding.go:81 + :117); rows stay at ctx->stride */
    const int64_t S_call = ceil_frac(block_len, d);
    /* single-pass fused encode+hash for HighwayHash geometries with a
     * compiled specialization (1.5 B HBM traffic per input byte vs 3.0
     * for the kernel pair) */
    if (sums_dev != nullptr && (algo == MEC_BITROT_HIGHWAYHASH256 ||
                                algo == MEC_BITROT_HIGHWAYHASH256S)) {
        FusedArgs fa{};
        fa.data = (const uint8_t *)data_dev;
        fa.parity = (uint8_t *)parity_dev;
        fa.sums = (uint8_t *)sums_dev;
        fa.row_stride = ctx->stride;
        fa.shard_len = S_call;
        fa.n = n;
        memcpy(fa.key, kMagicHHKey, 32);
        hipError_t he = mec_launch_fused3_encode_hh(d, p, &fa, ctx->stream);
        if (he == hipSuccess) return MEC_OK;
        if (he != hipErrorNotSupported) {
            set_err("fused3_encode_hh", he);
            return MEC_ERR_HIP;
        }
        he = mec_launch_fused2_encode_hh(d, p, &fa, ctx->stream);
        if (he == hipSuccess) return MEC_OK;
        if (he != hipErrorNotSupported) {
            set_err("fused2_encode_hh", he);
            return MEC_ERR_HIP;
        }
        he = mec_launch_fused_encode_hh(d, p, &fa, ctx->stream);
        if (he == hipSuccess) return MEC_OK;
        if (he != hipErrorNotSupported) {
            set_err("fused_encode_hh", he);
            return MEC_ERR_HIP;
        }
    }
    /* specialized straight-line kernel for common geometries */
    {
        GfEncArgs ea{};
        ea.data = (const uint8_t *)data_dev;
        ea.parity = (uint8_t *)parity_dev;
        ea.row_stride = ctx->stride;
        ea.shard_len = S_call;
        hipError_t he =
            mec_launch_gf_encode_spec(d, p, &ea, n, ctx->stream);
        if (he == hipSuccess) goto gf_done;
        if (he != hipErrorNotSupported) {
            set_err("gf_encode_spec", he);
            return MEC_ERR_HIP;
        }
    }
    /* generic fallback: GF parity rows, in groups of <= MEC_KMAX_E */
    for (int i0 = 0; i0 < p; i0 += MEC_KMAX_E) {
        int e = p - i0 > MEC_KMAX_E ? MEC_KMAX_E : p - i0;
        GfMatmulArgs a{};
        a.src = (const uint8_t *)data_dev;
        a.dst = (uint8_t *)parity_dev;
        a.src_item_stride = (int64_t)d * ctx->stride;
        a.dst_item_stride = (int64_t)p * ctx->stride;
        a.row_stride = ctx->stride;
        a.shard_len = S_call;
        a.d = d;
        for (int k = 0; k < d; k++) a.src_rows[k] = (uint8_t)k;
        for (int i = 0; i < e; i++) a.dst_rows[i] = (uint8_t)(i0 + i);
        for (int i = 0; i < e; i++)
            for (int k = 0; k < d; k++)
                a.mat[i * MEC_KMAX_D + k] =
                    ctx->enc_matrix[(size_t)(d + i0 + i) * d + k];
        {
            uint32_t masks[MEC_KMAX_E * MEC_KMAX_D * 2];
            for (int i = 0; i < e; i++)
                for (int k = 0; k < d; k++)
                    bs_pack_matrix(a.mat[i * MEC_KMAX_D + k],
                                   &masks[((size_t)i * d + k) * 2]);
            size_t mw = (size_t)e * d * 2;
            size_t mb = mw * sizeof(uint32_t);
            if (ctx->m_cached.size() != mw ||
                memcmp(ctx->m_cached.data(), masks, mb) != 0) {
                mec_status st2;
                if ((st2 = ctx->ensure(&ctx->dev_m, &ctx->cap_m, mb)) !=
                    MEC_OK)
                    return st2;
                /* prior launches may still read dev_m */
                HIP_TRY(hipStreamSynchronize(ctx->stream));
                HIP_TRY(hipMemcpy(ctx->dev_m, masks, mb,
                                  hipMemcpyHostToDevice));
                ctx->m_cached.assign(masks, masks + mw);
            }
            a.bs_masks = (const uint32_t *)ctx->dev_m;
        }
        HIP_TRY(mec_launch_gf_matmul(&a, e, n, ctx->stream));
    }
gf_done:
    if (sums_dev != nullptr) {
        if (!hash_size(algo)) return MEC_ERR_INVALID_ARG;
        /* single launch with every chain in flight: the hash is
         * latency-bound per chain, so splitting it (or overlapping it with
         * the GF kernel's 8-waves/SIMD occupancy) measurably regresses —
         * measured: overlapped split 2.04 ms/step vs sequential
         * single-launch 1.12 ms/step at EC8+4/1MiB/1024. */
        HashArgs h{};
        h.data = (const uint8_t *)data_dev;
        h.parity = (const uint8_t *)parity_dev;
        h.sums = (uint8_t *)sums_dev;
        h.row_stride = ctx->stride;
        h.msg_len = S_call;
        h.d = d;
        h.p = p;
        h.mode = MEC_HASH_ALL;
        h.n_chains = (int64_t)n * (d + p);
        memcpy(h.key, kMagicHHKey, 32);
        HIP_TRY(mec_launch_hash(algo, &h, ctx->stream));
    }
    return MEC_OK;
}

/* Pipelined encode: batch t's hash (on the context's second stream)
 * overlaps batch t+1's GF.  CONTRACT: the caller alternates between TWO
 * parity/sums buffer sets in strict round-robin and calls mec_pipe_sync
 * before reading any results.  Results are identical to
 * mec_encode_batch_dev per call; only cross-call scheduling differs
 * (independent batches — MinIO objects — pipeline the same way). */
mec_status mec_encode_batch_dev_pipe(mec_ctx *ctx, int n,
                                     const void *data_dev, int64_t block_len,
                                     void *parity_dev, int algo,
                                     void *sums_dev) {
    if (!ctx) return MEC_ERR_INVALID_ARG;
    std::lock_guard<std::mutex> lk(ctx->mu);
    if (n <= 0 || block_len <= 0 || block_len > ctx->block_size ||
        sums_dev == nullptr || !hash_size(algo))
        return MEC_ERR_INVALID_ARG;
    HIP_TRY(hipSetDevice(ctx->device));
    const int d = ctx->d, p = ctx->p;
    const int64_t S_call = ceil_frac(block_len, d);
    const int idx = ctx->pipe_idx & 1;
    ctx->pipe_idx ^= 1;
    /* this buffer slot's previous hash must be drained before GF rewrites
     * the parity buffer */
    HIP_TRY(hipStreamWaitEvent(ctx->stream, ctx->ev_pipe[idx], 0));
    /* single-pass fused kernel (r2 default): one launch does GF + hash at
     * 1.5 B/input byte — cross-batch pipelining degenerates to back-to-
     * back fused launches, which is what we want (the fused kernel is
     * memory-bound; overlapping two of them cannot beat sequential) */
    if (algo == MEC_BITROT_HIGHWAYHASH256 ||
        algo == MEC_BITROT_HIGHWAYHASH256S) {
        FusedArgs fa{};
        fa.data = (const uint8_t *)data_dev;
        fa.parity = (uint8_t *)parity_dev;
        fa.sums = (uint8_t *)sums_dev;
        fa.row_stride = ctx->stride;
        fa.shard_len = S_call;
        fa.n = n;
        memcpy(fa.key, kMagicHHKey, 32);
        hipError_t he = mec_launch_fused3_encode_hh(d, p, &fa, ctx->stream);
        if (he == hipSuccess) {
            HIP_TRY(hipEventRecord(ctx->ev_pipe[idx], ctx->stream));
            return MEC_OK;
        }
        if (he != hipErrorNotSupported) {
            set_err("fused3_encode_hh", he);
            return MEC_ERR_HIP;
        }
    }
    {
        GfEncArgs ea{};
        ea.data = (const uint8_t *)data_dev;
        ea.parity = (uint8_t *)parity_dev;
        ea.row_stride = ctx->stride;
        ea.shard_len = S_call;
        hipError_t he = mec_launch_gf_encode_spec(d, p, &ea, n, ctx->stream);
        if (he == hipErrorNotSupported)
            return MEC_ERR_INVALID_ARG; /* pipe mode needs a specialization */
        if (he != hipSuccess) {
            set_err("gf_encode_spec", he);
            return MEC_ERR_HIP;
        }
    }
    HIP_TRY(hipEventRecord(ctx->ev_gf, ctx->stream));
    HIP_TRY(hipStreamWaitEvent(ctx->stream2, ctx->ev_gf, 0));
    {
        HashArgs h{};
        h.data = (const uint8_t *)data_dev;
        h.parity = (const uint8_t *)parity_dev;
        h.sums = (uint8_t *)sums_dev;
        h.row_stride = ctx->stride;
        h.msg_len = S_call;
        h.d = d;
        h.p = p;
        h.mode = MEC_HASH_ALL;
        h.n_chains = (int64_t)n * (d + p);
        memcpy(h.key, kMagicHHKey, 32);
        HIP_TRY(mec_launch_hash(algo, &h, ctx->stream2));
    }
    HIP_TRY(hipEventRecord(ctx->ev_pipe[idx], ctx->stream2));
    return MEC_OK;
}

mec_status mec_pipe_sync(mec_ctx *ctx) {
    if (!ctx) return MEC_ERR_INVALID_ARG;
    std::lock_guard<std::mutex> lk(ctx->mu);
    HIP_TRY(hipSetDevice(ctx->device));
    HIP_TRY(hipStreamSynchronize(ctx->stream));
    HIP_TRY(hipStreamSynchronize(ctx->stream2));
    return MEC_OK;
}

mec_status mec_encode_batch_dev_async(mec_ctx *ctx, int n,
                                      const void *data_dev, int64_t block_len,
                                      void *parity_dev, int algo,
                                      void *sums_dev) {
    if (!ctx) return MEC_ERR_INVALID_ARG;
    std::lock_guard<std::mutex> lk(ctx->mu);
    return encode_dev_locked(ctx, n, data_dev, block_len, parity_dev, algo,
                             sums_dev);
}

mec_status mec_encode_batch_dev(mec_ctx *ctx, int n, const void *data_dev,
                                int64_t block_len, void *parity_dev, int algo,
                                void *sums_dev) {
    if (!ctx) return MEC_ERR_INVALID_ARG;
    std::lock_guard<std::mutex> lk(ctx->mu);
    mec_status s = encode_dev_locked(ctx, n, data_dev, block_len, parity_dev,
                                     algo, sums_dev);
    if (s != MEC_OK) return s;
    HIP_TRY(hipStreamSynchronize(ctx->stream));
    return MEC_OK;
}

mec_status mec_encode_batch(mec_ctx *ctx, int n, const uint8_t *data,
                            int64_t block_len, uint8_t *parity, int algo,
                            uint8_t *sums) {
    if (!ctx) return MEC_ERR_INVALID_ARG;
    if (n <= 0 || block_len <= 0 || block_len > ctx->block_size)
        return MEC_ERR_INVALID_ARG;
    std::lock_guard<std::mutex> lk(ctx->mu);
    HIP_TRY(hipSetDevice(ctx->device));
    const int d = ctx->d, p = ctx->p;
    const int real_hsz = sums ? hash_size(algo) : 0;
    const int64_t S = ceil_frac(block_len, d), stride = ctx->stride;
    size_t data_bytes = (size_t)n * d * stride;
    size_t par_bytes = (size_t)n * p * stride;
    size_t sum_bytes = (size_t)n * (d + p) * (size_t)real_hsz;
    mec_status st;
    if ((st = ctx->ensure(&ctx->dev_a, &ctx->cap_a, data_bytes)) != MEC_OK)
        return st;
    if ((st = ctx->ensure(&ctx->dev_b, &ctx->cap_b, par_bytes)) != MEC_OK)
        return st;
    if (sums &&
        (st = ctx->ensure(&ctx->dev_c, &ctx->cap_c, sum_bytes)) != MEC_OK)
        return st;
    if ((st = ctx->ensure_pin(data_bytes > par_bytes + sum_bytes
                                  ? data_bytes
                                  : par_bytes + sum_bytes)) != MEC_OK)
        return st;

    /* scatter packed object bytes into the padded strided shard layout
     * (Split semantics, cmd/erasure-coding.go:81: shard k gets bytes
     * [k*S, (k+1)*S) of the block, zero-padded) */
    uint8_t *pinb = (uint8_t *)ctx->pin;
    for (int b = 0; b < n; b++) {
        const uint8_t *blk = data + (size_t)b * block_len;
        for (int k = 0; k < d; k++) {
            uint8_t *dst = pinb + ((size_t)b * d + k) * stride;
            int64_t have = block_len - (int64_t)k * S;
            if (have < 0) have = 0;
            if (have > S) have = S;
            if (have) memcpy(dst, blk + (int64_t)k * S, (size_t)have);
            if (have < S) memset(dst + have, 0, (size_t)(S - have));
        }
    }
    HIP_TRY(hipMemcpyAsync(ctx->dev_a, pinb, data_bytes, hipMemcpyHostToDevice,
                           ctx->stream));
    st = encode_dev_locked(ctx, n, ctx->dev_a, block_len, ctx->dev_b, algo,
                           sums ? ctx->dev_c : nullptr);
    if (st != MEC_OK) return st;
    HIP_TRY(hipMemcpyAsync(pinb, ctx->dev_b, par_bytes, hipMemcpyDeviceToHost,
                           ctx->stream));
    if (sums)
        HIP_TRY(hipMemcpyAsync(pinb + par_bytes, ctx->dev_c, sum_bytes,
                               hipMemcpyDeviceToHost, ctx->stream));
    HIP_TRY(hipStreamSynchronize(ctx->stream));
    /* gather parity (packed, no stride padding) */
    for (int b = 0; b < n; b++)
        for (int i = 0; i < p; i++)
            memcpy(parity + ((size_t)b * p + i) * S,
                   pinb + ((size_t)b * p + i) * stride, (size_t)S);
    if (sums) memcpy(sums, pinb + par_bytes, sum_bytes);
    return MEC_OK;
}

/* ---- batch reconstruct ------------------------------------------------- */

static mec_status reconstruct_dev_locked(mec_ctx *ctx, int n,
                                         void *shards_dev,
                                         const uint8_t *present,
                                         int64_t shard_len, int data_only) {
    if (n <= 0 || shard_len <= 0 || shard_len > ctx->stride)
        return MEC_ERR_INVALID_ARG;
    HIP_TRY(hipSetDevice(ctx->device));
    const int d = ctx->d, p = ctx->p, total = d + p;
    int src_idx[mec::kMaxShards], dst_idx[mec::kMaxShards], n_dst = 0;
    std::vector<uint8_t> dec((size_t)total * d);
    /* all present -> nothing to do (Reconstruct fast path) */
    int n_present = 0;
    for (int i = 0; i < total; i++)
        if (present[i]) n_present++;
    if (n_present == total) return MEC_OK;
    if (n_present < d) return MEC_ERR_TOO_FEW_SHARDS;
    if (!mec::build_decode_plan(ctx->enc_matrix.data(), d, p, present,
                                data_only, src_idx, dst_idx, &n_dst,
                                dec.data()))
        return MEC_ERR_TOO_FEW_SHARDS;
    if (n_dst == 0) return MEC_OK;

    for (int t0 = 0; t0 < n_dst; t0 += MEC_KMAX_E) {
        int e = n_dst - t0 > MEC_KMAX_E ? MEC_KMAX_E : n_dst - t0;
        GfMatmulArgs a{};
        a.src = (const uint8_t *)shards_dev;
        a.dst = (uint8_t *)shards_dev;
        a.src_item_stride = (int64_t)total * ctx->stride;
        a.dst_item_stride = (int64_t)total * ctx->stride;
        a.row_stride = ctx->stride;
        a.shard_len = shard_len;
        a.d = d;
        for (int k = 0; k < d; k++) a.src_rows[k] = (uint8_t)src_idx[k];
        for (int i = 0; i < e; i++) a.dst_rows[i] = (uint8_t)dst_idx[t0 + i];
        for (int i = 0; i < e; i++)
            for (int k = 0; k < d; k++)
                a.mat[i * MEC_KMAX_D + k] = dec[(size_t)(t0 + i) * d + k];
        /* bit-sliced path: pack the 8x8 bit matrices and upload (a few
         * hundred bytes; synchronous — tiny next to the kernel) */
        {
            uint32_t masks[MEC_KMAX_E * MEC_KMAX_D * 2];
            for (int i = 0; i < e; i++)
                for (int k = 0; k < d; k++)
                    bs_pack_matrix(a.mat[i * MEC_KMAX_D + k],
                                   &masks[((size_t)i * d + k) * 2]);
            size_t mw = (size_t)e * d * 2;
            size_t mb = mw * sizeof(uint32_t);
            if (ctx->m_cached.size() != mw ||
                memcmp(ctx->m_cached.data(), masks, mb) != 0) {
                mec_status st2;
                if ((st2 = ctx->ensure(&ctx->dev_m, &ctx->cap_m, mb)) !=
                    MEC_OK)
                    return st2;
                /* prior launches may still read dev_m */
                HIP_TRY(hipStreamSynchronize(ctx->stream));
                HIP_TRY(hipMemcpy(ctx->dev_m, masks, mb,
                                  hipMemcpyHostToDevice));
                ctx->m_cached.assign(masks, masks + mw);
            }
            a.bs_masks = (const uint32_t *)ctx->dev_m;
        }
        HIP_TRY(mec_launch_gf_matmul(&a, e, n, ctx->stream));
    }
    return MEC_OK;
}

mec_status mec_reconstruct_batch_dev_async(mec_ctx *ctx, int n,
                                           void *shards_dev,
                                           const uint8_t *present,
                                           int64_t shard_len, int data_only) {
    if (!ctx) return MEC_ERR_INVALID_ARG;
    std::lock_guard<std::mutex> lk(ctx->mu);
    return reconstruct_dev_locked(ctx, n, shards_dev, present, shard_len,
                                  data_only);
}

mec_status mec_reconstruct_batch_dev(mec_ctx *ctx, int n, void *shards_dev,
                                     const uint8_t *present,
                                     int64_t shard_len, int data_only) {
    if (!ctx) return MEC_ERR_INVALID_ARG;
    std::lock_guard<std::mutex> lk(ctx->mu);
    mec_status s = reconstruct_dev_locked(ctx, n, shards_dev, present,
                                          shard_len, data_only);
    if (s != MEC_OK) return s;
    HIP_TRY(hipStreamSynchronize(ctx->stream));
    return MEC_OK;
}

mec_status mec_reconstruct_batch(mec_ctx *ctx, int n, uint8_t *shards,
                                 const uint8_t *present, int64_t shard_len,
                                 int data_only) {
    if (!ctx) return MEC_ERR_INVALID_ARG;
    if (n <= 0 || shard_len <= 0 || shard_len > ctx->stride)
        return MEC_ERR_INVALID_ARG;
    std::lock_guard<std::mutex> lk(ctx->mu);
    HIP_TRY(hipSetDevice(ctx->device));
    const int total = ctx->d + ctx->p;
    const int64_t stride = ctx->stride;
    size_t bytes = (size_t)n * total * stride;
    mec_status st;
    if ((st = ctx->ensure(&ctx->dev_a, &ctx->cap_a, bytes)) != MEC_OK)
        return st;
    if ((st = ctx->ensure_pin(bytes)) != MEC_OK) return st;
    uint8_t *pinb = (uint8_t *)ctx->pin;
    /* overlapped ingest (parallelReader-style): the batch is
     * fed to the GPU reconstruct queue in chunks — chunk c+1's host pack
     * and PCIe upload (stream2) run while chunk c's decode kernels run
     * (stream); the event handshake keeps kernel c after upload c. */
    const int64_t chunk = (n > 128) ? ((n + 3) / 4) : n;
    for (int64_t b0 = 0; b0 < n; b0 += chunk) {
        const int64_t nc = (b0 + chunk <= n) ? chunk : (n - b0);
        uint8_t *pc = pinb + (size_t)b0 * total * stride;
        for (int64_t b = 0; b < nc; b++)
            for (int s = 0; s < total; s++)
                memcpy(pc + ((size_t)b * total + s) * stride,
                       shards + ((size_t)(b0 + b) * total + s) * shard_len,
                       (size_t)shard_len);
        HIP_TRY(hipMemcpyAsync((uint8_t *)ctx->dev_a +
                                   (size_t)b0 * total * stride,
                               pc, (size_t)nc * total * stride,
                               hipMemcpyHostToDevice, ctx->stream2));
        HIP_TRY(hipEventRecord(ctx->ev_h, ctx->stream2));
        HIP_TRY(hipStreamWaitEvent(ctx->stream, ctx->ev_h, 0));
        st = reconstruct_dev_locked(
            ctx, (int)nc,
            (uint8_t *)ctx->dev_a + (size_t)b0 * total * stride, present,
            shard_len, data_only);
        if (st != MEC_OK) return st;
        HIP_TRY(hipMemcpyAsync(pc,
                               (uint8_t *)ctx->dev_a +
                                   (size_t)b0 * total * stride,
                               (size_t)nc * total * stride,
                               hipMemcpyDeviceToHost, ctx->stream));
    }
    HIP_TRY(hipStreamSynchronize(ctx->stream));
    for (int b = 0; b < n; b++)
        for (int s = 0; s < total; s++) {
            if (present[s]) continue; /* only missing rows were rebuilt */
            if (data_only && s >= ctx->d) continue;
            memcpy(shards + ((size_t)b * total + s) * shard_len,
                   pinb + ((size_t)b * total + s) * stride,
                   (size_t)shard_len);
        }
    return MEC_OK;
}

/* ---- batch hashing / verification -------------------------------------- */

mec_status mec_bitrot_sum_batch_dev(mec_ctx *ctx, int algo, int n,
                                    const void *msgs_dev, int64_t msg_len,
                                    int64_t msg_stride, void *sums_dev) {
    if (!ctx) return MEC_ERR_INVALID_ARG;
    if (n <= 0 || msg_len < 0 || !hash_size(algo)) return MEC_ERR_INVALID_ARG;
    std::lock_guard<std::mutex> lk(ctx->mu);
    HIP_TRY(hipSetDevice(ctx->device));
    HashArgs h{};
    h.data = (const uint8_t *)msgs_dev;
    h.parity = nullptr;
    h.sums = (uint8_t *)sums_dev;
    h.row_stride = msg_stride;
    h.msg_len = msg_len;
    h.n_chains = n;
    memcpy(h.key, kMagicHHKey, 32);
    HIP_TRY(mec_launch_hash(algo, &h, ctx->stream));
    HIP_TRY(hipStreamSynchronize(ctx->stream));
    return MEC_OK;
}

mec_status mec_bitrot_sum_batch(mec_ctx *ctx, int algo, int n,
                                const uint8_t *msgs, int64_t msg_len,
                                int64_t msg_stride, uint8_t *sums) {
    if (!ctx) return MEC_ERR_INVALID_ARG;
    if (n <= 0 || msg_len < 0 || !hash_size(algo)) return MEC_ERR_INVALID_ARG;
    const int hsz = hash_size(algo);
    std::lock_guard<std::mutex> lk(ctx->mu);
    HIP_TRY(hipSetDevice(ctx->device));
    int64_t dstride = (msg_len + 63) & ~int64_t(63);
    if (dstride == 0) dstride = 64;
    size_t in_bytes = (size_t)n * dstride;
    size_t out_bytes = (size_t)n * hsz;
    mec_status st;
    if ((st = ctx->ensure(&ctx->dev_a, &ctx->cap_a, in_bytes)) != MEC_OK)
        return st;
    if ((st = ctx->ensure(&ctx->dev_c, &ctx->cap_c, out_bytes)) != MEC_OK)
        return st;
    if ((st = ctx->ensure_pin(in_bytes > out_bytes ? in_bytes : out_bytes)) !=
        MEC_OK)
        return st;
    uint8_t *pinb = (uint8_t *)ctx->pin;
    /* overlapped ingest: chunk c+1's host pack + PCIe
     * upload (stream2) overlap chunk c's hash kernel (stream) — the
     * GPU-side analogue of parallelReader's overlapped shard reads
     * feeding verify-on-read (cmd/erasure-decode.go:127-235). */
    const int64_t chunk = (n > 512) ? ((n + 3) / 4) : n;
    for (int64_t i0 = 0; i0 < n; i0 += chunk) {
        const int64_t nc = (i0 + chunk <= n) ? chunk : (n - i0);
        uint8_t *pc = pinb + (size_t)i0 * dstride;
        for (int64_t i = 0; i < nc; i++)
            memcpy(pc + (size_t)i * dstride,
                   msgs + (size_t)(i0 + i) * msg_stride, (size_t)msg_len);
        HIP_TRY(hipMemcpyAsync((uint8_t *)ctx->dev_a + (size_t)i0 * dstride,
                               pc, (size_t)nc * dstride,
                               hipMemcpyHostToDevice, ctx->stream2));
        HIP_TRY(hipEventRecord(ctx->ev_h, ctx->stream2));
        HIP_TRY(hipStreamWaitEvent(ctx->stream, ctx->ev_h, 0));
        HashArgs h{};
        h.data = (const uint8_t *)ctx->dev_a + (size_t)i0 * dstride;
        h.parity = nullptr;
        h.sums = (uint8_t *)ctx->dev_c + (size_t)i0 * hsz;
        h.row_stride = dstride;
        h.msg_len = msg_len;
        h.n_chains = nc;
        memcpy(h.key, kMagicHHKey, 32);
        HIP_TRY(mec_launch_hash(algo, &h, ctx->stream));
    }
    HIP_TRY(hipMemcpyAsync(pinb, ctx->dev_c, out_bytes, hipMemcpyDeviceToHost,
                           ctx->stream));
    HIP_TRY(hipStreamSynchronize(ctx->stream));
    memcpy(sums, pinb, out_bytes);
    return MEC_OK;
}

mec_status mec_bitrot_verify_batch(mec_ctx *ctx, int algo, int n,
                                   const uint8_t *msgs, int64_t msg_len,
                                   int64_t msg_stride, const uint8_t *want,
                                   uint8_t *ok_out) {
    const int hsz = hash_size(algo);
    if (!hsz) return MEC_ERR_INVALID_ARG;
    std::vector<uint8_t> got((size_t)n * hsz);
    mec_status st = mec_bitrot_sum_batch(ctx, algo, n, msgs, msg_len,
                                         msg_stride, got.data());
    if (st != MEC_OK) return st;
    for (int i = 0; i < n; i++)
        ok_out[i] = memcmp(got.data() + (size_t)i * hsz,
                           want + (size_t)i * hsz, (size_t)hsz) == 0;
    return MEC_OK;
}

/* ---- device memory + timing helpers ------------------------------------ */

mec_status mec_dev_alloc(mec_ctx *ctx, size_t bytes, void **out) {
    if (!ctx) return MEC_ERR_INVALID_ARG;
    HIP_TRY(hipSetDevice(ctx->device));
    HIP_TRY(hipMalloc(out, bytes));
    return MEC_OK;
}

void mec_dev_free(mec_ctx *ctx, void *ptr) {
    (void)hipSetDevice(ctx->device);
    (void)hipFree(ptr);
}

mec_status mec_memcpy_h2d(mec_ctx *ctx, void *dst_dev, const void *src,
                          size_t bytes) {
    if (!ctx) return MEC_ERR_INVALID_ARG;
    HIP_TRY(hipSetDevice(ctx->device));
    HIP_TRY(hipMemcpyAsync(dst_dev, src, bytes, hipMemcpyHostToDevice,
                           ctx->stream));
    HIP_TRY(hipStreamSynchronize(ctx->stream));
    return MEC_OK;
}

mec_status mec_memcpy_d2h(mec_ctx *ctx, void *dst, const void *src_dev,
                          size_t bytes) {
    if (!ctx) return MEC_ERR_INVALID_ARG;
    HIP_TRY(hipSetDevice(ctx->device));
    HIP_TRY(hipMemcpyAsync(dst, src_dev, bytes, hipMemcpyDeviceToHost,
                           ctx->stream));
    HIP_TRY(hipStreamSynchronize(ctx->stream));
    return MEC_OK;
}

mec_status mec_memset_dev(mec_ctx *ctx, void *dst_dev, int value,
                          size_t bytes) {
    if (!ctx) return MEC_ERR_INVALID_ARG;
    HIP_TRY(hipSetDevice(ctx->device));
    HIP_TRY(hipMemsetAsync(dst_dev, value, bytes, ctx->stream));
    HIP_TRY(hipStreamSynchronize(ctx->stream));
    return MEC_OK;
}

mec_status mec_stream_sync(mec_ctx *ctx) {
    if (!ctx) return MEC_ERR_INVALID_ARG;
    HIP_TRY(hipSetDevice(ctx->device));
    HIP_TRY(hipStreamSynchronize(ctx->stream));
    return MEC_OK;
}

mec_status mec_timer_start(mec_ctx *ctx) {
    if (!ctx) return MEC_ERR_INVALID_ARG;
    HIP_TRY(hipSetDevice(ctx->device));
    HIP_TRY(hipEventRecord(ctx->ev_start, ctx->stream));
    return MEC_OK;
}

/* GPU-side streaming-format assembly for the full-block phase of
 * mec_encode_stream: upload packed object bytes once,
 * scatter to shard rows on device, fused encode+hash, interleave the
 * per-drive [hash||shard]* streams on device, copy one stream per drive
 * out.  Layout matches cmd/bitrot-streaming.go:57-75 bit-for-bit (pinned
 * by the stream round-trip tests). */
mec_status mec_encode_stream_gpu(mec_ctx *ctx, const uint8_t *src,
                                 int64_t n_full, int algo,
                                 uint8_t *const *drive_bufs,
                                 int64_t drive_off) {
    if (!ctx) return MEC_ERR_INVALID_ARG;
    std::lock_guard<std::mutex> lk(ctx->mu);
    HIP_TRY(hipSetDevice(ctx->device));
    const int d = ctx->d, p = ctx->p, total = d + p;
    const int64_t bs = ctx->block_size, S = ctx->S, stride = ctx->stride;
    const int64_t pitch = 32 + S;
    const size_t src_bytes = (size_t)n_full * bs;
    const size_t row_bytes = (size_t)n_full * d * stride;
    const size_t par_bytes = (size_t)n_full * p * stride;
    const size_t sum_bytes = (size_t)n_full * total * 32;
    const size_t out_bytes = (size_t)total * n_full * pitch;
    /* shard rows live after the stream-out region; rows must keep the 64-B
     * base alignment the row kernels' uint4 accesses assume, and out_bytes
     * is odd whenever pitch (32+S) is (ragged S, e.g. EC12+4) */
    const size_t out_aligned = (out_bytes + 63) & ~(size_t)63;
    mec_status st;
    if ((st = ctx->ensure(&ctx->dev_a, &ctx->cap_a,
                          row_bytes > src_bytes ? row_bytes : src_bytes)) !=
        MEC_OK)
        return st;
    if ((st = ctx->ensure(&ctx->dev_b, &ctx->cap_b, par_bytes)) != MEC_OK)
        return st;
    if ((st = ctx->ensure(&ctx->dev_c, &ctx->cap_c, sum_bytes)) != MEC_OK)
        return st;
    if ((st = ctx->ensure(&ctx->dev_d, &ctx->cap_d,
                          out_aligned + row_bytes)) != MEC_OK)
        return st;
    /* dev_d holds [stream out | shard rows]; dev_a stages the packed src */
    uint8_t *dev_rows = (uint8_t *)ctx->dev_d + out_aligned;
    HIP_TRY(hipMemcpyAsync(ctx->dev_a, src, src_bytes, hipMemcpyHostToDevice,
                           ctx->stream));
    ScatterArgs sa{};
    sa.src = (const uint8_t *)ctx->dev_a;
    sa.rows = dev_rows;
    sa.block_len = bs;
    sa.S = S;
    sa.row_stride = stride;
    sa.n = n_full;
    sa.d = d;
    HIP_TRY(mec_launch_scatter_rows(&sa, ctx->stream));
    st = encode_dev_locked(ctx, (int)n_full, dev_rows, bs, ctx->dev_b, algo,
                           ctx->dev_c);
    if (st != MEC_OK) return st;
    InterleaveArgs ia{};
    ia.data = dev_rows;
    ia.parity = (const uint8_t *)ctx->dev_b;
    ia.sums = (const uint8_t *)ctx->dev_c;
    ia.out = (uint8_t *)ctx->dev_d;
    ia.S = S;
    ia.row_stride = stride;
    ia.n = n_full;
    ia.d = d;
    ia.p = p;
    HIP_TRY(mec_launch_stream_interleave(&ia, ctx->stream));
    HIP_TRY(hipStreamSynchronize(ctx->stream));
    for (int s = 0; s < total; s++)
        HIP_TRY(hipMemcpy(drive_bufs[s] + drive_off,
                          (uint8_t *)ctx->dev_d + (int64_t)s * n_full * pitch,
                          (size_t)(n_full * pitch), hipMemcpyDeviceToHost));
    return MEC_OK;
}

mec_status mec_timer_stop(mec_ctx *ctx, float *ms_out) {
    if (!ctx) return MEC_ERR_INVALID_ARG;
    HIP_TRY(hipSetDevice(ctx->device));
    HIP_TRY(hipEventRecord(ctx->ev_stop, ctx->stream));
    HIP_TRY(hipEventSynchronize(ctx->ev_stop));
    HIP_TRY(hipEventElapsedTime(ms_out, ctx->ev_start, ctx->ev_stop));
    return MEC_OK;
}

} /* extern "C" */
