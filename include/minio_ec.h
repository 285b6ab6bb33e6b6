/* minio_ec.h — C-ABI drop-in boundary for MinIO's erasure-coding + bitrot
 * hot path, MI355X (gfx950) native implementation.
 *
 * This is the surface a thin cgo shim binds to satisfy the reference's Go
 * interfaces (see INTEGRATION.md for the shim):
 *
 *  - reedsolomon.Encoder methods used by the reference
 *    (cmd/erasure-coding.go:81 Split, :85 Encode, :106 ReconstructData,
 *     :112 Reconstruct)                       -> mec_encode_batch /
 *                                                mec_reconstruct_batch
 *  - bitrotWriter's per-shard hash step
 *    (cmd/bitrot-streaming.go:57-59)          -> fused into mec_encode_batch
 *                                                (checksums output)
 *  - bitrotReader verify-on-read
 *    (cmd/bitrot-streaming.go:185-197)        -> mec_bitrot_verify_batch
 *  - bitrotVerify / VerifyFile scrub
 *    (cmd/bitrot.go:164-216)                  -> mec_bitrot_verify_stream
 *  - Erasure.Encode block loop + streaming [hash||shard]* on-disk layout
 *    (cmd/erasure-encode.go:76-108, cmd/bitrot-streaming.go:44-75)
 *                                             -> mec_encode_stream
 *  - Erasure.Decode / Heal
 *    (cmd/erasure-decode.go:239-314, :317-364) -> mec_decode_stream /
 *                                                mec_heal_stream
 *  - shard-size math (cmd/erasure-coding.go:116-141, cmd/bitrot.go:156-161)
 *                                             -> mec_shard_size etc.
 *
 * Calls are synchronous and thread-safe per context; a context owns one GPU
 * (one HIP stream + staging buffers).  Batching across objects/blocks is the
 * caller's (shim's) job: the reference API is per-block, GPU efficiency
 * needs hundreds of blocks per call.
 *
 * All functions return mec_status unless documented otherwise.  Status codes
 * map 1:1 onto the Go error values the shim must surface
 * (reedsolomon.ErrInvShardNum etc., cmd/erasure-coding.go:45-49,
 *  errFileCorrupt for bitrot mismatches).
 */
#ifndef MINIO_EC_H
#define MINIO_EC_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef struct mec_ctx mec_ctx;

typedef enum {
    MEC_OK = 0,
    MEC_ERR_INV_SHARD_NUM = 1,  /* reedsolomon.ErrInvShardNum */
    MEC_ERR_MAX_SHARD_NUM = 2,  /* reedsolomon.ErrMaxShardNum */
    MEC_ERR_TOO_FEW_SHARDS = 3, /* reedsolomon.ErrTooFewShards */
    MEC_ERR_SHORT_DATA = 4,     /* reedsolomon.ErrShortData */
    MEC_ERR_FILE_CORRUPT = 5,   /* errFileCorrupt (bitrot mismatch) */
    MEC_ERR_INVALID_ARG = 6,
    MEC_ERR_HIP = 7,            /* HIP runtime failure (message via
                                   mec_last_error) */
    MEC_ERR_NO_GPU = 8,         /* no MI355X visible — the product path
                                   fails loudly, it never falls back */
    MEC_ERR_INTERNAL = 9,       /* invariant violation inside this library
                                   (a bug, not a caller/quorum condition) */
} mec_status;

/* Bitrot algorithm ids — values mirror the reference enum
 * (cmd/xl-storage-format-v1.go:146-153). */
typedef enum {
    MEC_BITROT_SHA256 = 1,
    MEC_BITROT_HIGHWAYHASH256 = 2,
    MEC_BITROT_HIGHWAYHASH256S = 3, /* DefaultBitrotAlgorithm */
    MEC_BITROT_BLAKE2B512 = 4,
} mec_bitrot_algo;

int mec_version(void);
const char *mec_last_error(void); /* thread-local message for MEC_ERR_HIP */
int mec_device_count(void);       /* number of visible GPUs */

/* ---- context ---------------------------------------------------------- */

/* Geometry checks mirror NewErasure (cmd/erasure-coding.go:42-50). */
mec_status mec_ctx_create(int data_shards, int parity_shards,
                          int64_t block_size, int device, mec_ctx **out);
void mec_ctx_destroy(mec_ctx *ctx);
int mec_ctx_d(mec_ctx *ctx);
int mec_ctx_p(mec_ctx *ctx);
int64_t mec_ctx_block_size(mec_ctx *ctx);

/* ---- shard-size math (host, exact int mirrors) ------------------------ */

int64_t mec_shard_size(int64_t block_size, int data_shards); /* ceilFrac */
int64_t mec_shard_file_size(int64_t block_size, int data_shards,
                            int64_t total_length);
int64_t mec_shard_file_offset(int64_t block_size, int data_shards,
                              int64_t start_offset, int64_t length,
                              int64_t total_length);
int64_t mec_bitrot_shard_file_size(int64_t size, int64_t shard_size,
                                   int algo);
/* Device-side stride for one shard (shard_size rounded up to 64 B). */
int64_t mec_shard_stride(int64_t block_size, int data_shards);

/* ---- batch encode (the north-star kernel) ------------------------------
 *
 * n independent blocks, each block_len bytes (1 <= block_len <= block_size).
 * Device layout (the _dev entry points):
 *   data:   n * d * stride bytes; shard k of block b at
 *           data + (b*d + k)*stride, holding
 *           min(max(block_len - k*S, 0), S) valid bytes, zero-padded to S
 *           (Split semantics, cmd/erasure-coding.go:81), bytes S..stride
 *           undefined; S = mec_shard_size, stride = mec_shard_stride.
 *   parity: n * p * stride bytes (output).
 *   sums:   n * (d+p) * hash_size bytes (output; per-shard digest over the
 *           S padded shard bytes, as streamingBitrotWriter hashes them).
 *           May be NULL to skip hashing (encode only).
 * Host-pointer version: data is the packed object bytes, n * block_len
 * contiguous (the Go Split aliasing shape); the library stages/scatters.
 * Runs on the context's stream; synchronous unless noted. */
mec_status mec_encode_batch_dev(mec_ctx *ctx, int n, const void *data_dev,
                                int64_t block_len, void *parity_dev,
                                int bitrot_algo, void *sums_dev);
mec_status mec_encode_batch(mec_ctx *ctx, int n, const uint8_t *data,
                            int64_t block_len, uint8_t *parity,
                            int bitrot_algo, uint8_t *sums);

/* Async variant for bench pipelining: does not synchronize the stream. */
mec_status mec_encode_batch_dev_async(mec_ctx *ctx, int n,
                                      const void *data_dev, int64_t block_len,
                                      void *parity_dev, int bitrot_algo,
                                      void *sums_dev);

/* Pipelined encode: batch t's hash overlaps batch t+1's GF kernel (the
 * hash is latency-bound per chain and leaves most SIMDs idle; independent
 * batches — i.e. different objects — fill them).  CONTRACT: alternate two
 * parity/sums buffer sets in strict round-robin and call mec_pipe_sync
 * before reading results.  Per-call results are identical to
 * mec_encode_batch_dev.  Requires a compiled (d,p) specialization and a
 * HighwayHash algorithm. */
mec_status mec_encode_batch_dev_pipe(mec_ctx *ctx, int n,
                                     const void *data_dev, int64_t block_len,
                                     void *parity_dev, int bitrot_algo,
                                     void *sums_dev);
mec_status mec_pipe_sync(mec_ctx *ctx);

/* ---- batch reconstruct (ReconstructData / Reconstruct / Heal kernel) ---
 *
 * shards_dev: n * (d+p) * stride bytes; present[i] != 0 marks shard row i
 * intact for ALL n items (the erasure pattern of one object's read).
 * Missing rows are reconstructed in place (data rows only when data_only).
 * Matrix inversion happens once on host per call. */
mec_status mec_reconstruct_batch_dev(mec_ctx *ctx, int n, void *shards_dev,
                                     const uint8_t *present,
                                     int64_t shard_len, int data_only);
mec_status mec_reconstruct_batch(mec_ctx *ctx, int n, uint8_t *shards,
                                 const uint8_t *present, int64_t shard_len,
                                 int data_only);
mec_status mec_reconstruct_batch_dev_async(mec_ctx *ctx, int n,
                                           void *shards_dev,
                                           const uint8_t *present,
                                           int64_t shard_len, int data_only);

/* ---- batch bitrot hash / verify ----------------------------------------
 * msgs: n messages of msg_len bytes at msg_stride; sums: n * hash_size.
 * verify: ok_out[i]=1 where digest matches want[i] (errFileCorrupt map). */
mec_status mec_bitrot_sum_batch_dev(mec_ctx *ctx, int algo, int n,
                                    const void *msgs_dev, int64_t msg_len,
                                    int64_t msg_stride, void *sums_dev);
mec_status mec_bitrot_sum_batch(mec_ctx *ctx, int algo, int n,
                                const uint8_t *msgs, int64_t msg_len,
                                int64_t msg_stride, uint8_t *sums);
mec_status mec_bitrot_verify_batch(mec_ctx *ctx, int algo, int n,
                                   const uint8_t *msgs, int64_t msg_len,
                                   int64_t msg_stride, const uint8_t *want,
                                   uint8_t *ok_out);

/* bitrotVerify (cmd/bitrot.go:164-216): one [hash||shard]* stream (or whole
 * file for non-streaming algos).  Returns MEC_OK or MEC_ERR_FILE_CORRUPT. */
mec_status mec_bitrot_verify_stream(mec_ctx *ctx, const uint8_t *stream,
                                    int64_t want_size, int64_t part_size,
                                    int algo, const uint8_t *want_sum,
                                    int64_t shard_size);

/* ---- streaming-format host mirrors (C++ host driver over the kernels) --
 *
 * mec_encode_stream: Erasure.Encode loop (cmd/erasure-encode.go:76-108)
 * fused with streamingBitrotWriter (cmd/bitrot-streaming.go:44-75): encodes
 * src (src_len bytes) into d+p per-drive streams in the on-disk
 * [hash||shard]* layout (HighwayHash256S) or raw shards (whole-file algos;
 * whole_sums then receives the d+p whole-file digests).  drive_bufs[i] must
 * hold mec_bitrot_shard_file_size(mec_shard_file_size(...), S, algo) bytes. */
mec_status mec_encode_stream(mec_ctx *ctx, const uint8_t *src,
                             int64_t src_len, int algo,
                             uint8_t *const *drive_bufs,
                             uint8_t *whole_sums);

/* mec_decode_stream: Erasure.Decode (cmd/erasure-decode.go:239-314) over
 * in-memory drive streams; NULL entries mark missing drives.  Verifies
 * every shard read (streamingBitrotReader.ReadAt), reconstructs when rows
 * are missing, writes [offset, offset+length) of the object into dst
 * (writeDataBlocks, cmd/erasure-utils.go:42-105).
 * whole_sums: for non-streaming algos, the d+p expected whole-file digests
 * (NULL entries skip verification), mirroring wholeBitrotReader. */
mec_status mec_decode_stream(mec_ctx *ctx, const uint8_t *const *drive_bufs,
                             const uint8_t *whole_sums, int algo,
                             int64_t total_length, int64_t offset,
                             int64_t length, uint8_t *dst);

/* mec_heal_stream: Erasure.Heal (cmd/erasure-decode.go:317-364): given any
 * >= d intact drive streams, regenerate the full set of d+p streams for the
 * missing drives (out_bufs[i] may be NULL to skip a drive). */
mec_status mec_heal_stream(mec_ctx *ctx, const uint8_t *const *drive_bufs,
                           int algo, int64_t total_length,
                           uint8_t *const *out_bufs);

/* ---- device memory + timing helpers (bench/test plumbing) ------------- */

mec_status mec_dev_alloc(mec_ctx *ctx, size_t bytes, void **out);
void mec_dev_free(mec_ctx *ctx, void *ptr);
mec_status mec_memcpy_h2d(mec_ctx *ctx, void *dst_dev, const void *src,
                          size_t bytes);
mec_status mec_memcpy_d2h(mec_ctx *ctx, void *dst, const void *src_dev,
                          size_t bytes);
mec_status mec_memset_dev(mec_ctx *ctx, void *dst_dev, int value,
                          size_t bytes);
mec_status mec_stream_sync(mec_ctx *ctx);
/* hipEvent timers on the context's stream (the stream kernels launch on). */
mec_status mec_timer_start(mec_ctx *ctx);
mec_status mec_timer_stop(mec_ctx *ctx, float *ms_out);

#ifdef __cplusplus
}
#endif
#endif /* MINIO_EC_H */
