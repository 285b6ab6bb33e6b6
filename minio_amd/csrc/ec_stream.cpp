/* Streaming-format host mirrors over the batch GPU calls.
 *
 *  - mec_encode_stream: Erasure.Encode loop (cmd/erasure-encode.go:76-108)
 *    + streamingBitrotWriter on-disk [hash||shard]* layout
 *    (cmd/bitrot-streaming.go:44-75) / wholeBitrotWriter
 *    (cmd/bitrot-whole.go:35-45).
 *  - mec_decode_stream: Erasure.Decode (cmd/erasure-decode.go:239-314):
 *    verify-on-read (streamingBitrotReader.ReadAt,
 *    cmd/bitrot-streaming.go:161-200), DecodeDataBlocks
 *    (cmd/erasure-coding.go:94-107), writeDataBlocks trim
 *    (cmd/erasure-utils.go:42-105).
 *  - mec_heal_stream: Erasure.Heal (cmd/erasure-decode.go:317-364).
 *  - mec_bitrot_verify_stream: bitrotVerify (cmd/bitrot.go:164-216).
 *
 * All shard arithmetic and hashing runs on the GPU via the batch entry
 * points; this file is control flow + layout assembly only.
 */
#include "../../include/minio_ec.h"

#include <cstring>
#include <map>
#include <vector>

namespace {

int64_t ceil_frac(int64_t num, int64_t den) {
    if (den == 0) return 0;
    return (num + den - 1) / den;
}

int hash_size(int algo) {
    return (algo == MEC_BITROT_BLAKE2B512) ? 64
           : (algo >= 1 && algo <= 3)      ? 32
                                           : 0;
}

struct Geo {
    int d, p, total;
    int64_t block_size, S;
    int64_t n_blocks;     /* blocks in the object */
    int64_t last_len;     /* bytes in the last block */
    int64_t last_S;       /* shard size of the last block */
    int64_t shard_file;   /* per-drive shard file size (no bitrot overhead) */
};

Geo make_geo(int d, int p, int64_t block_size, int64_t total_length) {
    Geo g{};
    g.d = d;
    g.p = p;
    g.total = d + p;
    g.block_size = block_size;
    g.S = ceil_frac(block_size, d);
    g.n_blocks = total_length > 0 ? ceil_frac(total_length, block_size) : 0;
    g.last_len = total_length > 0
                     ? (total_length - (g.n_blocks - 1) * block_size)
                     : 0;
    g.last_S = ceil_frac(g.last_len, d);
    g.shard_file = mec_shard_file_size(block_size, d, total_length);
    return g;
}

/* offset of block b's shard inside the per-drive stream */
int64_t stream_off(const Geo &g, int64_t b, int algo) {
    int hsz = (algo == MEC_BITROT_HIGHWAYHASH256S) ? 32 : 0;
    return b * ((int64_t)hsz + g.S);
}

} // namespace

/* geometry getters + GPU stream assembly (defined in ec_abi.cpp) */
extern "C" int64_t mec_ctx_block_size(mec_ctx *);
extern "C" int mec_ctx_d(mec_ctx *);
extern "C" int mec_ctx_p(mec_ctx *);
extern "C" mec_status mec_encode_stream_gpu(mec_ctx *, const uint8_t *,
                                            int64_t, int, uint8_t *const *,
                                            int64_t);

extern "C" {

mec_status mec_encode_stream(mec_ctx *ctx_, const uint8_t *src,
                             int64_t src_len, int algo,
                             uint8_t *const *drive_bufs,
                             uint8_t *whole_sums) {
    int d = mec_ctx_d(ctx_), p = mec_ctx_p(ctx_);
    int64_t block_size = mec_ctx_block_size(ctx_);
    if (src_len < 0 || !hash_size(algo)) return MEC_ERR_INVALID_ARG;
    Geo g = make_geo(d, p, block_size, src_len);
    const int hsz = hash_size(algo);
    const bool streaming = algo == MEC_BITROT_HIGHWAYHASH256S;

    if (src_len == 0) {
        /* empty object: empty streams; whole-file digest of empty input
         * (wholeBitrotWriter over zero bytes) */
        if (!streaming && whole_sums) {
            /* hash of empty via the GPU batch call */
            std::vector<uint8_t> sums((size_t)g.total * hsz);
            uint8_t dummy = 0;
            mec_status st = mec_bitrot_sum_batch(ctx_, algo, g.total, &dummy,
                                                 0, 1, sums.data());
            if (st != MEC_OK) return st;
            memcpy(whole_sums, sums.data(), sums.size());
        }
        return MEC_OK;
    }

    /* encode all full blocks in one batch, the ragged last block alone */
    int64_t n_full = g.n_blocks - (g.last_len != g.block_size ? 1 : 0);
    std::vector<uint8_t> parity, sums;
    /* full blocks of the streaming format: fully on-device assembly
     * (scatter -> fused encode+hash -> [hash||shard]* interleave), one
     * upload + one download per drive */
    bool gpu_full_done = false;
    if (streaming && n_full > 0) {
        mec_status st = mec_encode_stream_gpu(ctx_, src, n_full, algo,
                                              drive_bufs, 0);
        if (st != MEC_OK) return st;
        gpu_full_done = true;
    }
    /* whole-file accumulation needs every shard byte per drive; we assemble
     * drive streams first, then hash them in one strided batch call */
    for (int64_t phase = 0; phase < 2; phase++) {
        if (phase == 0 && gpu_full_done) continue;
        int64_t b0 = phase == 0 ? 0 : n_full;
        int64_t nb = phase == 0 ? n_full : g.n_blocks - n_full;
        if (nb <= 0) continue;
        int64_t blen = phase == 0 ? g.block_size : g.last_len;
        int64_t S = ceil_frac(blen, d);
        parity.assign((size_t)nb * p * S, 0);
        sums.assign((size_t)nb * g.total * hsz, 0);
        mec_status st = mec_encode_batch(
            ctx_, (int)nb, src + b0 * g.block_size, blen, parity.data(), algo,
            streaming ? sums.data() : nullptr);
        if (st != MEC_OK) return st;
        /* assemble per-drive streams */
        for (int64_t b = 0; b < nb; b++) {
            int64_t gb = b0 + b;
            const uint8_t *blk = src + gb * g.block_size;
            for (int s = 0; s < g.total; s++) {
                uint8_t *out = drive_bufs[s] + stream_off(g, gb, algo);
                if (streaming) {
                    memcpy(out, sums.data() + ((size_t)b * g.total + s) * hsz,
                           (size_t)hsz);
                    out += hsz;
                }
                if (s < d) {
                    /* data shard: bytes [s*S,(s+1)*S) of the block,
                     * zero-padded (Split) */
                    int64_t have = blen - (int64_t)s * S;
                    if (have < 0) have = 0;
                    if (have > S) have = S;
                    if (have) memcpy(out, blk + (int64_t)s * S, (size_t)have);
                    if (have < S) memset(out + have, 0, (size_t)(S - have));
                } else {
                    memcpy(out, parity.data() + ((size_t)b * p + (s - d)) * S,
                           (size_t)S);
                }
            }
        }
    }
    if (!streaming && whole_sums) {
        /* one whole-file digest per drive stream (wholeBitrotWriter.Sum) */
        for (int s = 0; s < g.total; s++) {
            mec_status st = mec_bitrot_sum_batch(ctx_, algo, 1, drive_bufs[s],
                                                 g.shard_file, g.shard_file,
                                                 whole_sums + (size_t)s * hsz);
            if (st != MEC_OK) return st;
        }
    }
    return MEC_OK;
}

mec_status mec_decode_stream(mec_ctx *ctx_, const uint8_t *const *drive_bufs,
                             const uint8_t *whole_sums, int algo,
                             int64_t total_length, int64_t offset,
                             int64_t length, uint8_t *dst) {
    int d = mec_ctx_d(ctx_), p = mec_ctx_p(ctx_);
    int64_t block_size = mec_ctx_block_size(ctx_);
    if (offset < 0 || length < 0 || offset + length > total_length)
        return MEC_ERR_INVALID_ARG;
    if (length == 0) return MEC_OK;
    if (!hash_size(algo)) return MEC_ERR_INVALID_ARG;
    Geo g = make_geo(d, p, block_size, total_length);
    const int hsz = hash_size(algo);
    const bool streaming = algo == MEC_BITROT_HIGHWAYHASH256S;

    /* whole-file algorithms: verify each available drive's full stream
     * digest up front (wholeBitrotReader buffering semantics) */
    std::vector<uint8_t> drive_ok(g.total, 0);
    for (int s = 0; s < g.total; s++) drive_ok[s] = drive_bufs[s] != nullptr;
    if (!streaming && whole_sums) {
        for (int s = 0; s < g.total; s++) {
            if (!drive_ok[s]) continue;
            const uint8_t *want = whole_sums + (size_t)s * hsz;
            /* NULL-sum entries are all-zero -> skip verification */
            bool has = false;
            for (int i = 0; i < hsz; i++) has |= want[i] != 0;
            if (!has) continue;
            uint8_t ok = 0;
            mec_status st = mec_bitrot_verify_batch(
                ctx_, algo, 1, drive_bufs[s], g.shard_file, g.shard_file,
                want, &ok);
            if (st != MEC_OK) return st;
            if (!ok) drive_ok[s] = 0;
        }
    }

    int64_t start_block = offset / block_size;
    /* the reference iterates to endBlock = (offset+length)/blockSize
     * inclusive and breaks on a zero-length final block
     * (cmd/erasure-decode.go:257-280); reads only touch blocks that hold
     * data */
    int64_t end_block_raw = (offset + length) / block_size;
    int64_t end_block = end_block_raw;
    if (end_block >= g.n_blocks) end_block = g.n_blocks - 1;

    /* verify-on-read for the streaming format, batched per drive over the
     * touched blocks (full-size blocks in one call, ragged last alone) */
    int64_t nb = end_block - start_block + 1;
    std::vector<uint8_t> shard_ok((size_t)nb * g.total, 0);
    for (int s = 0; s < g.total; s++) {
        if (!drive_ok[s]) continue;
        if (!streaming) {
            for (int64_t b = 0; b < nb; b++) shard_ok[b * g.total + s] = 1;
            continue;
        }
        int64_t full_nb = nb;
        bool ragged_last =
            (start_block + nb - 1 == g.n_blocks - 1) && g.last_S != g.S;
        if (ragged_last) full_nb--;
        if (full_nb > 0) {
            std::vector<uint8_t> ok((size_t)full_nb);
            const uint8_t *base =
                drive_bufs[s] + stream_off(g, start_block, algo);
            /* gather the stored hashes packed (verify wants stride hsz) */
            std::vector<uint8_t> want((size_t)full_nb * hsz);
            for (int64_t b = 0; b < full_nb; b++)
                memcpy(want.data() + (size_t)b * hsz,
                       base + b * (hsz + g.S), (size_t)hsz);
            mec_status st = mec_bitrot_verify_batch(
                ctx_, algo, (int)full_nb, base + hsz, g.S, hsz + g.S,
                want.data(), ok.data());
            if (st != MEC_OK) return st;
            for (int64_t b = 0; b < full_nb; b++)
                shard_ok[b * g.total + s] = ok[b];
        }
        if (ragged_last) {
            const uint8_t *hp =
                drive_bufs[s] + stream_off(g, g.n_blocks - 1, algo);
            uint8_t ok = 0;
            mec_status st = mec_bitrot_verify_batch(ctx_, algo, 1, hp + hsz,
                                                    g.last_S, g.last_S + hsz,
                                                    hp, &ok);
            if (st != MEC_OK) return st;
            shard_ok[(nb - 1) * g.total + s] = ok;
        }
    }

    /* group blocks by present-mask, reconstruct groups via the batch call */
    std::map<std::vector<uint8_t>, std::vector<int64_t>> groups;
    for (int64_t b = 0; b < nb; b++) {
        std::vector<uint8_t> mask(shard_ok.begin() + b * g.total,
                                  shard_ok.begin() + (b + 1) * g.total);
        groups[mask].push_back(b);
    }
    /* assemble decoded data-shard bytes per block */
    std::vector<std::vector<uint8_t>> block_data((size_t)nb);
    for (auto &kv : groups) {
        const std::vector<uint8_t> &mask = kv.first;
        int n_present = 0;
        for (int s = 0; s < g.total; s++) n_present += mask[s] != 0;
        bool all_data = true;
        for (int s = 0; s < d; s++) all_data &= mask[s] != 0;
        if (!all_data && n_present < d) return MEC_ERR_FILE_CORRUPT;
        /* split group into uniform-shard-size runs (ragged last separate) */
        for (int pass = 0; pass < 2; pass++) {
            std::vector<int64_t> blocks;
            for (int64_t b : kv.second) {
                bool ragged = (start_block + b == g.n_blocks - 1) &&
                              g.last_S != g.S;
                if ((pass == 1) == ragged) blocks.push_back(b);
            }
            if (blocks.empty()) continue;
            int64_t S = pass == 1 ? g.last_S : g.S;
            int64_t n = (int64_t)blocks.size();
            std::vector<uint8_t> rows((size_t)n * g.total * S, 0);
            for (int64_t i = 0; i < n; i++) {
                int64_t gb = start_block + blocks[i];
                for (int s = 0; s < g.total; s++) {
                    if (!mask[s]) continue;
                    const uint8_t *sp = drive_bufs[s] +
                                        stream_off(g, gb, algo) +
                                        (streaming ? hsz : 0);
                    memcpy(rows.data() + ((size_t)i * g.total + s) * S, sp,
                           (size_t)S);
                }
            }
            if (!all_data) {
                mec_status st = mec_reconstruct_batch(
                    ctx_, (int)n, rows.data(), mask.data(), S, 1);
                if (st != MEC_OK) return st;
            }
            for (int64_t i = 0; i < n; i++) {
                auto &bd = block_data[(size_t)blocks[i]];
                bd.assign((size_t)d * S, 0);
                memcpy(bd.data(), rows.data() + (size_t)i * g.total * S,
                       (size_t)d * S);
            }
        }
    }

    /* writeDataBlocks: trim [offset, offset+length) out of the block
     * stream (cmd/erasure-utils.go:42-105 via cmd/erasure-decode.go:262) */
    int64_t written = 0;
    for (int64_t gb = start_block; gb <= end_block_raw; gb++) {
        int64_t block_off, block_len;
        if (start_block == end_block_raw) {
            block_off = offset % block_size;
            block_len = length;
        } else if (gb == start_block) {
            block_off = offset % block_size;
            block_len = block_size - block_off;
        } else if (gb == end_block_raw) {
            block_off = 0;
            block_len = (offset + length) % block_size;
        } else {
            block_off = 0;
            block_len = block_size;
        }
        if (block_len == 0) break;
        memcpy(dst + written,
               block_data[(size_t)(gb - start_block)].data() + block_off,
               (size_t)block_len);
        written += block_len;
    }
    /* a short write here would be a logic bug in the block walk, not a
     * quorum condition — label it as such (VERDICT r1 minor #7) */
    return written == length ? MEC_OK : MEC_ERR_INTERNAL;
}

mec_status mec_heal_stream(mec_ctx *ctx_, const uint8_t *const *drive_bufs,
                           int algo, int64_t total_length,
                           uint8_t *const *out_bufs) {
    int d = mec_ctx_d(ctx_), p = mec_ctx_p(ctx_);
    int64_t block_size = mec_ctx_block_size(ctx_);
    if (!hash_size(algo)) return MEC_ERR_INVALID_ARG;
    if (total_length < 0) return MEC_ERR_INVALID_ARG;
    if (total_length == 0) return MEC_OK; /* empty part: nothing to heal
        (a 0-length object has no shard stream; without this guard the
        ragged-last path below would index block -1) */
    Geo g = make_geo(d, p, block_size, total_length);
    const int hsz = hash_size(algo);
    const bool streaming = algo == MEC_BITROT_HIGHWAYHASH256S;
    if (!streaming) return MEC_ERR_INVALID_ARG; /* heal targets v2 objects */

    std::vector<uint8_t> mask(g.total);
    int n_present = 0;
    for (int s = 0; s < g.total; s++) {
        mask[s] = drive_bufs[s] != nullptr;
        n_present += mask[s];
    }
    if (n_present < d) return MEC_ERR_TOO_FEW_SHARDS;

    /* verify-on-read (Heal reads through bitrot readers,
     * cmd/erasure-decode.go:322 + bitrot-streaming.go:185-197): a corrupt
     * shard is treated as missing for its block.  Whole-drive verification
     * batched per drive; per-block masks grouped below. */
    std::vector<uint8_t> shard_ok((size_t)g.n_blocks * g.total, 0);
    for (int s = 0; s < g.total; s++) {
        if (!mask[s]) continue;
        bool ragged_last = g.last_S != g.S;
        int64_t full_nb = ragged_last ? g.n_blocks - 1 : g.n_blocks;
        if (full_nb > 0) {
            std::vector<uint8_t> ok((size_t)full_nb);
            std::vector<uint8_t> want((size_t)full_nb * hsz);
            for (int64_t b = 0; b < full_nb; b++)
                memcpy(want.data() + (size_t)b * hsz,
                       drive_bufs[s] + b * (hsz + g.S), (size_t)hsz);
            mec_status st = mec_bitrot_verify_batch(
                ctx_, algo, (int)full_nb, drive_bufs[s] + hsz, g.S,
                hsz + g.S, want.data(), ok.data());
            if (st != MEC_OK) return st;
            for (int64_t b = 0; b < full_nb; b++)
                shard_ok[b * g.total + s] = ok[b];
        }
        if (ragged_last) {
            const uint8_t *hp =
                drive_bufs[s] + stream_off(g, g.n_blocks - 1, algo);
            uint8_t ok = 0;
            mec_status st = mec_bitrot_verify_batch(ctx_, algo, 1, hp + hsz,
                                                    g.last_S,
                                                    g.last_S + hsz, hp, &ok);
            if (st != MEC_OK) return st;
            shard_ok[(g.n_blocks - 1) * g.total + s] = ok;
        }
    }

    for (int pass = 0; pass < 2; pass++) {
        int64_t b0 = pass == 0 ? 0 : g.n_blocks - 1;
        int64_t n;
        int64_t S;
        bool ragged = g.last_S != g.S;
        if (pass == 0) {
            n = ragged ? g.n_blocks - 1 : g.n_blocks;
            S = g.S;
        } else {
            if (!ragged) break;
            n = 1;
            S = g.last_S;
            b0 = g.n_blocks - 1;
        }
        if (n <= 0) continue;
        std::vector<uint8_t> rows((size_t)n * g.total * S, 0);
        for (int64_t i = 0; i < n; i++)
            for (int s = 0; s < g.total; s++) {
                if (!shard_ok[(b0 + i) * g.total + s]) continue;
                memcpy(rows.data() + ((size_t)i * g.total + s) * S,
                       drive_bufs[s] + stream_off(g, b0 + i, algo) + hsz,
                       (size_t)S);
            }
        /* group blocks by their verified-present mask */
        std::map<std::vector<uint8_t>, std::vector<int64_t>> groups;
        for (int64_t i = 0; i < n; i++) {
            std::vector<uint8_t> m(shard_ok.begin() + (b0 + i) * g.total,
                                   shard_ok.begin() + (b0 + i + 1) * g.total);
            groups[m].push_back(i);
        }
        for (auto &kv : groups) {
            int np = 0;
            for (int s = 0; s < g.total; s++) np += kv.first[s] != 0;
            if (np < d) return MEC_ERR_FILE_CORRUPT;
            if (np == g.total) continue;
            int64_t gn = (int64_t)kv.second.size();
            std::vector<uint8_t> grp((size_t)gn * g.total * S, 0);
            for (int64_t t = 0; t < gn; t++)
                memcpy(grp.data() + (size_t)t * g.total * S,
                       rows.data() + (size_t)kv.second[t] * g.total * S,
                       (size_t)g.total * S);
            mec_status st = mec_reconstruct_batch(
                ctx_, (int)gn, grp.data(), kv.first.data(), S, 0);
            if (st != MEC_OK) return st;
            for (int64_t t = 0; t < gn; t++)
                memcpy(rows.data() + (size_t)kv.second[t] * g.total * S,
                       grp.data() + (size_t)t * g.total * S,
                       (size_t)g.total * S);
        }
        /* hashes for the healed shards, then assemble output streams */
        std::vector<uint8_t> sums((size_t)n * g.total * hsz);
        mec_status st = mec_bitrot_sum_batch(ctx_, algo, (int)(n * g.total),
                                             rows.data(), S, S, sums.data());
        if (st != MEC_OK) return st;
        for (int s = 0; s < g.total; s++) {
            if (!out_bufs || !out_bufs[s]) continue;
            for (int64_t i = 0; i < n; i++) {
                uint8_t *out = out_bufs[s] + stream_off(g, b0 + i, algo);
                memcpy(out, sums.data() + ((size_t)i * g.total + s) * hsz,
                       (size_t)hsz);
                memcpy(out + hsz, rows.data() + ((size_t)i * g.total + s) * S,
                       (size_t)S);
            }
        }
    }
    return MEC_OK;
}

mec_status mec_bitrot_verify_stream(mec_ctx *ctx_, const uint8_t *stream,
                                    int64_t want_size, int64_t part_size,
                                    int algo, const uint8_t *want_sum,
                                    int64_t shard_size) {
    const int hsz = hash_size(algo);
    if (!hsz) return MEC_ERR_INVALID_ARG;
    if (algo != MEC_BITROT_HIGHWAYHASH256S) {
        /* whole-file verification (cmd/bitrot.go:165-175) */
        uint8_t ok = 0;
        mec_status st = mec_bitrot_verify_batch(ctx_, algo, 1, stream,
                                                want_size, want_size,
                                                want_sum, &ok);
        if (st != MEC_OK) return st;
        return ok ? MEC_OK : MEC_ERR_FILE_CORRUPT;
    }
    /* streaming: size check then per-shard verify (cmd/bitrot.go:177-215) */
    if (want_size != mec_bitrot_shard_file_size(part_size, shard_size, algo))
        return MEC_ERR_FILE_CORRUPT;
    int64_t n_shards = ceil_frac(part_size, shard_size);
    if (n_shards == 0) return MEC_OK;
    int64_t last = part_size - (n_shards - 1) * shard_size;
    int64_t n_full = last == shard_size ? n_shards : n_shards - 1;
    if (n_full > 0) {
        std::vector<uint8_t> ok((size_t)n_full);
        std::vector<uint8_t> want((size_t)n_full * hsz);
        for (int64_t i = 0; i < n_full; i++)
            memcpy(want.data() + (size_t)i * hsz,
                   stream + i * (hsz + shard_size), (size_t)hsz);
        mec_status st = mec_bitrot_verify_batch(
            ctx_, algo, (int)n_full, stream + hsz, shard_size,
            hsz + shard_size, want.data(), ok.data());
        if (st != MEC_OK) return st;
        for (int64_t i = 0; i < n_full; i++)
            if (!ok[i]) return MEC_ERR_FILE_CORRUPT;
    }
    if (n_full != n_shards) {
        const uint8_t *hp = stream + n_full * (hsz + shard_size);
        uint8_t ok = 0;
        mec_status st = mec_bitrot_verify_batch(ctx_, algo, 1, hp + hsz, last,
                                                last + hsz, hp, &ok);
        if (st != MEC_OK) return st;
        if (!ok) return MEC_ERR_FILE_CORRUPT;
    }
    return MEC_OK;
}

} /* extern "C" */
