/* Product host-side GF(2^8) + matrix math for the MI355X erasure path.
 *
 * Builds the systematic-Vandermonde encode matrix ("rs-vandermonde",
 * reference cmd/erasure-metadata.go:40; construction per the published
 * klauspost/reedsolomon v1.12.4 default: vm[r][c] = r^c over GF(2^8)/0x11D,
 * encode = vm * inv(vm[0:d][0:d])) and the per-erasure-pattern decode
 * matrices (ReconstructData semantics, cmd/erasure-coding.go:94-113).
 * Host-only: the GPU kernels receive the coefficient rows.
 */
#ifndef MEC_GF_HOST_H
#define MEC_GF_HOST_H

#include <cstdint>
#include <cstddef>

namespace mec {

constexpr int kMaxShards = 256;

uint8_t gf_mul(uint8_t a, uint8_t b);
uint8_t gf_exp(uint8_t a, int n);

/* encode matrix, (d+p) x d row-major into out.  Returns false on bad dims
 * or singular top square (cannot happen for valid Vandermonde dims). */
bool build_encode_matrix(int d, int p, uint8_t *out);

/* Decode plan for one erasure pattern.
 * present: d+p flags.  Fills:
 *   src_idx[0..d): the first d present shard rows (klauspost order)
 *   dst_idx[0..*n_dst): missing rows to rebuild (data only, or all)
 *   dec[(*n_dst) x d]: coefficient rows, row t applies over src shards.
 * Returns false if fewer than d present or singular. */
bool build_decode_plan(const uint8_t *enc_matrix, int d, int p,
                       const uint8_t *present, int data_only, int *src_idx,
                       int *dst_idx, int *n_dst, uint8_t *dec);

} // namespace mec
#endif
