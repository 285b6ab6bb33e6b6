/* ORACLE — TEST INFRASTRUCTURE ONLY (see oracle.h).
 * Private HighwayHash state shared between the portable restatement
 * (hh256.c, the parity checker) and the SIMD bench leg (simd.c).  The
 * SIMD leg reuses the scalar remainder/finalization so the two differ
 * only in the 32-B packet main loop. */
#ifndef MO_HH_INTERNAL_H
#define MO_HH_INTERNAL_H
#include <stddef.h>
#include <stdint.h>

typedef struct {
    uint64_t v0[4], v1[4], mul0[4], mul1[4];
} mo_hh_state;

void mo_hh_reset_(mo_hh_state *s, const uint8_t key32[32]);
void mo_hh_update_packet_(mo_hh_state *s, const uint8_t *packet);
void mo_hh_finish_(mo_hh_state *s, const uint8_t *tail, size_t tail_len,
                   uint8_t out[32]);

#endif
