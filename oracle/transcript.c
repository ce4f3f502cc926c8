/* transcript.c — restatement of halo2_proofs' Blake2bWrite/Blake2bRead
 * Fiat–Shamir transcript over vesta::Affine (reference call sites:
 * taiga_halo2/src/proof.rs:32,52; the transcript implementation itself is
 * in the un-vendored halo2_proofs dep — SURVEY.md §8c; base: the public
 * zcash halo2_proofs 0.3 transcript module).
 *
 * ORACLE TEST INFRASTRUCTURE (see fd.h header note).
 *
 * Conventions restated (flagged in DESIGN.md §parity-assumptions):
 *   state = Blake2b-512 keyed with personalization "Halo2-Transcript"
 *   common_point:  update([0x01] ‖ x.to_repr() ‖ y.to_repr())   (affine)
 *   common_scalar: update([0x02] ‖ s.to_repr())
 *   squeeze_challenge: update([0x00]); digest = finalize(clone of state);
 *     challenge scalar = from_uniform_512(digest)  (in the curve's SCALAR
 *     field: Fp for vesta::Affine)
 *   write_point  = common_point  + append 32-B compressed point to proof
 *   write_scalar = common_scalar + append 32-B repr to proof
 */
#include "fd.h"
#include "curve.h"
#include <stdlib.h>
#include <string.h>

/* blake2b state from blake2b.c */
typedef struct {
    uint64_t h[8];
    uint64_t t[2];
    uint8_t buf[128];
    size_t buflen;
    size_t outlen;
} blake2b_state;
extern void orc_blake2b_init(blake2b_state* S, size_t outlen, const uint8_t* personal);
extern void orc_blake2b_update(blake2b_state* S, const uint8_t* in, size_t inlen);
extern void orc_blake2b_final(blake2b_state* S, uint8_t* out);

typedef struct {
    blake2b_state st;
    uint8_t* proof;    /* write mode: growing buffer; read mode: input */
    size_t len, cap;   /* write */
    size_t rpos, rlen; /* read */
    int reading;
} tg_transcript;

void orc_ts_init_write(tg_transcript* t) {
    memset(t, 0, sizeof(*t));
    orc_blake2b_init(&t->st, 64, (const uint8_t*)"Halo2-Transcript");
    t->cap = 8192;
    t->proof = (uint8_t*)malloc(t->cap);
    t->len = 0;
    t->reading = 0;
}

void orc_ts_init_read(tg_transcript* t, const uint8_t* proof, size_t len) {
    memset(t, 0, sizeof(*t));
    orc_blake2b_init(&t->st, 64, (const uint8_t*)"Halo2-Transcript");
    t->proof = (uint8_t*)proof;
    t->rlen = len;
    t->rpos = 0;
    t->reading = 1;
}

static void ts_append(tg_transcript* t, const uint8_t* b, size_t n) {
    if (t->len + n > t->cap) {
        while (t->len + n > t->cap) t->cap *= 2;
        t->proof = (uint8_t*)realloc(t->proof, t->cap);
    }
    memcpy(t->proof + t->len, b, n);
    t->len += n;
}

/* affine Vesta point (Mont coords, never identity on a transcript) */
void orc_ts_common_point(tg_transcript* t, const pt_aff* p) {
    uint8_t pre = 1, xb[32], yb[32];
    orc_blake2b_update(&t->st, &pre, 1);
    fd_to_bytes(xb, p->x, &FD_Q);
    fd_to_bytes(yb, p->y, &FD_Q);
    orc_blake2b_update(&t->st, xb, 32);
    orc_blake2b_update(&t->st, yb, 32);
}

/* scalar in Fp (Mont) */
void orc_ts_common_scalar(tg_transcript* t, const uint64_t s[4]) {
    uint8_t pre = 2, sb[32];
    orc_blake2b_update(&t->st, &pre, 1);
    fd_to_bytes(sb, s, &FD_P);
    orc_blake2b_update(&t->st, sb, 32);
}

int orc_ts_write_point(tg_transcript* t, const pt_aff* p) {
    if (p->inf) return -1;
    orc_ts_common_point(t, p);
    uint8_t cb[32];
    pt_compress(cb, p, &FD_Q);
    ts_append(t, cb, 32);
    return 0;
}

void orc_ts_write_scalar(tg_transcript* t, const uint64_t s[4]) {
    orc_ts_common_scalar(t, s);
    uint8_t sb[32];
    fd_to_bytes(sb, s, &FD_P);
    ts_append(t, sb, 32);
}

int orc_ts_read_point(tg_transcript* t, pt_aff* p) {
    if (t->rpos + 32 > t->rlen) return -1;
    if (pt_decompress(p, t->proof + t->rpos, &FD_Q)) return -1;
    if (p->inf) return -1;
    t->rpos += 32;
    orc_ts_common_point(t, p);
    return 0;
}

int orc_ts_read_scalar(tg_transcript* t, uint64_t s[4]) {
    if (t->rpos + 32 > t->rlen) return -1;
    if (fd_from_bytes(s, t->proof + t->rpos, &FD_P)) return -1;
    t->rpos += 32;
    orc_ts_common_scalar(t, s);
    return 0;
}

/* squeeze a challenge scalar in Fp (Mont out) */
void orc_ts_squeeze(tg_transcript* t, uint64_t out[4]) {
    uint8_t pre = 0;
    orc_blake2b_update(&t->st, &pre, 1);
    blake2b_state clone = t->st;
    uint8_t dig[64];
    orc_blake2b_final(&clone, dig);
    /* from_uniform_512 (same identity as orc_drbg_field) */
    uint64_t lo[4], hi[4];
    memcpy(lo, dig, 32);
    memcpy(hi, dig + 32, 32);
    fd_limbs lom, him, tt;
    const fd_ctx* f = &FD_P;
    fd_mul(lom, lo, f->r2, f);
    fd_mul(him, hi, f->r2, f);
    fd_mul(tt, him, f->r2, f);
    fd_add(out, lom, tt, f);
}
