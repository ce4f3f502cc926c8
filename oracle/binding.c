/* binding.c — RedDSA binding signatures over Pallas + transaction digest.
 *
 * ORACLE TEST INFRASTRUCTURE (usage contract in fd.h). Independent twin of
 * the product's binding_sig.hpp: restates the PUBLIC RedDSA algorithm
 * (Zcash §5.4.7 / the reddsa crate) as instantiated by the reference's
 * TaigaBinding (taiga_halo2/src/binding_signature.rs:23-31 — H* =
 * BLAKE2b-512 "Taiga_RedPallasH" wide-reduced into the Pallas scalar
 * field) and Transaction::digest (transaction.rs:116-158 — BLAKE2b-256
 * "TxBindingSigHash" over nullifiers ‖ cms ‖ delta commitments ‖
 * anchors). Basepoint: Pallas generator for round 1 (the reference's
 * sinsemilla-derived R generator, constant.rs:160, lands with the
 * group-hash chain — DESIGN.md §6).
 */
#include <stdint.h>
#include <string.h>

#include "curve.h"
#include "fd.h"

extern const fd_ctx FD_P, FD_Q;

typedef struct {
    uint8_t h[64];
} b2b_out;

void orc_blake2b(const uint8_t* in, long inlen, const uint8_t* personal16_or_null,
                 long outlen, uint8_t* out);

typedef struct blake2b_state_fwd blake2b_state_fwd;
/* use the streaming API via a local concat buffer instead (messages are
 * small: 80 + 32 + digest-size bytes) */

void orc_drbg_raw(const uint8_t seed[32], long n, uint8_t* out);

#define FP (&FD_P)
#define FQ (&FD_Q)

static void pallas_gen(pt_jac* g) {
    /* RESOURCE_COMMIT_DOMAIN.R() (constant.rs:160): sinsemilla
     * CommitDomain("Taiga-NoteCommit") R, derived via the restated pasta
     * group hash and pinned byte-for-byte against the reference's R_U/R_Z
     * window tables (tests/test_fixed_base_tables.py). Standard limbs. */
    static const uint64_t RX[4] = {0x8802ca95558f33acULL, 0x18eaf0144b63a217ULL,
                                   0xae99b6368e616ee5ULL, 0x108bffda6aff53a7ULL};
    static const uint64_t RY[4] = {0x8c0b30657333c5e1ULL, 0xb8b4f7c83326a380ULL,
                                   0x3ebb783b59d92092ULL, 0x1319b788fe5fec16ULL};
    pt_aff a;
    fd_to_mont(a.x, RX, FP);
    fd_to_mont(a.y, RY, FP);
    a.inf = 0;
    pt_from_aff(g, &a, FP);
}

/* H*(a ‖ b ‖ c) -> Fq scalar (Mont): BLAKE2b-512 + wide reduction */
static void hstar(uint64_t out_mont[4], const uint8_t* a, size_t alen,
                  const uint8_t* b, size_t blen, const uint8_t* c, size_t clen) {
    uint8_t buf[65536];
    size_t n = 0;
    memcpy(buf + n, a, alen); n += alen;
    memcpy(buf + n, b, blen); n += blen;
    if (clen) { memcpy(buf + n, c, clen); n += clen; }
    uint8_t dig[64];
    orc_blake2b(buf, (long)n, (const uint8_t*)"Taiga_RedPallasH", 64, dig);
    /* wide reduce: lo*R + hi*R^2 (Mont) = lo + hi*2^256 mod q */
    fd_limbs lo, hi, lom, him;
    memcpy(lo, dig, 32);
    memcpy(hi, dig + 32, 32);
    fd_mul(lom, lo, FD_Q.r2, FQ);
    fd_mul(him, hi, FD_Q.r2, FQ);
    fd_mul(him, him, FD_Q.r2, FQ);
    fd_add(out_mont, lom, him, FQ);
}

static int scalar_from_bytes(uint64_t out_mont[4], const uint8_t in[32]) {
    fd_limbs v;
    memcpy(v, in, 32);
    return fd_from_bytes(out_mont, in, FQ) == 0 ? 0 : -1;
    (void)v;
}

/* [s]P, s Mont Fq */
static void pmul(pt_jac* r, const pt_jac* p, const uint64_t s_mont[4]) {
    fd_limbs s;
    fd_from_mont(s, s_mont, FQ);
    pt_mul(r, p, s, FP);
}

int orc_binding_vk(const uint8_t sk[32], uint8_t vk_out[32]) {
    fd_limbs s;
    if (scalar_from_bytes(s, sk)) return -1;
    pt_jac g, v;
    pallas_gen(&g);
    pmul(&v, &g, s);
    pt_aff a;
    pt_to_aff(&a, &v, FP);
    pt_compress(vk_out, &a, FP);
    return 0;
}

int orc_delta_commit(const uint8_t r[32], uint8_t cv_out[32]) {
    return orc_binding_vk(r, cv_out);
}

int orc_binding_sign(const uint8_t sk[32], const uint8_t* msg, long msg_len,
                     const uint8_t rng_seed[32], uint8_t sig_out[64]) {
    if (msg_len < 0 || msg_len > 60000) return -2;
    fd_limbs s;
    if (scalar_from_bytes(s, sk)) return -1;
    uint8_t vk_bytes[32];
    orc_binding_vk(sk, vk_bytes);
    uint8_t T[80];
    orc_drbg_raw(rng_seed, 80, T);
    uint8_t tv[112];
    memcpy(tv, T, 80);
    memcpy(tv + 80, vk_bytes, 32);
    fd_limbs r;
    hstar(r, tv, 112, msg ? msg : (const uint8_t*)"", (size_t)msg_len, NULL, 0);
    pt_jac g, R;
    pallas_gen(&g);
    pmul(&R, &g, r);
    pt_aff Ra;
    pt_to_aff(&Ra, &R, FP);
    pt_compress(sig_out, &Ra, FP);
    uint8_t rv[64];
    memcpy(rv, sig_out, 32);
    memcpy(rv + 32, vk_bytes, 32);
    fd_limbs c, S;
    hstar(c, rv, 64, msg ? msg : (const uint8_t*)"", (size_t)msg_len, NULL, 0);
    fd_mul(S, c, s, FQ);
    fd_add(S, S, r, FQ);
    fd_limbs S_std;
    fd_from_mont(S_std, S, FQ);
    memcpy(sig_out + 32, S_std, 32);
    return 0;
}

int orc_binding_verify(const uint8_t vk_bytes[32], const uint8_t* msg, long msg_len,
                       const uint8_t sig[64]) {
    if (msg_len < 0 || msg_len > 60000) return -2;
    pt_aff vk_a, R_a;
    if (pt_decompress(&vk_a, vk_bytes, FP)) return -1;
    if (pt_decompress(&R_a, sig, FP)) return -1;
    fd_limbs S;
    if (scalar_from_bytes(S, sig + 32)) return -1;
    uint8_t rv[64];
    memcpy(rv, sig, 32);
    memcpy(rv + 32, vk_bytes, 32);
    fd_limbs c;
    hstar(c, rv, 64, msg ? msg : (const uint8_t*)"", (size_t)msg_len, NULL, 0);
    pt_jac g, lhs, vkj, cvk, R, rhs;
    pallas_gen(&g);
    pmul(&lhs, &g, S);
    pt_from_aff(&vkj, &vk_a, FP);
    pmul(&cvk, &vkj, c);
    pt_from_aff(&R, &R_a, FP);
    pt_add(&rhs, &R, &cvk, FP);
    pt_aff la, ra;
    pt_to_aff(&la, &lhs, FP);
    pt_to_aff(&ra, &rhs, FP);
    if (la.inf != ra.inf) return -1;
    if (!la.inf && (!fd_eq(la.x, ra.x) || !fd_eq(la.y, ra.y))) return -1;
    return 0;
}

int orc_binding_vk_from_deltas(const uint8_t* deltas, long n, uint8_t vk_out[32]) {
    pt_jac acc;
    pt_jac_identity(&acc);
    for (long i = 0; i < n; i++) {
        pt_aff a;
        if (pt_decompress(&a, deltas + 32 * i, FP)) return -1;
        pt_jac p;
        pt_from_aff(&p, &a, FP);
        pt_add(&acc, &acc, &p, FP);
    }
    pt_aff out;
    pt_to_aff(&out, &acc, FP);
    pt_compress(vk_out, &out, FP);
    return 0;
}

int orc_tx_digest(const uint8_t* nfs, long n_nf, const uint8_t* cms, long n_cm,
                  const uint8_t* deltas, long n_delta, const uint8_t* anchors,
                  long n_anchor, uint8_t out[32]) {
    uint8_t buf[65536];
    size_t n = 0;
    memcpy(buf + n, nfs, 32 * (size_t)n_nf); n += 32 * (size_t)n_nf;
    memcpy(buf + n, cms, 32 * (size_t)n_cm); n += 32 * (size_t)n_cm;
    memcpy(buf + n, deltas, 32 * (size_t)n_delta); n += 32 * (size_t)n_delta;
    memcpy(buf + n, anchors, 32 * (size_t)n_anchor); n += 32 * (size_t)n_anchor;
    orc_blake2b(buf, (long)n, (const uint8_t*)"TxBindingSigHash", 32, out);
    return 0;
}
