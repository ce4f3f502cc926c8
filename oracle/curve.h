/* curve.h — Pallas/Vesta short-Weierstrass (y^2 = x^3 + 5) point arithmetic.
 *
 * ORACLE TEST INFRASTRUCTURE (see fd.h header note). Restates the group law
 * and GroupEncoding of pasta_curves v0.5.1. Compression convention (x repr
 * with y-oddness in bit 255; identity = 32 zero bytes) is pinned against
 * the reference SRS (tests/test_srs_pin.py).
 *
 * Vesta (base field Fq, scalar field Fp) is the commitment curve of
 * Params<vesta::Affine> (reference proof.rs:25-42); Pallas is its cycle
 * partner. The base field is chosen by the fd_ctx passed in.
 */
#ifndef ORACLE_CURVE_H
#define ORACLE_CURVE_H

#include "fd.h"

/* affine point, coords in Montgomery form; inf flag for identity */
typedef struct {
    fd_limbs x, y;
    int32_t inf;
    int32_t _pad;
} pt_aff;

/* Jacobian point (X/Z^2, Y/Z^3); identity is Z == 0 */
typedef struct {
    fd_limbs x, y, z;
} pt_jac;

static inline void pt_jac_identity(pt_jac* r) {
    fd_zero(r->x); fd_zero(r->y); fd_zero(r->z);
    r->x[0] = 0; /* (0:0:0) treated as identity via z==0 */
}

static inline int pt_jac_is_identity(const pt_jac* p) { return fd_is_zero(p->z); }

static inline void pt_from_aff(pt_jac* r, const pt_aff* a, const fd_ctx* f) {
    if (a->inf) { pt_jac_identity(r); return; }
    fd_copy(r->x, a->x);
    fd_copy(r->y, a->y);
    fd_one_mont(r->z, f);
}

/* dbl-2009-l (a = 0): A=X^2 B=Y^2 C=B^2 D=2((X+B)^2-A-C) E=3A F=E^2 */
static inline void pt_dbl(pt_jac* r, const pt_jac* p, const fd_ctx* f) {
    if (pt_jac_is_identity(p) || fd_is_zero(p->y)) { pt_jac_identity(r); return; }
    fd_limbs A, Bv, C, D, E, F, t;
    fd_sqr(A, p->x, f);
    fd_sqr(Bv, p->y, f);
    fd_sqr(C, Bv, f);
    fd_add(t, p->x, Bv, f);
    fd_sqr(t, t, f);
    fd_sub(t, t, A, f);
    fd_sub(t, t, C, f);
    fd_add(D, t, t, f);
    fd_add(E, A, A, f);
    fd_add(E, E, A, f);
    fd_sqr(F, E, f);
    fd_limbs x3, y3, z3;
    fd_sub(x3, F, D, f);
    fd_sub(x3, x3, D, f);
    fd_add(t, p->y, p->z, f); /* z3 = (Y+Z)^2 - B - Z^2  (= 2YZ) */
    fd_sqr(t, t, f);
    fd_sub(t, t, Bv, f);
    fd_limbs zz;
    fd_sqr(zz, p->z, f);
    fd_sub(z3, t, zz, f);
    fd_sub(t, D, x3, f);
    fd_mul(t, t, E, f);
    fd_limbs c8;
    fd_add(c8, C, C, f);
    fd_add(c8, c8, c8, f);
    fd_add(c8, c8, c8, f);
    fd_sub(y3, t, c8, f);
    fd_copy(r->x, x3); fd_copy(r->y, y3); fd_copy(r->z, z3);
}

/* add-2007-bl full Jacobian add */
static inline void pt_add(pt_jac* r, const pt_jac* p, const pt_jac* q, const fd_ctx* f) {
    if (pt_jac_is_identity(p)) { *r = *q; return; }
    if (pt_jac_is_identity(q)) { *r = *p; return; }
    fd_limbs z1z1, z2z2, u1, u2, s1, s2, t;
    fd_sqr(z1z1, p->z, f);
    fd_sqr(z2z2, q->z, f);
    fd_mul(u1, p->x, z2z2, f);
    fd_mul(u2, q->x, z1z1, f);
    fd_mul(s1, p->y, q->z, f); fd_mul(s1, s1, z2z2, f);
    fd_mul(s2, q->y, p->z, f); fd_mul(s2, s2, z1z1, f);
    if (fd_eq(u1, u2)) {
        if (fd_eq(s1, s2)) { pt_dbl(r, p, f); return; }
        pt_jac_identity(r); return;
    }
    fd_limbs h, i, j, rr, v;
    fd_sub(h, u2, u1, f);
    fd_add(i, h, h, f); fd_sqr(i, i, f);
    fd_mul(j, h, i, f);
    fd_sub(rr, s2, s1, f); fd_add(rr, rr, rr, f);
    fd_mul(v, u1, i, f);
    fd_limbs x3, y3, z3;
    fd_sqr(x3, rr, f);
    fd_sub(x3, x3, j, f);
    fd_sub(x3, x3, v, f);
    fd_sub(x3, x3, v, f);
    fd_sub(t, v, x3, f);
    fd_mul(t, t, rr, f);
    fd_limbs s1j;
    fd_mul(s1j, s1, j, f);
    fd_add(s1j, s1j, s1j, f);
    fd_sub(y3, t, s1j, f);
    fd_add(t, p->z, q->z, f);
    fd_sqr(t, t, f);
    fd_sub(t, t, z1z1, f);
    fd_sub(t, t, z2z2, f);
    fd_mul(z3, t, h, f);
    fd_copy(r->x, x3); fd_copy(r->y, y3); fd_copy(r->z, z3);
}

/* mixed add (q affine): add-2008-g / madd-2007-bl */
static inline void pt_add_aff(pt_jac* r, const pt_jac* p, const pt_aff* q, const fd_ctx* f) {
    if (q->inf) { *r = *p; return; }
    if (pt_jac_is_identity(p)) { pt_from_aff(r, q, f); return; }
    fd_limbs z1z1, u2, s2, t;
    fd_sqr(z1z1, p->z, f);
    fd_mul(u2, q->x, z1z1, f);
    fd_mul(s2, q->y, p->z, f); fd_mul(s2, s2, z1z1, f);
    if (fd_eq(p->x, u2)) {
        if (fd_eq(p->y, s2)) { pt_dbl(r, p, f); return; }
        pt_jac_identity(r); return;
    }
    fd_limbs h, hh, i, j, rr, v;
    fd_sub(h, u2, p->x, f);
    fd_sqr(hh, h, f);
    fd_add(i, hh, hh, f); fd_add(i, i, i, f); /* 4*hh */
    fd_mul(j, h, i, f);
    fd_sub(rr, s2, p->y, f); fd_add(rr, rr, rr, f);
    fd_mul(v, p->x, i, f);
    fd_limbs x3, y3, z3;
    fd_sqr(x3, rr, f);
    fd_sub(x3, x3, j, f);
    fd_sub(x3, x3, v, f);
    fd_sub(x3, x3, v, f);
    fd_sub(t, v, x3, f);
    fd_mul(t, t, rr, f);
    fd_limbs yj;
    fd_mul(yj, p->y, j, f);
    fd_add(yj, yj, yj, f);
    fd_sub(y3, t, yj, f);
    fd_add(t, p->z, h, f);
    fd_sqr(t, t, f);
    fd_sub(t, t, z1z1, f);
    fd_sub(z3, t, hh, f);
    fd_copy(r->x, x3); fd_copy(r->y, y3); fd_copy(r->z, z3);
}

static inline void pt_neg(pt_jac* r, const pt_jac* p, const fd_ctx* f) {
    *r = *p;
    if (!pt_jac_is_identity(p)) fd_neg(r->y, p->y, f);
}

/* scalar mult, scalar = 4x64 LE standard integer (caller pre-reduces) */
static inline void pt_mul(pt_jac* r, const pt_jac* p, const uint64_t k[4], const fd_ctx* f) {
    pt_jac acc, base = *p;
    pt_jac_identity(&acc);
    for (int limb = 0; limb < 4; limb++) {
        uint64_t bits = k[limb];
        for (int b = 0; b < 64; b++) {
            if (bits & 1) pt_add(&acc, &acc, &base, f);
            pt_dbl(&base, &base, f);
            bits >>= 1;
        }
    }
    *r = acc;
}

static inline void pt_to_aff(pt_aff* r, const pt_jac* p, const fd_ctx* f) {
    if (pt_jac_is_identity(p)) {
        fd_zero(r->x); fd_zero(r->y); r->inf = 1; return;
    }
    fd_limbs zi, zi2, zi3;
    fd_inv(zi, p->z, f);
    fd_sqr(zi2, zi, f);
    fd_mul(zi3, zi2, zi, f);
    fd_mul(r->x, p->x, zi2, f);
    fd_mul(r->y, p->y, zi3, f);
    r->inf = 0;
}

/* batch affine normalization: Montgomery's trick, n inversions -> 1 */
static inline void pt_to_aff_batch(pt_aff* out, const pt_jac* in, long n, const fd_ctx* f) {
    if (n <= 0) return;
    fd_limbs* prefix = (fd_limbs*)__builtin_malloc(sizeof(fd_limbs) * (size_t)n);
    fd_limbs run;
    fd_one_mont(run, f);
    for (long i = 0; i < n; i++) {
        fd_copy(prefix[i], run);
        if (!pt_jac_is_identity(&in[i])) fd_mul(run, run, in[i].z, f);
    }
    fd_limbs inv_all;
    fd_inv(inv_all, run, f);
    for (long i = n - 1; i >= 0; i--) {
        if (pt_jac_is_identity(&in[i])) {
            fd_zero(out[i].x); fd_zero(out[i].y); out[i].inf = 1;
            continue;
        }
        fd_limbs zi, zi2, zi3;
        fd_mul(zi, inv_all, prefix[i], f);
        fd_mul(inv_all, inv_all, in[i].z, f);
        fd_sqr(zi2, zi, f);
        fd_mul(zi3, zi2, zi, f);
        fd_mul(out[i].x, in[i].x, zi2, f);
        fd_mul(out[i].y, in[i].y, zi3, f);
        out[i].inf = 0;
    }
    __builtin_free(prefix);
}

/* pasta GroupEncoding: 32-byte compressed. Returns 0 ok, -1 invalid. */
static inline int pt_decompress(pt_aff* r, const uint8_t in[32], const fd_ctx* f) {
    uint8_t buf[32];
    memcpy(buf, in, 32);
    int sign = buf[31] >> 7;
    buf[31] &= 0x7F;
    int allz = 1;
    for (int i = 0; i < 32; i++) if (buf[i]) { allz = 0; break; }
    if (allz && !sign) { fd_zero(r->x); fd_zero(r->y); r->inf = 1; return 0; }
    fd_limbs x;
    if (fd_from_bytes(x, buf, f)) return -1;
    fd_limbs rhs, b5;
    fd_sqr(rhs, x, f);
    fd_mul(rhs, rhs, x, f);
    uint64_t five[4] = {5, 0, 0, 0};
    fd_to_mont(b5, five, f);
    fd_add(rhs, rhs, b5, f);
    fd_limbs y;
    if (!fd_sqrt(y, rhs, f)) return -1;
    if (fd_is_odd(y, f) != sign) fd_neg(y, y, f);
    fd_copy(r->x, x);
    fd_copy(r->y, y);
    r->inf = 0;
    return 0;
}

static inline void pt_compress(uint8_t out[32], const pt_aff* p, const fd_ctx* f) {
    if (p->inf) { memset(out, 0, 32); return; }
    fd_to_bytes(out, p->x, f);
    out[31] |= (uint8_t)(fd_is_odd(p->y, f) << 7);
}

#endif
