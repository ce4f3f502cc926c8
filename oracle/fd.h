/* fd.h — 4x64 Montgomery field arithmetic for the Pasta fields (Fp, Fq).
 *
 * ORACLE TEST INFRASTRUCTURE. This is the CPU restatement of the arithmetic
 * of the pasta_curves crate v0.5.1 (un-vendored git dep of the reference,
 * /root/reference/Cargo.toml:10, taiga_halo2/Cargo.toml:9 — SURVEY.md §8c).
 * Only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may
 * link or call this library; the product path (taiga_amd/libtaiga_gpu.so)
 * must fail loudly rather than fall back to it.
 *
 * Representation: little-endian 4x64 limbs, Montgomery form (aR mod m,
 * R = 2^256). Canonical byte I/O matches pasta_curves to_repr()/from_repr()
 * (32-byte little-endian of the standard representative) — pinned against
 * the reference SRS bytes by tests/test_srs_pin.py.
 */
#ifndef ORACLE_FD_H
#define ORACLE_FD_H

#include <stdint.h>
#include <string.h>

typedef uint64_t fd_limbs[4];

typedef struct {
    fd_limbs mod;
    fd_limbs r2;       /* R^2 mod m  (to_mont) */
    fd_limbs r3;       /* R^3 mod m */
    uint64_t inv;      /* -m^{-1} mod 2^64 */
    fd_limbs root;     /* order-2^32 root of unity 5^((m-1)/2^32), STANDARD repr */
    fd_limbs root_inv; /* its inverse, STANDARD repr */
    fd_limbs t_odd;    /* (m-1)/2^32 */
    fd_limbs t1_2;     /* ((m-1)/2^32 + 1)/2 */
} fd_ctx;

typedef unsigned __int128 u128;

static inline int fd_is_zero(const uint64_t a[4]) {
    return (a[0] | a[1] | a[2] | a[3]) == 0;
}

static inline int fd_eq(const uint64_t a[4], const uint64_t b[4]) {
    return a[0] == b[0] && a[1] == b[1] && a[2] == b[2] && a[3] == b[3];
}

static inline void fd_copy(uint64_t r[4], const uint64_t a[4]) {
    r[0] = a[0]; r[1] = a[1]; r[2] = a[2]; r[3] = a[3];
}

static inline void fd_zero(uint64_t r[4]) { r[0] = r[1] = r[2] = r[3] = 0; }

/* r = r - m if r >= m */
static inline void fd_reduce_once(uint64_t r[4], const fd_ctx* f) {
    uint64_t t[4];
    uint64_t borrow = 0;
    for (int i = 0; i < 4; i++) {
        u128 d = (u128)r[i] - f->mod[i] - borrow;
        t[i] = (uint64_t)d;
        borrow = (d >> 64) ? 1 : 0; /* underflow wraps: high bits all ones */
    }
    if (!borrow) { fd_copy(r, t); }
}

static inline void fd_add(uint64_t r[4], const uint64_t a[4], const uint64_t b[4], const fd_ctx* f) {
    uint64_t carry = 0;
    for (int i = 0; i < 4; i++) {
        u128 s = (u128)a[i] + b[i] + carry;
        r[i] = (uint64_t)s;
        carry = (uint64_t)(s >> 64);
    }
    /* moduli < 2^255 so no overflow past 4 limbs when a,b < m */
    fd_reduce_once(r, f);
}

static inline void fd_sub(uint64_t r[4], const uint64_t a[4], const uint64_t b[4], const fd_ctx* f) {
    uint64_t borrow = 0, t[4];
    for (int i = 0; i < 4; i++) {
        u128 d = (u128)a[i] - b[i] - borrow;
        t[i] = (uint64_t)d;
        borrow = (d >> 64) ? 1 : 0;
    }
    if (borrow) {
        uint64_t carry = 0;
        for (int i = 0; i < 4; i++) {
            u128 s = (u128)t[i] + f->mod[i] + carry;
            t[i] = (uint64_t)s;
            carry = (uint64_t)(s >> 64);
        }
    }
    fd_copy(r, t);
}

static inline void fd_neg(uint64_t r[4], const uint64_t a[4], const fd_ctx* f) {
    if (fd_is_zero(a)) { fd_zero(r); return; }
    uint64_t borrow = 0;
    for (int i = 0; i < 4; i++) {
        u128 d = (u128)f->mod[i] - a[i] - borrow;
        r[i] = (uint64_t)d;
        borrow = (d >> 64) ? 1 : 0;
    }
}

/* CIOS Montgomery multiplication: r = a*b*R^{-1} mod m */
static inline void fd_mul(uint64_t r[4], const uint64_t a[4], const uint64_t b[4], const fd_ctx* f) {
    uint64_t t[6] = {0, 0, 0, 0, 0, 0};
    for (int i = 0; i < 4; i++) {
        uint64_t carry = 0;
        for (int j = 0; j < 4; j++) {
            u128 s = (u128)a[j] * b[i] + t[j] + carry;
            t[j] = (uint64_t)s;
            carry = (uint64_t)(s >> 64);
        }
        u128 s = (u128)t[4] + carry;
        t[4] = (uint64_t)s;
        t[5] = (uint64_t)(s >> 64);

        uint64_t m = t[0] * f->inv;
        u128 c = (u128)m * f->mod[0] + t[0];
        carry = (uint64_t)(c >> 64);
        for (int j = 1; j < 4; j++) {
            c = (u128)m * f->mod[j] + t[j] + carry;
            t[j - 1] = (uint64_t)c;
            carry = (uint64_t)(c >> 64);
        }
        c = (u128)t[4] + carry;
        t[3] = (uint64_t)c;
        t[4] = t[5] + (uint64_t)(c >> 64);
        t[5] = 0;
    }
    fd_copy(r, t);
    fd_reduce_once(r, f);
}

static inline void fd_sqr(uint64_t r[4], const uint64_t a[4], const fd_ctx* f) {
    fd_mul(r, a, a, f);
}

/* to Montgomery form: r = a * R mod m  (a in standard repr) */
static inline void fd_to_mont(uint64_t r[4], const uint64_t a[4], const fd_ctx* f) {
    fd_mul(r, a, f->r2, f);
}

/* from Montgomery form: r = a * R^{-1} mod m */
static inline void fd_from_mont(uint64_t r[4], const uint64_t a[4], const fd_ctx* f) {
    uint64_t one[4] = {1, 0, 0, 0};
    fd_mul(r, a, one, f);
}

static inline void fd_one_mont(uint64_t r[4], const fd_ctx* f) {
    uint64_t one[4] = {1, 0, 0, 0};
    fd_to_mont(r, one, f);
}

/* r = base^e mod m (base in Mont form, e standard 4x64 integer), result Mont */
static inline void fd_pow(uint64_t r[4], const uint64_t base[4], const uint64_t e[4], const fd_ctx* f) {
    uint64_t acc[4], b[4];
    fd_one_mont(acc, f);
    fd_copy(b, base);
    for (int limb = 0; limb < 4; limb++) {
        uint64_t bits = e[limb];
        for (int i = 0; i < 64; i++) {
            /* left-to-right would need known top bit; do right-to-left */
            if (bits & 1) fd_mul(acc, acc, b, f);
            fd_sqr(b, b, f);
            bits >>= 1;
        }
    }
    fd_copy(r, acc);
}

/* r = a^{-1} via Fermat: a^(m-2). a in Mont, result Mont. Zero maps to zero. */
static inline void fd_inv(uint64_t r[4], const uint64_t a[4], const fd_ctx* f) {
    uint64_t e[4];
    /* e = m - 2 */
    uint64_t borrow = 0;
    uint64_t two[4] = {2, 0, 0, 0};
    for (int i = 0; i < 4; i++) {
        u128 d = (u128)f->mod[i] - two[i] - borrow;
        e[i] = (uint64_t)d;
        borrow = (d >> 64) ? 1 : 0;
    }
    fd_pow(r, a, e, f);
}

/* canonical byte I/O: 32-byte little-endian standard repr (pasta to_repr) */
static inline void fd_to_bytes(uint8_t out[32], const uint64_t a[4], const fd_ctx* f) {
    uint64_t s[4];
    fd_from_mont(s, a, f);
    memcpy(out, s, 32);
}

/* returns 0 on success, -1 if not canonical (>= m) */
static inline int fd_from_bytes(uint64_t r[4], const uint8_t in[32], const fd_ctx* f) {
    uint64_t s[4];
    memcpy(s, in, 32);
    /* check s < m */
    for (int i = 3; i >= 0; i--) {
        if (s[i] < f->mod[i]) break;
        if (s[i] > f->mod[i]) return -1;
        if (i == 0) return -1; /* equal to m */
    }
    fd_to_mont(r, s, f);
    return 0;
}

/* Tonelli–Shanks square root for m ≡ 1 (mod 2^32), S=32.
 * a Mont in, r Mont out. Returns 1 if square (r set), 0 otherwise. */
static inline int fd_sqrt(uint64_t r[4], const uint64_t a[4], const fd_ctx* f) {
    if (fd_is_zero(a)) { fd_zero(r); return 1; }
    uint64_t z[4], u[4], x[4], b[4], tmp[4];
    /* z = ROOT (Mont), u = a^t, x = a^((t+1)/2) */
    fd_to_mont(z, f->root, f);
    fd_pow(u, a, f->t_odd, f);
    fd_pow(x, a, f->t1_2, f);
    int m = 32;
    uint64_t onem[4];
    fd_one_mont(onem, f);
    while (!fd_eq(u, onem)) {
        /* find least i with u^(2^i) == 1 */
        int i = 0;
        fd_copy(tmp, u);
        while (!fd_eq(tmp, onem)) {
            fd_sqr(tmp, tmp, f);
            i++;
            if (i > m) return 0; /* not a square */
        }
        if (i == m) return 0;
        fd_copy(b, z);
        for (int j = 0; j < m - i - 1; j++) fd_sqr(b, b, f);
        m = i;
        fd_sqr(z, b, f);
        fd_mul(u, u, z, f);
        fd_mul(x, x, b, f);
    }
    fd_copy(r, x);
    return 1;
}

/* is the STANDARD representative odd? (pasta sign bit for compression) */
static inline int fd_is_odd(const uint64_t a[4], const fd_ctx* f) {
    uint64_t s[4];
    fd_from_mont(s, a, f);
    return (int)(s[0] & 1);
}

extern const fd_ctx FD_P; /* Fp: pallas::Base = vesta::Scalar */
extern const fd_ctx FD_Q; /* Fq: vesta::Base  = pallas::Scalar */

#endif
