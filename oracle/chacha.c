/* chacha.c — ChaCha20 (RFC 8439) keystream + the deterministic field-element
 * DRBG used for all prover randomness (blinds, blinding rows).
 *
 * ORACLE TEST INFRASTRUCTURE (see fd.h header note).
 *
 * The reference draws prover randomness from a caller-supplied RngCore
 * (OsRng in every reference test/bench — taiga_halo2/benches/
 * compliance_proof.rs:22, so reference proofs are never byte-reproducible).
 * BASELINE.md fixes the seeded-replacement convention: ChaCha20, key =
 * 32-byte seed, 12-byte zero nonce, counter from 0. A field element draw
 * takes the next 64 keystream bytes as a little-endian 512-bit integer
 * reduced mod m (the from_uniform_bytes convention) — identical in the
 * oracle, the product host shim and the Python test harness.
 */
#include "fd.h"
#include <string.h>

typedef struct {
    uint32_t key[8];
    uint32_t counter;
    uint8_t buf[64];
    int pos; /* 64 = empty */
} tg_drbg;

static inline uint32_t rotl32_(uint32_t x, int n) { return (x << n) | (x >> (32 - n)); }

static void chacha_block(const uint32_t key[8], uint32_t counter, uint8_t out[64]) {
    uint32_t st[16] = {0x61707865, 0x3320646e, 0x79622d32, 0x6b206574,
                       key[0], key[1], key[2], key[3], key[4], key[5], key[6], key[7],
                       counter, 0, 0, 0};
    uint32_t x[16];
    memcpy(x, st, sizeof(x));
#define QR(a, b, c, d)                                  \
    x[a] += x[b]; x[d] = rotl32_(x[d] ^ x[a], 16);      \
    x[c] += x[d]; x[b] = rotl32_(x[b] ^ x[c], 12);      \
    x[a] += x[b]; x[d] = rotl32_(x[d] ^ x[a], 8);       \
    x[c] += x[d]; x[b] = rotl32_(x[b] ^ x[c], 7);
    for (int i = 0; i < 10; i++) {
        QR(0, 4, 8, 12) QR(1, 5, 9, 13) QR(2, 6, 10, 14) QR(3, 7, 11, 15)
        QR(0, 5, 10, 15) QR(1, 6, 11, 12) QR(2, 7, 8, 13) QR(3, 4, 9, 14)
    }
#undef QR
    for (int i = 0; i < 16; i++) {
        uint32_t v = x[i] + st[i];
        out[4 * i] = (uint8_t)v;
        out[4 * i + 1] = (uint8_t)(v >> 8);
        out[4 * i + 2] = (uint8_t)(v >> 16);
        out[4 * i + 3] = (uint8_t)(v >> 24);
    }
}

void orc_drbg_init(tg_drbg* d, const uint8_t seed[32]) {
    memset(d, 0, sizeof(*d));
    for (int i = 0; i < 8; i++) {
        d->key[i] = (uint32_t)seed[4 * i] | ((uint32_t)seed[4 * i + 1] << 8) |
                    ((uint32_t)seed[4 * i + 2] << 16) | ((uint32_t)seed[4 * i + 3] << 24);
    }
    d->counter = 0;
    d->pos = 64;
}

void orc_drbg_bytes(tg_drbg* d, uint8_t* out, size_t n) {
    while (n) {
        if (d->pos == 64) {
            chacha_block(d->key, d->counter++, d->buf);
            d->pos = 0;
        }
        size_t take = 64 - (size_t)d->pos;
        if (take > n) take = n;
        memcpy(out, d->buf + d->pos, take);
        d->pos += (int)take;
        out += take;
        n -= take;
    }
}

/* 512-bit LE wide reduction mod m: r = lo + hi * 2^256 = lo + hi * R (so in
 * Montgomery terms: to_mont(lo) + to_mont(hi)*R ... computed directly as
 * r_mont = lo*R + hi*R^2 ( = to_mont(lo) + mul(to_mont(hi), R) ) — we just
 * do it with fd ops: r = to_mont(lo) + to_mont(hi) * to_mont(R2? ) careful:
 * value v = lo + hi*2^256. to_mont(v) = v*R = lo*R + hi*R*2^256 =
 * to_mont(lo) + hi*R^2*... : hi*2^256*R mod m = hi*R^2 = mont_mul(to_mont(hi), to_mont(R))
 * simpler: to_mont(hi) = hi*R; mont_mul(to_mont(hi), R2_mont=R^2*R? ) — use:
 * t = to_mont(hi); t2 = fd_mul(t, r2_as_element?) where r2 field element
 * (R^2 std) in Mont form is R^3... Direct approach: result_mont =
 * to_mont(lo) + fd_mul(to_mont(hi), to_mont(2^256 mod m)).  2^256 mod m is
 * just to_mont(1) read as std? 2^256 mod m = R mod m = from_mont(r2) ...
 * We avoid confusion by computing with the identity:
 *   to_mont(v) = to_mont(lo) + mont_mul(to_mont(hi), R2)   since
 *   mont_mul(hi*R, R^2) = hi*R^2*R^{-1}*R ... = hi*R*R = (hi*2^256)*R. ✓
 */
void orc_drbg_field(tg_drbg* d, int fid, uint64_t out_mont[4]) {
    const fd_ctx* f = fid ? &FD_Q : &FD_P;
    uint8_t buf[64];
    orc_drbg_bytes(d, buf, 64);
    uint64_t lo[4], hi[4];
    memcpy(lo, buf, 32);
    memcpy(hi, buf + 32, 32);
    fd_limbs lom, him, t;
    /* to_mont via fd_mul(x, R2) works for ANY 256-bit x (CIOS handles x < 2^256) */
    fd_mul(lom, lo, f->r2, f);
    fd_mul(him, hi, f->r2, f);
    fd_mul(t, him, f->r2, f); /* = hi * 2^256 * R */
    fd_add(out_mont, lom, t, f);
}

/* ctypes-facing: draw n field elements as canonical reprs */
void orc_drbg_fields(const uint8_t seed[32], int fid, long n, uint8_t* out) {
    const fd_ctx* f = fid ? &FD_Q : &FD_P;
    tg_drbg d;
    orc_drbg_init(&d, seed);
    for (long i = 0; i < n; i++) {
        fd_limbs v;
        orc_drbg_field(&d, fid, v);
        fd_to_bytes(out + 32 * i, v, f);
    }
}

void orc_drbg_raw(const uint8_t seed[32], long n, uint8_t* out) {
    tg_drbg d;
    orc_drbg_init(&d, seed);
    orc_drbg_bytes(&d, out, (size_t)n);
}
