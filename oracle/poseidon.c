/* poseidon.c — Poseidon P128Pow5T3 over Fp (Pallas base field).
 *
 * ORACLE TEST INFRASTRUCTURE (see fd.h header for the usage contract).
 * Restates the PUBLIC Poseidon reference parameter generation (Grain LFSR,
 * generate_parameters_grain.sage of the Poseidon paper) and permutation in
 * the configuration the reference uses through
 * halo2_gadgets::poseidon::primitives::P128Pow5T3
 * (taiga_halo2/src/utils.rs:40-48, prf_nf utils.rs:37, resource commitment
 * resource.rs / resource_commitment.rs): t=3, rate 2, alpha=5, R_F=8,
 * R_P=56, ConstantLength<L> sponge (capacity tail = L<<64, zero padding,
 * squeeze = state[0]).
 *
 * The halo2_gadgets crate is un-vendored, so the derivation
 * is pinned by DOUBLE IMPLEMENTATION: this file (C, independent Grain) vs
 * tools/gen_poseidon.py (Python) — cross-checked in
 * tests/test_poseidon.py against the committed fixture
 * tests/golden/poseidon_p128t3.bin; parity status "assumed, restated"
 * per DESIGN.md §6.
 */
#include <stdint.h>
#include <string.h>

#include "fd.h"

extern const fd_ctx FD_P;

#define POS_T 3
#define POS_RF 8
#define POS_RP 56
#define POS_NB 255
#define POS_ROUNDS (POS_RF + POS_RP)

/* ---- Grain LFSR (80-bit, self-shrinking output) ---- */
typedef struct {
    uint8_t s[80];
} grain_t;

static int grain_update(grain_t* g) {
    int nb = g->s[62] ^ g->s[51] ^ g->s[38] ^ g->s[23] ^ g->s[13] ^ g->s[0];
    memmove(g->s, g->s + 1, 79);
    g->s[79] = (uint8_t)nb;
    return nb;
}

static void grain_init(grain_t* g) {
    /* init sequence, MSB-first fields: field type (2b)=1, sbox (4b)=0,
     * n (12b)=255, t (12b), R_F (10b), R_P (10b), then 30 ones */
    int pos = 0;
    memset(g->s, 0, sizeof(g->s));
#define PUTBITS(val, width)                                  \
    do {                                                     \
        for (int i = (width) - 1; i >= 0; i--)               \
            g->s[pos++] = (uint8_t)(((val) >> i) & 1);       \
    } while (0)
    PUTBITS(1, 2);
    PUTBITS(0, 4);
    PUTBITS(POS_NB, 12);
    PUTBITS(POS_T, 12);
    PUTBITS(POS_RF, 10);
    PUTBITS(POS_RP, 10);
#undef PUTBITS
    for (; pos < 80; pos++) g->s[pos] = 1;
    for (int i = 0; i < 160; i++) grain_update(g);
}

static int grain_bit(grain_t* g) {
    for (;;) {
        int b1 = grain_update(g);
        int b2 = grain_update(g);
        if (b1) return b2;
    }
}

/* 255 bits MSB-first into limbs (standard form, may exceed p) + top bit
 * overflow tracking is unnecessary: 255 bits < 2^255 fits 4 limbs */
static void grain_bits255(grain_t* g, uint64_t v[4]) {
    v[0] = v[1] = v[2] = v[3] = 0;
    for (int i = 0; i < POS_NB; i++) {
        /* shift left 1 */
        v[3] = (v[3] << 1) | (v[2] >> 63);
        v[2] = (v[2] << 1) | (v[1] >> 63);
        v[1] = (v[1] << 1) | (v[0] >> 63);
        v[0] = (v[0] << 1) | (uint64_t)grain_bit(g);
    }
}

static int fd_lt_mod(const uint64_t v[4], const fd_ctx* f) {
    for (int i = 3; i >= 0; i--) {
        if (v[i] < f->mod[i]) return 1;
        if (v[i] > f->mod[i]) return 0;
    }
    return 0; /* equal */
}

static void grain_field_rej(grain_t* g, uint64_t v[4]) {
    do {
        grain_bits255(g, v);
    } while (!fd_lt_mod(v, &FD_P));
}

static void grain_field_norej(grain_t* g, uint64_t v[4]) {
    grain_bits255(g, v);
    fd_reduce_once(v, &FD_P); /* value < 2^255 < 2p, one subtraction */
}

/* ---- generated parameters (Montgomery form), lazily initialized ---- */
static fd_limbs g_rc[POS_ROUNDS][POS_T];
static fd_limbs g_mds[POS_T][POS_T];
static int g_ready = 0;

static void pos_ensure(void) {
    if (g_ready) return;
    grain_t g;
    grain_init(&g);
    uint64_t v[4];
    for (int r = 0; r < POS_ROUNDS; r++)
        for (int i = 0; i < POS_T; i++) {
            grain_field_rej(&g, v);
            fd_to_mont(g_rc[r][i], v, &FD_P);
        }
    fd_limbs xs[POS_T], ys[POS_T];
    for (int i = 0; i < POS_T; i++) {
        grain_field_norej(&g, v);
        fd_to_mont(xs[i], v, &FD_P);
    }
    for (int i = 0; i < POS_T; i++) {
        grain_field_norej(&g, v);
        fd_to_mont(ys[i], v, &FD_P);
    }
    for (int i = 0; i < POS_T; i++)
        for (int j = 0; j < POS_T; j++) {
            fd_limbs s;
            fd_add(s, xs[i], ys[j], &FD_P);
            fd_inv(g_mds[i][j], s, &FD_P);
        }
    g_ready = 1;
}

/* export constants as canonical 32B LE reprs: 64x3 RC then 3x3 MDS
 * (row-major) — the tests/golden/poseidon_p128t3.bin layout */
int orc_poseidon_consts(uint8_t* out, size_t cap) {
    size_t need = (size_t)(POS_ROUNDS * POS_T + POS_T * POS_T) * 32;
    if (cap < need) return -1;
    pos_ensure();
    uint64_t std[4];
    uint8_t* p = out;
    for (int r = 0; r < POS_ROUNDS; r++)
        for (int i = 0; i < POS_T; i++) {
            fd_from_mont(std, g_rc[r][i], &FD_P);
            memcpy(p, std, 32);
            p += 32;
        }
    for (int i = 0; i < POS_T; i++)
        for (int j = 0; j < POS_T; j++) {
            fd_from_mont(std, g_mds[i][j], &FD_P);
            memcpy(p, std, 32);
            p += 32;
        }
    return (int)need;
}

static void sbox5(uint64_t x[4]) {
    fd_limbs x2, x4;
    fd_mul(x2, x, x, &FD_P);
    fd_mul(x4, x2, x2, &FD_P);
    fd_mul(x, x4, x, &FD_P);
}

static void pos_mix(fd_limbs st[POS_T]) {
    fd_limbs nw[POS_T], t;
    for (int i = 0; i < POS_T; i++) {
        fd_zero(nw[i]);
        for (int j = 0; j < POS_T; j++) {
            fd_mul(t, g_mds[i][j], st[j], &FD_P);
            fd_add(nw[i], nw[i], t, &FD_P);
        }
    }
    for (int i = 0; i < POS_T; i++) fd_copy(st[i], nw[i]);
}

/* permutation on a Montgomery-form state */
static void pos_permute_mont(fd_limbs st[POS_T]) {
    int r = 0;
    for (int h = 0; h < POS_RF / 2; h++, r++) {
        for (int i = 0; i < POS_T; i++) {
            fd_add(st[i], st[i], g_rc[r][i], &FD_P);
            sbox5(st[i]);
        }
        pos_mix(st);
    }
    for (int h = 0; h < POS_RP; h++, r++) {
        for (int i = 0; i < POS_T; i++) fd_add(st[i], st[i], g_rc[r][i], &FD_P);
        sbox5(st[0]);
        pos_mix(st);
    }
    for (int h = 0; h < POS_RF / 2; h++, r++) {
        for (int i = 0; i < POS_T; i++) {
            fd_add(st[i], st[i], g_rc[r][i], &FD_P);
            sbox5(st[i]);
        }
        pos_mix(st);
    }
}

/* permutation on canonical 3x32B state bytes (test entry) */
int orc_poseidon_permute(uint8_t state[96]) {
    pos_ensure();
    fd_limbs st[POS_T], v;
    for (int i = 0; i < POS_T; i++) {
        memcpy(v, state + 32 * i, 32);
        if (!fd_lt_mod(v, &FD_P)) return -1;
        fd_to_mont(st[i], v, &FD_P);
    }
    pos_permute_mont(st);
    for (int i = 0; i < POS_T; i++) {
        fd_from_mont(v, st[i], &FD_P);
        memcpy(state + 32 * i, v, 32);
    }
    return 0;
}

/* ConstantLength<L> hash: msg = L x 32B canonical, out = 32B canonical.
 * Initial capacity element = L << 64; zero padding to a rate multiple;
 * output = state[0] after the last permutation. */
int orc_poseidon_hash(const uint8_t* msg, int L, uint8_t out[32]) {
    if (L < 1 || L > 64) return -1;
    pos_ensure();
    fd_limbs st[POS_T], v;
    fd_zero(st[0]);
    fd_zero(st[1]);
    uint64_t capv[4] = {0, (uint64_t)L, 0, 0}; /* L << 64 */
    fd_to_mont(st[POS_T - 1], capv, &FD_P);
    int padded = L + (L & 1);
    for (int c = 0; c < padded; c += 2) {
        for (int i = 0; i < 2; i++) {
            if (c + i < L) {
                memcpy(v, msg + 32 * (c + i), 32);
                if (!fd_lt_mod(v, &FD_P)) return -2;
            } else {
                fd_zero(v);
            }
            fd_limbs m;
            fd_to_mont(m, v, &FD_P);
            fd_add(st[i], st[i], m, &FD_P);
        }
        pos_permute_mont(st);
    }
    fd_from_mont(v, st[0], &FD_P);
    memcpy(out, v, 32);
    return 0;
}
