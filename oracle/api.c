/* api.c — ctypes-facing helpers of the CPU oracle (liboracle.so).
 *
 * ORACLE TEST INFRASTRUCTURE (see fd.h header note): consumed only by
 * tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg.
 */
#include "curve.h"
#include <stdlib.h>
#include <string.h>

/* --- field element ops on canonical 32-byte LE reprs; fid 0=Fp 1=Fq ---
 * op: 0 add, 1 sub, 2 mul, 3 inv(a), 4 neg(a), 5 sqrt(a) (b unused for 3..5)
 * returns 0 ok; -1 non-canonical input; -2 sqrt of non-residue */
int orc_fd_op(int fid, int op, const uint8_t* a, const uint8_t* b, uint8_t* out) {
    const fd_ctx* f = fid ? &FD_Q : &FD_P;
    fd_limbs x, y, r;
    if (fd_from_bytes(x, a, f)) return -1;
    if (op <= 2 && fd_from_bytes(y, b, f)) return -1;
    switch (op) {
        case 0: fd_add(r, x, y, f); break;
        case 1: fd_sub(r, x, y, f); break;
        case 2: fd_mul(r, x, y, f); break;
        case 3: fd_inv(r, x, f); break;
        case 4: fd_neg(r, x, f); break;
        case 5:
            if (!fd_sqrt(r, x, f)) return -2;
            break;
        default: return -1;
    }
    fd_to_bytes(out, r, f);
    return 0;
}

/* a^e where e is a 32-byte LE integer (not reduced) */
int orc_fd_pow(int fid, const uint8_t* a, const uint8_t* e, uint8_t* out) {
    const fd_ctx* f = fid ? &FD_Q : &FD_P;
    fd_limbs x, r;
    uint64_t ee[4];
    if (fd_from_bytes(x, a, f)) return -1;
    memcpy(ee, e, 32);
    fd_pow(r, x, ee, f);
    fd_to_bytes(out, r, f);
    return 0;
}

/* domain omega for 2^k as canonical bytes */
void orc_get_omega(int fid, int k, int inverse, uint8_t* out) {
    const fd_ctx* f = fid ? &FD_Q : &FD_P;
    extern void orc_domain_omega(uint64_t out[4], int k, int inverse, const fd_ctx* f);
    fd_limbs w;
    orc_domain_omega(w, k, inverse, f);
    fd_to_bytes(out, w, f);
}

/* --- point ops on 64-byte affine reprs (identity = all zeros) --- */
static int aff_from_bytes(pt_aff* p, const uint8_t* in, const fd_ctx* f) {
    int allz = 1;
    for (int i = 0; i < 64; i++) if (in[i]) { allz = 0; break; }
    if (allz) { p->inf = 1; fd_zero(p->x); fd_zero(p->y); return 0; }
    p->inf = 0;
    if (fd_from_bytes(p->x, in, f)) return -1;
    if (fd_from_bytes(p->y, in + 32, f)) return -1;
    return 0;
}

static void aff_to_bytes(uint8_t* out, const pt_aff* p, const fd_ctx* f) {
    if (p->inf) { memset(out, 0, 64); return; }
    fd_to_bytes(out, p->x, f);
    fd_to_bytes(out + 32, p->y, f);
}

/* op: 0 add, 1 dbl(a), 2 neg(a), 3 mul(a, scalar=b as 32B int) */
int orc_pt_op(int fid, int op, const uint8_t* a, const uint8_t* b, uint8_t* out) {
    const fd_ctx* f = fid ? &FD_Q : &FD_P;
    pt_aff pa, pb, ra;
    pt_jac ja, jb, jr;
    if (aff_from_bytes(&pa, a, f)) return -1;
    pt_from_aff(&ja, &pa, f);
    switch (op) {
        case 0:
            if (aff_from_bytes(&pb, b, f)) return -1;
            pt_from_aff(&jb, &pb, f);
            pt_add(&jr, &ja, &jb, f);
            break;
        case 1: pt_dbl(&jr, &ja, f); break;
        case 2: pt_neg(&jr, &ja, f); break;
        case 3: {
            uint64_t k[4];
            memcpy(k, b, 32);
            pt_mul(&jr, &ja, k, f);
            break;
        }
        default: return -1;
    }
    pt_to_aff(&ra, &jr, f);
    aff_to_bytes(out, &ra, f);
    return 0;
}

/* is the 64-byte affine point on y^2 = x^3 + 5? */
int orc_pt_on_curve(int fid, const uint8_t* a) {
    const fd_ctx* f = fid ? &FD_Q : &FD_P;
    pt_aff p;
    if (aff_from_bytes(&p, a, f)) return 0;
    if (p.inf) return 1;
    fd_limbs lhs, rhs, b5;
    fd_sqr(lhs, p.y, f);
    fd_sqr(rhs, p.x, f);
    fd_mul(rhs, rhs, p.x, f);
    uint64_t five[4] = {5, 0, 0, 0};
    fd_to_mont(b5, five, f);
    fd_add(rhs, rhs, b5, f);
    return fd_eq(lhs, rhs);
}

/* generate the bench's synthetic distinct bases: out[i] = [k_i]G on
 * Vesta with k_i a splitmix64-derived 256-bit scalar, 64-byte canonical
 * affine — byte-identical to the product's tg_gen_bases (msm.hip
 * k_gen_bases; see the note there on why the multiples must be RANDOM,
 * not small/sequential). Used by bench.py's cpu_baseline leg and tests. */
static uint64_t orc_sm64(uint64_t x) {
    x += 0x9E3779B97F4A7C15ULL;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ULL;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBULL;
    return x ^ (x >> 31);
}
void orc_gen_bases(long n, uint64_t seed, uint8_t* out) {
    const fd_ctx* f = &FD_Q;
#ifdef _OPENMP
#pragma omp parallel for schedule(static)
#endif
    for (long i = 0; i < n; i++) {
        pt_aff G;
        fd_limbs one;
        fd_one_mont(one, f);
        fd_neg(G.x, one, f);
        fd_add(G.y, one, one, f);
        G.inf = 0;
        uint64_t k[4];
        for (int l = 0; l < 4; l++)
            k[l] = orc_sm64(seed * 0xD1B54A32D192ED03ULL + (uint64_t)i * 4 +
                            (uint64_t)l);
        pt_jac j, g;
        pt_from_aff(&g, &G, f);
        pt_mul(&j, &g, k, f);
        pt_aff a;
        pt_to_aff(&a, &j, f);
        fd_to_bytes(out + 64 * i, a.x, f);
        fd_to_bytes(out + 64 * i + 32, a.y, f);
    }
}

/* --- SRS pin: random-projection check of g_lagrange == group-iNTT(g) ---
 * params: the raw params_15 bytes. Uses `rounds` random projections drawn
 * from a seeded xorshift; full-strength pin of decompression, field, curve,
 * NTT and MSM against the reference's own SRS data in one identity:
 *     MSM(r, g_lagrange) == MSM(iNTT(r), g)
 * Returns 0 on success, <0 on failure. */
extern int orc_ntt(int fid, int dir, int k, uint8_t* data);
extern void orc_ntt_inplace(uint64_t (*a)[4], int k, int inverse, const fd_ctx* f);
extern void orc_msm_core(pt_jac* out, const uint64_t (*scalars)[4], const pt_aff* pts,
                         long n, const fd_ctx* f);
extern int orc_decompress(int fid, long n, const uint8_t* in, uint8_t* out);

static uint64_t xs_next(uint64_t* s) {
    uint64_t x = *s;
    x ^= x << 13; x ^= x >> 7; x ^= x << 17;
    *s = x;
    return x;
}

int orc_srs_project_check(const uint8_t* params, long len, int rounds) {
    if (len < 4) return -10;
    uint32_t k;
    memcpy(&k, params, 4);
    long n = 1L << k;
    if (len != 4 + 2 * n * 32 + 64) return -11;
    const fd_ctx* fb = &FD_Q; /* Vesta base field */
    const fd_ctx* fs = &FD_P; /* Vesta scalar field */
    pt_aff* g = (pt_aff*)malloc(sizeof(pt_aff) * (size_t)n);
    pt_aff* gl = (pt_aff*)malloc(sizeof(pt_aff) * (size_t)n);
    int bad = 0;
#ifdef _OPENMP
#pragma omp parallel for schedule(static) reduction(| : bad)
#endif
    for (long i = 0; i < n; i++) {
        bad |= pt_decompress(&g[i], params + 4 + 32 * i, fb) ? 1 : 0;
        bad |= pt_decompress(&gl[i], params + 4 + 32 * (n + i), fb) ? 1 : 0;
    }
    if (bad) { free(g); free(gl); return -12; }

    uint64_t seed = 0x5441494741ULL; /* "TAIGA" */
    int rc = 0;
    for (int round = 0; round < rounds && rc == 0; round++) {
        uint64_t(*r)[4] = (uint64_t(*)[4])malloc(sizeof(fd_limbs) * (size_t)n);
        uint64_t(*rm)[4] = (uint64_t(*)[4])malloc(sizeof(fd_limbs) * (size_t)n);
        for (long i = 0; i < n; i++) {
            /* ~253-bit uniform-ish scalars: top limb masked below p's top */
            for (int l = 0; l < 4; l++) r[i][l] = xs_next(&seed);
            r[i][3] &= 0x3FFFFFFFFFFFFFFFULL;
            if (r[i][3] >= 0x2000000000000000ULL) r[i][3] -= 0x2000000000000000ULL;
            fd_to_mont(rm[i], r[i], fs);
        }
        /* lhs = MSM(r, g_lagrange) */
        pt_jac lhs, rhs;
        orc_msm_core(&lhs, (const uint64_t(*)[4])r, gl, n, fb);
        /* s = iNTT(r) in the scalar field */
        orc_ntt_inplace(rm, (int)k, 1, fs);
        for (long i = 0; i < n; i++) fd_from_mont(r[i], rm[i], fs);
        orc_msm_core(&rhs, (const uint64_t(*)[4])r, g, n, fb);
        pt_aff la, ra;
        pt_to_aff(&la, &lhs, fb);
        pt_to_aff(&ra, &rhs, fb);
        if (la.inf != ra.inf || !fd_eq(la.x, ra.x) || !fd_eq(la.y, ra.y)) rc = -(20 + round);
        free(r);
        free(rm);
    }
    free(g);
    free(gl);
    return rc;
}
