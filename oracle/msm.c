/* msm.c — variable-base multi-scalar multiplication over Vesta/Pallas
 * (Pippenger buckets; CPU oracle; OpenMP over windows).
 *
 * ORACLE TEST INFRASTRUCTURE (see fd.h header note). Restates the
 * best_multiexp algorithm of halo2_proofs (un-vendored dep — SURVEY.md §8c;
 * any correct multiexp is equivalent: the output is the group element
 * sum_i s_i * P_i, checked against the reference SRS through the Lagrange
 * projection identity in tests/test_srs_pin.py).
 *
 * This is also the timed `cpu_baseline` (kind "port") leg of bench.py for
 * the MSM workload: OpenMP-parallel over windows, cores reported there.
 */
#include "curve.h"
#include <stdlib.h>

#ifdef _OPENMP
#include <omp.h>
#endif

/* window width: 255 bits / 13 -> 20 windows */
#define MSM_C 13
#define MSM_NWIN ((255 + MSM_C - 1) / MSM_C)

/* scalars: n x 4x64 LE standard-form (already < p); points: n affine Mont.
 * result jacobian. */
void orc_msm_core(pt_jac* out, const uint64_t (*scalars)[4], const pt_aff* pts,
                  long n, const fd_ctx* f) {
    pt_jac win_sum[MSM_NWIN];
#ifdef _OPENMP
#pragma omp parallel for schedule(dynamic, 1)
#endif
    for (int w = 0; w < MSM_NWIN; w++) {
        long nbuckets = 1L << MSM_C;
        pt_jac* buckets = (pt_jac*)malloc(sizeof(pt_jac) * (size_t)nbuckets);
        for (long i = 0; i < nbuckets; i++) pt_jac_identity(&buckets[i]);
        int bit0 = w * MSM_C;
        for (long i = 0; i < n; i++) {
            /* extract MSM_C bits starting at bit0 from 256-bit scalar */
            int limb = bit0 >> 6, sh = bit0 & 63;
            uint64_t d = scalars[i][limb] >> sh;
            if (sh && limb < 3) d |= scalars[i][limb + 1] << (64 - sh);
            d &= (1UL << MSM_C) - 1;
            if (bit0 + MSM_C > 255) d &= (1UL << (255 - bit0)) - 1;
            if (d) pt_add_aff(&buckets[d], &buckets[d], &pts[i], f);
        }
        /* bucket reduction: sum_d d * bucket[d] via running suffix sums */
        pt_jac run, tot;
        pt_jac_identity(&run);
        pt_jac_identity(&tot);
        for (long d = nbuckets - 1; d >= 1; d--) {
            pt_add(&run, &run, &buckets[d], f);
            pt_add(&tot, &tot, &run, f);
        }
        win_sum[w] = tot;
        free(buckets);
    }
    pt_jac acc;
    pt_jac_identity(&acc);
    for (int w = MSM_NWIN - 1; w >= 0; w--) {
        for (int b = 0; b < MSM_C && w != MSM_NWIN - 1; b++) pt_dbl(&acc, &acc, f);
        pt_add(&acc, &acc, &win_sum[w], f);
    }
    *out = acc;
}

/* byte interface.
 * fid: field of the CURVE's base coordinates (1 = Fq => Vesta, scalars Fp).
 * scalars: n*32 B canonical LE (standard form, in the curve's SCALAR field).
 * points:  n*64 B affine x||y canonical LE in base field; identity given as
 *          all-zero x and y.
 * out: 64 B affine x||y (all zeros for identity).
 * Returns 0 ok, -1 bad input. */
int orc_msm(int fid, long n, const uint8_t* scalars, const uint8_t* points, uint8_t* out) {
    const fd_ctx* fb = fid ? &FD_Q : &FD_P;
    uint64_t(*sc)[4] = (uint64_t(*)[4])malloc(sizeof(fd_limbs) * (size_t)n);
    pt_aff* pa = (pt_aff*)malloc(sizeof(pt_aff) * (size_t)n);
    int bad = 0;
#ifdef _OPENMP
#pragma omp parallel for schedule(static) reduction(| : bad)
#endif
    for (long i = 0; i < n; i++) {
        memcpy(sc[i], scalars + 32 * i, 32);
        const uint8_t* px = points + 64 * i;
        int allz = 1;
        for (int b = 0; b < 64; b++) if (px[b]) { allz = 0; break; }
        if (allz) {
            fd_zero(pa[i].x); fd_zero(pa[i].y); pa[i].inf = 1;
        } else {
            pa[i].inf = 0;
            bad |= fd_from_bytes(pa[i].x, px, fb) ? 1 : 0;
            bad |= fd_from_bytes(pa[i].y, px + 32, fb) ? 1 : 0;
        }
    }
    if (bad) { free(sc); free(pa); return -1; }
    pt_jac r;
    orc_msm_core(&r, (const uint64_t(*)[4])sc, pa, n, fb);
    pt_aff ra;
    pt_to_aff(&ra, &r, fb);
    if (ra.inf) {
        memset(out, 0, 64);
    } else {
        fd_to_bytes(out, ra.x, fb);
        fd_to_bytes(out + 32, ra.y, fb);
    }
    free(sc);
    free(pa);
    return 0;
}

/* decompress a batch of 32-B compressed points to 64-B affine repr
 * (identity -> all zeros). Returns 0 ok, -1 invalid encoding. */
int orc_decompress(int fid, long n, const uint8_t* in, uint8_t* out) {
    const fd_ctx* fb = fid ? &FD_Q : &FD_P;
    int bad = 0;
#ifdef _OPENMP
#pragma omp parallel for schedule(static) reduction(| : bad)
#endif
    for (long i = 0; i < n; i++) {
        pt_aff p;
        if (pt_decompress(&p, in + 32 * i, fb)) { bad |= 1; continue; }
        if (p.inf) {
            memset(out + 64 * i, 0, 64);
        } else {
            fd_to_bytes(out + 64 * i, p.x, fb);
            fd_to_bytes(out + 64 * i + 32, p.y, fb);
        }
    }
    return bad ? -1 : 0;
}

/* compress a batch of 64-B affine points to 32-B (inverse of the above) */
int orc_compress(int fid, long n, const uint8_t* in, uint8_t* out) {
    const fd_ctx* fb = fid ? &FD_Q : &FD_P;
    int bad = 0;
#ifdef _OPENMP
#pragma omp parallel for schedule(static) reduction(| : bad)
#endif
    for (long i = 0; i < n; i++) {
        const uint8_t* px = in + 64 * i;
        int allz = 1;
        for (int b = 0; b < 64; b++) if (px[b]) { allz = 0; break; }
        pt_aff p;
        if (allz) {
            p.inf = 1; fd_zero(p.x); fd_zero(p.y);
        } else {
            p.inf = 0;
            bad |= fd_from_bytes(p.x, px, fb) ? 1 : 0;
            bad |= fd_from_bytes(p.y, px + 32, fb) ? 1 : 0;
        }
        pt_compress(out + 32 * i, &p, fb);
    }
    return bad ? -1 : 0;
}
