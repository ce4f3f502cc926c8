/* ntt.c — radix-2 NTT over a Pasta field (CPU oracle; OpenMP).
 *
 * ORACLE TEST INFRASTRUCTURE (see fd.h header note).
 *
 * Restates halo2_proofs' best_fft / EvaluationDomain semantics (the
 * un-vendored heliaxdev/halo2 `taiga` dep — SURVEY.md §8c; base: zcash
 * halo2_proofs 0.3 public algorithm):
 *   fwd:  out[i] = sum_j a[j] * omega^{ij}     (bit-reverse + butterflies)
 *   inv:  out[j] = n^{-1} sum_i a[i] omega^{-ij}
 * with omega = root^(2^(32-k)), root = 5^((m-1)/2^32) — the generator-5
 * convention is pinned against the reference SRS by tests/test_srs_pin.py.
 */
#include "fd.h"
#include <stdlib.h>

#ifdef _OPENMP
#include <omp.h>
#endif

/* omega for domain size 2^k (Mont form) */
void orc_domain_omega(uint64_t out[4], int k, int inverse, const fd_ctx* f) {
    fd_limbs w;
    fd_to_mont(w, inverse ? f->root_inv : f->root, f);
    for (int i = k; i < 32; i++) fd_sqr(w, w, f);
    fd_copy(out, w);
}

/* in-place NTT on Mont-form limbs; a has n = 2^k elements */
void orc_ntt_inplace(uint64_t (*a)[4], int k, int inverse, const fd_ctx* f) {
    long n = 1L << k;
    /* bit-reverse permutation */
    for (long i = 0; i < n; i++) {
        long j = 0;
        for (int b = 0; b < k; b++) j |= ((i >> b) & 1L) << (k - 1 - b);
        if (j > i) {
            fd_limbs t;
            fd_copy(t, a[i]); fd_copy(a[i], a[j]); fd_copy(a[j], t);
        }
    }
    fd_limbs omega;
    orc_domain_omega(omega, k, inverse, f);
    /* twiddle table for the largest stage: w^t for t in [0, n/2) */
    fd_limbs* tw = (fd_limbs*)malloc(sizeof(fd_limbs) * (size_t)(n / 2 > 0 ? n / 2 : 1));
    fd_one_mont(tw[0], f);
    for (long t = 1; t < n / 2; t++) fd_mul(tw[t], tw[t - 1], omega, f);

    for (int s = 1; s <= k; s++) {
        long size = 1L << s;
        long half = size >> 1;
        long tstep = n >> s; /* twiddle stride into tw */
#ifdef _OPENMP
#pragma omp parallel for schedule(static) if (n >= 4096)
#endif
        for (long j = 0; j < n / 2; j++) {
            long start = (j / half) * size;
            long kk = j % half;
            fd_limbs t;
            fd_mul(t, a[start + kk + half], tw[kk * tstep], f);
            fd_sub(a[start + kk + half], a[start + kk], t, f);
            fd_add(a[start + kk], a[start + kk], t, f);
        }
    }
    free(tw);
    if (inverse) {
        /* scale by n^{-1} */
        fd_limbs ninv;
        uint64_t nstd[4] = {(uint64_t)n, 0, 0, 0};
        fd_to_mont(ninv, nstd, f);
        fd_inv(ninv, ninv, f);
#ifdef _OPENMP
#pragma omp parallel for schedule(static) if (n >= 4096)
#endif
        for (long i = 0; i < n; i++) fd_mul(a[i], a[i], ninv, f);
    }
}

/* byte interface: values are 32-byte LE canonical reprs, in-place.
 * fid: 0 = Fp, 1 = Fq.  dir: 0 = forward, 1 = inverse (includes 1/n).
 * returns 0 ok, -1 on non-canonical input. */
int orc_ntt(int fid, int dir, int k, uint8_t* data) {
    const fd_ctx* f = fid ? &FD_Q : &FD_P;
    long n = 1L << k;
    uint64_t(*a)[4] = (uint64_t(*)[4])malloc(sizeof(fd_limbs) * (size_t)n);
    int bad = 0;
#ifdef _OPENMP
#pragma omp parallel for schedule(static) reduction(| : bad)
#endif
    for (long i = 0; i < n; i++)
        bad |= fd_from_bytes(a[i], data + 32 * i, f) ? 1 : 0;
    if (bad) { free(a); return -1; }
    orc_ntt_inplace(a, k, dir, f);
#ifdef _OPENMP
#pragma omp parallel for schedule(static)
#endif
    for (long i = 0; i < n; i++) fd_to_bytes(data + 32 * i, a[i], f);
    free(a);
    return 0;
}
