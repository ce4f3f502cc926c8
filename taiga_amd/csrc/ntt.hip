// ntt.hip — radix-2 NTT over Fp for gfx950. PRODUCT CODE.
//
// MI355X-native replacement for halo2_proofs' best_fft on the create_proof
// hot path (SURVEY.md §8a "quotient h(X)" row; the dep is un-vendored —
// §8c). Semantics (bit-reverse + DIT butterflies, omega from the
// generator-5 convention) are pinned against the CPU oracle, which is
// pinned against the reference SRS (tests/test_srs_pin.py).
//
// HBM-bound integer work (32 B/element): all kernels are grid-stride with
// coalesced per-lane 32-byte accesses. v1 = one pass per stage; the fused
// LDS multi-stage path cuts passes to ceil(k/FUSE) (bytes model in
// BASELINE.md config 3).

#include "pasta_device.hpp"

namespace taiga {

// tw[i] = omega^i (Mont), i < half_n. log-cost pow per thread.
template <class C>
__global__ void __launch_bounds__(256) k_twiddles(Fd<C>* tw, u64 half_n, int k, bool inverse) {
  Fd<C> root;
#pragma unroll
  for (int i = 0; i < 4; i++) root.l[i] = inverse ? C::ROOT_INV[i] : C::ROOT[i];
  Fd<C> w = fd_to_mont(root);
  for (int i = k; i < 32; i++) w = fd_sqr(w);  // omega for 2^k
  for (u64 i = blockIdx.x * (u64)blockDim.x + threadIdx.x; i < half_n;
       i += (u64)gridDim.x * blockDim.x) {
    tw[i] = fd_pow_u64(w, i);
  }
}

__device__ __forceinline__ u64 bitrev(u64 x, int k) {
  u64 r = 0;
  for (int b = 0; b < k; b++) r |= ((x >> b) & 1ULL) << (k - 1 - b);
  return r;
}

// out[i] = to_mont(in[bitrev(i)]); sets *err if any input repr >= MOD.
template <class C>
__global__ void __launch_bounds__(256) k_bitrev_load(Fd<C>* out, const Fd<C>* in, int k, int to_mont_flag,
                              unsigned* err) {
  u64 n = 1ULL << k;
  for (u64 i = blockIdx.x * (u64)blockDim.x + threadIdx.x; i < n;
       i += (u64)gridDim.x * blockDim.x) {
    Fd<C> v = in[bitrev(i, k)];
    if (err) {
      // canonical check: v < MOD
      bool ge = true;  // ge stays true only if v == MOD prefix-equal path
      bool lt = false;
#pragma unroll
      for (int limb = 3; limb >= 0; limb--) {
        if (!lt && v.l[limb] > C::MOD[limb]) { atomicOr(err, 1u); break; }
        if (v.l[limb] < C::MOD[limb]) { lt = true; }
      }
      if (!lt) atomicOr(err, 1u);
      (void)ge;
    }
    out[i] = to_mont_flag ? fd_to_mont(v) : v;
  }
}

// one radix-2 DIT stage s (1-based): butterflies on pairs span 2^(s-1)
template <class C>
__global__ void __launch_bounds__(256) k_ntt_stage(Fd<C>* a, const Fd<C>* tw, int k, int s) {
  u64 nb = 1ULL << (k - 1);  // number of butterflies
  u64 half = 1ULL << (s - 1);
  int tshift = k - s;  // twiddle stride = 2^(k-s)
  for (u64 j = blockIdx.x * (u64)blockDim.x + threadIdx.x; j < nb;
       j += (u64)gridDim.x * blockDim.x) {
    u64 grp = j >> (s - 1);
    u64 kk = j & (half - 1);
    u64 i0 = (grp << s) | kk;
    u64 i1 = i0 + half;
    Fd<C> t = fd_mul(a[i1], tw[kk << tshift]);
    Fd<C> lo = a[i0];
    a[i1] = fd_sub(lo, t);
    a[i0] = fd_add(lo, t);
  }
}

// fused LDS stages: processes FUSE stages [s0+1 .. s0+FUSE] in one pass.
// Each block handles TILE = 2^FUSE_LOG consecutive butterfly-groups worth of
// elements gathered with stride 2^s0 elements.
// Tile size chosen so 256 threads * 2 elements/thread = 512 elements in LDS.
template <class C, int FUSE>
__global__ void __launch_bounds__(256) k_ntt_fused(Fd<C>* a, const Fd<C>* tw, int k, int s0) {
  // elements per tile
  constexpr int TILE = 1 << FUSE;           // e.g. 32.. but we use 512 = 2^9
  __shared__ Fd<C> lds[1 << FUSE];
  u64 n = 1ULL << k;
  u64 span = 1ULL << s0;                    // input butterfly span entering this pass
  // tile t handles elements idx = base + m*span for m in [0, TILE), where
  // tiles partition each contiguous "segment" of length span*TILE.
  u64 tiles_per_seg = span;                 // one tile per residue class r < span
  u64 seg_len = span << FUSE;
  u64 nseg = n / seg_len;
  u64 ntiles = nseg * tiles_per_seg;
  for (u64 tile = blockIdx.x; tile < ntiles; tile += gridDim.x) {
    u64 seg = tile / tiles_per_seg;
    u64 r = tile % tiles_per_seg;
    u64 base = seg * seg_len + r;
    // load TILE strided elements
    for (int m = threadIdx.x; m < TILE; m += blockDim.x) lds[m] = a[base + (u64)m * span];
    __syncthreads();
    // FUSE local stages; local stage ls corresponds to global stage s0+ls
    for (int ls = 1; ls <= FUSE; ls++) {
      int s = s0 + ls;
      u64 half = 1ULL << (ls - 1);
      int tshift = k - s;
      for (int j = threadIdx.x; j < (TILE >> 1); j += blockDim.x) {
        u64 grp = (u64)j >> (ls - 1);
        u64 kk = (u64)j & (half - 1);
        u64 i0 = (grp << ls) | kk;
        u64 i1 = i0 + half;
        // global butterfly index kk_global = kk*span + r, twiddle = kk_g << tshift
        u64 kkg = (kk << s0) | r;
        Fd<C> t = fd_mul(lds[i1], tw[kkg << tshift]);
        Fd<C> lo = lds[i0];
        lds[i1] = fd_sub(lo, t);
        lds[i0] = fd_add(lo, t);
      }
      __syncthreads();
    }
    for (int m = threadIdx.x; m < TILE; m += blockDim.x) a[base + (u64)m * span] = lds[m];
    __syncthreads();
  }
}

// first-pass variant with the bit-reversal fused into the load: reads the
// NATURAL-order input directly (a strided gather — source index =
// bitrev_FUSE(m) << (k-FUSE) | bitrev_{k-FUSE}(tile)) and writes the
// pass-1 result contiguously, replacing the separate k_bitrev_load sweep
// (one full n-read + n-write saved per direction).
template <class C, int FUSE>
__global__ void __launch_bounds__(256) k_ntt_fused_br(Fd<C>* out, const Fd<C>* in,
                                                      const Fd<C>* tw, int k) {
  constexpr int TILE = 1 << FUSE;
  __shared__ Fd<C> lds[1 << FUSE];
  u64 n = 1ULL << k;
  u64 ntiles = n >> FUSE;
  int kshift = k - FUSE;
  for (u64 tile = blockIdx.x; tile < ntiles; tile += gridDim.x) {
    u64 base = tile << FUSE;
    u64 src_off = bitrev(tile, kshift);
    for (int m = threadIdx.x; m < TILE; m += blockDim.x)
      lds[m] = in[(bitrev((u64)m, FUSE) << kshift) | src_off];
    __syncthreads();
    for (int ls = 1; ls <= FUSE; ls++) {
      int s = ls;  // s0 = 0, r = 0
      u64 half = 1ULL << (ls - 1);
      int tshift = k - s;
      for (int j = threadIdx.x; j < (TILE >> 1); j += blockDim.x) {
        u64 grp = (u64)j >> (ls - 1);
        u64 kk = (u64)j & (half - 1);
        u64 i0 = (grp << ls) | kk;
        u64 i1 = i0 + half;
        Fd<C> t = fd_mul(lds[i1], tw[kk << tshift]);
        Fd<C> lo = lds[i0];
        lds[i1] = fd_sub(lo, t);
        lds[i0] = fd_add(lo, t);
      }
      __syncthreads();
    }
    for (int m = threadIdx.x; m < TILE; m += blockDim.x) out[base + (u64)m] = lds[m];
    __syncthreads();
  }
}

// 2-D fused LDS stages for strided passes: each block processes RB
// consecutive residues r, so global accesses are RB*32-byte contiguous
// chunks (the 1-residue variant gathers single 32-B elements at stride
// 2^s0 — ~4.5x slower per pass at k=22). F local stages; span = 2^s0 must
// be a multiple of RB. LDS rows padded (+1 element) to spread bank groups.
template <class C, int F, int RB>
__global__ void __launch_bounds__(256) k_ntt_fused2(Fd<C>* a, const Fd<C>* tw, int k,
                                                    int s0) {
  constexpr int TILE = 1 << F;
  __shared__ Fd<C> lds[(TILE + 1) * RB];
  u64 n = 1ULL << k;
  u64 span = 1ULL << s0;
  u64 groups_per_span = span / RB;
  u64 seg_len = span << F;
  u64 nseg = n / seg_len;
  u64 ntiles = nseg * groups_per_span;
  for (u64 tile = blockIdx.x; tile < ntiles; tile += gridDim.x) {
    u64 seg = tile / groups_per_span;
    u64 r0 = (tile % groups_per_span) * RB;
    u64 base = seg * seg_len + r0;
    for (int t = threadIdx.x; t < TILE * RB; t += blockDim.x) {
      int m = t / RB, rr = t % RB;
      lds[rr * (TILE + 1) + m] = a[base + (u64)m * span + rr];
    }
    __syncthreads();
    for (int ls = 1; ls <= F; ls++) {
      int s = s0 + ls;
      u64 half = 1ULL << (ls - 1);
      int tshift = k - s;
      for (int t = threadIdx.x; t < (TILE / 2) * RB; t += blockDim.x) {
        int j = t / RB, rr = t % RB;
        u64 grp = (u64)j >> (ls - 1);
        u64 kk = (u64)j & (half - 1);
        int i0 = (int)((grp << ls) | kk);
        int i1 = i0 + (int)half;
        u64 kkg = (kk << s0) | (r0 + rr);
        Fd<C> tv = fd_mul(lds[rr * (TILE + 1) + i1], tw[kkg << tshift]);
        Fd<C> lo = lds[rr * (TILE + 1) + i0];
        lds[rr * (TILE + 1) + i1] = fd_sub(lo, tv);
        lds[rr * (TILE + 1) + i0] = fd_add(lo, tv);
      }
      __syncthreads();
    }
    for (int t = threadIdx.x; t < TILE * RB; t += blockDim.x) {
      int m = t / RB, rr = t % RB;
      a[base + (u64)m * span + rr] = lds[rr * (TILE + 1) + m];
    }
    __syncthreads();
  }
}

// pointwise scale (by a constant) and/or from-mont conversion
template <class C>
__global__ void __launch_bounds__(256) k_scale(Fd<C>* a, u64 n, Fd<C> c, int do_scale, int from_mont_flag) {
  for (u64 i = blockIdx.x * (u64)blockDim.x + threadIdx.x; i < n;
       i += (u64)gridDim.x * blockDim.x) {
    Fd<C> v = a[i];
    if (do_scale) v = fd_mul(v, c);
    if (from_mont_flag) v = fd_from_mont(v);
    a[i] = v;
  }
}

// pointwise multiply by powers of g: a[i] *= g^i  (coset enter/exit)
// a[i] *= g^i. One pow per thread for its first index, then an
// incremental multiply by g^stride per grid-stride step (precomputed on
// the host) — the per-element fd_pow_u64 this replaces was ~65 muls per
// element and 12x the runtime.
template <class C>
__global__ void __launch_bounds__(256) k_coset_scale(Fd<C>* a, u64 n, Fd<C> g, Fd<C> g_stride) {
  u64 stride = (u64)gridDim.x * blockDim.x;
  u64 i0 = blockIdx.x * (u64)blockDim.x + threadIdx.x;
  if (i0 >= n) return;
  Fd<C> cur = fd_pow_u64(g, i0);
  for (u64 i = i0; i < n; i += stride) {
    a[i] = fd_mul(a[i], cur);
    cur = fd_mul(cur, g_stride);
  }
}

// launch helper: ~8 elements per thread amortize the initial pow
static inline void coset_scale_launch(Fd<FpCfg>* a, u64 n, const Fd<FpCfg>& g,
                                      hipStream_t stream) {
  u64 blocks = (n + 256 * 8 - 1) / (256 * 8);
  if (blocks < 1) blocks = 1;
  if (blocks > 2048) blocks = 2048;
  Fd<FpCfg> gs = fd_pow_u64(g, blocks * 256);
  hipLaunchKernelGGL(k_coset_scale<FpCfg>, dim3((int)blocks), dim3(256), 0, stream,
                     a, n, g, gs);
}

// ---------- host-side launcher (internal; C ABI wraps it) ----------

struct NttPlan {
  Fd<FpCfg>* d_tw_fwd = nullptr;  // omega^i, i < n/2
  Fd<FpCfg>* d_tw_inv = nullptr;
  int k = -1;
};

static inline int ntt_grid(u64 work, int block = 256) {
  u64 blocks = (work + block - 1) / block;
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

// builds (or rebuilds) the twiddle tables for size 2^k
inline hipError_t ntt_plan_init(NttPlan& plan, int k, hipStream_t stream) {
  if (plan.k == k) return hipSuccess;
  if (plan.d_tw_fwd) { hipFree(plan.d_tw_fwd); plan.d_tw_fwd = nullptr; }
  if (plan.d_tw_inv) { hipFree(plan.d_tw_inv); plan.d_tw_inv = nullptr; }
  u64 half = k > 0 ? (1ULL << (k - 1)) : 1;
  hipError_t e;
  if ((e = hipMalloc(&plan.d_tw_fwd, half * sizeof(Fp))) != hipSuccess) return e;
  if ((e = hipMalloc(&plan.d_tw_inv, half * sizeof(Fp))) != hipSuccess) return e;
  hipLaunchKernelGGL(k_twiddles<FpCfg>, dim3(ntt_grid(half)), dim3(256), 0, stream,
                     plan.d_tw_fwd, half, k, false);
  hipLaunchKernelGGL(k_twiddles<FpCfg>, dim3(ntt_grid(half)), dim3(256), 0, stream,
                     plan.d_tw_inv, half, k, true);
  plan.k = k;
  return hipGetLastError();
}

// in-place NTT on device Montgomery-form data (d_a), size 2^k, using d_tmp
// as the working buffer (bitrev lands there; result copied back).
// inverse => scales by n^{-1}. ProfFn: callable int->RAII scope, bracketing
// each launch group for the HIP-event profiler (pass a no-op for none).
// Prof indices: 0 bitrev, 1 fused, 2 stage, 3 scale.
template <class ProfFn>
inline hipError_t ntt_run(Fd<FpCfg>* d_a, Fd<FpCfg>* d_tmp, const NttPlan& plan, int k,
                          bool inverse, hipStream_t stream, const Fd<FpCfg>* ninv_mont,
                          ProfFn&& prof) {
  const Fd<FpCfg>* tw = inverse ? plan.d_tw_inv : plan.d_tw_fwd;
  u64 n = 1ULL << k;
  constexpr int FUSE = 9;  // 512-element LDS tiles (16 KiB), first pass only
  constexpr int RB = 8;    // residues per block in the strided passes (256-B gathers)
  // bitrev fused into the first pass's load when that pass exists
  // (TG_NTT_FUSE_BR=0 restores the separate sweep for A/B)
  static int fuse_br = [] {
    const char* e = getenv("TG_NTT_FUSE_BR");
    return e ? (atoi(e) ? 1 : 0) : 1;
  }();
  int s = 1;
  if (fuse_br && k >= FUSE) {
    [[maybe_unused]] auto sc = prof(1);
    u64 ntiles = n >> FUSE;
    hipLaunchKernelGGL((k_ntt_fused_br<FpCfg, FUSE>),
                       dim3(ntiles > 2048 ? 2048 : (unsigned)ntiles), dim3(256), 0,
                       stream, d_tmp, d_a, tw, k);
    s = 1 + FUSE;
  } else {
    [[maybe_unused]] auto sc = prof(0);
    hipLaunchKernelGGL(k_bitrev_load<FpCfg>, dim3(ntt_grid(n)), dim3(256), 0, stream,
                       d_tmp, d_a, k, 0, nullptr);
  }
  while (s <= k) {
    int remaining = k - s + 1;
    if (s == 1 && remaining >= FUSE) {
      // first pass: span 1, tiles are contiguous -> 1-D variant
      [[maybe_unused]] auto sc = prof(1);
      u64 ntiles = n >> FUSE;
      hipLaunchKernelGGL((k_ntt_fused<FpCfg, FUSE>),
                         dim3(ntiles > 2048 ? 2048 : (unsigned)ntiles), dim3(256), 0,
                         stream, d_tmp, tw, k, s - 1);
      s += FUSE;
      continue;
    }
    int f = remaining < 7 ? remaining : 7;  // RB=8 at F=7: 33 KiB LDS
    u64 span = 1ULL << (s - 1);
    if (f >= 2 && span >= RB) {
      [[maybe_unused]] auto sc = prof(1);
      u64 ntiles = n >> f;
      unsigned grid = ntiles > 2048 ? 2048 : (unsigned)ntiles;
      // wider residue blocks for the F=7 passes (512-B / 1-KiB contiguous
      // gathers; gfx950 allows the 66/132 KiB LDS workgroups) — MEASURED
      // SLOWER at k=22 (RB=8: 4.20 G elem/s, 16: 3.95, 32: 3.35): the
      // 66/132 KiB workgroups drop residency to 2/1 per CU and the lost
      // latency hiding outweighs the wider gathers, so 256-B chunks were
      // not the bottleneck. Kept behind TG_NTT_RB={8,16,32} as the A/B
      // evidence; default stays 8.
      static int rb_wide = [] {
        const char* e = getenv("TG_NTT_RB");
        int v = e ? atoi(e) : 8;
        return (v == 8 || v == 16 || v == 32) ? v : 8;
      }();
      if (f == 7 && rb_wide == 32 && span >= 32) {
        hipLaunchKernelGGL((k_ntt_fused2<FpCfg, 7, 32>), dim3(grid), dim3(256), 0, stream, d_tmp, tw, k, s - 1);
        s += f;
        continue;
      }
      if (f == 7 && rb_wide == 16 && span >= 16) {
        hipLaunchKernelGGL((k_ntt_fused2<FpCfg, 7, 16>), dim3(grid), dim3(256), 0, stream, d_tmp, tw, k, s - 1);
        s += f;
        continue;
      }
      switch (f) {
        case 7: hipLaunchKernelGGL((k_ntt_fused2<FpCfg, 7, RB>), dim3(grid), dim3(256), 0, stream, d_tmp, tw, k, s - 1); break;
        case 6: hipLaunchKernelGGL((k_ntt_fused2<FpCfg, 6, RB>), dim3(grid), dim3(256), 0, stream, d_tmp, tw, k, s - 1); break;
        case 5: hipLaunchKernelGGL((k_ntt_fused2<FpCfg, 5, RB>), dim3(grid), dim3(256), 0, stream, d_tmp, tw, k, s - 1); break;
        case 4: hipLaunchKernelGGL((k_ntt_fused2<FpCfg, 4, RB>), dim3(grid), dim3(256), 0, stream, d_tmp, tw, k, s - 1); break;
        case 3: hipLaunchKernelGGL((k_ntt_fused2<FpCfg, 3, RB>), dim3(grid), dim3(256), 0, stream, d_tmp, tw, k, s - 1); break;
        default: hipLaunchKernelGGL((k_ntt_fused2<FpCfg, 2, RB>), dim3(grid), dim3(256), 0, stream, d_tmp, tw, k, s - 1); break;
      }
      s += f;
      continue;
    }
    {
      [[maybe_unused]] auto sc = prof(2);
      hipLaunchKernelGGL(k_ntt_stage<FpCfg>, dim3(ntt_grid(n >> 1)), dim3(256), 0, stream,
                         d_tmp, tw, k, s);
    }
    s += 1;
  }
  if (inverse && ninv_mont) {
    [[maybe_unused]] auto sc = prof(3);
    hipLaunchKernelGGL(k_scale<FpCfg>, dim3(ntt_grid(n)), dim3(256), 0, stream, d_tmp, n,
                       *ninv_mont, 1, 0);
  }
  hipMemcpyAsync(d_a, d_tmp, n * sizeof(Fp), hipMemcpyDeviceToDevice, stream);
  return hipGetLastError();
}

}  // namespace taiga
