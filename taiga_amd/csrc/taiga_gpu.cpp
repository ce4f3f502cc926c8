// taiga_gpu.cpp — C-ABI host shim of libtaiga_gpu.so. PRODUCT CODE.
//
// Thin extern "C" layer over the gfx950 kernels (msm.hip, ntt.hip); see
// include/taiga_gpu.h for the boundary contract and the reference
// interfaces each entry point replaces. No CPU fallback: every entry point
// fails with TG_ERR_HIP when no HIP device works.

#include "../../include/taiga_gpu.h"

#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstring>
#include <string>
#include <vector>

#include "msm.hip"
#include "ntt.hip"
#include "poseidon.hip"
#include "binding_sig.hpp"
#include "tx_wire.hpp"
#include "fd28.hpp"
#include "witness.hpp"
#include "prover_impl.hpp"
#include "hf_rtc.hpp"

namespace taiga {

// ---------- small device kernels for API ingestion ----------

// canonical bytes -> Mont, with canonicality check (err bitmask != 0 on bad)
template <class C>
__global__ void __launch_bounds__(256) k_to_mont_check(Fd<C>* out, const Fd<C>* in, u64 n, unsigned* err) {
  for (u64 i = blockIdx.x * (u64)blockDim.x + threadIdx.x; i < n;
       i += (u64)gridDim.x * blockDim.x) {
    Fd<C> v = in[i];
    bool lt = false;
#pragma unroll
    for (int limb = 3; limb >= 0; limb--) {
      if (!lt) {
        if (v.l[limb] > C::MOD[limb]) { atomicOr(err, 1u); break; }
        if (v.l[limb] < C::MOD[limb]) lt = true;
      }
    }
    if (!lt) atomicOr(err, 1u);
    out[i] = fd_to_mont(v);
  }
}

template <class C>
__global__ void __launch_bounds__(256) k_from_mont(Fd<C>* out, const Fd<C>* in, u64 n) {
  for (u64 i = blockIdx.x * (u64)blockDim.x + threadIdx.x; i < n;
       i += (u64)gridDim.x * blockDim.x)
    out[i] = fd_from_mont(in[i]);
}

// canonical affine x||y pairs -> Mont affine (identity = all-zero pair)
__global__ void __launch_bounds__(256) k_aff_to_mont_check(VestaAff* out, const VestaAff* in, u64 n,
                                    unsigned* err) {
  for (u64 i = blockIdx.x * (u64)blockDim.x + threadIdx.x; i < n;
       i += (u64)gridDim.x * blockDim.x) {
    VestaAff p = in[i];
    if (fd_is_zero(p.x) && fd_is_zero(p.y)) {
      out[i] = p;  // identity stays (0,0)
      continue;
    }
    bool bad = false;
#pragma unroll
    for (int c = 0; c < 2; c++) {
      const Fq& v = c ? p.y : p.x;
      bool lt = false;
#pragma unroll
      for (int limb = 3; limb >= 0; limb--) {
        if (!lt) {
          if (v.l[limb] > FqCfg::MOD[limb]) { bad = true; break; }
          if (v.l[limb] < FqCfg::MOD[limb]) lt = true;
        }
      }
      if (!lt) bad = true;
    }
    VestaAff m;
    m.x = fd_to_mont(p.x);
    m.y = fd_to_mont(p.y);
    // on-curve check: y^2 == x^3 + 5
    Fq five{{5, 0, 0, 0}};
    Fq rhs = fd_add(fd_mul(fd_sqr(m.x), m.x), fd_to_mont(five));
    if (!fd_eq(fd_sqr(m.y), rhs)) bad = true;
    if (bad) atomicOr(err, 1u);
    out[i] = m;
  }
}

// decompress 32-byte compressed points -> Mont affine
__global__ void __launch_bounds__(256) k_decompress(VestaAff* out, const uint8_t* in, u64 n, unsigned* err) {
  for (u64 i = blockIdx.x * (u64)blockDim.x + threadIdx.x; i < n;
       i += (u64)gridDim.x * blockDim.x) {
    const uint8_t* b = in + 32 * i;
    u64 l[4];
    memcpy(l, b, 32);
    unsigned sign = (unsigned)(l[3] >> 63);
    l[3] &= 0x7FFFFFFFFFFFFFFFULL;
    if ((l[0] | l[1] | l[2] | l[3]) == 0) {
      if (sign) { atomicOr(err, 1u); }
      out[i].x = fd_zero<FqCfg>();
      out[i].y = fd_zero<FqCfg>();
      continue;
    }
    Fq x{{l[0], l[1], l[2], l[3]}};
    // canonical check
    bool lt = false, bad = false;
#pragma unroll
    for (int limb = 3; limb >= 0; limb--) {
      if (!lt) {
        if (x.l[limb] > FqCfg::MOD[limb]) { bad = true; break; }
        if (x.l[limb] < FqCfg::MOD[limb]) lt = true;
      }
    }
    if (!lt) bad = true;
    Fq xm = fd_to_mont(x);
    Fq five{{5, 0, 0, 0}};
    Fq rhs = fd_add(fd_mul(fd_sqr(xm), xm), fd_to_mont(five));
    Fq y;
    if (!bad && !fd_sqrt(y, rhs)) bad = true;
    if (bad) {
      atomicOr(err, 1u);
      out[i].x = fd_zero<FqCfg>();
      out[i].y = fd_zero<FqCfg>();
      continue;
    }
    if (fd_is_odd_std(y) != (bool)sign) y = fd_neg(y);
    out[i].x = xm;
    out[i].y = y;
  }
}

// ---------- profiling ----------

struct ProfCounter {
  std::vector<hipEvent_t> starts, stops;
  double done_ms = 0;
  long done_n = 0;
};

static const char* const PROF_NAMES[] = {
    "msm_digits", "msm_scan",  "msm_scatter", "msm_bucket_acc", "msm_reduce",
    "msm_wsum",   "ntt_stage", "ntt_fused",   "ntt_bitrev",     "ntt_scale",
    "msm_total",  "ntt_total"};
constexpr int PROF_N = sizeof(PROF_NAMES) / sizeof(PROF_NAMES[0]);

struct PPk;

struct Ctx {
  int device = 0;
  hipStream_t stream = nullptr;
  std::string err;

  // SRS
  int k = -1;
  VestaAff* d_g = nullptr;
  VestaAff* d_gl = nullptr;
  VestaAff h_w, h_u;  // Mont form, host copies

  // custom bases / staged inputs
  VestaAff* d_bases = nullptr;
  u64 n_bases = 0;
  bool bases_distinct = false;  // gen_bases output: distinct, non-identity
  ScalarRepr* d_scalars = nullptr;
  u64 n_scalars = 0;
  Fp* d_poly = nullptr;
  Fp* d_poly_tmp = nullptr;
  int poly_k = -1;

  MsmWork msm;
  NttPlan ntt;
  struct PPk* ppk = nullptr;            // active proving key (prover_gpu.inc)
  std::vector<struct PPk*> ppk_slots;   // PK cache: one entry
                                        // per tg_keygen call; slot id = index

  bool prof = false;
  bool pos_ready = false;  // poseidon constants uploaded to __constant__
  ProfCounter prof_c[PROF_N];
};

static thread_local std::string g_err;

static int set_err(Ctx* c, const char* where, hipError_t e) {
  std::string msg = std::string(where) + ": " + hipGetErrorString(e);
  if (c)
    c->err = msg;
  else
    g_err = msg;
  return TG_ERR_HIP;
}

// bind the calling thread to this ctx's device: HIP device selection is
// per-thread, and callers may invoke entries from worker threads (one
// process per GPU with thread pools — the 8-GPU bench path)
static int tg_enter(Ctx* c) {
  hipError_t e = hipSetDevice(c->device);
  if (e != hipSuccess) return set_err(c, "hipSetDevice", e);
  return 0;
}

struct ProfScope {
  Ctx* c;
  int idx;
  ProfScope(Ctx* c_, int idx_) : c(c_), idx(idx_) {
    if (c->prof) {
      hipEvent_t ev;
      hipEventCreate(&ev);
      hipEventRecord(ev, c->stream);
      c->prof_c[idx].starts.push_back(ev);
    }
  }
  ~ProfScope() {
    if (c->prof) {
      hipEvent_t ev;
      hipEventCreate(&ev);
      hipEventRecord(ev, c->stream);
      c->prof_c[idx].stops.push_back(ev);
    }
  }
};

enum {
  P_MSM_DIGITS = 0, P_MSM_SCAN, P_MSM_SCATTER, P_MSM_ACC, P_MSM_REDUCE,
  P_MSM_WSUM, P_NTT_STAGE, P_NTT_FUSED, P_NTT_BITREV, P_NTT_SCALE,
  P_MSM_TOTAL, P_NTT_TOTAL,
};

}  // namespace taiga

#include "prover_gpu.inc"

using namespace taiga;

extern "C" {

int tg_device_count(void) {
  int n = 0;
  if (hipGetDeviceCount(&n) != hipSuccess) return 0;
  return n;
}

const char* tg_error_string(const tg_ctx* ctx) {
  const Ctx* c = (const Ctx*)ctx;
  return c ? c->err.c_str() : g_err.c_str();
}

int tg_init(int device, tg_ctx** out) {
  if (!out) return TG_ERR_BADARG;
  hipError_t e = hipSetDevice(device);
  if (e != hipSuccess) return set_err(nullptr, "hipSetDevice", e);
  Ctx* c = new Ctx();
  c->device = device;
  if ((e = hipStreamCreate(&c->stream)) != hipSuccess) {
    delete c;
    return set_err(nullptr, "hipStreamCreate", e);
  }
  *out = (tg_ctx*)c;
  return TG_OK;
}

void tg_destroy(tg_ctx* ctx) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return;
  if (!c) return;
  hipStreamSynchronize(c->stream);
  for (PPk* pk : c->ppk_slots) {
    pdev_free(pk->pd);
    delete pk;
  }
  c->ppk_slots.clear();
  c->ppk = nullptr;
#define TGF(p) \
  if (p) hipFree(p)
  TGF(c->d_g); TGF(c->d_gl); TGF(c->d_bases); TGF(c->d_scalars);
  TGF(c->d_poly); TGF(c->d_poly_tmp);
  TGF(c->msm.d_dig); TGF(c->msm.d_hist); TGF(c->msm.d_off); TGF(c->msm.d_end);
  TGF(c->msm.d_bsum); TGF(c->msm.d_sorted); TGF(c->msm.d_buckets);
  TGF(c->msm.d_partials); TGF(c->msm.d_wsums);
  TGF(c->ntt.d_tw_fwd); TGF(c->ntt.d_tw_inv);
#undef TGF
  hipStreamDestroy(c->stream);
  delete c;
}

int tg_synchronize(tg_ctx* ctx) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return TG_ERR_BADARG;
  hipError_t e = hipStreamSynchronize(c->stream);
  if (e != hipSuccess) return set_err(c, "sync", e);
  return TG_OK;
}

void tg_prof_enable(tg_ctx* ctx, int on) { ((Ctx*)ctx)->prof = on != 0; }

void tg_prof_reset(tg_ctx* ctx) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return;
  for (auto& pc : c->prof_c) {
    for (auto ev : pc.starts) hipEventDestroy(ev);
    for (auto ev : pc.stops) hipEventDestroy(ev);
    pc.starts.clear();
    pc.stops.clear();
    pc.done_ms = 0;
    pc.done_n = 0;
  }
}

int tg_prof_get(tg_ctx* ctx, const char* name, double* total_ms, long* count) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return TG_ERR_BADARG;
  hipStreamSynchronize(c->stream);
  for (int i = 0; i < PROF_N; i++) {
    if (strcmp(PROF_NAMES[i], name) != 0) continue;
    ProfCounter& pc = c->prof_c[i];
    for (size_t j = 0; j < pc.stops.size(); j++) {
      float ms = 0;
      hipEventElapsedTime(&ms, pc.starts[j], pc.stops[j]);
      pc.done_ms += ms;
      pc.done_n++;
      hipEventDestroy(pc.starts[j]);
      hipEventDestroy(pc.stops[j]);
    }
    pc.starts.clear();
    pc.stops.clear();
    *total_ms = pc.done_ms;
    *count = pc.done_n;
    return TG_OK;
  }
  return TG_ERR_BADARG;
}

// ---------- SRS ----------

int tg_load_srs(tg_ctx* ctx, const uint8_t* bytes, size_t len) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return TG_ERR_BADARG;
  if (len < 4) return TG_ERR_BADARG;
  uint32_t k;
  memcpy(&k, bytes, 4);
  if (k > 28) return TG_ERR_BADARG;
  u64 n = 1ULL << k;
  if (len != 4 + 2 * n * 32 + 64) return TG_ERR_BADARG;
  hipError_t e;
  if (c->d_g) { hipFree(c->d_g); c->d_g = nullptr; }
  if (c->d_gl) { hipFree(c->d_gl); c->d_gl = nullptr; }
  if ((e = hipMalloc(&c->d_g, n * sizeof(VestaAff))) != hipSuccess)
    return set_err(c, "srs alloc g", e);
  if ((e = hipMalloc(&c->d_gl, n * sizeof(VestaAff))) != hipSuccess)
    return set_err(c, "srs alloc gl", e);
  uint8_t* d_comp = nullptr;
  unsigned* d_err = nullptr;
  if ((e = hipMalloc(&d_comp, 2 * n * 32 + 64)) != hipSuccess)
    return set_err(c, "srs alloc comp", e);
  hipMalloc(&d_err, 4);
  hipMemsetAsync(d_err, 0, 4, c->stream);
  hipMemcpyAsync(d_comp, bytes + 4, 2 * n * 32 + 64, hipMemcpyHostToDevice, c->stream);
  hipLaunchKernelGGL(k_decompress, dim3(msm_grid(n)), dim3(256), 0, c->stream, c->d_g,
                     d_comp, n, d_err);
  hipLaunchKernelGGL(k_decompress, dim3(msm_grid(n)), dim3(256), 0, c->stream, c->d_gl,
                     d_comp + n * 32, n, d_err);
  // w and u: decompress on device into a scratch, copy to host Mont form
  VestaAff* d_wu = nullptr;
  hipMalloc(&d_wu, 2 * sizeof(VestaAff));
  hipLaunchKernelGGL(k_decompress, dim3(1), dim3(64), 0, c->stream, d_wu,
                     d_comp + 2 * n * 32, 2, d_err);
  unsigned h_err = 0;
  hipMemcpyAsync(&h_err, d_err, 4, hipMemcpyDeviceToHost, c->stream);
  VestaAff h_wu[2];
  hipMemcpyAsync(h_wu, d_wu, 2 * sizeof(VestaAff), hipMemcpyDeviceToHost, c->stream);
  e = hipStreamSynchronize(c->stream);
  hipFree(d_comp);
  hipFree(d_wu);
  hipFree(d_err);
  if (e != hipSuccess) return set_err(c, "srs decompress", e);
  if (h_err) return TG_ERR_ENCODING;
  c->h_w = h_wu[0];
  c->h_u = h_wu[1];
  c->k = (int)k;
  return TG_OK;
}

int tg_srs_k(const tg_ctx* ctx) {
  const Ctx* c = (const Ctx*)ctx;
  return c->k >= 0 ? c->k : TG_ERR_NOSRS;
}

// ---------- MSM ----------

int tg_bases_upload(tg_ctx* ctx, const uint8_t* points_xy, size_t n) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return TG_ERR_BADARG;
  if (!n) return TG_ERR_BADARG;
  hipError_t e;
  if (c->d_bases && c->n_bases < n) { hipFree(c->d_bases); c->d_bases = nullptr; }
  if (!c->d_bases &&
      (e = hipMalloc(&c->d_bases, n * sizeof(VestaAff))) != hipSuccess)
    return set_err(c, "bases alloc", e);
  VestaAff* d_raw = nullptr;
  unsigned* d_err = nullptr;
  if ((e = hipMalloc(&d_raw, n * sizeof(VestaAff))) != hipSuccess)
    return set_err(c, "bases raw alloc", e);
  hipMalloc(&d_err, 4);
  hipMemsetAsync(d_err, 0, 4, c->stream);
  hipMemcpyAsync(d_raw, points_xy, n * 64, hipMemcpyHostToDevice, c->stream);
  hipLaunchKernelGGL(k_aff_to_mont_check, dim3(msm_grid(n)), dim3(256), 0, c->stream,
                     c->d_bases, d_raw, n, d_err);
  unsigned h_err = 0;
  hipMemcpyAsync(&h_err, d_err, 4, hipMemcpyDeviceToHost, c->stream);
  e = hipStreamSynchronize(c->stream);
  hipFree(d_raw);
  hipFree(d_err);
  if (e != hipSuccess) return set_err(c, "bases upload", e);
  if (h_err) return TG_ERR_ENCODING;
  c->n_bases = n;
  c->bases_distinct = false;
  return TG_OK;
}

int tg_gen_bases(tg_ctx* ctx, size_t n, uint64_t seed) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return TG_ERR_BADARG;
  if (!n) return TG_ERR_BADARG;
  hipError_t e;
  if (c->d_bases && c->n_bases < n) { hipFree(c->d_bases); c->d_bases = nullptr; }
  if (!c->d_bases &&
      (e = hipMalloc(&c->d_bases, n * sizeof(VestaAff))) != hipSuccess)
    return set_err(c, "bases alloc", e);
  hipLaunchKernelGGL(k_gen_bases, dim3(msm_grid(n)), dim3(256), 0, c->stream, c->d_bases,
                     n, seed);
  e = hipStreamSynchronize(c->stream);
  if (e != hipSuccess) return set_err(c, "gen bases", e);
  c->n_bases = n;
  c->bases_distinct = true;
  return TG_OK;
}

int tg_bases_download(tg_ctx* ctx, size_t n, uint8_t* out_xy) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return TG_ERR_BADARG;
  if (!n || !c->d_bases || c->n_bases < n) return TG_ERR_BADARG;
  hipError_t e;
  Fq* tmp = nullptr;
  if ((e = hipMalloc(&tmp, n * sizeof(VestaAff))) != hipSuccess)
    return set_err(c, "bases dl alloc", e);
  hipLaunchKernelGGL(k_from_mont<FqCfg>, dim3(ntt_grid(2 * n)), dim3(256), 0, c->stream,
                     tmp, (const Fq*)c->d_bases, (u64)(2 * n));
  hipMemcpyAsync(out_xy, tmp, n * 64, hipMemcpyDeviceToHost, c->stream);
  e = hipStreamSynchronize(c->stream);
  hipFree(tmp);
  return e == hipSuccess ? TG_OK : set_err(c, "bases dl", e);
}

int tg_scalars_upload(tg_ctx* ctx, const uint8_t* scalars, size_t n) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return TG_ERR_BADARG;
  if (!n) return TG_ERR_BADARG;
  hipError_t e;
  if (c->d_scalars && c->n_scalars < n) { hipFree(c->d_scalars); c->d_scalars = nullptr; }
  if (!c->d_scalars &&
      (e = hipMalloc(&c->d_scalars, n * sizeof(ScalarRepr))) != hipSuccess)
    return set_err(c, "scalars alloc", e);
  hipMemcpyAsync(c->d_scalars, scalars, n * 32, hipMemcpyHostToDevice, c->stream);
  e = hipStreamSynchronize(c->stream);
  if (e != hipSuccess) return set_err(c, "scalars upload", e);
  c->n_scalars = n;
  return TG_OK;
}

static int msm_common(Ctx* c, size_t n, int base_set, uint8_t out_xy[64]) {
  // custom uploaded bases may contain identities/duplicates -> SAFE variant;
  // SRS bases are distinct non-identity points -> fast branchless variant
  const bool safe = base_set == 0 && !c->bases_distinct;
  const VestaAff* bases = nullptr;
  if (base_set == 0) {
    if (c->n_bases < n) return TG_ERR_STATE;
    bases = c->d_bases;
  } else {
    if (c->k < 0) return TG_ERR_NOSRS;
    if (n > (1ULL << c->k)) return TG_ERR_BADARG;
    bases = base_set == 1 ? c->d_g : c->d_gl;
  }
  hipError_t e;
  if ((e = msm_work_alloc(c->msm, n)) != hipSuccess) return set_err(c, "msm ws", e);
  MsmCfg cfg = msm_cfg((long)n);
  {
    ProfScope total(c, P_MSM_TOTAL);
    u64 m = (u64)cfg.nwin * cfg.nbuck;
    {
      ProfScope p(c, P_MSM_DIGITS);
      hipMemsetAsync(c->msm.d_hist, 0, m * 4, c->stream);
      hipLaunchKernelGGL(k_digits, dim3(msm_grid(n)), dim3(256), 0, c->stream,
                         c->d_scalars, n, n, cfg, c->msm.d_dig, c->msm.d_hist);
    }
    {
      ProfScope p(c, P_MSM_SCAN);
      msm_scan(c->msm.d_hist, c->msm.d_off, c->msm.d_bsum, m, c->stream);
      hipMemcpyAsync(c->msm.d_hist, c->msm.d_off, m * 4, hipMemcpyDeviceToDevice,
                     c->stream);
    }
    {
      ProfScope p(c, P_MSM_SCATTER);
      hipLaunchKernelGGL(k_scatter, dim3(msm_grid(n)), dim3(256), 0, c->stream,
                         c->msm.d_dig, n, n, cfg, c->msm.d_off, c->msm.d_sorted);
      hipMemcpyAsync(c->msm.d_end, c->msm.d_off, m * 4, hipMemcpyDeviceToDevice,
                     c->stream);
    }
    {
      ProfScope p(c, P_MSM_ACC);
      hipMemsetAsync(c->msm.d_big + m, 0, 4, c->stream);
      const uint32_t* ord = msm_len_sort(c->msm, cfg, m, c->stream);
      if (safe) {
        hipLaunchKernelGGL(k_bucket_acc<true>, dim3(msm_grid(m)), dim3(256), 0, c->stream,
                           c->msm.d_hist, c->msm.d_end, c->msm.d_sorted, bases,
                           c->msm.d_buckets, m, c->msm.d_big, c->msm.d_big + m, ord);
        hipLaunchKernelGGL(k_bucket_acc_big<true>, dim3(1024), dim3(MSM_BIG_LANES), 0, c->stream,
                           c->msm.d_hist, c->msm.d_end, c->msm.d_sorted, bases,
                           c->msm.d_buckets, c->msm.d_big, c->msm.d_big + m);
      } else {
        hipLaunchKernelGGL(k_bucket_acc<false>, dim3(msm_grid(m)), dim3(256), 0, c->stream,
                           c->msm.d_hist, c->msm.d_end, c->msm.d_sorted, bases,
                           c->msm.d_buckets, m, c->msm.d_big, c->msm.d_big + m, ord);
        hipLaunchKernelGGL(k_bucket_acc_big<false>, dim3(1024), dim3(MSM_BIG_LANES), 0, c->stream,
                           c->msm.d_hist, c->msm.d_end, c->msm.d_sorted, bases,
                           c->msm.d_buckets, c->msm.d_big, c->msm.d_big + m);
      }
    }
    {
      ProfScope p(c, P_MSM_REDUCE);
      hipLaunchKernelGGL(k_bucket_reduce, dim3(msm_grid((u64)cfg.nwin * cfg.nseg)),
                         dim3(256), 0, c->stream, c->msm.d_buckets, c->msm.d_partials,
                         (u64)cfg.nwin, cfg);
    }
    {
      ProfScope p(c, P_MSM_WSUM);
      hipLaunchKernelGGL(k_wsum, dim3(cfg.nwin), dim3(256), 0, c->stream,
                         c->msm.d_partials, c->msm.d_wsums, cfg);
    }
  }
  VestaJac wsums[32]; /* nwin <= 32 (c >= 8) */
  hipMemcpyAsync(wsums, c->msm.d_wsums, sizeof(VestaJac) * cfg.nwin,
                 hipMemcpyDeviceToHost, c->stream);
  hipError_t es = hipStreamSynchronize(c->stream);
  if (es != hipSuccess) return set_err(c, "msm run", es);
  VestaAff r = msm_host_combine(wsums, cfg);
  if (aff_is_identity(r)) {
    memset(out_xy, 0, 64);
  } else {
    Fq x = fd_from_mont(r.x), y = fd_from_mont(r.y);
    memcpy(out_xy, x.l, 32);
    memcpy(out_xy + 32, y.l, 32);
  }
  return TG_OK;
}

int tg_msm_resident(tg_ctx* ctx, size_t n, int base_set, uint8_t out_xy[64]) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return TG_ERR_BADARG;
  if (c->n_scalars < n) return TG_ERR_STATE;
  return msm_common(c, n, base_set, out_xy);
}

int tg_msm_pallas(tg_ctx* ctx, const uint8_t* scalars, size_t n, int base_set,
                  uint8_t out_xy[64]) {
  int rc = tg_scalars_upload(ctx, scalars, n);
  if (rc != TG_OK) return rc;
  return msm_common((Ctx*)ctx, n, base_set, out_xy);
}

// ---------- NTT ----------

int tg_poly_upload(tg_ctx* ctx, const uint8_t* poly, uint32_t k) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return TG_ERR_BADARG;
  if (k > 28) return TG_ERR_BADARG;
  u64 n = 1ULL << k;
  hipError_t e;
  if (c->poly_k != (int)k) {
    if (c->d_poly) { hipFree(c->d_poly); c->d_poly = nullptr; }
    if (c->d_poly_tmp) { hipFree(c->d_poly_tmp); c->d_poly_tmp = nullptr; }
    if ((e = hipMalloc(&c->d_poly, n * sizeof(Fp))) != hipSuccess)
      return set_err(c, "poly alloc", e);
    if ((e = hipMalloc(&c->d_poly_tmp, n * sizeof(Fp))) != hipSuccess)
      return set_err(c, "poly tmp alloc", e);
    c->poly_k = (int)k;
  }
  unsigned* d_err = nullptr;
  hipMalloc(&d_err, 4);
  hipMemsetAsync(d_err, 0, 4, c->stream);
  // upload canonical into tmp, convert to Mont into d_poly
  hipMemcpyAsync(c->d_poly_tmp, poly, n * 32, hipMemcpyHostToDevice, c->stream);
  hipLaunchKernelGGL(k_to_mont_check<FpCfg>, dim3(ntt_grid(n)), dim3(256), 0, c->stream,
                     c->d_poly, c->d_poly_tmp, n, d_err);
  unsigned h_err = 0;
  hipMemcpyAsync(&h_err, d_err, 4, hipMemcpyDeviceToHost, c->stream);
  e = hipStreamSynchronize(c->stream);
  hipFree(d_err);
  if (e != hipSuccess) return set_err(c, "poly upload", e);
  if (h_err) return TG_ERR_ENCODING;
  return TG_OK;
}

int tg_ntt_resident(tg_ctx* ctx, int dir, uint32_t k, int coset) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return TG_ERR_BADARG;
  if (coset) return TG_ERR_BADARG;  // arrives with the prover pipeline
  if (c->poly_k != (int)k) return TG_ERR_STATE;
  hipError_t e;
  if ((e = ntt_plan_init(c->ntt, (int)k, c->stream)) != hipSuccess)
    return set_err(c, "ntt plan", e);
  Fp ninv;
  if (dir) {
    // n^{-1} in Mont form, computed host-side with the shared primitives
    Fp nstd{{1ULL << k, 0, 0, 0}};
    ninv = fd_inv(fd_to_mont(nstd));
  }
  {
    ProfScope total(c, P_NTT_TOTAL);
    static const int NTT_PROF_MAP[4] = {P_NTT_BITREV, P_NTT_FUSED, P_NTT_STAGE,
                                        P_NTT_SCALE};
    auto prof = [&](int idx) { return ProfScope(c, NTT_PROF_MAP[idx]); };
    if ((e = ntt_run(c->d_poly, c->d_poly_tmp, c->ntt, (int)k, dir != 0, c->stream,
                     dir ? &ninv : nullptr, prof)) != hipSuccess)
      return set_err(c, "ntt run", e);
  }
  e = hipStreamSynchronize(c->stream);
  if (e != hipSuccess) return set_err(c, "ntt sync", e);
  return TG_OK;
}

int tg_poly_download(tg_ctx* ctx, uint8_t* poly, uint32_t k) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return TG_ERR_BADARG;
  if (c->poly_k != (int)k) return TG_ERR_STATE;
  u64 n = 1ULL << k;
  // from_mont into tmp, then D2H
  hipLaunchKernelGGL(k_from_mont<FpCfg>, dim3(ntt_grid(n)), dim3(256), 0, c->stream,
                     c->d_poly_tmp, c->d_poly, n);
  hipMemcpyAsync(poly, c->d_poly_tmp, n * 32, hipMemcpyDeviceToHost, c->stream);
  hipError_t e = hipStreamSynchronize(c->stream);
  if (e != hipSuccess) return set_err(c, "poly download", e);
  return TG_OK;
}

int tg_ntt_fp(tg_ctx* ctx, int dir, uint32_t k, int coset, uint8_t* poly) {
  int rc = tg_poly_upload(ctx, poly, k);
  if (rc != TG_OK) return rc;
  rc = tg_ntt_resident(ctx, dir, k, coset);
  if (rc != TG_OK) return rc;
  return tg_poly_download(ctx, poly, k);
}

}  // extern "C"

/* ---- prover ABI (implementation in prover_gpu.inc) ---- */
extern "C" {

/* builds a proving key and caches it; returns the non-negative slot id
 * (also leaves it active for the slotless entry points). PKs are cached for
 * the ctx lifetime — the reference's generic-macro path re-keygens per
 * proof (resource_logic_circuit.rs:578-580); the cache is §8f-1's fix. */
int tg_keygen(tg_ctx* ctx, const uint8_t* desc, size_t desc_len) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return TG_ERR_BADARG;
  if (c->k < 0) return TG_ERR_NOSRS;
  PPk* pk = new PPk();
  int rc = ppk_keygen(c, *pk, desc, desc_len);
  if (rc != 0) {
    delete pk;
    return rc;
  }
  c->ppk_slots.push_back(pk);
  c->ppk = pk;
  return (int)c->ppk_slots.size() - 1;
}

/* select a previously built proving key by slot id */
int tg_select_key(tg_ctx* ctx, int slot) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return TG_ERR_BADARG;
  if (slot < 0 || (size_t)slot >= c->ppk_slots.size()) return TG_ERR_BADARG;
  c->ppk = c->ppk_slots[slot];
  return TG_OK;
}

int tg_create_proof(tg_ctx* ctx, const uint8_t inst_seed[32], const uint8_t wit_seed[32],
                    const uint8_t rng_seed[32], uint8_t* proof_out, size_t cap,
                    size_t* out_len) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return TG_ERR_BADARG;
  if (!c->ppk || !c->ppk->ready) return TG_ERR_STATE;
  std::vector<uint8_t> proof;
  int rc = pprove(c, *c->ppk, inst_seed, wit_seed, rng_seed, proof);
  if (rc != 0) return rc;
  if (proof.size() > cap) return TG_ERR_BADARG;
  memcpy(proof_out, proof.data(), proof.size());
  *out_len = proof.size();
  return TG_OK;
}

int tg_witness_hash(tg_ctx* ctx, const uint8_t inst_seed[32], const uint8_t wit_seed[32],
                    uint8_t out[32]) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return TG_ERR_BADARG;
  if (!c->ppk || !c->ppk->ready) return TG_ERR_STATE;
  PDesc& d = c->ppk->d;
  std::vector<Fp> inst;
  cs1_instance(d, inst_seed, inst);
  std::vector<std::vector<Fp>> adv;
  cs1_witness(d, wit_seed, inst, adv);
  Blake2b h(32);
  std::vector<uint8_t> buf(32 * (size_t)d.n);
  for (int col = 0; col < d.n_advice; col++) {
#ifdef _OPENMP
#pragma omp parallel for schedule(static)
#endif
    for (long i = 0; i < d.n; i++) {
      Fp v = fd_from_mont(adv[col][i]);
      memcpy(buf.data() + 32 * i, v.l, 32);
    }
    h.update(buf.data(), buf.size());
  }
  h.final(out);
  return TG_OK;
}

}  /* extern "C" */

extern "C" {

int tg_verify_proof(tg_ctx* ctx, const uint8_t inst_seed[32], const uint8_t* proof,
                    size_t proof_len) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return TG_ERR_BADARG;
  if (!c->ppk || !c->ppk->ready) return TG_ERR_STATE;
  return pverify(c, *c->ppk, inst_seed, proof, proof_len);
}

/* raw-instance verification: instance = n_instance_rows x 32B canonical
 * reprs (the drop-in shape of plonk::verify_proof's instance slices —
 * proof.rs:45-54). */
int tg_verify_proof_raw(tg_ctx* ctx, const uint8_t* instance, const uint8_t* proof,
                        size_t proof_len) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return TG_ERR_BADARG;
  if (!c->ppk || !c->ppk->ready) return TG_ERR_STATE;
  if (!instance) return TG_ERR_BADARG;
  return pverify_raw(c, *c->ppk, instance, proof, proof_len);
}

/* batch verification (the halo2 BatchVerifier / Guard
 * accumulation pattern behind proof.rs:45-54): each proof contributes a
 * final-check guard; guards are combined with random weights so that all
 * m IPA final checks collapse into ONE g-sized GPU MSM plus a few hundred
 * variable points. Weights are derived by hashing the whole batch (a
 * prover cannot predict them). All proofs must be for the active key.
 * Returns TG_OK iff every proof is valid; -1 if the combined check fails;
 * -1xx on the first structurally malformed proof. */
int tg_verify_batch(tg_ctx* ctx, size_t m, const uint8_t* inst_seeds,
                    const uint8_t* proofs, const size_t* proof_lens) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return TG_ERR_BADARG;
  if (!c->ppk || !c->ppk->ready) return TG_ERR_STATE;
  if (m == 0 || m > 4096 || !inst_seeds || !proofs || !proof_lens) return TG_ERR_BADARG;
  std::vector<PVGuard> gds(m);
  size_t off = 0;
  for (size_t i = 0; i < m; i++) {
    std::vector<Fp> inst_lag;
    cs1_instance(c->ppk->d, inst_seeds + 32 * i, inst_lag);
    int rc = pverify_guard(c, *c->ppk, inst_lag, proofs + off,
                           proof_lens[i], gds[i]);
    if (rc != 0) return rc;
    off += proof_lens[i];
  }
  /* weights: rho_0 = 1, rho_i = DRBG(Blake2b(seeds ‖ proofs)) field draws */
  std::vector<Fp> rho(m);
  rho[0] = fd_one_mont<FpCfg>();
  if (m > 1) {
    Blake2b h(32, (const uint8_t*)"TaigaGPU-BatchVf");
    uint64_t mle = (uint64_t)m;
    h.update((const uint8_t*)&mle, 8);
    h.update(inst_seeds, 32 * m);
    h.update(proofs, off);
    uint8_t seed[32];
    h.final(seed);
    Drbg rng(seed);
    for (size_t i = 1; i < m; i++) rho[i] = rng.field<FpCfg>();
  }
  return pverify_eval(c, *c->ppk, gds.data(), (int)m, rho.data());
}

/* raw-instance batch verification: the ptx-bundle shape (§8f-3 + §8f-4
 * composed): instances = concatenated n_instance_rows x 32B blocks, one
 * per proof, for the ACTIVE key. Same combined-check semantics as
 * tg_verify_batch. */
int tg_verify_batch_raw(tg_ctx* ctx, size_t m, const uint8_t* instances,
                        const uint8_t* proofs, const size_t* proof_lens) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return TG_ERR_BADARG;
  if (!c->ppk || !c->ppk->ready) return TG_ERR_STATE;
  if (m == 0 || m > 4096 || !instances || !proofs || !proof_lens) return TG_ERR_BADARG;
  size_t inst_stride = 32 * (size_t)c->ppk->d.n_instance_rows;
  std::vector<PVGuard> gds(m);
  size_t off = 0;
  for (size_t i = 0; i < m; i++) {
    std::vector<Fp> inst_lag;
    int rc = pinst_from_raw(c->ppk->d, instances + inst_stride * i, inst_lag);
    if (rc != 0) return rc;
    rc = pverify_guard(c, *c->ppk, inst_lag, proofs + off, proof_lens[i], gds[i]);
    if (rc != 0) return rc;
    off += proof_lens[i];
  }
  std::vector<Fp> rho(m);
  rho[0] = fd_one_mont<FpCfg>();
  if (m > 1) {
    Blake2b h(32, (const uint8_t*)"TaigaGPU-BatchVf");
    uint64_t mle = (uint64_t)m;
    h.update((const uint8_t*)&mle, 8);
    h.update(instances, inst_stride * m);
    h.update(proofs, off);
    uint8_t seed[32];
    h.final(seed);
    Drbg rng(seed);
    for (size_t i = 1; i < m; i++) rho[i] = rng.field<FpCfg>();
  }
  return pverify_eval(c, *c->ppk, gds.data(), (int)m, rho.data());
}

/* experiment self-test (fd28.hpp — carry-chain-free radix-2^28
 * Montgomery multiply, round-2 kernel candidate): host-side evaluation of
 * out = a * b * 2^-280 mod p over canonical Fp bytes. The arithmetic is
 * host/device-shared TG_HD code, so this pins the device semantics
 * numerically without a GPU (tests/test_fd28.py). Not a product entry. */
int tg_dbg_fd28_mul(const uint8_t a_bytes[32], const uint8_t b_bytes[32],
                    uint8_t out[32]) {
  Fp a, b;
  memcpy(a.l, a_bytes, 32);
  memcpy(b.l, b_bytes, 32);
  auto canon = [](const Fp& v) {
    for (int limb = 3; limb >= 0; limb--) {
      if (v.l[limb] > FpCfg::MOD[limb]) return false;
      if (v.l[limb] < FpCfg::MOD[limb]) return true;
    }
    return false;  // equal to p
  };
  if (!canon(a) || !canon(b)) return TG_ERR_ENCODING;
  Fd28<FpCfg> r = fd28_mul(fd28_from<FpCfg>(a), fd28_from<FpCfg>(b));
  Fp v = fd28_norm(r);
  memcpy(out, v.l, 32);
  return TG_OK;
}

/* ---- binding signatures + transaction digest (host-side wire layer,
 * SURVEY §8f-4; binding_signature.rs / transaction.rs:116-158) ----
 * RedDSA over Pallas, H* = BLAKE2b-512("Taiga_RedPallasH"), ctx-free.
 * See binding_sig.hpp for the basepoint pin status. */
int tg_binding_vk(const uint8_t sk[32], uint8_t vk_out[32]) {
  return bs_derive_vk(vk_out, sk) ? TG_ERR_ENCODING : TG_OK;
}

int tg_delta_commit(const uint8_t r[32], uint8_t cv_out[32]) {
  Fq s;
  if (!bs_scalar_from_bytes(s, r)) return TG_ERR_ENCODING;
  pallas_compress(cv_out, pallas_mul(pallas_basepoint(), s));
  return TG_OK;
}

int tg_binding_sign(const uint8_t sk[32], const uint8_t* msg, size_t msg_len,
                    const uint8_t rng_seed[32], uint8_t sig_out[64]) {
  if (!msg && msg_len) return TG_ERR_BADARG;
  return bs_sign(sig_out, sk, msg, msg_len, rng_seed) ? TG_ERR_ENCODING : TG_OK;
}

int tg_binding_verify(const uint8_t vk[32], const uint8_t* msg, size_t msg_len,
                      const uint8_t sig[64]) {
  if (!msg && msg_len) return TG_ERR_BADARG;
  return bs_verify(vk, msg, msg_len, sig) ? -1 : TG_OK;
}

int tg_binding_vk_from_deltas(const uint8_t* deltas, size_t n, uint8_t vk_out[32]) {
  if (n && !deltas) return TG_ERR_BADARG;
  return bs_vk_from_deltas(vk_out, deltas, n) ? TG_ERR_ENCODING : TG_OK;
}

int tg_tx_digest(const uint8_t* nfs, size_t n_nf, const uint8_t* cms, size_t n_cm,
                 const uint8_t* deltas, size_t n_delta, const uint8_t* anchors,
                 size_t n_anchor, uint8_t out[32]) {
  bs_tx_digest(out, nfs, n_nf, cms, n_cm, deltas, n_delta, anchors, n_anchor);
  return TG_OK;
}

/* ---- transaction wire format (borsh layouts cited in
 * tx_wire.hpp) ---- */

/* ctx-free wire check: parse a borsh Transaction, recompute
 * Transaction::digest from the compliance instances, aggregate delta
 * commitments into the binding vk and verify the binding signature.
 * vk_len = byte length of one embedded resource-logic VerifyingKey
 * (32*(n_fixed+n_perm) for TGD1 circuits). Returns TG_OK; -1 on a bad
 * binding signature; -2xx on structural decode errors. */
int tg_tx_wire_check(const uint8_t* tx, size_t len, uint32_t vk_len,
                     uint32_t* n_sptx, uint32_t* n_proofs) {
  if (!tx) return TG_ERR_BADARG; /* short inputs fail structurally in the parser */
  return tx_check(tx, len, vk_len, n_sptx, n_proofs, nullptr);
}

/* full transaction verification against the ACTIVE key: wire check +
 * binding signature + ONE combined batch verification of every compliance
 * proof in the bundle. Round-1 instance mapping: the first
 * n_instance_rows fields of each 192-byte compliance instance block
 * (requires n_instance_rows <= 6). The REAL compliance circuit expands
 * the 6 borsh fields into 9 instance rows — [nf, anchor, cm, delta.x,
 * delta.y, rl_in x2 halves, rl_out x2 halves], constant.rs:54-62 — which
 * replaces this mapping in round 2. */
int tg_tx_verify(tg_ctx* ctx, const uint8_t* tx, size_t len) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return TG_ERR_BADARG;
  if (!c->ppk || !c->ppk->ready) return TG_ERR_STATE;
  PDesc& d = c->ppk->d;
  if (d.n_instance_rows > 6) return TG_ERR_BADARG;
  uint32_t vk_len = 32u * (uint32_t)(d.n_fixed + d.n_perm);
  TxDigestStreams st;
  int rc = tx_check(tx, len, vk_len, nullptr, nullptr, &st);
  if (rc) return rc;
  size_t m = st.proof_ptr.size();
  if (m == 0) return TG_OK;
  std::vector<PVGuard> gds(m);
  for (size_t i = 0; i < m; i++) {
    std::vector<Fp> inst_lag;
    rc = pinst_from_raw(d, st.inst_ptr[i], inst_lag);
    if (rc) return rc;
    rc = pverify_guard(c, *c->ppk, inst_lag, st.proof_ptr[i], st.proof_len[i], gds[i]);
    if (rc) return rc;
  }
  std::vector<Fp> rho(m);
  rho[0] = fd_one_mont<FpCfg>();
  if (m > 1) {
    Blake2b h(32, (const uint8_t*)"TaigaGPU-BatchVf");
    h.update(tx, len);
    uint8_t seed[32];
    h.final(seed);
    Drbg rng(seed);
    for (size_t i = 1; i < m; i++) rho[i] = rng.field<FpCfg>();
  }
  return pverify_eval(c, *c->ppk, gds.data(), (int)m, rho.data());
}

static bool compliance_rows_expand(const uint8_t* inst192, uint8_t rows[288]);
static std::vector<uint8_t> pk_vk_bytes(PPk* k);

/* FULL transaction verification (round 2; supersedes tg_tx_verify's
 * compliance-only check — ADVICE.md item 2): borsh Transaction parse +
 * Transaction::digest + binding signature (tx_check), then batch-verify
 * EVERY compliance proof (slot_compliance; 192-B instances expanded to
 * the real 9 rows) and EVERY resource-logic proof (slot_rl; vk bytes
 * must match the slot), plus the per-sptx execute() consistency checks.
 * TG_OK only when the whole transaction is valid. */
int tg_tx_verify_full(tg_ctx* ctx, int slot_compliance, int slot_rl,
                      const uint8_t* tx, size_t len) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return TG_ERR_BADARG;
  int rc = tg_select_key(ctx, slot_rl);
  if (rc) return rc;
  PPk* rlk = c->ppk;
  uint32_t vk_len = 32u * (uint32_t)(rlk->fixed_commits.size() +
                                     rlk->sigma_commits.size());
  TxDigestStreams st;
  rc = tx_check(tx, len, vk_len, nullptr, nullptr, &st);
  if (rc) return rc;
  // vk match + consistency checks per sptx
  std::vector<uint8_t> vk_ref = pk_vk_bytes(rlk);
  for (const uint8_t* vkb : st.rl_vk_ptr)
    if (memcmp(vkb, vk_ref.data(), vk_len) != 0) return -207;
  st.sptx_comp_begin.push_back(st.n_compliance);
  st.sptx_rl_begin.push_back(st.n_rl);
  std::vector<uint8_t> cinst;
  for (uint32_t i = 0; i < st.n_compliance; i++) {
    uint8_t rows[288];
    if (!compliance_rows_expand(st.inst_ptr[i], rows)) return -204;
    cinst.insert(cinst.end(), rows, rows + 288);
  }
  for (uint32_t s2 = 0; s2 < st.n_sptx; s2++) {
    uint32_t c0 = st.sptx_comp_begin[s2], c1 = st.sptx_comp_begin[s2 + 1];
    uint32_t r0 = st.sptx_rl_begin[s2], r1 = st.sptx_rl_begin[s2 + 1];
    if (r1 == r0) continue;
    const uint8_t* root0 = st.rl_inst_ptr[r0];
    uint32_t in_seen = 0, out_seen = 0;
    for (uint32_t r = r0; r < r1; r++) {
      if (memcmp(st.rl_inst_ptr[r], root0, 32) != 0) return -301;
      const uint8_t* self_id = st.rl_inst_ptr[r] + 32;
      if (st.rl_is_input[r]) {
        uint32_t ci = c0 + in_seen++;
        if (ci >= c1 || memcmp(self_id, cinst.data() + 288ul * ci, 32) != 0)
          return -303;
      } else {
        uint32_t ci = c0 + out_seen++;
        if (ci >= c1 || memcmp(self_id, cinst.data() + 288ul * ci + 64, 32) != 0)
          return -304;
      }
    }
  }
  // batch-verify RL proofs (RL key active)
  if (st.n_rl) {
    std::vector<uint8_t> insts;
    std::vector<uint8_t> proofs;
    std::vector<size_t> lens;
    for (uint32_t r = 0; r < st.n_rl; r++) {
      insts.insert(insts.end(), st.rl_inst_ptr[r], st.rl_inst_ptr[r] + 22 * 32);
      proofs.insert(proofs.end(), st.rl_proof_ptr[r],
                    st.rl_proof_ptr[r] + st.rl_proof_len[r]);
      lens.push_back(st.rl_proof_len[r]);
    }
    rc = tg_verify_batch_raw(ctx, st.n_rl, insts.data(), proofs.data(),
                             lens.data());
    if (rc) return rc == TG_ERR_BADARG ? rc : -1;
  }
  // batch-verify compliance proofs
  rc = tg_select_key(ctx, slot_compliance);
  if (rc) return rc;
  if (st.n_compliance) {
    std::vector<uint8_t> proofs;
    for (uint32_t i = 0; i < st.n_compliance; i++)
      proofs.insert(proofs.end(), st.proof_ptr[i],
                    st.proof_ptr[i] + st.proof_len[i]);
    rc = tg_verify_batch_raw(ctx, st.n_compliance, cinst.data(), proofs.data(),
                             st.proof_len.data());
    if (rc) return rc == TG_ERR_BADARG ? rc : -1;
  }
  return TG_OK;
}

/* batched Poseidon P128Pow5T3 ConstantLength<L> hashing (GPU witness
 * synthesis; replaces host-side halo2_gadgets poseidon
 * hashing — utils.rs:40-48): msgs = n x L x 32B canonical reprs,
 * out = n x 32B canonical digests. One hash per GPU thread. */
int tg_poseidon_hash(tg_ctx* ctx, const uint8_t* msgs, size_t n, int L,
                     uint8_t* out) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return TG_ERR_BADARG;
  if (!msgs || !out || n == 0 || L < 1 || L > 64) return TG_ERR_BADARG;
  hipError_t e = hipSetDevice(c->device);
  if (e != hipSuccess) return set_err(c, "hipSetDevice", e);
  if (!c->pos_ready) {
    if ((e = pos_upload_consts()) != hipSuccess)
      return set_err(c, "poseidon consts", e);
    c->pos_ready = true;
  }
  Fp* d_in = nullptr;
  Fp* d_out = nullptr;
  unsigned* d_err = nullptr;
  size_t in_bytes = n * (size_t)L * 32;
  if ((e = hipMalloc(&d_in, in_bytes)) != hipSuccess) return TG_ERR_NOMEM;
  if ((e = hipMalloc(&d_out, n * 32)) != hipSuccess) { hipFree(d_in); return TG_ERR_NOMEM; }
  if ((e = hipMalloc(&d_err, 4)) != hipSuccess) { hipFree(d_in); hipFree(d_out); return TG_ERR_NOMEM; }
  hipMemsetAsync(d_err, 0, 4, c->stream);
  hipMemcpyAsync(d_in, msgs, in_bytes, hipMemcpyHostToDevice, c->stream);
  u64 grid = (n + 255) / 256;
  if (grid > 16384) grid = 16384;
  hipLaunchKernelGGL(k_poseidon_hash, dim3(grid), dim3(256), 0, c->stream,
                     d_out, d_in, (u64)n, L, d_err);
  unsigned h_err = 0;
  hipMemcpyAsync(&h_err, d_err, 4, hipMemcpyDeviceToHost, c->stream);
  std::vector<uint8_t> h_out(n * 32);
  hipMemcpyAsync(h_out.data(), d_out, n * 32, hipMemcpyDeviceToHost, c->stream);
  e = hipStreamSynchronize(c->stream);
  hipFree(d_in);
  hipFree(d_out);
  hipFree(d_err);
  if (e != hipSuccess) return set_err(c, "poseidon", e);
  if (h_err) return TG_ERR_ENCODING;
  memcpy(out, h_out.data(), n * 32);
  return TG_OK;
}

}  /* extern "C" */

extern "C" {

/* raw-witness proving (SURVEY §8b witness-blob shape): instance =
 * n_instance_rows x 32B canonical reprs; advice = n_advice x 2^k x 32B
 * canonical, column-major (rows beyond usable = n-(bf+1) are replaced by
 * blinding). */
/* attach a TGW1 witness-synthesis program (tools/circuit/emit.py) to the
 * ACTIVE key slot; required by tg_compliance_prove / tg_rl_prove */
int tg_witness_program_load(tg_ctx* ctx, const uint8_t* tgw, size_t len) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return TG_ERR_BADARG;
  if (!c->ppk || !c->ppk->ready) return TG_ERR_STATE;
  if (!c->ppk->tgw.parse(tgw, len)) return TG_ERR_BADARG;
  if ((int)c->ppk->tgw.k != c->ppk->d.k) return TG_ERR_BADARG;
  return TG_OK;
}

/* drop-in ComplianceInfo prove (replaces ComplianceInfo::build +
 * Proof::create — compliance.rs:190-233 + proof.rs:25-42): parses the
 * borsh ComplianceInfo, synthesizes the exact compliance circuit witness
 * via the attached program, reads the circuit-computed public inputs back
 * (nf, cm, delta, RL commitments; anchor from the blob), proves on the
 * GPU. instance_out = 9 x 32B rows (CompliancePublicInputs::to_instance
 * order). */
int tg_compliance_prove(tg_ctx* ctx, const uint8_t* info_borsh, size_t len,
                        const uint8_t rng_seed[32], uint8_t* proof_out,
                        size_t cap, size_t* out_len, uint8_t instance_out[288]) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return TG_ERR_BADARG;
  if (!c->ppk || !c->ppk->ready || !c->ppk->tgw.ready) return TG_ERR_STATE;
  PDesc& d = c->ppk->d;
  if (d.n_instance_rows != 9 || c->ppk->tgw.n_inputs != 124) return TG_ERR_STATE;
  std::vector<Fp> inputs;
  if (!compliance_inputs(info_borsh, len, inputs)) return TG_ERR_ENCODING;
  std::vector<std::vector<Fp>> advice_lag;
  if (!c->ppk->tgw.run_mont(inputs, d.n_advice, advice_lag)) return TG_ERR_BADARG;
  std::vector<Fp> inst_rows(9, fd_zero<FpCfg>());
  inst_rows[1] = inputs[1];  // anchor from the blob
  c->ppk->tgw.read_instance_mont(advice_lag, inst_rows);
  std::vector<Fp> inst_lag(d.n, fd_zero<FpCfg>());
  for (int r = 0; r < 9; r++) inst_lag[r] = inst_rows[r];
  std::vector<uint8_t> proof;
  int rc = pprove_core(c, *c->ppk, inst_lag, advice_lag, rng_seed, proof);
  if (rc != 0) return rc;
  if (proof.size() > cap) return TG_ERR_BADARG;
  memcpy(proof_out, proof.data(), proof.size());
  *out_len = proof.size();
  if (instance_out)
    for (int r = 0; r < 9; r++) {
      Fp v = fd_from_mont(inst_rows[r]);
      memcpy(instance_out + 32 * r, v.l, 32);
    }
  return TG_OK;
}

/* drop-in TrivialRL prove (resource_logic_examples.rs get_verifying_info):
 * witness = borsh ResourceExistenceWitness; pad_rseed drives the random
 * instance padding (rows 6..21). instance_out = 22 x 32B. */
int tg_rl_prove(tg_ctx* ctx, const uint8_t* witness_borsh, size_t len,
                const uint8_t pad_rseed[32], const uint8_t rng_seed[32],
                uint8_t* proof_out, size_t cap, size_t* out_len,
                uint8_t instance_out[704]) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return TG_ERR_BADARG;
  if (!c->ppk || !c->ppk->ready || !c->ppk->tgw.ready) return TG_ERR_STATE;
  PDesc& d = c->ppk->d;
  if (d.n_instance_rows != 22 || c->ppk->tgw.n_inputs != 41) return TG_ERR_STATE;
  std::vector<Fp> inputs;
  uint8_t pad[16 * 32];
  if (!rl_inputs(witness_borsh, len, pad_rseed, inputs, pad))
    return TG_ERR_ENCODING;
  std::vector<std::vector<Fp>> advice_lag;
  if (!c->ppk->tgw.run_mont(inputs, d.n_advice, advice_lag)) return TG_ERR_BADARG;
  std::vector<Fp> inst_rows(22, fd_zero<FpCfg>());
  for (int r = 0; r < 16; r++) {
    Fp v;
    memcpy(v.l, pad + 32 * r, 32);
    inst_rows[6 + r] = fd_to_mont(v);
  }
  c->ppk->tgw.read_instance_mont(advice_lag, inst_rows);
  std::vector<Fp> inst_lag(d.n, fd_zero<FpCfg>());
  for (int r = 0; r < 22; r++) inst_lag[r] = inst_rows[r];
  std::vector<uint8_t> proof;
  int rc = pprove_core(c, *c->ppk, inst_lag, advice_lag, rng_seed, proof);
  if (rc != 0) return rc;
  if (proof.size() > cap) return TG_ERR_BADARG;
  memcpy(proof_out, proof.data(), proof.size());
  *out_len = proof.size();
  if (instance_out)
    for (int r = 0; r < 22; r++) {
      Fp v = fd_from_mont(inst_rows[r]);
      memcpy(instance_out + 32 * r, v.l, 32);
    }
  return TG_OK;
}

/* synthesized-advice export for parity tests: runs the attached program on
 * a borsh witness (kind 0 = compliance, 1 = RL) and writes the advice
 * column bytes (n_advice x 2^k x 32). */
int tg_witness_synthesize(tg_ctx* ctx, int kind, const uint8_t* borsh, size_t len,
                          const uint8_t pad_rseed[32], uint8_t* advice_out,
                          uint8_t* instance_out) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return TG_ERR_BADARG;
  if (!c->ppk || !c->ppk->ready || !c->ppk->tgw.ready) return TG_ERR_STATE;
  PDesc& d = c->ppk->d;
  std::vector<Fp> inputs;
  size_t ninst = (size_t)d.n_instance_rows * 32;
  std::vector<uint8_t> inst(ninst, 0);
  if (kind == 0) {
    if (!compliance_inputs(borsh, len, inputs)) return TG_ERR_ENCODING;
    Fp a = fd_from_mont(inputs[1]);
    memcpy(inst.data() + 32, a.l, 32);
  } else {
    if (ninst < 22 * 32) return TG_ERR_STATE;
    if (!rl_inputs(borsh, len, pad_rseed, inputs, inst.data() + 6 * 32))
      return TG_ERR_ENCODING;
  }
  memset(advice_out, 0, 32ul * d.n_advice * d.n);
  if (!c->ppk->tgw.run(inputs, d.n_advice, advice_out)) return TG_ERR_BADARG;
  c->ppk->tgw.read_instance(d.n_advice, advice_out, inst.data());
  if (instance_out) memcpy(instance_out, inst.data(), ninst);
  return TG_OK;
}

/* ---- ShieldedPartialTransaction::build / verify (shielded_ptx.rs:98-137,
 * BASELINE configs[3]) ----
 * Builds one borsh ShieldedPartialTransaction: n_compliance compliance
 * proofs (borsh ComplianceInfo units, 1528 B each; key slot
 * slot_compliance) + per input/output resource one TrivialRL proof (borsh
 * ResourceExistenceWitness units, 334 B each; slot_rl), binding_sig_r =
 * sum of the units' rcv (Some), empty hints. Per-proof randomness is
 * drawn from ONE ChaCha20 stream over rng_seed in build order:
 * compliance proof seeds first, then per RL (pad_rseed, proof seed) —
 * the determinized stand-in for the reference's single &mut rng
 * (shielded_ptx.rs:105-125; DESIGN.md randomness ledger). */
int tg_ptx_build(tg_ctx* ctx, int slot_compliance, int slot_rl,
                 uint32_t n_compliance, const uint8_t* compliance_units,
                 uint32_t n_in, uint32_t n_out, const uint8_t* rl_units,
                 const uint8_t rng_seed[32], uint8_t* ptx_out, size_t cap,
                 size_t* out_len) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return TG_ERR_BADARG;
  if (n_compliance > 64 || n_in + n_out > 128) return TG_ERR_BADARG;
  Drbg stream(rng_seed);
  std::vector<uint8_t> out;
  auto put_u32 = [&](uint32_t v) {
    out.push_back((uint8_t)v);
    out.push_back((uint8_t)(v >> 8));
    out.push_back((uint8_t)(v >> 16));
    out.push_back((uint8_t)(v >> 24));
  };
  // compliances
  put_u32(n_compliance);
  Fq rcv_sum = fd_zero<FqCfg>();
  int rc = tg_select_key(ctx, slot_compliance);
  if (rc) return rc;
  for (uint32_t i = 0; i < n_compliance; i++) {
    const uint8_t* unit = compliance_units + 1528ul * i;
    uint8_t seed[32];
    stream.bytes(seed, 32);
    uint8_t proof[1 << 14];
    size_t plen = 0;
    uint8_t inst[288];
    rc = tg_compliance_prove(ctx, unit, 1528, seed, proof, sizeof(proof), &plen,
                             inst);
    if (rc) return rc;
    // rcv from the unit's rseed (last 32 bytes)
    rcv_sum = fd_add(rcv_sum, prf_expand<FqCfg>(unit + 1528 - 32, 3));
    put_u32((uint32_t)plen);
    out.insert(out.end(), proof, proof + plen);
    // CompliancePublicInputs borsh (compliance.rs:82-93):
    // anchor ‖ nf ‖ cm ‖ delta(compressed) ‖ rlcm_in ‖ rlcm_out
    out.insert(out.end(), inst + 32, inst + 64);   // anchor (row 1)
    out.insert(out.end(), inst, inst + 32);        // nf (row 0)
    out.insert(out.end(), inst + 64, inst + 96);   // cm (row 2)
    {  // delta rows 3,4 -> compressed pallas point
      Fp x, y;
      memcpy(x.l, inst + 96, 32);
      memcpy(y.l, inst + 128, 32);
      uint8_t comp[32];
      if (fd_is_zero(x) && fd_is_zero(y)) {
        memset(comp, 0, 32);
      } else {
        memcpy(comp, x.l, 32);
        comp[31] |= (uint8_t)((y.l[0] & 1) << 7);
      }
      out.insert(out.end(), comp, comp + 32);
    }
    for (int half = 0; half < 2; half++) {  // rlcm halves -> 32B commitment
      uint8_t cmb[32];
      memcpy(cmb, inst + (160 + 64 * half), 16);
      memcpy(cmb + 16, inst + (192 + 64 * half), 16);
      out.insert(out.end(), cmb, cmb + 32);
    }
  }
  // RL sets (inputs then outputs)
  rc = tg_select_key(ctx, slot_rl);
  if (rc) return rc;
  PPk* rlk = c->ppk;
  std::vector<uint8_t> vk_bytes;
  for (const VestaAff& p : rlk->fixed_commits) {
    uint8_t b[32];
    Transcript::compress(b, p);
    vk_bytes.insert(vk_bytes.end(), b, b + 32);
  }
  for (const VestaAff& p : rlk->sigma_commits) {
    uint8_t b[32];
    Transcript::compress(b, p);
    vk_bytes.insert(vk_bytes.end(), b, b + 32);
  }
  for (int grp = 0; grp < 2; grp++) {
    uint32_t cnt = grp == 0 ? n_in : n_out;
    uint32_t base = grp == 0 ? 0 : n_in;
    put_u32(cnt);
    for (uint32_t i = 0; i < cnt; i++) {
      const uint8_t* unit = rl_units + 334ul * (base + i);
      uint8_t pad[32], seed[32];
      stream.bytes(pad, 32);
      stream.bytes(seed, 32);
      uint8_t proof[1 << 14];
      size_t plen = 0;
      uint8_t inst[704];
      rc = tg_rl_prove(ctx, unit, 334, pad, seed, proof, sizeof(proof), &plen,
                       inst);
      if (rc) return rc;
      out.insert(out.end(), vk_bytes.begin(), vk_bytes.end());
      put_u32((uint32_t)plen);
      out.insert(out.end(), proof, proof + plen);
      out.insert(out.end(), inst, inst + 704);
      put_u32(0);  // app_dynamic_resource_logic_verifying_info: empty vec
    }
  }
  out.push_back(1);  // Some(binding_sig_r)
  {
    Fq s = fd_from_mont(rcv_sum);
    uint8_t b[32];
    memcpy(b, s.l, 32);
    out.insert(out.end(), b, b + 32);
  }
  put_u32(0);  // hints: empty
  if (out.size() > cap) return TG_ERR_BADARG;
  memcpy(ptx_out, out.data(), out.size());
  *out_len = out.size();
  return TG_OK;
}

/* expand the 192-byte borsh CompliancePublicInputs block into the real
 * 9 instance rows [nf, anchor, cm, delta_x, delta_y, rlcm halves]
 * (compliance.rs to_instance order; closes the round-1 placeholder
 * mapping flagged in ADVICE.md item 3) */
static bool compliance_rows_expand(const uint8_t* inst192, uint8_t rows[288]) {
  memcpy(rows, inst192 + 32, 32);       // nf
  memcpy(rows + 32, inst192, 32);       // anchor
  memcpy(rows + 64, inst192 + 64, 32);  // cm
  bool all0 = true;
  for (int b = 0; b < 32; b++)
    if (inst192[96 + b]) all0 = false;
  if (all0) {
    memset(rows + 96, 0, 64);
  } else {
    PallasJac d;
    if (!pallas_decompress(d, inst192 + 96)) return false;
    PallasAff a = jac_to_aff(d);
    Fp x = fd_from_mont(a.x), y = fd_from_mont(a.y);
    memcpy(rows + 96, x.l, 32);
    memcpy(rows + 128, y.l, 32);
  }
  for (int half = 0; half < 2; half++) {
    memset(rows + 160 + 64 * half, 0, 64);
    memcpy(rows + 160 + 64 * half, inst192 + 128 + 32 * half, 16);
    memcpy(rows + 192 + 64 * half + 0, inst192 + 128 + 32 * half + 16, 16);
  }
  return true;
}

static std::vector<uint8_t> pk_vk_bytes(PPk* k) {
  std::vector<uint8_t> out;
  uint8_t b[32];
  for (const VestaAff& p : k->fixed_commits) {
    Transcript::compress(b, p);
    out.insert(out.end(), b, b + 32);
  }
  for (const VestaAff& p : k->sigma_commits) {
    Transcript::compress(b, p);
    out.insert(out.end(), b, b + 32);
  }
  return out;
}

/* verify one borsh ShieldedPartialTransaction end-to-end
 * (ShieldedPartialTransaction::execute, shielded_ptx.rs:232-240):
 * batch-verify ALL compliance proofs (slot_compliance) and ALL RL proofs
 * (slot_rl, vk bytes must match the slot's key), then the consistency
 * checks: every RL root equals the others (resource merkle root), input
 * RL self-ids == compliance nfs, output RL self-ids == compliance cms.
 * Returns TG_OK, -1 proof failure, -3xx consistency, -2xx structure. */
int tg_ptx_verify(tg_ctx* ctx, int slot_compliance, int slot_rl,
                  const uint8_t* ptx, size_t len) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return TG_ERR_BADARG;
  int rc = tg_select_key(ctx, slot_rl);
  if (rc) return rc;
  PPk* rlk = c->ppk;
  uint32_t vk_len = 32u * (uint32_t)(rlk->fixed_commits.size() +
                                     rlk->sigma_commits.size());
  TxCursor cur{ptx, len};
  uint32_t n_cvi;
  if (!cur.u32(n_cvi) || n_cvi > 64) return -201;
  std::vector<const uint8_t*> cproof;
  std::vector<size_t> cplen;
  std::vector<uint8_t> cinst;  // 9 rows x 32 per proof
  std::vector<uint8_t> nfs, cms;
  for (uint32_t i = 0; i < n_cvi; i++) {
    uint32_t plen;
    const uint8_t* p;
    if (!cur.u32(plen) || plen > (1u << 20) || !cur.take(p, plen)) return -202;
    cproof.push_back(p);
    cplen.push_back(plen);
    const uint8_t* inst;
    if (!cur.take(inst, 192)) return -203;
    uint8_t rows[288];
    if (!compliance_rows_expand(inst, rows)) return -204;
    cinst.insert(cinst.end(), rows, rows + 288);
    nfs.insert(nfs.end(), rows, rows + 32);
    cms.insert(cms.end(), rows + 64, rows + 96);
  }
  uint32_t n_in, n_out;
  std::vector<const uint8_t*> rproof;
  std::vector<size_t> rplen;
  std::vector<const uint8_t*> rinst;
  std::vector<uint8_t> self_ids, roots;
  for (int grp = 0; grp < 2; grp++) {
    uint32_t cnt;
    if (!cur.u32(cnt) || cnt > 128) return -205;
    (grp == 0 ? n_in : n_out) = cnt;
    for (uint32_t i = 0; i < cnt; i++) {
      const uint8_t* vkb;
      if (!cur.take(vkb, vk_len)) return -206;
      static thread_local std::vector<uint8_t> vk_ref;
      if (vk_ref.size() != vk_len) vk_ref = pk_vk_bytes(rlk);
      if (memcmp(vkb, vk_ref.data(), vk_len) != 0)
        return -207;  // unknown RL vk (only the TrivialRL key is loaded)
      uint32_t plen;
      const uint8_t* p;
      if (!cur.u32(plen) || plen > (1u << 20) || !cur.take(p, plen)) return -208;
      const uint8_t* inst;
      if (!cur.take(inst, 22 * 32)) return -209;
      rproof.push_back(p);
      rplen.push_back(plen);
      rinst.push_back(inst);
      roots.insert(roots.end(), inst, inst + 32);
      self_ids.insert(self_ids.end(), inst + 32, inst + 64);
      uint32_t ndyn;
      if (!cur.u32(ndyn) || ndyn != 0) return -210;  // dynamic RLs: not built
    }
  }
  const uint8_t* b;
  if (!cur.take(b, 1)) return -211;
  if (*b == 1 && !cur.take(b, 32)) return -212;
  uint32_t n_hints;
  if (!cur.u32(n_hints) || !cur.take(b, n_hints)) return -213;
  if (cur.left != 0) return -214;
  // consistency (shielded_ptx.rs:156-230): roots all equal; input self-ids
  // match nfs in order; output self-ids match cms
  for (uint32_t i = 1; i < n_in + n_out; i++)
    if (memcmp(roots.data(), roots.data() + 32ul * i, 32) != 0) return -301;
  if (n_in != n_cvi || n_out != n_cvi) return -302;
  for (uint32_t i = 0; i < n_in; i++)
    if (memcmp(self_ids.data() + 32ul * i, nfs.data() + 32ul * i, 32) != 0)
      return -303;
  for (uint32_t i = 0; i < n_out; i++)
    if (memcmp(self_ids.data() + 32ul * (n_in + i), cms.data() + 32ul * i, 32) != 0)
      return -304;
  // batch-verify RL proofs on the RL key
  {
    std::vector<uint8_t> insts;
    for (const uint8_t* ip : rinst) insts.insert(insts.end(), ip, ip + 22 * 32);
    std::vector<uint8_t> proofs;
    for (size_t i = 0; i < rproof.size(); i++)
      proofs.insert(proofs.end(), rproof[i], rproof[i] + rplen[i]);
    rc = tg_verify_batch_raw(ctx, rproof.size(), insts.data(), proofs.data(),
                             rplen.data());
    if (rc) return rc == TG_ERR_BADARG ? rc : -1;
  }
  // batch-verify compliance proofs on the compliance key
  rc = tg_select_key(ctx, slot_compliance);
  if (rc) return rc;
  {
    std::vector<uint8_t> proofs;
    for (size_t i = 0; i < cproof.size(); i++)
      proofs.insert(proofs.end(), cproof[i], cproof[i] + cplen[i]);
    rc = tg_verify_batch_raw(ctx, cproof.size(), cinst.data(), proofs.data(),
                             cplen.data());
    if (rc) return rc == TG_ERR_BADARG ? rc : -1;
  }
  return TG_OK;
}

int tg_create_proof_raw(tg_ctx* ctx, const uint8_t* instance, const uint8_t* advice,
                        const uint8_t rng_seed[32], uint8_t* proof_out, size_t cap,
                        size_t* out_len) {
  Ctx* c = (Ctx*)ctx;
  if (!c || tg_enter(c)) return TG_ERR_BADARG;
  if (!c->ppk || !c->ppk->ready) return TG_ERR_STATE;
  std::vector<uint8_t> proof;
  int rc = pprove_raw(c, *c->ppk, instance, advice, rng_seed, proof);
  if (rc != 0) return rc;
  if (proof.size() > cap) return TG_ERR_BADARG;
  memcpy(proof_out, proof.data(), proof.size());
  *out_len = proof.size();
  return TG_OK;
}

}  /* extern "C" */
