// hf_rtc.hpp — runtime-specialized quotient (h-fold) gate evaluation.
// PRODUCT CODE.
//
// The generic k_h_fold interpreter walks the desc's postfix programs per
// extended-domain row; with the real compliance circuit that is ~5k ops /
// 95 constraints per row, and the interpreter's runtime-indexed value
// cache lives in scratch — measured at ~24 ms per 2^19-row launch
// (gpurun_out/prof_r2), load/issue bound, not math bound.
//
// This module generates HIP C++ for the desc's EXACT gate set at keygen
// and compiles it with hipRTC for gfx950: every queried (column,rotation)
// value becomes a named register-resident local, the 95 constraint
// expressions become straight-line field arithmetic the compiler CSEs and
// schedules. Results are bit-identical to the interpreter (same exact
// integer field ops in the same fold order; scheduling cannot change
// exact arithmetic). The interpreter remains as the fallback when
// compilation is unavailable (TG_NO_RTC=1 forces it for A/B).
//
// One compilation per distinct desc per process (content-hash cache).
#pragma once

#include <hip/hiprtc.h>

#include <cstdio>
#include <algorithm>
#include <map>
#include <mutex>
#include <array>
#include <sstream>
#include <string>
#include <vector>

#include "pasta_device.hpp"
#include "prover_impl.hpp"

namespace taiga {

struct HfRtcKernel {
  hipModule_t mod = nullptr;
  hipFunction_t fn = nullptr;
  bool ready = false;
};

inline std::string hf_rtc_prelude() {
  std::ostringstream o;
  o << "typedef unsigned long long u64;\n"
       "typedef unsigned __int128 u128;\n"
       "struct Fp { u64 l[4]; };\n";
  auto arr = [&](const char* name, const u64 v[4]) {
    char buf[256];
    snprintf(buf, sizeof(buf),
             "__device__ __constant__ u64 %s[4] = {%lluULL,%lluULL,%lluULL,%lluULL};\n",
             name, (unsigned long long)v[0], (unsigned long long)v[1],
             (unsigned long long)v[2], (unsigned long long)v[3]);
    o << buf;
  };
  arr("MODV", FpCfg::MOD);
  char buf[128];
  snprintf(buf, sizeof(buf), "#define FPINV 0x%llxULL\n",
           (unsigned long long)FpCfg::INV);
  o << buf;
  o << R"RTC(
__device__ __forceinline__ void reduce_once(Fp& r) {
  u64 t[4]; u64 borrow = 0;
#pragma unroll
  for (int i = 0; i < 4; i++) {
    u128 d = (u128)r.l[i] - MODV[i] - borrow;
    t[i] = (u64)d; borrow = (u64)(d >> 64) ? 1 : 0;
  }
  if (!borrow) { r.l[0]=t[0]; r.l[1]=t[1]; r.l[2]=t[2]; r.l[3]=t[3]; }
}
__device__ __forceinline__ Fp f_add(Fp a, Fp b) {
  u64 carry = 0; Fp r;
#pragma unroll
  for (int i = 0; i < 4; i++) {
    u128 s = (u128)a.l[i] + b.l[i] + carry;
    r.l[i] = (u64)s; carry = (u64)(s >> 64);
  }
  reduce_once(r); return r;
}
__device__ __forceinline__ Fp f_sub(Fp a, Fp b) {
  u64 borrow = 0; Fp r;
#pragma unroll
  for (int i = 0; i < 4; i++) {
    u128 d = (u128)a.l[i] - b.l[i] - borrow;
    r.l[i] = (u64)d; borrow = (u64)(d >> 64) ? 1 : 0;
  }
  if (borrow) {
    u64 c = 0;
#pragma unroll
    for (int i = 0; i < 4; i++) {
      u128 s = (u128)r.l[i] + MODV[i] + c;
      r.l[i] = (u64)s; c = (u64)(s >> 64);
    }
  }
  return r;
}
__device__ __forceinline__ Fp f_neg(Fp a) {
  bool z = !(a.l[0] | a.l[1] | a.l[2] | a.l[3]);
  if (z) return a;
  u64 borrow = 0; Fp r;
#pragma unroll
  for (int i = 0; i < 4; i++) {
    u128 d = (u128)MODV[i] - a.l[i] - borrow;
    r.l[i] = (u64)d; borrow = (u64)(d >> 64) ? 1 : 0;
  }
  return r;
}
__device__ __forceinline__ Fp f_mul(Fp a, Fp b) {
  u64 t0=0,t1=0,t2=0,t3=0,t4=0,t5=0;
#pragma unroll
  for (int i = 0; i < 4; i++) {
    u64 bi = b.l[i]; u64 carry = 0; u128 s;
    s = (u128)a.l[0]*bi + t0 + carry; t0=(u64)s; carry=(u64)(s>>64);
    s = (u128)a.l[1]*bi + t1 + carry; t1=(u64)s; carry=(u64)(s>>64);
    s = (u128)a.l[2]*bi + t2 + carry; t2=(u64)s; carry=(u64)(s>>64);
    s = (u128)a.l[3]*bi + t3 + carry; t3=(u64)s; carry=(u64)(s>>64);
    s = (u128)t4 + carry; t4=(u64)s; t5=(u64)(s>>64);
    u64 m = t0 * FPINV;
    u128 c = (u128)m*MODV[0] + t0; carry=(u64)(c>>64);
    c = (u128)m*MODV[1] + t1 + carry; t0=(u64)c; carry=(u64)(c>>64);
    c = (u128)m*MODV[2] + t2 + carry; t1=(u64)c; carry=(u64)(c>>64);
    c = (u128)m*MODV[3] + t3 + carry; t2=(u64)c; carry=(u64)(c>>64);
    c = (u128)t4 + carry; t3=(u64)c;
    t4 = t5 + (u64)(c>>64); t5 = 0;
  }
  Fp r; r.l[0]=t0; r.l[1]=t1; r.l[2]=t2; r.l[3]=t3;
  reduce_once(r); return r;
}
)RTC";
  return o.str();
}

// generate the hf_gates kernel source for this desc's gate constraints.
// One __noinline__ __device__ function PER GATE (a single fused basic
// block of ~5k inlined Montgomery muls sent the optimizer into a
// half-hour register-allocation spiral; per-gate functions compile in
// seconds and still keep each gate's queries register-resident with
// full common-subexpression elimination inside the gate).
inline std::string hf_rtc_source(const PDesc& d) {
  std::ostringstream o;
  o << hf_rtc_prelude();
  o << "__device__ __constant__ Fp CONSTS[" << (d.consts.empty() ? 1 : d.consts.size())
    << "] = {";
  for (size_t i = 0; i < d.consts.size(); i++) {
    const Fp& c = d.consts[i];
    o << "{{" << c.l[0] << "ULL," << c.l[1] << "ULL," << c.l[2] << "ULL,"
      << c.l[3] << "ULL}},";
  }
  if (d.consts.empty()) o << "{{0,0,0,0}}";
  o << "};\n";

  auto colbase = [&](uint32_t tag) {
    return tag == XFIXED ? 0 : tag == XADVICE ? d.n_fixed : d.n_fixed + d.n_advice;
  };

  // constraint clusters (8 per device function): consecutive desc
  // constraints come from the same halo2 gate, so clustering recovers the
  // cross-constraint subexpression sharing (complete-add's 12 constraints
  // share most of their terms) while keeping each function small enough
  // for sane compile times
  const size_t GRP = 8;
  size_t n_groups = (d.gates.size() + GRP - 1) / GRP;
  for (size_t grp = 0; grp < n_groups; grp++) {
    size_t g0 = grp * GRP, g1 = std::min(d.gates.size(), g0 + GRP);
    o << "__device__ __noinline__ Fp g" << grp
      << "(Fp* const* cols, long i, long rs, long mask, Fp acc, Fp y) {\n";
    // distinct queries of this cluster
    std::vector<std::array<int, 3>> qs;
    for (size_t gi = g0; gi < g1; gi++)
    for (const auto& op : d.gates[gi].ops)
      if (op.tag == XFIXED || op.tag == XADVICE || op.tag == XINSTANCE) {
        std::array<int, 3> k{(int)op.tag, (int)op.a, op.b};
        bool seen = false;
        for (auto& q : qs)
          if (q == k) { seen = true; break; }
        if (!seen) qs.push_back(k);
      }
    auto qname = [&](uint32_t tag, uint32_t col, int rot) {
      std::ostringstream n;
      n << "q" << tag << "_" << col << "_" << (rot < 0 ? "m" : "p")
        << (rot < 0 ? -rot : rot);
      return n.str();
    };
    for (auto& q : qs) {
      o << "  const Fp " << qname(q[0], q[1], q[2]) << " = cols["
        << (colbase(q[0]) + q[1]) << "][(i + (long)(" << q[2] << ")*rs) & mask];\n";
    }
    // postfix -> temporaries with hash-consing (per cluster)
    std::map<std::string, std::string> cse;
    int tmp = 0;
    std::vector<std::string> stk;
    for (size_t gi = g0; gi < g1; gi++) {
    stk.clear();
    auto emit = [&](const std::string& expr) {
      auto it = cse.find(expr);
      if (it != cse.end()) return it->second;
      std::ostringstream tn;
      tn << "t" << tmp++;
      o << "  const Fp " << tn.str() << " = " << expr << ";\n";
      cse[expr] = tn.str();
      return cse[expr];
    };
    for (const auto& op : d.gates[gi].ops) {
      switch (op.tag) {
        case XCONST: {
          std::ostringstream e;
          e << "CONSTS[" << op.a << "]";
          stk.push_back(e.str());
          break;
        }
        case XFIXED:
        case XADVICE:
        case XINSTANCE:
          stk.push_back(qname(op.tag, op.a, op.b));
          break;
        case XADD:
        case XSUB:
        case XMUL: {
          std::string b = stk.back(); stk.pop_back();
          std::string a = stk.back(); stk.pop_back();
          const char* fn = op.tag == XADD ? "f_add" : op.tag == XSUB ? "f_sub" : "f_mul";
          stk.push_back(emit(std::string(fn) + "(" + a + "," + b + ")"));
          break;
        }
        case XNEG: {
          std::string a = stk.back(); stk.pop_back();
          stk.push_back(emit("f_neg(" + a + ")"));
          break;
        }
        case XSCALE: {
          std::string a = stk.back(); stk.pop_back();
          std::ostringstream e;
          e << "f_mul(" << a << ",CONSTS[" << op.a << "])";
          stk.push_back(emit(e.str()));
          break;
        }
        default:
          return std::string();
      }
    }
    o << "  acc = f_add(f_mul(acc, y), " << stk.back() << ");\n";
    }  // gi
    o << "  return acc;\n}\n";
  }

  o << "extern \"C\" __global__ void __launch_bounds__(256) hf_gates("
       "Fp* out, Fp* const* cols, Fp y, long ext_n, long rs) {\n"
       "  long mask = ext_n - 1;\n"
       "  for (long i = blockIdx.x*(long)blockDim.x + threadIdx.x; i < ext_n;"
       " i += (long)gridDim.x*blockDim.x) {\n"
       "    Fp acc = {{0,0,0,0}};\n";
  for (size_t grp = 0; grp < n_groups; grp++)
    o << "    acc = g" << grp << "(cols, i, rs, mask, acc, y);\n";
  o << "    out[i] = acc;\n  }\n}\n";
  return o.str();
}

// content-hash -> compiled kernel (process-lifetime cache: 8 contexts
// keygen the same two descs)
inline uint64_t hf_desc_hash(const PDesc& d) {
  uint64_t h = 1469598103934665603ULL;
  for (uint8_t b : d.blob) h = (h ^ b) * 1099511628211ULL;
  return h;
}

inline std::string hf_cache_path(uint64_t h) {
  const char* dir = getenv("TG_HF_CACHE");
  if (!dir) dir = "tests/golden";
  char buf[512];
  snprintf(buf, sizeof(buf), "%s/hf_%016llx.hsaco", dir, (unsigned long long)h);
  return buf;
}

// compile the desc's gate kernel; returns the code object (empty on
// failure). Pure hiprtc — usable without a GPU (the build container
// pre-compiles and ships the .hsaco files so GPU boxes never pay the
// ~1.5-3 min compile).
inline std::vector<char> hf_rtc_compile(const PDesc& d) {
  std::string src = hf_rtc_source(d);
  if (src.empty()) return {};
  hiprtcProgram prog;
  if (hiprtcCreateProgram(&prog, src.c_str(), "hf_gates.hip", 0, nullptr, nullptr) !=
      HIPRTC_SUCCESS)
    return {};
  const char* opts[] = {"--offload-arch=gfx950", "-O3", "-std=c++17"};
  hiprtcResult rc = hiprtcCompileProgram(prog, 3, opts);
  if (rc != HIPRTC_SUCCESS) {
    size_t lsz = 0;
    hiprtcGetProgramLogSize(prog, &lsz);
    std::string log(lsz, 0);
    hiprtcGetProgramLog(prog, log.data());
    fprintf(stderr, "taiga hf_rtc: compile failed, using interpreter:\n%.2000s\n",
            log.c_str());
    hiprtcDestroyProgram(&prog);
    return {};
  }
  size_t csz = 0;
  hiprtcGetCodeSize(prog, &csz);
  std::vector<char> code(csz);
  hiprtcGetCode(prog, code.data());
  hiprtcDestroyProgram(&prog);
  return code;
}

inline HfRtcKernel* hf_rtc_get(const PDesc& d) {
  static std::mutex mu;
  static std::map<uint64_t, HfRtcKernel> cache;
  if (getenv("TG_NO_RTC")) return nullptr;
  std::lock_guard<std::mutex> lk(mu);
  uint64_t h = hf_desc_hash(d);
  auto it = cache.find(h);
  if (it != cache.end()) return it->second.ready ? &it->second : nullptr;
  HfRtcKernel& k = cache[h];  // default: not ready (negative-cache failures)
  std::vector<char> code;
  // disk cache first (shipped by the build step)
  {
    FILE* f = fopen(hf_cache_path(h).c_str(), "rb");
    if (f) {
      fseek(f, 0, SEEK_END);
      long sz = ftell(f);
      fseek(f, 0, SEEK_SET);
      code.resize(sz);
      if (fread(code.data(), 1, sz, f) != (size_t)sz) code.clear();
      fclose(f);
    }
  }
  if (code.empty()) {
    code = hf_rtc_compile(d);
    if (code.empty()) return nullptr;
    FILE* f = fopen(hf_cache_path(h).c_str(), "wb");
    if (f) {
      fwrite(code.data(), 1, code.size(), f);
      fclose(f);
    }
  }
  if (hipModuleLoadData(&k.mod, code.data()) != hipSuccess) return nullptr;
  if (hipModuleGetFunction(&k.fn, k.mod, "hf_gates") != hipSuccess) return nullptr;
  k.ready = true;
  return &k;
}

}  // namespace taiga
