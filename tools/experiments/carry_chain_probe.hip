// Compile-only probes for the Montgomery-multiply instruction stream
// (see profiles/r01_bucket_acc_hazard_analysis.txt for the findings):
//   hipcc --offload-arch=gfx950 -O3 -I../../taiga_amd/csrc -c carry_chain_probe.hip
// then llvm-objdump the gfx950 bundle and count s_nop wait states.
// k1/k2: current u128 CIOS (one / two independent muls)
// k3: __builtin_addcll lowering (dead end: more real instructions)
// k4/k5: fd28 lazy-carry radix-2^28 (32% fewer slots; measured EQUAL wall
//        time at full occupancy -> multiply is real-op bound, not hazard
//        bound; see fd28_bench.hip for the dynamic A/B)
#include <hip/hip_runtime.h>
typedef unsigned long long u64;
typedef unsigned __int128 u128;
struct Fd { u64 l[4]; };
__constant__ u64 MOD[4] = {1,2,3,4};
__constant__ u64 INV = 0x123456789abcdefULL;

// current style: u128 CIOS
__device__ Fd mul_u128(const Fd& a, const Fd& b) {
  u64 t[5] = {0,0,0,0,0};
  for (int i = 0; i < 4; i++) {
    u128 c = 0;
    for (int j = 0; j < 4; j++) {
      u128 p = (u128)a.l[j] * b.l[i] + t[j] + (u64)c;
      t[j] = (u64)p; c = p >> 64;
    }
    u64 t4 = t[4] + (u64)c;
    u64 m = t[0] * INV;
    c = ((u128)m * MOD[0] + t[0]) >> 64;
    for (int j = 1; j < 4; j++) {
      u128 p = (u128)m * MOD[j] + t[j] + (u64)c;
      t[j-1] = (u64)p; c = p >> 64;
    }
    t[3] = t4 + (u64)c;
    t[4] = 0;
  }
  Fd r; for (int i=0;i<4;i++) r.l[i]=t[i]; return r;
}

__global__ void k1(Fd* out, const Fd* in, int n) {
  int i = blockIdx.x*256+threadIdx.x;
  Fd a = in[2*i], b = in[2*i+1];
  out[i] = mul_u128(a, b);
}

// two independent muls in one kernel (can the scheduler interleave?)
__global__ void k2(Fd* out, const Fd* in, int n) {
  int i = blockIdx.x*256+threadIdx.x;
  Fd a = in[4*i], b = in[4*i+1], c = in[4*i+2], d = in[4*i+3];
  Fd r0 = mul_u128(a, b);
  Fd r1 = mul_u128(c, d);
  out[2*i] = r0; out[2*i+1] = r1;
}

// addcll variant of the carry chains
__device__ Fd mul_addc(const Fd& a, const Fd& b) {
  u64 t[5] = {0,0,0,0,0};
  for (int i = 0; i < 4; i++) {
    u64 lo[4], hi[4];
    for (int j = 0; j < 4; j++) {
      u128 p = (u128)a.l[j] * b.l[i];
      lo[j] = (u64)p; hi[j] = (u64)(p >> 64);
    }
    unsigned long long cc = 0;
    t[0] = __builtin_addcll(t[0], lo[0], 0, &cc);
    t[1] = __builtin_addcll(t[1], lo[1], cc, &cc);
    t[2] = __builtin_addcll(t[2], lo[2], cc, &cc);
    t[3] = __builtin_addcll(t[3], lo[3], cc, &cc);
    u64 t4 = t[4] + cc;
    cc = 0;
    t[1] = __builtin_addcll(t[1], hi[0], 0, &cc);
    t[2] = __builtin_addcll(t[2], hi[1], cc, &cc);
    t[3] = __builtin_addcll(t[3], hi[2], cc, &cc);
    t4 += hi[3] + cc;
    u64 m = t[0] * INV, mlo[4], mhi[4];
    for (int j = 0; j < 4; j++) {
      u128 p = (u128)m * MOD[j];
      mlo[j] = (u64)p; mhi[j] = (u64)(p >> 64);
    }
    cc = 0;
    (void)__builtin_addcll(t[0], mlo[0], 0, &cc);
    t[0] = __builtin_addcll(t[1], mlo[1], cc, &cc);
    t[1] = __builtin_addcll(t[2], mlo[2], cc, &cc);
    t[2] = __builtin_addcll(t[3], mlo[3], cc, &cc);
    t[3] = t4 + cc;
    cc = 0;
    t[0] = __builtin_addcll(t[0], mhi[0], 0, &cc);
    t[1] = __builtin_addcll(t[1], mhi[1], cc, &cc);
    t[2] = __builtin_addcll(t[2], mhi[2], cc, &cc);
    t[3] += mhi[3] + cc;
    t[4] = 0;
  }
  Fd r; for (int i=0;i<4;i++) r.l[i]=t[i]; return r;
}

__global__ void k3(Fd* out, const Fd* in, int n) {
  int i = blockIdx.x*256+threadIdx.x;
  Fd a = in[4*i], b = in[4*i+1], c = in[4*i+2], d = in[4*i+3];
  Fd r0 = mul_addc(a, b);
  Fd r1 = mul_addc(c, d);
  out[2*i] = r0; out[2*i+1] = r1;
}

// k4: the fd28 carry-chain-free Montgomery multiply (taiga_amd/csrc/
// fd28.hpp) on the same workload shape as k1 — objdump slot comparison.
#include "../../taiga_amd/csrc/fd28.hpp"
using namespace taiga;
__global__ void k4(Fd28<FpCfg>* out, const Fd28<FpCfg>* in, int n) {
  int i = blockIdx.x * 256 + threadIdx.x;
  Fd28<FpCfg> a = in[2 * i], b = in[2 * i + 1];
  out[i] = fd28_mul(a, b);
}

// k5: two independent fd28 muls (interleaving headroom without VCC)
__global__ void k5(Fd28<FpCfg>* out, const Fd28<FpCfg>* in, int n) {
  int i = blockIdx.x * 256 + threadIdx.x;
  Fd28<FpCfg> a = in[4 * i], b = in[4 * i + 1], c = in[4 * i + 2], d = in[4 * i + 3];
  Fd28<FpCfg> r0 = fd28_mul(a, b);
  Fd28<FpCfg> r1 = fd28_mul(c, d);
  out[2 * i] = r0;
  out[2 * i + 1] = r1;
}
