// fd28.hpp — EXPERIMENT (round-2 kernel candidate): carry-chain-free
// Montgomery multiplication in radix 2^28.
//
// Why: k_bucket_acc's instruction stream is ~38% s_nop hazard padding
// (profiles/r01_bucket_acc_hazard_analysis.txt) because gfx950 pads a wait
// state between a VALU write of VCC and the next VALU reading it, and the
// 4x64-limb CIOS multiply is one long VCC carry chain. Ten 28-bit digits
// held in u64 lanes absorb ~2^8 unpropagated product additions
// (products are <= 2^56, column accumulators stay < 2^61 over the 10 CIOS
// steps), so the whole multiply runs on v_mad_u64_u32 + plain 64-bit adds
// with NO carry flags; carries are propagated once at the end (~10 short
// serial ops instead of ~150 hazard-padded ones). Slot model: ~200 mads +
// ~80 glue ≈ 290 issue slots vs ~848 measured for the current fd_mul
// (tools/experiments/carry_chain_probe.hip k1). Cost: +25% register
// footprint per resident value (10 vs 8 VGPRs) — integrating into the
// bucket kernel without dropping below 4 waves/SIMD is the round-2 work.
//
// Semantics: mul28(a, b) = a * b * 2^-280 mod p for canonical 4x64 inputs
// (its own Montgomery domain, R28 = 2^280; domain-change constants are a
// per-field one-time precompute). Validated bit-exactly on the HOST
// against python bignum identities (tests/test_fd28.py) — the arithmetic
// is host/device-shared TG_HD code, so host validation covers the device
// semantics up to compiler bugs; on-GPU A/B is round-2.
#pragma once

#include "pasta_device.hpp"

namespace taiga {

constexpr int FD28_ND = 10;          // 10 x 28 = 280 bits >= 255
constexpr u64 FD28_MASK = (1ULL << 28) - 1;

template <class C>
struct Fd28 {
  u64 d[FD28_ND];  // digit i holds bits [28i, 28i+28); may carry excess
};

// extract 28-bit digits of a canonical (standard-form) 4x64 value
template <class C>
TG_HD Fd28<C> fd28_from(const Fd<C>& a) {
  Fd28<C> r;
#pragma unroll
  for (int i = 0; i < FD28_ND; i++) {
    int bit = 28 * i;
    int limb = bit >> 6, off = bit & 63;
    u64 v = a.l[limb] >> off;
    if (off > 36 && limb < 3) v |= a.l[limb + 1] << (64 - off);
    r.d[i] = v & FD28_MASK;
  }
  return r;
}

// p as 28-bit digits
template <class C>
TG_HD Fd28<C> fd28_mod() {
  Fd<C> m;
#pragma unroll
  for (int i = 0; i < 4; i++) m.l[i] = C::MOD[i];
  return fd28_from<C>(m);
}

// propagate carries and fully reduce to a canonical 4x64 value.
// Precondition: the represented value is < 2p (the CIOS output bound).
template <class C>
TG_HD Fd<C> fd28_norm(const Fd28<C>& a) {
  u64 t[FD28_ND];
  u64 carry = 0;
#pragma unroll
  for (int i = 0; i < FD28_ND; i++) {
    u64 v = a.d[i] + carry;
    t[i] = v & FD28_MASK;
    carry = v >> 28;
  }
  // pack into 4x64 (value < 2^280 with carry==0 guaranteed for < 2p)
  Fd<C> r{{0, 0, 0, 0}};
#pragma unroll
  for (int i = 0; i < FD28_ND; i++) {
    int bit = 28 * i;
    int limb = bit >> 6, off = bit & 63;
    r.l[limb] |= t[i] << off;
    if (off > 36 && limb < 3) r.l[limb + 1] |= t[i] >> (64 - off);
  }
  // value may exceed p (but < 2p): one conditional subtract
  bool ge = true;  // r >= MOD ?
  bool decided = false;
#pragma unroll
  for (int limb = 3; limb >= 0; limb--) {
    if (!decided) {
      if (r.l[limb] > C::MOD[limb]) decided = true;
      else if (r.l[limb] < C::MOD[limb]) { ge = false; decided = true; }
    }
  }
  if (ge) {
    u64 borrow = 0;
#pragma unroll
    for (int i = 0; i < 4; i++) {
      u64 mi = C::MOD[i];
      u64 v = r.l[i] - mi - borrow;
      borrow = (r.l[i] < mi + borrow || (mi + borrow < mi)) ? 1 : 0;
      r.l[i] = v;
    }
  }
  return r;
}

// CIOS Montgomery multiply in radix 2^28 with LAZY column carries:
// result value = a * b * 2^-280 mod p (< 2p before norm; fd28_norm
// finishes the reduction). No carry chains: every addition targets a
// 64-bit accumulator that cannot overflow (columns stay < 2^61).
template <class C>
TG_HD Fd28<C> fd28_mul(const Fd28<C>& a, const Fd28<C>& b) {
  const Fd28<C> p = fd28_mod<C>();
  const u64 inv28 = C::INV & FD28_MASK;  // -p^-1 mod 2^28 (mod-2^64 inverse truncates)
  u64 t[FD28_ND + 1];
#pragma unroll
  for (int i = 0; i <= FD28_ND; i++) t[i] = 0;
#pragma unroll
  for (int i = 0; i < FD28_ND; i++) {
    u64 bi = b.d[i];
#pragma unroll
    for (int j = 0; j < FD28_ND; j++) t[j] += a.d[j] * bi;  // <= 2^56 each
    u64 m = ((t[0] & FD28_MASK) * inv28) & FD28_MASK;
#pragma unroll
    for (int j = 0; j < FD28_ND; j++) t[j] += m * p.d[j];
    // t[0] now has zero low 28 bits (exact: t[0]'s low bits are carry-free)
    u64 sh = t[0] >> 28;
#pragma unroll
    for (int j = 0; j < FD28_ND; j++) t[j] = t[j + 1];
    t[0] += sh;
    t[FD28_ND] = 0;
  }
  Fd28<C> r;
#pragma unroll
  for (int i = 0; i < FD28_ND; i++) r.d[i] = t[i];
  return r;
}

}  // namespace taiga
