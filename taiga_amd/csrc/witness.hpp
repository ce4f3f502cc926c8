// witness.hpp — TGW1 witness-synthesis program interpreter + borsh witness
// builders for the exact compliance / trivial-RL circuits (product side).
//
// Independent C++ implementation of the same contracts the CPU oracle
// restates in oracle/witness.c (GPU-vs-oracle advice parity is
// byte-compared in tests): the TGW1 program format
// (tools/circuit/emit.py), the ComplianceInfo / ResourceExistenceWitness
// borsh layouts (compliance.rs:51-59, resource_tree.rs:70-81,
// resource.rs:296-328, merkle_tree.rs LR) and the RandomSeed / psi / rcm /
// npk derivations (resource.rs:217-293,402-448).
#pragma once

#include <cstring>
#include <vector>

#include "pasta_device.hpp"
#include "host_crypto.hpp"

namespace taiga {

enum WOpcode : uint8_t {
  W_LOADI, W_CONST, W_ADD, W_SUB, W_MUL, W_INV0, W_NEG, W_SQRT0, W_BIT, W_BYTE
};

struct TgwOp {
  uint8_t op;
  uint32_t a, b;
};

struct TgwProgram {
  uint32_t n_inputs = 0, n_expose = 0, k = 0;
  std::vector<Fp> consts;  // Mont
  std::vector<TgwOp> ops;
  std::vector<uint32_t> stores;  // col,row,reg
  std::vector<uint32_t> expose;  // irow,col,row
  bool ready = false;

  bool parse(const uint8_t* blob, size_t len) {
    if (len < 28 || memcmp(blob, "TGW1", 4) != 0) return false;
    const uint8_t* p = blob + 4;
    uint32_t hdr[6];
    memcpy(hdr, p, 24);
    p += 24;
    n_inputs = hdr[0];
    uint32_t n_consts = hdr[1], n_ops = hdr[2], n_stores = hdr[3];
    n_expose = hdr[4];
    k = hdr[5];
    if (len != 28 + 32ul * n_consts + 12ul * n_ops + 12ul * n_stores +
                   12ul * n_expose)
      return false;
    consts.resize(n_consts);
    for (uint32_t i = 0; i < n_consts; i++) {
      Fp v;
      memcpy(v.l, p, 32);
      consts[i] = fd_to_mont(v);
      p += 32;
    }
    ops.resize(n_ops);
    for (uint32_t i = 0; i < n_ops; i++) {
      ops[i].op = p[0];
      memcpy(&ops[i].a, p + 4, 4);
      memcpy(&ops[i].b, p + 8, 4);
      p += 12;
    }
    stores.resize(3ul * n_stores);
    memcpy(stores.data(), p, 12ul * n_stores);
    p += 12ul * n_stores;
    expose.resize(3ul * n_expose);
    memcpy(expose.data(), p, 12ul * n_expose);
    ready = true;
    return true;
  }

  // inputs: n_inputs Mont values; advice: canonical-bytes column-major
  // buffer (n_advice x 2^k x 32), caller-zeroed.
  bool run(const std::vector<Fp>& inputs, int n_advice, uint8_t* advice) const {
    if (inputs.size() != n_inputs) return false;
    long n = 1L << k;
    std::vector<Fp> regs(ops.size());
    for (size_t i = 0; i < ops.size(); i++) {
      const TgwOp& o = ops[i];
      switch (o.op) {
        case W_LOADI: regs[i] = inputs[o.a]; break;
        case W_CONST: regs[i] = consts[o.a]; break;
        case W_ADD: regs[i] = fd_add(regs[o.a], regs[o.b]); break;
        case W_SUB: regs[i] = fd_sub(regs[o.a], regs[o.b]); break;
        case W_MUL: regs[i] = fd_mul(regs[o.a], regs[o.b]); break;
        case W_INV0: regs[i] = fd_inv(regs[o.a]); break;
        case W_NEG: regs[i] = fd_neg(regs[o.a]); break;
        case W_SQRT0:
          if (!fd_sqrt(regs[i], regs[o.a])) regs[i] = fd_zero<FpCfg>();
          break;
        case W_BIT: {
          Fp s = fd_from_mont(regs[o.a]);
          u64 bit = (s.l[o.b >> 6] >> (o.b & 63)) & 1;
          Fp v{{bit, 0, 0, 0}};
          regs[i] = fd_to_mont(v);
          break;
        }
        case W_BYTE: {
          Fp s = fd_from_mont(regs[o.a]);
          u64 byte = (s.l[o.b >> 3] >> (8 * (o.b & 7))) & 0xFF;
          Fp v{{byte, 0, 0, 0}};
          regs[i] = fd_to_mont(v);
          break;
        }
        default:
          return false;
      }
    }
    for (size_t s = 0; s < stores.size(); s += 3) {
      uint32_t col = stores[s], row = stores[s + 1], reg = stores[s + 2];
      if ((int)col >= n_advice || row >= (uint32_t)n || reg >= ops.size())
        return false;
      Fp v = fd_from_mont(regs[reg]);
      memcpy(advice + (32ul * n) * col + 32ul * row, v.l, 32);
    }
    return true;
  }

  // Montgomery-direct variant: synthesize straight into per-column Mont
  // vectors (the prover core's native witness form — skips the canonical
  // round-trip of run()); exposure values are returned Mont as well.
  bool run_mont(const std::vector<Fp>& inputs, int n_advice,
                std::vector<std::vector<Fp>>& advice_lag,
                std::vector<Fp>* regs_out = nullptr) const {
    if (inputs.size() != n_inputs) return false;
    long n = 1L << k;
    std::vector<Fp> regs(ops.size());
    for (size_t i = 0; i < ops.size(); i++) {
      const TgwOp& o = ops[i];
      switch (o.op) {
        case W_LOADI: regs[i] = inputs[o.a]; break;
        case W_CONST: regs[i] = consts[o.a]; break;
        case W_ADD: regs[i] = fd_add(regs[o.a], regs[o.b]); break;
        case W_SUB: regs[i] = fd_sub(regs[o.a], regs[o.b]); break;
        case W_MUL: regs[i] = fd_mul(regs[o.a], regs[o.b]); break;
        case W_INV0: regs[i] = fd_inv(regs[o.a]); break;
        case W_NEG: regs[i] = fd_neg(regs[o.a]); break;
        case W_SQRT0:
          if (!fd_sqrt(regs[i], regs[o.a])) regs[i] = fd_zero<FpCfg>();
          break;
        case W_BIT: {
          Fp s = fd_from_mont(regs[o.a]);
          u64 bit = (s.l[o.b >> 6] >> (o.b & 63)) & 1;
          Fp v{{bit, 0, 0, 0}};
          regs[i] = fd_to_mont(v);
          break;
        }
        case W_BYTE: {
          Fp s = fd_from_mont(regs[o.a]);
          u64 byte = (s.l[o.b >> 3] >> (8 * (o.b & 7))) & 0xFF;
          Fp v{{byte, 0, 0, 0}};
          regs[i] = fd_to_mont(v);
          break;
        }
        default:
          return false;
      }
    }
    advice_lag.assign(n_advice, std::vector<Fp>(n, fd_zero<FpCfg>()));
    for (size_t s2 = 0; s2 < stores.size(); s2 += 3) {
      uint32_t col = stores[s2], row = stores[s2 + 1], reg = stores[s2 + 2];
      if ((int)col >= n_advice || row >= (uint32_t)n || reg >= ops.size())
        return false;
      advice_lag[col][row] = regs[reg];
    }
    if (regs_out) *regs_out = std::move(regs);
    return true;
  }

  // instance rows (Mont) from synthesized Mont columns
  void read_instance_mont(const std::vector<std::vector<Fp>>& advice_lag,
                          std::vector<Fp>& inst_rows) const {
    long n = 1L << k;
    for (size_t i = 0; i < expose.size(); i += 3) {
      uint32_t irow = expose[i], col = expose[i + 1], row = expose[i + 2];
      if (col < advice_lag.size() && row < (uint32_t)n && irow < inst_rows.size())
        inst_rows[irow] = advice_lag[col][row];
    }
  }

  // read circuit-computed instance rows from synthesized advice bytes
  void read_instance(int n_advice, const uint8_t* advice,
                     uint8_t* instance_out) const {
    long n = 1L << k;
    for (size_t i = 0; i < expose.size(); i += 3) {
      uint32_t irow = expose[i], col = expose[i + 1], row = expose[i + 2];
      if ((int)col < n_advice && row < (uint32_t)n)
        memcpy(instance_out + 32ul * irow, advice + (32ul * n) * col + 32ul * row,
               32);
    }
  }
};

// ---------------- host Poseidon P128Pow5T3 (constants: poseidon_const.inc,
// Grain-derived, pinned vs the oracle's independent derivation) ----------

namespace hostpos {
#include "poseidon_const.inc"

inline void permute(Fp s[3]) {
  auto mds = [&](Fp x[3]) {
    Fp r[3];
    for (int i = 0; i < 3; i++) {
      r[i] = fd_zero<FpCfg>();
      for (int j = 0; j < 3; j++) {
        Fp m{{POS_MDS[i][j][0], POS_MDS[i][j][1], POS_MDS[i][j][2],
              POS_MDS[i][j][3]}};
        r[i] = fd_add(r[i], fd_mul(fd_to_mont(m), x[j]));
      }
    }
    for (int i = 0; i < 3; i++) x[i] = r[i];
  };
  auto pow5 = [](const Fp& x) {
    Fp x2 = fd_sqr(x);
    return fd_mul(fd_sqr(x2), x);
  };
  auto rc = [&](int r, int i) {
    Fp v{{POS_RC[r][i][0], POS_RC[r][i][1], POS_RC[r][i][2], POS_RC[r][i][3]}};
    return fd_to_mont(v);
  };
  int r = 0;
  for (int f = 0; f < 4; f++, r++) {
    for (int i = 0; i < 3; i++) s[i] = pow5(fd_add(s[i], rc(r, i)));
    mds(s);
  }
  for (int p = 0; p < 56; p++, r++) {
    for (int i = 0; i < 3; i++) s[i] = fd_add(s[i], rc(r, i));
    s[0] = pow5(s[0]);
    mds(s);
  }
  for (int f = 0; f < 4; f++, r++) {
    for (int i = 0; i < 3; i++) s[i] = pow5(fd_add(s[i], rc(r, i)));
    mds(s);
  }
}

inline Fp hash_n(const Fp* msg, int L) {
  Fp cap{{0, (u64)L, 0, 0}};  // L << 64
  Fp st[3] = {fd_zero<FpCfg>(), fd_zero<FpCfg>(), fd_to_mont(cap)};
  int padded = L + (L & 1);
  for (int c = 0; c < padded; c += 2) {
    st[0] = fd_add(st[0], c < L ? msg[c] : fd_zero<FpCfg>());
    st[1] = fd_add(st[1], c + 1 < L ? msg[c + 1] : fd_zero<FpCfg>());
    permute(st);
  }
  return st[0];
}
}  // namespace hostpos

// ---------------- borsh parsing + input building ----------------

struct ResourceB {
  Fp logic, label, value, nonce, rseed, nk;  // Mont
  u64 quantity = 0;
  bool nk_is_key = false, is_ephemeral = false;
};

inline bool parse_resource(const uint8_t* p, ResourceB& r) {
  auto rd = [&](const uint8_t* q, Fp& out) {
    Fp v;
    memcpy(v.l, q, 32);
    // canonicality: v < MOD
    for (int i = 3; i >= 0; i--) {
      if (v.l[i] < FpCfg::MOD[i]) break;
      if (v.l[i] > FpCfg::MOD[i] || i == 0) return false;
    }
    out = fd_to_mont(v);
    return true;
  };
  if (!rd(p, r.logic) || !rd(p + 32, r.label) || !rd(p + 64, r.value))
    return false;
  memcpy(&r.quantity, p + 96, 8);
  r.nk_is_key = p[104] == 2;
  if (!rd(p + 105, r.nk) || !rd(p + 137, r.nonce)) return false;
  r.is_ephemeral = p[169] == 1;
  return rd(p + 170, r.rseed);
}

inline Fp prf_personal_field() {
  // to_field_elements("Taiga_ExpandSeed")[0]
  Fp v = fd_zero<FpCfg>();
  uint8_t b[32] = {0};
  memcpy(b, "Taiga_ExpandSeed", 16);
  memcpy(v.l, b, 32);
  return fd_to_mont(v);
}

inline void res_psi_rcm(const ResourceB& r, Fp& psi, Fp& rcm) {
  Fp tag0 = fd_zero<FpCfg>();
  Fp one{{1, 0, 0, 0}};
  Fp msg[4] = {prf_personal_field(), tag0, r.rseed, r.nonce};
  psi = hostpos::hash_n(msg, 4);
  msg[1] = fd_to_mont(one);
  rcm = hostpos::hash_n(msg, 4);
}

inline Fp res_npk(const ResourceB& r) {
  if (!r.nk_is_key) return r.nk;
  Fp msg[2] = {r.nk, fd_zero<FpCfg>()};
  return hostpos::hash_n(msg, 2);
}

// RandomSeed::get_* expansions: blake2b-512 "Taiga_ExpandSeed" over
// [tag] || rseed, wide-reduced into Fp or Fq
template <class C>
inline Fd<C> prf_expand(const uint8_t rseed[32], uint8_t tag) {
  Blake2b h(64, (const uint8_t*)"Taiga_ExpandSeed");
  uint8_t buf[33];
  buf[0] = tag;
  memcpy(buf + 1, rseed, 32);
  h.update(buf, 33);
  uint8_t wide[64];
  h.final(wide);
  Fd<C> lo, hi;
  memcpy(lo.l, wide, 32);
  memcpy(hi.l, wide + 32, 32);
  // lo + hi*2^256 mod m
  Fd<C> mlo = fd_to_mont(lo);
  Fd<C> mhi = fd_to_mont(fd_to_mont(hi));
  return fd_add(mlo, mhi);
}

inline Fp u64_fp(u64 v) {
  Fp x{{v, 0, 0, 0}};
  return fd_to_mont(x);
}

// ComplianceInfo borsh -> the 124 witness-program inputs (Mont); layout in
// tools/circuit/compliance.py. Returns false on malformed blob.
inline bool compliance_inputs(const uint8_t* borsh, size_t len,
                              std::vector<Fp>& out) {
  const int DEPTH = 32;
  if (len != 202 + 4 + 33ul * DEPTH + 32 + 202 + 32) return false;
  const uint8_t* p = borsh;
  ResourceB rin, rout;
  if (!parse_resource(p, rin)) return false;
  p += 202;
  uint32_t plen;
  memcpy(&plen, p, 4);
  p += 4;
  if (plen != DEPTH) return false;
  const uint8_t* path = p;
  p += 33ul * DEPTH;
  Fp anchor;
  {
    Fp v;
    memcpy(v.l, p, 32);
    anchor = fd_to_mont(v);
  }
  p += 32;
  if (!parse_resource(p, rout)) return false;
  p += 202;
  const uint8_t* rseed = p;
  if (!rin.nk_is_key) return false;

  out.assign(124, fd_zero<FpCfg>());
  out[1] = anchor;
  out[9] = rin.nk;
  out[10] = rin.logic;
  out[11] = rin.label;
  out[12] = rin.value;
  out[13] = u64_fp(rin.quantity);
  out[14] = rin.nonce;
  out[15] = rin.rseed;
  res_psi_rcm(rin, out[16], out[17]);
  out[18] = u64_fp(rin.is_ephemeral ? 1 : 0);
  for (int i = 0; i < DEPTH; i++) {
    Fp v;
    memcpy(v.l, path + 33ul * i, 32);
    out[19 + i] = fd_to_mont(v);
    out[51 + i] = u64_fp(path[33ul * i + 32] == 1 ? 1 : 0);
  }
  out[83] = res_npk(rout);
  out[84] = rout.logic;
  out[85] = rout.label;
  out[86] = rout.value;
  out[87] = u64_fp(rout.quantity);
  out[88] = rout.rseed;
  out[89] = u64_fp(rout.is_ephemeral ? 1 : 0);
  Fq rcv = prf_expand<FqCfg>(rseed, 3);
  Fq rcv_std = fd_from_mont(rcv);
  uint8_t rb[32];
  memcpy(rb, rcv_std.l, 32);
  for (int i = 0; i < 32; i++) out[90 + i] = u64_fp(rb[i]);
  out[122] = prf_expand<FpCfg>(rseed, 4);
  out[123] = prf_expand<FpCfg>(rseed, 5);
  return true;
}

// ResourceExistenceWitness borsh -> the 41 RL inputs + the 16 instance
// padding rows (canonical bytes) from pad_rseed.
inline bool rl_inputs(const uint8_t* borsh, size_t len,
                      const uint8_t pad_rseed[32], std::vector<Fp>& out,
                      uint8_t padding_out[16 * 32]) {
  const int DEPTH = 4;
  if (len != 202 + 33ul * DEPTH) return false;
  ResourceB res;
  if (!parse_resource(borsh, res)) return false;
  const uint8_t* path = borsh + 202;
  bool is_input = !(path[32] == 1);
  out.assign(41, fd_zero<FpCfg>());
  out[22] = u64_fp(is_input ? 1 : 0);
  if (is_input) {
    if (!res.nk_is_key) return false;
    out[23] = res.nk;
  } else {
    out[23] = res_npk(res);
  }
  out[24] = res.logic;
  out[25] = res.label;
  out[26] = res.value;
  out[27] = u64_fp(res.quantity);
  out[28] = res.nonce;
  out[29] = res.rseed;
  res_psi_rcm(res, out[30], out[31]);
  out[32] = u64_fp(res.is_ephemeral ? 1 : 0);
  for (int i = 0; i < DEPTH; i++) {
    Fp v;
    memcpy(v.l, path + 33ul * i, 32);
    out[33 + i] = fd_to_mont(v);
    out[37 + i] = u64_fp(path[33ul * i + 32] == 1 ? 1 : 0);
  }
  for (int i = 0; i < 16; i++) {
    Blake2b h(64, (const uint8_t*)"Taiga_ExpandSeed");
    uint8_t buf[34];
    buf[0] = 2;  // PRF_EXPAND_PUBLIC_INPUT_PADDING
    buf[1] = (uint8_t)i;
    memcpy(buf + 2, pad_rseed, 32);
    h.update(buf, 34);
    uint8_t wide[64];
    h.final(wide);
    Fp lo, hi;
    memcpy(lo.l, wide, 32);
    memcpy(hi.l, wide + 32, 32);
    Fp v = fd_add(fd_to_mont(lo), fd_to_mont(fd_to_mont(hi)));
    Fp s = fd_from_mont(v);
    memcpy(padding_out + 32ul * i, s.l, 32);
  }
  return true;
}

}  // namespace taiga
