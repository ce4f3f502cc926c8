// tx_wire.hpp — borsh wire format of Taiga transactions.
// PRODUCT CODE (host side): the decode/verify direction a
// `verify_transaction` service needs, mirroring the reference's borsh
// layouts field for field:
//
//   Transaction            transaction.rs:24-33 (derived borsh)
//     = Vec<ShieldedPartialTransaction>   (u32 LE count + items)
//     ‖ Vec<TransparentPartialTransaction> (round 1: must be empty)
//     ‖ BindingSignature = 64 raw bytes   (binding_signature.rs:59-73)
//   ShieldedPartialTransaction   shielded_ptx.rs:272-320 (manual borsh)
//     = Vec<ComplianceVerifyingInfo>
//     ‖ Vec<ResourceLogicVerifyingInfoSet>   (inputs)
//     ‖ Vec<ResourceLogicVerifyingInfoSet>   (outputs)
//     ‖ u8 option tag (0|1) [+ 32B binding_sig_r scalar]
//     ‖ Vec<u8> hints
//   ComplianceVerifyingInfo      shielded_ptx.rs:47-50 (derived)
//     = Proof(Vec<u8>) (u32 len + bytes)
//     ‖ CompliancePublicInputs = 6 x 32B raw (compliance.rs:82-93:
//       anchor, nf, cm, delta(compressed point), rl_cm_in, rl_cm_out)
//   ResourceLogicVerifyingInfoSet  shielded_ptx.rs:57-60 (derived)
//     = ResourceLogicVerifyingInfo ‖ Vec<ResourceLogicVerifyingInfo>
//   ResourceLogicVerifyingInfo   resource_logic_circuit.rs:175-188
//     = vk.write() bytes (no length prefix; vk_len is a per-circuit
//       constant the caller supplies — for our TGD1 keys it is
//       32*(n_fixed + n_perm) compressed commitments, the restated
//       halo2-0.3 VerifyingKey::write layout)
//     ‖ Proof(Vec<u8>)
//     ‖ 22 x 32B public inputs (RESOURCE_LOGIC_CIRCUIT_PUBLIC_INPUT_NUM
//       = 6 mandatory + 2 custom + 14 encryption — constant.rs:68-75)
//
// The checker recomputes Transaction::digest (transaction.rs:116-158)
// from the parsed compliance instances (bundle-order nullifiers, then
// cms, then delta commitments, then anchors), aggregates the delta
// commitments into the binding verification key (transaction.rs:99-114)
// and verifies the RedDSA binding signature.
#pragma once

#include <vector>

#include "binding_sig.hpp"

namespace taiga {

struct TxCursor {
  const uint8_t* p;
  size_t left;
  bool take(const uint8_t*& out, size_t n) {
    if (left < n) return false;
    out = p;
    p += n;
    left -= n;
    return true;
  }
  bool u32(uint32_t& v) {
    const uint8_t* b;
    if (!take(b, 4)) return false;
    v = (uint32_t)b[0] | ((uint32_t)b[1] << 8) | ((uint32_t)b[2] << 16) |
        ((uint32_t)b[3] << 24);
    return true;
  }
};

constexpr uint32_t TX_RL_PUBLIC_INPUTS = 22;  // constant.rs:68-75
constexpr uint32_t TX_MAX_ITEMS = 4096;       // structural sanity bound

// parsed compliance instance streams (digest order)
struct TxDigestStreams {
  std::vector<uint8_t> nfs, cms, deltas, anchors;
  uint32_t n_sptx = 0, n_compliance = 0, n_rl = 0;
  // per-compliance-proof views into the tx buffer (for proof verification)
  std::vector<const uint8_t*> proof_ptr;
  std::vector<size_t> proof_len;
  std::vector<const uint8_t*> inst_ptr;  // 192B each
  const uint8_t* sig = nullptr;          // 64B binding signature
  // resource-logic proof views (round 2: RL proofs are verified too)
  std::vector<const uint8_t*> rl_vk_ptr, rl_proof_ptr, rl_inst_ptr;  // inst 22x32
  std::vector<size_t> rl_proof_len;
  std::vector<uint8_t> rl_is_input;  // 1 = inputs group
  // per-sptx boundaries: [compliance_begin, rl_begin] index pairs
  std::vector<uint32_t> sptx_comp_begin, sptx_rl_begin;
};

// one ResourceLogicVerifyingInfo; records views when st != nullptr
inline bool tx_parse_rl_info(TxCursor& c, uint32_t vk_len, TxDigestStreams* st,
                             uint8_t is_input) {
  const uint8_t* vk;
  if (!c.take(vk, vk_len)) return false;
  uint32_t plen;
  if (!c.u32(plen) || plen > (1u << 20)) return false;
  const uint8_t* proof;
  if (!c.take(proof, plen)) return false;
  const uint8_t* inst;
  if (!c.take(inst, 32 * TX_RL_PUBLIC_INPUTS)) return false;
  if (st) {
    st->rl_vk_ptr.push_back(vk);
    st->rl_proof_ptr.push_back(proof);
    st->rl_proof_len.push_back(plen);
    st->rl_inst_ptr.push_back(inst);
    st->rl_is_input.push_back(is_input);
  }
  return true;
}

inline bool tx_parse_rl_set(TxCursor& c, uint32_t vk_len, uint32_t& n_rl,
                            TxDigestStreams* st, uint8_t is_input) {
  if (!tx_parse_rl_info(c, vk_len, st, is_input)) return false;
  n_rl++;
  uint32_t n;
  if (!c.u32(n) || n > TX_MAX_ITEMS) return false;
  for (uint32_t i = 0; i < n; i++) {
    if (!tx_parse_rl_info(c, vk_len, st, is_input)) return false;
    n_rl++;
  }
  return true;
}

// full Transaction parse; fills the digest streams. Returns 0, or a
// negative structural error (-2xx, distinct from proof errors).
inline int tx_parse(const uint8_t* tx, size_t len, uint32_t vk_len,
                    TxDigestStreams& out) {
  TxCursor c{tx, len};
  uint32_t n_sptx;
  if (!c.u32(n_sptx) || n_sptx > TX_MAX_ITEMS) return -201;
  out.n_sptx = n_sptx;
  for (uint32_t s = 0; s < n_sptx; s++) {
    out.sptx_comp_begin.push_back(out.n_compliance);
    out.sptx_rl_begin.push_back(out.n_rl);
    uint32_t n_cvi;
    if (!c.u32(n_cvi) || n_cvi > TX_MAX_ITEMS) return -202;
    for (uint32_t i = 0; i < n_cvi; i++) {
      uint32_t plen;
      if (!c.u32(plen) || plen > (1u << 20)) return -203;
      const uint8_t* proof;
      if (!c.take(proof, plen)) return -204;
      const uint8_t* inst;
      if (!c.take(inst, 6 * 32)) return -205;
      out.proof_ptr.push_back(proof);
      out.proof_len.push_back(plen);
      out.inst_ptr.push_back(inst);
      // digest streams: anchor(0) nf(1) cm(2) delta(3) — compliance.rs order
      out.anchors.insert(out.anchors.end(), inst, inst + 32);
      out.nfs.insert(out.nfs.end(), inst + 32, inst + 64);
      out.cms.insert(out.cms.end(), inst + 64, inst + 96);
      out.deltas.insert(out.deltas.end(), inst + 96, inst + 128);
      out.n_compliance++;
    }
    uint32_t n_in, n_out;
    if (!c.u32(n_in) || n_in > TX_MAX_ITEMS) return -206;
    for (uint32_t i = 0; i < n_in; i++)
      if (!tx_parse_rl_set(c, vk_len, out.n_rl, &out, 1)) return -207;
    if (!c.u32(n_out) || n_out > TX_MAX_ITEMS) return -208;
    for (uint32_t i = 0; i < n_out; i++)
      if (!tx_parse_rl_set(c, vk_len, out.n_rl, &out, 0)) return -209;
    const uint8_t* b;
    if (!c.take(b, 1)) return -210;
    if (*b == 1) {
      if (!c.take(b, 32)) return -211;  // retained binding_sig_r (unfinalized)
    } else if (*b != 0) {
      return -212;
    }
    uint32_t n_hints;
    if (!c.u32(n_hints) || n_hints > (1u << 24)) return -213;
    if (!c.take(b, n_hints)) return -214;
  }
  uint32_t n_transparent;
  if (!c.u32(n_transparent)) return -215;
  if (n_transparent != 0) return -216;  // transparent ptx: round-2 layout
  if (!c.take(out.sig, 64)) return -217;
  if (c.left != 0) return -218;  // trailing bytes
  return 0;
}

// wire check: parse + recompute Transaction::digest + aggregate delta
// commitments -> binding vk + verify the binding signature.
inline int tx_check(const uint8_t* tx, size_t len, uint32_t vk_len,
                    uint32_t* n_sptx, uint32_t* n_proofs, TxDigestStreams* keep) {
  TxDigestStreams st;
  int rc = tx_parse(tx, len, vk_len, st);
  if (rc) return rc;
  if (n_sptx) *n_sptx = st.n_sptx;
  if (n_proofs) *n_proofs = st.n_compliance;
  uint8_t digest[32];
  bs_tx_digest(digest, st.nfs.data(), st.n_compliance, st.cms.data(),
               st.n_compliance, st.deltas.data(), st.n_compliance,
               st.anchors.data(), st.n_compliance);
  uint8_t vk[32];
  if (bs_vk_from_deltas(vk, st.deltas.data(), st.n_compliance)) return -220;
  if (bs_verify(vk, digest, 32, st.sig)) return -1;
  if (keep) *keep = std::move(st);
  return 0;
}

}  // namespace taiga
