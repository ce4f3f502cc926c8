/* taiga_gpu.h — C ABI of the MI355X-native Halo2/Pasta proving backend
 * (libtaiga_gpu.so).
 *
 * This is the drop-in boundary of SURVEY.md §8(b): the entry points are what
 * a Rust-side binding for Taiga's proof layer would bind, replacing the
 * internals of:
 *   Proof::create / Proof::verify      — taiga_halo2/src/proof.rs:25-42,45-54
 *   Params::<vesta::Affine>::read      — taiga_halo2/src/constant.rs:128-139
 *   halo2_proofs::plonk::create_proof  — un-vendored heliaxdev/halo2 dep
 *                                        (call site proof.rs:33)
 *   best_multiexp / best_fft           — the MSM/NTT engine inside that dep
 *                                        (microbench entries below)
 * The Rust-side cbindgen/FFI stub a maintainer would add is shown in
 * INTEGRATION.md.
 *
 * Conventions (identical to the reference wire formats):
 *   - field elements: 32-byte little-endian canonical repr
 *     (pasta_curves to_repr(); values >= modulus are rejected)
 *   - affine points: 64-byte x||y canonical repr; identity = 64 zero bytes
 *   - compressed points: 32 bytes, sign of y (oddness) in bit 255
 *     (pasta_curves GroupEncoding; pinned against the bundled SRS)
 *   - SRS bytes: the exact params_15 file format
 *     u32(k) ‖ 2^k x 32B g ‖ 2^k x 32B g_lagrange ‖ 32B w ‖ 32B u
 *
 * All buffers are caller-owned. A tg_ctx is bound to one GPU and one HIP
 * stream; use one ctx per device (proofs shard embarrassingly across GPUs —
 * SURVEY.md §8e, no collective). Calls on one ctx are not thread-safe;
 * different ctxs are independent.
 *
 * Every function returns 0 on success or a negative TG_ERR code.
 * There is NO CPU fallback anywhere behind this ABI: without a working HIP
 * device every entry point fails loudly with TG_ERR_HIP.
 */
#ifndef TAIGA_GPU_H
#define TAIGA_GPU_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

enum {
  TG_OK = 0,
  TG_ERR_HIP = -1,       /* HIP runtime failure (incl. no device) */
  TG_ERR_BADARG = -2,    /* invalid argument / size */
  TG_ERR_ENCODING = -3,  /* non-canonical field repr or invalid point */
  TG_ERR_NOSRS = -4,     /* SRS not loaded */
  TG_ERR_NOMEM = -5,     /* device allocation failed */
  TG_ERR_STATE = -6,     /* missing resident inputs for a _resident call */
};

typedef struct tg_ctx tg_ctx;

/* ---- lifecycle ---- */
int tg_init(int device_ordinal, tg_ctx** out);
void tg_destroy(tg_ctx* ctx);
int tg_device_count(void);
/* last HIP error string for this ctx (valid until next call) */
const char* tg_error_string(const tg_ctx* ctx);

/* ---- SRS (replaces Params::read + commit bases residency;
 *      constant.rs:128-139) ----
 * Parses the params_15 byte format, uploads and decompresses both base sets
 * into HBM (they stay resident; g_lagrange never changes between proofs). */
int tg_load_srs(tg_ctx* ctx, const uint8_t* params_bytes, size_t len);
int tg_srs_k(const tg_ctx* ctx); /* k, or TG_ERR_NOSRS */

/* ---- variable-base MSM over Vesta (microbench entry) ----
 * base_set: 0 = custom bases (tg_bases_upload), 1 = SRS g,
 *           2 = SRS g_lagrange.
 * scalars: n x 32B canonical (vesta::Scalar = pallas::Base = Fp).
 * out_xy: 64B canonical affine result. */
int tg_bases_upload(tg_ctx* ctx, const uint8_t* points_xy, size_t n);
/* synthetic distinct bases generated on-device ([seed+i+1]G) for benches */
int tg_gen_bases(tg_ctx* ctx, size_t n, uint64_t seed);
/* read back the current base set as n x 64B canonical affine (test/debug:
 * lets parity tests diff device-generated bases against the oracle's) */
int tg_bases_download(tg_ctx* ctx, size_t n, uint8_t* out_xy);
int tg_msm_pallas(tg_ctx* ctx, const uint8_t* scalars, size_t n, int base_set,
                  uint8_t out_xy[64]);

/* resident-input variant for measurement: scalars staged ahead of the timed
 * region (bench.py's contract: inputs already in HBM when timing starts). */
int tg_scalars_upload(tg_ctx* ctx, const uint8_t* scalars, size_t n);
int tg_msm_resident(tg_ctx* ctx, size_t n, int base_set, uint8_t out_xy[64]);

/* ---- radix-2 NTT over Fp (microbench entry) ----
 * dir: 0 forward, 1 inverse (inverse includes the n^{-1} scaling).
 * coset: 0 plain domain (coset variants arrive with the prover pipeline).
 * poly: in-place 2^k x 32B canonical. */
int tg_ntt_fp(tg_ctx* ctx, int dir, uint32_t k, int coset, uint8_t* poly);

/* resident-input variant: upload once, run on device data, download. */
int tg_poly_upload(tg_ctx* ctx, const uint8_t* poly, uint32_t k);
int tg_ntt_resident(tg_ctx* ctx, int dir, uint32_t k, int coset);
int tg_poly_download(tg_ctx* ctx, uint8_t* poly, uint32_t k);

/* ---- proving (replaces plonk::create_proof behind Proof::create;
 *      proof.rs:25-42 / SURVEY §8a) ----
 * tg_keygen: parses a TGD1 circuit-description blob (tools/gen_cs1.py) and
 * builds the proving key on this ctx's GPU (fixed/sigma transforms and
 * commitments via the NTT/MSM kernels). Requires tg_load_srs first.
 * tg_create_proof: produces one proof. Round-1 inputs are deterministic
 * seeds: instance (public input), witness, and prover-randomness streams
 * (ChaCha20 — BASELINE.md convention; the reference draws from caller RNG,
 * proof.rs:30). Proof bytes are bit-identical to the CPU oracle on the same
 * seeds (tests/test_prover_parity.py). */
/* returns the slot id (>= 0) of the cached PK and makes it active; PKs are
 * cached per ctx (SURVEY §8f-1 — the reference's generic macro re-keygens
 * every proof, resource_logic_circuit.rs:578-580). */
int tg_keygen(tg_ctx* ctx, const uint8_t* desc, size_t desc_len);
int tg_select_key(tg_ctx* ctx, int slot);
int tg_create_proof(tg_ctx* ctx, const uint8_t inst_seed[32],
                    const uint8_t wit_seed[32], const uint8_t rng_seed[32],
                    uint8_t* proof_out, size_t cap, size_t* out_len);
/* ---- drop-in circuit proving (round 2: the EXACT compliance and
 * TrivialRL circuits; tools/circuit generates the TGD1 desc for tg_keygen
 * and the TGW1 witness-synthesis program attached here) ----
 * tg_witness_program_load: attach the TGW1 blob to the ACTIVE key.
 * tg_compliance_prove: replaces ComplianceInfo::build + Proof::create
 *   (compliance.rs:190-233, proof.rs:25-42): input = borsh ComplianceInfo
 *   (input_resource 202B ‖ merkle path 4+33x32 ‖ anchor 32 ‖
 *   output_resource 202B ‖ rseed 32); witness synthesized via the program;
 *   instance_out = the 9 public-input rows (to_instance order).
 * tg_rl_prove: TrivialResourceLogicCircuit::get_verifying_info
 *   (resource_logic_examples.rs:116-134): witness = borsh
 *   ResourceExistenceWitness (202 + 33x4); pad_rseed draws the 16 random
 *   padding rows; instance_out = 22 rows.
 * tg_witness_synthesize: advice export for parity tests (kind 0/1). */
int tg_witness_program_load(tg_ctx* ctx, const uint8_t* tgw, size_t len);
int tg_compliance_prove(tg_ctx* ctx, const uint8_t* info_borsh, size_t len,
                        const uint8_t rng_seed[32], uint8_t* proof_out,
                        size_t cap, size_t* out_len, uint8_t instance_out[288]);
int tg_rl_prove(tg_ctx* ctx, const uint8_t* witness_borsh, size_t len,
                const uint8_t pad_rseed[32], const uint8_t rng_seed[32],
                uint8_t* proof_out, size_t cap, size_t* out_len,
                uint8_t instance_out[704]);
int tg_witness_synthesize(tg_ctx* ctx, int kind, const uint8_t* borsh, size_t len,
                          const uint8_t pad_rseed[32], uint8_t* advice_out,
                          uint8_t* instance_out);
/* raw-witness variant: instance = n_instance_rows x 32B canonical reprs,
 * advice = n_advice x 2^k x 32B canonical column-major (rows beyond
 * usable are replaced by prover blinding). Same proof bytes as the seeded
 * path when fed the generated witness. */
int tg_create_proof_raw(tg_ctx* ctx, const uint8_t* instance, const uint8_t* advice,
                        const uint8_t rng_seed[32], uint8_t* proof_out, size_t cap,
                        size_t* out_len);
/* verify one proof (replaces Proof::verify / plonk::verify_proof with the
 * SingleVerifier strategy — proof.rs:45-54). Returns TG_OK iff valid. */
int tg_verify_proof(tg_ctx* ctx, const uint8_t inst_seed[32], const uint8_t* proof,
                    size_t proof_len);
/* raw-instance verification: instance = n_instance_rows x 32B canonical
 * reprs (the drop-in shape of plonk::verify_proof's instance slices —
 * proof.rs:45-54). */
int tg_verify_proof_raw(tg_ctx* ctx, const uint8_t* instance, const uint8_t* proof,
                        size_t proof_len);
/* batch verification (halo2's BatchVerifier accumulation for
 * a bundle's proofs — shielded_ptx.rs:137-153): all m final IPA checks
 * collapse into one randomly-weighted combined check sharing a single
 * g-sized GPU MSM. inst_seeds = m x 32B; proofs = concatenated proof bytes
 * with per-proof lengths in proof_lens. All proofs must use the active key.
 * TG_OK iff all valid; -1 if the combined check fails; -1xx on the first
 * malformed proof. */
int tg_verify_batch(tg_ctx* ctx, size_t m, const uint8_t* inst_seeds,
                    const uint8_t* proofs, const size_t* proof_lens);
/* raw-instance batch verification (the real ptx-bundle shape): instances =
 * m concatenated n_instance_rows x 32B blocks for the active key. */
int tg_verify_batch_raw(tg_ctx* ctx, size_t m, const uint8_t* instances,
                        const uint8_t* proofs, const size_t* proof_lens);
/* batched Poseidon P128Pow5T3 ConstantLength<L> hashing over Fp (GPU
 * witness synthesis — SURVEY §8f-2; replaces host halo2_gadgets poseidon
 * hashing, utils.rs:40-48 / prf_nf utils.rs:37). msgs = n x L x 32B
 * canonical reprs; out = n x 32B digests. One hash per GPU thread;
 * parameters derived by tools/gen_poseidon.py (Grain LFSR), pinned against
 * the oracle's independent derivation. */
int tg_poseidon_hash(tg_ctx* ctx, const uint8_t* msgs, size_t n, int L,
                     uint8_t* out);
/* blake2b-256 of the generated advice matrix (witness-spec cross-check) */
int tg_witness_hash(tg_ctx* ctx, const uint8_t inst_seed[32],
                    const uint8_t wit_seed[32], uint8_t out[32]);

/* ---- binding signatures + transaction digest (host-side wire layer,
 *      SURVEY §8f-4) ----
 * RedDSA over Pallas as instantiated by TaigaBinding
 * (binding_signature.rs:23-31): H* = BLAKE2b-512 with personalization
 * "Taiga_RedPallasH" wide-reduced to the Pallas scalar field; signature =
 * 32B compressed R ‖ 32B LE scalar S; sign nonce = H*(T ‖ vk ‖ msg) with
 * T = 80 ChaCha20-DRBG bytes from rng_seed. Ctx-free host calls. The
 * basepoint is the Pallas generator pending the sinsemilla group-hash
 * chain for RESOURCE_COMMIT_DOMAIN.R() (constant.rs:160).
 * Scalars are 32B LE canonical mod the Pallas group order. */
int tg_binding_vk(const uint8_t sk[32], uint8_t vk_out[32]);
int tg_delta_commit(const uint8_t r[32], uint8_t cv_out[32]);
int tg_binding_sign(const uint8_t sk[32], const uint8_t* msg, size_t msg_len,
                    const uint8_t rng_seed[32], uint8_t sig_out[64]);
/* TG_OK iff the signature verifies */
int tg_binding_verify(const uint8_t vk[32], const uint8_t* msg, size_t msg_len,
                      const uint8_t sig[64]);
/* binding vk = sum of a bundle's delta commitments (transaction.rs:99-114) */
int tg_binding_vk_from_deltas(const uint8_t* deltas, size_t n, uint8_t vk_out[32]);
/* Transaction::digest (transaction.rs:116-158): BLAKE2b-256
 * "TxBindingSigHash" over nullifiers ‖ output cms ‖ delta commitments ‖
 * anchors (32B each; append the transparent bundle's streams after the
 * shielded ones). */
int tg_tx_digest(const uint8_t* nfs, size_t n_nf, const uint8_t* cms, size_t n_cm,
                 const uint8_t* deltas, size_t n_delta, const uint8_t* anchors,
                 size_t n_anchor, uint8_t out[32]);

/* ---- transaction wire format (layout citations in
 *      taiga_amd/csrc/tx_wire.hpp) ----
 * tg_tx_wire_check: ctx-free — parse a borsh Transaction, recompute
 * Transaction::digest from the compliance instances, aggregate delta
 * commitments into the binding vk, verify the binding signature.
 * vk_len = byte length of one embedded RL VerifyingKey (32*(n_fixed +
 * n_perm) for TGD1 keys). TG_OK; -1 bad binding signature; -2xx
 * structural decode error.
 * tg_tx_verify: LEGACY/PARTIAL (round 1): verifies compliance proofs +
 * the binding signature only, with the round-1 first-n-rows instance
 * mapping (requires n_instance_rows <= 6); resource-logic proofs are
 * parsed but NOT verified — TG_OK from this entry does NOT mean the
 * transaction is fully valid.
 * tg_tx_verify_full: the COMPLETE check (round 2): Transaction::execute
 * semantics — wire parse + Transaction::digest + binding signature +
 * batch verification of EVERY compliance proof (real 9-row instance
 * expansion: delta decompressed, RL-commitment halves split —
 * compliance.rs to_instance) and EVERY resource-logic proof (vk bytes
 * must match slot_rl's key), plus the per-sptx nf/cm/self-id/root
 * consistency checks. */
int tg_tx_wire_check(const uint8_t* tx, size_t len, uint32_t vk_len,
                     uint32_t* n_sptx, uint32_t* n_proofs);
int tg_tx_verify(tg_ctx* ctx, const uint8_t* tx, size_t len);
int tg_tx_verify_full(tg_ctx* ctx, int slot_compliance, int slot_rl,
                      const uint8_t* tx, size_t len);
/* ShieldedPartialTransaction build/verify (shielded_ptx.rs:98-137;
 * BASELINE configs[3]): see taiga_gpu.cpp headers for unit layouts. */
int tg_ptx_build(tg_ctx* ctx, int slot_compliance, int slot_rl,
                 uint32_t n_compliance, const uint8_t* compliance_units,
                 uint32_t n_in, uint32_t n_out, const uint8_t* rl_units,
                 const uint8_t rng_seed[32], uint8_t* ptx_out, size_t cap,
                 size_t* out_len);
int tg_ptx_verify(tg_ctx* ctx, int slot_compliance, int slot_rl,
                  const uint8_t* ptx, size_t len);

/* ---- kernel profiling (HIP events on the ctx stream) ----
 * names: "msm_digits", "msm_scan", "msm_scatter", "msm_bucket_acc",
 *        "msm_reduce", "msm_wsum", "ntt_stage", "ntt_fused", "ntt_bitrev",
 *        "ntt_scale", "msm_total", "ntt_total" */
void tg_prof_enable(tg_ctx* ctx, int on);
void tg_prof_reset(tg_ctx* ctx);
int tg_prof_get(tg_ctx* ctx, const char* name, double* total_ms, long* count);

/* ---- sync ---- */
int tg_synchronize(tg_ctx* ctx);

#ifdef __cplusplus
}
#endif

#endif /* TAIGA_GPU_H */
