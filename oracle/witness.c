/* witness.c — TGW1 witness-synthesis program interpreter + borsh witness
 * input builders for the EXACT compliance / trivial-RL circuits.
 *
 * ORACLE TEST INFRASTRUCTURE (see fd.h header note). Restates, per the
 * reference: Circuit::synthesize witness computation (driven by the TGW1
 * program tools/circuit/emit.py generates from the restated circuits),
 * ComplianceInfo borsh layout (compliance.rs:51-59, resource.rs:296-328,
 * merkle_tree.rs MerklePath/LR), RandomSeed PRF expansions
 * (resource.rs:402-448) and the psi/rcm/npk/nf poseidon derivations
 * (resource.rs:217-293). The product implements the same independently in
 * taiga_amd/csrc/witness.hpp; parity is byte-compared in tests.
 */
#include "curve.h"
#include <stdlib.h>
#include <string.h>

#ifdef _OPENMP
#include <omp.h>
#endif

#define FP (&FD_P)
#define FQ (&FD_Q)

enum { W_LOADI, W_CONST, W_ADD, W_SUB, W_MUL, W_INV0, W_NEG, W_SQRT0, W_BIT, W_BYTE };

typedef struct {
    uint8_t op;
    uint32_t a, b;
} WOp;

typedef struct {
    uint32_t n_inputs, n_consts, n_ops, n_stores, n_expose, k;
    fd_limbs* consts; /* Mont */
    WOp* ops;
    uint32_t* stores;  /* col,row,reg triples */
    uint32_t* expose;  /* irow,col,row triples */
} TgwProg;

int orc_tgw_load(const uint8_t* blob, long len, void** out) {
    if (len < 28 || memcmp(blob, "TGW1", 4) != 0) return -1;
    const uint8_t* p = blob + 4;
    TgwProg* w = (TgwProg*)calloc(1, sizeof(TgwProg));
    uint32_t hdr[6];
    memcpy(hdr, p, 24);
    p += 24;
    w->n_inputs = hdr[0];
    w->n_consts = hdr[1];
    w->n_ops = hdr[2];
    w->n_stores = hdr[3];
    w->n_expose = hdr[4];
    w->k = hdr[5];
    long need = 28 + 32L * w->n_consts + 12L * w->n_ops + 12L * w->n_stores +
                12L * w->n_expose;
    if (len != need) { free(w); return -2; }
    w->consts = (fd_limbs*)malloc(sizeof(fd_limbs) * w->n_consts);
    for (uint32_t i = 0; i < w->n_consts; i++) {
        if (fd_from_bytes(w->consts[i], p, FP)) { free(w->consts); free(w); return -3; }
        p += 32;
    }
    w->ops = (WOp*)malloc(sizeof(WOp) * w->n_ops);
    for (uint32_t i = 0; i < w->n_ops; i++) {
        w->ops[i].op = p[0];
        memcpy(&w->ops[i].a, p + 4, 4);
        memcpy(&w->ops[i].b, p + 8, 4);
        p += 12;
    }
    w->stores = (uint32_t*)malloc(12L * w->n_stores);
    memcpy(w->stores, p, 12L * w->n_stores);
    p += 12L * w->n_stores;
    w->expose = (uint32_t*)malloc(12L * w->n_expose);
    memcpy(w->expose, p, 12L * w->n_expose);
    *out = w;
    return 0;
}

void orc_tgw_free(void* prog) {
    TgwProg* w = (TgwProg*)prog;
    if (!w) return;
    free(w->consts);
    free(w->ops);
    free(w->stores);
    free(w->expose);
    free(w);
}

/* Run the program. inputs: n_inputs x 32B canonical. advice: n_advice
 * column-major 2^k x 32B canonical buffers, caller-zeroed; regs buffer is
 * internal. Returns 0 or <0 on malformed input. */
int orc_tgw_run(const void* prog, const uint8_t* inputs, int n_advice,
                uint8_t* advice) {
    const TgwProg* w = (const TgwProg*)prog;
    long n = 1L << w->k;
    fd_limbs* in = (fd_limbs*)malloc(sizeof(fd_limbs) * w->n_inputs);
    for (uint32_t i = 0; i < w->n_inputs; i++)
        if (fd_from_bytes(in[i], inputs + 32L * i, FP)) { free(in); return -1; }
    fd_limbs* regs = (fd_limbs*)malloc(sizeof(fd_limbs) * w->n_ops);
    uint64_t std[4];
    for (uint32_t i = 0; i < w->n_ops; i++) {
        const WOp* o = &w->ops[i];
        switch (o->op) {
            case W_LOADI: fd_copy(regs[i], in[o->a]); break;
            case W_CONST: fd_copy(regs[i], w->consts[o->a]); break;
            case W_ADD: fd_add(regs[i], regs[o->a], regs[o->b], FP); break;
            case W_SUB: fd_sub(regs[i], regs[o->a], regs[o->b], FP); break;
            case W_MUL: fd_mul(regs[i], regs[o->a], regs[o->b], FP); break;
            case W_INV0: fd_inv(regs[i], regs[o->a], FP); break;
            case W_NEG: fd_neg(regs[i], regs[o->a], FP); break;
            case W_SQRT0:
                if (!fd_sqrt(regs[i], regs[o->a], FP)) fd_zero(regs[i]);
                break;
            case W_BIT: {
                fd_from_mont(std, regs[o->a], FP);
                uint64_t bit = (std[o->b >> 6] >> (o->b & 63)) & 1;
                uint64_t v[4] = {bit, 0, 0, 0};
                fd_to_mont(regs[i], v, FP);
                break;
            }
            case W_BYTE: {
                fd_from_mont(std, regs[o->a], FP);
                uint64_t byte = (std[o->b >> 3] >> (8 * (o->b & 7))) & 0xFF;
                uint64_t v[4] = {byte, 0, 0, 0};
                fd_to_mont(regs[i], v, FP);
                break;
            }
            default:
                free(in);
                free(regs);
                return -2;
        }
    }
    for (uint32_t s = 0; s < w->n_stores; s++) {
        uint32_t col = w->stores[3 * s], row = w->stores[3 * s + 1],
                 reg = w->stores[3 * s + 2];
        if ((int)col >= n_advice || row >= (uint32_t)n || reg >= w->n_ops) {
            free(in); free(regs); return -3;
        }
        fd_to_bytes(advice + (32L * n) * col + 32L * row, regs[reg], FP);
    }
    free(in);
    free(regs);
    return 0;
}

/* read the circuit-computed instance rows out of a synthesized advice
 * buffer (rows not exposed are left untouched in instance_out) */
int orc_tgw_instance(const void* prog, int n_advice, const uint8_t* advice,
                     uint8_t* instance_out) {
    const TgwProg* w = (const TgwProg*)prog;
    long n = 1L << w->k;
    for (uint32_t i = 0; i < w->n_expose; i++) {
        uint32_t irow = w->expose[3 * i], col = w->expose[3 * i + 1],
                 row = w->expose[3 * i + 2];
        if ((int)col >= n_advice || row >= (uint32_t)n) return -1;
        memcpy(instance_out + 32L * irow, advice + (32L * n) * col + 32L * row, 32);
    }
    return 0;
}

/* ---------------- borsh witness-input builders ---------------- */

extern int orc_poseidon_hash(const uint8_t* msg, int L, uint8_t out[32]);
extern void orc_blake2b_init(void* S, size_t outlen, const uint8_t* personal);
/* blake2b helpers come from blake2b.c via a simple one-shot wrapper */
extern void orc_blake2b(const uint8_t* in, long len, const uint8_t* personal,
                        size_t outlen, uint8_t* out);

static const uint8_t PRF_PERSONAL[16] = "Taiga_ExpandSeed";

/* wide-reduce 64 LE bytes into field fid; out canonical 32B */
static void wide_reduce(const uint8_t wide[64], const fd_ctx* f, uint8_t out[32]) {
    /* lo + hi * 2^256: Mont tricks — lo*R^-1... do simple: treat as two
     * 4-limb values: v = lo + hi*2^256 mod m = lo + hi*R mod m where
     * R = 2^256. hi*R mod m: to_mont(hi) IS hi*R. */
    uint64_t lo[4], hi[4];
    memcpy(lo, wide, 32);
    memcpy(hi, wide + 32, 32);
    fd_limbs mlo, mhi, r;
    /* reduce lo mod m first: to_mont then from_mont is identity mod m */
    fd_to_mont(mlo, lo, f);
    fd_from_mont(mlo, mlo, f);
    fd_to_mont(mlo, mlo, f); /* mlo = lo (Mont) */
    fd_to_mont(mhi, hi, f);  /* mhi = hi*R (Mont of hi) */
    /* hi*2^256 mod m = from_mont(to_mont(to_mont(hi))) ... to_mont(hi) is
     * hi*R mod m as a VALUE in Mont form it's hi*R*R... careful:
     * to_mont(x) stores x*R; interpreting stored limbs as a Mont value
     * means value = stored/R = x. We want VALUE hi*R: store hi*R*R =
     * to_mont applied twice. */
    fd_to_mont(mhi, mhi, f);
    fd_add(r, mlo, mhi, f);
    fd_to_bytes(out, r, f);
}

static void prf_expand(const uint8_t rseed[32], uint8_t tag, const fd_ctx* f,
                       uint8_t out[32]) {
    uint8_t buf[33];
    buf[0] = tag;
    memcpy(buf + 1, rseed, 32);
    uint8_t wide[64];
    orc_blake2b(buf, 33, PRF_PERSONAL, 64, wide);
    wide_reduce(wide, f, out);
}

/* Resource (202 B borsh; resource.rs:296-328) field offsets */
typedef struct {
    uint8_t logic[32], label[32], value[32], quantity8[8];
    uint8_t nk_is_key; /* borsh tag: 1 = PublicKey, 2 = Key */
    uint8_t nk[32], nonce[32];
    uint8_t is_ephemeral;
    uint8_t rseed[32];
} ResourceB;

static int parse_resource(const uint8_t* p, ResourceB* r) {
    memcpy(r->logic, p, 32);
    memcpy(r->label, p + 32, 32);
    memcpy(r->value, p + 64, 32);
    memcpy(r->quantity8, p + 96, 8);
    r->nk_is_key = p[104] == 2;
    memcpy(r->nk, p + 105, 32);
    memcpy(r->nonce, p + 137, 32);
    r->is_ephemeral = p[169] == 1;
    memcpy(r->rseed, p + 170, 32);
    return 0;
}

static void u64_to_canon(uint64_t v, uint8_t out[32]) {
    memset(out, 0, 32);
    memcpy(out, &v, 8);
}

static void res_psi_rcm(const ResourceB* r, uint8_t psi[32], uint8_t rcm[32]) {
    /* poseidon([PRF_PERSONALIZATION_TO_FIELD, tag, rseed, nonce]) */
    uint8_t msg[4 * 32];
    /* to_field_elements("Taiga_ExpandSeed")[0]: 16 bytes zero-padded */
    memset(msg, 0, 32);
    memcpy(msg, PRF_PERSONAL, 16);
    memset(msg + 32, 0, 32); /* tag 0 = PSI */
    memcpy(msg + 64, r->rseed, 32);
    memcpy(msg + 96, r->nonce, 32);
    orc_poseidon_hash(msg, 4, psi);
    msg[32] = 1; /* tag 1 = RCM */
    orc_poseidon_hash(msg, 4, rcm);
}

static void res_npk(const ResourceB* r, uint8_t npk[32]) {
    if (!r->nk_is_key) {
        memcpy(npk, r->nk, 32);
        return;
    }
    uint8_t msg[64];
    memcpy(msg, r->nk, 32);
    memset(msg + 32, 0, 32);
    orc_poseidon_hash(msg, 2, npk);
}

/* ComplianceInfo borsh (compliance.rs:51-59):
 *   input_resource(202) ‖ merkle_path(4 + 33*32) ‖ anchor(32) ‖
 *   output_resource(202) ‖ rseed(32)   = 1528 bytes.
 * Builds the 124-element witness-input vector of
 * tools/circuit/compliance.py (anchor into slot 1; the other instance
 * slots are zero — the program does not read them; the real instance is
 * read back from the synthesized advice via orc_tgw_instance). */
int orc_compliance_inputs(const uint8_t* borsh, long len, uint8_t* inputs_out) {
    const int DEPTH = 32;
    long need = 202 + 4 + 33L * DEPTH + 32 + 202 + 32;
    if (len != need) return -1;
    const uint8_t* p = borsh;
    ResourceB rin, rout;
    parse_resource(p, &rin);
    p += 202;
    uint32_t plen;
    memcpy(&plen, p, 4);
    p += 4;
    if (plen != DEPTH) return -2;
    const uint8_t* path = p;
    p += 33L * DEPTH;
    const uint8_t* anchor = p;
    p += 32;
    parse_resource(p, &rout);
    p += 202;
    const uint8_t* rseed = p;

    uint8_t* o = inputs_out;
    memset(o, 0, 124L * 32);
#define SLOT(i) (o + 32L * (i))
    memcpy(SLOT(1), anchor, 32);
    if (!rin.nk_is_key) return -3; /* input resource must hold the key */
    memcpy(SLOT(9), rin.nk, 32);
    memcpy(SLOT(10), rin.logic, 32);
    memcpy(SLOT(11), rin.label, 32);
    memcpy(SLOT(12), rin.value, 32);
    uint64_t q;
    memcpy(&q, rin.quantity8, 8);
    u64_to_canon(q, SLOT(13));
    memcpy(SLOT(14), rin.nonce, 32);
    memcpy(SLOT(15), rin.rseed, 32);
    res_psi_rcm(&rin, SLOT(16), SLOT(17));
    u64_to_canon(rin.is_ephemeral, SLOT(18));
    for (int i = 0; i < DEPTH; i++) {
        memcpy(SLOT(19 + i), path + 33L * i, 32);
        /* LR borsh enum: 0 = R, 1 = L; is_left == (tag == 1) */
        u64_to_canon(path[33L * i + 32] == 1, SLOT(51 + i));
    }
    res_npk(&rout, SLOT(83));
    memcpy(SLOT(84), rout.logic, 32);
    memcpy(SLOT(85), rout.label, 32);
    memcpy(SLOT(86), rout.value, 32);
    memcpy(&q, rout.quantity8, 8);
    u64_to_canon(q, SLOT(87));
    memcpy(SLOT(88), rout.rseed, 32);
    u64_to_canon(rout.is_ephemeral, SLOT(89));
    /* rcv = PRF_EXPAND_VCM_R(3) wide-reduced into Fq; slots 90..121 are
     * its repr BYTES as field values */
    uint8_t rcv[32];
    prf_expand(rseed, 3, FQ, rcv);
    for (int i = 0; i < 32; i++) u64_to_canon(rcv[i], SLOT(90 + i));
    prf_expand(rseed, 4, FP, SLOT(122));
    prf_expand(rseed, 5, FP, SLOT(123));
#undef SLOT
    return 0;
}

/* ResourceExistenceWitness borsh (resource_tree.rs:70-81):
 *   resource(202) ‖ 4 x (node(32) ‖ lr(1))   = 334 bytes.
 * Builds the 41-element vector of tools/circuit/trivial_rl.py; instance
 * rows 6..21 (random padding) are derived from pad_rseed
 * (ResourceLogicPublicInputs::get_public_input_padding ->
 * RandomSeed::get_random_padding, resource.rs:413-426) into padding_out
 * (16 x 32B). */
int orc_rl_inputs(const uint8_t* borsh, long len, const uint8_t pad_rseed[32],
                  uint8_t* inputs_out, uint8_t* padding_out) {
    const int DEPTH = 4;
    if (len != 202 + 33L * DEPTH) return -1;
    ResourceB res;
    parse_resource(borsh, &res);
    const uint8_t* path = borsh + 202;
    /* is_input = !path[0].is_left (resource_tree.rs:41-43) */
    int is_input = !(path[32] == 1);
    uint8_t* o = inputs_out;
    memset(o, 0, 41L * 32);
#define SLOT(i) (o + 32L * (i))
    u64_to_canon(is_input, SLOT(22));
    if (is_input) {
        if (!res.nk_is_key) return -3;
        memcpy(SLOT(23), res.nk, 32);
    } else {
        res_npk(&res, SLOT(23));
    }
    memcpy(SLOT(24), res.logic, 32);
    memcpy(SLOT(25), res.label, 32);
    memcpy(SLOT(26), res.value, 32);
    uint64_t q;
    memcpy(&q, res.quantity8, 8);
    u64_to_canon(q, SLOT(27));
    memcpy(SLOT(28), res.nonce, 32);
    memcpy(SLOT(29), res.rseed, 32);
    res_psi_rcm(&res, SLOT(30), SLOT(31));
    u64_to_canon(res.is_ephemeral, SLOT(32));
    for (int i = 0; i < DEPTH; i++) {
        memcpy(SLOT(33 + i), path + 33L * i, 32);
        u64_to_canon(path[33L * i + 32] == 1, SLOT(37 + i));
    }
#undef SLOT
    /* instance padding rows 6..21 */
    for (int i = 0; i < 16; i++) {
        uint8_t buf[34];
        buf[0] = 2; /* PRF_EXPAND_PUBLIC_INPUT_PADDING */
        buf[1] = (uint8_t)i;
        memcpy(buf + 2, pad_rseed, 32);
        uint8_t wide[64];
        orc_blake2b(buf, 34, PRF_PERSONAL, 64, wide);
        wide_reduce(wide, FP, padding_out + 32L * i);
    }
    return 0;
}
