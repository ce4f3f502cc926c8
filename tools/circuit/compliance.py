"""The EXACT Taiga Compliance (Action) circuit, restated.

configure mirrors compliance_circuit.rs:77-172 (column/chip allocation
order identical); synthesize mirrors compliance_circuit.rs:174-327 +
integrity.rs (check_input_resource :51-190, check_output_resource
:193-325, compute_delta_commitment :546-630, quantity_range_check
:632-651). Witnesses are traced into a TGW1 program; the constraint
system is emitted as a TGD2 desc (tools/circuit/emit.py).

Witness-program input layout (the C/C++ interpreters build the same
vector from the borsh ComplianceInfo + host crypto — see
oracle/witness.c / taiga_amd/csrc/witness.hpp):
   0..8    instance rows [nf, anchor, cm, delta_x, delta_y,
                          rlcm_in_1, rlcm_in_2, rlcm_out_1, rlcm_out_2]
   9       input.nk            10 input.logic       11 input.label
   12      input.value         13 input.quantity    14 input.nonce
   15      input.rseed         16 input.psi (host)  17 input.rcm (host)
   18      input.is_ephemeral
   19..50  merkle path node values (32)
   51..82  merkle path lr bits (1 = sibling is left)
   83      output.npk (host)   84 output.logic      85 output.label
   86      output.value        87 output.quantity   88 output.rseed
   89      output.is_ephemeral
   90..121 rcv scalar repr bytes (32, little-endian)
   122     input_resource_logic_cm_r
   123     output_resource_logic_cm_r
"""
from .plonkish import (ConstraintSystem, Prog, assign_free_advice,
                       assign_free_constant)
from .chips.pow5 import Pow5Config, poseidon_hash_gadget
from .chips.lookup_range import LookupRangeCheckConfig
from .chips.gadgets import (CondSwapConfig, ResourceCommitConfig,
                            merkle_poseidon_gadget, quantity_range_check)
from .chips.blake2s import Blake2sConfig, Blake2sChip, resource_logic_commitment_gadget
from .chips.ecc import EccConfig
from .chips.swu import HashToCurveConfig
from . import hostcrypto as hc
from . import fields as F

K = 15
N_INPUTS = 124
TREE_DEPTH = 32

# instance row indices (constant.rs:54-62)
NF_ROW, ANCHOR_ROW, CM_ROW, DX_ROW, DY_ROW = 0, 1, 2, 3, 4
RLIN1, RLIN2, RLOUT1, RLOUT2 = 5, 6, 7, 8


class ComplianceModel:
    def __init__(self):
        cs = ConstraintSystem(K, "compliance")
        self.cs = cs
        self.instances = cs.instance_column()
        cs.enable_equality(self.instances)
        self.advices = [cs.advice_column() for _ in range(10)]
        for a in self.advices:
            cs.enable_equality(a)
        self.table_idx = cs.lookup_table_column()
        self.range_check = LookupRangeCheckConfig(cs, self.advices[9], self.table_idx)
        self.lagrange = [cs.fixed_column() for _ in range(8)]
        cs.enable_constant(self.lagrange[0])
        self.ecc = EccConfig(cs, self.advices, self.lagrange, self.range_check)
        self.poseidon = Pow5Config(cs, self.advices[6:9], self.advices[5],
                                   self.lagrange[2:5], self.lagrange[5:8])
        self.merkle_path_selector = cs.selector()
        cs.create_gate("merkle path check", self.merkle_path_selector, [
            ("is_ephemeral or root = anchor",
             (1 - self.advices[0].cur()) * (self.advices[2].cur() - self.advices[1].cur())),
        ])
        self.cond_swap = CondSwapConfig(cs, self.advices[:5])
        self.h2c = HashToCurveConfig(cs, self.advices, self.poseidon)
        self.blake2s = Blake2sConfig(cs, self.advices)
        self.resource_commit = ResourceCommitConfig(
            cs, self.advices[0:3], self.poseidon, self.range_check)

    def synthesize(self, inputs_int):
        """Run the traced synthesis with the given concrete inputs."""
        cs = self.cs
        prog = Prog(N_INPUTS)
        prog.input_vals = [v % F.P for v in inputs_int]
        inst_v = [prog.load_input(i) for i in range(9)]
        cs.start_synth(prog, inst_v)
        inp = lambda i: prog.load_input(i)
        adv = self.advices

        # lookup table
        cs.assign_table(self.table_idx, list(range(1 << 10)))

        # ---- check_input_resource (integrity.rs:51-190) ----
        nk = assign_free_advice(cs, adv[0], inp(9))
        zero_c = assign_free_constant(cs, adv[0], 0)
        npk_in = poseidon_hash_gadget(self.poseidon, [nk, zero_c])
        value_in = assign_free_advice(cs, adv[0], inp(12))
        logic_in = assign_free_advice(cs, adv[0], inp(10))
        label_in = assign_free_advice(cs, adv[0], inp(11))
        quantity_in = quantity_range_check(self.range_check, inp(13))
        nonce_in = assign_free_advice(cs, adv[0], inp(14))
        assign_free_advice(cs, adv[0], inp(15))  # rseed (witnessed, unused)
        psi_in = assign_free_advice(cs, adv[0], inp(16))
        rcm_in = assign_free_advice(cs, adv[0], inp(17))
        is_eph_in = assign_free_advice(cs, adv[0], inp(18))
        cm_in = self.resource_commit.resource_commit(
            logic_in, label_in, value_in, npk_in, nonce_in, psi_in,
            quantity_in, is_eph_in, rcm_in)
        nf = poseidon_hash_gadget(self.poseidon, [nk, nonce_in, psi_in, cm_in])
        cs.constrain_instance(nf, self.instances, NF_ROW)

        # ---- merkle root ----
        path = [(inp(19 + i), inp(51 + i)) for i in range(TREE_DEPTH)]
        root = merkle_poseidon_gadget(self.cond_swap, self.poseidon, cm_in, path)

        # ---- check_output_resource (integrity.rs:193-325) ----
        npk_out = assign_free_advice(cs, adv[0], inp(83))
        value_out = assign_free_advice(cs, adv[0], inp(86))
        logic_out = assign_free_advice(cs, adv[0], inp(84))
        label_out = assign_free_advice(cs, adv[0], inp(85))
        quantity_out = quantity_range_check(self.range_check, inp(87))
        rseed_out = assign_free_advice(cs, adv[0], inp(88))
        pers = assign_free_constant(cs, adv[0], hc.PRF_EXPAND_PERSONALIZATION_TO_FIELD)
        prf_rcm = assign_free_constant(cs, adv[0], hc.PRF_EXPAND_RCM)
        rcm_out = poseidon_hash_gadget(self.poseidon, [pers, prf_rcm, rseed_out, nf])
        prf_psi = assign_free_constant(cs, adv[0], hc.PRF_EXPAND_PSI)
        psi_out = poseidon_hash_gadget(self.poseidon, [pers, prf_psi, rseed_out, nf])
        is_eph_out = assign_free_advice(cs, adv[0], inp(89))
        cm_out = self.resource_commit.resource_commit(
            logic_out, label_out, value_out, npk_out, nf, psi_out,
            quantity_out, is_eph_out, rcm_out)
        cs.constrain_instance(cm_out, self.instances, CM_ROW)

        # ---- delta commitment (integrity.rs:546-630) ----
        def derive_kind(logic, label):
            pt = self.h2c.hash_to_curve(self.ecc, [logic, label])
            nid = self.ecc.witness_point(pt.x.reg, pt.y.reg, non_identity=True)
            cs.copies.append(((pt.x.col, pt.x.row), (nid.x.col, nid.x.row)))
            cs.copies.append(((pt.y.col, pt.y.row), (nid.y.col, nid.y.row)))
            return nid

        kind_in = derive_kind(logic_in, label_in)
        v_in = self.ecc.mul_var_base(quantity_in, kind_in)
        kind_out = derive_kind(logic_out, label_out)
        v_out = self.ecc.mul_var_base(quantity_out, kind_out)
        neg_v_out = self.ecc.witness_point(v_out.x.reg, 0 - v_out.y.reg,
                                           non_identity=False)
        cs.copies.append(((v_out.x.col, v_out.x.row), (neg_v_out.x.col, neg_v_out.x.row)))
        zero_pt = self.ecc.add(v_out, neg_v_out)
        with cs.region("constrain zero point") as r:
            r.constrain_constant(zero_pt.x, 0)
            r.constrain_constant(zero_pt.y, 0)
        commitment_v = self.ecc.add(v_in, neg_v_out)
        rcv_bytes = [inp(90 + i) for i in range(32)]
        R = hc.sinsemilla_commit_domain_r("Taiga-NoteCommit")
        blind = self.ecc.mul_fixed_full(rcv_bytes, "resource_commit_r", R)
        delta = self.ecc.add(commitment_v, blind)
        cs.constrain_instance(delta.x, self.instances, DX_ROW)
        cs.constrain_instance(delta.y, self.instances, DY_ROW)

        # ---- merkle path check (compliance_circuit.rs:261-278) ----
        with cs.region("merkle path check") as r:
            self.merkle_path_selector.enable(r, 0)
            r.copy_advice(is_eph_in, adv[0], 0)
            r.assign_advice_from_instance(self.instances, ANCHOR_ROW, adv[1], 0)
            r.copy_advice(root, adv[2], 0)

        # ---- resource logic commitments (blake2s) ----
        chip = Blake2sChip(self.blake2s)
        r_in = assign_free_advice(cs, adv[0], inp(122))
        cm1 = resource_logic_commitment_gadget(chip, logic_in, r_in)
        cs.constrain_instance(cm1[0], self.instances, RLIN1)
        cs.constrain_instance(cm1[1], self.instances, RLIN2)
        r_out = assign_free_advice(cs, adv[0], inp(123))
        cm2 = resource_logic_commitment_gadget(chip, logic_out, r_out)
        cs.constrain_instance(cm2[0], self.instances, RLOUT1)
        cs.constrain_instance(cm2[1], self.instances, RLOUT2)
        return cs


def build_inputs(input_res: hc.Resource, merkle_path, anchor,
                 output_res: hc.Resource, rcv_repr32: bytes,
                 rlcm_r_in: int, rlcm_r_out: int):
    """Host-side mirror of ComplianceInfo::build (compliance.rs:190-233):
    computes the public inputs + the witness-program input vector."""
    import pypasta as pp
    nf = input_res.get_nf()
    assert output_res.nonce == nf, "output nonce must be input nf"
    cm_out = output_res.commitment()
    # delta = [q_in]K_in - [q_out]K_out + [rcv]R   (delta_commitment.rs)
    k_in = hc.poseidon_to_curve([input_res.logic, input_res.label])
    k_out = hc.poseidon_to_curve([output_res.logic, output_res.label])
    R = hc.sinsemilla_commit_domain_r("Taiga-NoteCommit")
    rcv = int.from_bytes(rcv_repr32, "little")
    assert rcv < F.Q
    P_in = pp.Point(k_in[0], k_in[1], F.P).mul(input_res.quantity)
    P_out = pp.Point(k_out[0], k_out[1], F.P).mul(output_res.quantity)
    D = P_in + (-P_out) + pp.Point(R[0], R[1], F.P).mul(rcv)
    dx, dy = (0, 0) if D.inf else (D.x, D.y)
    rl_in = hc.rlcm_to_public_inputs(
        hc.resource_logic_commitment(input_res.logic, rlcm_r_in))
    rl_out = hc.rlcm_to_public_inputs(
        hc.resource_logic_commitment(output_res.logic, rlcm_r_out))
    instance = [nf, anchor, cm_out, dx, dy, rl_in[0], rl_in[1], rl_out[0], rl_out[1]]
    inputs = list(instance)
    inputs += [input_res.nk, input_res.logic, input_res.label, input_res.value,
               input_res.quantity, input_res.nonce, input_res.rseed,
               input_res.get_psi(), input_res.get_rcm(),
               1 if input_res.is_ephemeral else 0]
    inputs += [node for node, _ in merkle_path]
    inputs += [1 if is_left else 0 for _, is_left in merkle_path]
    inputs += [output_res.get_npk(), output_res.logic, output_res.label,
               output_res.value, output_res.quantity, output_res.rseed,
               1 if output_res.is_ephemeral else 0]
    inputs += list(rcv_repr32)
    inputs += [rlcm_r_in, rlcm_r_out]
    assert len(inputs) == N_INPUTS
    return instance, inputs
