"""The TrivialResourceLogicCircuit, restated.

configure mirrors ResourceLogicConfig::configure
(resource_logic_circuit.rs:321-410); synthesize mirrors the
resource_logic_circuit_impl! default flow (basic_constraints
:424-475 -> load_resource integrity.rs:328-512, then the Trivial
custom_constraints = publicize_default_dynamic_resource_logic_commitments
blake2s.rs:37-76, which exposes the DEFAULT (all-zero) commitment halves).

Witness-program input layout:
   0..21   instance rows (RESOURCE_LOGIC_CIRCUIT_PUBLIC_INPUT_NUM = 22:
           [resource_merkle_root, self_resource_id, 0,0, 0,0,
            16 random padding rows])
   22      is_input            23 nk_or_npk
   24 logic  25 label  26 value  27 quantity  28 nonce  29 rseed
   30 psi    31 rcm    32 is_ephemeral
   33..36   resource-tree path node values (TAIGA_RESOURCE_TREE_DEPTH = 4)
   37..40   path lr bits
"""
from .plonkish import (ConstraintSystem, Prog, assign_free_advice,
                       assign_free_constant)
from .chips.pow5 import Pow5Config, poseidon_hash_gadget
from .chips.lookup_range import LookupRangeCheckConfig
from .chips.gadgets import (CondSwapConfig, ConditionalSelectConfig,
                            ConditionalEqualConfig, ArithConfig,
                            ExtendedOrRelationConfig, ResourceCommitConfig,
                            merkle_poseidon_gadget, quantity_range_check)
from .chips.blake2s import Blake2sConfig
from .chips.ecc import EccConfig
from . import hostcrypto as hc
from . import fields as F

K = 15
N_INPUTS = 41
TREE_DEPTH = 4
N_PUB = 22
ROOT_ROW, SELF_ID_ROW = 0, 1
D1_CM1, D1_CM2, D2_CM1, D2_CM2 = 2, 3, 4, 5


class TrivialRLModel:
    def __init__(self):
        cs = ConstraintSystem(K, "trivial_rl")
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
        self.cond_equal = ConditionalEqualConfig(cs, self.advices[0:3])
        self.cond_select = ConditionalSelectConfig(cs, self.advices[0:2])
        self.add_cfg = ArithConfig(cs, self.advices[0:2], "add")
        self.sub_cfg = ArithConfig(cs, self.advices[0:2], "sub")
        self.mul_cfg = ArithConfig(cs, self.advices[0:2], "mul")
        self.ext_or = ExtendedOrRelationConfig(cs, self.advices[0:3])
        self.blake2s = Blake2sConfig(cs, self.advices)
        self.resource_commit = ResourceCommitConfig(
            cs, self.advices[0:3], self.poseidon, self.range_check)
        self.cond_swap = CondSwapConfig(cs, self.advices[:5])

    def synthesize(self, inputs_int):
        cs = self.cs
        prog = Prog(N_INPUTS)
        prog.input_vals = [v % F.P for v in inputs_int]
        inst_v = [prog.load_input(i) for i in range(N_PUB)]
        cs.start_synth(prog, inst_v)
        inp = lambda i: prog.load_input(i)
        adv = self.advices

        cs.assign_table(self.table_idx, list(range(1 << 10)))

        # ---- load_resource (integrity.rs:328-512) ----
        is_input = assign_free_advice(cs, adv[0], inp(22))
        nk_or_npk = assign_free_advice(cs, adv[0], inp(23))
        zero_c = assign_free_constant(cs, adv[0], 0)
        input_npk = poseidon_hash_gadget(self.poseidon, [nk_or_npk, zero_c])
        npk = self.cond_select.assign(is_input, input_npk, nk_or_npk)
        value = assign_free_advice(cs, adv[0], inp(26))
        logic = assign_free_advice(cs, adv[0], inp(24))
        label = assign_free_advice(cs, adv[0], inp(25))
        quantity = quantity_range_check(self.range_check, inp(27))
        nonce = assign_free_advice(cs, adv[0], inp(28))
        assign_free_advice(cs, adv[0], inp(29))  # rseed (witnessed, unused)
        psi = assign_free_advice(cs, adv[0], inp(30))
        rcm = assign_free_advice(cs, adv[0], inp(31))
        is_eph = assign_free_advice(cs, adv[0], inp(32))
        cm = self.resource_commit.resource_commit(
            logic, label, value, npk, nonce, psi, quantity, is_eph, rcm)
        nf = poseidon_hash_gadget(self.poseidon, [nk_or_npk, nonce, psi, cm])
        self_id = self.cond_select.assign(is_input, nf, cm)
        path = [(inp(33 + i), inp(37 + i)) for i in range(TREE_DEPTH)]
        root = merkle_poseidon_gadget(self.cond_swap, self.poseidon, self_id, path)

        cs.constrain_instance(root, self.instances, ROOT_ROW)
        cs.constrain_instance(self_id, self.instances, SELF_ID_ROW)

        # ---- custom_constraints: default dynamic RL commitments ----
        d = hc.rlcm_to_public_inputs(b"\x00" * 32)  # default = zero bytes
        c1 = assign_free_advice(cs, adv[0], cs.prog.const(d[0]))
        c2 = assign_free_advice(cs, adv[0], cs.prog.const(d[1]))
        cs.constrain_instance(c1, self.instances, D1_CM1)
        cs.constrain_instance(c2, self.instances, D1_CM2)
        cs.constrain_instance(c1, self.instances, D2_CM1)
        cs.constrain_instance(c2, self.instances, D2_CM2)
        return cs


def build_inputs(resource: hc.Resource, merkle_path, is_input: bool,
                 padding16):
    """Host-side mirror of TrivialResourceLogicCircuit::get_public_inputs
    (resource_logic_examples.rs:95-107) + the witness input vector."""
    nk_or_npk = resource.nk if is_input else resource.get_npk()
    cm = resource.commitment()
    if is_input:
        self_id = resource.get_nf()
    else:
        self_id = cm
    root = hc.merkle_root(self_id, merkle_path)
    d = hc.rlcm_to_public_inputs(b"\x00" * 32)
    instance = [root, self_id, d[0], d[1], d[0], d[1]] + list(padding16)
    assert len(instance) == N_PUB
    inputs = list(instance)
    inputs += [1 if is_input else 0, nk_or_npk,
               resource.logic, resource.label, resource.value,
               resource.quantity, resource.nonce, resource.rseed,
               resource.get_psi(), resource.get_rcm(),
               1 if resource.is_ephemeral else 0]
    inputs += [node for node, _ in merkle_path]
    inputs += [1 if is_left else 0 for _, is_left in merkle_path]
    assert len(inputs) == N_INPUTS
    return instance, inputs
