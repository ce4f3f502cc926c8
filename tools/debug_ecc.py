#!/usr/bin/env python3
"""Unit-debug the ECC chip witness paths against pypasta."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from circuit import fields as F
from circuit import hostcrypto as hc
from circuit.plonkish import ConstraintSystem, Prog, assign_free_advice
from circuit.chips.lookup_range import LookupRangeCheckConfig
from circuit.chips.ecc import EccConfig
from circuit.chips.pow5 import Pow5Config
from circuit.chips.swu import HashToCurveConfig
import pypasta as pp


def mk():
    cs = ConstraintSystem(15, "dbg")
    inst = cs.instance_column()
    cs.enable_equality(inst)
    adv = [cs.advice_column() for _ in range(10)]
    for a in adv:
        cs.enable_equality(a)
    tab = cs.lookup_table_column()
    rc = LookupRangeCheckConfig(cs, adv[9], tab)
    lag = [cs.fixed_column() for _ in range(8)]
    cs.enable_constant(lag[0])
    ecc = EccConfig(cs, adv, lag, rc)
    pos = Pow5Config(cs, adv[6:9], adv[5], lag[2:5], lag[5:8])
    h2c = HashToCurveConfig(cs, adv, pos)
    return cs, adv, ecc, h2c


def main():
    cs, adv, ecc, h2c = mk()
    prog = Prog(4)
    kx, ky = hc.poseidon_to_curve([123, 456])
    q = 0xDEADBEEFCAFE1234
    prog.input_vals = [kx, ky, q, 0]
    cs.start_synth(prog, [])

    # witness the kind point
    kxc = assign_free_advice(cs, adv[0], prog.load_input(0))
    kyc = assign_free_advice(cs, adv[0], prog.load_input(1))
    from circuit.chips.ecc import EccPoint
    base = EccPoint(kxc, kyc)

    # 1) complete add: P + P = [2]P
    d = ecc.add(base, base)
    exp = pp.Point(kx, ky, F.P).mul(2)
    print("add(double):", (d.x.reg.v, d.y.reg.v) == (exp.x, exp.y))

    # 2) incomplete add: [2]P + P = [3]P
    t = ecc.add_incomplete(d, base)
    exp3 = pp.Point(kx, ky, F.P).mul(3)
    print("incomplete:", (t.x.reg.v, t.y.reg.v) == (exp3.x, exp3.y))

    # 3) var-base mul
    qc = assign_free_advice(cs, adv[0], prog.load_input(2))
    m = ecc.mul_var_base(qc, base)
    expm = pp.Point(kx, ky, F.P).mul(q)
    print("var mul:", (m.x.reg.v, m.y.reg.v) == (expm.x, expm.y))
    if (m.x.reg.v, m.y.reg.v) != (expm.x, expm.y):
        # small scalar probes
        for qq in (1, 2, 3, 5, 8):
            cs2, adv2, ecc2, _ = mk()
            prog2 = Prog(3)
            prog2.input_vals = [kx, ky, qq]
            cs2.start_synth(prog2, [])
            b2 = EccPoint(assign_free_advice(cs2, adv2[0], prog2.load_input(0)),
                          assign_free_advice(cs2, adv2[0], prog2.load_input(1)))
            m2 = ecc2.mul_var_base(assign_free_advice(cs2, adv2[0], prog2.load_input(2)), b2)
            e2 = pp.Point(kx, ky, F.P).mul(qq)
            print(f"  [q={qq}]:", (m2.x.reg.v, m2.y.reg.v) == (e2.x, e2.y))

    # 4) fixed-base mul by rcv on R
    R = hc.sinsemilla_commit_domain_r("Taiga-NoteCommit")
    rcv = 0x1234567890ABCDEF1122334455667788 % F.Q
    rb = [b for b in rcv.to_bytes(32, "little")]
    cs3, adv3, ecc3, _ = mk()
    prog3 = Prog(32)
    prog3.input_vals = rb
    cs3.start_synth(prog3, [])
    fb = ecc3.mul_fixed_full([prog3.load_input(i) for i in range(32)],
                             "resource_commit_r", R)
    expf = pp.Point(R[0], R[1], F.P).mul(rcv)
    print("fixed mul:", (fb.x.reg.v, fb.y.reg.v) == (expf.x, expf.y))

    # 5) hash_to_curve vs host poseidon_to_curve
    cs4, adv4, ecc4, h2c4 = mk()
    prog4 = Prog(2)
    prog4.input_vals = [123, 456]
    cs4.start_synth(prog4, [])
    l = assign_free_advice(cs4, adv4[0], prog4.load_input(0))
    b = assign_free_advice(cs4, adv4[0], prog4.load_input(1))
    htc = h2c4.hash_to_curve(ecc4, [l, b])
    print("hash_to_curve:", (htc.x.reg.v, htc.y.reg.v) == (kx, ky))


if __name__ == "__main__":
    main()
