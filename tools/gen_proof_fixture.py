"""Freeze the exact-circuit proof bytes as committed fixtures.

The parity tests compare GPU vs oracle computed fresh each run; these
fixtures additionally pin the ABSOLUTE bytes across rounds/refactors, so
a change that shifts both provers together (shared design drift) still
trips a test. Regenerate ONLY for an intentional, documented change:
    python tools/gen_proof_fixture.py --write
"""
import ctypes
import hashlib
import json
import os
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
GOLDEN = os.path.join(REPO, "tests", "golden")
sys.path.insert(0, os.path.join(REPO, "tests"))

RNG = bytes([9]) + bytes(31)  # same seeds as tests/test_compliance_circuit.py


def oracle_proofs():
    sys.path.insert(0, REPO)
    lib = ctypes.CDLL(os.path.join(REPO, "oracle", "liboracle.so"))
    lib.orc_prove_raw.restype = ctypes.c_long
    srs = open(os.path.join(GOLDEN, "params_15"), "rb").read()
    out = {}
    n = 1 << 15
    for name, n_adv, n_inputs, ninst in (("compliance", 10, 124, 9),
                                         ("trivial_rl", 10, 41, 22)):
        desc = open(os.path.join(GOLDEN, f"{name}.desc"), "rb").read()
        tgw = open(os.path.join(GOLDEN, f"{name}.tgw"), "rb").read()
        sample = json.load(open(os.path.join(GOLDEN, f"{name}_sample.json")))
        borsh = bytes.fromhex(sample["witness_borsh"])
        prog = ctypes.c_void_p()
        assert lib.orc_tgw_load(tgw, ctypes.c_long(len(tgw)), ctypes.byref(prog)) == 0
        adv = ctypes.create_string_buffer(n_adv * n * 32)
        inst = bytearray(ninst * 32)
        if name == "compliance":
            inputs = ctypes.create_string_buffer(n_inputs * 32)
            assert lib.orc_compliance_inputs(borsh, ctypes.c_long(len(borsh)),
                                             inputs) == 0
            inst[32:64] = int(sample["instance"][1], 16).to_bytes(32, "little")
        else:
            inputs = ctypes.create_string_buffer(n_inputs * 32)
            padding = ctypes.create_string_buffer(16 * 32)
            pad = bytes.fromhex(sample["pad_rseed"])
            assert lib.orc_rl_inputs(borsh, ctypes.c_long(len(borsh)), pad,
                                     inputs, padding) == 0
            inst[6 * 32:] = padding.raw
        assert lib.orc_tgw_run(prog, inputs, n_adv, adv) == 0
        buf = (ctypes.c_char * len(inst)).from_buffer(inst)
        assert lib.orc_tgw_instance(prog, n_adv, adv, buf) == 0
        lib.orc_tgw_free(prog)
        lib.orc_prover_reset()
        assert lib.orc_prover_init(desc, ctypes.c_long(len(desc)), srs,
                                   ctypes.c_long(len(srs))) == 0
        pout = ctypes.create_string_buffer(1 << 16)
        plen = lib.orc_prove_raw(bytes(inst), adv, RNG, pout, ctypes.c_long(1 << 16))
        assert plen > 0, f"{name}: oracle prove failed {plen}"
        out[name] = pout.raw[:plen]
    lib.orc_prover_reset()
    return out


def main():
    proofs = oracle_proofs()
    for name, p in proofs.items():
        print(f"{name}: {len(p)} B blake2b={hashlib.blake2b(p, digest_size=16).hexdigest()}")
    if "--write" in sys.argv:
        for name, p in proofs.items():
            open(os.path.join(GOLDEN, f"{name}_proof_pin.bin"), "wb").write(p)
        print("fixtures written")


if __name__ == "__main__":
    main()
