#!/usr/bin/env python3
"""Verify a proof produced by the REAL Rust reference against this repo's
restated verifier (companion to tools/make_reference_vectors.rs).

    python tools/check_reference_proof.py <instance_hex> <proof_hex>

instance_hex = 9 concatenated 32-byte little-endian field reprs (the
to_instance order), proof_hex = the create_proof bytes. Accepts iff the
oracle verifier (compliance desc) accepts; also checks that a 1-bit
mutation is rejected. NOTE: byte-for-byte PROOF equality additionally
requires matching blinding-draw order; verification is
order-independent and is the cross-implementation check this script
provides.
"""
import ctypes
import os
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def main():
    inst = bytes.fromhex(sys.argv[1])
    proof = bytes.fromhex(sys.argv[2])
    assert len(inst) == 9 * 32, "expected 9 instance rows"
    lib = ctypes.CDLL(os.path.join(REPO, "oracle", "liboracle.so"))
    desc = open(os.path.join(REPO, "tests", "golden", "compliance.desc"), "rb").read()
    srs = open(os.path.join(REPO, "tests", "golden", "params_15"), "rb").read()
    lib.orc_prover_reset()
    rc = lib.orc_prover_init(desc, ctypes.c_long(len(desc)), srs, ctypes.c_long(len(srs)))
    assert rc == 0, rc
    rc = lib.orc_verify_raw(inst, proof, ctypes.c_long(len(proof)))
    print(f"verify: {'ACCEPT' if rc == 0 else f'REJECT ({rc})'}")
    bad = bytearray(proof)
    bad[64] ^= 1
    rc2 = lib.orc_verify_raw(inst, bytes(bad), ctypes.c_long(len(proof)))
    print(f"mutation: {'rejected (good)' if rc2 != 0 else 'ACCEPTED (BUG)'}")
    sys.exit(0 if rc == 0 and rc2 != 0 else 1)


if __name__ == "__main__":
    main()
