"""Prover parity fuzzing over RANDOM satisfiable circuits
(tools/gen_rand_circuit.py): shapes beyond the CS1 fixture — varied column
counts, gates, lookups on/off, chunk counts, blinding factors — through
the raw-witness path. The oracle must prove+verify each circuit, and (GPU)
the product prover's proof bytes must equal the oracle's bit-for-bit with
cross-verification both ways."""
import ctypes
import os
import sys

import pytest

from conftest import GOLDEN, REPO

sys.path.insert(0, os.path.join(REPO, "tools"))

RNG = bytes([7]) + bytes(31)


def load_oracle():
    lib = ctypes.CDLL(os.path.join(REPO, "oracle", "liboracle.so"))
    lib.orc_prove_raw.restype = ctypes.c_long
    return lib


def oracle_keygen(lib, desc):
    srs = open(os.path.join(GOLDEN, "params_15"), "rb").read()
    lib.orc_prover_reset()
    assert lib.orc_prover_init(desc, len(desc), srs, len(srs)) == 0


def oracle_prove(lib, inst, adv):
    out = ctypes.create_string_buffer(1 << 15)
    n = lib.orc_prove_raw(inst, adv, RNG, out, 1 << 15)
    assert n > 0, f"oracle raw prove failed rc={n}"
    return out.raw[:n]


def test_oracle_random_circuit_roundtrip():
    """CPU: one random circuit proves and verifies on the oracle; tampered
    proofs and wrong instances are rejected."""
    from gen_rand_circuit import gen

    lib = load_oracle()
    desc, inst, adv, meta = gen(11)
    oracle_keygen(lib, desc)
    proof = oracle_prove(lib, inst, adv)
    assert lib.orc_verify_raw(inst, proof, len(proof)) == 0
    bad = bytearray(proof)
    bad[60] ^= 1
    assert lib.orc_verify_raw(inst, bytes(bad), len(proof)) != 0
    wrong_inst = bytes(32 * meta["n_instance_rows"])
    assert lib.orc_verify_raw(wrong_inst, proof, len(proof)) != 0
    lib.orc_prover_reset()


def test_generator_shapes_differ():
    """the generator actually varies the stressed dimensions"""
    from gen_rand_circuit import gen

    metas = [gen(s)[3] for s in (1, 2)]
    assert metas[0] != metas[1]


@pytest.mark.gpu
def test_gpu_random_circuit_parity():
    """GPU vs oracle on 28 random circuit shapes (VERDICT round-1 item 6:
    widened from 5, then 20) — including the structural edges seed 69 (SINGLE
    permutation chunk, no lookups: no last_z chaining at all) and seed 8
    (3 chunks, no lookups) — with bit-identical proofs,
    cross-verification both ways, tamper rejection."""
    import taiga_amd
    from gen_rand_circuit import gen

    lib = load_oracle()
    g = taiga_amd.TaigaGpu(0)
    g.load_srs(open(os.path.join(GOLDEN, "params_15"), "rb").read())
    try:
        for seed in (21, 22, 23, 69, 8, 31, 32, 33, 34, 35, 36, 37, 38,
                     101, 102, 103, 104, 105, 106, 107,
                     201, 202, 203, 204, 205, 206, 207, 208):
            desc, inst, adv, meta = gen(seed)
            oracle_keygen(lib, desc)
            slot = g.keygen(desc)
            g.select_key(slot)
            o_proof = oracle_prove(lib, inst, adv)
            g_proof = g.create_proof_raw(inst, adv, RNG)
            assert g_proof == o_proof, f"seed {seed} ({meta}): proof bytes differ"
            # cross verification
            assert g.verify_proof_raw(inst, o_proof)
            assert lib.orc_verify_raw(inst, g_proof, len(g_proof)) == 0
            bad = bytearray(g_proof)
            bad[-1] ^= 1
            assert not g.verify_proof_raw(inst, bytes(bad))
            # raw-instance BATCH verify on this circuit: two valid proofs
            # accepted together, rejected when one is tampered
            p2 = g.create_proof_raw(inst, adv, bytes([8]) + bytes(31))
            assert g.verify_batch_raw([(inst, g_proof), (inst, p2)])
            assert not g.verify_batch_raw([(inst, g_proof), (inst, bytes(bad))])
        lib.orc_prover_reset()
    finally:
        g.close()
