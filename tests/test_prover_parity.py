"""Prover parity: the product (GPU) create_proof must be byte-identical to
the CPU oracle on the same SRS / circuit description / seeds, and the oracle
verifier must accept the GPU proof (bit-exact parity bar — SURVEY.md §8c)."""
import ctypes
import os

import pytest

from conftest import GOLDEN, REPO

pytestmark = pytest.mark.gpu

INST = bytes(32)
WIT = bytes([1]) + bytes(31)
RNG = bytes([2]) + bytes(31)


@pytest.fixture(scope="module")
def oracle_pk():
    lib = ctypes.CDLL(os.path.join(REPO, "oracle", "liboracle.so"))
    desc = open(os.path.join(GOLDEN, "cs1.desc"), "rb").read()
    srs = open(os.path.join(GOLDEN, "params_15"), "rb").read()
    lib.orc_prover_reset()  # other modules init the global PK with other descs
    rc = lib.orc_prover_init(desc, len(desc), srs, len(srs))
    assert rc in (0, 1)
    lib.orc_prove_cs1.restype = ctypes.c_long
    return lib


@pytest.fixture(scope="module")
def gpu_pk(params15):
    import taiga_amd

    g = taiga_amd.TaigaGpu(0)
    g.load_srs(params15)
    desc = open(os.path.join(GOLDEN, "cs1.desc"), "rb").read()
    g.keygen(desc)
    yield g
    g.close()


def oracle_prove(lib, inst, wit, rng):
    out = ctypes.create_string_buffer(1 << 14)
    n = lib.orc_prove_cs1(inst, wit, rng, out, 1 << 14)
    assert n > 0, f"oracle prove failed rc={n}"
    return out.raw[:n]


def test_witness_hash_matches(oracle_pk, gpu_pk):
    got = gpu_pk.witness_hash(INST, WIT)
    exp = ctypes.create_string_buffer(32)
    oracle_pk.orc_cs1_witness_hash(INST, WIT, exp)
    assert got == exp.raw


def test_proof_bytes_identical(oracle_pk, gpu_pk):
    oracle_proof = oracle_prove(oracle_pk, INST, WIT, RNG)
    gpu_proof = gpu_pk.create_proof(INST, WIT, RNG)
    assert len(gpu_proof) == len(oracle_proof)
    assert gpu_proof == oracle_proof


def test_product_verifier(oracle_pk, gpu_pk):
    """tg_verify_proof accepts oracle and GPU proofs, rejects tampering."""
    oracle_proof = oracle_prove(oracle_pk, INST, WIT, RNG)
    assert gpu_pk.verify_proof(INST, oracle_proof)
    gpu_proof = gpu_pk.create_proof(INST, WIT, RNG)
    assert gpu_pk.verify_proof(INST, gpu_proof)
    bad = bytearray(gpu_proof)
    bad[50] ^= 1
    assert not gpu_pk.verify_proof(INST, bytes(bad))
    assert not gpu_pk.verify_proof(bytes([9]) + bytes(31), gpu_proof)
    assert not gpu_pk.verify_proof(INST, gpu_proof[:-32])


def test_gpu_proof_verifies_and_seeds_differ(oracle_pk, gpu_pk):
    for seed_idx in range(2):
        inst = bytes([10 + seed_idx]) + bytes(31)
        wit = bytes([20 + seed_idx]) + bytes(31)
        rng = bytes([30 + seed_idx]) + bytes(31)
        p = gpu_pk.create_proof(inst, wit, rng)
        # oracle verifier accepts the GPU proof
        assert oracle_pk.orc_verify_cs1(inst, p, len(p)) == 0
        # and rejects it against a different instance
        assert oracle_pk.orc_verify_cs1(bytes([99]) + bytes(31), p, len(p)) != 0


def test_raw_witness_path_matches_seeded(oracle_pk, gpu_pk):
    """tg_create_proof_raw fed the exported CS1 witness must produce the
    SAME proof bytes as the seeded path (proof depends only on
    SRS/desc/witness/rng — the §8b witness-blob ABI shape)."""
    n = 1 << 15
    inst = ctypes.create_string_buffer(9 * 32)
    adv = ctypes.create_string_buffer(10 * n * 32)
    assert oracle_pk.orc_cs1_export_witness(INST, WIT, inst, adv) == 0
    raw_proof = gpu_pk.create_proof_raw(inst.raw, adv.raw, RNG)
    seeded_proof = gpu_pk.create_proof(INST, WIT, RNG)
    assert raw_proof == seeded_proof
    # and still verifies
    assert gpu_pk.verify_proof(INST, raw_proof)


def test_proof_parity_more_seeds(oracle_pk, gpu_pk):
    for s in (101, 202):
        inst = bytes([s % 251]) + bytes(31)
        wit = bytes([(s * 3) % 251]) + bytes(31)
        rng = bytes([(s * 7) % 251]) + bytes(31)
        assert gpu_pk.create_proof(inst, wit, rng) == oracle_prove(oracle_pk, inst, wit, rng)

def test_key_slots(gpu_pk):
    """PK cache: a second keygen gets a new slot, both slots
    stay selectable, and proofs from either slot of the same circuit are
    identical. Invalid slots are rejected."""
    import taiga_amd
    desc = open(os.path.join(GOLDEN, "cs1.desc"), "rb").read()
    slot1 = gpu_pk.keygen(desc)
    assert slot1 == 1  # module fixture already built slot 0
    p1 = gpu_pk.create_proof(INST, WIT, RNG)
    gpu_pk.select_key(0)
    p0 = gpu_pk.create_proof(INST, WIT, RNG)
    assert p0 == p1
    with pytest.raises(taiga_amd.TaigaGpuError):
        gpu_pk.select_key(2)
    with pytest.raises(taiga_amd.TaigaGpuError):
        gpu_pk.select_key(-1)
    gpu_pk.select_key(slot1)

def test_batch_verify(oracle_pk, gpu_pk):
    """tg_verify_batch: one combined IPA check accepts a
    bundle of valid proofs (GPU- and oracle-produced, distinct instances)
    and rejects the bundle when any single proof is tampered with or
    mismatched against its instance."""
    gpu_pk.select_key(0)
    items = []
    for s in range(4):
        inst = bytes([40 + s]) + bytes(31)
        wit = bytes([50 + s]) + bytes(31)
        rng = bytes([60 + s]) + bytes(31)
        p = gpu_pk.create_proof(inst, wit, rng) if s % 2 == 0 else \
            oracle_prove(oracle_pk, inst, wit, rng)
        items.append((inst, p))
    assert gpu_pk.verify_batch(items)
    assert gpu_pk.verify_batch(items[:1])  # m=1 path == single verifier
    # tamper one proof's bytes -> whole batch rejected
    bad = bytearray(items[2][1])
    bad[100] ^= 1
    assert not gpu_pk.verify_batch(items[:2] + [(items[2][0], bytes(bad))] + items[3:])
    # swap one instance -> rejected
    assert not gpu_pk.verify_batch(items[:3] + [(items[0][0], items[3][1])])
    # truncated member -> rejected (structural)
    assert not gpu_pk.verify_batch(items[:3] + [(items[3][0], items[3][1][:-32])])


def test_batch_verify_agrees_with_single(gpu_pk):
    """randomized agreement: for bundles of valid proofs the batch verdict
    matches per-proof verification."""
    gpu_pk.select_key(0)
    items = []
    for s in (7, 8, 9, 11, 12, 13):
        inst = bytes([s]) + bytes(31)
        p = gpu_pk.create_proof(inst, bytes([s + 1]) + bytes(31), bytes([s + 2]) + bytes(31))
        assert gpu_pk.verify_proof(inst, p)
        items.append((inst, p))
    assert gpu_pk.verify_batch(items)
