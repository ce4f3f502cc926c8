"""Transaction wire layer: the python builders
(taiga_amd/wire.py) construct borsh transactions that the product's
ctx-free parser/checker (tg_tx_wire_check) must accept — binding signature
recomputed from the parsed compliance instances — and reject under
tampering. The GPU test composes the whole stack: real proofs in the
bundle, one combined batch verification + binding check (tg_tx_verify)."""
import ctypes
import os
import secrets
import sys

import pytest

from conftest import GOLDEN, REPO

sys.path.insert(0, os.path.join(REPO, "tools"))

import taiga_amd
from taiga_amd import wire

Q = 0x40000000000000000000000000000000224698FC0994A8DD8C46EB2100000001
SEED = bytes([3]) * 32
VK_LEN = 99  # arbitrary for the ctx-free tests (dummy RL vks of that size)


def delta_commit(r: int) -> bytes:
    lib = taiga_amd.load_library()
    lib.tg_delta_commit.argtypes = [ctypes.c_char_p, ctypes.c_char_p]
    cv = ctypes.create_string_buffer(32)
    assert lib.tg_delta_commit(r.to_bytes(32, "little"), cv) == 0
    return cv.raw


def build_tx(n_sptx=2, n_cvi=2, with_rl=True, sk_override=None):
    """a structurally full transaction: dummy ZK proofs, real deltas and a
    real binding signature over the recomputed digest"""
    rs, ptxs = [], []
    nfs, cms, deltas, anchors = [], [], [], []
    for s in range(n_sptx):
        cvis = []
        for i in range(n_cvi):
            r = 1000 + 7 * s + i
            rs.append(r)
            anchor = (100 + s * 10 + i).to_bytes(32, "little")
            nf = (200 + s * 10 + i).to_bytes(32, "little")
            cm = (300 + s * 10 + i).to_bytes(32, "little")
            cv = delta_commit(r)
            inst = wire.compliance_instance(anchor, nf, cm, cv)
            cvis.append(wire.compliance_info(secrets.token_bytes(640), inst))
            anchors.append(anchor), nfs.append(nf), cms.append(cm), deltas.append(cv)
        rl_sets = []
        if with_rl:
            info = wire.rl_info(secrets.token_bytes(VK_LEN), secrets.token_bytes(512),
                                [(i).to_bytes(32, "little") for i in range(22)])
            rl_sets = [wire.rl_set(info, [info])]
        ptxs.append(wire.shielded_ptx(cvis, rl_sets, rl_sets))
    digest = taiga_amd.tx_digest(nfs, cms, deltas, anchors)
    sk = (sum(rs) % Q) if sk_override is None else sk_override
    sig = taiga_amd.binding_sign(sk.to_bytes(32, "little"), digest, SEED)
    return wire.transaction(ptxs, sig)


def test_wire_check_accepts_and_counts():
    tx = build_tx()
    ok, n_sptx, n_proofs = taiga_amd.tx_wire_check(tx, VK_LEN)
    assert ok and n_sptx == 2 and n_proofs == 4


def test_wire_check_no_rl_and_single():
    tx = build_tx(n_sptx=1, n_cvi=1, with_rl=False)
    ok, n_sptx, n_proofs = taiga_amd.tx_wire_check(tx, VK_LEN)
    assert ok and n_sptx == 1 and n_proofs == 1


def test_wire_check_rejects_tampering():
    tx = build_tx()
    # flip one instance byte -> digest changes -> binding sig fails
    bad = bytearray(tx)
    idx = tx.index((100).to_bytes(32, "little")[:4])
    bad[idx] ^= 1
    assert taiga_amd.tx_wire_check(bytes(bad), VK_LEN)[0] is False
    # truncated
    assert taiga_amd.tx_wire_check(tx[:-10], VK_LEN)[0] is False
    # wrong vk_len shifts the whole parse
    assert taiga_amd.tx_wire_check(tx, VK_LEN + 1)[0] is False
    # wrong-key signature
    assert taiga_amd.tx_wire_check(build_tx(sk_override=123456), VK_LEN)[0] is False
    # trailing garbage
    assert taiga_amd.tx_wire_check(tx + b"\x00", VK_LEN)[0] is False


def test_wire_check_unfinalized_r_tag():
    """the Option<binding_sig_r> = Some branch parses"""
    tx = build_tx(n_sptx=1, n_cvi=1, with_rl=False)
    # rebuild the single ptx with a retained r (tag 1 + 32B)
    r = (42).to_bytes(32, "little")
    # splice: the ptx ends ...[tag 0][hints u32=0]; locate and replace
    assert tx[-4 - 64 - 1 - 4] == 0  # tag byte before hints len + empty tp vec + sig
    # simpler: rebuild via the builder
    cv = delta_commit(1000)
    inst = wire.compliance_instance((100).to_bytes(32, "little"),
                                    (200).to_bytes(32, "little"),
                                    (300).to_bytes(32, "little"), cv)
    cvi = wire.compliance_info(secrets.token_bytes(64), inst)
    ptx = wire.shielded_ptx([cvi], [], [], binding_sig_r=r, hints=b"hint")
    digest = taiga_amd.tx_digest([(200).to_bytes(32, "little")],
                                 [(300).to_bytes(32, "little")], [cv],
                                 [(100).to_bytes(32, "little")])
    sig = taiga_amd.binding_sign((1000).to_bytes(32, "little"), digest, SEED)
    tx2 = wire.transaction([ptx], sig)
    ok, n_sptx, n_proofs = taiga_amd.tx_wire_check(tx2, VK_LEN)
    assert ok and n_sptx == 1 and n_proofs == 1


@pytest.mark.gpu
def test_tx_verify_end_to_end():
    """the whole §8 stack composed: a 2-proof shielded bundle with REAL
    GPU proofs over a random circuit (instances = the compliance public
    input block), real delta commitments, a real binding signature — one
    tg_tx_verify call parses the wire bytes, batch-verifies both proofs in
    one combined IPA check and checks the binding signature."""
    from gen_rand_circuit import gen

    g = taiga_amd.TaigaGpu(0)
    g.load_srs(open(os.path.join(GOLDEN, "params_15"), "rb").read())
    try:
        desc = None
        cvis, rs = [], []
        nfs, cms, deltas, anchors = [], [], [], []
        for i in range(2):
            r = 5000 + i
            while True:  # even-y delta so its repr is a canonical Fp value
                cv = delta_commit(r)
                if cv[31] & 0x80 == 0:
                    break
                r += 100
            rs.append(r)
            anchor = (1100 + i).to_bytes(32, "little")
            nf = (1200 + i).to_bytes(32, "little")
            cm = (1300 + i).to_bytes(32, "little")
            fields = [int.from_bytes(b, "little") for b in (anchor, nf, cm, cv, bytes(32))]
            d, inst_b, adv, meta = gen(21, inst_override=fields + [0])
            assert meta["n_instance_rows"] == 5
            if desc is None:
                desc = d
                g.keygen(desc)
            proof = g.create_proof_raw(inst_b, adv, bytes([60 + i]) + bytes(31))
            assert g.verify_proof_raw(inst_b, proof)
            inst192 = wire.compliance_instance(anchor, nf, cm, cv)
            cvis.append(wire.compliance_info(proof, inst192))
            anchors.append(anchor), nfs.append(nf), cms.append(cm), deltas.append(cv)
        ptx = wire.shielded_ptx(cvis, [], [])
        digest = taiga_amd.tx_digest(nfs, cms, deltas, anchors)
        sk = sum(rs) % Q
        sig = taiga_amd.binding_sign(sk.to_bytes(32, "little"), digest, SEED)
        tx = wire.transaction([ptx], sig)
        assert g.tx_verify(tx)
        # tampered proof byte -> combined batch check fails
        bad = bytearray(tx)
        bad[60] ^= 1
        assert not g.tx_verify(bytes(bad))
        # tampered signature -> binding check fails
        bad2 = bytearray(tx)
        bad2[-1] ^= 1
        assert not g.tx_verify(bytes(bad2))
    finally:
        g.close()


def test_wire_random_shapes_roundtrip():
    """property: randomly shaped (but structurally valid) transactions all
    parse with the right counts; every prefix truncation is rejected
    structurally."""
    import random

    rng = random.Random(9)
    for trial in range(6):
        n_sptx = rng.randint(1, 3)
        shapes = []
        rs, nfs, cms, deltas, anchors = [], [], [], [], []
        ptxs = []
        total_proofs = 0
        for s in range(n_sptx):
            n_cvi = rng.randint(1, 3)
            total_proofs += n_cvi
            cvis = []
            for i in range(n_cvi):
                r = rng.randrange(1, 1 << 30)
                rs.append(r)
                a, nf, cm = (rng.randrange(1 << 64).to_bytes(32, "little") for _ in range(3))
                cv = delta_commit(r)
                cvis.append(wire.compliance_info(secrets.token_bytes(rng.randrange(0, 900)),
                                                 wire.compliance_instance(a, nf, cm, cv)))
                anchors.append(a), nfs.append(nf), cms.append(cm), deltas.append(cv)
            rl = []
            for _ in range(rng.randint(0, 2)):
                info = wire.rl_info(secrets.token_bytes(VK_LEN), secrets.token_bytes(100),
                                    [bytes(32)] * 22)
                rl.append(wire.rl_set(info, [info] * rng.randint(0, 2)))
            ptxs.append(wire.shielded_ptx(cvis, rl, rl[:1],
                                          hints=secrets.token_bytes(rng.randrange(0, 40))))
        digest = taiga_amd.tx_digest(nfs, cms, deltas, anchors)
        sk = sum(rs) % Q
        sig = taiga_amd.binding_sign(sk.to_bytes(32, "little"), digest, SEED)
        tx = wire.transaction(ptxs, sig)
        ok, got_sptx, got_proofs = taiga_amd.tx_wire_check(tx, VK_LEN)
        assert ok and got_sptx == n_sptx and got_proofs == total_proofs
        # truncations never parse as valid
        for cut in (1, 7, len(tx) // 2, len(tx) - 1):
            assert taiga_amd.tx_wire_check(tx[:cut], VK_LEN)[0] is False
