"""ShieldedPartialTransaction::build parity (BASELINE configs[3]).

One 2-in/2-out ptx = 2 compliance + 4 TrivialRL proofs over the ptx
resource tree (resource_tree.rs ResourceMerkleTreeLeaves: leaves
[nf_1, cm_1, nf_2, cm_2], depth 4, zero-padded). The GPU's tg_ptx_build
bundle bytes must equal the oracle's orc_ptx_build bit-for-bit on the
same units + rng seed, and tg_ptx_verify must accept the bundle (batch
verification of all 6 proofs + the execute() consistency checks,
shielded_ptx.rs:232-240) and reject tampering.
"""
import ctypes
import hashlib
import os
import sys

import pytest

from conftest import GOLDEN, REPO

sys.path.insert(0, os.path.join(REPO, "tools"))

RNG = bytes([17]) + bytes(31)


def _build_units(n_compliance=2):
    from circuit import fields as F
    from circuit import hostcrypto as hc

    def det(seed, n):
        return int.from_bytes(hashlib.blake2b(seed, digest_size=64).digest(),
                              "little") % n

    def mkres(tag, nk_is_key=True, nonce=None):
        fp = lambda s: det(tag + s, F.P)
        return hc.Resource(
            logic=fp(b"logic"), label=fp(b"label"), value=fp(b"value"),
            quantity=det(tag + b"q", 1 << 64), nk=fp(b"nk"), nk_is_key=nk_is_key,
            nonce=nonce if nonce is not None else fp(b"nonce"),
            is_ephemeral=False, rseed=fp(b"rseed"))

    comp_units = []
    rl_wits_in, rl_wits_out = [], []
    leaves = []
    resources = []
    for i in range(n_compliance):
        tag = b"ptx%d" % i
        rin = mkres(tag + b"in")
        nf = rin.get_nf()
        # balanced transfer: the output carries the input's kind and
        # quantity, so sum(delta) = [sum rcv]R and the binding signature
        # verifies (delta_commitment.rs / transaction.rs:99-114)
        rout = mkres(tag + b"out", nonce=nf)
        rout.logic, rout.label, rout.quantity = rin.logic, rin.label, rin.quantity
        # commitment-tree path (depth 32) for the compliance proof
        path = [(det(tag + b"n%d" % j, F.P), bool(det(tag + b"l%d" % j, 2)))
                for j in range(32)]
        anchor = hc.merkle_root(rin.commitment(), path)
        rseed = hashlib.blake2b(tag + b"rs", digest_size=32).digest()
        borsh = rin.borsh()
        import struct
        borsh += struct.pack("<I", 32)
        for node, is_left in path:
            borsh += F.to_repr(node) + bytes([1 if is_left else 0])
        borsh += F.to_repr(anchor)
        borsh += rout.borsh()
        borsh += rseed
        comp_units.append(borsh)
        leaves += [nf, rout.commitment()]
        resources.append((rin, rout))

    # resource tree: depth 4, leaves padded with zeros (ragged for
    # n_compliance != 2: 2*n leaves used of 16)
    layer = leaves + [0] * (16 - len(leaves))
    layers = [layer]
    while len(layer) > 1:
        layer = [hc.poseidon_hash(layer[i], layer[i + 1])
                 for i in range(0, len(layer), 2)]
        layers.append(layer)

    def path_of(pos):
        out = []
        p = pos
        for lvl in range(4):
            sib = p ^ 1
            is_left = sib < p  # sibling is the left child
            out.append((layers[lvl][sib], is_left))
            p >>= 1
        return out

    for i, (rin, rout) in enumerate(resources):
        for j, res in ((0, rin), (1, rout)):
            pos = 2 * i + j
            path = path_of(pos)
            # sanity: is_input convention (resource_tree.rs:41-43)
            assert (not path[0][1]) == (j == 0)
            wb = res.borsh()
            for node, is_left in path:
                wb += F.to_repr(node) + bytes([1 if is_left else 0])
            (rl_wits_in if j == 0 else rl_wits_out).append(wb)
    return comp_units, rl_wits_in, rl_wits_out


@pytest.fixture(scope="module")
def units():
    return _build_units()


def test_oracle_ptx_build(units):
    lib = ctypes.CDLL(os.path.join(REPO, "oracle", "liboracle.so"))
    lib.orc_ptx_build.restype = ctypes.c_long
    comp, rin, rout = units
    out = ctypes.create_string_buffer(1 << 18)
    n = _orc_ptx(lib, comp, rin, rout, out)
    assert n > 0, f"orc_ptx_build failed: {n}"
    # structure: 2 compliance + 4 RL; vk_len = 32*(21+12)
    import struct
    (nc,) = struct.unpack_from("<I", out.raw, 0)
    assert nc == 2


def _orc_ptx(lib, comp, rin, rout, out):
    cdesc = open(os.path.join(GOLDEN, "compliance.desc"), "rb").read()
    rdesc = open(os.path.join(GOLDEN, "trivial_rl.desc"), "rb").read()
    srs = open(os.path.join(GOLDEN, "params_15"), "rb").read()
    ctgw = open(os.path.join(GOLDEN, "compliance.tgw"), "rb").read()
    rtgw = open(os.path.join(GOLDEN, "trivial_rl.tgw"), "rb").read()
    return lib.orc_ptx_build(
        cdesc, ctypes.c_long(len(cdesc)), rdesc, ctypes.c_long(len(rdesc)),
        srs, ctypes.c_long(len(srs)), ctgw, ctypes.c_long(len(ctgw)),
        rtgw, ctypes.c_long(len(rtgw)),
        len(comp), b"".join(comp), len(rin), len(rout),
        b"".join(rin) + b"".join(rout),
        RNG, out, ctypes.c_long(len(out)))


@pytest.mark.gpu
@pytest.mark.parametrize("n", [1, 3])
def test_gpu_ptx_ragged_parity(n, params15):
    """GPU == oracle bit-for-bit on ragged bundles too (1 and 3
    compliance units), and the bundles verify."""
    import taiga_amd

    comp, rin, rout = _build_units(n)
    g = taiga_amd.TaigaGpu(0)
    try:
        g.load_srs(params15)
        slot_c = g.keygen(open(os.path.join(GOLDEN, "compliance.desc"), "rb").read())
        g.witness_program_load(open(os.path.join(GOLDEN, "compliance.tgw"), "rb").read())
        slot_r = g.keygen(open(os.path.join(GOLDEN, "trivial_rl.desc"), "rb").read())
        g.witness_program_load(open(os.path.join(GOLDEN, "trivial_rl.tgw"), "rb").read())
        lib = taiga_amd.api.load_library()
        out = ctypes.create_string_buffer(1 << 19)
        out_len = ctypes.c_size_t()
        rc = lib.tg_ptx_build(g._h, slot_c, slot_r, n, b"".join(comp),
                              n, n, b"".join(rin) + b"".join(rout), RNG, out,
                              len(out), ctypes.byref(out_len))
        assert rc == 0, f"tg_ptx_build(n={n}) rc={rc}"
        ptx_gpu = out.raw[:out_len.value]
        olib = ctypes.CDLL(os.path.join(REPO, "oracle", "liboracle.so"))
        olib.orc_ptx_build.restype = ctypes.c_long
        oout = ctypes.create_string_buffer(1 << 19)
        m = _orc_ptx(olib, comp, rin, rout, oout)
        assert m == len(ptx_gpu)
        assert oout.raw[:m] == ptx_gpu, f"n={n}: GPU bundle != oracle bundle"
        assert lib.tg_ptx_verify(g._h, slot_c, slot_r, ptx_gpu, len(ptx_gpu)) == 0
    finally:
        g.close()


@pytest.mark.gpu
def test_gpu_ptx_build_parity_and_verify(units, params15):
    import taiga_amd

    comp, rin, rout = units
    g = taiga_amd.TaigaGpu(0)
    g.load_srs(params15)
    cdesc = open(os.path.join(GOLDEN, "compliance.desc"), "rb").read()
    rdesc = open(os.path.join(GOLDEN, "trivial_rl.desc"), "rb").read()
    slot_c = g.keygen(cdesc)
    g.witness_program_load(open(os.path.join(GOLDEN, "compliance.tgw"), "rb").read())
    slot_r = g.keygen(rdesc)
    g.witness_program_load(open(os.path.join(GOLDEN, "trivial_rl.tgw"), "rb").read())
    lib = taiga_amd.api.load_library()
    out = ctypes.create_string_buffer(1 << 18)
    out_len = ctypes.c_size_t()
    rc = lib.tg_ptx_build(g._h, slot_c, slot_r, 2, b"".join(comp), 2, 2,
                          b"".join(rin) + b"".join(rout), RNG, out,
                          len(out), ctypes.byref(out_len))
    assert rc == 0, f"tg_ptx_build rc={rc}"
    ptx_gpu = out.raw[:out_len.value]

    olib = ctypes.CDLL(os.path.join(REPO, "oracle", "liboracle.so"))
    olib.orc_ptx_build.restype = ctypes.c_long
    oout = ctypes.create_string_buffer(1 << 18)
    n = _orc_ptx(olib, comp, rin, rout, oout)
    assert n == len(ptx_gpu)
    assert oout.raw[:n] == ptx_gpu, "GPU ptx bundle != oracle ptx bundle"

    assert lib.tg_ptx_verify(g._h, slot_c, slot_r, ptx_gpu, len(ptx_gpu)) == 0
    bad = bytearray(ptx_gpu)
    bad[40] ^= 1  # inside the first compliance proof
    assert lib.tg_ptx_verify(g._h, slot_c, slot_r, bytes(bad), len(bad)) != 0
    g.close()


@pytest.mark.gpu
def test_gpu_full_transaction_verify(units, params15):
    """tg_tx_verify_full: a real Transaction (one 2-in/2-out ptx from
    tg_ptx_build + RedDSA binding signature over the recomputed digest)
    verifies end to end — ALL 6 proofs batch-verified on their keys, RL
    vks matched, consistency checks, binding check (Transaction::execute
    semantics — closes ADVICE round-1 item 2)."""
    import struct

    import taiga_amd
    from taiga_amd import wire

    comp, rin, rout = units
    g = taiga_amd.TaigaGpu(0)
    g.load_srs(params15)
    slot_c = g.keygen(open(os.path.join(GOLDEN, "compliance.desc"), "rb").read())
    g.witness_program_load(open(os.path.join(GOLDEN, "compliance.tgw"), "rb").read())
    slot_r = g.keygen(open(os.path.join(GOLDEN, "trivial_rl.desc"), "rb").read())
    g.witness_program_load(open(os.path.join(GOLDEN, "trivial_rl.tgw"), "rb").read())
    lib = taiga_amd.api.load_library()
    out = ctypes.create_string_buffer(1 << 18)
    out_len = ctypes.c_size_t()
    rc = lib.tg_ptx_build(g._h, slot_c, slot_r, 2, b"".join(comp), 2, 2,
                          b"".join(rin) + b"".join(rout), RNG, out,
                          len(out), ctypes.byref(out_len))
    assert rc == 0
    ptx = out.raw[:out_len.value]
    # binding_sig_r = the Some() scalar before the empty hints vec
    assert ptx[-37] == 1
    r = ptx[-36:-4]
    # digest streams from the bundle's compliance instances
    nfs, cms, deltas, anchors = [], [], [], []
    (n_cvi,) = struct.unpack_from("<I", ptx, 0)
    off = 4
    for _ in range(n_cvi):
        (plen,) = struct.unpack_from("<I", ptx, off)
        off += 4 + plen
        inst = ptx[off:off + 192]
        off += 192
        anchors.append(inst[0:32])
        nfs.append(inst[32:64])
        cms.append(inst[64:96])
        deltas.append(inst[96:128])
    digest = taiga_amd.tx_digest(nfs, cms, deltas, anchors)
    sig = taiga_amd.binding_sign(r, digest, bytes([5]) * 32)
    tx = wire.transaction([ptx], sig)
    rc = lib.tg_tx_verify_full(g._h, slot_c, slot_r, tx, len(tx))
    assert rc == 0, f"tg_tx_verify_full rc={rc}"
    # tamper: flip a byte inside an RL proof -> batch verify must fail
    bad = bytearray(tx)
    bad[len(tx) - 2000] ^= 1
    assert lib.tg_tx_verify_full(g._h, slot_c, slot_r, bytes(bad), len(bad)) != 0
    # wrong binding signature
    bad2 = bytearray(tx)
    bad2[-1] ^= 1
    assert lib.tg_tx_verify_full(g._h, slot_c, slot_r, bytes(bad2), len(bad2)) != 0
    g.close()


@pytest.mark.gpu
def test_gpu_two_ptx_transaction_verify(params15):
    """A Transaction aggregating TWO ShieldedPartialTransactions: the
    binding signing key is the sum of the per-ptx binding_sig_r scalars
    and the digest streams span both bundles (transaction.rs:99-158 —
    Transaction::sign over all ptxs)."""
    import struct

    import taiga_amd
    from taiga_amd import wire

    sys.path.insert(0, os.path.join(REPO, "oracle"))
    import pypasta as pp

    g = taiga_amd.TaigaGpu(0)
    try:
        g.load_srs(params15)
        slot_c = g.keygen(open(os.path.join(GOLDEN, "compliance.desc"), "rb").read())
        g.witness_program_load(open(os.path.join(GOLDEN, "compliance.tgw"), "rb").read())
        slot_r = g.keygen(open(os.path.join(GOLDEN, "trivial_rl.desc"), "rb").read())
        g.witness_program_load(open(os.path.join(GOLDEN, "trivial_rl.tgw"), "rb").read())
        lib = taiga_amd.api.load_library()

        ptxs, rs = [], []
        for n, rng0 in ((1, 31), (2, 32)):
            comp, rin, rout = _build_units(n)
            out = ctypes.create_string_buffer(1 << 19)
            out_len = ctypes.c_size_t()
            rc = lib.tg_ptx_build(g._h, slot_c, slot_r, n, b"".join(comp),
                                  n, n, b"".join(rin) + b"".join(rout),
                                  bytes([rng0]) + bytes(31), out, len(out),
                                  ctypes.byref(out_len))
            assert rc == 0
            ptx = out.raw[:out_len.value]
            assert ptx[-37] == 1
            rs.append(int.from_bytes(ptx[-36:-4], "little"))
            ptxs.append(ptx)

        nfs, cms, deltas, anchors = [], [], [], []
        for ptx in ptxs:
            (n_cvi,) = struct.unpack_from("<I", ptx, 0)
            off = 4
            for _ in range(n_cvi):
                (plen,) = struct.unpack_from("<I", ptx, off)
                off += 4 + plen
                inst = ptx[off:off + 192]
                off += 192
                anchors.append(inst[0:32])
                nfs.append(inst[32:64])
                cms.append(inst[64:96])
                deltas.append(inst[96:128])
        digest = taiga_amd.tx_digest(nfs, cms, deltas, anchors)
        r_total = ((rs[0] + rs[1]) % pp.Q).to_bytes(32, "little")
        sig = taiga_amd.binding_sign(r_total, digest, bytes([6]) * 32)
        tx = wire.transaction(ptxs, sig)
        rc = lib.tg_tx_verify_full(g._h, slot_c, slot_r, tx, len(tx))
        assert rc == 0, f"two-ptx tg_tx_verify_full rc={rc}"
        bad = bytearray(tx)
        bad[-1] ^= 1
        assert lib.tg_tx_verify_full(g._h, slot_c, slot_r, bytes(bad), len(bad)) != 0
    finally:
        g.close()
