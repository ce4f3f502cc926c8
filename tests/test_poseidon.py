"""Poseidon P128Pow5T3 over Fp: parameter derivation (Grain LFSR) pinned by
DOUBLE IMPLEMENTATION — tools/gen_poseidon.py (Python) vs oracle/poseidon.c
(C) must agree bit-for-bit on all 201 constants — plus sponge semantics
tests. The halo2_gadgets crate the reference uses (taiga_halo2/src/
utils.rs:40-48) is un-vendored, so this derivation is an "assumed,
restated" convention per DESIGN.md §6; the GPU kernel is separately
pinned against the oracle in test_gpu_parity.py."""
import ctypes
import os
import sys

import pytest

from conftest import GOLDEN, REPO

sys.path.insert(0, os.path.join(REPO, "tools"))


@pytest.fixture(scope="module")
def orc():
    return ctypes.CDLL(os.path.join(REPO, "oracle", "liboracle.so"))


@pytest.fixture(scope="module")
def pygen():
    import gen_poseidon as gp
    rc, mds = gp.generate()
    return gp, rc, mds


def test_constants_double_implementation(orc, pygen):
    gp, rc, mds = pygen
    blob = b"".join(v.to_bytes(32, "little") for row in rc for v in row)
    blob += b"".join(mds[i][j].to_bytes(32, "little") for i in range(3) for j in range(3))
    buf = ctypes.create_string_buffer(len(blob))
    assert orc.orc_poseidon_consts(buf, len(blob)) == len(blob)
    assert buf.raw == blob


def test_constants_golden_fixture(orc):
    golden = open(os.path.join(GOLDEN, "poseidon_p128t3.bin"), "rb").read()
    buf = ctypes.create_string_buffer(len(golden))
    assert orc.orc_poseidon_consts(buf, len(golden)) == len(golden)
    assert buf.raw == golden


def test_constants_shape(pygen):
    gp, rc, mds = pygen
    P = gp.P
    assert len(rc) == 64 and all(len(r) == 3 for r in rc)
    assert all(0 < v < P for row in rc for v in row)  # canonical, nonzero w.h.p.
    # MDS is Cauchy: entries nonzero, rows distinct, invertible
    assert all(mds[i][j] != 0 for i in range(3) for j in range(3))
    det = (
        mds[0][0] * (mds[1][1] * mds[2][2] - mds[1][2] * mds[2][1])
        - mds[0][1] * (mds[1][0] * mds[2][2] - mds[1][2] * mds[2][0])
        + mds[0][2] * (mds[1][0] * mds[2][1] - mds[1][1] * mds[2][0])
    ) % P
    assert det != 0


def test_hash_oracle_vs_python(orc, pygen):
    gp, rc, mds = pygen
    import random

    rng = random.Random(42)
    for L in (1, 2, 3, 5, 9):
        msg = [rng.randrange(gp.P) for _ in range(L)]
        exp = gp.hash_n(msg, rc, mds)
        raw = b"".join(v.to_bytes(32, "little") for v in msg)
        out = ctypes.create_string_buffer(32)
        assert orc.orc_poseidon_hash(raw, L, out) == 0
        assert int.from_bytes(out.raw, "little") == exp


def test_hash_domain_separation(orc):
    """ConstantLength<L> capacity element = L<<64: the same field elements
    hashed under different L must differ (L=1 pads to [m, 0] which would
    collide with L=2 of (m, 0) without the domain tag)."""
    m = (7).to_bytes(32, "little")
    two = m + bytes(32)
    o1 = ctypes.create_string_buffer(32)
    o2 = ctypes.create_string_buffer(32)
    assert orc.orc_poseidon_hash(m, 1, o1) == 0
    assert orc.orc_poseidon_hash(two, 2, o2) == 0
    assert o1.raw != o2.raw


def test_hash_rejects_noncanonical(orc):
    import gen_poseidon as gp
    bad = gp.P.to_bytes(32, "little") + bytes(32)
    out = ctypes.create_string_buffer(32)
    assert orc.orc_poseidon_hash(bad, 2, out) != 0


def test_permute_matches_python(orc, pygen):
    gp, rc, mds = pygen
    st = [3, 4, 5]
    exp = gp.permute(st, rc, mds)
    buf = ctypes.create_string_buffer(b"".join(v.to_bytes(32, "little") for v in st), 96)
    assert orc.orc_poseidon_permute(buf) == 0
    got = [int.from_bytes(buf.raw[32 * i : 32 * i + 32], "little") for i in range(3)]
    assert got == exp
