"""CPU-oracle unit tests: C oracle (liboracle.so) vs the independent
pure-Python big-int restatement (oracle/pypasta.py) plus public KATs.

Oracle provenance and the reference-file citations are in oracle/fd.h,
oracle/ntt.c, oracle/msm.c headers.
"""
import hashlib
import random

import oracle_ct as oc
import pypasta as pp


def test_field_ops_vs_python():
    rng = random.Random(1234)
    for fid, mod in ((oc.FP, pp.P), (oc.FQ, pp.Q)):
        for _ in range(40):
            a, b = rng.randrange(mod), rng.randrange(mod)
            ab, bb = a.to_bytes(32, "little"), b.to_bytes(32, "little")
            assert oc.fd_op(fid, 0, ab, bb) == ((a + b) % mod).to_bytes(32, "little")
            assert oc.fd_op(fid, 1, ab, bb) == ((a - b) % mod).to_bytes(32, "little")
            assert oc.fd_op(fid, 2, ab, bb) == (a * b % mod).to_bytes(32, "little")
            assert oc.fd_op(fid, 3, ab) == pow(a, -1, mod).to_bytes(32, "little")
            assert oc.fd_op(fid, 4, ab) == ((-a) % mod).to_bytes(32, "little")


def test_field_edge_cases():
    for fid, mod in ((oc.FP, pp.P), (oc.FQ, pp.Q)):
        z = (0).to_bytes(32, "little")
        one = (1).to_bytes(32, "little")
        pm1 = (mod - 1).to_bytes(32, "little")
        assert oc.fd_op(fid, 0, pm1, one) == z  # (m-1) + 1 == 0
        assert oc.fd_op(fid, 1, z, one) == pm1  # 0 - 1 == m-1
        assert oc.fd_op(fid, 2, pm1, pm1) == one  # (-1)^2 == 1
        assert oc.fd_op(fid, 4, z) == z
        # non-canonical input rejected
        try:
            oc.fd_op(fid, 0, mod.to_bytes(32, "little"), one)
            assert False, "expected rejection of repr >= modulus"
        except ValueError:
            pass


def test_sqrt():
    rng = random.Random(99)
    for fid, mod in ((oc.FP, pp.P), (oc.FQ, pp.Q)):
        for _ in range(8):
            a = rng.randrange(mod)
            sq = a * a % mod
            r = int.from_bytes(oc.fd_op(fid, 5, sq.to_bytes(32, "little")), "little")
            assert r * r % mod == sq
        # a known non-residue must be rejected: 5 generates F*, so 5 is a
        # non-residue mod both p and q (odd group order factor)
        nonres = pow(5, 1, mod)
        assert pow(nonres, (mod - 1) // 2, mod) == mod - 1
        try:
            oc.fd_op(fid, 5, nonres.to_bytes(32, "little"))
            assert False, "sqrt of non-residue must fail"
        except ValueError:
            pass


def test_blake2_kats():
    # RFC 7693 appendix A vector + hashlib cross-checks incl. personalization
    assert (
        oc.blake2b(b"abc").hex()
        == "ba80a53f981c4d0d6a2797b69f12f6e94c212f14685ac4b74b12bb6fdbffa2d1"
        "7d87c5392aab792dc252d5de4533cc9518d38aa8dbf1925ab92386edd4009923"
    )
    assert oc.blake2b(b"", outlen=64) == hashlib.blake2b(b"").digest()
    for n in (0, 1, 127, 128, 129, 300, 1000):
        data = bytes((i * 7 + n) & 0xFF for i in range(n))
        assert oc.blake2b(data, personal=b"Halo2-Transcript") == hashlib.blake2b(
            data, person=b"Halo2-Transcript"
        ).digest()
        assert oc.blake2s(data, personal=b"VPCommit") == hashlib.blake2s(
            data, person=b"VPCommit"
        ).digest()


def test_point_ops_vs_python():
    rng = random.Random(7)
    G = pp.Point.generator(pp.Q)

    def enc(P):
        if P.inf:
            return b"\x00" * 64
        return P.x.to_bytes(32, "little") + P.y.to_bytes(32, "little")

    gb = enc(G)
    assert oc.pt_on_curve(oc.FQ, gb)
    assert oc.pt_op(oc.FQ, 1, gb) == enc(G.double())
    P1 = G.mul(rng.randrange(pp.P))
    P2 = G.mul(rng.randrange(pp.P))
    assert oc.pt_op(oc.FQ, 0, enc(P1), enc(P2)) == enc(P1 + P2)
    # add with identity, doubling identity, P + (-P)
    assert oc.pt_op(oc.FQ, 0, enc(P1), b"\x00" * 64) == enc(P1)
    assert oc.pt_op(oc.FQ, 0, b"\x00" * 64, enc(P1)) == enc(P1)
    assert oc.pt_op(oc.FQ, 1, b"\x00" * 64) == b"\x00" * 64
    assert oc.pt_op(oc.FQ, 0, enc(P1), enc(-P1)) == b"\x00" * 64
    # scalar mult matches python; scalar 0 and order-p wrap
    k = rng.randrange(pp.P)
    assert oc.pt_op(oc.FQ, 3, gb, k.to_bytes(32, "little")) == enc(G.mul(k))
    assert oc.pt_op(oc.FQ, 3, gb, (0).to_bytes(32, "little")) == b"\x00" * 64
    # Vesta group order is p: [p]G == identity
    assert oc.pt_op(oc.FQ, 3, gb, pp.P.to_bytes(32, "little")) == b"\x00" * 64
    # Pallas side too (base field Fp, scalar field Fq): [q]G == identity
    Gp = pp.Point.generator(pp.P)
    gpb = Gp.x.to_bytes(32, "little") + Gp.y.to_bytes(32, "little")
    assert oc.pt_op(oc.FP, 3, gpb, pp.Q.to_bytes(32, "little")) == b"\x00" * 64


def test_compress_roundtrip():
    rng = random.Random(11)
    G = pp.Point.generator(pp.Q)
    pts = [G.mul(rng.randrange(pp.P)) for _ in range(16)] + [pp.Point.identity(pp.Q)]
    aff = b"".join(
        (b"\x00" * 64) if P.inf else P.x.to_bytes(32, "little") + P.y.to_bytes(32, "little")
        for P in pts
    )
    comp = oc.compress(oc.FQ, aff)
    # python compression agrees
    assert comp == b"".join(P.to_bytes() for P in pts)
    assert oc.decompress(oc.FQ, comp) == aff


def test_ntt_vs_python_and_roundtrip():
    rng = random.Random(5)
    for k in (0, 1, 3, 6, 10):
        n = 1 << k
        vals = [rng.randrange(pp.P) for _ in range(n)]
        data = b"".join(v.to_bytes(32, "little") for v in vals)
        got = oc.ntt(oc.FP, 0, k, data)
        exp = pp.ntt(vals, pp.root_of_unity(pp.P, k), pp.P)
        assert got == b"".join(v.to_bytes(32, "little") for v in exp)
        assert oc.ntt(oc.FP, 1, k, got) == data


def test_ntt_linearity_large():
    # size-independent property at a bench-class size: NTT(a + c*b) == NTT(a) + c*NTT(b)
    rng = random.Random(17)
    k = 15
    n = 1 << k
    a = [rng.randrange(pp.P) for _ in range(n)]
    b = [rng.randrange(pp.P) for _ in range(n)]
    c = rng.randrange(pp.P)
    enc = lambda v: b"".join(x.to_bytes(32, "little") for x in v)
    dec = lambda d: [int.from_bytes(d[32 * i : 32 * i + 32], "little") for i in range(n)]
    fa = dec(oc.ntt(oc.FP, 0, k, enc(a)))
    fb = dec(oc.ntt(oc.FP, 0, k, enc(b)))
    comb = [(x + c * y) % pp.P for x, y in zip(a, b)]
    fcomb = dec(oc.ntt(oc.FP, 0, k, enc(comb)))
    assert all((x + c * y) % pp.P == z for x, y, z in zip(fa, fb, fcomb))


def test_msm_vs_python():
    rng = random.Random(3)
    G = pp.Point.generator(pp.Q)
    n = 48
    pts = [G.mul(rng.randrange(pp.P)) for _ in range(n)]
    sc = [rng.randrange(pp.P) for _ in range(n)]
    # mix in edge cases: zero scalar, scalar 1, identity point
    sc[0] = 0
    sc[1] = 1
    pts[2] = pp.Point.identity(pp.Q)
    ptb = b"".join(
        (b"\x00" * 64) if P.inf else P.x.to_bytes(32, "little") + P.y.to_bytes(32, "little")
        for P in pts
    )
    scb = b"".join(s.to_bytes(32, "little") for s in sc)
    exp = pp.msm(sc, pts)
    got = oc.msm(oc.FQ, scb, ptb)
    assert got == exp.x.to_bytes(32, "little") + exp.y.to_bytes(32, "little")


def test_msm_linearity():
    # MSM(s + t, P) == MSM(s, P) + MSM(t, P) via oracle point add
    rng = random.Random(21)
    G = pp.Point.generator(pp.Q)
    n = 32
    pts = [G.mul(rng.randrange(pp.P)) for _ in range(n)]
    ptb = b"".join(P.x.to_bytes(32, "little") + P.y.to_bytes(32, "little") for P in pts)
    s = [rng.randrange(pp.P) for _ in range(n)]
    t = [rng.randrange(pp.P) for _ in range(n)]
    u = [(a + b) % pp.P for a, b in zip(s, t)]
    enc = lambda v: b"".join(x.to_bytes(32, "little") for x in v)
    lhs = oc.msm(oc.FQ, enc(u), ptb)
    rhs = oc.pt_op(oc.FQ, 0, oc.msm(oc.FQ, enc(s), ptb), oc.msm(oc.FQ, enc(t), ptb))
    assert lhs == rhs
