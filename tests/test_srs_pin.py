"""SRS golden-fixture pin (SURVEY.md §8c fixture #1).

tests/golden/params_15 is a byte-for-byte copy of the reference's bundled
public SRS parameter file (/root/reference/taiga_halo2/params/params_15,
loaded by constant.rs:128-139; sha256
e1fb29749c7bd0870768044d5329b4e293cb2d44dae24db2554605427b19d0dd). It is
DATA (the shared proving parameters), not code, and it is what pins the
oracle's conventions in-container:

  * layout u32(k=15) ‖ 32768×32B g ‖ 32768×32B g_lagrange ‖ 32B w ‖ 32B u
  * point compression: x.to_repr() little-endian + (y odd) in bit 255
  * g_lagrange[i] == n^{-1} · Σ_j ω^{-ij} g[j] with ω = 5^((p-1)/2^15) mod p

The last identity simultaneously validates decompression (any sign flip on
any of the 65536 points breaks it), field arithmetic, curve addition, the
generator-5 root-of-unity convention, the oracle NTT and the oracle MSM —
all against the reference's own bytes.
"""
import random

import oracle_ct as oc
import pypasta as pp


def test_layout_and_spot_decompression(params15):
    k = int.from_bytes(params15[:4], "little")
    assert k == 15
    n = 1 << k
    assert len(params15) == 4 + 2 * n * 32 + 64
    # python spot-checks a sample of points (full pass done by the C oracle)
    rng = random.Random(42)
    for idx in [0, 1, n - 1, n, 2 * n - 1] + [rng.randrange(2 * n) for _ in range(64)]:
        pt = pp.Point.from_bytes(params15[4 + 32 * idx : 4 + 32 * idx + 32], pp.Q)
        assert pt is not None and pt.is_on_curve()
    w = pp.Point.from_bytes(params15[-64:-32], pp.Q)
    u = pp.Point.from_bytes(params15[-32:], pp.Q)
    assert w.is_on_curve() and u.is_on_curve()


def test_lagrange_row0_python(params15):
    # cheap independent-python pin: g_lagrange[0] == n^{-1} * sum(g)
    n = 1 << 15
    acc = pp.Point.identity(pp.Q)
    for i in range(n):
        acc = acc + pp.Point.from_bytes(params15[4 + 32 * i : 36 + 32 * i], pp.Q)
    lhs = acc.mul(pow(n, -1, pp.P))
    gl0 = pp.Point.from_bytes(params15[4 + 32 * n : 36 + 32 * n], pp.Q)
    assert lhs == gl0


def test_srs_projection_identity(params15):
    # the full-strength random-projection pin (C oracle, all 2^15 rows):
    #   MSM(r, g_lagrange) == MSM(iNTT(r), g)  for random r
    assert oc.srs_project_check(params15, rounds=2) == 0
