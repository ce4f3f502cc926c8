#!/usr/bin/env python3
"""Derive the Iso-Pallas -> Pallas 3-isogeny constants from first principles.

The reference circuit's iso_map gate hardcodes pallas::Point::ISOGENY_CONSTANTS
(circuit/curve/map_to_curve.rs:53) from the un-vendored pasta_curves crate.
Rather than restate them from memory, this derives them exactly:

  1. The iso curve is y^2 = x^3 + A x + B (A = ISO_A, B = 1265, public
     parameters restated in tools/circuit/hostcrypto.py and pinned by the
     reference's in-tree map_to_curve KAT at u=0).
  2. Its division polynomial psi_3 = 3x^4 + 6A x^2 + 12B x - A^2 has exactly
     one Fp-rational root x0 (computed via gcd(x^p - x, psi_3)); the kernel
     {O, (x0, +-y0)} is Fp-rational as a group even though y0 lives in Fp^2
     (Velu's formulas only use y0^2 = x0^3 + A x0 + B).
  3. Velu: v = 2(3 x0^2 + A), u = 4 y0^2, w = u + x0 v;
     target curve a' = A - 5v = 0, b' = B - 7w = 3645 = 5 * 3^6, so the
     isomorphism (x, y) -> (x/9, y/27) lands exactly on Pallas
     y^2 = x^3 + 5.
  4. Canonical (monic-denominator) rational-map coefficients are unique and
     match the gate's layout:
       num_x = (iso0 x^3 + iso1 x^2 + iso2 x + iso3),  div_x = x^2 + iso4 x + iso5
       num_y = (iso6 x^3 + iso7 x^2 + iso8 x + iso9) y, div_y = (x^3 + iso10 x^2 + iso11 x + iso12) z^3
     with num_x = N/9, num_y = (N'(x-x0) - 2N)/27, N the monic Velu cubic.

Validated end-to-end by the R_U/R_Z window-table byte pins
(tests/test_fixed_base_tables.py): the tables only match if the isogeny —
and everything else in the group-hash chain — is exact.
"""
import sys
import os

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from circuit import hostcrypto as hc, fields as F  # noqa: E402

P = F.P
A, B = hc.ISO_A, hc.ISO_B


def poly_gcd_roots():
    def pmulmod(a, b, f):
        r = [0] * (len(a) + len(b) - 1)
        for i, x in enumerate(a):
            if x:
                for j, y in enumerate(b):
                    r[i + j] = (r[i + j] + x * y) % P
        df = len(f) - 1
        while len(r) > df:
            c = r[-1]
            if c:
                for i in range(df + 1):
                    r[len(r) - 1 - df + i] = (r[len(r) - 1 - df + i] - c * f[i]) % P
            r.pop()
        return r

    def ppowmod(base, e, f):
        r = [1]
        b = base[:]
        while e:
            if e & 1:
                r = pmulmod(r, b, f)
            b = pmulmod(b, b, f)
            e >>= 1
        return r

    f = [(-A * A) % P, (12 * B) % P, (6 * A) % P, 0, 3]
    inv3 = pow(3, P - 2, P)
    f = [c * inv3 % P for c in f]
    xp = ppowmod([0, 1], P, f)
    g = [(xp[i] if i < len(xp) else 0) - (1 if i == 1 else 0) for i in range(max(len(xp), 2))]
    g = [c % P for c in g]
    while g and g[-1] == 0:
        g.pop()

    def pgcd(a, b):
        a, b = a[:], b[:]
        while b:
            binv = pow(b[-1], P - 2, P)
            while len(a) >= len(b):
                c = a[-1] * binv % P
                if c:
                    for i in range(len(b)):
                        a[len(a) - len(b) + i] = (a[len(a) - len(b) + i] - c * b[i]) % P
                a.pop()
                while a and a[-1] == 0:
                    a.pop()
                if not a:
                    break
            a, b = b, a
        return a

    h = pgcd(f, g)
    assert len(h) == 2, "expected exactly one rational 3-torsion x"
    return (-h[0] * pow(h[1], P - 2, P)) % P


def derive():
    x0 = poly_gcd_roots()
    y2 = (x0 * x0 % P * x0 + A * x0 + B) % P
    gx = (3 * x0 * x0 + A) % P
    v = 2 * gx % P
    u = 4 * y2 % P
    w = (u + x0 * v) % P
    assert (A - 5 * v) % P == 0, "Velu target a' != 0"
    assert (B - 7 * w) % P == 3645, "Velu target b' != 5*3^6"
    inv9 = pow(9, P - 2, P)
    inv27 = pow(27, P - 2, P)
    N = [(u - v * x0) % P, (x0 * x0 + v) % P, (-2 * x0) % P, 1]
    Np = [(x0 * x0 + v) % P, (-4 * x0) % P, 3]
    T = [(-x0 * Np[0]) % P, (Np[0] - x0 * Np[1]) % P, (Np[1] - x0 * Np[2]) % P, Np[2]]
    M = [(T[i] - 2 * N[i]) % P for i in range(4)]
    iso = [
        inv9, N[2] * inv9 % P, N[1] * inv9 % P, N[0] * inv9 % P,
        (-2 * x0) % P, (x0 * x0) % P,
        M[3] * inv27 % P, M[2] * inv27 % P, M[1] * inv27 % P, M[0] * inv27 % P,
        (-3 * x0) % P, (3 * x0 * x0) % P, (-pow(x0, 3, P)) % P,
    ]
    return iso


if __name__ == "__main__":
    iso = derive()
    ok = iso == hc.ISOGENY_CONSTANTS
    for i, v in enumerate(iso):
        print(f"{i:2d} 0x{v:064x}")
    print("matches hostcrypto.ISOGENY_CONSTANTS:", ok)
    sys.exit(0 if ok else 1)
