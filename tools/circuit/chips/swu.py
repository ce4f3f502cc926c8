"""Hash-to-curve circuit chips (taiga-owned sources, fully readable):
  MapToCurveConfig  — circuit/curve/iso_map.rs (simplified SWU, 2 rows)
  IsoMapConfig      — circuit/curve/map_to_curve.rs (3-isogeny, 2 rows)
  ToAffineConfig    — circuit/curve/to_affine.rs (Jacobian -> affine, 2 rows)
  hash_to_curve_circuit — circuit/hash_to_curve.rs (poseidon u_0/u_1 +
    map + iso + affine + complete add)

Constraint expressions are restated 1:1 from those files (they are in the
reference tree, unlike the halo2_gadgets chips); the isogeny constants are
the Velu-derived pinned set (hostcrypto.ISOGENY_CONSTANTS).
"""
from .pow5 import poseidon_hash_gadget
from .ecc import EccPoint
from .gadgets import bool_check
from .. import fields as F
from ..hostcrypto import (ISO_A, ISO_B, SWU_Z, THETA, ISOGENY_CONSTANTS,
                          POSEIDON_TO_FIELD_U_0_POSTFIX,
                          POSEIDON_TO_FIELD_U_1_POSTFIX)

TWO_INV = pow(2, F.P - 2, F.P)


def _ternary(f, a, b):
    return f * a + (1 - f) * b


class MapToCurveConfig:
    """10 columns: u, x, y, u_sgn0, u_other_bits, alpha, beta, gamma,
    delta, epsilon (iso_map.rs:30-231)."""

    def __init__(self, cs, cols):
        (self.u, self.x, self.y, self.u_sgn0, self.u_other, self.alpha,
         self.beta, self.gamma, self.delta, self.epsilon) = cols
        self.cs = cs
        cs.enable_equality(self.u)
        cs.enable_equality(self.x)
        cs.enable_equality(self.y)
        self.q = cs.selector()

        u = self.u.cur()
        alpha = self.alpha.cur()
        ta = self.alpha.next()
        beta = self.beta.cur()
        gx1_square = self.epsilon.next()
        sqrt_a = self.x.next()
        delta = self.delta.cur()
        sqrt_b = self.y.next()
        epsilon = self.epsilon.cur()
        num_x1 = self.beta.next()
        div = self.gamma.next()
        num_gx1 = self.delta.next()
        gamma = self.gamma.cur()
        u_sgn0 = self.u_sgn0.cur()
        u_other = self.u_other.cur()
        y_sgn0 = self.u_sgn0.next()
        y_other = self.u_other.next()
        x_jac = self.x.cur()
        y_jac = self.y.cur()
        z_jac = self.u.next()

        a = ISO_A
        b = ISO_B
        z = SWU_Z
        z_u2 = u.square() * z
        ta_poly = z_u2.square() + z_u2 - ta
        num_x1_poly = (ta + 1) * b - num_x1
        ta_is_zero = 1 - alpha * ta
        poly1 = ta * ta_is_zero
        div_poly = _ternary(ta_is_zero, 0 * u + z, 0 - ta) * a - div
        num2_x1 = num_x1.square()
        div2 = div.square()
        div3 = div2 * div
        num_gx1_poly = (num2_x1 + div2 * a) * num_x1 + div3 * b - num_gx1
        num_x2 = z_u2 * num_x1
        div3_is_zero = 1 - div3 * beta
        poly2 = div3 * div3_is_zero
        aa = beta * num_gx1
        bb = aa * F.ROOT_OF_UNITY
        num_gx1_is_zero = 1 - num_gx1 * gamma
        poly3 = num_gx1 * num_gx1_is_zero
        a_is_sqrt_value = aa - sqrt_a * sqrt_a
        a_is_sqrt = 1 - a_is_sqrt_value * delta
        poly4 = a_is_sqrt_value * a_is_sqrt
        b_is_sqrt_value = bb - sqrt_b * sqrt_b
        b_is_sqrt = 1 - b_is_sqrt_value * epsilon
        poly5 = b_is_sqrt_value * b_is_sqrt
        a_xor_b = a_is_sqrt + b_is_sqrt - a_is_sqrt * b_is_sqrt * 2
        poly6 = (num_gx1 * gamma) * (div3 * beta) * (1 - a_xor_b)
        gx1_square_poly = a_is_sqrt * (1 - (1 - num_gx1_is_zero) * div3_is_zero) - gx1_square
        y1 = _ternary(a_is_sqrt, sqrt_a, sqrt_b)
        y2 = y1 * z_u2 * u * THETA
        num_x = _ternary(gx1_square, num_x1, num_x2)
        y_sel = _ternary(gx1_square, y1, y2)
        u_check = u - (u_other * 2 + u_sgn0)
        y_check = y_sel - (y_other * 2 + y_sgn0)
        u_xor_y = u_sgn0 + y_sgn0 - u_sgn0 * y_sgn0 * 2
        poly7 = x_jac - num_x * div
        poly8 = y_jac - _ternary(u_xor_y, 0 - y_sel, y_sel) * div3
        poly9 = z_jac - div

        cs.create_gate("map to curve", self.q, [
            ("ta is zero", poly1),
            ("ta", ta_poly),
            ("num_x1", num_x1_poly),
            ("div", div_poly),
            ("div3 is zero", poly2),
            ("num_gx1", num_gx1_poly),
            ("num_gx1 is zero", poly3),
            ("a is sqrt", poly4),
            ("b is sqrt", poly5),
            ("gx1_square", gx1_square_poly),
            ("sqrt exists", poly6),
            ("bool u_sgn0", bool_check(u_sgn0)),
            ("bool y_sgn0", bool_check(y_sgn0)),
            ("u check", u_check),
            ("y check", y_check),
            ("x", poly7),
            ("y", poly8),
            ("z", poly9),
        ])

    def assign(self, u_cell):
        """2-row region; returns Jacobian (x, y, z) cells (on the ISO curve).
        Witness values mirror iso_map.rs:233-349 branch-free."""
        cs = self.cs
        z = SWU_Z
        a, b = ISO_A, ISO_B
        with cs.region("map_to_curve") as r:
            self.q.enable(r, 0)
            u = r.copy_advice(u_cell, self.u, 0)
            uv = u.reg
            z_u2 = uv * uv * z
            ta = z_u2 * z_u2 + z_u2
            alpha = ta.inv0()
            r.assign_advice(self.alpha, 0, alpha)
            r.assign_advice(self.alpha, 1, ta)
            ta_zero = ta.is_zero()
            div = ta_zero.select(cs.prog.const(z), 0 - ta) * a
            r.assign_advice(self.gamma, 1, div)
            div3 = div * div * div
            beta = div3.inv0()
            r.assign_advice(self.beta, 0, beta)
            num_x1 = (ta + 1) * b
            r.assign_advice(self.beta, 1, num_x1)
            num_gx1 = (num_x1 * num_x1 + div * div * a) * num_x1 + div3 * b
            r.assign_advice(self.delta, 1, num_gx1)
            gamma = num_gx1.inv0()
            r.assign_advice(self.gamma, 0, gamma)
            aa = div3.inv0() * num_gx1
            sqrt_a = aa.sqrt0()
            r.assign_advice(self.x, 1, sqrt_a)
            delta = (aa - sqrt_a * sqrt_a).inv0()
            r.assign_advice(self.delta, 0, delta)
            bb = aa * F.ROOT_OF_UNITY
            sqrt_b = bb.sqrt0()
            r.assign_advice(self.y, 1, sqrt_b)
            epsilon = (bb - sqrt_b * sqrt_b).inv0()
            r.assign_advice(self.epsilon, 0, epsilon)
            u_sgn0 = uv.bit(0)
            r.assign_advice(self.u_sgn0, 0, u_sgn0)
            r.assign_advice(self.u_other, 0, (uv - u_sgn0) * TWO_INV)
            a_is_sqrt = (aa - sqrt_a * sqrt_a).is_zero()
            num_gx1_is_zero = num_gx1.is_zero()
            div3_is_zero = div3.is_zero()
            gx1_square = a_is_sqrt * (1 - (1 - num_gx1_is_zero) * div3_is_zero)
            r.assign_advice(self.epsilon, 1, gx1_square)
            y1 = a_is_sqrt.select(sqrt_a, sqrt_b)
            y2 = y1 * z_u2 * uv * THETA
            num_x = gx1_square.select(num_x1, z_u2 * num_x1)
            y_sel = gx1_square.select(y1, y2)
            y_sgn0 = y_sel.bit(0)
            r.assign_advice(self.u_sgn0, 1, y_sgn0)
            r.assign_advice(self.u_other, 1, (y_sel - y_sgn0) * TWO_INV)
            u_xor_y = u_sgn0 + y_sgn0 - u_sgn0 * y_sgn0 * 2
            x_out = num_x * div
            y_out = u_xor_y.select(0 - y_sel, y_sel) * div3
            xc = r.assign_advice(self.x, 0, x_out)
            yc = r.assign_advice(self.y, 0, y_out)
            zc = r.assign_advice(self.u, 1, div)
        return xc, yc, zc


class IsoMapConfig:
    """3 columns x, y, z (map_to_curve.rs:19-121): 2-row region applying
    the 3-isogeny on Jacobian coords."""

    def __init__(self, cs, x, y, z):
        self.cs = cs
        self.x, self.y, self.z = x, y, z
        cs.enable_equality(x)
        cs.enable_equality(y)
        cs.enable_equality(z)
        self.q = cs.selector()
        iso = ISOGENY_CONSTANTS
        xq = x.cur()
        yq = y.cur()
        zq = z.cur()
        xo = x.next()
        yo = y.next()
        zo = z.next()
        z2 = zq.square()
        z3 = z2 * zq
        z4 = z2.square()
        z6 = z3.square()
        num_x = ((xq * iso[0] + z2 * iso[1]) * xq + z4 * iso[2]) * xq + z6 * iso[3]
        div_x = (z2 * xq + z4 * iso[4]) * xq + z6 * iso[5]
        num_y = (((xq * iso[6] + z2 * iso[7]) * xq + z4 * iso[8]) * xq + z6 * iso[9]) * yq
        div_y = (((xq + z2 * iso[10]) * xq + z4 * iso[11]) * xq + z6 * iso[12]) * z3
        cs.create_gate("iso map", self.q, [
            ("z", div_x * div_y - zo),
            ("x", num_x * div_y * zo - xo),
            ("y", num_y * div_x * zo.square() - yo),
        ])

    def assign(self, xc, yc, zc):
        cs = self.cs
        iso = ISOGENY_CONSTANTS
        with cs.region("iso map") as r:
            self.q.enable(r, 0)
            x = r.copy_advice(xc, self.x, 0).reg
            y = r.copy_advice(yc, self.y, 0).reg
            z = r.copy_advice(zc, self.z, 0).reg
            z2 = z * z
            z3 = z2 * z
            z4 = z2 * z2
            z6 = z3 * z3
            num_x = ((x * iso[0] + z2 * iso[1]) * x + z4 * iso[2]) * x + z6 * iso[3]
            div_x = (z2 * x + z4 * iso[4]) * x + z6 * iso[5]
            num_y = (((x * iso[6] + z2 * iso[7]) * x + z4 * iso[8]) * x + z6 * iso[9]) * y
            div_y = (((x + z2 * iso[10]) * x + z4 * iso[11]) * x + z6 * iso[12]) * z3
            zo = div_x * div_y
            xo = num_x * div_y * zo
            yo = num_y * div_x * zo * zo
            xoc = r.assign_advice(self.x, 1, xo)
            yoc = r.assign_advice(self.y, 1, yo)
            zoc = r.assign_advice(self.z, 1, zo)
        return xoc, yoc, zoc


class ToAffineConfig:
    """3 columns (to_affine.rs:24-130): Jacobian -> affine with identity
    mapped to (0,0)."""

    def __init__(self, cs, x, y, z):
        self.cs = cs
        self.x, self.y, self.z = x, y, z
        cs.enable_equality(x)
        cs.enable_equality(y)
        cs.enable_equality(z)
        self.q = cs.selector()
        xj = x.cur()
        yj = y.cur()
        zj = z.cur()
        xa = x.next()
        ya = y.next()
        zinv = z.next()
        z_is_zero = 1 - zj * zinv
        zinv2 = zinv.square()
        zinv3 = zinv2 * zinv
        cs.create_gate("to affine", self.q, [
            ("z is zero", zj * z_is_zero),
            ("x id", z_is_zero * xa),
            ("y id", z_is_zero * ya),
            ("x", z_is_zero * (xj * zinv2 - xa)),
            ("y", z_is_zero * (yj * zinv3 - ya)),
        ])

    def assign(self, xc, yc, zc):
        cs = self.cs
        with cs.region("to affine") as r:
            self.q.enable(r, 0)
            x = r.copy_advice(xc, self.x, 0).reg
            y = r.copy_advice(yc, self.y, 0).reg
            z = r.copy_advice(zc, self.z, 0).reg
            zinv = z.inv0()
            zi2 = zinv * zinv
            xa = x * zi2
            ya = y * zi2 * zinv
            xac = r.assign_advice(self.x, 1, xa)
            yac = r.assign_advice(self.y, 1, ya)
            r.assign_advice(self.z, 1, zinv)
        return EccPoint(xac, yac)


class HashToCurveConfig:
    """hash_to_curve.rs: poseidon u_0/u_1 + 2x(map + iso + affine) +
    complete add."""

    def __init__(self, cs, advices, poseidon_config):
        self.cs = cs
        self.adv = advices
        self.poseidon = poseidon_config
        self.map_to_curve = MapToCurveConfig(cs, advices)
        self.iso_map = IsoMapConfig(cs, advices[0], advices[1], advices[2])
        self.to_affine = ToAffineConfig(cs, advices[3], advices[4], advices[5])

    def hash_to_curve(self, ecc, message_cells):
        """hash_to_curve_circuit (2 messages + 1 postfix element = L 3)."""
        from ..plonkish import assign_free_constant
        cs = self.cs
        pts = []
        for postfix, col in ((POSEIDON_TO_FIELD_U_0_POSTFIX, self.adv[0]),
                             (POSEIDON_TO_FIELD_U_1_POSTFIX, self.adv[1])):
            post_cells = [assign_free_constant(cs, col, v) for v in postfix]
            u = poseidon_hash_gadget(self.poseidon, list(message_cells) + post_cells)
            q = self.map_to_curve.assign(u)
            rr = self.iso_map.assign(*q)
            pts.append(self.to_affine.assign(*rr))
        return ecc.add(pts[0], pts[1])


def ternary_gate_fix():  # pragma: no cover
    pass
