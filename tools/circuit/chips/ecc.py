"""ECC chip over Pallas (halo2_gadgets 0.3 ecc/chip/*, un-vendored).

Restated from the public halo2 book's documented constraint systems
("Elliptic curves" chapters: witness point, incomplete/complete addition,
variable-base scalar mul, fixed-base windowed mul). Where the book fixes
the algebra, the constraints below are verbatim restatements; layout
choices (column roles, region shapes) are our own. DOCUMENTED DEVIATIONS
(DESIGN.md §circuit-fidelity):
  - variable-base mul runs one 249-round incomplete chain + 3 complete
    rounds + LSB (Orchard splits hi/lo halves across column sets to halve
    region height; same algorithm, same formulas);
  - the scalar-overflow check is the simplified sound form z_0 = alpha
    with bits 254/253 forced to zero (complete for alpha < 2^253; the
    compliance circuit only muls 64-bit quantities — Orchard's
    2^130-decomposition check admits full-range alpha);
  - fixed-base mul accumulates window points with incomplete adds and a
    final complete add; the window tables, the per-window Lagrange
    x-interpolation columns, the z/u y-pin and the [(k+2)*8^w]B convention
    are EXACTLY the reference's (pinned byte-for-byte against constant.rs
    R_U/R_Z + GENERATOR_U/GENERATOR_Z — tests/test_fixed_base_tables.py).

The unused-by-compliance mul_fixed variants (base-field-element, short)
contribute gates to the reference CS; restated gate shells are included
(never enabled) so selector compression sees the same degree classes.
"""
from ..plonkish import assign_free_advice, assign_free_constant
from .. import fields as F
from ..hostcrypto import compute_window_table, find_zs_and_us

TWO_INV = pow(2, F.P - 2, F.P)
T_Q = F.Q - (1 << 254)  # q = 2^254 + t_q


class EccPoint:
    """(x, y) cells; identity = (0, 0)."""

    def __init__(self, x_cell, y_cell):
        self.x = x_cell
        self.y = y_cell


class EccConfig:
    def __init__(self, cs, advices, lagrange_coeffs, lookup_config):
        self.cs = cs
        self.adv = advices
        self.lagrange = lagrange_coeffs
        self.lookup = lookup_config
        self.fixed_z = cs.fixed_column()

        a = advices

        # --- witness point (book: "Witnessing points") ---
        self.q_point = cs.selector()
        self.q_point_non_id = cs.selector()
        x, y = a[0].cur(), a[1].cur()
        on_curve = y * y - x * x * x - 5
        cs.create_gate("witness point", self.q_point, [
            ("x on-curve-or-id", x * on_curve),
            ("y on-curve-or-id", y * on_curve),
        ])
        cs.create_gate("witness non-identity point", self.q_point_non_id,
                       [("on curve", on_curve)])

        # --- incomplete addition (book: "Incomplete addition") ---
        # x_p=a0, y_p=a1, x_q=a2 cur / x_r=a2 next, y_q=a3 cur / y_r=a3 next
        self.q_add_incomplete = cs.selector()
        x_p, y_p = a[0].cur(), a[1].cur()
        x_q, y_q = a[2].cur(), a[3].cur()
        x_r, y_r = a[2].next(), a[3].next()
        cs.create_gate("incomplete addition", self.q_add_incomplete, [
            ("x_r", (x_r + x_q + x_p) * (x_p - x_q) * (x_p - x_q) - (y_p - y_q) * (y_p - y_q)),
            ("y_r", (y_r + y_q) * (x_p - x_q) - (y_p - y_q) * (x_q - x_r)),
        ])

        # --- complete addition (book: "Complete addition") ---
        # x_p=a0, y_p=a1, x_q=a2 cur/x_r next, y_q=a3 cur/y_r next,
        # lambda=a4, alpha=a5, beta=a6, gamma=a7, delta=a8
        self.q_add = cs.selector()
        lam = a[4].cur()
        alpha = a[5].cur()
        beta = a[6].cur()
        gamma = a[7].cur()
        delta = a[8].cur()
        xq_m_xp = x_q - x_p
        yq_p_yp = y_q + y_p
        if_alpha = 1 - xq_m_xp * alpha
        if_beta = 1 - x_p * beta
        if_gamma = 1 - x_q * gamma
        if_id = 1 - xq_m_xp * alpha - yq_p_yp * delta
        slope_ok = lam * lam - x_p - x_q - x_r
        yslope_ok = lam * (x_p - x_r) - y_p - y_r
        cs.create_gate("complete addition", self.q_add, [
            ("1", xq_m_xp * (xq_m_xp * lam - (y_q - y_p))),
            ("2", if_alpha * (lam * y_p * 2 - x_p * x_p * 3)),
            ("3", x_p * x_q * xq_m_xp * slope_ok),
            ("4", x_p * x_q * xq_m_xp * yslope_ok),
            ("5", x_p * x_q * yq_p_yp * slope_ok),
            ("6", x_p * x_q * yq_p_yp * yslope_ok),
            ("7", if_beta * (x_r - x_q)),
            ("8", if_beta * (y_r - y_q)),
            ("9", if_gamma * (x_r - x_p)),
            ("10", if_gamma * (y_r - y_p)),
            ("11", if_id * x_r),
            ("12", if_id * y_r),
        ])

        # --- variable-base mul: incomplete rounds ---
        # region columns: x_T=a0, y_T=a1, z=a2, x_A=a3, l1=a4, l2=a5, y_w=a6
        self.q_mul_round = cs.selector()
        self.q_mul_last = cs.selector()
        self.q_mul_first = cs.selector()
        x_T, y_T = a[0].cur(), a[1].cur()
        z_cur, z_prev = a[2].cur(), a[2].prev()
        x_A, x_A_next = a[3].cur(), a[3].next()
        l1, l2 = a[4].cur(), a[5].cur()
        l1n, l2n = a[4].next(), a[5].next()
        y_w_prev, y_w_next = a[6].prev(), a[6].next()
        k_bit = z_cur - z_prev * 2

        def y_a(xa, lam1, lam2, xt):
            x_r = lam1 * lam1 - xa - xt
            return (lam1 + lam2) * (xa - x_r) * TWO_INV

        x_R = l1 * l1 - x_A - x_T
        y_A_cur = y_a(x_A, l1, l2, x_T)
        y_A_next = y_a(x_A_next, l1n, l2n, x_T)
        cs.create_gate("var mul round", self.q_mul_round, [
            ("k bool", k_bit * (k_bit - 1)),
            ("l1", l1 * (x_A - x_T) - (y_A_cur - (k_bit * 2 - 1) * y_T)),
            ("x_A next", x_A_next - (l2 * l2 - x_A - x_R)),
        ])
        self.q_mul_chain = cs.selector()
        cs.create_gate("var mul chain", self.q_mul_chain, [
            ("y chain", l2 * (x_A - x_A_next) - (y_A_cur + y_A_next)),
        ])
        cs.create_gate("var mul first", self.q_mul_first, [
            ("y init", y_A_cur - y_w_prev),
        ])
        cs.create_gate("var mul last", self.q_mul_last, [
            ("y final", l2 * (x_A - x_A_next) - (y_A_cur + y_w_next)),
        ])

        # --- variable-base mul: low-bit decomposition + point selection ---
        # z continuation: z col = a2; k col = a3
        self.q_mul_dec = cs.selector()
        k_cur = a[3].cur()
        cs.create_gate("var mul low decompose", self.q_mul_dec, [
            ("z", z_cur - (z_prev * 2 + k_cur)),
            ("k bool", k_cur * (k_cur - 1)),
        ])
        # scalar binding: the 255-slot chain witnesses k = alpha + t_q
        # (so [2^254 + k]T = [q + alpha]T = [alpha]T); alpha copied to a4
        # on the binding row. Unique k given k < 2^254 (bit 254 forced 0).
        self.q_mul_bind = cs.selector()
        cs.create_gate("var mul scalar bind", self.q_mul_bind, [
            ("z = alpha + t_q", z_cur - a[4].cur() - T_Q),
        ])
        # bit point: x_P=a0 (=x_T copied), y_T=a1, k=a2, x_Pout=a3, y_Pout=a4
        self.q_mul_bitpt = cs.selector()
        kb = a[2].cur()
        x_po, y_po = a[3].cur(), a[4].cur()
        cs.create_gate("var mul bit point", self.q_mul_bitpt, [
            ("x", x_po - a[0].cur()),
            ("y", y_po - (kb * 2 - 1) * a[1].cur()),
        ])
        # lsb point: Q = k0 ? identity : -T
        self.q_mul_lsb = cs.selector()
        cs.create_gate("var mul lsb point", self.q_mul_lsb, [
            ("x", x_po - (1 - kb) * a[0].cur()),
            ("y", y_po + (1 - kb) * a[1].cur()),
        ])

        # --- fixed-base mul (full-width windows) ---
        # row: window k=a0, x_p=a1, y_p=a2, u=a3; fixed: lagrange[0..8), fixed_z
        self.q_mul_fixed = cs.selector()
        kw = a[0].cur()
        xp_f, yp_f, u_f = a[1].cur(), a[2].cur(), a[3].cur()
        rng = kw
        for b in range(1, 8):
            rng = rng * (kw - b)
        interp = lagrange_coeffs[7].cur()
        for j in range(6, -1, -1):
            interp = interp * kw + lagrange_coeffs[j].cur()
        cs.create_gate("fixed-base mul window", self.q_mul_fixed, [
            ("window range", rng),
            ("x interp", xp_f - interp),
            ("u pins y", u_f * u_f - (yp_f + self.fixed_z.cur())),
        ])

        # --- mul_fixed base-field / short variants: gate shells only.
        # Restated degree classes of halo2_gadgets' canonicity/sign gates;
        # NEVER enabled (compliance/RL witnesses don't use these paths) —
        # present so the selector-compression input matches the reference's
        # configure() selector inventory.
        self.q_mul_fixed_base_field = cs.selector()
        a0, a1c, a2c = a[0].cur(), a[1].cur(), a[2].cur()
        cs.create_gate("fixed-base mul base-field canonicity",
                       self.q_mul_fixed_base_field, [
                           ("canon hi", a0 * (a1c - 1) * a2c * a2c),
                           ("canon lo", a0 * a1c * (a2c - 1)),
                       ])
        self.q_mul_fixed_short = cs.selector()
        cs.create_gate("fixed-base mul short", self.q_mul_fixed_short, [
            ("sign bool", (a1c - 1) * (a1c + 1)),
            ("y flip", a2c - a1c * a0),
        ])
        self.q_range_check_win = cs.selector()
        word3 = z_cur - a[2].next() * 8
        rng3 = word3
        for b in range(1, 8):
            rng3 = rng3 * (word3 - b)
        cs.create_gate("running-sum 3-bit window range", self.q_range_check_win,
                       [("range", rng3)])

    # ------------------------------------------------------------ witness

    def witness_point(self, x_v, y_v, non_identity):
        cs = self.cs
        with cs.region("witness point") as r:
            (self.q_point_non_id if non_identity else self.q_point).enable(r, 0)
            xc = r.assign_advice(self.adv[0], 0, x_v)
            yc = r.assign_advice(self.adv[1], 0, y_v)
        return EccPoint(xc, yc)

    def add_incomplete(self, p: EccPoint, q: EccPoint):
        cs = self.cs
        with cs.region("incomplete add") as r:
            self.q_add_incomplete.enable(r, 0)
            xp = r.copy_advice(p.x, self.adv[0], 0)
            yp = r.copy_advice(p.y, self.adv[1], 0)
            xq = r.copy_advice(q.x, self.adv[2], 0)
            yq = r.copy_advice(q.y, self.adv[3], 0)
            lam = (yq.reg - yp.reg) * (xq.reg - xp.reg).inv0()
            xr = lam * lam - xp.reg - xq.reg
            yr = lam * (xq.reg - xr) - yq.reg
            xrc = r.assign_advice(self.adv[2], 1, xr)
            yrc = r.assign_advice(self.adv[3], 1, yr)
        return EccPoint(xrc, yrc)

    def add(self, p: EccPoint, q: EccPoint):
        """Complete addition; handles identities, doubling, inverses."""
        cs = self.cs
        with cs.region("complete add") as r:
            self.q_add.enable(r, 0)
            xp = r.copy_advice(p.x, self.adv[0], 0)
            yp = r.copy_advice(p.y, self.adv[1], 0)
            xq = r.copy_advice(q.x, self.adv[2], 0)
            yq = r.copy_advice(q.y, self.adv[3], 0)
            xpv, ypv, xqv, yqv = xp.reg, yp.reg, xq.reg, yq.reg
            dx = xqv - xpv
            alpha = dx.inv0()
            beta = xpv.inv0()
            gamma = xqv.inv0()
            eq = dx.is_zero()
            sy = yqv + ypv
            delta = eq * sy.inv0()
            lam_add = (yqv - ypv) * alpha
            lam_dbl = (xpv * xpv * 3) * (ypv + ypv).inv0()
            lam = eq.select(lam_dbl, lam_add)
            xr_g = lam * lam - xpv - xqv
            yr_g = lam * (xpv - xr_g) - ypv
            zp = xpv.is_zero()
            zq = xqv.is_zero()
            opp = eq * sy.is_zero()
            xr = zp.select(xqv, zq.select(xpv, opp.select(0 * xpv, xr_g)))
            yr = zp.select(yqv, zq.select(ypv, opp.select(0 * xpv, yr_g)))
            r.assign_advice(self.adv[4], 0, lam)
            r.assign_advice(self.adv[5], 0, alpha)
            r.assign_advice(self.adv[6], 0, beta)
            r.assign_advice(self.adv[7], 0, gamma)
            r.assign_advice(self.adv[8], 0, delta)
            xrc = r.assign_advice(self.adv[2], 1, xr)
            yrc = r.assign_advice(self.adv[3], 1, yr)
        return EccPoint(xrc, yrc)

    def mul_var_base(self, alpha_cell, base: EccPoint):
        """[alpha]T for a base-field alpha cell.
        Double-and-add (book algorithm): acc = [2]T; witness the 255 bits
        of k = alpha + t_q with k_254 forced to 0; per bit i from 254 down
        to 1: acc = (acc + P_i) + acc with P_i = k_i ? T : -T (incomplete
        adds for bits 254..4, complete for 3..1); if k_0 = 0 the final acc
        adds -T. Result = [2^254 + k]T = [q + alpha]T = [alpha]T.
        Binding: z_0 = alpha + t_q (gate) with k < 2^254 making k unique.
        Complete for alpha < 2^254 - t_q (the compliance path muls 64-bit
        quantities)."""
        cs = self.cs
        a = self.adv
        alpha_v = alpha_cell.reg
        scalar_v = alpha_v + T_Q  # k as an integer (< p for our range)
        # init acc = [2]T via complete add (handles the doubling branch)
        acc = self.add(base, base)
        NR = 251  # bits 254..4 incomplete (k_254 forced 0)
        with cs.region("var-base mul incomplete") as r:
            # row 0: carrier row (z = 0 const, y_w = init acc y copy)
            z_c = r.assign_advice_from_constant(a[2], 0, 0)
            r.copy_advice(acc.y, a[6], 0)
            x_av = acc.x.reg
            y_av = acc.y.reg
            z_v = z_c.reg
            zero_cell = z_c
            for j in range(NR):
                bit = 254 - j
                row = 1 + j
                self.q_mul_round.enable(r, row)
                if j == 0:
                    self.q_mul_first.enable(r, row)
                    r.copy_advice(acc.x, a[3], row)
                else:
                    r.assign_advice(a[3], row, x_av)
                if j < NR - 1:
                    self.q_mul_chain.enable(r, row)
                else:
                    self.q_mul_last.enable(r, row)
                k = scalar_v.bit(bit)
                z_v = z_v * 2 + k
                r.copy_advice(base.x, a[0], row)
                r.copy_advice(base.y, a[1], row)
                zc_round = r.assign_advice(a[2], row, z_v)
                if j == 0:
                    # k_254 = 0: force the first z to zero (copy to const)
                    cs._pending_eq.append((zc_round, zero_cell))
                # round: R = acc + P (P = k ? T : -T), acc' = R + acc
                y_p = (k * 2 - 1) * base.y.reg
                l1 = (y_av - y_p) * (x_av - base.x.reg).inv0()
                x_r = l1 * l1 - x_av - base.x.reg
                l2 = (y_av + y_av) * (x_av - x_r).inv0() - l1
                x_next = l2 * l2 - x_av - x_r
                y_next = l2 * (x_av - x_next) - y_av
                r.assign_advice(a[4], row, l1)
                r.assign_advice(a[5], row, l2)
                x_av, y_av = x_next, y_next
            # final row
            frow = NR + 1
            xa_f = r.assign_advice(a[3], frow, x_av)
            yw_f = r.assign_advice(a[6], frow, y_av)
            z_after4 = z_v
            z_f = r.assign_advice(a[2], frow, z_v)
        acc = EccPoint(xa_f, yw_f)
        # low bits 3..0: running-sum continuation (z col a2, k col a3),
        # with the scalar-binding gate on the last row (alpha copied to a4)
        with cs.region("var-base mul low bits") as r:
            r.copy_advice(z_f, a[2], 0)
            z_v = z_after4
            k_cells = []
            for i, bit in enumerate([3, 2, 1, 0]):
                self.q_mul_dec.enable(r, 1 + i)
                k = scalar_v.bit(bit)
                z_v = z_v * 2 + k
                k_cells.append(r.assign_advice(a[3], 1 + i, k))
                zc = r.assign_advice(a[2], 1 + i, z_v)
            self.q_mul_bind.enable(r, 4)
            r.copy_advice(alpha_cell, a[4], 4)
        # complete rounds for bits 3,2,1: acc = (acc + P_i) + acc
        for i in range(3):
            with cs.region("var mul bit point") as r:
                self.q_mul_bitpt.enable(r, 0)
                r.copy_advice(base.x, a[0], 0)
                r.copy_advice(base.y, a[1], 0)
                kc = r.copy_advice(k_cells[i], a[2], 0)
                k = kc.reg
                xpo = r.assign_advice(a[3], 0, base.x.reg * 1)
                ypo = r.assign_advice(a[4], 0, (k * 2 - 1) * base.y.reg)
            pt = EccPoint(xpo, ypo)
            tmp = self.add(acc, pt)
            acc = self.add(tmp, acc)
        # LSB: Q = k0 ? identity : -T; result = acc + Q
        with cs.region("var mul lsb point") as r:
            self.q_mul_lsb.enable(r, 0)
            r.copy_advice(base.x, a[0], 0)
            r.copy_advice(base.y, a[1], 0)
            kc = r.copy_advice(k_cells[3], a[2], 0)
            k = kc.reg
            xpo = r.assign_advice(a[3], 0, (1 - k) * base.x.reg)
            ypo = r.assign_advice(a[4], 0, (0 - (1 - k)) * base.y.reg)
        q_pt = EccPoint(xpo, ypo)
        return self.add(acc, q_pt)

    def mul_fixed_full(self, scalar_byte_vs, base_name, base_xy):
        """[scalar]B for a fixed base with full-width Fq scalar given as 32
        repr-byte traced values. Window tables/z/u exactly per the pinned
        reference convention. Returns the result EccPoint."""
        cs = self.cs
        a = self.adv
        table = _fixed_base_cache(base_name, base_xy)
        win_pts = []
        with cs.region(f"fixed-base mul {base_name}") as r:
            for w in range(85):
                self.q_mul_fixed.enable(r, w)
                coeffs = table["coeffs"][w]
                z = table["z"][w]
                for j in range(8):
                    r.assign_fixed(self.lagrange[j], w, coeffs[j])
                r.assign_fixed(self.fixed_z, w, z)
                # window value from scalar repr bits (3-bit LE windows)
                b0 = scalar_byte_vs[(3 * w) // 8].bit((3 * w) % 8)
                b1 = scalar_byte_vs[(3 * w + 1) // 8].bit((3 * w + 1) % 8) if 3 * w + 1 < 256 else None
                b2 = scalar_byte_vs[(3 * w + 2) // 8].bit((3 * w + 2) % 8) if 3 * w + 2 < 256 else None
                kv = b0
                if b1 is not None:
                    kv = kv + b1 * 2
                if b2 is not None:
                    kv = kv + b2 * 4
                r.assign_advice(a[0], w, kv)
                # x_p by Lagrange interpolation over the window value
                xv = cs.prog.const(coeffs[7])
                for j in range(6, -1, -1):
                    xv = xv * kv + coeffs[j]
                # y_p: the root of x^3+5 with (z + y) square
                yy = (xv * xv * xv + 5).sqrt0()
                s = (yy + z).sqrt0()
                flag = (s * s - (yy + z)).is_zero()
                yv = flag.select(yy, 0 - yy)
                uv = (yv + z).sqrt0()
                xc = r.assign_advice(a[1], w, xv)
                yc = r.assign_advice(a[2], w, yv)
                r.assign_advice(a[3], w, uv)
                win_pts.append(EccPoint(xc, yc))
        acc = win_pts[0]
        for w in range(1, 84):
            acc = self.add_incomplete(acc, win_pts[w])
        return self.add(acc, win_pts[84])


_FIXED_BASE_CACHE = {}


def _fixed_base_cache(name, base_xy):
    """Per-base window data: z values, u arrays, Lagrange x-interpolation
    coefficients (the fixed-column data of the mul_fixed rows). The z/u
    values come from the byte-pinned golden fixtures when available
    (tests/test_fixed_base_tables.py validates them against recomputation
    — re-searching z here would cost ~10 min for nothing)."""
    if name in _FIXED_BASE_CACHE:
        return _FIXED_BASE_CACHE[name]
    import os
    import struct
    tab = compute_window_table(base_xy)
    fixture = {
        "resource_commit_r": "fixed_base_resource_commit_r.bin",
        "generator": "fixed_base_generator.bin",
    }.get(name)
    path = None
    if fixture:
        path = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                            "..", "..", "..", "tests", "golden", fixture)
    if path and os.path.exists(path):
        blob = open(path, "rb").read()
        zs, us = [], []
        for w in range(85):
            off = w * 264
            (z,) = struct.unpack("<Q", blob[off:off + 8])
            zs.append(z)
            uu = [int.from_bytes(blob[off + 8 + k * 32: off + 8 + (k + 1) * 32],
                                 "little") for k in range(8)]
            us.append(uu)
            for k in range(8):
                assert (uu[k] * uu[k] - z - tab[w][k][1]) % F.P == 0, \
                    (name, w, k, "fixture/table mismatch")
    else:
        zs_us = find_zs_and_us(base_xy)
        zs = [zu[0] for zu in zs_us]
        us = [zu[1] for zu in zs_us]
    coeffs = []
    for w in range(85):
        xs = [tab[w][k][0] for k in range(8)]
        coeffs.append(_lagrange_coeffs_8(xs))
    data = {"table": tab, "z": zs, "u": us, "coeffs": coeffs}
    _FIXED_BASE_CACHE[name] = data
    return data


def _lagrange_coeffs_8(ys):
    """Coefficients c_0..c_7 of the unique degree-7 polynomial with
    p(k) = ys[k] for k = 0..7 (over Fp)."""
    P = F.P
    # Newton -> monomial (small fixed size; direct Lagrange accumulation)
    coeffs = [0] * 8
    for k in range(8):
        # basis poly prod_{j != k} (x - j) / (k - j)
        denom = 1
        poly = [1]
        for j in range(8):
            if j == k:
                continue
            denom = denom * (k - j) % P
            new = [0] * (len(poly) + 1)
            for i, c in enumerate(poly):
                new[i] = (new[i] - j * c) % P
                new[i + 1] = (new[i + 1] + c) % P
            poly = new
        dinv = pow(denom, P - 2, P)
        for i in range(8):
            coeffs[i] = (coeffs[i] + ys[k] * dinv % P * poly[i]) % P
    return coeffs
