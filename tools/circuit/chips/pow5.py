"""Pow5 Poseidon chip (halo2_gadgets 0.3 poseidon/pow5.rs, un-vendored;
restated). WIDTH=3, RATE=2, P128Pow5T3 (R_F=8, R_P=56).

Layout (configure args mirror PoseidonChip::configure at
compliance_circuit.rs:117-123): state = 3 advice, partial_sbox = 1 advice,
rc_a/rc_b = 3+3 fixed (the round constants are ASSIGNED PER REGION ROW —
the rc columns carry values only on permute rows).

Gates:
  full round   (s_full):   for j: sum_i M[j][i]*(state_i + rc_a_i)^5 - state_j(next)
  partial pair (s_partial): two partial rounds per row via the witnessed
    mid_0 = (state_0 + rc_a_0)^5 and the inverse-MDS trick:
      (mid(0) + rc_b_0)^5 = (M^-1 next)(0);  mid(i) + rc_b_i = (M^-1 next)(i)
    with mid(i) = M[i][0]*mid_0 + M[i][1]*(state_1+rc_a_1) + M[i][2]*(state_2+rc_a_2)
  pad_and_add  (s_pad):    init_i(prev) + input_i(cur) - output_i(next) for
    the rate cells; init_2(prev) - output_2(next) for the capacity.

ConstantLength<L> sponge (primitives.rs): initial state [0,0,L<<64], chunks
of 2 zero-padded, squeeze = state[0] (matches tools/gen_poseidon.py hash_n,
pinned by the Grain fixture + oracle double-implementation).
"""
from ..plonkish import assign_free_constant
from ..hostcrypto import POS_RC, POS_MDS, _mds_inv
from .. import fields as F

RF, RP, T = 8, 56, 3


class Pow5Config:
    def __init__(self, cs, state, partial_sbox, rc_a, rc_b):
        self.cs = cs
        self.state = state
        self.partial_sbox = partial_sbox
        self.rc_a = rc_a
        self.rc_b = rc_b
        self.s_full = cs.selector()
        self.s_partial = cs.selector()
        self.s_pad = cs.selector()
        M = POS_MDS
        Minv = _mds_inv()

        def pow5(e):
            e2 = e * e
            return e2 * e2 * e

        cur = [state[i].cur() for i in range(T)]
        nxt = [state[i].next() for i in range(T)]
        prv = [state[i].prev() for i in range(T)]
        ra = [rc_a[i].cur() for i in range(T)]
        rb = [rc_b[i].cur() for i in range(T)]

        cs.create_gate("full round", self.s_full, [
            (f"state_{j}", sum(
                (pow5(cur[i] + ra[i]) * M[j][i] for i in range(T)),
                start=0 * cur[0],
            ) - nxt[j])
            for j in range(T)
        ])

        mid0 = partial_sbox.cur()

        def mid(i):
            return (mid0 * POS_MDS[i][0]
                    + (cur[1] + ra[1]) * POS_MDS[i][1]
                    + (cur[2] + ra[2]) * POS_MDS[i][2])

        def minv_next(i):
            return sum((nxt[j] * Minv[i][j] for j in range(T)), start=0 * cur[0])

        cs.create_gate("partial rounds", self.s_partial, [
            ("mid_0", pow5(cur[0] + ra[0]) - mid0),
            ("r_0", pow5(mid(0) + rb[0]) - minv_next(0)),
            ("r_1", (mid(1) + rb[1]) - minv_next(1)),
            ("r_2", (mid(2) + rb[2]) - minv_next(2)),
        ])

        cs.create_gate("pad and add", self.s_pad, [
            ("rate_0", prv[0] + cur[0] - nxt[0]),
            ("rate_1", prv[1] + cur[1] - nxt[1]),
            ("capacity", prv[2] - nxt[2]),
        ])


def _permute_region(cfg, init_cells):
    """One Poseidon permutation: 37-row region; returns final-state cells."""
    cs = cfg.cs
    M = POS_MDS

    def pow5v(v):
        v2 = v * v
        return v2 * v2 * v

    with cs.region("permute state") as r:
        cells = [r.copy_advice(c, cfg.state[i], 0) for i, c in enumerate(init_cells)]
        state = [c.reg for c in cells]
        row = 0
        rnd = 0
        for _ in range(RF // 2):
            cfg.s_full.enable(r, row)
            for i in range(T):
                r.assign_fixed(cfg.rc_a[i], row, POS_RC[rnd][i])
            sb = [pow5v(state[i] + POS_RC[rnd][i]) for i in range(T)]
            state = [sum((sb[j] * M[i][j] for j in range(T)), start=sb[0] * 0) for i in range(T)]
            row += 1
            rnd += 1
            cells = [r.assign_advice(cfg.state[i], row, state[i]) for i in range(T)]
        for _ in range(RP // 2):
            cfg.s_partial.enable(r, row)
            for i in range(T):
                r.assign_fixed(cfg.rc_a[i], row, POS_RC[rnd][i])
                r.assign_fixed(cfg.rc_b[i], row, POS_RC[rnd + 1][i])
            mid0 = pow5v(state[0] + POS_RC[rnd][0])
            r.assign_advice(cfg.partial_sbox, row, mid0)
            m = [mid0, state[1] + POS_RC[rnd][1], state[2] + POS_RC[rnd][2]]
            r1 = [sum((m[j] * M[i][j] for j in range(T)), start=m[0] * 0) for i in range(T)]
            m2 = [pow5v(r1[0] + POS_RC[rnd + 1][0]),
                  r1[1] + POS_RC[rnd + 1][1],
                  r1[2] + POS_RC[rnd + 1][2]]
            state = [sum((m2[j] * M[i][j] for j in range(T)), start=m2[0] * 0) for i in range(T)]
            row += 1
            rnd += 2
            cells = [r.assign_advice(cfg.state[i], row, state[i]) for i in range(T)]
        for _ in range(RF // 2):
            cfg.s_full.enable(r, row)
            for i in range(T):
                r.assign_fixed(cfg.rc_a[i], row, POS_RC[rnd][i])
            sb = [pow5v(state[i] + POS_RC[rnd][i]) for i in range(T)]
            state = [sum((sb[j] * M[i][j] for j in range(T)), start=sb[0] * 0) for i in range(T)]
            row += 1
            rnd += 1
            cells = [r.assign_advice(cfg.state[i], row, state[i]) for i in range(T)]
        assert rnd == RF + RP and row == 36
    return cells


def poseidon_hash_gadget(cfg, message_cells):
    """gadgets/poseidon_hash.rs poseidon_hash_gadget::<L> — returns the
    squeeze cell (state[0] after the final permute)."""
    cs = cfg.cs
    L = len(message_cells)
    cap = (L << 64) % F.P
    # initial state region
    with cs.region("poseidon init") as r:
        init = [
            r.assign_advice_from_constant(cfg.state[0], 0, 0),
            r.assign_advice_from_constant(cfg.state[1], 0, 0),
            r.assign_advice_from_constant(cfg.state[2], 0, cap),
        ]
    chunks = [message_cells[i:i + 2] for i in range(0, max(L, 1), 2)]
    state = init
    for chunk in chunks:
        with cs.region("pad and add") as r:
            cfg.s_pad.enable(r, 1)
            prev = [r.copy_advice(state[i], cfg.state[i], 0) for i in range(T)]
            inputs = []
            for i in range(2):
                if i < len(chunk):
                    inputs.append(r.copy_advice(chunk[i], cfg.state[i], 1))
                else:
                    inputs.append(r.assign_advice_from_constant(cfg.state[i], 1, 0))
            out = [
                r.assign_advice(cfg.state[0], 2, prev[0].reg + inputs[0].reg),
                r.assign_advice(cfg.state[1], 2, prev[1].reg + inputs[1].reg),
                r.assign_advice(cfg.state[2], 2, prev[2].reg + 0),
            ]
        state = _permute_region(cfg, out)
    return state[0]
