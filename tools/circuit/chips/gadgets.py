"""Small taiga gadget chips + halo2_gadgets CondSwap, restated.

Sources (fully readable in-reference; constraints copied by meaning, code
from scratch):
  - CondSwapChip: halo2_gadgets utilities/cond_swap.rs (un-vendored;
    restated: a' = swap?b:a, b' = swap?a:b, bool_check(swap))
  - ConditionalSelect / ConditionalEqual: circuit/gadgets/conditional_*.rs
  - Add/Sub/Mul: circuit/gadgets/{add,sub,mul}.rs
  - ExtendedOrRelation: circuit/gadgets/extended_or_relation.rs
  - ComposeIsEphemeralQuantity + resource_commit:
    circuit/resource_commitment.rs
"""
from .pow5 import poseidon_hash_gadget


def bool_check(e):
    return e * (e - 1)


class CondSwapConfig:
    """5 advice columns: a, b, a_swapped, b_swapped, swap (one row)."""

    def __init__(self, cs, advices5):
        self.cs = cs
        self.a, self.b, self.a_sw, self.b_sw, self.swap_col = advices5
        for c in advices5:
            cs.enable_equality(c)
        self.q_swap = cs.selector()
        a, b = self.a.cur(), self.b.cur()
        a_sw, b_sw = self.a_sw.cur(), self.b_sw.cur()
        sw = self.swap_col.cur()
        cs.create_gate("cond swap", self.q_swap, [
            ("a_swapped", a_sw - (sw * b + (1 - sw) * a)),
            ("b_swapped", b_sw - (sw * a + (1 - sw) * b)),
            ("swap bool", bool_check(sw)),
        ])

    def swap(self_cfg, pair_cell, pair_val_v, swap_v):
        """chip.swap(pair=(cell, value), swap) -> (a', b') cells."""
        cs = self_cfg.cs
        with cs.region("merkle swap") as r:
            self_cfg.q_swap.enable(r, 0)
            a = r.copy_advice(pair_cell, self_cfg.a, 0)
            b = r.assign_advice(self_cfg.b, 0, pair_val_v)
            a_sw = swap_v.select(b.reg, a.reg)
            b_sw = swap_v.select(a.reg, b.reg)
            ca = r.assign_advice(self_cfg.a_sw, 0, a_sw)
            cb = r.assign_advice(self_cfg.b_sw, 0, b_sw)
            r.assign_advice(self_cfg.swap_col, 0, swap_v)
        return ca, cb


class ConditionalSelectConfig:
    """conditional_select.rs: 2 advice cols, 2 rows:
    flag(a0,r0), ret(a0,r1), lhs(a1,r0), rhs(a1,r1);
    ret = flag*lhs + (1-flag)*rhs."""

    def __init__(self, cs, advices2):
        self.cs = cs
        self.adv = advices2
        self.q = cs.selector()
        flag = advices2[0].cur()
        ret = advices2[0].next()
        lhs = advices2[1].cur()
        rhs = advices2[1].next()
        cs.create_gate("conditional select", self.q,
                       [("select", flag * lhs + (1 - flag) * rhs - ret)])

    def assign(self, flag_cell, lhs_cell, rhs_cell):
        cs = self.cs
        with cs.region("conditional select") as r:
            self.q.enable(r, 0)
            f = r.copy_advice(flag_cell, self.adv[0], 0)
            lhs = r.copy_advice(lhs_cell, self.adv[1], 0)
            rhs = r.copy_advice(rhs_cell, self.adv[1], 1)
            ret = f.reg.select(lhs.reg, rhs.reg)
            return r.assign_advice(self.adv[0], 1, ret)


class ConditionalEqualConfig:
    """conditional_equal.rs: flag * (lhs - rhs) = 0, one row, 3 cols."""

    def __init__(self, cs, advices3):
        self.cs = cs
        self.adv = advices3
        self.q = cs.selector()
        flag, lhs, rhs = (c.cur() for c in advices3)
        cs.create_gate("conditional equal", self.q,
                       [("flag*(lhs-rhs)", flag * (lhs - rhs))])

    def assign(self, flag_cell, lhs_cell, rhs_cell):
        cs = self.cs
        with cs.region("conditional equal") as r:
            self.q.enable(r, 0)
            r.copy_advice(flag_cell, self.adv[0], 0)
            r.copy_advice(lhs_cell, self.adv[1], 0)
            r.copy_advice(rhs_cell, self.adv[2], 0)


class ArithConfig:
    """add.rs / sub.rs / mul.rs pattern: lhs(a0,r0) op rhs(a1,r0) =
    out(a0,r1)."""

    def __init__(self, cs, advices2, op):
        self.cs = cs
        self.adv = advices2
        self.op = op
        self.q = cs.selector()
        lhs = advices2[0].cur()
        rhs = advices2[1].cur()
        out = advices2[0].next()
        expr = {"add": lhs + rhs, "sub": lhs - rhs, "mul": lhs * rhs}[op]
        cs.create_gate(op, self.q, [(op, expr - out)])

    def assign(self, a_cell, b_cell):
        cs = self.cs
        with cs.region(self.op) as r:
            self.q.enable(r, 0)
            a = r.copy_advice(a_cell, self.adv[0], 0)
            b = r.copy_advice(b_cell, self.adv[1], 0)
            if self.op == "add":
                out = a.reg + b.reg
            elif self.op == "sub":
                out = a.reg - b.reg
            else:
                out = a.reg * b.reg
            return r.assign_advice(self.adv[0], 1, out)


class ExtendedOrRelationConfig:
    """extended_or_relation.rs: C==A or C==B over pairs, gated by flag."""

    def __init__(self, cs, advices3):
        self.cs = cs
        self.adv = advices3
        self.q = cs.selector()
        a1 = advices3[0].prev()
        a2 = advices3[1].prev()
        b1 = advices3[0].cur()
        b2 = advices3[1].cur()
        c1 = advices3[0].next()
        c2 = advices3[1].next()
        flag = advices3[2].cur()
        cs.create_gate("extended or relation", self.q, [
            ("(c1-a1)(c1-b1)", flag * (c1 - a1) * (c1 - b1)),
            ("(c2-a2)(c2-b2)", flag * (c2 - a2) * (c2 - b2)),
            ("(c1-a1)(c2-b2)", flag * (c1 - a1) * (c2 - b2)),
            ("(c1-b1)(c2-a2)", flag * (c1 - b1) * (c2 - a2)),
        ])


class ComposeQuantityConfig:
    """resource_commitment.rs ComposeIsEphemeralQuantity:
    compose = is_ephemeral * 2^128 + quantity, bool_check(is_ephemeral)."""

    def __init__(self, cs, col_l, col_m, col_r):
        self.cs = cs
        self.col_l, self.col_m, self.col_r = col_l, col_m, col_r
        self.q = cs.selector()
        compose = col_l.cur()
        is_eph = col_m.cur()
        quantity = col_r.cur()
        cs.create_gate("compose is_ephemeral and quantity", self.q, [
            ("bool is_ephemeral", bool_check(is_eph)),
            ("composition", compose - (quantity + is_eph * (1 << 128))),
        ])

    def assign(self, is_eph_cell, quantity_cell):
        cs = self.cs
        with cs.region("compose is_ephemeral and quantity") as r:
            self.q.enable(r, 0)
            compose = quantity_cell.reg + is_eph_cell.reg * (1 << 128)
            r.copy_advice(is_eph_cell, self.col_m, 0)
            r.copy_advice(quantity_cell, self.col_r, 0)
            return r.assign_advice(self.col_l, 0, compose)


class ResourceCommitConfig:
    """resource_commitment.rs ResourceCommitChip."""

    def __init__(self, cs, advices3, poseidon_config, lookup_config):
        self.compose = ComposeQuantityConfig(cs, advices3[0], advices3[1], advices3[2])
        self.poseidon = poseidon_config
        self.lookup = lookup_config

    def resource_commit(self, logic, label, value, npk, nonce, psi, quantity,
                        is_ephemeral, rcm):
        compose = self.compose.assign(is_ephemeral, quantity)
        return poseidon_hash_gadget(
            self.poseidon,
            [logic, label, value, npk, nonce, psi, compose, rcm],
        )


def merkle_poseidon_gadget(cond_swap_cfg, poseidon_cfg, leaf_cell, path):
    """merkle_circuit.rs merkle_poseidon_gadget: path = [(node_v, lr_v)]
    with node_v/lr_v traced values (lr = 1 when the sibling is LEFT)."""
    cur = leaf_cell
    for node_v, lr_v in path:
        a, b = cond_swap_cfg.swap(cur, node_v, lr_v)
        cur = poseidon_hash_gadget(poseidon_cfg, [a, b])
    return cur


def quantity_range_check(lookup_cfg, quantity_v):
    """integrity.rs:632-651: 6x10-bit running sum + 4-bit short check."""
    zs = lookup_cfg.witness_check(quantity_v, 6)
    lookup_cfg.copy_short_check(zs[6], 4)
    return zs[0]
