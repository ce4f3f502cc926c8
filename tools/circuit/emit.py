"""Finalize a synthesized circuit: selector compression, mock verification,
TGD2 circuit-description emission and TGW1 witness-program emission.

The desc blob uses the TGD1 format (tools/gen_cs1.py) unchanged — the
real circuits just use larger counts (95 constraints, 17+ fixed columns,
postfix stack depth up to 8), which the provers' capacities cover.
(Selector-gated skipping is NOT possible in the quotient evaluation: on
the extended coset the selector polynomials are dense.)

TGW1 blob (witness-synthesis program, interpreted by oracle/witness.c and
taiga_amd/csrc/witness.hpp at prove time):
  magic "TGW1", u32: n_inputs, n_consts, n_ops, n_stores, n_expose, k
  consts:  n_consts x 32B canonical Fp
  ops:     n_ops x { u8 opcode, u8 pad, u16 pad2, u32 a, u32 b }  (12 B)
           opcodes 0..9 = LOADI CONST ADD SUB MUL INV0 NEG SQRT0 BIT BYTE
           (plonkish.py W_* enums); a/b are reg indices except
           LOADI/CONST (a = input/const index) and BIT/BYTE (b = position)
  stores:  n_stores x { u32 col, u32 row, u32 reg }   (advice cells)
  expose:  n_expose x { u32 instance_row, u32 col, u32 row } — instance
           rows whose value the circuit computes (a copy constraint ties
           the instance row to advice (col,row)); the prove-time builder
           reads them from the synthesized advice instead of re-deriving
           the public inputs host-side. Rows not listed (the compliance
           anchor, the RL random padding) come from the witness blob.
Register i is defined by op i (SSA; single pass).
"""
from __future__ import annotations

import struct

from . import fields as F
from .plonkish import (ConstraintSystem, PostfixEmitter, subst_selectors,
                       collect_queries, Prod, expr_degree,
                       W_LOADI, W_CONST, W_ADD, W_SUB, W_MUL, W_INV0, W_NEG,
                       W_SQRT0, W_BIT, W_BYTE)


class Finalized:
    pass


def finalize(cs: ConstraintSystem, instance_vals_int, name="circuit"):
    """Compress selectors, compute degrees/queries, run the mock check.
    Returns a Finalized carrying everything emission needs."""
    out = Finalized()
    out.cs = cs
    cs.compress_selectors()
    mapping = cs._selector_exprs()
    out.degree = cs.degree()
    out.ext_k = cs.k + max(1, (out.degree - 1 - 1).bit_length())
    out.chunk_len = out.degree - 2
    out.bf = cs.blinding_factors()
    out.usable = cs.n - (out.bf + 1)

    # flattened constraints with substituted selectors
    out.constraints = []  # (name, expr, gating fixed col)
    for g in cs.gates:
        sel_expr = mapping[g.sel]
        gate_col = _gating_col(sel_expr)
        out.constraints.append(
            (f"{g.name}/{g.cname}", Prod(sel_expr, g.content), gate_col))
    out.lookups = []
    for lname, ins, tabs in cs.lookups:
        out.lookups.append((lname,
                            [subst_selectors(e, mapping) for e in ins],
                            [subst_selectors(e, mapping) for e in tabs]))

    # query sets, ordered: first use scanning constraints then lookups,
    # then permutation columns at cur
    aq, fq, iq = [], [], []

    def note(lst, key):
        if key not in lst:
            lst.append(key)

    tmp = {"advice": set(), "fixed": set(), "instance": set()}
    ordered = []
    for _, e, _ in out.constraints:
        _collect_ordered(e, ordered)
    for _, ins, tabs in out.lookups:
        for e in ins + tabs:
            _collect_ordered(e, ordered)
    for kind, col, rot in ordered:
        if kind == "advice":
            note(aq, (col, rot))
        elif kind == "fixed":
            note(fq, (col, rot))
        else:
            note(iq, (col, rot))
    for col in cs.eq_cols:
        if col.kind == "advice":
            note(aq, (col, 0))
        elif col.kind in ("fixed", "table"):
            note(fq, (col, 0))
        else:
            note(iq, (col, 0))
    out.advice_q, out.fixed_q, out.instance_q = aq, fq, iq
    out.instance_vals = [v % F.P for v in instance_vals_int]
    return out


def _collect_ordered(e, out):
    from .plonkish import FixedQ, AdviceQ, InstanceQ, Sum, Sub, Prod, Neg, Scaled
    if isinstance(e, FixedQ):
        out.append(("fixed", e.col, e.rot))
    elif isinstance(e, AdviceQ):
        out.append(("advice", e.col, e.rot))
    elif isinstance(e, InstanceQ):
        out.append(("instance", e.col, e.rot))
    elif isinstance(e, (Sum, Sub, Prod)):
        _collect_ordered(e.a, out)
        _collect_ordered(e.b, out)
    elif isinstance(e, (Neg, Scaled)):
        _collect_ordered(e.a, out)


def _gating_col(sel_expr):
    """The fixed column whose zero value kills the selector expression."""
    from .plonkish import FixedQ, Prod, Scaled
    e = sel_expr
    while isinstance(e, (Scaled,)):
        e = e.a
    while isinstance(e, Prod):
        e = e.a
        while isinstance(e, Scaled):
            e = e.a
    assert isinstance(e, FixedQ), sel_expr
    return e.col


# ------------------------------------------------------------- mock verify


def mock_verify(fin: Finalized, sample_rows=64):
    cs = fin.cs
    n = cs.n
    adv = [[0] * n for _ in cs.advice_cols]
    for ci, colvals in enumerate(cs.advice_vals):
        for row, reg in colvals.items():
            adv[ci][row] = reg.v
    fx = [[0] * n for _ in cs.fixed_cols]
    for ci, colvals in enumerate(cs.fixed_vals):
        for row, v in colvals.items():
            fx[ci][row] = v
    inst = [fin.instance_vals + [0] * (n - len(fin.instance_vals))]

    def ev(e, row):
        from .plonkish import (Const, FixedQ, AdviceQ, InstanceQ, Sum, Sub,
                               Prod, Neg, Scaled)
        if isinstance(e, Const):
            return e.v
        if isinstance(e, FixedQ):
            return fx[e.col.index][(row + e.rot) % n]
        if isinstance(e, AdviceQ):
            return adv[e.col.index][(row + e.rot) % n]
        if isinstance(e, InstanceQ):
            return inst[e.col.index][(row + e.rot) % n]
        if isinstance(e, Sum):
            return (ev(e.a, row) + ev(e.b, row)) % F.P
        if isinstance(e, Sub):
            return (ev(e.a, row) - ev(e.b, row)) % F.P
        if isinstance(e, Prod):
            va = ev(e.a, row)
            if va == 0:
                return 0
            return va * ev(e.b, row) % F.P
        if isinstance(e, Neg):
            return (-ev(e.a, row)) % F.P
        if isinstance(e, Scaled):
            return ev(e.a, row) * e.v % F.P
        raise TypeError(e)

    failures = []
    # gates: check on every row where the gating fixed column is nonzero,
    # plus a sample of other rows
    for cname, e, gcol in fin.constraints:
        rows = [r for r, v in cs.fixed_vals[gcol.index].items() if v]
        rows += list(range(0, n, max(1, n // sample_rows)))
        for r in set(rows):
            if ev(e, r) != 0:
                failures.append((cname, r))
                break
    # lookups: every usable row's input must be a table value
    for lname, ins, tabs in fin.lookups:
        assert len(ins) == 1 and len(tabs) == 1, "single-expr lookups only"
        tset = set()
        for r in range(n):
            tset.add(ev(tabs[0], r))
        for r in range(fin.usable):
            v = ev(ins[0], r)
            if v not in tset:
                failures.append((lname, r, v))
                break
    # copies
    def cellval(col, row):
        if col.kind == "advice":
            return adv[col.index][row]
        if col.kind == "instance":
            return inst[col.index][row]
        return fx[col.index][row]

    for (c1, r1), (c2, r2) in cs.copies:
        v1 = cellval(c1, r1)
        v2 = cellval(c2, r2)
        if v1 != v2:
            failures.append(("copy", c1, r1, c2, r2, v1, v2))
            if len(failures) > 10:
                break
    # row budget
    maxrow = max(
        [max(d.keys(), default=0) for d in cs.advice_vals] +
        [max(d.keys(), default=0) for d in cs.fixed_vals])
    if maxrow >= fin.usable:
        failures.append(("row budget", maxrow, fin.usable))
    return failures


# ------------------------------------------------------------- TGD2 blob


def emit_desc(fin: Finalized):
    cs = fin.cs
    n = cs.n
    const_tab = []
    cmap = {}

    def ci(v):
        v %= F.P
        if v not in cmap:
            cmap[v] = len(const_tab)
            const_tab.append(v)
        return cmap[v]

    def fidx(col):
        return col.index

    def aidx(col):
        return col.index

    def iidx(col):
        return col.index

    def emit_expr(e):
        em = PostfixEmitter(ci, fidx, aidx, iidx)
        em.emit(e)
        assert em.max_depth <= 8, f"stack depth {em.max_depth}"
        return em.ops

    gate_ops = []
    gate_gating = []
    for cname, e, gcol in fin.constraints:
        gate_ops.append(emit_expr(e))
        gate_gating.append((gcol.index, 0))
    lk_enc = []
    for lname, ins, tabs in fin.lookups:
        lk_enc.append(([emit_expr(e) for e in ins], [emit_expr(e) for e in tabs]))

    # permutation sigma from copies (halo2 Assembly::copy splicing)
    perm_cols = []
    for col in cs.eq_cols:
        kind = {"advice": 0, "fixed": 1, "table": 1, "instance": 2}[col.kind]
        perm_cols.append((kind, col.index, col))
    pc_index = {}
    for j, (_, _, col) in enumerate(perm_cols):
        pc_index[col] = j
    # sigma as (col, row) pairs; identity then spliced per copy
    sig = [[(j, i) for i in range(n)] for j in range(len(perm_cols))]
    for (c1, r1), (c2, r2) in cs.copies:
        j1, j2 = pc_index[c1], pc_index[c2]
        x, y = sig[j1][r1], sig[j2][r2]
        sig[j1][r1], sig[j2][r2] = y, x

    # fixed columns dense
    fixed_dense = []
    for cix in range(len(cs.fixed_cols)):
        colvals = cs.fixed_vals[cix]
        arr = bytearray(32 * n)
        for row, v in colvals.items():
            arr[row * 32:(row + 1) * 32] = (v % F.P).to_bytes(32, "little")
        fixed_dense.append(bytes(arr))

    out = b"TGD1"
    out += struct.pack(
        "<15I", cs.k, fin.ext_k, len(cs.fixed_cols), len(cs.advice_cols),
        len(cs.instance_cols), fin.bf, len(gate_ops), len(perm_cols),
        fin.chunk_len, len(fin.lookups), len(const_tab),
        len(fin.advice_q), len(fin.fixed_q), len(fin.instance_q),
        len(fin.instance_vals),
    )
    for v in const_tab:
        out += v.to_bytes(32, "little")
    for col, rot in fin.advice_q:
        out += struct.pack("<Ii", col.index, rot)
    for col, rot in fin.fixed_q:
        out += struct.pack("<Ii", col.index, rot)
    for col, rot in fin.instance_q:
        out += struct.pack("<Ii", col.index, rot)
    for kind, idx, _ in perm_cols:
        out += struct.pack("<II", kind, idx)
    for ops in gate_ops:
        out += struct.pack("<I", len(ops))
        for tag, a, b in ops:
            out += struct.pack("<IIi", tag, a, b)
    for ins, tabs in lk_enc:
        out += struct.pack("<II", len(ins), len(tabs))
        for ops in ins + tabs:
            out += struct.pack("<I", len(ops))
            for tag, a, b in ops:
                out += struct.pack("<IIi", tag, a, b)
    for j in range(len(perm_cols)):
        row = bytearray()
        for i in range(n):
            cj, ri = sig[j][i]
            row += struct.pack("<II", cj, ri)
        out += bytes(row)
    for blob in fixed_dense:
        out += blob
    return out


# ------------------------------------------------------------- TGW1 blob


def emit_witness_program(cs: ConstraintSystem):
    prog = cs.prog
    stores = []
    for ci, colvals in enumerate(cs.advice_vals):
        for row, reg in sorted(colvals.items()):
            stores.append((ci, row, reg.r))
    expose = {}
    for (c1, r1), (c2, r2) in cs.copies:
        if c1.kind == "instance" and c2.kind == "advice" and r1 not in expose:
            expose[r1] = (c2.index, r2)
        elif c2.kind == "instance" and c1.kind == "advice" and r2 not in expose:
            expose[r2] = (c1.index, r1)
    out = b"TGW1"
    out += struct.pack("<6I", prog.n_inputs, len(prog.consts), len(prog.ops),
                       len(stores), len(expose), cs.k)
    for v in prog.consts:
        out += v.to_bytes(32, "little")
    for (op, a, b) in prog.ops:
        out += struct.pack("<BBHII", op, 0, 0, a & 0xFFFFFFFF, b & 0xFFFFFFFF)
    for (col, row, reg) in stores:
        out += struct.pack("<III", col, row, reg)
    for irow in sorted(expose):
        col, row = expose[irow]
        out += struct.pack("<III", irow, col, row)
    return out
