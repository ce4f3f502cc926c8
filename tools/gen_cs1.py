#!/usr/bin/env python3
"""Generates tests/golden/cs1.desc — the round-1 "compliance-shaped" circuit
description blob parsed by BOTH the CPU oracle (oracle/prover.c) and the
product prover (taiga_amd/csrc), so the static circuit data has a single
source of truth.

CS1 mirrors the reference compliance circuit's SHAPE (k=15,
n=2^15 rows, 10 advice + 1 instance columns, fixed columns incl. a 2^10
lookup table, ~12 equality-enabled columns in 2 permutation chunks, max
gate degree 9 -> extended domain 2^18), exercising every prover stage
(custom gates with ±1 rotations, instance exposure via copies, one lookup,
two permutation chunks). Exact witness/constraint fidelity to
compliance_circuit.rs is the round-2 work item; proofs
over CS1 are parity-checked GPU-vs-oracle bit-for-bit.

Blob format (little-endian):
  magic "TGD1", u32 each: k, ext_k, n_fixed, n_advice, n_instance, bf,
  n_gates, n_perm_cols, chunk_len, n_lookups, n_consts,
  n_advice_q, n_fixed_q, n_instance_q, n_instance_rows
  consts:       n_consts * 32B (canonical Fp)
  advice_q:     n_advice_q * (u32 col, i32 rot)
  fixed_q, instance_q likewise
  perm_cols:    n_perm_cols * (u32 kind, u32 idx)    kind 0=advice 1=fixed 2=instance
  gates:        n_gates * { u32 n_ops; n_ops * (u32 tag, u32 a, i32 b) }
  lookups:      n_lookups * { u32 n_in, u32 n_tab; exprs... } (same expr format)
  sigma map:    n_perm_cols * n * (u32 col', u32 row')  (column index in
                perm-column order)
  fixed values: n_fixed * n * 32B (canonical Fp)

Expression ops (postfix stack machine):
  0 CONST(a=const idx)  1 FIXED(a=col,b=rot)  2 ADVICE  3 INSTANCE
  4 ADD  5 SUB  6 MUL  7 NEG  8 SCALE(a=const idx)
"""
import os
import struct
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", "oracle"))
import pypasta as pp  # noqa: E402

K = 15
N = 1 << K
EXT_K = 18
N_FIXED = 5  # q0 q1 q2 f_table f_c
N_ADVICE = 10
N_INSTANCE = 1
BF = 5  # blinding factors; usable = N - (BF+1)
CHUNK = 7  # degree(9) - 2
N_INSTANCE_ROWS = 9  # public-input rows (compliance has 9 — constant.rs:54-62)
USABLE = N - (BF + 1)

Q0, Q1, Q2, FTAB, FC = range(5)
REGION = 8192

OP_CONST, OP_FIXED, OP_ADVICE, OP_INSTANCE, OP_ADD, OP_SUB, OP_MUL, OP_NEG, OP_SCALE = range(9)


def expr(ops):
    return ops


def enc_expr(ops):
    out = struct.pack("<I", len(ops))
    for tag, a, b in ops:
        out += struct.pack("<IIi", tag, a, b)
    return out


def main():
    # ---- gates ----
    A = lambda c, r=0: (OP_ADVICE, c, r)
    F = lambda c, r=0: (OP_FIXED, c, r)
    I = lambda c, r=0: (OP_INSTANCE, c, r)
    MUL = (OP_MUL, 0, 0)
    ADD = (OP_ADD, 0, 0)
    SUB = (OP_SUB, 0, 0)

    gates = [
        # q0 * (a0*a1 - a2)
        [F(Q0), A(0), A(1), MUL, A(2), SUB, MUL],
        # q0 * (a0 + a1 - a3)
        [F(Q0), A(0), A(1), ADD, A(3), SUB, MUL],
        # q1 * (a0*a1*a4*a5*a6*a7*a8*a9 - a2(w))  — degree 9
        [F(Q1), A(0), A(1), MUL, A(4), MUL, A(5), MUL, A(6), MUL, A(7), MUL,
         A(8), MUL, A(9), MUL, A(2, 1), SUB, MUL],
        # q2 * (a3(-1) - a3*i0)
        [F(Q2), A(3, -1), A(3), I(0), MUL, SUB, MUL],
    ]
    # lookup: input [q2 * a4], table [f_table]
    lookups = [([[F(Q2), A(4), MUL]], [[F(FTAB)]])]

    # queries (order = first use while scanning gates, then lookups; every
    # advice column must appear)
    advice_q, fixed_q, instance_q = [], [], []

    def note(lst, key):
        if key not in lst:
            lst.append(key)

    for g in gates:
        for tag, a, b in g:
            if tag == OP_ADVICE:
                note(advice_q, (a, b))
            elif tag == OP_FIXED:
                note(fixed_q, (a, b))
            elif tag == OP_INSTANCE:
                note(instance_q, (a, b))
    for ins, tabs in lookups:
        for e in ins + tabs:
            for tag, a, b in e:
                if tag == OP_ADVICE:
                    note(advice_q, (a, b))
                elif tag == OP_FIXED:
                    note(fixed_q, (a, b))
                elif tag == OP_INSTANCE:
                    note(instance_q, (a, b))

    # ---- permutation ----
    perm_cols = [(0, i) for i in range(10)] + [(2, 0), (1, FC)]  # a0..a9, i0, f_c
    # every permutation column participates in the transition constraint at
    # the current rotation -> it must be queried (halo2 adds these itself)
    for kind, idx in perm_cols:
        note({0: advice_q, 1: fixed_q, 2: instance_q}[kind], (idx, 0))
    pc_index = {c: j for j, c in enumerate(perm_cols)}
    # sigma starts as identity: sigma[j][i] = (j, i)
    sigma = [[(j, i) for i in range(N)] for j in range(len(perm_cols))]

    def copy(c1, r1, c2, r2):
        j1, j2 = pc_index[c1], pc_index[c2]
        # splice cycles (halo2 permutation::Assembly::copy semantics)
        a, b = sigma[j1][r1], sigma[j2][r2]
        sigma[j1][r1], sigma[j2][r2] = b, a

    for r in range(N_INSTANCE_ROWS):
        copy((2, 0), r, (0, 0), r)  # i0[r] = a0[r]
    for j in range(4096):
        copy((0, 5), j, (0, 4), j + 1)  # a5[j] = a4[j+1]
    copy((0, 7), 0, (1, FC), 0)  # a7[0] = f_c[0]

    # ---- fixed columns ----
    fixed = [[0] * N for _ in range(N_FIXED)]
    for i in range(0, REGION):
        fixed[Q0][i] = 1
    for i in range(REGION, 2 * REGION):
        fixed[Q1][i] = 1
    for i in range(2 * REGION, 3 * REGION):
        fixed[Q2][i] = 1
    for i in range(N):
        fixed[FTAB][i] = i & 1023
    fixed[FC][0] = 42

    consts = []  # none needed

    # ---- serialize ----
    out = b"TGD1"
    out += struct.pack(
        "<15I", K, EXT_K, N_FIXED, N_ADVICE, N_INSTANCE, BF, len(gates),
        len(perm_cols), CHUNK, len(lookups), len(consts),
        len(advice_q), len(fixed_q), len(instance_q), N_INSTANCE_ROWS,
    )
    for c in consts:
        out += c.to_bytes(32, "little")
    for col, rot in advice_q:
        out += struct.pack("<Ii", col, rot)
    for col, rot in fixed_q:
        out += struct.pack("<Ii", col, rot)
    for col, rot in instance_q:
        out += struct.pack("<Ii", col, rot)
    for kind, idx in perm_cols:
        out += struct.pack("<II", kind, idx)
    for g in gates:
        out += enc_expr(g)
    for ins, tabs in lookups:
        out += struct.pack("<II", len(ins), len(tabs))
        for e in ins:
            out += enc_expr(e)
        for e in tabs:
            out += enc_expr(e)
    for j in range(len(perm_cols)):
        row = bytearray()
        for i in range(N):
            cj, ri = sigma[j][i]
            row += struct.pack("<II", cj, ri)
        out += bytes(row)
    for c in range(N_FIXED):
        col = bytearray()
        for i in range(N):
            col += fixed[c][i].to_bytes(32, "little")
        out += bytes(col)

    path = os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", "tests",
                        "golden", "cs1.desc")
    with open(path, "wb") as fh:
        fh.write(out)
    print(f"wrote {path}: {len(out)} bytes; queries: adv={advice_q} fix={fixed_q} inst={instance_q}")
    assert pp  # imported for future golden additions


if __name__ == "__main__":
    main()
