#!/usr/bin/env python3
"""Random satisfiable circuit generator for prover parity fuzzing.

Produces (desc_bytes, instance_bytes, advice_bytes) triples: a random TGD1
circuit description (same blob format as tools/gen_cs1.py) together with a
witness that satisfies it BY CONSTRUCTION, fed to both provers through the
raw-witness entries (orc_prove_raw / tg_create_proof_raw) whose proof bytes
must match bit-for-bit and verify through both raw-instance verifiers.

Shapes varied per seed (within the engine limits: expression stack depth
<= 4 — the device evaluator's s0..s3 register stack —, advice <= 16,
gates <= 16, lookups <= 4, permutation chunks <= 8):
  - advice column count, gate count/expressions (rotations ±1, degree <= 9)
  - lookup count 0..2 (selector-gated membership in a fixed table)
  - permutation column set and chunk length (1..4 chunks), random equality
    cycles, instance-exposure copies
  - blinding factor count, instance row count

Satisfiability scheme: gate j is sel_j * (E_j - target_j) with a dedicated
0/1 selector fixed column; E_j only references advice columns with index
below target_j's, so targets are assigned left-to-right by evaluating E_j
(free columns are random). Selector-active rows keep every referenced row
below `usable` (rows >= usable are replaced by prover blinding and must
only be touched with zero selectors — the halo2 rule that gates vanish on
the WHOLE domain)."""
import os
import random
import struct
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", "oracle"))

P = 0x40000000000000000000000000000000224698FC094CF91B992D30ED00000001

K = 15
N = 1 << K
EXT_K = 18

OP_CONST, OP_FIXED, OP_ADVICE, OP_INSTANCE, OP_ADD, OP_SUB, OP_MUL, OP_NEG, OP_SCALE = range(9)


def enc_expr(ops):
    out = struct.pack("<I", len(ops))
    for tag, a, b in ops:
        out += struct.pack("<IIi", tag, a, b)
    return out


def eval_expr(ops, row, advice, fixed, inst_col, consts):
    st = []
    for tag, a, b in ops:
        if tag == OP_CONST:
            st.append(consts[a])
        elif tag == OP_FIXED:
            st.append(fixed[a][(row + b) % N])
        elif tag == OP_ADVICE:
            st.append(advice[a][(row + b) % N])
        elif tag == OP_INSTANCE:
            st.append(inst_col[(row + b) % N])
        elif tag == OP_ADD:
            st[-2:] = [(st[-2] + st[-1]) % P]
        elif tag == OP_SUB:
            st[-2:] = [(st[-2] - st[-1]) % P]
        elif tag == OP_MUL:
            st[-2:] = [(st[-2] * st[-1]) % P]
        elif tag == OP_NEG:
            st[-1] = (-st[-1]) % P
        elif tag == OP_SCALE:
            st[-1] = st[-1] * consts[a] % P
    assert len(st) == 1
    return st[0]


def stack_peak(ops):
    d = peak = 0
    for tag, _, _ in ops:
        d += 1 if tag in (OP_CONST, OP_FIXED, OP_ADVICE, OP_INSTANCE) else (
            -1 if tag in (OP_ADD, OP_SUB, OP_MUL) else 0)
        peak = max(peak, d)
    return peak


def gen(seed, inst_override=None):
    """inst_override: optional list of field values replacing the drawn
    instance values (same rng stream is consumed either way, so the DESC
    bytes are identical for any override — only the witness changes).
    Extra override entries beyond n_instance_rows are ignored."""
    rng = random.Random(seed)
    bf = rng.choice([4, 5, 6])
    usable = N - (bf + 1)
    n_gates = rng.randint(2, 6)
    n_free = rng.randint(3, max(3, 10 - n_gates))
    n_advice = n_free + n_gates
    n_lookups = rng.randint(0, 2)
    n_instance_rows = rng.randint(1, 9)
    # fixed columns: one selector per gate, one table+selector per lookup,
    # one constant column
    n_fixed = n_gates + 2 * n_lookups + 1
    fc = n_fixed - 1
    consts = [rng.randrange(P) for _ in range(rng.randint(0, 3))]

    fixed = [[0] * N for _ in range(n_fixed)]
    advice = [[0] * N for _ in range(n_advice)]
    inst_vals = [rng.randrange(P) for _ in range(n_instance_rows)]
    if inst_override is not None:
        assert len(inst_override) >= n_instance_rows
        inst_vals = [v % P for v in inst_override[:n_instance_rows]]
    inst_col = [0] * N
    for r, v in enumerate(inst_vals):
        inst_col[r] = v

    # free columns: random on [0, usable); rows >= usable stay 0 (replaced
    # by prover blinding anyway)
    for c in range(n_free):
        col = advice[c]
        for i in range(usable):
            col[i] = rng.randrange(P)

    # ---- permutation: all advice cols + instance + the const fixed col
    perm_cols = [(0, i) for i in range(n_advice)] + [(2, 0), (1, fc)]
    pc_index = {c: j for j, c in enumerate(perm_cols)}
    sigma = [[(j, i) for i in range(N)] for j in range(len(perm_cols))]

    def copy(c1, r1, c2, r2):
        j1, j2 = pc_index[c1], pc_index[c2]
        a, b = sigma[j1][r1], sigma[j2][r2]
        sigma[j1][r1], sigma[j2][r2] = b, a

    # instance exposure: a0[r] = inst[r] (value + copy)
    for r in range(n_instance_rows):
        advice[0][r] = inst_vals[r]
        copy((2, 0), r, (0, 0), r)
    # const-col copy: a2[0] = fc[0]
    fixed[fc][0] = rng.randrange(P)
    advice[2][0] = fixed[fc][0]
    copy((0, 2), 0, (1, fc), 0)
    # random equality cycles within free column 2 (not used by lookups)
    for _ in range(rng.randint(1, 3)):
        r1 = rng.randrange(1, usable - 1)
        r2 = rng.randrange(1, usable - 1)
        if r1 == r2:
            continue
        advice[2][r2] = advice[2][r1]
        copy((0, 2), r1, (0, 2), r2)
    chunk_len = rng.randint(3, 7)

    # ---- lookups: input [q_lk * a1], table [f_table]. All lookups share
    # one value pool S (their tables hold the same SET in different row
    # arrangements) so overlapping active regions on the shared input
    # column a1 stay members of every table; 0 in S covers idle rows.
    lookups = []
    tset = [0] + [rng.randrange(P) for _ in range(63)]
    for l in range(n_lookups):
        qcol = n_gates + 2 * l
        tcol = n_gates + 2 * l + 1
        off = rng.randrange(64)
        for i in range(N):
            fixed[tcol][i] = tset[(i + off) % 64]
        lo = rng.randrange(0, usable // 2)
        hi = rng.randrange(lo + 1, usable - 1)
        for i in range(lo, hi):
            fixed[qcol][i] = 1
            advice[1][i] = rng.choice(tset)
        lookups.append(([[(OP_FIXED, qcol, 0), (OP_ADVICE, 1, 0), (OP_MUL, 0, 0)]],
                        [[(OP_FIXED, tcol, 0)]]))

    # ---- gates: sel_j * (E_j - target_j), target col = n_free + j
    gates = []
    for j in range(n_gates):
        tcol = n_free + j
        qcol = j
        trot = rng.choice([0, 0, 1])
        # selector-active region: every referenced row (rot in [-1, +1],
        # target rot in [0, +1]) must stay below usable
        lo = rng.randrange(1, usable // 2)
        hi = rng.randrange(lo + 1, usable - 2)
        for i in range(lo, hi):
            fixed[qcol][i] = 1
        # E: left-assoc chain of 1..4 factors over cols < tcol (peak
        # stack: sel + acc + 2 factor operands = 4 = device limit)
        nfac = rng.randint(1, 4)
        E = []
        deg = 0
        for f in range(nfac):
            kind = rng.randrange(4 if consts else 3)
            c1 = rng.randrange(tcol)
            r1 = rng.choice([-1, 0, 1])
            if kind == 0:  # single cell
                fac = [(OP_ADVICE, c1, r1)]
                d = 1
            elif kind == 1:  # a +/- b
                c2 = rng.randrange(tcol)
                fac = [(OP_ADVICE, c1, r1), (OP_ADVICE, c2, rng.choice([-1, 0, 1])),
                       (rng.choice([OP_ADD, OP_SUB]), 0, 0)]
                d = 1
            elif kind == 2:  # a * b
                c2 = rng.randrange(tcol)
                fac = [(OP_ADVICE, c1, r1), (OP_ADVICE, c2, rng.choice([-1, 0, 1])),
                       (OP_MUL, 0, 0)]
                d = 2
            else:  # a + const (SCALE exercises the const path)
                fac = [(OP_ADVICE, c1, r1), (OP_SCALE, rng.randrange(len(consts)), 0)]
                d = 1
            if deg + d > 7:
                break
            E += fac
            if f > 0:
                E.append((OP_MUL, 0, 0))
            deg += d
        if rng.random() < 0.3:
            E.append((OP_NEG, 0, 0))
        gate = [(OP_FIXED, qcol, 0)] + E + [(OP_ADVICE, tcol, trot), (OP_SUB, 0, 0),
                                            (OP_MUL, 0, 0)]
        assert stack_peak(gate) <= 4, stack_peak(gate)
        gates.append(gate)
        # assign the target so the gate vanishes on active rows
        for i in range(lo, hi):
            v = eval_expr(E, i, advice, fixed, inst_col, consts)
            advice[tcol][(i + trot) % N] = v

    # ---- queries (first-use order over gates then lookups, then perm
    # cols at rotation 0 — mirrors gen_cs1)
    advice_q, fixed_q, instance_q = [], [], []

    def note(lst, key):
        if key not in lst:
            lst.append(key)

    for ops in gates + [e for ins, tabs in lookups for e in ins + tabs]:
        for tag, a, b in ops:
            if tag == OP_ADVICE:
                note(advice_q, (a, b))
            elif tag == OP_FIXED:
                note(fixed_q, (a, b))
            elif tag == OP_INSTANCE:
                note(instance_q, (a, b))
    for kind, idx in perm_cols:
        note({0: advice_q, 1: fixed_q, 2: instance_q}[kind], (idx, 0))
    if not instance_q:
        note(instance_q, (0, 0))

    # ---- mock check (python-side MockProver equivalent) ----
    for g_i, ops in enumerate(gates):
        for i in range(0, usable, 97):  # stride-sampled; active rows exact below
            assert eval_expr(ops, i, advice, fixed, inst_col, consts) == 0, (g_i, i)
    for ops in gates:
        qcol = ops[0][1]
        for i in range(N):
            if fixed[qcol][i]:
                assert eval_expr(ops, i, advice, fixed, inst_col, consts) == 0
    for (ins, tabs), l in zip(lookups, range(n_lookups)):
        tcol = n_gates + 2 * l + 1
        tvals = set(fixed[tcol][:usable])
        for i in range(usable):
            v = eval_expr(ins[0], i, advice, fixed, inst_col, consts)
            assert v in tvals
    for j, scol in enumerate(sigma):
        for i in range(0, N, 251):
            cj, ri = scol[i]
            kind, idx = perm_cols[j]
            kind2, idx2 = perm_cols[cj]
            src = {0: lambda c, r: advice[c][r], 1: lambda c, r: fixed[c][r],
                   2: lambda c, r: inst_col[r]}
            assert src[kind](idx, i) == src[kind2](idx2, ri)

    # ---- serialize desc ----
    out = b"TGD1"
    out += struct.pack(
        "<15I", K, EXT_K, n_fixed, n_advice, 1, bf, len(gates),
        len(perm_cols), chunk_len, len(lookups), len(consts),
        len(advice_q), len(fixed_q), len(instance_q), n_instance_rows,
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
    for c in range(n_fixed):
        col = bytearray()
        for v in fixed[c]:
            col += v.to_bytes(32, "little")
        out += bytes(col)

    inst_bytes = b"".join(v.to_bytes(32, "little") for v in inst_vals)
    adv_bytes = b"".join(v.to_bytes(32, "little") for col in advice for v in col)
    meta = dict(n_advice=n_advice, n_free=n_free, n_gates=n_gates,
                n_lookups=n_lookups, n_fixed=n_fixed, bf=bf, chunk_len=chunk_len,
                n_perm=len(perm_cols), n_instance_rows=n_instance_rows)
    return out, inst_bytes, adv_bytes, meta


if __name__ == "__main__":
    for seed in (int(a) for a in (sys.argv[1:] or ["1"])):
        desc, inst, adv, meta = gen(seed)
        print(f"seed={seed}: desc {len(desc)} bytes, {meta}")
