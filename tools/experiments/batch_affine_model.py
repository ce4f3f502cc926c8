#!/usr/bin/env python3
"""Round-2 groundwork: phase-scheduled batch-affine bucket accumulation.

The bucket-accumulation kernel is VALU real-instruction bound
(profiles/r01_bucket_acc_hazard_analysis.txt): a mixed Jacobian add costs
~7M+4S per gathered point. The known cheaper scheme keeps buckets AFFINE
and uses the affine chord add (2M+1S+1I) with the inversion amortized to
~3M by a GLOBAL batched inversion per phase (Montgomery trick over every
active bucket at once), i.e. ~6M per add — ~1.8x fewer multiplies — at the
price of ~3x more HBM traffic per add (bucket x/y read+write per phase vs
register residency). At the measured 272 GB/s actual traffic of the
current kernel this trade stays far below the 8 TB/s roofline, so the
op-count win should be realizable (round-2 A/B will decide).

This file MODELS the phase algorithm exactly as a kernel would run it and
validates it bit-for-bit against naive accumulation over the oracle's
Python curve (oracle/pypasta.py), INCLUDING the exceptional lanes a GPU
implementation must route around:
  - empty buckets and buckets with one point
  - duplicate points inside one bucket (doubling: chord formula division
    by zero -> use the tangent formula, still one batched inverse of 2y)
  - inverse points inside one bucket (P + (-P) = identity: bucket resets
    to empty and continues)
  - identity never enters buckets (the scatter layout excludes zero
    digits), but accumulators pass through the identity state

Phase schedule (per window; mirrors the sorted-bucket layout the real
pipeline already produces):
  while any bucket has pending points:
    for each active bucket b: pick its next point P_b
      classify: acc empty -> plain assign (no inverse needed)
                x(P) != x(acc) -> chord add, denom = x_P - x_acc
                P == acc       -> double, denom = 2*y_acc
                P == -acc      -> annihilate (no inverse needed)
    batch-invert all denominators of the phase (one inversion total)
    complete the adds with the shared inverses
Validation: randomized buckets + adversarial constructions, exact match
against the naive per-bucket sum for every case.
"""
import os
import random
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", "..",
                                "oracle"))
import pypasta as pp

P = pp.Q  # Vesta base field (point coordinates)


def batch_inverse(vals):
    """Montgomery trick: one inversion for the whole list (zeros excluded
    by the caller's classification)."""
    n = len(vals)
    if n == 0:
        return []
    pref = [1] * (n + 1)
    for i, v in enumerate(vals):
        assert v % P != 0
        pref[i + 1] = pref[i] * v % P
    inv_all = pow(pref[n], P - 2, P)
    out = [0] * n
    acc = inv_all
    for i in range(n - 1, -1, -1):
        out[i] = acc * pref[i] % P
        acc = acc * vals[i] % P
    return out


def affine_phase_accumulate(buckets):
    """buckets: list of lists of affine (x, y) tuples (no identities).
    Returns the list of bucket sums as affine points or None (identity),
    running the exact phase schedule a kernel would."""
    nb = len(buckets)
    acc = [None] * nb      # None = identity accumulator
    idx = [0] * nb         # next pending point per bucket
    phases = 0
    while True:
        work = []  # (bucket, kind, denom) kinds: chord, dbl
        done = True
        for b in range(nb):
            if idx[b] >= len(buckets[b]):
                continue
            done = False
            px, py = buckets[b][idx[b]]
            if acc[b] is None:
                acc[b] = (px, py)       # plain assign, consume
                idx[b] += 1
            elif px != acc[b][0]:
                work.append((b, "chord", (px - acc[b][0]) % P))
            elif py == acc[b][1]:
                work.append((b, "dbl", (2 * acc[b][1]) % P))
            else:
                acc[b] = None           # P + (-P): annihilate, consume
                idx[b] += 1
        if done:
            return acc, phases
        phases += 1
        invs = batch_inverse([d for _, _, d in work])
        for (b, kind, _), inv in zip(work, invs):
            ax, ay = acc[b]
            px, py = buckets[b][idx[b]]
            if kind == "chord":
                lam = (py - ay) * inv % P
            else:  # tangent: lambda = 3x^2 / 2y  (a = 0 on Pasta curves)
                lam = 3 * ax * ax * inv % P
            x3 = (lam * lam - ax - px) % P
            y3 = (lam * (ax - x3) - ay) % P
            acc[b] = (x3, y3)
            idx[b] += 1


def naive_sum(points):
    s = pp.Point.identity(P)
    for x, y in points:
        s = s + pp.Point(x, y, P)
    return s


def to_xy(p):
    return None if p.inf else (p.x, p.y)


def run_case(name, buckets):
    got, phases = affine_phase_accumulate(buckets)
    for b, pts in enumerate(buckets):
        exp = to_xy(naive_sum(pts))
        assert got[b] == exp, f"{name}: bucket {b} mismatch"
    return phases


def main():
    rng = random.Random(99)
    G = pp.Point.generator(P)

    def pt(k):
        q = G.mul(k)
        return (q.x, q.y)


    def neg(p):
        return (p[0], (-p[1]) % P)

    # randomized buckets (uneven sizes, like real digit histograms)
    buckets = [[pt(rng.randrange(1, 1 << 20)) for _ in range(rng.randrange(0, 12))]
               for _ in range(64)]
    ph = run_case("random", buckets)
    print(f"random 64 buckets: OK ({ph} phases)")

    # adversarial: duplicates (doubling), annihilations, long same-point runs
    p1, p2 = pt(7), pt(11)
    adversarial = [
        [],                                 # empty
        [p1],                               # single
        [p1, p1],                           # double
        [p1, neg(p1)],                      # annihilate to identity
        [p1, neg(p1), p2],                  # annihilate then continue
        [p1, p1, p1, p1, p1],               # repeated doubling/chord mix
        [p1, p2, neg(p1), neg(p2)],         # full cancellation
        [pt(3), pt(3), neg(pt(6))],         # double then annihilate
        [p1] * 9 + [neg(p1)] * 9,           # long run then full unwind
        [pt(k % 5 + 1) for k in range(20)],  # heavy duplicates
    ]
    ph = run_case("adversarial", adversarial)
    print(f"adversarial buckets: OK ({ph} phases)")

    # op accounting at the real shape: n=2^15 points into 4096 buckets
    # (the prover's c=13 window): phases ~= max bucket size
    big = [[] for _ in range(4096)]
    for _ in range(1 << 15):
        big[rng.randrange(4096)].append(pt(rng.randrange(1, 1 << 30)))
    got, phases = affine_phase_accumulate(big)
    for b in rng.sample(range(4096), 24):
        assert got[b] == to_xy(naive_sum(big[b]))
    maxlen = max(len(b) for b in big)
    print(f"2^15 -> 4096 buckets: OK; phases={phases} (max bucket {maxlen})")
    print("model validated: phase-scheduled batch-affine accumulation is "
          "bit-exact vs naive sums incl. doubling/annihilation lanes")


if __name__ == "__main__":
    main()
