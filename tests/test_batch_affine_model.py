"""Round-2 groundwork stays validated: the phase-scheduled batch-affine
bucket accumulation model (tools/experiments/batch_affine_model.py — the
planned op-count reduction for the VALU-bound bucket kernel) must remain
bit-exact against naive bucket sums, including the doubling and
annihilation lanes a kernel must route around."""
import os
import random
import sys

from conftest import REPO

sys.path.insert(0, os.path.join(REPO, "tools", "experiments"))
sys.path.insert(0, os.path.join(REPO, "oracle"))


def test_phase_model_bit_exact():
    import pypasta as pp
    from batch_affine_model import affine_phase_accumulate, naive_sum, to_xy

    P = pp.Q
    G = pp.Point.generator(P)
    rng = random.Random(5)

    def pt(k):
        q = G.mul(k)
        return (q.x, q.y)

    def neg(p):
        return (p[0], (-p[1]) % P)

    p1 = pt(7)
    buckets = [[pt(rng.randrange(1, 1 << 16)) for _ in range(rng.randrange(0, 8))]
               for _ in range(24)]
    buckets += [[], [p1], [p1, p1], [p1, neg(p1)], [p1, neg(p1), pt(11)],
                [p1] * 5 + [neg(p1)] * 5, [pt(k % 3 + 1) for k in range(9)]]
    got, _ = affine_phase_accumulate(buckets)
    for b, pts in enumerate(buckets):
        assert got[b] == to_xy(naive_sum(pts)), b
