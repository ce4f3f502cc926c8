"""Discriminate the 2^20 linearity failure: oracle-generated bases
(known good, uploaded) vs tg_gen_bases; optionally full oracle MSM."""
import os
import random
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(REPO, "oracle"))
sys.path.insert(0, REPO)
import numpy as np
import oracle_ct as oc
import pypasta as pp
import taiga_amd

n = int(sys.argv[1]) if len(sys.argv) > 1 else (1 << 20)
g = taiga_amd.TaigaGpu(0)
rng = random.Random(99)
raw = np.frombuffer(rng.getrandbits(2 * n * 256).to_bytes(2 * n * 32, "little"),
                    dtype=np.uint8).copy()
raw[31::32] &= 0x3F
a, b = raw[: n * 32].tobytes(), raw[n * 32:].tobytes()
arr = np.frombuffer(raw, dtype="<u8").reshape(2 * n, 4).astype(object)
vals = arr[:, 0] + (arr[:, 1] << 64) + (arr[:, 2] << 128) + (arr[:, 3] << 192)
s = (vals[:n] + vals[n:]) % pp.P
ab = b"".join(int(v).to_bytes(32, "little") for v in s)


def pt(r):
    return pp.Point(int.from_bytes(r[:32], "little"),
                    int.from_bytes(r[32:], "little"), pp.Q)


def lin_check(tag):
    ra, rb, rab = g.msm(a, base_set=0), g.msm(b, base_set=0), g.msm(ab, base_set=0)
    pa, pb, pab = pt(ra), pt(rb), pt(rab)
    got = pa + pb
    ok = (got.x, got.y) == (pab.x, pab.y)
    print(f"{tag}: linearity {'OK' if ok else 'BROKEN'} "
          f"(on-curve: {pa.is_on_curve()},{pb.is_on_curve()},{pab.is_on_curve()})")
    return ra


print("generating oracle bases...")
bases = oc.gen_bases(n, 42)
g.bases_upload(bases)
r_up = lin_check("uploaded-oracle-bases")

g.gen_bases(n, seed=42)
r_gen = lin_check("tg_gen_bases")
print("uploaded vs generated msm(a) equal:", r_up == r_gen)

if "--oracle" in sys.argv:
    print("oracle full msm (slow)...")
    want = oc.msm(oc.FQ, a, bases)
    print("gpu == oracle at n=%d:" % n, r_up == want)
g.close()
