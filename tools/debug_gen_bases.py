"""Pinpoint bad k_gen_bases outputs: extract individual generated points
through the (oracle-validated) MSM with unit scalars and diff against the
oracle's gen_bases bytes."""
import os
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(REPO, "oracle"))
sys.path.insert(0, REPO)
import oracle_ct as oc
import pypasta as pp
import taiga_amd

n = int(sys.argv[1]) if len(sys.argv) > 1 else (1 << 20)
g = taiga_amd.TaigaGpu(0)
ref = oc.gen_bases(n, 42)
g.gen_bases(n, seed=42)
bad = []
idxs = [0, 1, 2, 63, 64, 255, 256, 4095, 4096, 65535, 65536,
        262143, 262144, 524287, 524288, 1048574, 1048575]
for i in [j for j in idxs if j < n]:
    sc = bytearray(32 * n)
    sc[32 * i] = 1
    r = g.msm(bytes(sc), base_set=0)
    want = ref[64 * i:64 * i + 64]
    ok = r == want
    if not ok:
        x = int.from_bytes(r[:32], "little")
        y = int.from_bytes(r[32:], "little")
        p = pp.Point(x, y, pp.Q)
        bad.append(i)
        print(f"i={i}: MISMATCH on_curve={p.is_on_curve()}")
    else:
        print(f"i={i}: ok")
print("bad indices:", bad)
g.close()
