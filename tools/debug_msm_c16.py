"""Bisect the MSM window-config bug the 2^20 linearity test exposed:
force each small-MSM window config via TG_MSM_SMALL_C/SEG and compare
GPU vs oracle bit-exactly at an oracle-checkable size."""
import os
import random
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

if len(sys.argv) > 1 and sys.argv[1] == "one":
    sys.path.insert(0, os.path.join(REPO, "oracle"))
    sys.path.insert(0, REPO)
    import oracle_ct as oc
    import pypasta as pp
    import taiga_amd

    n = int(sys.argv[2])
    g = taiga_amd.TaigaGpu(0)
    rng = random.Random(1)
    G = pp.Point.generator(pp.Q)
    pts = b"".join(
        (P := G.mul(rng.randrange(1, pp.P))).x.to_bytes(32, "little")
        + P.y.to_bytes(32, "little") for _ in range(n))
    sc = b"".join(rng.randrange(pp.P).to_bytes(32, "little") for _ in range(n))
    g.bases_upload(pts)
    got = g.msm(sc)
    want = oc.msm(oc.FQ, sc, pts)
    print(f"c={os.environ.get('TG_MSM_SMALL_C','auto')} "
          f"seg={os.environ.get('TG_MSM_SMALL_SEG','auto')} n={n}: "
          f"{'OK' if got == want else 'MISMATCH'}")
    g.close()
    sys.exit(0)

n = sys.argv[1] if len(sys.argv) > 1 else "4096"
for c, seg in (("13", "4"), ("14", "8"), ("15", "16"), ("16", "32"), ("16", "16")):
    env = dict(os.environ, TG_MSM_SMALL_C=c, TG_MSM_SMALL_SEG=seg)
    subprocess.run([sys.executable, __file__, "one", n], env=env, cwd=REPO)
