import os, random, sys
REPO = os.getcwd()
sys.path.insert(0, os.path.join(REPO, "oracle")); sys.path.insert(0, REPO)
import numpy as np, oracle_ct as oc, taiga_amd
n = 1 << 20
g = taiga_amd.TaigaGpu(0)
ref = oc.gen_bases(n, 42)
g.gen_bases(n, seed=42)
got = g.bases_download(n)
if got == ref:
    print("bases identical -> bug is in the FAST bucket kernels")
else:
    a = np.frombuffer(got, dtype=np.uint8).reshape(n, 64)
    b = np.frombuffer(ref, dtype=np.uint8).reshape(n, 64)
    bad = np.nonzero((a != b).any(axis=1))[0]
    print(f"{len(bad)} mismatching points; first 20 idx: {bad[:20].tolist()}")
    for i in bad[:5]:
        print(i, got[64*i:64*i+64].hex()[:48], "vs", ref[64*i:64*i+64].hex()[:48])
g.close()
