import ctypes, os, sys
REPO = os.getcwd()
sys.path.insert(0, os.path.join(REPO, "tools")); sys.path.insert(0, REPO)
from gen_rand_circuit import gen
import taiga_amd
lib = ctypes.CDLL(os.path.join(REPO, "oracle", "liboracle.so"))
lib.orc_prove_raw.restype = ctypes.c_long
srs = open("tests/golden/params_15", "rb").read()
g = taiga_amd.TaigaGpu(0)
g.load_srs(srs)
RNG = bytes([7]) + bytes(31)
bad = []
import time
t0 = time.time()
seeds = list(range(300, 400))
for seed in seeds:
    if time.time() - t0 > 420: 
        print(f"time-capped after seed {seed}")
        break
    desc, inst, adv, meta = gen(seed)
    lib.orc_prover_reset()
    assert lib.orc_prover_init(desc, len(desc), srs, len(srs)) == 0
    out = ctypes.create_string_buffer(1 << 15)
    n = lib.orc_prove_raw(inst, adv, RNG, out, 1 << 15)
    if n <= 0:
        bad.append((seed, "oracle", n)); continue
    slot = g.keygen(desc); g.select_key(slot)
    gp = g.create_proof_raw(inst, adv, RNG)
    if gp != out.raw[:n]:
        bad.append((seed, "mismatch", meta))
    print(f"seed {seed}: {'MISMATCH' if gp != out.raw[:n] else 'ok'}", flush=True)
print(f"swept {seeds[0]}..{seed}, bad: {bad if bad else 'none'}")
g.close()
