#!/usr/bin/env python3
"""Generate the EXACT compliance + trivial-RL circuit artifacts:

  tests/golden/compliance.desc    TGD2 constraint-system blob
  tests/golden/compliance.tgw     TGW1 witness-synthesis program
  tests/golden/trivial_rl.desc    TGD2
  tests/golden/trivial_rl.tgw     TGW1
  tests/golden/compliance_sample.json   seeded sample (inputs, instance,
                                        witness column hashes) for
                                        interpreter parity tests

Run with --check to also run the MockProver-equivalent (slow-ish).
"""
import argparse
import hashlib
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from circuit import fields as F  # noqa: E402
from circuit import hostcrypto as hc  # noqa: E402
from circuit import emit  # noqa: E402
from circuit.compliance import ComplianceModel, build_inputs as build_compliance_inputs  # noqa: E402
from circuit.trivial_rl import TrivialRLModel, build_inputs as build_rl_inputs  # noqa: E402

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
GOLDEN = os.path.join(REPO, "tests", "golden")


def det(seed: bytes, n: int) -> int:
    return int.from_bytes(hashlib.blake2b(seed, digest_size=64).digest(), "little") % n


def sample_compliance_inputs(tag=b"sample0"):
    """Deterministic sample ComplianceInfo-equivalent."""
    fp = lambda s: det(tag + s, F.P)
    input_res = hc.Resource(
        logic=fp(b"ilogic"), label=fp(b"ilabel"), value=fp(b"ivalue"),
        quantity=det(tag + b"iq", 1 << 64), nk=fp(b"ink"), nk_is_key=True,
        nonce=fp(b"inonce"), is_ephemeral=False, rseed=fp(b"irseed"))
    nf = input_res.get_nf()
    output_res = hc.Resource(
        logic=fp(b"ologic"), label=fp(b"olabel"), value=fp(b"ovalue"),
        quantity=det(tag + b"oq", 1 << 64), nk=fp(b"onpk"), nk_is_key=False,
        nonce=nf, is_ephemeral=False, rseed=fp(b"orseed"))
    path = [(fp(b"node%d" % i), bool(det(tag + b"lr%d" % i, 2)))
            for i in range(32)]
    anchor = hc.merkle_root(input_res.commitment(), path)
    # rseed-driven randomness exactly like ComplianceInfo (compliance.rs):
    rseed = hashlib.blake2b(tag + b"rseed", digest_size=32).digest()
    rcv_int = hc.prf_expand_field(rseed, hc.PRF_EXPAND_VCM_R, F.Q)
    rcv = rcv_int.to_bytes(32, "little")
    r_in = hc.prf_expand_field(rseed, hc.PRF_EXPAND_INPUT_RESOURCE_LOGIC_CM_R)
    r_out = hc.prf_expand_field(rseed, hc.PRF_EXPAND_OUTPUT_RESOURCE_LOGIC_CM_R)
    instance, inputs = build_compliance_inputs(
        input_res, path, anchor, output_res, rcv, r_in, r_out)
    # borsh ComplianceInfo (compliance.rs:51-59)
    import struct as _s
    borsh = input_res.borsh()
    borsh += _s.pack("<I", 32)
    for node, is_left in path:
        borsh += F.to_repr(node) + bytes([1 if is_left else 0])
    borsh += F.to_repr(anchor)
    borsh += output_res.borsh()
    borsh += rseed
    return instance, inputs, borsh


def sample_rl_inputs(tag=b"rlsample0"):
    fp = lambda s: det(tag + s, F.P)
    res = hc.Resource(
        logic=fp(b"logic"), label=fp(b"label"), value=fp(b"value"),
        quantity=det(tag + b"q", 1 << 64), nk=fp(b"nk"), nk_is_key=True,
        nonce=fp(b"nonce"), is_ephemeral=False, rseed=fp(b"rseed"))
    path = [(fp(b"node%d" % i), bool(det(tag + b"lr%d" % i, 2)))
            for i in range(4)]
    # is_input must match !path[0].is_left (resource_tree.rs is_input())
    path[0] = (path[0][0], False)
    pad_rseed = hashlib.blake2b(tag + b"padseed", digest_size=32).digest()
    padding = hc.random_seed_padding(pad_rseed, 16)
    instance, inputs = build_rl_inputs(res, path, True, padding)
    # borsh ResourceExistenceWitness (resource_tree.rs:70-81)
    borsh = res.borsh()
    for node, is_left in path:
        borsh += F.to_repr(node) + bytes([1 if is_left else 0])
    return instance, inputs, borsh, pad_rseed


def run(model_cls, builder, name, check):
    t0 = time.time()
    model = model_cls()
    built = builder()
    instance, inputs = built[0], built[1]
    extra = built[2:]
    cs = model.synthesize(inputs)
    t1 = time.time()
    fin = emit.finalize(cs, instance, name)
    print(f"[{name}] synth {t1-t0:.1f}s; regions={cs.regions} ops={len(cs.prog.ops)} "
          f"degree={fin.degree} ext_k={fin.ext_k} bf={fin.bf} "
          f"n_fixed={len(cs.fixed_cols)} constraints={len(fin.constraints)} "
          f"aq={len(fin.advice_q)} fq={len(fin.fixed_q)}")
    maxrow = max(
        [max(d.keys(), default=0) for d in cs.advice_vals] +
        [max(d.keys(), default=0) for d in cs.fixed_vals])
    print(f"[{name}] max row used = {maxrow} (usable {fin.usable})")
    if check:
        t0 = time.time()
        fails = emit.mock_verify(fin)
        print(f"[{name}] mock verify {time.time()-t0:.1f}s: "
              f"{'OK' if not fails else fails[:5]}")
        if fails:
            return None, None, None
    desc = emit.emit_desc(fin)
    tgw = emit.emit_witness_program(cs)
    open(os.path.join(GOLDEN, f"{name}.desc"), "wb").write(desc)
    open(os.path.join(GOLDEN, f"{name}.tgw"), "wb").write(tgw)
    print(f"[{name}] desc {len(desc)} B, tgw {len(tgw)} B")
    # sample fixture: inputs + instance + per-column advice blake2b hashes
    colhash = []
    n = cs.n
    for ci in range(len(cs.advice_cols)):
        h = hashlib.blake2b(digest_size=32)
        col = cs.advice_vals[ci]
        buf = bytearray(32 * n)
        for row, reg in col.items():
            buf[row * 32:(row + 1) * 32] = (reg.v).to_bytes(32, "little")
        h.update(bytes(buf))
        colhash.append(h.hexdigest())
    fix = {
        "inputs": [hex(v) for v in inputs],
        "instance": [hex(v) for v in instance],
        "advice_col_blake2b": colhash,
    }
    if extra:
        fix["witness_borsh"] = extra[0].hex()
    if len(extra) > 1:
        fix["pad_rseed"] = extra[1].hex()
    open(os.path.join(GOLDEN, f"{name}_sample.json"), "w").write(
        json.dumps(fix, indent=1))
    return cs, fin, instance


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--check", action="store_true")
    ap.add_argument("--only", choices=["compliance", "trivial_rl"])
    args = ap.parse_args()
    if args.only in (None, "compliance"):
        run(ComplianceModel, sample_compliance_inputs, "compliance", args.check)
    if args.only in (None, "trivial_rl"):
        run(TrivialRLModel, sample_rl_inputs, "trivial_rl", args.check)


if __name__ == "__main__":
    main()
