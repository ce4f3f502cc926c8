"""Driver-contract checks on the committed bench lines (profiles/*.json):
catches accidental drift in bench.py's output schema without needing a
GPU (the lines were produced by the real bench on MI355X boxes)."""
import json
import os

import pytest

from conftest import REPO

PROFILES = os.path.join(REPO, "profiles")
REQUIRED = ["metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
            "higher_is_better", "scaling", "vs_baseline", "dtype", "data",
            "config", "roofline", "cpu_baseline"]


def _final_lines():
    import glob

    names = {os.path.splitext(os.path.basename(p))[0]
             for p in glob.glob(os.path.join(PROFILES, "r0*_final_*.json"))}
    # the round-1 four must always exist; later rounds' lines join the sweep
    names |= {"r01_final_proof", "r01_final_msm", "r01_final_ntt",
              "r01_final_verify"}
    return sorted(names)


@pytest.mark.parametrize("name", _final_lines())
def test_bench_line_schema(name):
    path = os.path.join(PROFILES, f"{name}.json")
    d = json.loads(open(path).read().strip().splitlines()[-1])
    for k in REQUIRED:
        assert k in d, f"{name}: missing {k}"
    assert d["value"] > 0 and d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["data"] == "synthetic"
    assert "workload" in d["config"]
    r = d["roofline"]
    assert r["bound"] in ("hbm", "mfma") and 0 < r["frac"] < 1
    assert r["peak"] == 8000.0 and r["unit"] == "GB/s"
    assert abs(r["achieved"] / r["peak"] - r["frac"]) < 1e-3
    c = d["cpu_baseline"]
    assert c["kind"] in ("port", "reference") and c["value"] > 0 and c["cores"] >= 1
    assert c["unit"] == d["unit"]
    # GPU beats the CPU oracle on every workload
    assert d["value"] > c["value"]


def test_bench_script_defaults_parse():
    """bench.py must keep the contract flags and default to N=1"""
    import ast

    src = open(os.path.join(REPO, "bench.py")).read()
    ast.parse(src)
    for flag in ["--gpus", "--steps", "--warmup", "--workload", "--streams"]:
        assert flag in src
    assert '"metric"' in src and '"vs_baseline"' in src
