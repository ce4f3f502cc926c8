"""Pasta field helpers for the circuit model (plain-int arithmetic mod p/q).

Part of the round-2 circuit restatement toolchain (tools/circuit): the model
that generates the EXACT Taiga compliance / resource-logic constraint systems
(TGD2 desc blobs) and their witness-synthesis programs (TGW1) from a
from-scratch restatement of the reference circuits
(/root/reference/taiga_halo2/src/circuit/*) and the public halo2_gadgets 0.3
chips (un-vendored dep — SURVEY.md §8c).

Build-time tooling only: nothing here ships in the product path; the product
prover consumes the emitted blobs.
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", "..", "oracle"))
import pypasta as pp  # noqa: E402

P = pp.P  # Fp = pallas::Base = vesta::Scalar (circuit field)
Q = pp.Q  # Fq = pallas::Scalar

TWO_INV = pow(2, P - 2, P)
# pasta_curves Fp::ROOT_OF_UNITY = 5^((p-1)/2^32) (2-adicity 32)
ROOT_OF_UNITY = pow(5, (P - 1) >> 32, P)


def inv0(x: int) -> int:
    """halo2's Assigned::invert() semantics: inv0(0) = 0."""
    x %= P
    return 0 if x == 0 else pow(x, P - 2, P)


def sqrt0(x: int) -> int:
    """Deterministic Tonelli-Shanks root, or 0 if x is a non-residue.

    Matches pasta_curves Fp::sqrt()'s deterministic output (same 2-Sylow
    generator 5^t); the choice is pinned by the fixed-base u-table
    byte-compare (tests/test_fixed_base_tables.py).
    """
    r = pp.sqrt_mod(x % P, P)
    return 0 if r is None else r


def is_square(x: int, mod: int = P) -> bool:
    """Legendre symbol via Jacobi (binary) — fast for the z-table search."""
    a = x % mod
    if a == 0:
        return True
    n = mod
    t = 1
    while a != 0:
        while a & 1 == 0:
            a >>= 1
            if n & 7 in (3, 5):
                t = -t
        a, n = n, a
        if a & 3 == 3 and n & 3 == 3:
            t = -t
        a %= n
    return t == 1


def fbit(x: int, i: int) -> int:
    return (x >> i) & 1


def fbyte(x: int, i: int) -> int:
    return (x >> (8 * i)) & 0xFF


def to_repr(x: int) -> bytes:
    return (x % P).to_bytes(32, "little")


def from_repr(b: bytes) -> int:
    x = int.from_bytes(b, "little")
    assert x < P, "non-canonical repr"
    return x
