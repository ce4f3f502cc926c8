#!/usr/bin/env python3
"""Regenerates the constants in oracle/fd_const.c (and the device copies in
taiga_amd/csrc/fd_device.hpp) from the published Pasta moduli.

Provenance: p, q are the Pasta field moduli (pasta_curves v0.5.1, the
un-vendored dep of /root/reference — SURVEY.md §8c); every derived value
(R^2, R^3, -m^-1 mod 2^64, 2-adic root of unity from generator 5) is
computed here, and the generator-5 / omega convention is pinned against the
reference SRS by tests/test_srs_pin.py.
"""
P = 0x40000000000000000000000000000000224698FC094CF91B992D30ED00000001
Q = 0x40000000000000000000000000000000224698FC0994A8DD8C46EB2100000001
S = 32
GEN = 5


def limbs(x, n=4):
    return [(x >> (64 * i)) & 0xFFFFFFFFFFFFFFFF for i in range(n)]


def fmt(x):
    return "{" + ", ".join(f"0x{v:016x}ULL" for v in limbs(x)) + "}"


for name, m in (("P", P), ("Q", Q)):
    R = (1 << 256) % m
    print(f"/* field {name} = 0x{m:064x} */")
    print(f"mod      {fmt(m)}")
    print(f"r2       {fmt(R * R % m)}")
    print(f"r3       {fmt(R * R % m * R % m)}")
    print(f"inv      0x{(-pow(m, -1, 1 << 64)) % (1 << 64):016x}ULL")
    root = pow(GEN, (m - 1) >> S, m)
    print(f"root     {fmt(root)}")
    print(f"root_inv {fmt(pow(root, -1, m))}")
    t = (m - 1) >> S
    print(f"t_odd    {fmt(t)}")
    print(f"t1_2     {fmt((t + 1) // 2)}")
    print()
