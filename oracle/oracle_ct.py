"""ctypes wrapper over liboracle.so (CPU oracle).

ORACLE TEST INFRASTRUCTURE — importable only from tests/,
__graft_entry__.smoke() and bench.py's cpu_baseline leg (see fd.h).
Builds the library on first use if missing (gcc, seconds).
"""
import ctypes
import os
import subprocess

_DIR = os.path.dirname(os.path.abspath(__file__))
_LIB = os.path.join(_DIR, "liboracle.so")


def build():
    subprocess.run(["make", "-s", "-C", _DIR], check=True)


_lib = None


def lib():
    global _lib
    if _lib is None:
        if not os.path.exists(_LIB):
            build()
        _lib = ctypes.CDLL(_LIB)
        _lib.orc_srs_project_check.restype = ctypes.c_int
        _lib.orc_srs_project_check.argtypes = [ctypes.c_char_p, ctypes.c_long, ctypes.c_int]
    return _lib


FP, FQ = 0, 1


def fd_op(fid, op, a: bytes, b: bytes = b"\x00" * 32) -> bytes:
    out = ctypes.create_string_buffer(32)
    rc = lib().orc_fd_op(fid, op, a, b, out)
    if rc != 0:
        raise ValueError(f"orc_fd_op rc={rc}")
    return out.raw


def fd_pow(fid, a: bytes, e: bytes) -> bytes:
    out = ctypes.create_string_buffer(32)
    rc = lib().orc_fd_pow(fid, a, e, out)
    if rc != 0:
        raise ValueError(f"orc_fd_pow rc={rc}")
    return out.raw


def get_omega(fid, k, inverse=False) -> bytes:
    out = ctypes.create_string_buffer(32)
    lib().orc_get_omega(fid, k, 1 if inverse else 0, out)
    return out.raw


def pt_op(fid, op, a: bytes, b: bytes = b"\x00" * 32) -> bytes:
    out = ctypes.create_string_buffer(64)
    rc = lib().orc_pt_op(fid, op, a, b, out)
    if rc != 0:
        raise ValueError(f"orc_pt_op rc={rc}")
    return out.raw


def pt_on_curve(fid, a: bytes) -> bool:
    return bool(lib().orc_pt_on_curve(fid, a))


def ntt(fid, direction, k, data: bytes) -> bytes:
    buf = ctypes.create_string_buffer(data, len(data))
    rc = lib().orc_ntt(fid, direction, k, buf)
    if rc != 0:
        raise ValueError(f"orc_ntt rc={rc}")
    return buf.raw


def msm(fid, scalars: bytes, points: bytes) -> bytes:
    n = len(scalars) // 32
    assert len(points) == 64 * n
    out = ctypes.create_string_buffer(64)
    rc = lib().orc_msm(fid, n, scalars, points, out)
    if rc != 0:
        raise ValueError(f"orc_msm rc={rc}")
    return out.raw


def decompress(fid, data: bytes) -> bytes:
    n = len(data) // 32
    out = ctypes.create_string_buffer(64 * n)
    rc = lib().orc_decompress(fid, n, data, out)
    if rc != 0:
        raise ValueError(f"orc_decompress rc={rc}")
    return out.raw


def compress(fid, data: bytes) -> bytes:
    n = len(data) // 64
    out = ctypes.create_string_buffer(32 * n)
    rc = lib().orc_compress(fid, n, data, out)
    if rc != 0:
        raise ValueError(f"orc_compress rc={rc}")
    return out.raw


def blake2b(data: bytes, personal: bytes = None, outlen: int = 64) -> bytes:
    out = ctypes.create_string_buffer(outlen)
    lib().orc_blake2b(data, len(data), personal, outlen, out)
    return out.raw


def blake2s(data: bytes, personal: bytes = None, outlen: int = 32) -> bytes:
    out = ctypes.create_string_buffer(outlen)
    lib().orc_blake2s(data, len(data), personal, outlen, out)
    return out.raw


def gen_bases(n: int, seed: int = 0) -> bytes:
    out = ctypes.create_string_buffer(64 * n)
    lib().orc_gen_bases(n, ctypes.c_uint64(seed), out)
    return out.raw


def srs_project_check(params: bytes, rounds: int = 2) -> int:
    return lib().orc_srs_project_check(params, len(params), rounds)
