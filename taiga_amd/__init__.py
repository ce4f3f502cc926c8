"""taiga_amd — MI355X-native Halo2/Pasta proving backend for Taiga.

PRODUCT package. The compute path is libtaiga_gpu.so (hand-written HIP/CDNA4
kernels behind the C ABI in include/taiga_gpu.h — the drop-in boundary for
the reference's Proof::create path, taiga_halo2/src/proof.rs:25-42).

There is NO CPU fallback here: if the extension is missing or no HIP device
is usable, calls raise. The CPU restatement used by the tests lives in
oracle/ and is test infrastructure only.
"""
from . import wire  # noqa: F401
from .api import (  # noqa: F401
    binding_sign,
    binding_verify,
    binding_vk,
    tx_digest,
    tx_wire_check,
    TaigaGpu,
    TaigaGpuError,
    lib_path,
    load_library,
)

__all__ = ["TaigaGpu", "TaigaGpuError", "load_library", "lib_path", "wire",
           "binding_sign", "binding_verify", "binding_vk", "tx_digest",
           "tx_wire_check"]
