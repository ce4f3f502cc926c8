"""taiga_amd — MI355X-native Halo2/Pasta proving backend for Taiga.

PRODUCT package. The compute path is libtaiga_gpu.so (hand-written HIP/CDNA4
kernels behind the C ABI in include/taiga_gpu.h — the drop-in boundary for
the reference's Proof::create path, taiga_halo2/src/proof.rs:25-42).

There is NO CPU fallback here: if the extension is missing or no HIP device
is usable, calls raise. The CPU restatement used by the tests lives in
oracle/ and is test infrastructure only.
"""
from .api import (  # noqa: F401
    TaigaGpu,
    TaigaGpuError,
    lib_path,
    load_library,
)

__all__ = ["TaigaGpu", "TaigaGpuError", "load_library", "lib_path"]
