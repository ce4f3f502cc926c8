"""ctypes binding over libtaiga_gpu.so (see include/taiga_gpu.h).

Fails loudly when the extension or a HIP device is missing — no silent
eager/CPU fallback (the CPU oracle is test-only infrastructure).
"""
import ctypes
import os

_DIR = os.path.dirname(os.path.abspath(__file__))
_LIB = os.path.join(_DIR, "csrc", "libtaiga_gpu.so")

ERR_NAMES = {
    0: "TG_OK",
    -1: "TG_ERR_HIP",
    -2: "TG_ERR_BADARG",
    -3: "TG_ERR_ENCODING",
    -4: "TG_ERR_NOSRS",
    -5: "TG_ERR_NOMEM",
    -6: "TG_ERR_STATE",
}


class TaigaGpuError(RuntimeError):
    def __init__(self, rc, detail=""):
        self.rc = rc
        super().__init__(f"{ERR_NAMES.get(rc, rc)}: {detail}")


def lib_path():
    return _LIB


_lib = None


def load_library():
    """Load libtaiga_gpu.so; raises if not built (no fallback)."""
    global _lib
    if _lib is None:
        if not os.path.exists(_LIB):
            raise TaigaGpuError(
                -1,
                f"{_LIB} not built — run `make -C taiga_amd/csrc` "
                "(hipcc --offload-arch=gfx950); there is no CPU fallback",
            )
        lib = ctypes.CDLL(_LIB)
        lib.tg_init.argtypes = [ctypes.c_int, ctypes.POINTER(ctypes.c_void_p)]
        lib.tg_error_string.restype = ctypes.c_char_p
        lib.tg_error_string.argtypes = [ctypes.c_void_p]
        lib.tg_destroy.argtypes = [ctypes.c_void_p]
        lib.tg_load_srs.argtypes = [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_size_t]
        lib.tg_srs_k.argtypes = [ctypes.c_void_p]
        lib.tg_bases_upload.argtypes = [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_size_t]
        lib.tg_scalars_upload.argtypes = [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_size_t]
        lib.tg_gen_bases.argtypes = [ctypes.c_void_p, ctypes.c_size_t, ctypes.c_uint64]
        lib.tg_bases_download.argtypes = [ctypes.c_void_p, ctypes.c_size_t,
                                          ctypes.c_char_p]
        lib.tg_msm_pallas.argtypes = [
            ctypes.c_void_p, ctypes.c_char_p, ctypes.c_size_t, ctypes.c_int, ctypes.c_char_p,
        ]
        lib.tg_msm_resident.argtypes = [
            ctypes.c_void_p, ctypes.c_size_t, ctypes.c_int, ctypes.c_char_p,
        ]
        lib.tg_ntt_fp.argtypes = [
            ctypes.c_void_p, ctypes.c_int, ctypes.c_uint32, ctypes.c_int, ctypes.c_char_p,
        ]
        lib.tg_poly_upload.argtypes = [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_uint32]
        lib.tg_ntt_resident.argtypes = [
            ctypes.c_void_p, ctypes.c_int, ctypes.c_uint32, ctypes.c_int,
        ]
        lib.tg_poly_download.argtypes = [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_uint32]
        lib.tg_keygen.argtypes = [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_size_t]
        lib.tg_select_key.argtypes = [ctypes.c_void_p, ctypes.c_int]
        lib.tg_create_proof.argtypes = [
            ctypes.c_void_p, ctypes.c_char_p, ctypes.c_char_p, ctypes.c_char_p,
            ctypes.c_char_p, ctypes.c_size_t, ctypes.POINTER(ctypes.c_size_t),
        ]
        lib.tg_create_proof_raw.argtypes = [
            ctypes.c_void_p, ctypes.c_char_p, ctypes.c_char_p, ctypes.c_char_p,
            ctypes.c_char_p, ctypes.c_size_t, ctypes.POINTER(ctypes.c_size_t),
        ]
        lib.tg_verify_proof.argtypes = [
            ctypes.c_void_p, ctypes.c_char_p, ctypes.c_char_p, ctypes.c_size_t,
        ]
        lib.tg_verify_proof_raw.argtypes = [
            ctypes.c_void_p, ctypes.c_char_p, ctypes.c_char_p, ctypes.c_size_t,
        ]
        lib.tg_verify_batch.argtypes = [
            ctypes.c_void_p, ctypes.c_size_t, ctypes.c_char_p, ctypes.c_char_p,
            ctypes.POINTER(ctypes.c_size_t),
        ]
        lib.tg_verify_batch_raw.argtypes = [
            ctypes.c_void_p, ctypes.c_size_t, ctypes.c_char_p, ctypes.c_char_p,
            ctypes.POINTER(ctypes.c_size_t),
        ]
        lib.tg_poseidon_hash.argtypes = [
            ctypes.c_void_p, ctypes.c_char_p, ctypes.c_size_t, ctypes.c_int,
            ctypes.c_char_p,
        ]
        lib.tg_witness_hash.argtypes = [
            ctypes.c_void_p, ctypes.c_char_p, ctypes.c_char_p, ctypes.c_char_p,
        ]
        lib.tg_binding_vk.argtypes = [ctypes.c_char_p, ctypes.c_char_p]
        lib.tg_delta_commit.argtypes = [ctypes.c_char_p, ctypes.c_char_p]
        lib.tg_binding_sign.argtypes = [
            ctypes.c_char_p, ctypes.c_char_p, ctypes.c_size_t, ctypes.c_char_p,
            ctypes.c_char_p,
        ]
        lib.tg_binding_verify.argtypes = [
            ctypes.c_char_p, ctypes.c_char_p, ctypes.c_size_t, ctypes.c_char_p,
        ]
        lib.tg_binding_vk_from_deltas.argtypes = [
            ctypes.c_char_p, ctypes.c_size_t, ctypes.c_char_p,
        ]
        lib.tg_tx_digest.argtypes = [ctypes.c_char_p, ctypes.c_size_t] * 4 + [ctypes.c_char_p]
        lib.tg_tx_wire_check.argtypes = [
            ctypes.c_char_p, ctypes.c_size_t, ctypes.c_uint32,
            ctypes.POINTER(ctypes.c_uint32), ctypes.POINTER(ctypes.c_uint32),
        ]
        lib.tg_tx_verify.argtypes = [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_size_t]
        lib.tg_prof_enable.argtypes = [ctypes.c_void_p, ctypes.c_int]
        lib.tg_prof_enable.restype = None
        lib.tg_prof_reset.argtypes = [ctypes.c_void_p]
        lib.tg_prof_reset.restype = None
        lib.tg_prof_get.argtypes = [
            ctypes.c_void_p, ctypes.c_char_p,
            ctypes.POINTER(ctypes.c_double), ctypes.POINTER(ctypes.c_long),
        ]
        lib.tg_synchronize.argtypes = [ctypes.c_void_p]
        _lib = lib
    return _lib


def binding_sign(sk: bytes, msg: bytes, rng_seed: bytes) -> bytes:
    """RedDSA binding signature (TaigaBinding instantiation) — ctx-free
    host call (wire layer)."""
    lib = load_library()
    sig = ctypes.create_string_buffer(64)
    rc = lib.tg_binding_sign(sk, msg, len(msg), rng_seed, sig)
    if rc != 0:
        raise TaigaGpuError(rc)
    return sig.raw


def binding_verify(vk: bytes, msg: bytes, sig: bytes) -> bool:
    return load_library().tg_binding_verify(vk, msg, len(msg), sig) == 0


def binding_vk(sk: bytes) -> bytes:
    lib = load_library()
    out = ctypes.create_string_buffer(32)
    rc = lib.tg_binding_vk(sk, out)
    if rc != 0:
        raise TaigaGpuError(rc)
    return out.raw


def tx_digest(nullifiers, output_cms, delta_cms, anchors) -> bytes:
    """Transaction::digest layout (transaction.rs:116-158)."""
    lib = load_library()
    out = ctypes.create_string_buffer(32)
    lib.tg_tx_digest(b"".join(nullifiers), len(nullifiers),
                     b"".join(output_cms), len(output_cms),
                     b"".join(delta_cms), len(delta_cms),
                     b"".join(anchors), len(anchors), out)
    return out.raw


def tx_wire_check(tx: bytes, vk_len: int):
    """Parse a borsh Transaction + verify its binding signature (ctx-free).
    Returns (n_sptx, n_compliance_proofs); raises on structural errors;
    returns None counts with valid=False on a bad signature."""
    lib = load_library()
    n_sptx = ctypes.c_uint32()
    n_proofs = ctypes.c_uint32()
    rc = lib.tg_tx_wire_check(tx, len(tx), vk_len, ctypes.byref(n_sptx),
                              ctypes.byref(n_proofs))
    if rc == 0:
        return True, n_sptx.value, n_proofs.value
    if rc == -1 or rc <= -200:
        return False, None, None
    raise TaigaGpuError(rc)


def device_count():
    return load_library().tg_device_count()


class TaigaGpu:
    """One proving context on one GPU (one ctx per device)."""

    def __init__(self, device=0):
        lib = load_library()
        h = ctypes.c_void_p()
        rc = lib.tg_init(device, ctypes.byref(h))
        if rc != 0:
            raise TaigaGpuError(rc, lib.tg_error_string(None) or b"")
        self._lib = lib
        self._h = h

    def _ck(self, rc):
        if rc != 0:
            raise TaigaGpuError(rc, (self._lib.tg_error_string(self._h) or b"").decode())

    def close(self):
        if self._h:
            self._lib.tg_destroy(self._h)
            self._h = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass

    # --- SRS ---
    def load_srs(self, params_bytes: bytes):
        self._ck(self._lib.tg_load_srs(self._h, params_bytes, len(params_bytes)))

    @property
    def srs_k(self):
        k = self._lib.tg_srs_k(self._h)
        if k < 0:
            raise TaigaGpuError(k)
        return k

    # --- MSM ---
    def bases_upload(self, points_xy: bytes):
        self._ck(self._lib.tg_bases_upload(self._h, points_xy, len(points_xy) // 64))

    def scalars_upload(self, scalars: bytes):
        self._ck(self._lib.tg_scalars_upload(self._h, scalars, len(scalars) // 32))

    def gen_bases(self, n: int, seed: int = 0):
        self._ck(self._lib.tg_gen_bases(self._h, n, seed))

    def bases_download(self, n: int) -> bytes:
        out = ctypes.create_string_buffer(64 * n)
        self._ck(self._lib.tg_bases_download(self._h, n, out))
        return out.raw

    def msm(self, scalars: bytes, base_set=0) -> bytes:
        out = ctypes.create_string_buffer(64)
        self._ck(self._lib.tg_msm_pallas(self._h, scalars, len(scalars) // 32, base_set, out))
        return out.raw

    def msm_resident(self, n: int, base_set=0) -> bytes:
        out = ctypes.create_string_buffer(64)
        self._ck(self._lib.tg_msm_resident(self._h, n, base_set, out))
        return out.raw

    # --- NTT ---
    def ntt(self, data: bytes, k: int, inverse=False, coset=False) -> bytes:
        buf = ctypes.create_string_buffer(data, len(data))
        self._ck(self._lib.tg_ntt_fp(self._h, 1 if inverse else 0, k, 1 if coset else 0, buf))
        return buf.raw

    def poly_upload(self, data: bytes, k: int):
        self._ck(self._lib.tg_poly_upload(self._h, data, k))

    def ntt_resident(self, k: int, inverse=False, coset=False):
        self._ck(self._lib.tg_ntt_resident(self._h, 1 if inverse else 0, k, 1 if coset else 0))

    def poly_download(self, k: int) -> bytes:
        buf = ctypes.create_string_buffer(32 << k)
        self._ck(self._lib.tg_poly_download(self._h, buf, k))
        return buf.raw

    # --- proving ---
    def keygen(self, desc: bytes) -> int:
        rc = self._lib.tg_keygen(self._h, desc, len(desc))
        if rc < 0:
            raise TaigaGpuError(rc, (self._lib.tg_error_string(self._h) or b"").decode())
        return rc  # slot id

    def select_key(self, slot: int):
        self._ck(self._lib.tg_select_key(self._h, slot))

    def create_proof(self, inst_seed: bytes, wit_seed: bytes, rng_seed: bytes) -> bytes:
        out = ctypes.create_string_buffer(1 << 16)
        out_len = ctypes.c_size_t()
        self._ck(self._lib.tg_create_proof(
            self._h, inst_seed, wit_seed, rng_seed, out, len(out), ctypes.byref(out_len)))
        return out.raw[: out_len.value]

    def create_proof_raw(self, instance: bytes, advice: bytes, rng_seed: bytes) -> bytes:
        out = ctypes.create_string_buffer(1 << 16)
        out_len = ctypes.c_size_t()
        self._ck(self._lib.tg_create_proof_raw(
            self._h, instance, advice, rng_seed, out, len(out), ctypes.byref(out_len)))
        return out.raw[: out_len.value]

    def witness_program_load(self, tgw: bytes):
        """Attach a TGW1 witness-synthesis program to the active key."""
        self._ck(self._lib.tg_witness_program_load(self._h, tgw, len(tgw)))

    def compliance_prove(self, info_borsh: bytes, rng_seed: bytes):
        """Drop-in ComplianceInfo::build + Proof::create; returns
        (proof_bytes, instance_bytes 9x32)."""
        out = ctypes.create_string_buffer(1 << 16)
        out_len = ctypes.c_size_t()
        inst = ctypes.create_string_buffer(288)
        self._ck(self._lib.tg_compliance_prove(
            self._h, info_borsh, len(info_borsh), rng_seed, out, len(out),
            ctypes.byref(out_len), inst))
        return out.raw[: out_len.value], inst.raw

    def rl_prove(self, witness_borsh: bytes, pad_rseed: bytes, rng_seed: bytes):
        """Drop-in TrivialRL get_verifying_info; returns
        (proof_bytes, instance_bytes 22x32)."""
        out = ctypes.create_string_buffer(1 << 16)
        out_len = ctypes.c_size_t()
        inst = ctypes.create_string_buffer(704)
        self._ck(self._lib.tg_rl_prove(
            self._h, witness_borsh, len(witness_borsh), pad_rseed, rng_seed,
            out, len(out), ctypes.byref(out_len), inst))
        return out.raw[: out_len.value], inst.raw

    def witness_synthesize(self, kind: int, borsh: bytes, pad_rseed: bytes,
                           n_advice: int, n: int, n_inst_rows: int):
        adv = ctypes.create_string_buffer(32 * n_advice * n)
        inst = ctypes.create_string_buffer(32 * n_inst_rows)
        self._ck(self._lib.tg_witness_synthesize(
            self._h, kind, borsh, len(borsh), pad_rseed, adv, inst))
        return adv.raw, inst.raw

    def verify_proof(self, inst_seed: bytes, proof: bytes) -> bool:
        rc = self._lib.tg_verify_proof(self._h, inst_seed, proof, len(proof))
        if rc == 0:
            return True
        # -1 = final check failed; -1xx = malformed/truncated transcript
        if rc == -1 or rc <= -100:
            return False
        raise TaigaGpuError(rc, (self._lib.tg_error_string(self._h) or b"").decode())

    def verify_proof_raw(self, instance: bytes, proof: bytes) -> bool:
        rc = self._lib.tg_verify_proof_raw(self._h, instance, proof, len(proof))
        if rc == 0:
            return True
        if rc == -1 or rc <= -100:
            return False
        raise TaigaGpuError(rc, (self._lib.tg_error_string(self._h) or b"").decode())

    def verify_batch(self, items) -> bool:
        """Batch-verify [(inst_seed, proof), ...] in one combined IPA check
        (one shared g-sized GPU MSM for all proofs — SURVEY §8f-3)."""
        m = len(items)
        seeds = b"".join(i for i, _ in items)
        proofs = b"".join(p for _, p in items)
        lens = (ctypes.c_size_t * m)(*[len(p) for _, p in items])
        rc = self._lib.tg_verify_batch(self._h, m, seeds, proofs, lens)
        if rc == 0:
            return True
        if rc == -1 or rc <= -100:
            return False
        raise TaigaGpuError(rc, (self._lib.tg_error_string(self._h) or b"").decode())

    def verify_batch_raw(self, items) -> bool:
        """Batch-verify [(instance_rows_bytes, proof), ...] (raw instances,
        active key) in one combined IPA check."""
        m = len(items)
        insts = b"".join(i for i, _ in items)
        proofs = b"".join(p for _, p in items)
        lens = (ctypes.c_size_t * m)(*[len(p) for _, p in items])
        rc = self._lib.tg_verify_batch_raw(self._h, m, insts, proofs, lens)
        if rc == 0:
            return True
        if rc == -1 or rc <= -100 or rc == -3:
            return False
        raise TaigaGpuError(rc, (self._lib.tg_error_string(self._h) or b"").decode())

    def tx_verify(self, tx: bytes) -> bool:
        """Full transaction verification against the active key: wire
        check + binding signature + one combined batch verification of all
        compliance proofs (SURVEY §8f-4 + §8f-3 composed)."""
        rc = self._lib.tg_tx_verify(self._h, tx, len(tx))
        if rc == 0:
            return True
        if rc == -1 or rc <= -100:
            return False
        if rc == -3:
            return False
        raise TaigaGpuError(rc, (self._lib.tg_error_string(self._h) or b"").decode())

    def poseidon_hash(self, msgs: bytes, n: int, L: int) -> bytes:
        """Batched Poseidon P128Pow5T3 ConstantLength<L>: n messages of L
        canonical 32B field reprs -> n 32B digests (one hash per thread)."""
        out = ctypes.create_string_buffer(32 * n)
        self._ck(self._lib.tg_poseidon_hash(self._h, msgs, n, L, out))
        return out.raw

    def witness_hash(self, inst_seed: bytes, wit_seed: bytes) -> bytes:
        out = ctypes.create_string_buffer(32)
        self._ck(self._lib.tg_witness_hash(self._h, inst_seed, wit_seed, out))
        return out.raw

    # --- profiling ---
    def prof_enable(self, on=True):
        self._lib.tg_prof_enable(self._h, 1 if on else 0)

    def prof_reset(self):
        self._lib.tg_prof_reset(self._h)

    def prof_get(self, name: str):
        ms = ctypes.c_double()
        cnt = ctypes.c_long()
        self._ck(self._lib.tg_prof_get(self._h, name.encode(), ctypes.byref(ms), ctypes.byref(cnt)))
        return ms.value, cnt.value

    def synchronize(self):
        self._ck(self._lib.tg_synchronize(self._h))
