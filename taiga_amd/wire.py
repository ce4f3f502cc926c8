"""Borsh wire-format builders for Taiga transactions.

Encode-side mirror of the reference layouts (decode/verify lives in
taiga_amd/csrc/tx_wire.hpp — see its header for the field-by-field layout
citations): Transaction (transaction.rs:24-33), ShieldedPartialTransaction
(shielded_ptx.rs:272-320), ComplianceVerifyingInfo (shielded_ptx.rs:47-50),
ResourceLogicVerifyingInfo[Set] (resource_logic_circuit.rs:175-188,
shielded_ptx.rs:57-60), CompliancePublicInputs (compliance.rs:82-93).

All vectors are borsh u32-LE-count prefixed; field/point/hash encodings are
raw 32-byte blocks; the binding signature is 64 raw bytes.
"""
import struct


def _vec(items: list) -> bytes:
    return struct.pack("<I", len(items)) + b"".join(items)


def _bytes_vec(b: bytes) -> bytes:
    return struct.pack("<I", len(b)) + b


def compliance_instance(anchor: bytes, nf: bytes, cm: bytes, delta: bytes,
                        rl_cm_in: bytes = b"\x00" * 32,
                        rl_cm_out: bytes = b"\x00" * 32) -> bytes:
    """CompliancePublicInputs: 6 x 32B, compliance.rs:82-93 order."""
    parts = [anchor, nf, cm, delta, rl_cm_in, rl_cm_out]
    assert all(len(p) == 32 for p in parts)
    return b"".join(parts)


def compliance_info(proof: bytes, instance192: bytes) -> bytes:
    """ComplianceVerifyingInfo = Proof(Vec<u8>) ‖ 192B instance."""
    assert len(instance192) == 6 * 32
    return _bytes_vec(proof) + instance192


def rl_info(vk_bytes: bytes, proof: bytes, public_inputs: list) -> bytes:
    """ResourceLogicVerifyingInfo = raw vk ‖ Proof(Vec<u8>) ‖ 22 x 32B."""
    assert len(public_inputs) == 22 and all(len(p) == 32 for p in public_inputs)
    return vk_bytes + _bytes_vec(proof) + b"".join(public_inputs)


def rl_set(app_info: bytes, dynamic_infos: list = ()) -> bytes:
    return app_info + _vec(list(dynamic_infos))


def shielded_ptx(compliances: list, inputs: list, outputs: list,
                 binding_sig_r: bytes = None, hints: bytes = b"") -> bytes:
    out = _vec(compliances) + _vec(inputs) + _vec(outputs)
    if binding_sig_r is None:
        out += b"\x00"
    else:
        assert len(binding_sig_r) == 32
        out += b"\x01" + binding_sig_r
    out += _bytes_vec(hints)
    return out


def transaction(shielded_ptxs: list, signature64: bytes) -> bytes:
    """Transaction = shielded bundle ‖ transparent bundle (empty) ‖ 64B sig."""
    assert len(signature64) == 64
    return _vec(shielded_ptxs) + _vec([]) + signature64
