"""Golden-byte pins for the wire formats (regression guards: the byte
layouts are the protocol contract — reference
rust/xaynet-core/src/mask/object/serialization/, message/message.rs).

Layout cross-checks are structural (documented offsets); the hex snapshots
pin this implementation's bytes so codec changes can't slip through."""
import numpy as np

from xaynet_amd import _core

mk = _core.mask
msgmod = _core.message


def test_mask_config_wire_4_bytes():
    c = mk.MaskConfig(1, 0, 0, 3)  # Prime/F32/B0/M3
    b = bytes(c.to_bytes())
    assert b == bytes.fromhex("01000003")  # group, dtype, bound, model
    assert mk.MaskConfig.from_bytes(b).order == c.order


def test_mask_object_wire_layout_and_golden():
    pair = mk.MaskConfigPair(mk.MaskConfig(1, 0, 0, 3), mk.MaskConfig(1, 0, 0, 3))
    m = mk.derive_mask(b"\x2a" * 32, 4, pair)
    wire = bytes(m.serialize())
    # MaskVect: config(4) | count u32 BE | count*bpn LE limbs; bpn=6
    assert wire[:4] == bytes.fromhex("01000003")
    assert int.from_bytes(wire[4:8], "big") == 4
    # MaskUnit: config(4) | one limb
    unit_off = 8 + 4 * 6
    assert wire[unit_off : unit_off + 4] == bytes.fromhex("01000003")
    assert len(wire) == unit_off + 4 + 6
    # golden snapshot (ChaCha20 stream is fixed by the seed)
    assert wire.hex() == (
        "0100000300000004e498746805083987e832d90fa3942b06a601618525b5a40a"
        "01000003445436978803"
    )


def test_sum_message_wire_golden():
    payload = b"\x33" * 64 + b"\x44" * 32
    wire = bytes(msgmod.encode(msgmod.TAG_SUM, payload, b"\x11" * 32, b"\x22" * 32)[0])
    # header: sig(64) | participant_pk(32) | coordinator_pk(32) |
    # length u32 BE | tag | flags | reserved(2)
    assert len(wire) == 136 + 96
    assert wire[96:128] == b"\x22" * 32  # coordinator pk
    assert int.from_bytes(wire[128:132], "big") == len(wire)
    assert wire[132] == 1 and wire[133] == 0  # Sum, no flags
    assert wire[136 : 136 + 64] == b"\x33" * 64
    assert msgmod.verify(wire)
    # deterministic signature (Ed25519 from the fixed seed)
    assert wire[:16].hex() == "a63b86f0df443b5765a6774e4689dafc"


def test_masked_model_wire_golden():
    pair = mk.MaskConfigPair(mk.MaskConfig(1, 0, 0, 3), mk.MaskConfig(1, 0, 0, 3))
    w = np.array([0.5, -0.25], dtype=np.float32)
    masked = mk.mask_model(b"\x07" * 32, mk.Scalar(1, 2), w, pair)
    assert bytes(masked.serialize()).hex() == (
        "010000030000000212ef63003f0570b0a8b8e40c01000003335479363f07"
    )
    # oracle agreement pins the fast path too
    oracle = mk.mask_model_oracle(b"\x07" * 32, mk.Scalar(1, 2), w, pair)
    assert bytes(oracle.serialize()) == bytes(masked.serialize())


def test_local_seed_dict_entry_is_112_bytes():
    # update payload layout invariant (message/traits.rs: 32 pk + 80 sealed
    # seed; sealed = 48 sealbox overhead + 32 seed)
    from xaynet_amd import _core as c

    pk, sk = c.crypto.box_keypair()
    sealed = c.crypto.sealbox_seal(b"\x00" * 32, pk)
    assert len(sealed) == 80
