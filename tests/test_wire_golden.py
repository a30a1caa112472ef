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


def test_fast_model_codec_byte_identical_to_rational_path():
    """encode_model (direct dyadic emit) must produce byte-identical bincode
    to the exact-rational path (decode -> generic re-encode round-trips to
    the same bytes), across adversarial values and dtypes."""
    import numpy as np

    sdk = _core.sdk
    rng = np.random.default_rng(13)
    cases = [
        np.array([0.0, -0.0, 1.0, -1.0, 0.5, -0.75, 1 / 3, 2e38, -2e38,
                  1e-45, -1e-45, 2.0, -65536.0, 123.456], dtype=np.float32),
        rng.uniform(-1e6, 1e6, 4000).astype(np.float32),
        (rng.uniform(-1, 1, 1000) * 1e-30).astype(np.float32),
        rng.uniform(-1e12, 1e12, 2000).astype(np.float64),
        np.array([5e-324, -5e-324, 1e300], dtype=np.float64),
        rng.integers(-2**31, 2**31, 1000).astype(np.int32),
        rng.integers(-2**62, 2**62, 1000).astype(np.int64),
        np.array([0, -1, 1, -2**63, 2**63 - 1], dtype=np.int64),
    ]
    for arr in cases:
        fast = bytes(sdk.encode_model(arr))
        assert bytes(sdk.reencode_model_slow(fast)) == fast, arr.dtype
        if arr.dtype in (np.float32, np.float64):
            back = sdk.decode_model(fast, 0 if arr.dtype == np.float32 else 1)
            assert np.array_equal(back, arr)


def test_decode_model_adversarial_bodies():
    """decode_model parses untrusted HTTP bodies: truncations, corrupt
    lengths and random bit flips over a large (MT-decode-path) body must
    never crash — they either fail cleanly (None) or fall back to the
    generic rational decoder. The multi-thread decoder's boundary scan
    must reject structurally-broken bodies before any chunk decode runs."""
    import numpy as np

    from xaynet_amd import _core

    sdk = _core.sdk
    rng = np.random.default_rng(99)
    # large enough to take the multi-thread scan+decode path (>= 2^20 elems)
    w = rng.uniform(-1, 1, 1 << 20).astype(np.float32)
    body = bytearray(sdk.encode_model(w))

    # truncations at assorted depths
    for cut in [0, 1, 8, 9, 100, len(body) // 2, len(body) - 1]:
        out = sdk.decode_model(bytes(body[:cut]), 0)
        assert out is None or len(out) != len(w) or not np.array_equal(out, w) or cut == len(body)

    # corrupt the element count (header u64 at offset 1)
    b = bytearray(body)
    b[1:9] = (2**40).to_bytes(8, "little")
    assert sdk.decode_model(bytes(b), 0) is None

    # random byte corruptions sprinkled through digit-length fields
    for _ in range(32):
        b = bytearray(body)
        pos = int(rng.integers(9, len(b)))
        b[pos] ^= int(rng.integers(1, 256))
        sdk.decode_model(bytes(b), 0)  # must not crash; value may differ

    # oversized digit count in the first element's numerator
    b = bytearray(body)
    b[13:21] = (2**20).to_bytes(8, "little")  # nd > 2^16 bound
    assert sdk.decode_model(bytes(b), 0) is None

    # intact body still round-trips after all this
    assert np.array_equal(sdk.decode_model(bytes(body), 0), w)


def test_encode_model_mt_threshold_identity():
    """The chunked multi-thread encoder must stay byte-identical to the
    serial encoder across its activation threshold and at sizes that leave
    uneven final chunks."""
    import numpy as np

    from xaynet_amd import _core

    sdk = _core.sdk
    rng = np.random.default_rng(5)
    for n in [(1 << 19) - 1, 1 << 19, (1 << 19) + 1, (1 << 19) + 17 * 13]:
        w = rng.uniform(-1e3, 1e3, n).astype(np.float32)
        assert bytes(sdk.encode_model(w)) == bytes(sdk.encode_model_f32(w))
