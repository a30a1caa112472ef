"""Property-based tests (hypothesis): the fast masker tracks the exact
oracle on arbitrary inputs, and wire codecs round-trip."""
import numpy as np
from hypothesis import given, settings, strategies as st

from xaynet_amd import _core

mk = _core.mask
msgmod = _core.message


@st.composite
def f32_arrays(draw):
    n = draw(st.integers(1, 64))
    elems = st.one_of(
        st.floats(-1e6, 1e6, width=32),
        st.floats(-1.5, 1.5, width=32),
        st.sampled_from([0.0, -0.0, 1.0, -1.0, float("inf"), float("-inf"),
                         float("nan"), 1e-45, -1e-45, 3.4e38]),
    )
    return np.array(draw(st.lists(elems, min_size=n, max_size=n)), dtype=np.float32)


@settings(max_examples=120, deadline=None)
@given(w=f32_arrays(), num=st.integers(1, 2**32), den=st.integers(1, 2**32),
       group=st.integers(0, 2), bound=st.sampled_from([0, 2, 4]))
def test_fast_masker_equals_oracle_property(w, num, den, group, bound):
    if num > den:
        num, den = den, num  # scalar in (0, 1]
    c = mk.MaskConfig(group, 0, bound, 3)
    pair = mk.MaskConfigPair(c, c)
    seed = (num * 2654435761 % 2**128).to_bytes(16, "little") * 2
    fast = mk.mask_model(seed, mk.Scalar(num, den), w, pair)
    oracle = mk.mask_model_oracle(seed, mk.Scalar(num, den), w, pair)
    assert bytes(fast.serialize()) == bytes(oracle.serialize())


@settings(max_examples=60, deadline=None)
@given(seed=st.binary(min_size=32, max_size=32), n=st.integers(0, 40),
       dtype=st.integers(0, 3))
def test_mask_object_wire_roundtrip_property(seed, n, dtype):
    c = mk.MaskConfig(1, dtype, 0, 3)
    pair = mk.MaskConfigPair(c, c)
    obj = mk.derive_mask(seed, n, pair)
    wire = bytes(obj.serialize())
    back = mk.MaskObject.deserialize(wire)
    assert back is not None
    assert bytes(back.serialize()) == wire
    for i in range(0, n, 7):
        assert int(back.element(i)) == int(obj.element(i))


@settings(max_examples=60, deadline=None)
@given(payload_extra=st.integers(0, 200), max_payload=st.integers(16, 300),
       seed=st.binary(min_size=32, max_size=32))
def test_message_chunking_roundtrips_header_property(payload_extra, max_payload, seed):
    payload = bytes(range(256))[:96] + b"\x55" * 0  # sum payload is fixed 96 B
    parts = msgmod.encode(msgmod.TAG_SUM, payload, seed, b"\x10" * 32,
                          max_payload=max_payload)
    assert len(parts) >= 1
    total = 0
    for part in parts:
        hdr = msgmod.parse_header(part)
        assert hdr["length"] == len(part)
        assert msgmod.verify(part)
        if len(parts) > 1:
            assert hdr["flags"] & msgmod.FLAG_MULTIPART
            total += len(part) - 136 - msgmod.CHUNK_OVERHEAD
    if len(parts) > 1:
        assert total == 96  # chunk data re-assembles exactly to the payload


@settings(max_examples=200, deadline=None)
@given(data=st.binary(min_size=0, max_size=600))
def test_message_parser_never_crashes_on_garbage(data):
    """Random bytes through the full ingest pipeline: typed errors only."""
    from xaynet_amd import _core as c

    co = c.coordinator
    s = co.Settings()
    s.model_length = 4
    cfg = c.mask.MaskConfig(1, 0, 0, 3)
    s.mask_cfg = c.mask.MaskConfigPair(cfg, cfg)
    coord = _garbage_coord(co, s)
    r1 = coord.handle_message_bytes(data)
    r2 = coord.handle_encrypted_message(data)
    assert r1 != int(co.PipelineError.Ok)
    assert r2 != int(co.PipelineError.Ok)


def _garbage_coord(co, s, _cache={}):
    if "c" not in _cache:
        s.set_sum(1, 10, 0.05, 5.0)
        s.set_update(3, 10, 0.05, 5.0)
        s.set_sum2(1, 10, 0.05, 5.0)
        c = co.Coordinator(s, co.InMemoryStorage(), co.InMemoryModels(), False)
        c.run_one_phase()
        _cache["c"] = c
    return _cache["c"]


@settings(max_examples=120, deadline=None)
@given(flip=st.integers(0, 10_000), seed=st.binary(min_size=32, max_size=32))
def test_mutated_valid_message_never_crashes(flip, seed):
    """Bit-flipped VALID messages (the adversarial surface): typed errors,
    no crash, no acceptance of a corrupted signature."""
    from xaynet_amd import _core as c

    payload = b"\x33" * 64 + b"\x44" * 32
    wire = bytearray(c.message.encode(c.message.TAG_SUM, payload, seed, b"\x22" * 32)[0])
    pos = flip % (len(wire) * 8)
    wire[pos // 8] ^= 1 << (pos % 8)
    co = c.coordinator
    s = co.Settings()
    s.model_length = 4
    cfg = c.mask.MaskConfig(1, 0, 0, 3)
    s.mask_cfg = c.mask.MaskConfigPair(cfg, cfg)
    coord = _garbage_coord(co, s)
    r = coord.handle_message_bytes(bytes(wire))
    # flipped anywhere in sig/pk/payload: never accepted (coordinator pk
    # check fires first here since the test coordinator has its own keys)
    assert r != int(co.PipelineError.Ok)
