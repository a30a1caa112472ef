"""Multipart end-to-end: a model large enough that update messages exceed the
max message size must be chunked by the SDK encoder (reference
message_encoder/chunker.rs) and reassembled by the coordinator's multipart
handler (services/messages/multipart) — the round still completes and the
unmasked model matches."""
import time

import numpy as np

from xaynet_amd import _core

co = _core.coordinator
sdk = _core.sdk
mk = _core.mask


def test_round_with_multipart_updates():
    # model_length 600 at bpn=6 -> update payload ~3.7 KB plus seed dict;
    # max_message_size 1024 forces >= 4 chunks per update message
    n, length = 8, 600
    s = co.Settings()
    s.sum_prob = 0.5
    s.update_prob = 1.0
    s.model_length = length
    c = mk.MaskConfig(1, 0, 0, 3)
    s.mask_cfg = mk.MaskConfigPair(c, c)
    s.set_sum(1, 100, 0.05, 10.0)
    s.set_update(3, 100, 0.05, 10.0)
    s.set_sum2(1, 100, 0.05, 10.0)
    coord = co.Coordinator(s, co.InMemoryStorage(), co.InMemoryModels(), False)
    client = sdk.InProcessClient(coord)
    rng = np.random.default_rng(31)
    participants = [
        sdk.Participant(
            bytes(rng.integers(0, 256, 32, dtype=np.uint8)), 1, 1, client,
            max_message_size=1024,
        )
        for _ in range(n)
    ]
    weights = [rng.uniform(-1, 1, length).astype(np.float32) for _ in range(n)]

    coord.start()
    t0 = time.time()
    model = None
    try:
        while time.time() - t0 < 90.0:
            for i, p in enumerate(participants):
                p.tick()
                if p.should_set_model:
                    p.set_model(weights[i])
            body = coord.fetch_model()
            if body and body[0] == 1:
                model = sdk.decode_model(body, 0)
                break
            time.sleep(0.005)
    finally:
        coord.stop()
    assert model is not None, "no global model with multipart updates"
    assert model.shape == (length,)
    # sanity: within the clamp bound and non-degenerate
    assert np.abs(model).max() <= 1.0 + 1e-6
    assert np.abs(model).mean() > 1e-3


def test_encode_message_chunk_shapes():
    """The encoder's chunk framing: 136 B header per chunk message, 8 B chunk
    payload header, FLAG_MULTIPART on every chunk, sizes bounded by
    max_payload (reference message_encoder/{encoder,chunker}.rs)."""
    msgmod = _core.message
    payload = b"\x33" * 64 + b"\x44" * 32  # a sum payload (sig + ephm pk)
    seed, cpk = b"\x11" * 32, b"\x22" * 32

    # fits: single signed message, no multipart flag
    parts = msgmod.encode(msgmod.TAG_SUM, payload, seed, cpk, max_payload=512)
    assert len(parts) == 1
    hdr = msgmod.parse_header(parts[0])
    assert hdr["tag"] == msgmod.TAG_SUM and hdr["flags"] & msgmod.FLAG_MULTIPART == 0
    assert hdr["length"] == len(parts[0]) == 136 + 96
    assert msgmod.verify(parts[0])

    # forced chunking: payload 96 B with max_payload 40 -> 3 chunks
    parts = msgmod.encode(msgmod.TAG_SUM, payload, seed, cpk, max_payload=40)
    assert len(parts) == 3
    for part in parts:
        hdr = msgmod.parse_header(part)
        assert hdr["flags"] & msgmod.FLAG_MULTIPART
        assert len(part) <= 136 + msgmod.CHUNK_OVERHEAD + 40
        assert msgmod.verify(part)
