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


def test_multipart_chunks_out_of_order():
    """Chunks may arrive in any order (reference MessageBuilder keeps a
    BTreeMap and completes when all ids are present, multipart/service.rs):
    deliver the LAST-flagged chunk first and the message must still
    reassemble and register the sum participant."""
    import threading

    msgmod = _core.message
    cr = _core.crypto

    s = co.Settings()
    s.sum_prob = 0.99
    s.update_prob = 0.99
    s.model_length = 8
    c = mk.MaskConfig(1, 0, 0, 3)
    s.mask_cfg = mk.MaskConfigPair(c, c)
    s.set_sum(1, 10, 5.0, 30.0)
    s.set_update(1, 10, 5.0, 30.0)
    s.set_sum2(1, 10, 5.0, 30.0)
    coord = co.Coordinator(s, co.InMemoryStorage(), co.InMemoryModels(), False)
    coord.run_one_phase()  # Idle -> Sum
    params = bytes(coord.fetch_round_params())
    cpk, seed_round = params[:32], params[48:80]
    E = co.PipelineError

    rng = np.random.default_rng(17)
    # a sum-eligible participant (sum_prob=0.99: nearly any seed works)
    for _ in range(200):
        sgn = bytes(rng.integers(0, 256, 32, dtype=np.uint8))
        pk, sk = _core.crypto.sign_keypair_from_seed(sgn)
        sum_sig = cr.sign_detached(seed_round + b"sum", sk)
        if cr.is_eligible(sum_sig, 0.99):
            break
    ephm_pk, _ = cr.box_keypair()
    payload = bytes(sum_sig) + ephm_pk  # 96 B
    parts = msgmod.encode(msgmod.TAG_SUM, payload, sgn, cpk, max_payload=40)
    assert len(parts) == 3

    t = threading.Thread(target=coord.run_one_phase, daemon=True)
    t.start()
    time.sleep(0.05)
    # deliver LAST chunk first, then 0, then 1 — the final delivery completes
    order = [2, 0, 0, 1]  # duplicate of chunk 0 must be harmless
    results = [coord.handle_message_bytes(bytes(parts[i])) for i in order]
    assert all(r == int(E.Ok) for r in results[:-1])  # buffered (+dup)
    assert results[-1] == int(E.Ok)  # reassembled + accepted
    t.join(15)
    assert coord.phase == co.PhaseId.Update  # the summer was registered
    coord.stop()
