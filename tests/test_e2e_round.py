"""End-to-end PET rounds: C++ coordinator + C++ participants, in-process.

This is BASELINE.json config #1 ("2 in-process participants, 1k-weight f32
vector, one Sum/Update/Sum2 round on CPU coordinator") generalized: a full
Sum -> Update -> Sum2 -> Unmask round with real crypto, wire messages and
masking; the unmasked global model must equal the mean of the accepted
updaters' models (scalar_sum division, reference semantics).
"""
import time

import numpy as np
import pytest

from xaynet_amd import _core

co = _core.coordinator
sdk = _core.sdk
mk = _core.mask


def make_coordinator(model_length=32, n_expect=3, staged=False):
    s = co.Settings()
    s.sum_prob = 0.5
    s.update_prob = 1.0
    s.model_length = model_length
    c = mk.MaskConfig(1, 0, 0, 3)  # Prime/F32/B0/M3
    s.mask_cfg = mk.MaskConfigPair(c, c)
    # counts: at least 1 summer, n_expect updaters; generous time caps
    s.set_sum(1, 100, 0.05, 10.0)
    s.set_update(n_expect, 100, 0.05, 10.0)
    s.set_sum2(1, 100, 0.05, 10.0)
    store = co.InMemoryStorage()
    models = co.InMemoryModels()
    coord = co.Coordinator(s, store, models, staged)
    return coord, store, models


def run_round(coord, participants, weights, timeout_s=30.0):
    """Tick participants until a global model is published; returns it."""
    coord.start()
    t0 = time.time()
    model = None
    try:
        while time.time() - t0 < timeout_s:
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
    return model


def decode_sum_dict(body: bytes):
    assert body[0] == 1
    n = int.from_bytes(body[1:9], "little")
    out = {}
    off = 9
    for _ in range(n):
        out[body[off : off + 32]] = body[off + 32 : off + 64]
        off += 64
    return out


def decode_update_seed_dict(body: bytes):
    assert body[0] == 1
    n = int.from_bytes(body[1:9], "little")
    out = {}
    off = 9
    for _ in range(n):
        pk = body[off : off + 32]
        off += 32
        slen = int.from_bytes(body[off : off + 8], "little")
        assert slen == 80
        out[pk] = body[off + 8 : off + 88]
        off += 88
    return out


def test_full_round_in_process():
    n, length = 12, 64
    coord, store, models = make_coordinator(model_length=length, n_expect=3)
    client = sdk.InProcessClient(coord)
    rng = np.random.default_rng(7)
    participants = [
        sdk.Participant(bytes(rng.integers(0, 256, 32, dtype=np.uint8)), 1, 1, client)
        for _ in range(n)
    ]
    weights = [rng.uniform(-1, 1, length).astype(np.float32) for _ in range(n)]

    # drive the round, snapshotting the dicts before the next Idle clears them
    coord.start()
    t0 = time.time()
    model, sums_body, seeds_body = None, None, None
    try:
        while time.time() - t0 < 30.0:
            for i, p in enumerate(participants):
                p.tick()
                if p.should_set_model:
                    p.set_model(weights[i])
            sb = coord.fetch_sum_dict()
            if sb and sb[0] == 1:
                sums_body = sb
                some = next(iter(decode_sum_dict(sb)))
                eb = coord.fetch_seeds(some)
                if eb and eb[0] == 1:
                    seeds_body = eb
            body = coord.fetch_model()
            if body and body[0] == 1:
                model = sdk.decode_model(body, 0)
                break
            time.sleep(0.005)
    finally:
        coord.stop()
    assert model is not None, "no global model published within timeout"
    assert model.shape == (length,)

    # the accepted updater set is exactly the seed dict's update pks
    assert sums_body is not None and seeds_body is not None
    accepted = set(decode_update_seed_dict(seeds_body))
    assert len(accepted) >= 3
    by_pk = {p.pk: i for i, p in enumerate(participants)}
    idx = [by_pk[pk] for pk in accepted]
    expect = np.mean([weights[i].astype(np.float64) for i in idx], axis=0)
    assert np.abs(model.astype(np.float64) - expect).max() < 1e-4


def test_round_robin_multiple_rounds():
    n, length = 8, 16
    coord, store, models = make_coordinator(model_length=length, n_expect=2)
    client = sdk.InProcessClient(coord)
    rng = np.random.default_rng(3)
    participants = [
        sdk.Participant(bytes(rng.integers(0, 256, 32, dtype=np.uint8)), 1, 1, client)
        for _ in range(n)
    ]
    weights = [np.full(length, float(i), dtype=np.float32) for i in range(n)]

    coord.start()
    seen_models = []
    t0 = time.time()
    try:
        while time.time() - t0 < 60.0 and len(seen_models) < 2:
            for i, p in enumerate(participants):
                p.tick()
                if p.should_set_model:
                    p.set_model(weights[i])
            body = coord.fetch_model()
            if body and body[0] == 1 and (not seen_models or body != seen_models[-1]):
                seen_models.append(body)
            time.sleep(0.005)
    finally:
        coord.stop()
    assert len(seen_models) >= 2, "coordinator did not complete two rounds"


def test_checkpoint_roundtrip():
    coord, store, models = make_coordinator()
    st = coord.checkpoint_state()
    assert len(st) > 100
    coord2, _, _ = make_coordinator()
    assert coord2.restore_state(st)
    assert coord2.checkpoint_state()[64:] == st[64:]  # keys regenerated? no: restored
