"""Redis coordinator/model storage backend against a live RESP server (the
mini-RESP stub; a real redis-server speaks the same protocol). Covers the
reference's atomic Lua-script semantics (redis/mod.rs:208-339) expressed as
WATCH/MULTI/EXEC, full e2e rounds on Redis state, auto-reconnect, and
kill-and-restore."""
import time

import numpy as np
import pytest

from xaynet_amd import _core

from resp_stub import RespStubServer

co = _core.coordinator
mk = _core.mask
sdk = _core.sdk


@pytest.fixture
def stub():
    s = RespStubServer()
    s.start()
    yield s
    s.stop()


def make_settings(length=16):
    s = co.Settings()
    s.sum_prob = 0.5
    s.update_prob = 1.0
    s.model_length = length
    c = mk.MaskConfig(1, 0, 0, 3)
    s.mask_cfg = mk.MaskConfigPair(c, c)
    s.set_sum(1, 100, 0.05, 10.0)
    s.set_update(3, 100, 0.05, 10.0)
    s.set_sum2(1, 100, 0.05, 10.0)
    return s


def test_storage_ops_roundtrip(stub):
    st = co.RedisStorage("127.0.0.1", stub.port)
    assert st.is_ready()
    # coordinator state bytes
    assert st.set_coordinator_state(b"\x01\x02\xff\x00state")
    assert bytes(st.coordinator_state()) == b"\x01\x02\xff\x00state"
    # sum dict: HSETNX semantics
    pk1, e1 = b"\x01" * 32, b"\xa1" * 32
    pk2, e2 = b"\x02" * 32, b"\xa2" * 32
    assert st.add_sum_participant(pk1, e1) == co.SumPartAddError.Ok
    assert st.add_sum_participant(pk1, e1) == co.SumPartAddError.AlreadyExists
    assert st.add_sum_participant(pk2, e2) == co.SumPartAddError.Ok
    sd = st.sum_dict()
    assert len(sd) == 2 and bytes(sd[pk1]) == e1
    # local seed dicts with the protocol error taxonomy
    upd1 = b"\x11" * 32
    seed1 = b"\xb1" * 80
    E = co.SeedDictAddError
    assert st.add_local_seed_dict(upd1, [(pk1, seed1)]) == E.LengthMisMatch
    assert st.add_local_seed_dict(
        upd1, [(pk1, seed1), (b"\x09" * 32, seed1)]) == E.UnknownSumParticipant
    assert st.add_local_seed_dict(upd1, [(pk1, seed1), (pk2, seed1)]) == E.Ok
    assert st.add_local_seed_dict(upd1, [(pk1, seed1), (pk2, seed1)]) == E.UpdatePkAlreadySubmitted
    upd2 = b"\x12" * 32
    assert st.add_local_seed_dict(upd2, [(pk1, seed1), (pk2, seed1)]) == E.Ok
    seeds = st.seed_dict()
    assert len(seeds) == 2 and len(seeds[pk1]) == 2
    # mask votes
    M = co.MaskScoreIncrError
    assert st.incr_mask_score(b"\x09" * 32, b"maskA") == M.UnknownSumParticipant
    assert st.incr_mask_score(pk1, b"maskA") == M.Ok
    assert st.incr_mask_score(pk1, b"maskA") == M.MaskAlreadySubmitted
    assert st.incr_mask_score(pk2, b"maskA") == M.Ok
    assert st.number_of_unique_masks() == 1
    best = st.best_masks(2)
    assert len(best) == 1 and bytes(best[0][0]) == b"maskA" and best[0][1] == 2
    # model id + dict teardown
    assert st.set_latest_global_model_id("3_abc")
    assert st.latest_global_model_id() == "3_abc"
    assert st.delete_dicts()
    assert st.sum_dict() is not None and len(st.sum_dict()) == 0
    assert st.number_of_unique_masks() == 0
    # state survives delete_dicts (reference semantics)
    assert bytes(st.coordinator_state()) == b"\x01\x02\xff\x00state"


def test_model_storage_refuses_overwrite(stub):
    ms = co.RedisModels("127.0.0.1", stub.port)
    assert ms.is_ready()
    seed = b"\x07" * 32
    mid = ms.set_global_model(4, seed, b"MODELBYTES")
    assert mid == "4_" + "07" * 32
    assert bytes(ms.global_model(mid)) == b"MODELBYTES"
    # same id again -> refused (reference s3.rs:190-198)
    assert ms.set_global_model(4, seed, b"OTHER") is None
    assert bytes(ms.global_model(mid)) == b"MODELBYTES"


def test_auto_reconnect(stub):
    st = co.RedisStorage("127.0.0.1", stub.port, timeout_s=2.0)
    assert st.is_ready()
    # sever every live connection; the next command must transparently
    # reconnect (reference ConnectionManager, redis/mod.rs:95-101)
    stub.kill_connections()
    assert st.set_coordinator_state(b"after-reconnect")
    assert bytes(st.coordinator_state()) == b"after-reconnect"
    # and again, mid-stream
    stub.kill_connections()
    assert st.is_ready()


def test_full_round_on_redis_and_restore(stub):
    """Live PET rounds with all protocol state in Redis, then a coordinator
    'crash': a NEW coordinator restores round id + keys from Redis state."""
    length, n = 16, 8
    s = make_settings(length)
    store = co.RedisStorage("127.0.0.1", stub.port)
    models = co.RedisModels("127.0.0.1", stub.port)
    coord = co.Coordinator(s, store, models, False)
    client = sdk.InProcessClient(coord)
    rng = np.random.default_rng(5)
    participants = [
        sdk.Participant(bytes(rng.integers(0, 256, 32, dtype=np.uint8)), 1, 1, client)
        for _ in range(n)
    ]
    weights = [rng.uniform(-1, 1, length).astype(np.float32) for _ in range(n)]
    coord.start()
    t0 = time.time()
    model = None
    try:
        while time.time() - t0 < 60.0 and model is None:
            for i, p in enumerate(participants):
                p.tick()
                if p.should_set_model:
                    p.set_model(weights[i])
            body = coord.fetch_model()
            if body and body[0] == 1:
                model = np.asarray(sdk.decode_model(body, 0))
            time.sleep(0.005)
    finally:
        rid = coord.round_id
        coord.stop()
    assert model is not None and model.shape == (length,)
    assert np.abs(model).max() <= 1.0 + 1e-5

    # model persisted in Redis under roundid_seedhex
    mid = store.latest_global_model_id()
    assert mid is not None
    model_round = int(mid.split("_")[0])
    assert 1 <= model_round <= rid
    assert models.global_model(mid) is not None

    # crash + restore on a fresh coordinator against the same Redis
    s2 = make_settings(length)
    s2.restore = True
    store2 = co.RedisStorage("127.0.0.1", stub.port)
    coord2 = co.Coordinator(s2, store2, co.RedisModels("127.0.0.1", stub.port), False)
    # restored state resumes the round counter (next Idle continues from it,
    # never restarting at 1)
    coord2.run_one_phase()
    assert coord2.round_id > model_round
    # the restored coordinator re-broadcast the persisted global model
    body = coord2.fetch_model()
    assert body and body[0] == 1
    assert np.allclose(np.asarray(sdk.decode_model(body, 0)), model)
    coord2.stop()
