"""Failure-phase behavior: storage faults push the round into Failure, the
coordinator restarts a fresh round once storage is healthy, and participants
follow via the round-freshness check (reference phases/failure.rs:31-112 +
mockall-storage phase tests)."""
import time

import numpy as np
import pytest

from xaynet_amd import _core

co = _core.coordinator
sdk = _core.sdk
mk = _core.mask


def make_coord(store, n_expect=3, length=16):
    s = co.Settings()
    s.sum_prob = 0.5
    s.update_prob = 1.0
    s.model_length = length
    c = mk.MaskConfig(1, 0, 0, 3)
    s.mask_cfg = mk.MaskConfigPair(c, c)
    s.set_sum(1, 100, 0.02, 0.5)
    s.set_update(n_expect, 100, 0.02, 0.5)
    s.set_sum2(1, 100, 0.02, 0.5)
    return co.Coordinator(s, store, co.InMemoryModels(), False)


def test_idle_state_persist_failure_goes_failure_then_recovers():
    store = co.FaultInjectionStorage()
    coord = make_coord(store)
    store.fail_state = 1
    # Idle fails to persist -> Failure
    assert coord.run_one_phase() == co.PhaseId.Failure
    rid = coord.round_id
    # Failure waits for readiness then restarts the round (-> Idle)
    assert coord.run_one_phase() == co.PhaseId.Idle
    assert coord.run_one_phase() == co.PhaseId.Sum  # Idle succeeds now
    assert coord.round_id == rid + 1


def test_sum_dict_fetch_failure_recovers_mid_round():
    store = co.FaultInjectionStorage()
    coord = make_coord(store)
    store.fail_sum_dict = 1
    assert coord.run_one_phase() == co.PhaseId.Sum  # Idle ok
    # no sum messages + dict fetch fails -> Failure (timeout path also maps
    # to Failure; both restart the round)
    assert coord.run_one_phase() == co.PhaseId.Failure
    assert coord.run_one_phase() == co.PhaseId.Idle


def test_full_round_completes_after_transient_faults():
    """End-to-end: inject one state-persist fault; participants keep ticking
    through the aborted round and the next round still produces a model."""
    store = co.FaultInjectionStorage()
    coord = make_coord(store)
    store.fail_state = 1  # first Idle persist fails -> round restart
    client = sdk.InProcessClient(coord)
    rng = np.random.default_rng(17)
    participants = [
        sdk.Participant(bytes(rng.integers(0, 256, 32, dtype=np.uint8)), 1, 1, client)
        for _ in range(10)
    ]
    w = np.full(16, 0.125, dtype=np.float32)
    coord.start()
    model = None
    t0 = time.time()
    try:
        while time.time() - t0 < 30.0 and model is None:
            for p in participants:
                p.tick()
                if p.should_set_model:
                    p.set_model(w)
            body = coord.fetch_model()
            if body and body[0] == 1:
                model = sdk.decode_model(body, 0)
            time.sleep(0.005)
    finally:
        coord.stop()
    assert model is not None, "round never completed after transient fault"
    assert np.allclose(model, 0.125, atol=1e-4)
