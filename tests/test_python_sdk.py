"""xaynet_sdk package: reference-API-compatible participant runtime
(spawn_participant / ParticipantABC / AsyncParticipant) against a live
coordinator served over HTTP (reference bindings/python/xaynet_sdk)."""
import threading
import time

import numpy as np
import pytest

from xaynet_amd import _core
import xaynet_sdk

co = _core.coordinator
mk = _core.mask
rest = _core.rest


def serve_coordinator(model_length=24, n_expect=3):
    s = co.Settings()
    s.sum_prob = 0.5
    s.update_prob = 1.0
    s.model_length = model_length
    c = mk.MaskConfig(1, 0, 0, 3)
    s.mask_cfg = mk.MaskConfigPair(c, c)
    s.set_sum(1, 100, 0.05, 10.0)
    s.set_update(n_expect, 100, 0.05, 10.0)
    s.set_sum2(1, 100, 0.05, 10.0)
    coord = co.Coordinator(s, co.InMemoryStorage(), co.InMemoryModels(), False)
    server = rest.RestServer(coord, "127.0.0.1", 0, 4)
    assert server.start()
    coord.start()
    return coord, server


class NumpyParticipant(xaynet_sdk.ParticipantABC):
    """Minimal user participant: 'trains' by emitting a fixed vector."""

    def __init__(self, value, length, log):
        self.value = value
        self.length = length
        self.log = log

    def train_round(self, training_input):
        self.log.append(("train", training_input))
        return np.full(self.length, self.value, dtype=np.float32)

    def serialize_training_result(self, training_result) -> list:
        return training_result.tolist()

    def deserialize_training_input(self, global_model):
        return np.asarray(global_model, dtype=np.float32)

    def on_new_global_model(self, global_model):
        self.log.append(("global", global_model))

    def on_stop(self):
        self.log.append(("stop", None))


def test_spawn_participants_full_round():
    length, n = 24, 10
    coord, server = serve_coordinator(model_length=length, n_expect=3)
    url = f"http://127.0.0.1:{server.port}"
    logs = [[] for _ in range(n)]
    handles = []
    try:
        for i in range(n):
            handles.append(
                xaynet_sdk.spawn_participant(
                    url, NumpyParticipant, args=(float(i), length, logs[i])
                )
            )
        # wait until some participant observes a global model
        t0 = time.time()
        got = None
        while time.time() - t0 < 45.0 and got is None:
            for lg in logs:
                for kind, payload in lg:
                    if kind == "global" and payload is not None:
                        got = payload
                        break
                if got is not None:
                    break
            time.sleep(0.05)
        assert got is not None, "no participant observed a global model"
        assert got.shape == (length,)
        # the aggregate is a mean of a subset of the constant vectors 0..n-1
        assert 0.0 - 1e-4 <= float(got.min()) and float(got.max()) <= n - 1 + 1e-4
        trained = [lg for lg in logs if any(k == "train" for k, _ in lg)]
        assert len(trained) >= 3, "fewer than UPDATE_COUNT_MIN participants trained"
    finally:
        states = [h.stop() for h in handles]
        coord.stop()
        server.stop()
    # stop() returns a restorable serialized state
    assert all(isinstance(st, list) and len(st) > 100 for st in states)


def test_participant_save_restore_identity():
    coord, server = serve_coordinator()
    url = f"http://127.0.0.1:{server.port}"
    try:
        from xaynet_sdk.xaynet_sdk import Participant, UninitializedParticipant

        p = Participant(url, 1.0, None)
        p.tick()
        state = p.save()
        with pytest.raises(UninitializedParticipant):
            p.tick()
        p2 = Participant(url, 1.0, state)
        assert p2._sign_seed == p._sign_seed  # identity survives restore
        p2.tick()
    finally:
        coord.stop()
        server.stop()


def test_async_participant_round():
    length = 16
    coord, server = serve_coordinator(model_length=length, n_expect=3)
    url = f"http://127.0.0.1:{server.port}"
    handles = []
    try:
        participants = [
            xaynet_sdk.spawn_async_participant(url) for _ in range(8)
        ]
        handles = [p for p, _ in participants]
        model = np.ones(length, dtype=np.float32).tolist()
        t0 = time.time()
        got = None
        while time.time() - t0 < 45.0 and got is None:
            for p, notifier in participants:
                p.set_local_model(model)
                if notifier.is_set():
                    gm = p.get_global_model()
                    if gm is not None:
                        got = gm
                        break
            time.sleep(0.05)
        assert got is not None, "async participants produced no global model"
        assert np.allclose(np.asarray(got), 1.0, atol=1e-4)
    finally:
        for h in handles:
            h.stop()
        coord.stop()
        server.stop()


def test_run_concurrently_bounded():
    import threading
    import time as _t

    from xaynet_sdk.utils import run_concurrently

    active = [0]
    peak = [0]
    lock = threading.Lock()

    def task(i):
        def run():
            with lock:
                active[0] += 1
                peak[0] = max(peak[0], active[0])
            _t.sleep(0.02)
            with lock:
                active[0] -= 1
            if i == 7:
                raise ValueError("boom")
            return i * 10
        return run

    results = list(run_concurrently([task(i) for i in range(20)], max_concurrency=4))
    assert peak[0] <= 4
    assert len(results) == 20
    ok = {i: r for i, r, e in results if e is None}
    errs = [i for i, r, e in results if e is not None]
    assert errs == [7] and ok[3] == 30
