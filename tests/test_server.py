"""Settings (TOML + env overrides + validation), daemon serve(), durable
storage restore, and metrics recorder (reference settings/mod.rs, bin/main.rs,
initializer.rs, metrics/)."""
import os
import threading
import time

import numpy as np
import pytest

from xaynet_amd import _core
from xaynet_amd.server import Settings, SettingsError, build_coordinator, serve
from xaynet_amd.server.settings import _apply_env_overrides

co = _core.coordinator
mk = _core.mask


# ----------------------------------------------------------------- settings


def test_settings_defaults_and_example_config():
    s = Settings.load(os.path.join(os.path.dirname(__file__), "..", "configs", "config.toml"))
    assert s.sum.prob == 0.5 and s.update.count.min == 3
    assert s.mask.group_type == "Prime" and s.model_length == 4
    assert s.bind_host_port() == ("127.0.0.1", 8081)


def test_settings_env_overrides():
    raw = {"pet": {"sum": {"prob": 0.5}}}
    env = {
        "XAYNET__PET__SUM__PROB": "0.25",
        "XAYNET__API__BIND_ADDRESS": "0.0.0.0:9000",
        "XAYNET__RESTORE__ENABLE": "true",
        "XAYNET__MODEL__LENGTH": "128",
        "OTHER": "ignored",
    }
    _apply_env_overrides(raw, env)
    s = Settings._from_dict(raw)
    assert s.sum.prob == 0.25
    assert s.api.bind_address == "0.0.0.0:9000"
    assert s.restore_enable is True
    assert s.model_length == 128


@pytest.mark.parametrize(
    "mutate",
    [
        lambda s: setattr(s.sum, "prob", 0.0),          # sum prob must be > 0
        lambda s: setattr(s.sum, "prob", 1.0),          # sum prob must be < 1
        lambda s: setattr(s.update, "prob", 1.5),       # update prob must be <= 1
        lambda s: setattr(s.update.count, "min", 2),    # UPDATE_COUNT_MIN=3
        lambda s: setattr(s.sum.count, "min", 0),       # SUM_COUNT_MIN=1
        lambda s: setattr(s.sum.count, "max", 0),       # min <= max
        lambda s: setattr(s.sum.time, "max", 1.0),      # time min <= max (min=5)
        lambda s: setattr(s.sum2.count, "max", 101),    # sum2 max <= sum max
        lambda s: setattr(s.mask, "data_type", "F16"),  # unknown data type
        lambda s: setattr(s, "model_length", 0),
        lambda s: setattr(s.api, "bind_address", "nope"),
    ],
)
def test_settings_validation_rejects(mutate):
    s = Settings()
    mutate(s)
    with pytest.raises(SettingsError):
        s.validate()


# ----------------------------------------------------------------- daemon


def _daemon_settings(tmp_path=None, port=0):
    s = Settings()
    s.api.bind_address = f"127.0.0.1:{port}"
    s.sum.prob, s.update.prob = 0.5, 0.999
    s.sum.count.min, s.sum.count.max = 1, 100
    s.sum.time.min, s.sum.time.max = 0.05, 10.0
    s.update.count.min, s.update.count.max = 3, 100
    s.update.time.min, s.update.time.max = 0.05, 10.0
    s.sum2.count.min, s.sum2.count.max = 1, 100
    s.sum2.time.min, s.sum2.time.max = 0.05, 10.0
    s.model_length = 16
    if tmp_path is not None:
        s.storage_path = str(tmp_path)
    s.validate()
    return s


def test_serve_runs_round_over_http(tmp_path):
    import socket

    with socket.socket() as sk:
        sk.bind(("127.0.0.1", 0))
        port = sk.getsockname()[1]
    settings = _daemon_settings(port=port)
    ready, stop = threading.Event(), threading.Event()
    t = threading.Thread(target=serve, args=(settings, ready, stop), daemon=True)
    t.start()
    assert ready.wait(10)
    try:
        import xaynet_sdk

        url = f"http://127.0.0.1:{port}"
        logs = [[] for _ in range(8)]

        class P(xaynet_sdk.ParticipantABC):
            def __init__(self, i):
                self.i = i

            def train_round(self, ti):
                logs[self.i].append("t")
                return np.full(16, 1.0, dtype=np.float32)

            def serialize_training_result(self, r):
                return r.tolist()

            def deserialize_training_input(self, gm):
                return np.asarray(gm, dtype=np.float32)

            def on_new_global_model(self, gm):
                if gm is not None:
                    logs[self.i].append(gm)

        hs = [xaynet_sdk.spawn_participant(url, P, args=(i,)) for i in range(8)]
        t0 = time.time()
        got = None
        while time.time() - t0 < 45 and got is None:
            got = next(
                (x for lg in logs for x in lg if isinstance(x, np.ndarray)), None
            )
            time.sleep(0.05)
        for h in hs:
            h.stop()
        assert got is not None and np.allclose(got, 1.0, atol=1e-4)
    finally:
        stop.set()
        t.join(10)


def test_file_storage_checkpoint_restore(tmp_path):
    settings = _daemon_settings(tmp_path=tmp_path)
    coordinator, store, models = build_coordinator(settings)
    coordinator.run_one_phase()  # Idle: persists state, bumps round
    rid = coordinator.round_id
    params = coordinator.fetch_round_params()
    coordinator.stop()
    assert (tmp_path / "coordinator" / "coordinator_state.bin").exists()

    # a new daemon with restore=enable resumes the round counter + params
    settings.restore_enable = True
    coordinator2, _, _ = build_coordinator(settings)
    assert coordinator2.round_id == rid
    assert coordinator2.fetch_round_params() == params
    coordinator2.stop()


def test_file_model_storage_no_overwrite(tmp_path):
    models = co.FileModels(str(tmp_path / "m"))
    body = b"\x01" + b"\x00" * 16
    mid = None
    # write via the coordinator path: use the python binding surface directly
    # (set_global_model isn't exposed; exercise read path + id hygiene)
    assert models.global_model("no_such_id") is None
    assert models.global_model("../escape") is None


def test_metrics_file_sink(tmp_path):
    path = str(tmp_path / "metrics.lp")
    co.install_metrics_file(path)
    try:
        settings = _daemon_settings()
        coordinator, _, _ = build_coordinator(settings)
        coordinator.run_one_phase()  # Idle emits phase + round metrics
        coordinator.stop()
        co.metrics_flush()
        time.sleep(0.1)
        with open(path) as f:
            lines = f.read().strip().splitlines()
        names = {ln.split(",")[0] for ln in lines}
        assert "phase" in names and "round_total_number" in names
        assert "round_param_sum" in names and "round_param_update" in names
        # influx line protocol shape: name,tags value=... timestamp
        assert all(" value=" in ln and ln.rsplit(" ", 1)[1].isdigit() for ln in lines)
    finally:
        co.uninstall_metrics()
