"""Examples double as smoke tests (reference: python examples are the manual
e2e harness): run the torch regression participant end-to-end against a
served coordinator and check the global model converges toward the true
weights."""
import sys
import threading
import time
from pathlib import Path

import numpy as np
import pytest

torch = pytest.importorskip("torch")

sys.path.insert(0, str(Path(__file__).parent.parent / "examples" / "torch_regression"))

from xaynet_amd import _core  # noqa: E402

co = _core.coordinator
mk = _core.mask
rest = _core.rest


def test_torch_regression_example_converges():
    import xaynet_sdk
    from participant import N_FEATURES, TRUE_W, RegressionParticipant

    s = co.Settings()
    s.sum_prob = 0.4
    s.update_prob = 1.0
    s.model_length = N_FEATURES + 1
    c = mk.MaskConfig(1, 0, 0, 3)
    s.mask_cfg = mk.MaskConfigPair(c, c)
    s.set_sum(1, 100, 0.05, 10.0)
    s.set_update(3, 100, 0.05, 10.0)
    s.set_sum2(1, 100, 0.05, 10.0)
    coord = co.Coordinator(s, co.InMemoryStorage(), co.InMemoryModels(), False)
    server = rest.RestServer(coord, "127.0.0.1", 0, 4)
    assert server.start()
    coord.start()
    url = f"http://127.0.0.1:{server.port}"

    handles = [
        xaynet_sdk.spawn_participant(url, RegressionParticipant, args=(seed,))
        for seed in range(8)
    ]
    raw = rest.HttpClient("127.0.0.1", server.port)
    try:
        t0 = time.time()
        best = None
        while time.time() - t0 < 60.0:
            status, body = raw.request("GET", "/model")
            if status == 200:
                m = _core.sdk.decode_model(b"\x01" + body, 0)
                err = float(np.abs(np.asarray(m[:N_FEATURES]) - TRUE_W).mean())
                best = err if best is None else min(best, err)
                if best < 0.05:
                    break
            time.sleep(0.2)
    finally:
        for h in handles:
            h.stop()
        coord.stop()
        server.stop()
    assert best is not None, "no global model produced by the example"
    assert best < 0.2, f"federated regression did not converge (err {best})"


def test_example_modules_import_cleanly():
    """Every example is importable (the runnable part is __main__-guarded) —
    guards the examples against API rot."""
    import importlib.util

    for name in ("hello_world", "hello_world_async", "multiple_participants",
                 "participate_in_update", "restore", "download_global_model",
                 "download_global_model_async"):
        p = Path(__file__).parent.parent / "examples" / f"{name}.py"
        spec = importlib.util.spec_from_file_location(f"ex_{name}", p)
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)


def test_hello_world_participant_completes_round():
    """Drive the hello_world example's participant class against a live
    coordinator: a global model must appear."""
    import importlib.util
    import time

    import numpy as np

    from xaynet_amd import _core
    import xaynet_sdk

    p = Path(__file__).parent.parent / "examples" / "hello_world.py"
    spec = importlib.util.spec_from_file_location("ex_hw", p)
    hw = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(hw)

    co, mk, rest = _core.coordinator, _core.mask, _core.rest
    s = co.Settings()
    s.sum_prob = 0.6
    s.update_prob = 1.0
    s.model_length = 4
    c = mk.MaskConfig(1, 0, 0, 3)
    s.mask_cfg = mk.MaskConfigPair(c, c)
    s.set_sum(1, 50, 0.05, 10.0)
    s.set_update(3, 50, 0.05, 10.0)
    s.set_sum2(1, 50, 0.05, 10.0)
    coord = co.Coordinator(s, co.InMemoryStorage(), co.InMemoryModels(), False)
    server = rest.RestServer(coord, "127.0.0.1", 0, 4)
    assert server.start()
    coord.start()
    url = f"http://127.0.0.1:{server.port}"

    got = []
    orig = hw.FixedModelParticipant.on_new_global_model
    hw.FixedModelParticipant.on_new_global_model = (
        lambda self, gm: got.append(gm) if gm is not None else None)
    handles = [xaynet_sdk.spawn_participant(url, hw.FixedModelParticipant)
               for _ in range(8)]
    try:
        t0 = time.time()
        while time.time() - t0 < 45.0 and not got:
            time.sleep(0.05)
        assert got, "hello_world participants produced no global model"
        assert np.allclose(got[0], hw.FixedModelParticipant.WEIGHTS, atol=1e-4)
    finally:
        hw.FixedModelParticipant.on_new_global_model = orig
        for h in handles:
            h.stop()
        coord.stop()
        server.stop()
