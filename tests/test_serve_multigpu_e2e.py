"""End-to-end PET rounds through the MULTI-worker serve plane on CPU: staged
C++ coordinator + MultiGpuServeDriver (2 worker processes, gloo collective
unmask) + real participants. This is the exact production code path of
`python -m xaynet_amd.server` on an 8-GPU node, with CpuPlaneAggregator
standing in for the HIP engine (tests/test_serve_plane.py pins engine
equivalence)."""
import time

import numpy as np
import pytest

torch = pytest.importorskip("torch")

from xaynet_amd import _core  # noqa: E402

co = _core.coordinator
sdk = _core.sdk
mk = _core.mask


def test_live_rounds_multiworker_serve_plane():
    from xaynet_amd.parallel.serve import MultiGpuServeDriver

    n, length = 8, 256
    s = co.Settings()
    s.sum_prob = 0.5
    s.update_prob = 1.0
    s.model_length = length
    c = mk.MaskConfig(1, 0, 0, 3)  # Prime/F32/B0/M3
    s.mask_cfg = mk.MaskConfigPair(c, c)
    s.set_sum(1, 100, 0.05, 10.0)
    s.set_update(3, 100, 0.05, 10.0)
    s.set_sum2(1, 100, 0.05, 10.0)
    coord = co.Coordinator(s, co.InMemoryStorage(), co.InMemoryModels(), True)  # staged
    driver = MultiGpuServeDriver(coord, c, c, length, n_workers=2, device_kind="cpu",
                                 slots_per_worker=4, batch=2)
    driver.start()

    client = sdk.InProcessClient(coord)
    rng = np.random.default_rng(29)
    participants = [
        sdk.Participant(bytes(rng.integers(0, 256, 32, dtype=np.uint8)), 1, 1, client)
        for _ in range(n)
    ]
    weights = [rng.uniform(-1, 1, length).astype(np.float32) for _ in range(n)]

    coord.start()
    t0 = time.time()
    models = []
    last_unmasked = 0
    try:
        while time.time() - t0 < 240.0 and len(models) < 2:
            for i, p in enumerate(participants):
                p.tick()
                if p.should_set_model:
                    p.set_model(weights[i])
            # collect one model per ACTUAL driver unmask (fetch_model keeps
            # serving the previous round's body, so keying on round_id alone
            # can double-count a single unmask)
            if driver.rounds_unmasked > last_unmasked:
                body = coord.fetch_model()
                if body and body[0] == 1:
                    last_unmasked = driver.rounds_unmasked
                    models.append(np.asarray(sdk.decode_model(body, 0)))
            time.sleep(0.005)
    finally:
        coord.stop()
        driver.stop()

    assert len(models) >= 2, "serve plane did not complete 2 rounds"
    assert driver.rounds_unmasked >= 2
    for m in models:
        assert m.shape == (length,)
        assert np.isfinite(m).all()
        assert np.abs(m).max() <= 1.0 + 1e-5
        assert np.abs(m).mean() > 1e-3  # non-degenerate
