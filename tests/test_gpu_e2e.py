"""End-to-end PET round with the GPU data plane: staged C++ coordinator +
MI355X driver (K3 digit-plane aggregation, K6 unpack, K4 unmask) + real
participants over the native HTTP stack. The unmasked global model must match
the mean of the accepted updaters' models — same acceptance as the CPU-plane
e2e (tests/test_e2e_round.py), proving plane equivalence."""
import time

import numpy as np
import pytest

torch = pytest.importorskip("torch")

from xaynet_amd import _core  # noqa: E402

co = _core.coordinator
sdk = _core.sdk
mk = _core.mask
rest = _core.rest

pytestmark = pytest.mark.gpu


def _gpu():
    from xaynet_amd.ops import gpu_available

    if not gpu_available():
        pytest.skip("no MI355X visible")


def test_full_round_gpu_staged_plane():
    _gpu()
    from xaynet_amd.ops import make_coordinator_driver

    n, length = 12, 4096
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
    server = rest.RestServer(coord, "127.0.0.1", 0, 4)
    assert server.start()
    driver = make_coordinator_driver(coord, c, c, length, pool_size=8)
    driver.start()

    client = rest.HttpXaynetClient("127.0.0.1", server.port)
    rng = np.random.default_rng(23)
    participants = [
        sdk.Participant(bytes(rng.integers(0, 256, 32, dtype=np.uint8)), 1, 1, client)
        for _ in range(n)
    ]
    weights = [rng.uniform(-1, 1, length).astype(np.float32) for _ in range(n)]

    coord.start()
    t0 = time.time()
    model, seeds_body = None, None
    raw = rest.HttpClient("127.0.0.1", server.port)
    try:
        while time.time() - t0 < 60.0:
            for i, p in enumerate(participants):
                p.tick()
                if p.should_set_model:
                    p.set_model(weights[i])
            status, sums = raw.request("GET", "/sums")
            if status == 200:
                npk = int.from_bytes(sums[:8], "little")
                if npk:
                    import base64

                    enc = (
                        base64.b64encode(sums[8:40]).decode()
                        .replace("+", "%2B").replace("/", "%2F").replace("=", "%3D")
                    )
                    st2, b2 = raw.request("GET", f"/seeds?pk={enc}")
                    if st2 == 200:
                        seeds_body = b2
            status, body = raw.request("GET", "/model")
            if status == 200:
                model = sdk.decode_model(b"\x01" + body, 0)
                break
            time.sleep(0.005)
    finally:
        coord.stop()
        driver.stop()
        server.stop()

    assert model is not None, "no global model published (GPU plane)"
    assert driver.rounds_unmasked >= 1, "model did not come from the GPU driver"
    assert seeds_body is not None

    nsd = int.from_bytes(seeds_body[:8], "little")
    accepted, off = set(), 8
    for _ in range(nsd):
        accepted.add(bytes(seeds_body[off : off + 32]))
        off += 40 + 80
    by_pk = {p.pk: i for i, p in enumerate(participants)}
    idx = [by_pk[pk] for pk in accepted]
    assert len(idx) >= 3
    expect = np.mean([weights[i].astype(np.float64) for i in idx], axis=0)
    assert np.abs(model.astype(np.float64) - expect).max() < 1e-4


def test_full_round_multigpu_serve_plane_cuda_worker():
    """The production serve plane with a CUDA worker process (pinned shm
    ring + async H2D + collective-unmask code path at world=1): live HTTP
    round, exact model out."""
    _gpu()
    from xaynet_amd.parallel.serve import MultiGpuServeDriver

    n, length = 10, 100_000
    s = co.Settings()
    s.sum_prob = 0.5
    s.update_prob = 1.0
    s.model_length = length
    c = mk.MaskConfig(1, 0, 0, 6)
    s.mask_cfg = mk.MaskConfigPair(c, c)
    s.set_sum(1, 100, 0.05, 10.0)
    s.set_update(3, 100, 0.05, 10.0)
    s.set_sum2(1, 100, 0.05, 10.0)
    coord = co.Coordinator(s, co.InMemoryStorage(), co.InMemoryModels(), True)  # staged
    driver = MultiGpuServeDriver(coord, c, c, length, n_workers=1, device_kind="cuda",
                                 slots_per_worker=8, batch=4)
    driver.start()
    client = sdk.InProcessClient(coord)
    rng = np.random.default_rng(53)
    participants = [
        sdk.Participant(bytes(rng.integers(0, 256, 32, dtype=np.uint8)), 1, 1, client)
        for _ in range(n)
    ]
    weights = [rng.uniform(-1, 1, length).astype(np.float32) for _ in range(n)]
    coord.start()
    t0 = time.time()
    model = None
    try:
        while time.time() - t0 < 120.0 and model is None:
            for i, p in enumerate(participants):
                p.tick()
                if p.should_set_model:
                    p.set_model(weights[i])
            body = coord.fetch_model()
            if body and body[0] == 1:
                model = np.asarray(sdk.decode_model(body, 0))
            time.sleep(0.002)
    finally:
        coord.stop()
        driver.stop()
    assert model is not None
    assert np.isfinite(model).all()
    assert np.abs(model).max() <= 1.0 + 1e-5
    assert np.abs(model).mean() > 1e-3
