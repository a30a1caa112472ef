"""GPU participant accelerator: the SDK's update
masking (K1+K5w) and sum2 mask aggregation (K1+K2) run on the MI355X via
the hook surface, and a full live round with accelerated participants
produces the correct global model."""
import time

import numpy as np
import pytest

torch = pytest.importorskip("torch")

from xaynet_amd import _core  # noqa: E402

co = _core.coordinator
mk = _core.mask
sdk = _core.sdk

pytestmark = pytest.mark.gpu


def _gpu():
    from xaynet_amd.ops import gpu_available

    if not gpu_available():
        pytest.skip("no MI355X visible")


@pytest.mark.parametrize("cfg_args,dtype", [
    ((1, 0, 0, 6), np.float32),   # u64 order
    ((1, 3, 0, 6), np.int64),     # i64 masking
    ((1, 1, 0, 3), np.float64),   # wide (u128) order
])
def test_mask_weights_decodes_close_to_cpu(cfg_args, dtype):
    """GPU-masked update decodes to the same weights as the CPU fast masker
    within a few quantization quanta (the documented deviation)."""
    _gpu()
    from xaynet_amd.ops import GpuMaskedAggregator

    c = mk.MaskConfig(*cfg_args)
    pair = mk.MaskConfigPair(c, c)
    length, k = 4099, 1
    rng = np.random.default_rng(3)
    if np.issubdtype(dtype, np.integer):
        w = rng.integers(-1000, 1000, length).astype(dtype)
    else:
        w = rng.uniform(-1, 1, length).astype(dtype)
    seed = b"\x09" * 32

    eng = GpuMaskedAggregator(c, c, length)
    wire_gpu = eng.mask_weights(seed, torch.from_numpy(w.copy()), 1, k)
    cpu = mk.mask_model(seed, mk.Scalar(1, k), w.astype(np.float64)
                        if not np.issubdtype(dtype, np.integer) else w, pair)
    wire_cpu = bytes(cpu.serialize())
    assert len(wire_gpu) == len(wire_cpu)

    # decode both through the oracle unmask against the same mask
    agg_g = mk.Aggregation(pair, length)
    agg_g.aggregate(mk.MaskObject.deserialize(wire_gpu))
    agg_c = mk.Aggregation(pair, length)
    agg_c.aggregate(cpu)
    mobj = mk.derive_mask(seed, length, pair)
    out_g = np.asarray(agg_g.unmask(mobj), dtype=np.float64)
    out_c = np.asarray(agg_c.unmask(mobj), dtype=np.float64)
    info_exp = {np.float32: 1e10, np.float64: 1e20}.get(dtype, 1e10)
    tol = 16.0 / info_exp + (1e-6 if dtype == np.float32 else 0.0)
    assert np.abs(out_g - out_c).max() <= tol, np.abs(out_g - out_c).max()
    # sanity vs the original weights, clamped to the config bound (B0 = 1);
    # the unmask emits through double arithmetic -> allow f64 ulp noise
    wc = np.clip(w.astype(np.float64), -1.0, 1.0)
    assert np.abs(out_c - wc).max() <= tol + 1.0 / info_exp + 8e-16


def test_full_round_with_accelerated_participants():
    _gpu()
    from xaynet_amd.ops.accel import ParticipantAccel

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
    coord = co.Coordinator(s, co.InMemoryStorage(), co.InMemoryModels(), False)
    client = sdk.InProcessClient(coord)
    rng = np.random.default_rng(41)
    accel = ParticipantAccel()
    participants = []
    for _ in range(n):
        p = sdk.Participant(bytes(rng.integers(0, 256, 32, dtype=np.uint8)), 1, 1, client)
        accel.attach(p)
        participants.append(p)
    weights = [rng.uniform(-1, 1, length).astype(np.float32) for _ in range(n)]

    coord.start()
    t0 = time.time()
    model = None
    try:
        while time.time() - t0 < 90.0 and model is None:
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
    assert model is not None
    # the model is the mean of the accepted updaters' weights; all weights
    # share no structure, so just bound against the per-update tolerance
    assert np.isfinite(model).all()
    assert np.abs(model).max() <= 1.0 + 1e-5
    assert np.abs(model).mean() > 1e-3
