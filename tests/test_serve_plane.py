"""Multi-process serve plane on CPU (gloo world=2): the same worker loop,
shared-memory ingest ring and ShardedAggregation composition the multi-GPU
coordinator uses in production (xaynet_amd/parallel/serve.py), proven here
without hardware. Replaces the reference's single-threaded in-RAM
accumulator (rust/xaynet-server/src/state_machine/phases/update.rs:35-40)."""
import time

import numpy as np
import pytest

from xaynet_amd import _core

mk = _core.mask
co = _core.coordinator

CFG_ARGS = (1, 0, 0, 6)  # Prime/F32/B0/M6, bpn=7


def _forge_round(length, k, seed0=1, rng_seed=7):
    """k masked updates + the aggregated mask, via the exact oracle."""
    cfg = mk.MaskConfig(*CFG_ARGS)
    pair = mk.MaskConfigPair(cfg, cfg)
    order = int(cfg.order)
    bpn = cfg.bytes_per_number
    rng = np.random.default_rng(rng_seed)
    weights = [rng.uniform(-1, 1, length) for _ in range(k)]
    sc = mk.Scalar(1, k)
    magg = mk.Aggregation(pair, length)
    updates, unit_acc = [], 0
    for p in range(k):
        seed = bytes([seed0 + p]) * 32
        masked = mk.mask_model(seed, sc, weights[p].astype(np.float64), pair)
        wire = masked.serialize()
        vect = wire[8 : 8 + length * bpn]
        unit = int.from_bytes(wire[8 + length * bpn + 4 :], "little")
        updates.append(vect)
        unit_acc = (unit_acc + unit) % order
        magg.aggregate(mk.derive_mask(seed, length, pair))
    mask_wire = magg.object.serialize()
    mask_vect = mask_wire[8 : 8 + length * bpn]
    mask_unit = int.from_bytes(mask_wire[8 + length * bpn + 4 :], "little")
    return weights, updates, unit_acc, mask_vect, mask_unit


@pytest.mark.parametrize("world,length", [(2, 120), (4, 120), (2, 121)])
def test_serve_plane_round(world, length):
    """world=2/4 sharded rounds; length 121 exercises the all_reduce
    fallback (uneven shard)."""
    from xaynet_amd.parallel.serve import ServePlane

    k = 6
    plane = ServePlane(CFG_ARGS, length, world, "cpu", slots_per_worker=4, batch=2)
    try:
        weights, updates, unit_acc, mask_vect, mask_unit = _forge_round(length, k)
        for v in updates:
            plane.put_update(v)
        out = plane.unmask(mask_vect, mask_unit, unit_acc, k, round_id=1)
        ref = np.mean(weights, axis=0)
        assert np.abs(out - ref).max() < 1e-6
        # second round through the same plane (workers reset after unmask)
        weights, updates, unit_acc, mask_vect, mask_unit = _forge_round(
            length, k, seed0=50, rng_seed=11)
        for v in updates:
            plane.put_update(v)
        out = plane.unmask(mask_vect, mask_unit, unit_acc, k, round_id=2)
        ref = np.mean(weights, axis=0)
        assert np.abs(out - ref).max() < 1e-6
    finally:
        plane.stop()


def test_serve_plane_ring_backpressure():
    """More in-flight updates than ring slots: put_update must dispatch and
    recycle slots instead of deadlocking."""
    from xaynet_amd.parallel.serve import ServePlane

    length, k = 64, 12
    plane = ServePlane(CFG_ARGS, length, 1, "cpu", slots_per_worker=2, batch=1)
    try:
        weights, updates, unit_acc, mask_vect, mask_unit = _forge_round(length, k)
        t0 = time.time()
        for v in updates:
            plane.put_update(v)
        out = plane.unmask(mask_vect, mask_unit, unit_acc, k, round_id=1)
        assert time.time() - t0 < 60
        assert np.abs(out - np.mean(weights, axis=0)).max() < 1e-6
    finally:
        plane.stop()


def test_serve_plane_worker_death_detected():
    """A dead worker must surface as an error, not a hang."""
    from xaynet_amd.parallel.serve import ServePlane

    length = 64
    plane = ServePlane(CFG_ARGS, length, 1, "cpu", slots_per_worker=2, batch=1)
    try:
        plane.procs[0].terminate()
        plane.procs[0].join()
        weights, updates, unit_acc, mask_vect, mask_unit = _forge_round(length, 4)
        with pytest.raises(RuntimeError, match="died"):
            # ring has 2 slots and nobody recycles them -> 3rd put detects
            for v in updates:
                plane.put_update(v)
    finally:
        plane.stop()


def test_serve_plane_world8_planes_rs():
    """world=8 with a 61-bit group order: 61 + log2(8) > 63 forces the
    digit-plane reduce-scatter strategy (the xGMI path the 8-GPU node
    takes for near-u64 orders) — proven here on gloo."""
    from xaynet_amd.ops.cpu_engine import CpuPlaneAggregator
    from xaynet_amd.parallel import choose_reduce_strategy
    from xaynet_amd.parallel.serve import ServePlane

    cfg_args = (1, 2, 2, 6)  # Prime/I32/B2/M6: order ~2^61, bpn 8
    length, k, world = 160, 8, 8
    cfg = mk.MaskConfig(*cfg_args)
    assert choose_reduce_strategy(world, length, int(cfg.order)) == "planes_rs"

    pair = mk.MaskConfigPair(cfg, cfg)
    order = int(cfg.order)
    bpn = cfg.bytes_per_number
    rng = np.random.default_rng(23)
    weights = [rng.integers(-50, 50, length).astype(np.int32) for _ in range(k)]
    sc = mk.Scalar(1, k)
    magg = mk.Aggregation(pair, length)
    updates, unit_acc = [], 0
    for p in range(k):
        seed = bytes([p + 1]) * 32
        masked = mk.mask_model(seed, sc, weights[p].astype(np.float64), pair)
        wire = masked.serialize()
        updates.append(wire[8 : 8 + length * bpn])
        unit_acc = (unit_acc + int.from_bytes(wire[8 + length * bpn + 4 :], "little")) % order
        magg.aggregate(mk.derive_mask(seed, length, pair))
    mask_wire = magg.object.serialize()
    mask_vect = mask_wire[8 : 8 + length * bpn]
    mask_unit = int.from_bytes(mask_wire[8 + length * bpn + 4 :], "little")

    plane = ServePlane(cfg_args, length, world, "cpu", slots_per_worker=2, batch=1)
    try:
        for v in updates:
            plane.put_update(v)
        out = plane.unmask(mask_vect, mask_unit, unit_acc, k, round_id=1)
        # I32 output truncates the (fractional) mean toward zero
        ref = np.mean(weights, axis=0)
        assert out.dtype == np.int32
        assert np.abs(out - ref).max() < 1.0
    finally:
        plane.stop()


def test_driver_parallel_drain_backlog():
    """A staged BACKLOG (staged_count > 1 at drain time) triggers the
    driver's helper drain threads; the aggregate must stay exact and every
    update must be counted. Uses a stub coordinator so the backlog exists
    before the driver thread ever runs."""
    import ctypes

    from xaynet_amd.parallel.serve import MultiGpuServeDriver

    length, k, world = 96, 24, 2
    cfg = mk.MaskConfig(*CFG_ARGS)
    pair = mk.MaskConfigPair(cfg, cfg)
    bpn = cfg.bytes_per_number
    rng = np.random.default_rng(3)
    weights = [rng.uniform(-1, 1, length) for _ in range(k)]
    sc = mk.Scalar(1, k)
    magg = mk.Aggregation(pair, length)
    staged = []
    for p_ in range(k):
        seed = bytes([1 + p_]) * 32
        wire = bytes(mk.mask_model(seed, sc, weights[p_].astype(np.float64), pair).serialize())
        staged.append((wire[8 : 8 + length * bpn], wire[8 + length * bpn + 4 :]))
        magg.aggregate(mk.derive_mask(seed, length, pair))
    mask_wire = bytes(magg.object.serialize())

    class StubCoordinator:
        round_id = 7

        def __init__(self):
            self.staged = list(staged)
            self.model = None
            self._pu_sent = False

        def staged_count(self):
            return len(self.staged)

        def pop_staged_vect(self, ptr, cap):
            try:
                vect, unit = self.staged.pop()
            except IndexError:
                return None
            assert len(vect) <= cap
            ctypes.memmove(ptr, vect, len(vect))
            return (len(vect), unit)

        def pending_unmask(self):
            if self.staged or self._pu_sent:
                return None
            self._pu_sent = True
            return (mask_wire, k)

        def supply_unmasked_model(self, body):
            self.model = bytes(body)

    stub = StubCoordinator()
    driver = MultiGpuServeDriver(stub, cfg, cfg, length, n_workers=world,
                                 device_kind="cpu", slots_per_worker=4, batch=2)
    driver.start()
    t0 = time.time()
    while stub.model is None and time.time() - t0 < 120:
        time.sleep(0.002)
    try:
        assert stub.model is not None, "driver never supplied a model"
        out = np.asarray(_core.sdk.decode_model(stub.model, 0))
        ref = np.mean(weights, axis=0)
        assert np.abs(out - ref).max() < 1e-5
    finally:
        driver.stop()
