"""Masking / aggregation / unmasking tests.

Mirrors the reference test matrix (rust/xaynet-core/src/mask/masking.rs:
458-1014): mask -> derive -> unmask round-trips with tolerance 1/exp_shift,
group membership of masked weights, and mask-aggregate-unmask vs the plain
average with tolerance n/exp_shift, over GroupType x DataType x BoundType.
"""
import numpy as np
import pytest

from xaynet_amd import _core

mk = _core.mask

GROUPS = [0, 1, 2]  # Integer, Prime, Power2
DTYPES = {0: np.float32, 1: np.float64, 2: np.int32, 3: np.int64}
BOUNDS = [0, 2, 4, 6]  # B0, B2, B4, B6 (Bmax covered separately)
M3 = 3


def make_pair(group, dtype, bound):
    c = mk.MaskConfig(group, dtype, bound, M3)
    return c, mk.MaskConfigPair(c, c)


def gen_weights(dtype, bound, n, rng):
    lim = {0: 1.0, 2: 100.0, 4: 10_000.0, 6: 1_000_000.0}[bound]
    if dtype in (0, 1):
        return rng.uniform(-lim, lim, n).astype(DTYPES[dtype])
    return rng.integers(-int(lim), int(lim), n).astype(DTYPES[dtype])


@pytest.mark.parametrize("group", GROUPS)
@pytest.mark.parametrize("dtype", [0, 1, 2, 3])
@pytest.mark.parametrize("bound", BOUNDS)
def test_mask_unmask_roundtrip(group, dtype, bound):
    rng = np.random.default_rng(group * 100 + dtype * 10 + bound)
    cfg, pair = make_pair(group, dtype, bound)
    n = 10
    w = gen_weights(dtype, bound, n, rng)
    seed = bytes(rng.integers(0, 256, 32, dtype=np.uint8))
    masked = mk.mask_model(seed, mk.Scalar.unit(), w, pair)
    assert masked.is_valid()

    mask = mk.derive_mask(seed, n, pair)
    assert mask.is_valid()

    agg = mk.Aggregation(pair, n)
    assert agg.validate_aggregation(masked) == 0
    agg.aggregate(masked)
    assert agg.validate_unmasking(mask.__class__.deserialize(mask.serialize())) == 0
    out = agg.unmask(mask)
    tol = 10.0**-10 if dtype in (0, 2, 3) else 10.0**-20
    if dtype in (2, 3):
        assert (out == w).all(), (out[:5], w[:5])
    else:
        assert np.abs(out.astype(np.float64) - w.astype(np.float64)).max() <= tol * 10 + 1e-7 * np.abs(w).max()


@pytest.mark.parametrize("group", GROUPS)
@pytest.mark.parametrize("dtype", [0, 1])
def test_mask_aggregate_average(group, dtype):
    rng = np.random.default_rng(7 + group + dtype)
    cfg, pair = make_pair(group, dtype, 0)
    n, k = 50, 5
    aggm = mk.Aggregation(pair, n)
    aggmask = mk.Aggregation(pair, n)
    ws = []
    for i in range(k):
        w = rng.uniform(-1, 1, n).astype(DTYPES[dtype])
        ws.append(w)
        seed = bytes(rng.integers(0, 256, 32, dtype=np.uint8))
        masked = mk.mask_model(seed, mk.Scalar(1, k), w, pair)
        assert aggm.validate_aggregation(masked) == 0
        aggm.aggregate(masked)
        aggmask.aggregate(mk.derive_mask(seed, n, pair))
    out = aggm.unmask(aggmask.object)
    expect = np.mean([w.astype(np.float64) for w in ws], axis=0)
    assert np.abs(out.astype(np.float64) - expect).max() < k * 10.0**-9 + 1e-6


def test_masked_elements_in_group():
    cfg, pair = make_pair(1, 0, 0)
    order = int(cfg.order)
    rng = np.random.default_rng(0)
    w = rng.uniform(-1, 1, 20).astype(np.float32)
    masked = mk.mask_model(bytes(32), mk.Scalar.unit(), w, pair)
    for i in range(20):
        assert 0 <= int(masked.element(i)) < order


def test_scalar_weighting():
    # weighted fedavg: two models with weights 0.25 / 0.75
    cfg, pair = make_pair(1, 0, 0)
    n = 16
    rng = np.random.default_rng(3)
    w1 = rng.uniform(-1, 1, n).astype(np.float32)
    w2 = rng.uniform(-1, 1, n).astype(np.float32)
    aggm = mk.Aggregation(pair, n)
    aggmask = mk.Aggregation(pair, n)
    for w, (num, den) in [(w1, (1, 4)), (w2, (3, 4))]:
        seed = bytes(rng.integers(0, 256, 32, dtype=np.uint8))
        aggm.aggregate(mk.mask_model(seed, mk.Scalar(num, den), w, pair))
        aggmask.aggregate(mk.derive_mask(seed, n, pair))
    out = aggm.unmask(aggmask.object)
    expect = 0.25 * w1.astype(np.float64) + 0.75 * w2.astype(np.float64)
    assert np.abs(out - expect).max() < 1e-6


def test_config_catalogue_spots():
    # spot values verified against the reference catalogue
    assert mk.MaskConfig(0, 0, 0, 3).order == "20000000000001"
    assert mk.MaskConfig(1, 0, 0, 3).order == "20000000000021"
    assert mk.MaskConfig(2, 0, 0, 3).order == str(2**45)
    # largest catalogued order is 268 bytes? reference says <= 173 for
    # serialization... Bmax f64 M12 Integer:
    c = mk.MaskConfig(0, 1, 255, 12)
    assert c.bytes_per_number >= 170


def test_validation_errors():
    _, pair = make_pair(1, 0, 0)
    _, pair2 = make_pair(1, 1, 0)
    agg = mk.Aggregation(pair, 10)
    w = np.zeros(10, np.float32)
    masked = mk.mask_model(bytes(32), mk.Scalar.unit(), w, pair)
    other_len = mk.mask_model(bytes(32), mk.Scalar.unit(), np.zeros(5, np.float32), pair)
    assert agg.validate_aggregation(other_len) != 0  # length mismatch
    wrong_cfg = mk.mask_model(bytes(32), mk.Scalar.unit(), np.zeros(10, np.float64), pair2)
    assert agg.validate_aggregation(wrong_cfg) != 0  # config mismatch
    # unmask before any model
    assert agg.validate_unmasking(masked) != 0


def test_mask_object_serialization_format():
    cfg, pair = make_pair(1, 0, 0)
    w = np.zeros(3, np.float32)
    masked = mk.mask_model(bytes(32), mk.Scalar.unit(), w, pair)
    b = masked.serialize()
    # vect: config(4) + count(4) + 3*6 limbs; unit: config(4) + 6
    assert len(b) == 4 + 4 + 3 * 6 + 4 + 6
    assert b[0:4] == bytes([1, 0, 0, 3])
    assert int.from_bytes(b[4:8], "big") == 3
    m2 = mk.MaskObject.deserialize(b)
    assert m2.serialize() == b


def test_derive_mask_golden():
    """Self-golden derive_mask values (regression anchor; the GPU K1 kernel
    must reproduce these exactly — same ChaCha20 stream & rejection walk)."""
    c = mk.MaskConfig(1, 0, 0, 3)
    pair = mk.MaskConfigPair(c, c)
    m = mk.derive_mask(bytes(32), 5, pair)
    assert m.unit_value == "11343681809713"
    assert [m.element(i) for i in range(5)] == [
        "5166163054878",
        "19802247713817",
        "13039488148403",
        "12150543332323",
        "1814065813462",
    ]
    c2 = mk.MaskConfig(0, 2, 6, 3)  # Integer/I32/B6/M3, 9-byte draws
    p2 = mk.MaskConfigPair(c2, c2)
    m2 = mk.derive_mask(bytes(range(32)), 3, p2)
    assert m2.unit_value == "5011989261627366700"
    assert [m2.element(i) for i in range(3)] == [
        "7999609015010378104",
        "6103775104622568852",
        "8836670439873085217",
    ]


@pytest.mark.parametrize("group", [0, 1, 2])
@pytest.mark.parametrize("dtype", [0, 1, 2, 3])
@pytest.mark.parametrize("bound", [0, 2])
def test_fast_masker_matches_oracle(group, dtype, bound):
    """The typed fast masker (fixed-width exact arithmetic) is bit-identical
    to the exact-rational oracle, including clamp/rounding boundaries, tiny
    and subnormal values, and non-dyadic scalars."""
    import numpy as np

    c = mk.MaskConfig(group, dtype, bound, 3)
    pair = mk.MaskConfigPair(c, c)
    bnd = {0: 1.0, 2: 100.0}[bound]
    rng = np.random.default_rng(100 * group + 10 * dtype + bound)

    if dtype in (0, 1):
        np_dt = np.float32 if dtype == 0 else np.float64
        adversarial = [
            0.0, -0.0, 1e-300 if dtype == 1 else 1e-38, -1e-30, bnd, -bnd,
            bnd * 0.999999999, -bnd * 1.000000001, bnd * 2, -bnd * 1e6,
            1 / 3, -2 / 3, 5e-324 if dtype == 1 else 1e-45, 0.1, -0.7,
        ]
        w = np.concatenate([
            np.array(adversarial, dtype=np_dt),
            rng.uniform(-2 * bnd, 2 * bnd, 200).astype(np_dt),
            (rng.uniform(-1, 1, 50) * 1e-12).astype(np_dt),
        ])
    else:
        np_dt = np.int32 if dtype == 2 else np.int64
        big = 2**62 if dtype == 3 else 2**31 - 1
        adversarial = [0, 1, -1, int(bnd), -int(bnd), int(bnd) + 1, big, -big]
        if dtype == 3:
            adversarial += [2**53, 2**53 + 1, -(2**53) - 3]
        w = np.concatenate([
            np.array(adversarial, dtype=np_dt),
            rng.integers(-3 * int(bnd), 3 * int(bnd), 200).astype(np_dt),
        ])

    for num, den in [(1, 1), (1, 2), (1, 3), (7, 8), (3, 7), (1, 10**6)]:
        seed = bytes([num * 13 % 256, den % 256]) * 16
        fast = mk.mask_model(seed, mk.Scalar(num, den), w, pair)
        oracle = mk.mask_model_oracle(seed, mk.Scalar(num, den), w, pair)
        assert fast.serialize() == oracle.serialize(), (
            f"fast masker diverges from oracle (scalar {num}/{den})"
        )


def test_mixed_config_pair_roundtrip():
    """MaskConfigPair with DIFFERENT vect and unit configs (the reference
    keeps them independent, mask/config/mod.rs MaskConfigPair): mask ->
    aggregate -> unmask still averages correctly."""
    import numpy as np

    vect = mk.MaskConfig(1, 0, 0, 6)   # Prime/F32/B0/M6
    unit = mk.MaskConfig(0, 0, 2, 3)   # Integer/F32/B2/M3 for the scalar
    pair = mk.MaskConfigPair(vect, unit)
    n, k = 64, 4
    rng = np.random.default_rng(77)
    ws = [rng.uniform(-1, 1, n).astype(np.float32) for _ in range(k)]
    agg = mk.Aggregation(pair, n)
    mask_agg = mk.Aggregation(pair, n)
    for i, w in enumerate(ws):
        seed = bytes([i + 9]) * 32
        agg.aggregate(mk.mask_model(seed, mk.Scalar(1, k), w, pair))
        mask_agg.aggregate(mk.derive_mask(seed, n, pair))
    out = agg.unmask(mask_agg.object)
    expect = np.mean([w.astype(np.float64) for w in ws], axis=0)
    assert np.abs(out.astype(np.float64) - expect).max() < 1e-5
