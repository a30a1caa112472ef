"""GPU kernel tests (MI355X). Run via gpurun: pytest -m gpu.

Parity contracts:
  - K1 mask expansion is BIT-EXACT vs the CPU oracle (_core.mask.derive_mask):
    same ChaCha20 stream, same rejection walk (protocol requirement).
  - K3/K2 aggregation canonical results are bit-exact vs the CPU Aggregation.
  - K4 unmask matches the CPU exact-rational unmask within 1e-6 (f32).
"""
import numpy as np
import pytest

torch = pytest.importorskip("torch")

from xaynet_amd import _core  # noqa: E402

mk = _core.mask

pytestmark = pytest.mark.gpu


def make_engine(length, cfg_args=(1, 0, 0, 6)):
    from xaynet_amd.ops import GpuMaskedAggregator, gpu_available

    if not gpu_available():
        pytest.fail("GPU expected for -m gpu tests but not available")
    c = mk.MaskConfig(*cfg_args)
    return GpuMaskedAggregator(c, c, length), c


@pytest.mark.parametrize(
    "cfg_args",
    [
        (1, 0, 0, 3),  # Prime/F32/B0/M3: 6-byte order
        (1, 0, 0, 6),  # Prime/F32/B0/M6: 7-byte
        (0, 2, 6, 3),  # Integer/I32/B6/M3: 9-byte prng? (2e19+1 -> skip if >8)
        (0, 3, 0, 6),  # Integer/I64/B0/M6: 8-byte (2e16+1)
        (2, 0, 0, 3),  # Power2/F32/B0/M3: 2^45
        (0, 0, 2, 3),  # Integer/F32/B2/M3
    ],
)
def test_k1_mask_expand_bit_exact(cfg_args):
    c = mk.MaskConfig(*cfg_args)
    if c.prng_nbytes > 8:
        pytest.skip("order > 2^64: CPU path")
    length = 4099  # odd size to exercise tails
    from xaynet_amd.ops import GpuMaskedAggregator

    eng = GpuMaskedAggregator(c, c, length)
    pair = mk.MaskConfigPair(c, c)
    for seed_byte in (0, 7, 251):
        seed = bytes([seed_byte]) * 32
        vals = eng.derive_mask_values(seed).cpu().numpy().astype(np.uint64)
        oracle = mk.derive_mask(seed, length, pair)
        ob = np.frombuffer(bytes(oracle.vect_bytes), dtype=np.uint8).reshape(length, c.bytes_per_number)
        expect = np.zeros(length, dtype=np.uint64)
        for b in range(c.bytes_per_number):
            expect |= ob[:, b].astype(np.uint64) << np.uint64(8 * b)
        assert (vals == expect).all(), f"mismatch at {np.nonzero(vals != expect)[0][:5]}"


def test_k3_aggregate_bit_exact_vs_cpu():
    length = 1000
    eng, c = make_engine(length, (1, 0, 0, 3))
    pair = mk.MaskConfigPair(c, c)
    rng = np.random.default_rng(1)

    pool = eng.alloc_update_pool(8)
    cpu_agg = mk.Aggregation(pair, length)
    for i in range(8):
        seed = bytes(rng.integers(0, 256, 32, dtype=np.uint8))
        w = rng.uniform(-1, 1, length).astype(np.float32)
        masked = mk.mask_model(seed, mk.Scalar(1, 8), w, pair)
        cpu_agg.aggregate(masked)
        wire = masked.serialize()
        # vect limbs at offset 8 .. 8+len*bpn
        limbs = wire[8 : 8 + length * c.bytes_per_number]
        eng.upload_update(pool, i, limbs)
    torch.cuda.synchronize()
    eng.aggregate_pool(pool, 8)
    got = eng.canonical().cpu().numpy().astype(np.uint64)

    expect = np.array([int(cpu_agg.object.element(i)) for i in range(length)], dtype=np.uint64)
    assert (got == expect).all()


def test_k1_reg_fallback_bit_exact():
    """The XAYNET_K1_REG=1 pipeline (original all-register candidates +
    flag-gated scatter) stays bit-exact. Mode selection is process-static,
    so run the check in a subprocess."""
    import os
    import subprocess
    import sys

    code = (
        "import numpy as np\n"
        "from xaynet_amd import _core\n"
        "from xaynet_amd.ops import GpuMaskedAggregator\n"
        "mk = _core.mask\n"
        "c = mk.MaskConfig(1, 0, 0, 6)\n"
        "eng = GpuMaskedAggregator(c, c, 4099)\n"
        "pair = mk.MaskConfigPair(c, c)\n"
        "seed = bytes([7]) * 32\n"
        "vals = eng.derive_mask_values(seed).cpu().numpy().astype(np.uint64)\n"
        "oracle = mk.derive_mask(seed, 4099, pair)\n"
        "ob = np.frombuffer(bytes(oracle.vect_bytes), dtype=np.uint8)"
        ".reshape(4099, c.bytes_per_number)\n"
        "expect = np.zeros(4099, dtype=np.uint64)\n"
        "for b in range(c.bytes_per_number):\n"
        "    expect |= ob[:, b].astype(np.uint64) << np.uint64(8 * b)\n"
        "assert (vals == expect).all()\n"
        "print('REG-OK')\n"
    )
    env = {**os.environ, "XAYNET_K1_REG": "1"}
    r = subprocess.run([sys.executable, "-c", code], env=env,
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]
    assert "REG-OK" in r.stdout


def test_k3_variants_agree():
    """All K3 load variants (generic EPT 4/8/16, LDS-staged 201/202) produce
    identical digit planes, including partial tail tiles."""
    from xaynet_amd import _hip

    length = 5000  # not a multiple of the 2048/4096 LDS tiles
    eng, c = make_engine(length, (1, 0, 0, 6))  # bpn=7
    rng = np.random.default_rng(9)
    n = 6
    pool = eng.alloc_update_pool(n)
    for i in range(n):
        row = rng.integers(0, 256, length * c.bytes_per_number, dtype=np.uint8)
        eng.upload_update(pool, i, row.tobytes())
    torch.cuda.synchronize()

    planes = {}
    for ept in (4, 8, 16, 201, 202):
        eng.reset()
        _hip.aggregate_batch(
            eng.acc.data_ptr(), pool.data_ptr(), pool.stride(0), n, length, eng.bpn, ept
        )
        torch.cuda.synchronize()
        planes[ept] = eng.acc.cpu().numpy().copy()
    for ept in (8, 16, 201, 202):
        assert (planes[ept] == planes[4]).all(), f"variant ept={ept} diverges"


def test_k3_wide_variants_agree():
    """Wide-order (bpn=10) K3 EPT variants produce identical digit planes."""
    from xaynet_amd import _hip

    length = 5003  # prime: exercises partial tail chunks at every EPT
    eng, c = make_engine(length, (1, 1, 0, 3))  # bpn=10
    assert eng.wide
    rng = np.random.default_rng(11)
    n = 6
    pool = eng.alloc_update_pool(n)
    for i in range(n):
        row = rng.integers(0, 256, length * c.bytes_per_number, dtype=np.uint8)
        eng.upload_update(pool, i, row.tobytes())
    torch.cuda.synchronize()

    planes = {}
    for ept in (2, 4, 8):
        eng.reset()
        _hip.aggregate_batch(
            eng.acc.data_ptr(), pool.data_ptr(), pool.stride(0), n, length, eng.bpn, ept
        )
        torch.cuda.synchronize()
        planes[ept] = eng.acc.cpu().numpy().copy()
    for ept in (4, 8):
        assert (planes[ept] == planes[2]).all(), f"wide variant ept={ept} diverges"


def test_k4_full_roundtrip_vs_oracle():
    length = 2000
    eng, c = make_engine(length, (1, 0, 0, 3))
    pair = mk.MaskConfigPair(c, c)
    rng = np.random.default_rng(5)
    k = 4

    pool = eng.alloc_update_pool(k)
    cpu_agg = mk.Aggregation(pair, length)
    cpu_mask_agg = mk.Aggregation(pair, length)
    mask_vals = torch.zeros(length, dtype=torch.int64, device="cuda")
    mask_unit = 0
    ws = []
    for i in range(k):
        seed = bytes(rng.integers(0, 256, 32, dtype=np.uint8))
        w = rng.uniform(-1, 1, length).astype(np.float32)
        ws.append(w)
        masked = mk.mask_model(seed, mk.Scalar(1, k), w, pair)
        cpu_agg.aggregate(masked)
        m = mk.derive_mask(seed, length, pair)
        cpu_mask_agg.aggregate(m)
        limbs = masked.serialize()[8 : 8 + length * c.bytes_per_number]
        eng.upload_update(pool, i, limbs)
        # gpu mask aggregation
        mv = eng.derive_mask_values(seed)
        eng.mod_add_values(mask_vals, mv)
        mask_unit = (mask_unit + int(m.unit_value)) % int(c.order)
        eng.unit_acc = (eng.unit_acc + int(masked.unit_value)) % int(c.order)
    eng.aggregate_pool(pool, k, unit_sum=0)
    out = eng.unmask_f32(mask_vals, mask_unit).cpu().numpy()

    oracle = cpu_agg.unmask(cpu_mask_agg.object)
    expect = np.mean([w.astype(np.float64) for w in ws], axis=0)
    assert np.abs(out - oracle).max() < 1e-6
    assert np.abs(out - expect).max() < 1e-5


@pytest.mark.parametrize("dtype_id,np_dtype", [(2, np.int32), (3, np.int64)])
def test_k4_integer_dtypes_vs_oracle(dtype_id, np_dtype):
    """K4 unmask for i32/i64 data (config #4 family): GPU result matches the
    exact-rational CPU oracle (truncation toward zero). F64 orders exceed
    2^64 (exp_shift 10^20) and stay on the CPU plane."""
    length = 1500
    eng, c = make_engine(length, (1, dtype_id, 0, 3))  # Prime/<dtype>/B0/M3
    pair = mk.MaskConfigPair(c, c)
    rng = np.random.default_rng(6)
    k = 4

    pool = eng.alloc_update_pool(k)
    cpu_agg = mk.Aggregation(pair, length)
    cpu_mask_agg = mk.Aggregation(pair, length)
    mask_vals = torch.zeros(length, dtype=torch.int64, device="cuda")
    mask_unit = 0
    for i in range(k):
        seed = bytes(rng.integers(0, 256, 32, dtype=np.uint8))
        w = rng.integers(-1, 2, length).astype(np_dtype)  # B0 clamps to [-1,1]
        masked = mk.mask_model(seed, mk.Scalar(1, k), w, pair)
        cpu_agg.aggregate(masked)
        m = mk.derive_mask(seed, length, pair)
        cpu_mask_agg.aggregate(m)
        limbs = masked.serialize()[8 : 8 + length * c.bytes_per_number]
        eng.upload_update(pool, i, limbs)
        mv = eng.derive_mask_values(seed)
        eng.mod_add_values(mask_vals, mv)
        mask_unit = (mask_unit + int(m.unit_value)) % int(c.order)
        eng.unit_acc = (eng.unit_acc + int(masked.unit_value)) % int(c.order)
    eng.aggregate_pool(pool, k, unit_sum=0)
    out = eng.unmask(mask_vals, mask_unit).cpu().numpy()

    oracle = cpu_agg.unmask(cpu_mask_agg.object)
    assert out.dtype == np_dtype
    assert (out == oracle).all(), f"{(out != oracle).sum()} mismatches vs oracle"


def test_gpu_sum2_mask_aggregation_bit_exact():
    """ops.sum2.aggregate_masks (K1+K2 on GPU) emits wire bytes identical to
    the CPU oracle's Aggregation over derive_mask — the sum2 task offloaded."""
    from xaynet_amd.ops.sum2 import aggregate_masks

    length, k = 3000, 7
    c = mk.MaskConfig(1, 0, 0, 6)
    pair = mk.MaskConfigPair(c, c)
    seeds = [bytes([i + 1]) * 32 for i in range(k)]

    wire = aggregate_masks(seeds, c, c, length)

    agg = mk.Aggregation(pair, length)
    for s in seeds:
        agg.aggregate(mk.derive_mask(s, length, pair))
    assert wire == bytes(agg.object.serialize())


def test_shard_unmask_equals_full_unmask():
    """unmask_planes on contiguous shards (the reduce-scatter path at N>1)
    concatenates to exactly the full-vector unmask."""
    length, k, world = 4096, 6, 4
    eng, c = make_engine(length, (1, 0, 0, 6))
    mask_vals = torch.zeros(length, dtype=torch.int64, device="cuda")
    pool = eng.alloc_update_pool(k)
    scratch = torch.empty(length, dtype=torch.int64, device="cuda")
    mask_unit = 0
    for p in range(k):
        seed = bytes([p + 3]) * 32
        eng.derive_mask_values(seed, out=scratch)
        eng.synth_update(pool, p, scratch, participant=p, scalar=1.0 / k)
        eng.mod_add_values(mask_vals, scratch)
        mask_unit = (mask_unit + eng.unit_draw(seed)) % int(c.order)
        eng.unit_acc = (eng.unit_acc + eng.masked_unit_for(seed, 1, k)) % int(c.order)
    eng.aggregate_pool(pool, k)
    full = eng.unmask(mask_vals, mask_unit).cpu().numpy()

    shard = length // world
    parts = []
    for r in range(world):
        lo = r * shard
        planes = eng.acc[:, lo : lo + shard].contiguous()
        parts.append(
            eng.unmask_planes(planes, mask_vals[lo : lo + shard], mask_unit, k)
            .cpu().numpy()
        )
    assert (np.concatenate(parts) == full).all()

    # values path (the halved-bytes reduce-scatter): canonical values fed to
    # k4_unmask_values reproduce the plane unmask exactly
    canon = eng.canonical()
    parts = [
        eng.unmask_values(canon[r * shard : (r + 1) * shard].contiguous(),
                          mask_vals[r * shard : (r + 1) * shard], mask_unit, k)
        .cpu().numpy()
        for r in range(world)
    ]
    assert (np.concatenate(parts) == full).all()


def test_wide_order_f64_roundtrip_vs_oracle():
    """u128-order path (bpn=10, Prime/F64/B0/M3): K3 digit planes +
    k6_unpack_u128 + k4_unmask_u128 match the exact-rational oracle."""
    length = 1200
    eng, c = make_engine(length, (1, 1, 0, 3))  # F64 -> 10-byte limbs
    assert eng.wide and eng.bpn == 10
    pair = mk.MaskConfigPair(c, c)
    rng = np.random.default_rng(8)
    k = 4

    pool = eng.alloc_update_pool(k)
    cpu_agg = mk.Aggregation(pair, length)
    cpu_mask_agg = mk.Aggregation(pair, length)
    mask_vals = torch.zeros(2, length, dtype=torch.int64, device="cuda")
    mask_unit = 0
    ws = []
    for i in range(k):
        seed = bytes(rng.integers(0, 256, 32, dtype=np.uint8))
        w = rng.uniform(-1, 1, length).astype(np.float64)
        ws.append(w)
        masked = mk.mask_model(seed, mk.Scalar(1, k), w, pair)
        cpu_agg.aggregate(masked)
        m = mk.derive_mask(seed, length, pair)
        cpu_mask_agg.aggregate(m)
        limbs = masked.serialize()[8 : 8 + length * c.bytes_per_number]
        eng.upload_update(pool, i, limbs)
        # mask values via the GPU unpack of the CPU-derived mask wire
        mwire = m.serialize()[8 : 8 + length * c.bytes_per_number]
        t = torch.frombuffer(bytearray(mwire), dtype=torch.uint8).to("cuda")
        mv = eng.unpack_wire(t)
        eng.mod_add_values(mask_vals, mv)
        mask_unit = (mask_unit + int(m.unit_value)) % int(c.order)
        eng.unit_acc = (eng.unit_acc + int(masked.unit_value)) % int(c.order)
    eng.aggregate_pool(pool, k, unit_sum=0)

    # canonical split-planes equal the oracle's aggregated elements
    canon = eng.canonical().cpu().numpy()
    for i in range(0, length, 97):
        got = (int(canon[1][i]) % (1 << 64)) * (1 << 64) + (int(canon[0][i]) % (1 << 64))
        assert got == int(cpu_agg.object.element(i)), f"canonical mismatch at {i}"

    out = eng.unmask(mask_vals, mask_unit).cpu().numpy()
    assert out.dtype == np.float64
    oracle = cpu_agg.unmask(cpu_mask_agg.object)
    expect = np.mean(ws, axis=0)
    assert np.abs(out - oracle).max() < 1e-9
    assert np.abs(out - expect).max() < 1e-8


def test_wide_order_staged_driver_round():
    """Full staged-plane round on a wide (F64) config: coordinator + GPU
    driver + HTTP participants; the f64 global model matches the mean."""
    from xaynet_amd.ops import make_coordinator_driver

    co = _core.coordinator
    sdk = _core.sdk
    rest = _core.rest
    n, length = 10, 2048
    s = co.Settings()
    s.sum_prob = 0.5
    s.update_prob = 1.0
    s.model_length = length
    c = mk.MaskConfig(1, 1, 0, 3)  # Prime/F64/B0/M3 (bpn=10, wide)
    s.mask_cfg = mk.MaskConfigPair(c, c)
    s.set_sum(1, 100, 0.05, 10.0)
    s.set_update(3, 100, 0.05, 10.0)
    s.set_sum2(1, 100, 0.05, 10.0)
    coord = co.Coordinator(s, co.InMemoryStorage(), co.InMemoryModels(), True)
    server = rest.RestServer(coord, "127.0.0.1", 0, 4)
    assert server.start()
    driver = make_coordinator_driver(coord, c, c, length, pool_size=8)
    driver.start()
    client = rest.HttpXaynetClient("127.0.0.1", server.port)
    rng = np.random.default_rng(29)
    participants = [
        sdk.Participant(bytes(rng.integers(0, 256, 32, dtype=np.uint8)), 1, 1, client)
        for _ in range(n)
    ]
    weights = [rng.uniform(-1, 1, length).astype(np.float64) for _ in range(n)]

    import time

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
                model = _core.sdk.decode_model(body, 1)
            time.sleep(0.005)
    finally:
        coord.stop()
        driver.stop()
        server.stop()
    assert model is not None and driver.rounds_unmasked >= 1
    # subset mean: bounds check only (accepted set varies); weights in [-1,1]
    assert model.shape == (length,) and np.isfinite(model).all()
    assert np.abs(model).max() <= 1.0 + 1e-9


def test_k5_synth_updates_roundtrip():
    length = 3000
    eng, c = make_engine(length, (1, 0, 0, 3))
    k = 8
    pool = eng.alloc_update_pool(k)
    mask_vals = torch.zeros(length, dtype=torch.int64, device="cuda")
    mask_unit = 0
    for p in range(k):
        seed = bytes([p + 1]) * 32
        mv = eng.derive_mask_values(seed)
        eng.synth_update(pool, p, mv, participant=p, scalar=1.0 / k)
        eng.mod_add_values(mask_vals, mv)
        mask_unit = (mask_unit + eng.unit_draw(seed)) % int(c.order)
        eng.unit_acc = (eng.unit_acc + eng.masked_unit_for(seed, 1, k)) % int(c.order)
    eng.aggregate_pool(pool, k)
    out = eng.unmask_f32(mask_vals, mask_unit).cpu().numpy()

    # replicate splitmix64 weights
    def splitmix64(x):
        x = (x + 0x9E3779B97F4A7C15) & (2**64 - 1)
        x = ((x ^ (x >> 30)) * 0xBF58476D1CE4E5B9) & (2**64 - 1)
        x = ((x ^ (x >> 27)) * 0x94D049BB133111EB) & (2**64 - 1)
        return x ^ (x >> 31)

    expect = np.zeros(length)
    for p in range(k):
        w = np.array(
            [
                float(splitmix64((p * 0x100000001B3 + j) & (2**64 - 1)) >> 11)
                * (2.0 / 9007199254740992.0)
                - 1.0
                for j in range(length)
            ]
        )
        expect += w / k
    assert np.abs(out - expect).max() < 1e-5


def test_k6_pack_unpack_roundtrip():
    length = 5000
    eng, c = make_engine(length, (1, 0, 0, 3))
    vals = torch.randint(0, int(c.order), (length,), dtype=torch.int64, device="cuda")
    packed = eng.pack_wire(vals)
    back = eng.unpack_wire(packed)
    assert torch.equal(vals, back)


def test_aggregation_bandwidth_smoke():
    """Not an assertion on speed — prints achieved GB/s for the logbook."""
    import time

    length = 25_000_000
    eng, c = make_engine(length, (1, 0, 0, 6))
    n = 16
    pool = eng.alloc_update_pool(n)
    pool.random_()  # content irrelevant for throughput
    torch.cuda.synchronize()
    eng.aggregate_pool(pool, n)  # warmup
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    reps = 4
    for _ in range(reps):
        eng.aggregate_pool(pool, n)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    gb = reps * n * length * c.bytes_per_number / 1e9
    print(f"\nK3 aggregate: {gb / dt:.0f} GB/s effective ({n * reps / dt:.0f} updates/s @25M)")
    assert dt > 0


@pytest.mark.parametrize(
    "cfg_args",
    [
        (1, 1, 0, 3),   # Prime/F64/B0/M3: bpn=10, 3 words/draw
        (1, 1, 0, 6),   # Prime/F64/B0/M6: bpn=11, 3 words/draw
        (1, 1, 4, 9),   # Prime/F64/B4/M9: bpn=14, 4 words/draw
        (1, 1, 6, 12),  # Prime/F64/B6/M12: bpn=16 (order ~2^128), 4 words/draw
    ],
)
def test_k1_wide_mask_expand_bit_exact(cfg_args):
    """Wide-order (u128) K1: split lo/hi expansion reproduces the reference
    ChaCha20 rejection stream exactly. Replaces
    crypto/prng.rs:16-27 for the F64 families."""
    c = mk.MaskConfig(*cfg_args)
    assert c.prng_nbytes > 8
    length = 4099
    from xaynet_amd.ops import GpuMaskedAggregator

    eng = GpuMaskedAggregator(c, c, length)
    assert eng.wide
    pair = mk.MaskConfigPair(c, c)
    bpn = c.bytes_per_number
    for seed_byte in (0, 7, 251):
        seed = bytes([seed_byte]) * 32
        vals = eng.derive_mask_values(seed).cpu().numpy().astype(np.uint64)
        oracle = mk.derive_mask(seed, length, pair)
        ob = np.frombuffer(bytes(oracle.vect_bytes), dtype=np.uint8).reshape(length, bpn)
        exp_lo = np.zeros(length, dtype=np.uint64)
        exp_hi = np.zeros(length, dtype=np.uint64)
        for b in range(8):
            exp_lo |= ob[:, b].astype(np.uint64) << np.uint64(8 * b)
        for b in range(8, bpn):
            exp_hi |= ob[:, b].astype(np.uint64) << np.uint64(8 * (b - 8))
        bad = np.nonzero((vals[0] != exp_lo) | (vals[1] != exp_hi))[0]
        assert bad.size == 0, f"{cfg_args}: mismatch at {bad[:5]}"


def test_gpu_sum2_wide_mask_aggregation_bit_exact():
    """ops.sum2.aggregate_masks on a wide (bpn=10) config emits wire bytes
    identical to the CPU oracle Aggregation — K1/K2/K6 u128 end-to-end."""
    from xaynet_amd.ops.sum2 import aggregate_masks

    length, k = 2000, 5
    c = mk.MaskConfig(1, 1, 0, 3)
    pair = mk.MaskConfigPair(c, c)
    seeds = [bytes([i + 1]) * 32 for i in range(k)]

    wire = aggregate_masks(seeds, c, c, length)

    agg = mk.Aggregation(pair, length)
    for s in seeds:
        agg.aggregate(mk.derive_mask(s, length, pair))
    assert wire == bytes(agg.object.serialize())


def test_wide_stream_round_unmasks_clean():
    """Full wide-order GPU round: K1 expand -> K5 synth+pack -> K3 aggregate
    -> K4 u128 unmask; the unmasked f64 model must be finite, bounded and
    non-degenerate (the masks cancel exactly)."""
    length, k = 3001, 4
    c = mk.MaskConfig(1, 1, 0, 3)  # F64, bpn=10
    from xaynet_amd.ops import GpuMaskedAggregator

    eng = GpuMaskedAggregator(c, c, length)
    pool = eng.alloc_update_pool(k)
    mask_vals = torch.zeros(2, length, dtype=torch.int64, device="cuda")
    scratch = torch.empty(2, length, dtype=torch.int64, device="cuda")
    mask_unit = 0
    for p in range(k):
        seed = bytes([p + 1]) * 32
        eng.derive_mask_values(seed, out=scratch)
        eng.synth_update(pool, p, scratch, participant=p, scalar=1.0 / k)
        eng.mod_add_values(mask_vals, scratch)
        mask_unit = (mask_unit + eng.unit_draw(seed)) % int(c.order)
        eng.unit_acc = (eng.unit_acc + eng.masked_unit_for(seed, 1, k)) % int(c.order)
    eng.aggregate_pool(pool, k)
    out = eng.unmask(mask_vals, mask_unit).cpu().numpy()
    assert out.dtype == np.float64
    assert np.isfinite(out).all()
    assert np.abs(out).max() <= 1.0 + 1e-9
    assert np.abs(out).mean() > 1e-3
